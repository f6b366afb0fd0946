"""Runtime configuration.

The reference hardcodes everything (vendor "10de" at
`pkg/device_plugin/device_plugin.go:19`, CDI path `:20`, sysfs base `:36`,
pci.ids path `:37`, resource namespace `generic_device_plugin.go:26`,
device-list strategy `:58-66`).  Here every knob is a dataclass field that
can be set from CLI flags or `KXDP_*` environment variables (SURVEY.md §5
"Config / flag system": the new build gets a proper config system).
"""
from __future__ import annotations

import argparse
import dataclasses
import os
from dataclasses import dataclass, field
from typing import Optional, Sequence

# Device-list strategies (reference: cdi/constant.go:8-12).
STRATEGY_CDI_CRI = "cdi-cri"          # emit AllocateResponse.cdi_devices
STRATEGY_CDI_ANNOTATIONS = "cdi-annotations"  # emit cdi.k8s.io/... annotations
STRATEGY_DEVICE_NODES = "device-nodes"  # emit raw /dev/vfio DeviceSpecs (no CDI)
STRATEGIES = (STRATEGY_CDI_CRI, STRATEGY_CDI_ANNOTATIONS, STRATEGY_DEVICE_NODES)

DEFAULT_VENDOR_ALLOWLIST = (0x1002,)  # AMD/ATI GPUs (AMD CPU-side IP is 0x1022)


def bundled_pci_ids_path() -> str:
    """Pinned pci.ids snapshot shipped inside the package."""
    return os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "data", "pci.ids")


def _env(name: str, default: str) -> str:
    return os.environ.get(f"KXDP_{name}", default)


def _env_int(name: str, default: int) -> int:
    return int(os.environ.get(f"KXDP_{name}", default))


def _env_float(name: str, default: float) -> float:
    return float(os.environ.get(f"KXDP_{name}", default))


def _env_bool(name: str, default: bool) -> bool:
    raw = os.environ.get(f"KXDP_{name}")
    if raw is None:
        return default
    return raw.strip().lower() in ("1", "true", "yes", "on")


@dataclass
class Config:
    # --- filesystem roots (overridable so the whole stack runs against a
    # mock node tree: SURVEY.md §4 test plan item 1) ---
    sysfs_root: str = field(default_factory=lambda: _env("SYSFS_ROOT", "/sys"))
    dev_root: str = field(default_factory=lambda: _env("DEV_ROOT", "/dev"))

    # --- discovery ---
    vendor_allowlist: Sequence[int] = field(
        default_factory=lambda: tuple(
            int(v, 16) for v in _env("VENDORS", "1002").split(",") if v
        )
    )
    # Optional explicit device-ID allowlist ("75a3,75a0"); empty = any GPU
    # class function from an allowed vendor.
    device_allowlist: Sequence[int] = field(
        default_factory=lambda: tuple(
            int(v, 16) for v in _env("DEVICES", "").split(",") if v
        )
    )
    # Only devices bound to this driver are schedulable (VFIO passthrough).
    required_driver: str = field(default_factory=lambda: _env("DRIVER", "vfio-pci"))
    # Schedulable PCI classes: 0x12xxxx "Processing accelerator" (MI355X
    # enumerates as 0x120000 — live-node confirmed) and 0x03xxxx display
    # (consumer AMD GPUs). The reference filtered on vendor only
    # (device_plugin.go:142-149), which would also match vendor audio/bridge
    # functions; we keep non-GPU functions of the same IOMMU group as
    # passthrough companions but never schedule them.
    gpu_class_prefixes: Sequence[int] = (0x03, 0x12)

    # --- naming ---
    resource_namespace: str = field(default_factory=lambda: _env("NAMESPACE", "amd.com"))
    # pci.ids fallback chain, tried in order after the curated Instinct
    # table. First entry: the BUNDLED pinned snapshot (data/pci.ids) so an
    # air-gapped node without a distro pci.ids still names known silicon
    # deterministically; host copies follow for anything missing there.
    pci_ids_paths: Sequence[str] = field(
        default_factory=lambda: tuple(
            p for p in _env(
                "PCI_IDS",
                bundled_pci_ids_path()
                + ":/usr/share/misc/pci.ids:/usr/share/hwdata/pci.ids:/usr/pci.ids",
            ).split(":")
            if p
        )
    )
    # Collapse all discovered models into one resource name (e.g. "GPU") if
    # set; default: one resource per model like the reference
    # (device_plugin.go:83-119 starts one plugin per deviceMap entry).
    unified_resource_name: str = field(default_factory=lambda: _env("UNIFIED_RESOURCE", ""))

    # --- CDI ---
    cdi_dir: str = field(default_factory=lambda: _env("CDI_DIR", "/var/run/cdi"))
    cdi_kind: str = field(default_factory=lambda: _env("CDI_KIND", "amd.com/gpu"))
    cdi_spec_name: str = field(default_factory=lambda: _env("CDI_SPEC_NAME", "kxdp-vfio"))
    cdi_format: str = field(default_factory=lambda: _env("CDI_FORMAT", "yaml"))  # yaml|json
    # Emitted cdiVersion. 0.8.0 default; turn down for runtimes predating
    # it (the reference wrote frozen 0.6.0, cdi/spec.go:12 — everything
    # this plugin emits is already expressible at 0.6.0).
    cdi_version: str = field(default_factory=lambda: _env("CDI_VERSION", "0.8.0"))
    device_list_strategy: str = field(default_factory=lambda: _env("STRATEGY", STRATEGY_CDI_CRI))

    # --- kubelet ---
    kubelet_socket_dir: str = field(
        default_factory=lambda: _env("KUBELET_DIR", "/var/lib/kubelet/device-plugins")
    )
    kubelet_socket_name: str = "kubelet.sock"
    plugin_socket_prefix: str = field(default_factory=lambda: _env("SOCKET_PREFIX", "kata-xpu-amd"))
    # How kubelet learns about us: "legacy" self-registration against
    # kubelet.sock (the reference's only mode), "watcher" = serve the
    # pluginregistration.v1 service under plugins_registry/ for kubelet's
    # plugin watcher, or "both".
    registration_mode: str = field(default_factory=lambda: _env("REGISTRATION_MODE", "legacy"))
    plugins_registry_dir: str = field(
        default_factory=lambda: _env("PLUGINS_REGISTRY", "/var/lib/kubelet/plugins_registry")
    )
    grpc_timeout_s: float = field(default_factory=lambda: _env_float("GRPC_TIMEOUT_S", 5.0))

    # --- topology ---
    # Optional topology snapshot (JSON) taken while GPUs were amdgpu-bound;
    # needed because KFD does not enumerate vfio-bound GPUs. See
    # topology/hive.py and `python -m kata_xpu_device_plugin_amd.tools.topo`.
    topology_hint_path: str = field(
        default_factory=lambda: _env("TOPOLOGY_HINT", "/etc/kata-xpu-amd/topology.json")
    )

    # --- health ---
    health_poll_interval_s: float = field(default_factory=lambda: _env_float("HEALTH_POLL_S", 10.0))
    amdsmi_health: bool = field(default_factory=lambda: _env_bool("AMDSMI_HEALTH", True))
    # Reject Allocate() for devices currently marked Unhealthy (kubelet
    # should not send them, but a stale scheduler view can; defensive
    # default on). The reference allocates regardless of health.
    reject_unhealthy: bool = field(default_factory=lambda: _env_bool("REJECT_UNHEALTHY", True))
    # Periodic in-daemon GPU probing of HIP-visible (amdgpu-bound) devices;
    # 0 = off (the default — pure-VFIO nodes have no HIP-visible GPUs).
    gpu_probe_interval_s: float = field(default_factory=lambda: _env_float("GPU_PROBE_S", 0.0))
    # Periodic re-discovery (picks up SR-IOV VF count changes, rebinds);
    # 0 = off. Rescans that find no change are no-ops (no CDI rewrite).
    rescan_interval_s: float = field(default_factory=lambda: _env_float("RESCAN_S", 0.0))

    # --- diagnostics ---
    # When set, every plugin server records per-RPC handler wall time and
    # event-loop scheduling lag, and dumps a JSON breakdown to this path
    # on shutdown (tail-latency attribution; zero cost when unset).
    rpc_timing_path: str = field(default_factory=lambda: _env("RPC_TIMING", ""))

    # --- performance ---
    # Freeze the startup object graph + raise GC thresholds after start()
    # (kills multi-ms generational-GC pauses in Allocate's p99 tail).
    gc_tuning: bool = field(default_factory=lambda: _env_bool("GC_TUNING", True))
    # Pin the daemon to a CPU set, e.g. "2-3" or "0,4" (empty = no pin).
    # On busy nodes this keeps admission latency off noisy-neighbor cores;
    # matches a DaemonSet cpuset without needing static CPU policy.
    cpu_affinity: str = field(default_factory=lambda: _env("CPU_AFFINITY", ""))

    # --- observability ---
    metrics_port: int = field(default_factory=lambda: _env_int("METRICS_PORT", 0))  # 0 = off
    log_level: str = field(default_factory=lambda: _env("LOG_LEVEL", "INFO"))

    # --- native fast path ---
    # "auto": use the C++ extension when importable (required on GPU nodes),
    # "require": fail loudly without it, "off": pure-Python (tests only).
    native: str = field(default_factory=lambda: _env("NATIVE", "auto"))

    def validate(self) -> None:
        if self.device_list_strategy not in STRATEGIES:
            raise ValueError(
                f"device_list_strategy must be one of {STRATEGIES}, "
                f"got {self.device_list_strategy!r}"
            )
        if self.cdi_format not in ("yaml", "json"):
            raise ValueError(f"cdi_format must be yaml|json, got {self.cdi_format!r}")
        from .cdi.schema import KNOWN_CDI_VERSIONS
        if self.cdi_version not in KNOWN_CDI_VERSIONS:
            raise ValueError(
                f"cdi_version must be one of {KNOWN_CDI_VERSIONS}, "
                f"got {self.cdi_version!r}")
        if "/" not in self.cdi_kind:
            raise ValueError(f"cdi_kind must look like vendor/class, got {self.cdi_kind!r}")
        if self.native not in ("auto", "require", "off"):
            raise ValueError(f"native must be auto|require|off, got {self.native!r}")
        if self.registration_mode not in ("legacy", "watcher", "both"):
            raise ValueError(
                f"registration_mode must be legacy|watcher|both, "
                f"got {self.registration_mode!r}")

    @property
    def kubelet_socket(self) -> str:
        return os.path.join(self.kubelet_socket_dir, self.kubelet_socket_name)

    @classmethod
    def add_args(cls, parser: argparse.ArgumentParser) -> None:
        g = parser.add_argument_group("plugin config (env: KXDP_*)")
        g.add_argument("--sysfs-root", dest="sysfs_root")
        g.add_argument("--dev-root", dest="dev_root")
        g.add_argument("--cdi-dir", dest="cdi_dir")
        g.add_argument("--cdi-kind", dest="cdi_kind")
        g.add_argument("--strategy", dest="device_list_strategy", choices=STRATEGIES)
        g.add_argument("--kubelet-dir", dest="kubelet_socket_dir")
        g.add_argument("--namespace", dest="resource_namespace")
        g.add_argument("--driver", dest="required_driver")
        g.add_argument("--topology-hint", dest="topology_hint_path")
        g.add_argument("--metrics-port", dest="metrics_port", type=int)
        g.add_argument("--log-level", dest="log_level")
        g.add_argument("--native", dest="native", choices=("auto", "require", "off"))
        g.add_argument("--registration-mode", dest="registration_mode",
                       choices=("legacy", "watcher", "both"))
        g.add_argument("--plugins-registry", dest="plugins_registry_dir")
        g.add_argument("--unified-resource", dest="unified_resource_name")
        g.add_argument("--cdi-format", dest="cdi_format", choices=("yaml", "json"))
        g.add_argument("--gpu-probe-interval", dest="gpu_probe_interval_s",
                       type=float)

    @classmethod
    def from_args(cls, args: Optional[argparse.Namespace] = None) -> "Config":
        cfg = cls()
        if args is not None:
            for f in dataclasses.fields(cls):
                v = getattr(args, f.name, None)
                if v is not None:
                    setattr(cfg, f.name, v)
        cfg.validate()
        return cfg
