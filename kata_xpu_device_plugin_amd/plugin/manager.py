"""Lifecycle orchestrator: discovery → topology → CDI → plugins → watch.

Reference analog: ``InitiateDevicePlugin`` + ``createDevicePlugins``
(`pkg/device_plugin/device_plugin.go:44-53, :83-119`) which block forever
on a never-closed channel with no signal handling (`:114` — SURVEY.md §7
quirk list). This manager owns a proper shutdown path (SIGTERM/SIGINT),
a single health watcher, and re-registration on kubelet restart.
"""
from __future__ import annotations

import os
import signal
import threading
from typing import Dict, List, Optional

from ..cdi.spec import build_spec, write_spec
from ..config import Config
from ..discovery.naming import resource_name
from ..discovery.sysfs import NodeInventory, XPUDevice, scan_node
from ..health.watcher import NodeWatcher
from ..topology.hive import GPUTopology, load_topology
from ..utils.log import configure as configure_logging, get_logger
from .server import XPUDevicePlugin
from .state import DeviceState

log = get_logger(__name__)


class PluginManager:
    def __init__(self, cfg: Config):
        cfg.validate()
        self.cfg = cfg
        self.inventory: Optional[NodeInventory] = None
        self.topology: Optional[GPUTopology] = None
        self.plugins: Dict[str, XPUDevicePlugin] = {}   # resource name → plugin
        self.states: Dict[str, DeviceState] = {}
        self.watcher: Optional[NodeWatcher] = None
        self.cdi_spec_path: Optional[str] = None
        self._stop = threading.Event()
        self._register = True          # start()'s registration choice,
        self._rescan_lock = threading.Lock()  # reused by dynamic rescan
        self._metrics = None
        self._amdsmi = None
        self._gpu_probe = None
        self._rescan_thread = None

    # ------------------------------------------------------------------
    def _group_by_resource(self, inv: NodeInventory) -> Dict[str, Dict[str, XPUDevice]]:
        out: Dict[str, Dict[str, XPUDevice]] = {}
        for gid, dev in inv.devices.items():
            rname = resource_name(
                dev.model_device_id,
                namespace=self.cfg.resource_namespace,
                pci_ids_paths=tuple(self.cfg.pci_ids_paths),
                is_vf=dev.is_vf,
                unified=self.cfg.unified_resource_name,
            )
            out.setdefault(rname, {})[gid] = dev
        return out

    def _write_cdi(self, inv: NodeInventory) -> Optional[str]:
        """Write the CDI spec for this inventory; a zero-device spec is
        schema-invalid (upstream CDI rejects empty `devices`), so empty
        nodes get no spec file — and any stale one is removed."""
        if not inv.devices:
            from ..cdi.spec import spec_path
            for fmt in ("yaml", "json"):
                stale = spec_path(self.cfg.cdi_dir, self.cfg.cdi_spec_name, fmt)
                if os.path.exists(stale):
                    try:
                        os.unlink(stale)
                    except OSError:
                        pass
            log.info("no devices: no CDI spec written")
            return None
        spec = build_spec(inv, self.cfg.cdi_kind, self.cfg.dev_root,
                          self.cfg.cdi_version)
        return write_spec(
            spec, self.cfg.cdi_dir, self.cfg.cdi_spec_name, self.cfg.cdi_format
        )

    def setup(self) -> None:
        """Discovery pass → CDI spec → per-resource plugin objects."""
        inv = scan_node(self.cfg)
        self.inventory = inv
        log.info(
            "discovered %d xPU device(s) in %.1f ms (%d %s function(s) total)",
            len(inv.devices), inv.scan_wall_s * 1e3, len(inv.all_functions),
            "/".join(f"{v:04x}" for v in self.cfg.vendor_allowlist),
        )
        self.topology = load_topology(self.cfg, inv)
        log.info("xGMI topology source: %s (%d hive-mapped GPUs)",
                 self.topology.source, len(self.topology.hive_of))

        self.cdi_spec_path = self._write_cdi(inv)

        for rname, devs in sorted(self._group_by_resource(inv).items()):
            state = DeviceState(devs)
            self.states[rname] = state
            self.plugins[rname] = XPUDevicePlugin(
                self.cfg, rname, state, self.topology
            )
            log.info("resource %s: %d device(s): %s", rname, len(devs),
                     ",".join(sorted(devs)))

    # ------------------------------------------------------------------
    def start(self, register: bool = True) -> None:
        self._register = register
        if not self.plugins:
            log.warning("no devices discovered; serving nothing (will still "
                        "watch for kubelet restarts)")
        for plugin in self.plugins.values():
            plugin.start(register=register)
        self.watcher = NodeWatcher(
            self.cfg,
            self.states,
            on_socket_removed=self._on_socket_removed,
            on_kubelet_restarted=self._on_kubelet_restarted,
            plugin_socket_names={p.socket_name for p in self.plugins.values()},
            on_cdi_spec_removed=self._on_cdi_spec_removed,
            cdi_spec_names={os.path.basename(self.cdi_spec_path)}
            if self.cdi_spec_path else set(),
        )
        self.watcher.start()
        self.watcher.wait_ready()
        if self.cfg.amdsmi_health:
            from ..health.amdsmi_health import AmdSmiPoller
            self._amdsmi = AmdSmiPoller(
                self.cfg.health_poll_interval_s, self._on_amdsmi_health,
                on_xgmi=self._on_xgmi_telemetry,
            )
            self._amdsmi.start()
        if self.cfg.gpu_probe_interval_s > 0:
            from ..health.probe_poller import GpuProbePoller
            self._gpu_probe = GpuProbePoller(
                self.cfg.gpu_probe_interval_s, self._on_probe_health
            )
            self._gpu_probe.start()
        if self.cfg.rescan_interval_s > 0:
            self._rescan_thread = threading.Thread(
                target=self._rescan_loop, name="kxdp-rescan", daemon=True)
            self._rescan_thread.start()
        if self.cfg.metrics_port:
            from ..metrics import MetricsExporter
            self._metrics = MetricsExporter(self)
            self._metrics.start(self.cfg.metrics_port)
        if self.cfg.gc_tuning:
            self._tune_gc()
        if self.cfg.cpu_affinity:
            self._pin_cpus(self.cfg.cpu_affinity)

    @staticmethod
    def _pin_cpus(spec: str) -> None:
        """Pin this process (all threads) to the given CPU list spec."""
        cpus = set()
        try:
            for part in spec.split(","):
                part = part.strip()
                if not part:
                    continue
                if "-" in part:
                    lo, hi = part.split("-", 1)
                    cpus.update(range(int(lo), int(hi) + 1))
                else:
                    cpus.add(int(part))
            if cpus:
                os.sched_setaffinity(0, cpus)
                log.info("pinned daemon to CPUs %s", sorted(cpus))
        except (ValueError, OSError) as e:
            log.warning("cpu_affinity %r not applied: %s", spec, e)

    @staticmethod
    def _tune_gc() -> None:
        """Tail-latency hygiene: after startup the steady-state RPC path
        allocates only short-lived, acyclic protobuf/response objects that
        die by refcount — but CPython's generational GC still stops the
        world every ~700 container allocations, which lands multi-ms pauses
        in Allocate's p99. Freeze the long-lived
        startup object graph out of the collector and raise gen0's
        threshold so collections are rare; cycles remain collectable, just
        on a coarser cadence."""
        import gc
        gc.collect()
        gc.freeze()
        gc.set_threshold(50_000, 20, 20)

    def _on_socket_removed(self, socket_name: str) -> None:
        for plugin in list(self.plugins.values()):
            if plugin.socket_name == socket_name and not self._stop.is_set():
                # Skip removals the plugin inflicted on itself (stop or
                # restart); only an EXTERNAL wipe (kubelet cleanup) should
                # trigger recovery.
                if plugin.consume_expected_removal():
                    log.info("socket removal of %s was self-inflicted; skipping", socket_name)
                    continue
                if os.path.exists(plugin.socket_path) and plugin.serving:
                    continue
                try:
                    log.warning("EXTERNAL socket removal; restarting %s", plugin.resource_name)
                    plugin.restart()
                except Exception:
                    log.exception("restart of %s failed", plugin.resource_name)

    def _on_cdi_spec_removed(self, name: str) -> None:
        if self._stop.is_set() or self.inventory is None:
            return
        try:
            self.cdi_spec_path = self._write_cdi(self.inventory)
        except Exception:
            log.exception("CDI spec regeneration failed")

    def _on_amdsmi_health(self, bdf: str, healthy: bool, reasons) -> None:
        """amd-smi verdict → DeviceState (matches any function BDF of a
        schedulable group; normally only fires on hybrid/pre-flight nodes
        since vfio-bound GPUs are invisible to amd-smi)."""
        for state in list(self.states.values()):
            for gid in state.device_ids():
                dev = state.device(gid)
                if dev and bdf in (fn.bdf.lower() for fn in dev.functions):
                    if not healthy:
                        log.warning("amd-smi: %s unhealthy: %s", bdf, reasons)
                    state.set_health(gid, healthy, source="amdsmi")
                    return

    def _on_xgmi_telemetry(self, snap) -> None:
        """Per-link xGMI health → hive degradation:
        a GPU with link errors or a down link keeps serving but loses its
        hive membership in placement scoring, so multi-GPU pods prefer
        GPUs whose fabric is intact."""
        if self.topology is None:
            return
        sick = {bdf for bdf, dh in snap.items() if dh.xgmi_sick}
        if sick != set(self.topology.degraded):
            if sick:
                log.warning("xGMI degraded link(s) on %s — excluded from "
                            "hive placement", sorted(sick))
            else:
                log.info("xGMI links recovered; hive placement restored")
            self.topology.set_degraded(sick)

    def _rescan_loop(self) -> None:
        while not self._stop.wait(self.cfg.rescan_interval_s):
            try:
                self.rescan()
            except Exception:
                log.exception("periodic rescan failed")

    def _on_probe_health(self, bdf: str, healthy: bool) -> None:
        """In-daemon GPU probe verdict → DeviceState (source "probe")."""
        bdf = bdf.lower()
        for state in list(self.states.values()):
            for gid in state.device_ids():
                dev = state.device(gid)
                if dev and bdf in (fn.bdf.lower() for fn in dev.functions):
                    if not healthy:
                        log.warning("gpu probe: %s unhealthy", bdf)
                    state.set_health(gid, healthy, source="probe")
                    return

    def _on_kubelet_restarted(self) -> None:
        if self._stop.is_set():
            return
        for plugin in list(self.plugins.values()):
            try:
                plugin.register_with_kubelet()
            except Exception:
                log.exception("re-registration of %s failed", plugin.resource_name)

    # ------------------------------------------------------------------
    def install_signal_handlers(self) -> None:
        """Install SIGTERM/SIGINT → orderly-stop handlers. Call EARLY
        (before setup/serve): a signal during startup must still shut the
        daemon down cleanly, not kill it mid-registration."""
        def _sig(signum, frame):
            log.info("signal %d — shutting down", signum)
            self._stop.set()

        def _hup(signum, frame):
            # operator-triggered live rescan (e.g. right after enabling
            # SR-IOV VFs) — run off the signal frame; rescan() serializes
            # itself via _rescan_lock
            log.info("SIGHUP — rescanning inventory")
            threading.Thread(target=self._safe_rescan, name="kxdp-hup-rescan",
                             daemon=True).start()

        signal.signal(signal.SIGTERM, _sig)
        signal.signal(signal.SIGINT, _sig)
        signal.signal(signal.SIGHUP, _hup)

    def _safe_rescan(self) -> None:
        try:
            if not self._stop.is_set():
                self.rescan()
        except Exception:
            log.exception("SIGHUP rescan failed")

    def run_forever(self) -> None:
        """Block until SIGTERM/SIGINT (the reference blocks on a channel it
        never closes and installs no signal handler)."""
        self.install_signal_handlers()   # idempotent; EARLY install in main()
        self._stop.wait()
        self.stop()

    def stop(self) -> None:
        self._stop.set()
        if self.watcher is not None:
            self.watcher.stop()
            self.watcher = None
        if self._amdsmi is not None:
            self._amdsmi.stop()
            self._amdsmi = None
        if self._gpu_probe is not None:
            self._gpu_probe.stop()
            self._gpu_probe = None
        for plugin in list(self.plugins.values()):
            plugin.stop()
        if self._metrics is not None:
            self._metrics.stop()
            self._metrics = None

    # ------------------------------------------------------------------
    @staticmethod
    def _inventory_signature(inv: NodeInventory):
        return {
            gid: (tuple(dev.bdfs), dev.model_device_id, dev.numa_node,
                  dev.is_vf)
            for gid, dev in inv.devices.items()
        }

    def rescan(self) -> bool:
        """Re-run discovery and swap device sets in place (e.g. after VF
        count changes); plugins keep serving. Returns True if anything
        changed (no-change rescans skip the CDI rewrite and state swap).

        Fully dynamic: topology is reloaded (new VFs
        get hive/NUMA locality immediately — advisor r1 finding), plugins
        for brand-new resource names (e.g. first SR-IOV enable creating
        *_VF) are created, started and registered in place, and plugins
        whose resource vanished entirely are stopped and retired."""
        with self._rescan_lock:
            return self._rescan_locked()

    def _rescan_locked(self) -> bool:
        inv = scan_node(self.cfg)
        if self.inventory is not None and \
                self._inventory_signature(inv) == \
                self._inventory_signature(self.inventory):
            return False
        log.info("rescan: inventory changed (%d → %d devices)",
                 len(self.inventory.devices) if self.inventory else 0,
                 len(inv.devices))
        self.inventory = inv
        prev_degraded = self.topology.degraded if self.topology else frozenset()
        self.topology = load_topology(self.cfg, inv)
        self.topology.degraded = prev_degraded  # carry link-health across rescan
        self.cdi_spec_path = self._write_cdi(inv)
        grouped = self._group_by_resource(inv)

        # Existing resources: swap device sets + refresh topology handle.
        for rname, state in list(self.states.items()):
            if rname in grouped:
                state.replace_devices(grouped[rname])
                self.plugins[rname].topo = self.topology

        # Retire plugins whose resource has no devices left (kubelet drops
        # the resource when the plugin's socket goes away). Drop their
        # socket names from the watcher FIRST so the self-inflicted socket
        # removal isn't mistaken for an external wipe.
        retiring = [r for r in self.plugins if r not in grouped]
        if retiring and self.watcher is not None:
            self.watcher.plugin_socket_names = {
                p.socket_name for r, p in self.plugins.items()
                if r not in retiring}
        for rname in retiring:
            plugin = self.plugins.pop(rname)
            self.states.pop(rname, None)
            try:
                plugin.stop()
            except Exception:
                log.exception("stopping retired plugin %s failed", rname)
            log.info("resource %s retired (no devices after rescan)", rname)

        # Brand-new resource names: serve them now, no daemon restart.
        for rname in sorted(set(grouped) - set(self.plugins)):
            state = DeviceState(grouped[rname])
            plugin = XPUDevicePlugin(self.cfg, rname, state, self.topology)
            self.states[rname] = state
            self.plugins[rname] = plugin
            if self.watcher is not None:  # daemon is live → start serving
                try:
                    plugin.start(register=self._register)
                    log.info("resource %s: started dynamically (%d devices)",
                             rname, len(grouped[rname]))
                except Exception:
                    log.exception("dynamic start of %s failed", rname)

        if self.watcher is not None:
            self.watcher.plugin_socket_names = {
                p.socket_name for p in self.plugins.values()}
        return True


def main(argv: Optional[List[str]] = None) -> int:
    import argparse

    parser = argparse.ArgumentParser(
        prog="kata-xpu-device-plugin-amd",
        description="MI355X-native Kubernetes device plugin for VFIO/Kata GPU passthrough",
    )
    Config.add_args(parser)
    args = parser.parse_args(argv)
    cfg = Config.from_args(args)
    configure_logging(cfg.log_level)
    mgr = PluginManager(cfg)
    mgr.install_signal_handlers()   # before serving: no startup kill window
    mgr.setup()
    mgr.start()
    mgr.run_forever()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
