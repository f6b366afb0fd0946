"""amd-smi based device health (the AMD-native NVML analog).

The reference has no vendor-library health at all (fsnotify only —
SURVEY.md §2.2 row "fsnotify-on-/dev/vfio health": amd-smi is additive).
Scope note: amd-smi (like KFD) only sees **amdgpu-bound** GPUs. For a
passthrough node whose GPUs live behind vfio-pci, amd-smi covers:

* pre-flight health gating before GPUs are handed to the vfio driver,
* "hybrid" nodes where some GPUs stay on amdgpu,
* post-return validation when a GPU comes back from a VM.

The poller maps amd-smi processors to PCI BDFs and reports per-BDF
verdicts (RAS error counts, thermal/power faults); the manager marks
matching DeviceState entries unhealthy. All amdsmi calls are defensive —
a missing/broken libamd_smi never takes the daemon down.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from ..utils.log import get_logger

log = get_logger(__name__)

# GPU is unhealthy if any of these RAS blocks report uncorrectable errors.
_FATAL_TEMP_C = 105.0

# RAS IP blocks worth attributing faults to (AmdSmiGpuBlock names):
# UMC = HBM memory controller, GFX = compute, SDMA = copy engines,
# XGMI_WAFL = fabric, MMHUB = memory hub, PCIE_BIF = host link.
_RAS_BLOCKS = ("UMC", "GFX", "SDMA", "XGMI_WAFL", "MMHUB", "PCIE_BIF")


@dataclass
class XgmiLinkHealth:
    index: int
    status: str = "unknown"      # up | down | disabled | unknown
    errors: int = 0              # cumulative link error count


@dataclass
class DeviceHealth:
    bdf: str
    healthy: bool = True
    reasons: List[str] = field(default_factory=list)
    temperature_c: Optional[float] = None
    uncorrectable_errors: int = 0
    # Per-xGMI-link telemetry. A sick link does NOT
    # make the GPU Unhealthy (it still computes) — it degrades the hive in
    # placement (topology/hive.py::GPUTopology.degraded) and is exported
    # per-link to Prometheus.
    xgmi_links: List[XgmiLinkHealth] = field(default_factory=list)
    # Per-RAS-block uncorrectable counts (fault ATTRIBUTION — which IP
    # block is dying: HBM controller vs compute vs fabric), best-effort.
    ecc_by_block: Dict[str, int] = field(default_factory=dict)

    @property
    def xgmi_sick(self) -> bool:
        return any(l.errors > 0 or l.status == "down" for l in self.xgmi_links)


def _normalize_bdf(raw: str) -> str:
    # amdsmi returns "0000:0a:00.0" already; be liberal in what we accept.
    return raw.strip().lower()


def _read_xgmi_links(amdsmi, h) -> List[XgmiLinkHealth]:
    """Best-effort per-link xGMI status/error read. The amdsmi surface for
    this moved across ROCm releases, so try the dedicated APIs first and
    fall back to gpu_metrics' xgmi_link_status array; every call is
    defensive (missing API ⇒ fewer fields, never an exception out)."""
    links: List[XgmiLinkHealth] = []

    def status_name(v) -> str:
        name = str(v).rsplit(".", 1)[-1].strip().lower()
        if name.isdigit():  # numeric enum: 0=disabled,1=up,2=down (smi)
            name = {0: "disabled", 1: "up", 2: "down"}.get(int(name),
                                                           "unknown")
        return name if name in ("up", "down", "disabled") else "unknown"

    # 1) per-link status (amdsmi_get_gpu_xgmi_link_status, ROCm >= 6.3).
    # Return shape varies by release: list of enums, dict {"status": [..]}
    # or a single scalar/string — only a real sequence is per-link
    # (live-MI355X lesson: a str here would iterate as characters).
    try:
        st = amdsmi.amdsmi_get_gpu_xgmi_link_status(h)
        raw = st.get("status") if isinstance(st, dict) else st
        if isinstance(raw, (list, tuple)):
            for i, v in enumerate(raw):
                links.append(XgmiLinkHealth(index=i, status=status_name(v)))
        elif raw is not None and not isinstance(raw, (str, bytes)):
            links.append(XgmiLinkHealth(index=0, status=status_name(raw)))
    except Exception:
        pass
    if not links:
        # 2) gpu_metrics fallback
        try:
            gm = amdsmi.amdsmi_get_gpu_metrics_info(h)
            raw = gm.get("xgmi_link_status", []) if isinstance(gm, dict) else []
            if not isinstance(raw, (list, tuple)):
                raw = []
            for i, v in enumerate(raw):
                try:
                    iv = int(v)
                except (TypeError, ValueError):
                    continue
                if iv in (0xFFFF, 0xFFFFFFFF):  # UNSUPPORTED sentinel
                    continue
                links.append(XgmiLinkHealth(
                    index=i, status={0: "down", 1: "up"}.get(iv, "unknown")))
        except Exception:
            pass
    # 3) cumulative error status (per-device; attribute to all links when
    # the per-link error API is absent)
    err_total = 0
    try:
        es = amdsmi.amdsmi_gpu_xgmi_error_status(h)
        # enum: 0 NO_ERRORS, 1 ERROR, 2 MULTIPLE_ERRORS
        name = str(es).rsplit(".", 1)[-1].upper()
        if name in ("ERROR",):
            err_total = 1
        elif name in ("MULTIPLE_ERRORS",):
            err_total = 2
        elif name.isdigit():
            err_total = int(name)
    except Exception:
        pass
    if err_total and not links:
        links.append(XgmiLinkHealth(index=0, status="unknown",
                                    errors=err_total))
    elif err_total:
        links[0].errors = err_total
    return links


def snapshot() -> Dict[str, DeviceHealth]:
    """One amd-smi pass → {bdf: DeviceHealth}. Empty dict if amd-smi is
    unavailable or sees no devices (e.g. all GPUs vfio-bound)."""
    try:
        import amdsmi
    except ImportError:
        return {}
    out: Dict[str, DeviceHealth] = {}
    try:
        amdsmi.amdsmi_init()
    except Exception as e:
        log.debug("amdsmi init failed: %s", e)
        return {}
    try:
        try:
            handles = amdsmi.amdsmi_get_processor_handles()
        except Exception as e:
            log.debug("amdsmi enumerate failed: %s", e)
            return {}
        for h in handles:
            try:
                bdf = _normalize_bdf(str(amdsmi.amdsmi_get_gpu_device_bdf(h)))
            except Exception:
                continue
            dh = DeviceHealth(bdf=bdf)
            # RAS / ECC error counts
            try:
                ecc = amdsmi.amdsmi_get_gpu_total_ecc_count(h)
                ue = int(ecc.get("uncorrectable_count", 0) or 0)
                dh.uncorrectable_errors = ue
                if ue > 0:
                    dh.healthy = False
                    dh.reasons.append(f"{ue} uncorrectable ECC errors")
            except Exception:
                pass
            # Per-block attribution (which IP block is failing); also a
            # fallback verdict when the totals API is unavailable.
            try:
                blocks = getattr(amdsmi, "AmdSmiGpuBlock", None)
                for bname in _RAS_BLOCKS if blocks is not None else ():
                    blk = getattr(blocks, bname, None)
                    if blk is None:
                        continue
                    try:
                        c = amdsmi.amdsmi_get_gpu_ecc_count(h, blk)
                        bue = int(c.get("uncorrectable_count", 0) or 0)
                    except Exception:
                        continue
                    dh.ecc_by_block[bname] = bue
                    if bue > 0:
                        dh.healthy = False
                        reason = f"{bue} uncorrectable in {bname}"
                        if reason not in dh.reasons:
                            dh.reasons.append(reason)
            except Exception:
                pass
            # Thermals (edge/junction)
            try:
                t = amdsmi.amdsmi_get_temp_metric(
                    h,
                    amdsmi.AmdSmiTemperatureType.JUNCTION,
                    amdsmi.AmdSmiTemperatureMetric.CURRENT,
                )
                dh.temperature_c = float(t)
                if dh.temperature_c >= _FATAL_TEMP_C:
                    dh.healthy = False
                    dh.reasons.append(f"junction {dh.temperature_c:.0f}°C")
            except Exception:
                pass
            dh.xgmi_links = _read_xgmi_links(amdsmi, h)
            out[dh.bdf] = dh
    finally:
        try:
            amdsmi.amdsmi_shut_down()
        except Exception:
            pass
    return out


class AmdSmiPoller(threading.Thread):
    """Background poller calling `on_health(bdf, healthy, reasons)` on
    state transitions."""

    def __init__(
        self,
        interval_s: float,
        on_health: Callable[[str, bool, List[str]], None],
        snapshot_fn: Callable[[], Dict[str, DeviceHealth]] = snapshot,
        on_xgmi: Optional[Callable[[Dict[str, DeviceHealth]], None]] = None,
    ):
        super().__init__(name="kxdp-amdsmi-poller", daemon=True)
        self.interval_s = interval_s
        self.on_health = on_health
        self.on_xgmi = on_xgmi
        self.snapshot_fn = snapshot_fn
        self._stop_evt = threading.Event()
        self._last: Dict[str, bool] = {}
        self.last_snapshot: Dict[str, DeviceHealth] = {}  # metrics export

    def poll_once(self) -> None:
        snap = self.snapshot_fn()
        self.last_snapshot = snap
        for bdf, dh in snap.items():
            prev = self._last.get(bdf)
            if prev is None or prev != dh.healthy:
                self._last[bdf] = dh.healthy
                if prev is not None or not dh.healthy:
                    self.on_health(bdf, dh.healthy, dh.reasons)
        if self.on_xgmi is not None and snap:
            self.on_xgmi(snap)

    def run(self) -> None:
        while not self._stop_evt.wait(self.interval_s):
            try:
                self.poll_once()
            except Exception:
                log.exception("amdsmi poll failed")

    def stop(self) -> None:
        self._stop_evt.set()
        if self.is_alive():
            self.join(timeout=2.0)
