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


@dataclass
class DeviceHealth:
    bdf: str
    healthy: bool = True
    reasons: List[str] = field(default_factory=list)
    temperature_c: Optional[float] = None
    uncorrectable_errors: int = 0


def _normalize_bdf(raw: str) -> str:
    # amdsmi returns "0000:0a:00.0" already; be liberal in what we accept.
    return raw.strip().lower()


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
    ):
        super().__init__(name="kxdp-amdsmi-poller", daemon=True)
        self.interval_s = interval_s
        self.on_health = on_health
        self.snapshot_fn = snapshot_fn
        self._stop_evt = threading.Event()
        self._last: Dict[str, bool] = {}

    def poll_once(self) -> None:
        for bdf, dh in self.snapshot_fn().items():
            prev = self._last.get(bdf)
            if prev is None or prev != dh.healthy:
                self._last[bdf] = dh.healthy
                if prev is not None or not dh.healthy:
                    self.on_health(bdf, dh.healthy, dh.reasons)

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
