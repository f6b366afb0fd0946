"""Periodic in-daemon GPU probing (third health source, "probe").

On hybrid nodes where some schedulable devices are amdgpu-bound (e.g.
``--driver amdgpu`` development mode, or PF-resident VFs whose PF stays on
amdgpu), the daemon can periodically run a light _gpuprobe pass per
visible GPU and mark the matching DeviceState entry unhealthy on failure.
HIP device → PCI BDF mapping comes from hipDeviceProp
(pci_domain_id/pci_bus_id/pci_device_id, exposed by _gpuprobe.device_info).

Off by default (``gpu_probe_interval_s = 0``): on a pure-VFIO passthrough
node no GPU is HIP-visible and the poller would idle; pre-flight burn-in
(tools/burnin) is the passthrough-node path.
"""
from __future__ import annotations

import threading
from typing import Callable, Dict, List, Optional

from ..utils.log import get_logger

log = get_logger(__name__)


def _device_bdf(info: Dict) -> str:
    return (f"{info['pci_domain_id']:04x}:{info['pci_bus_id']:02x}:"
            f"{info['pci_device_id']:02x}.0")


def probe_snapshot(quick_bytes: int = 64 << 20,
                   burn_iters: int = 1000) -> Dict[str, bool]:
    """One light probe pass over HIP-visible GPUs → {bdf: healthy}."""
    try:
        from .. import _gpuprobe as g
    except ImportError:
        return {}
    out: Dict[str, bool] = {}
    try:
        n = g.device_count()
    except Exception:
        return {}
    for dev in range(n):
        try:
            info = g.device_info(dev)
            bdf = _device_bdf(info)
            f32 = g.mfma_probe_f32(dev)
            mt = g.memtest(dev, quick_bytes)
            bf16 = g.mfma_probe_bf16(dev, burn_iters)
            out[bdf] = bool(f32["ok"]) and int(mt["mismatches"]) == 0 \
                and bool(bf16["ok"])
        except Exception as e:
            log.warning("probe of HIP device %d failed: %s", dev, e)
            try:
                out[_device_bdf(g.device_info(dev))] = False
            except Exception:
                pass
    return out


class GpuProbePoller(threading.Thread):
    """Calls `on_health(bdf, healthy)` on probe-state transitions."""

    def __init__(
        self,
        interval_s: float,
        on_health: Callable[[str, bool], None],
        snapshot_fn: Callable[[], Dict[str, bool]] = probe_snapshot,
    ):
        super().__init__(name="kxdp-gpu-probe", daemon=True)
        self.interval_s = interval_s
        self.on_health = on_health
        self.snapshot_fn = snapshot_fn
        self._stop_evt = threading.Event()
        self._last: Dict[str, bool] = {}

    def poll_once(self) -> None:
        for bdf, healthy in self.snapshot_fn().items():
            prev = self._last.get(bdf)
            if prev is None or prev != healthy:
                self._last[bdf] = healthy
                if prev is not None or not healthy:
                    self.on_health(bdf, healthy)

    def run(self) -> None:
        while not self._stop_evt.wait(self.interval_s):
            try:
                self.poll_once()
            except Exception:
                log.exception("gpu probe poll failed")

    def stop(self) -> None:
        self._stop_evt.set()
        if self.is_alive():
            self.join(timeout=2.0)
