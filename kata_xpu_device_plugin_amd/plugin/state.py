"""Thread-safe device state shared by discovery, health and ListAndWatch.

The reference mutates ``devs[].Health`` from the ListAndWatch goroutine
while other goroutines read it, and its package-global maps are
unsynchronized (SURVEY.md §5 "real races exist"). Here all mutation goes
through one lock and ListAndWatch streams receive immutable snapshots via
per-stream queues.
"""
from __future__ import annotations

import queue
import threading
from typing import Dict, List, Optional, Tuple

from ..discovery.sysfs import XPUDevice
from ..utils.log import get_logger

log = get_logger(__name__)


class DeviceState:
    """Health-tracked set of XPUDevices for one resource name."""

    def __init__(self, devices: Dict[str, XPUDevice]):
        self._lock = threading.Lock()
        self._devices: Dict[str, XPUDevice] = dict(devices)
        self._healthy: Dict[str, bool] = {gid: True for gid in devices}
        self._watchers: List[queue.Queue] = []
        self._generation = 0

    # -- snapshots ------------------------------------------------------
    def snapshot(self) -> List[Tuple[XPUDevice, bool]]:
        with self._lock:
            return [(self._devices[g], self._healthy[g]) for g in sorted(
                self._devices, key=_gkey)]

    def device(self, gid: str) -> Optional[XPUDevice]:
        with self._lock:
            return self._devices.get(gid)

    def device_ids(self) -> List[str]:
        with self._lock:
            return sorted(self._devices, key=_gkey)

    def healthy_ids(self) -> List[str]:
        with self._lock:
            return sorted((g for g, h in self._healthy.items() if h), key=_gkey)

    def is_healthy(self, gid: str) -> bool:
        with self._lock:
            return self._healthy.get(gid, False)

    # -- mutation -------------------------------------------------------
    def set_health(self, gid: str, healthy: bool) -> bool:
        """Returns True if the state changed (and watchers were notified)."""
        with self._lock:
            if gid not in self._devices or self._healthy.get(gid) == healthy:
                return False
            self._healthy[gid] = healthy
            self._generation += 1
            self._notify_locked()
        log.info("device %s → %s", gid, "Healthy" if healthy else "Unhealthy")
        return True

    def replace_devices(self, devices: Dict[str, XPUDevice]) -> None:
        """Swap in a fresh discovery result (rescan)."""
        with self._lock:
            old_health = self._healthy
            self._devices = dict(devices)
            self._healthy = {g: old_health.get(g, True) for g in devices}
            self._generation += 1
            self._notify_locked()

    def _notify_locked(self) -> None:
        for q in self._watchers:
            try:
                q.put_nowait(self._generation)
            except queue.Full:
                pass  # watcher will resync from snapshot anyway

    # -- watch ----------------------------------------------------------
    def watch(self) -> queue.Queue:
        q: queue.Queue = queue.Queue(maxsize=64)
        with self._lock:
            self._watchers.append(q)
        return q

    def unwatch(self, q: queue.Queue) -> None:
        with self._lock:
            try:
                self._watchers.remove(q)
            except ValueError:
                pass


def _gkey(g: str):
    return (0, int(g)) if g.isdigit() else (1, g)
