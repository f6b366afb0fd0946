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
from typing import Callable, Dict, List, Optional, Tuple

from ..discovery.sysfs import XPUDevice
from ..utils.log import get_logger

log = get_logger(__name__)


class DeviceState:
    """Health-tracked set of XPUDevices for one resource name."""

    def __init__(self, devices: Dict[str, XPUDevice]):
        self._lock = threading.Lock()
        self._devices: Dict[str, XPUDevice] = dict(devices)
        # Health is tracked per SOURCE ("vfio" watcher, "amdsmi" poller,
        # "probe" burn-in, ...); a device is Healthy iff no source holds it
        # unhealthy — so an amd-smi recovery cannot mask a missing
        # /dev/vfio node and vice versa.
        self._unhealthy_by: Dict[str, set] = {gid: set() for gid in devices}
        self._watchers: List[queue.Queue] = []
        self._callbacks: List[Callable[[], None]] = []
        self._generation = 0

    # -- snapshots ------------------------------------------------------
    def snapshot(self) -> List[Tuple[XPUDevice, bool]]:
        with self._lock:
            return [(self._devices[g], not self._unhealthy_by[g])
                    for g in sorted(self._devices, key=_gkey)]

    def device(self, gid: str) -> Optional[XPUDevice]:
        with self._lock:
            return self._devices.get(gid)

    def device_ids(self) -> List[str]:
        with self._lock:
            return sorted(self._devices, key=_gkey)

    def healthy_ids(self) -> List[str]:
        with self._lock:
            return sorted((g for g, u in self._unhealthy_by.items() if not u),
                          key=_gkey)

    def is_healthy(self, gid: str) -> bool:
        with self._lock:
            u = self._unhealthy_by.get(gid)
            return u is not None and not u

    @property
    def generation(self) -> int:
        """Monotonic change counter (device set or effective health);
        lets hot paths cache derived maps keyed on it."""
        with self._lock:
            return self._generation

    def bdf_map(self) -> Dict[str, str]:
        """gid → primary BDF for the current device set (one lock trip)."""
        with self._lock:
            return {gid: dev.primary.bdf for gid, dev in self._devices.items()}

    # -- mutation -------------------------------------------------------
    def set_health(self, gid: str, healthy: bool, source: str = "vfio") -> bool:
        """Record one source's verdict; returns True if the device's
        EFFECTIVE health changed (and watchers were notified)."""
        with self._lock:
            if gid not in self._devices:
                return False
            u = self._unhealthy_by[gid]
            before = not u
            if healthy:
                u.discard(source)
            else:
                u.add(source)
            after = not u
            if before == after:
                return False
            self._generation += 1
            self._notify_locked()
        log.info("device %s → %s (source %s)", gid,
                 "Healthy" if after else "Unhealthy", source)
        return True

    def replace_devices(self, devices: Dict[str, XPUDevice]) -> None:
        """Swap in a fresh discovery result (rescan)."""
        with self._lock:
            old = self._unhealthy_by
            self._devices = dict(devices)
            self._unhealthy_by = {g: set(old.get(g, set())) for g in devices}
            self._generation += 1
            self._notify_locked()

    def _notify_locked(self) -> None:
        for q in self._watchers:
            try:
                q.put_nowait(self._generation)
            except queue.Full:
                pass  # watcher will resync from snapshot anyway
        for cb in self._callbacks:
            try:
                cb()
            except Exception:
                pass  # a dead subscriber must not break state mutation

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

    def subscribe(self, cb: Callable[[], None]) -> Callable[[], None]:
        """Register a zero-arg notification callback fired after every
        effective change (and on poke()). The callback runs under the state
        lock from the mutating thread: it MUST be non-blocking — the
        intended use is ``loop.call_soon_threadsafe(event.set)`` from the
        asyncio ListAndWatch bridge. Returns an idempotent unsubscriber."""
        with self._lock:
            self._callbacks.append(cb)

        def unsubscribe() -> None:
            with self._lock:
                try:
                    self._callbacks.remove(cb)
                except ValueError:
                    pass
        return unsubscribe

    def poke(self) -> None:
        """Wake all watchers without a state change (shutdown path: lets
        event-driven streams observe their stop flag immediately instead of
        waiting out the gRPC shutdown grace)."""
        with self._lock:
            self._notify_locked()


def _gkey(g: str):
    return (0, int(g)) if g.isdigit() else (1, g)
