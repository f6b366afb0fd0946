"""Filesystem health watcher: /dev/vfio group nodes + kubelet restarts.

Reference analog: ``healthCheck`` (`generic_device_plugin.go:389-457`):
fsnotify Create on a /dev/vfio/<group> path → healthy, Remove/Rename →
unhealthy, plugin-socket removal → full restart/re-register.

Differences: one watcher thread for the whole daemon (the reference runs
one goroutine per plugin each with its own fsnotify instance); watching
the *directories* (/dev/vfio and the kubelet dir) rather than individual
file paths, which also catches groups that appear later; and detection of
kubelet.sock re-creation (kubelet restart) to trigger re-registration even
if our socket file survived.
"""
from __future__ import annotations

import os
import threading
from typing import Callable, Dict, Optional

from ..config import Config
from ..plugin.state import DeviceState
from ..utils import inotify
from ..utils.log import get_logger

log = get_logger(__name__)

_MASK_DIR = (
    inotify.IN_CREATE
    | inotify.IN_DELETE
    | inotify.IN_MOVED_FROM
    | inotify.IN_MOVED_TO
)


class NodeWatcher(threading.Thread):
    """Watches /dev/vfio and the kubelet socket dir; drives DeviceState
    health and plugin restart callbacks."""

    def __init__(
        self,
        cfg: Config,
        states: Dict[str, DeviceState],           # resource name → state
        on_socket_removed: Optional[Callable[[str], None]] = None,   # socket name
        on_kubelet_restarted: Optional[Callable[[], None]] = None,
        plugin_socket_names: Optional[set] = None,
        on_cdi_spec_removed: Optional[Callable[[str], None]] = None,  # file name
        cdi_spec_names: Optional[set] = None,
    ):
        super().__init__(name="kxdp-node-watcher", daemon=True)
        self.cfg = cfg
        self.states = states
        self.on_socket_removed = on_socket_removed
        self.on_kubelet_restarted = on_kubelet_restarted
        self.plugin_socket_names = plugin_socket_names or set()
        self.on_cdi_spec_removed = on_cdi_spec_removed
        self.cdi_spec_names = cdi_spec_names or set()
        self._stop_evt = threading.Event()
        self._ready = threading.Event()

    # gid → state that owns it
    def _owner(self, gid: str) -> Optional[DeviceState]:
        for st in self.states.values():
            if st.device(gid) is not None:
                return st
        return None

    def run(self) -> None:
        vfio_dir = os.path.join(self.cfg.dev_root, "vfio")
        os.makedirs(vfio_dir, exist_ok=True)
        with inotify.Inotify() as ino:
            wd_vfio = ino.add_watch(vfio_dir, _MASK_DIR)
            wd_kubelet = wd_cdi = -1
            if os.path.isdir(self.cfg.kubelet_socket_dir):
                wd_kubelet = ino.add_watch(self.cfg.kubelet_socket_dir, _MASK_DIR)
            if os.path.isdir(self.cfg.cdi_dir):
                wd_cdi = ino.add_watch(self.cfg.cdi_dir, _MASK_DIR)
            self._ready.set()
            while not self._stop_evt.is_set():
                for ev in ino.read_events(timeout=0.2):
                    try:
                        self._handle(ev, wd_vfio, wd_kubelet, wd_cdi)
                    except Exception:  # watcher must never die silently
                        log.exception("health watcher event error: %s", ev)

    def _handle(self, ev: inotify.Event, wd_vfio: int, wd_kubelet: int,
                wd_cdi: int = -1) -> None:
        if ev.wd == wd_vfio:
            st = self._owner(ev.name)
            if st is None:
                return
            if ev.removed:
                st.set_health(ev.name, False)
            elif ev.created:
                st.set_health(ev.name, True)
        elif ev.wd == wd_kubelet:
            if ev.name in self.plugin_socket_names and ev.removed:
                log.warning("plugin socket %s removed; restarting plugin", ev.name)
                if self.on_socket_removed:
                    self.on_socket_removed(ev.name)
            elif ev.name == self.cfg.kubelet_socket_name and ev.created:
                log.warning("kubelet.sock re-created; kubelet restarted")
                if self.on_kubelet_restarted:
                    self.on_kubelet_restarted()
        elif ev.wd == wd_cdi:
            # self-heal: the CDI spec is the runtime's source of truth for
            # resolving our device names; if an operator or tmp-cleaner
            # removes it, regenerate immediately.
            if ev.name in self.cdi_spec_names and ev.removed:
                log.warning("CDI spec %s removed; regenerating", ev.name)
                if self.on_cdi_spec_removed:
                    self.on_cdi_spec_removed(ev.name)

    def wait_ready(self, timeout: float = 5.0) -> None:
        if not self._ready.wait(timeout):
            raise TimeoutError("health watcher did not start")

    def stop(self) -> None:
        self._stop_evt.set()
        if self.is_alive():
            self.join(timeout=2.0)
