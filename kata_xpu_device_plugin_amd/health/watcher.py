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
    | inotify.IN_DELETE_SELF
    | inotify.IN_MOVE_SELF
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
        for st in list(self.states.values()):
            if st.device(gid) is not None:
                return st
        return None

    # -- re-armable watch set -----------------------
    # A target dir that does not exist yet (kubelet dir created after
    # daemon start, CDI dir wiped by a tmp-cleaner) is covered by a watch
    # on its nearest EXISTING ancestor; any creation under an ancestor or
    # an IN_IGNORED (watched dir deleted) triggers re-arming, and arming a
    # target late reconciles the state the missed events would have built.

    @staticmethod
    def _nearest_existing_ancestor(path: str) -> str:
        p = os.path.dirname(os.path.abspath(path))
        while p and not os.path.isdir(p):
            parent = os.path.dirname(p)
            if parent == p:
                break
            p = parent
        return p or "/"

    def _arm_all(self, ino: inotify.Inotify) -> None:
        """(Re)compute the watch set; idempotent — inotify returns the
        same wd for an already-watched path."""
        vfio_dir = os.path.join(self.cfg.dev_root, "vfio")
        os.makedirs(vfio_dir, exist_ok=True)   # our own dir: always direct
        targets = {
            "vfio": vfio_dir,
            "kubelet": self.cfg.kubelet_socket_dir,
            "cdi": self.cfg.cdi_dir,
        }
        # An entirely-missing CDI dir is the spec-removed case writ large:
        # regeneration (write_spec) recreates dir + spec, and the loop
        # below then arms the fresh dir directly.
        if not self._initial_arm and not os.path.isdir(targets["cdi"]) \
                and self.on_cdi_spec_removed:
            for name in sorted(self.cdi_spec_names):
                try:
                    self.on_cdi_spec_removed(name)
                except Exception:
                    log.exception("CDI regeneration after dir wipe failed")
        new_roles: Dict[int, str] = {}
        for role, path in targets.items():
            armed_before = role in self._armed
            if os.path.isdir(path):
                try:
                    wd = ino.add_watch(path, _MASK_DIR)
                except OSError:
                    continue  # raced with deletion; next event re-arms
                new_roles[wd] = role
                self._armed.add(role)
                if not armed_before:
                    self._reconcile(role, path)
            else:
                self._armed.discard(role)
                anc = self._nearest_existing_ancestor(path)
                try:
                    wd = ino.add_watch(anc, _MASK_DIR)
                except OSError:
                    continue
                new_roles.setdefault(wd, "parent")
        self._roles = new_roles

    def _reconcile(self, role: str, path: str) -> None:
        """A target watch was armed LATE: apply the state its missed
        events would have produced."""
        if role == "vfio":
            # re-derive per-device health from node presence
            for st in list(self.states.values()):
                for gid in st.device_ids():
                    st.set_health(gid, os.path.exists(os.path.join(path, gid)))
        elif role == "kubelet":
            if self._initial_arm:
                return
            # events were missed while we were blind: sockets that should
            # exist but don't were wiped; kubelet.sock present means
            # kubelet may have (re)started without us seeing it.
            self._check_missing_sockets()
            if os.path.exists(self.cfg.kubelet_socket) and \
                    self.on_kubelet_restarted:
                log.warning("kubelet dir appeared after daemon start; "
                            "re-registering")
                self.on_kubelet_restarted()
        elif role == "cdi":
            for name in self.cdi_spec_names:
                if not os.path.exists(os.path.join(path, name)) and \
                        self.on_cdi_spec_removed and not self._initial_arm:
                    self.on_cdi_spec_removed(name)

    def run(self) -> None:
        self._roles: Dict[int, str] = {}
        self._armed: set = set()
        with inotify.Inotify() as ino:
            self._ino = ino
            self._initial_arm = True
            self._arm_all(ino)
            self._initial_arm = False
            self._ready.set()
            while not self._stop_evt.is_set():
                rearm = False
                for ev in ino.read_events(timeout=0.2):
                    try:
                        rearm |= self._handle(ev)
                    except Exception:  # watcher must never die silently
                        log.exception("health watcher event error: %s", ev)
                # While any target dir is unarmed, retry every tick: a deep
                # mkdir -p chain outruns event-driven re-arming (each dir
                # watch only sees DIRECT children, and the next path
                # component may exist before its ancestor watch is armed).
                rearm |= len(self._armed) < 3
                if rearm and not self._stop_evt.is_set():
                    try:
                        self._arm_all(ino)
                    except Exception:
                        log.exception("watch re-arm failed")

    def _handle(self, ev: inotify.Event) -> bool:
        """Process one event; returns True if the watch set must re-arm."""
        role = self._roles.get(ev.wd)
        if role is None:
            return False
        if ev.ignored:
            # the watched dir itself is gone (deleted or renamed away) —
            # drop the stale watch and re-arm via ancestor
            self._roles.pop(ev.wd, None)
            self._armed.discard(role)
            try:
                self._ino.rm_watch(ev.wd)  # MOVE_SELF: watch still live on
            except OSError:                # the moved inode — detach it
                pass
            if role == "kubelet":
                # the dir vanished wholesale: per-file removal events may
                # never arrive — treat every now-missing plugin socket as
                # wiped so its plugin restarts (and recreates the dir).
                self._check_missing_sockets()
            return True
        return self._handle_role(ev, role)

    def _check_missing_sockets(self) -> None:
        if not self.on_socket_removed:
            return
        for name in sorted(self.plugin_socket_names):
            if not os.path.exists(os.path.join(self.cfg.kubelet_socket_dir,
                                               name)):
                log.warning("plugin socket %s gone with its dir; restarting",
                            name)
                self.on_socket_removed(name)

    def _handle_role(self, ev: inotify.Event, role: str) -> bool:
        # While any target dir is still unarmed, every create/remove event
        # is a re-arm hint — the missing dir's nearest existing ancestor
        # may itself be one of the watched target dirs.
        rearm_hint = (ev.created or ev.removed) and len(self._armed) < 3
        if role == "vfio":
            st = self._owner(ev.name)
            if st is None:
                return rearm_hint
            if ev.removed:
                st.set_health(ev.name, False)
            elif ev.created:
                st.set_health(ev.name, True)
        elif role == "kubelet":
            if ev.name in self.plugin_socket_names and ev.removed:
                log.warning("plugin socket %s removed; restarting plugin", ev.name)
                if self.on_socket_removed:
                    self.on_socket_removed(ev.name)
            elif ev.name == self.cfg.kubelet_socket_name and ev.created:
                log.warning("kubelet.sock re-created; kubelet restarted")
                if self.on_kubelet_restarted:
                    self.on_kubelet_restarted()
        elif role == "cdi":
            # self-heal: the CDI spec is the runtime's source of truth for
            # resolving our device names; if an operator or tmp-cleaner
            # removes it, regenerate immediately.
            if ev.name in self.cdi_spec_names and ev.removed:
                log.warning("CDI spec %s removed; regenerating", ev.name)
                if self.on_cdi_spec_removed:
                    self.on_cdi_spec_removed(ev.name)
        elif role == "parent":
            # something changed under an ancestor of a missing target —
            # the target dir may exist now (dir creation, rename into
            # place); re-arm to find out.
            return ev.created or ev.removed
        return rearm_hint

    def wait_ready(self, timeout: float = 5.0) -> None:
        if not self._ready.wait(timeout):
            raise TimeoutError("health watcher did not start")

    def stop(self) -> None:
        self._stop_evt.set()
        if self.is_alive():
            self.join(timeout=2.0)
