"""Minimal Linux inotify binding (ctypes — no external deps).

Python's stdlib has no inotify interface and this image has no watchdog/
pyinotify package; the daemon needs filesystem events for device health
(/dev/vfio group nodes appearing/disappearing) and kubelet-restart
detection (plugin socket removal), the same mechanisms the reference gets
from fsnotify (`generic_device_plugin.go:396-426`).
"""
from __future__ import annotations

import ctypes
import ctypes.util
import errno
import os
import select
import struct
from dataclasses import dataclass
from typing import List, Optional

IN_CREATE = 0x00000100
IN_DELETE = 0x00000200
IN_MOVED_FROM = 0x00000040
IN_MOVED_TO = 0x00000080
IN_DELETE_SELF = 0x00000400
IN_ATTRIB = 0x00000004
IN_ONLYDIR = 0x01000000
IN_IGNORED = 0x00008000   # kernel dropped the watch (dir deleted)
IN_MOVE_SELF = 0x00000800  # watched dir renamed away (watch follows inode!)

_EVENT_HDR = struct.Struct("iIII")  # wd, mask, cookie, len

_libc = ctypes.CDLL(ctypes.util.find_library("c") or "libc.so.6", use_errno=True)


@dataclass(frozen=True)
class Event:
    wd: int
    mask: int
    name: str

    @property
    def created(self) -> bool:
        return bool(self.mask & (IN_CREATE | IN_MOVED_TO))

    @property
    def removed(self) -> bool:
        return bool(self.mask & (IN_DELETE | IN_MOVED_FROM | IN_DELETE_SELF))

    @property
    def ignored(self) -> bool:
        """Watch no longer observes the original path: the dir was deleted
        (IN_IGNORED) or renamed away (IN_MOVE_SELF — inotify watches follow
        the inode, so events would silently track the MOVED dir)."""
        return bool(self.mask & (IN_IGNORED | IN_MOVE_SELF))


class Inotify:
    def __init__(self):
        self.fd = _libc.inotify_init1(os.O_NONBLOCK)
        if self.fd < 0:
            raise OSError(ctypes.get_errno(), "inotify_init1 failed")
        self._watches = {}  # wd → path

    def add_watch(self, path: str, mask: int) -> int:
        wd = _libc.inotify_add_watch(self.fd, path.encode(), mask)
        if wd < 0:
            raise OSError(ctypes.get_errno(), f"inotify_add_watch({path}) failed")
        self._watches[wd] = path
        return wd

    def rm_watch(self, wd: int) -> None:
        _libc.inotify_rm_watch(self.fd, wd)
        self._watches.pop(wd, None)

    def path_of(self, wd: int) -> Optional[str]:
        return self._watches.get(wd)

    def read_events(self, timeout: Optional[float] = None) -> List[Event]:
        """Block up to `timeout` seconds; return all pending events."""
        r, _, _ = select.select([self.fd], [], [], timeout)
        if not r:
            return []
        events: List[Event] = []
        while True:
            try:
                data = os.read(self.fd, 65536)
            except OSError as e:
                if e.errno in (errno.EAGAIN, errno.EWOULDBLOCK):
                    break
                raise
            off = 0
            while off + _EVENT_HDR.size <= len(data):
                wd, mask, _cookie, nlen = _EVENT_HDR.unpack_from(data, off)
                off += _EVENT_HDR.size
                name = data[off:off + nlen].split(b"\0", 1)[0].decode(
                    "utf-8", "replace")
                off += nlen
                events.append(Event(wd=wd, mask=mask, name=name))
            if len(data) < 65536:
                break
        return events

    def close(self) -> None:
        if self.fd >= 0:
            os.close(self.fd)
            self.fd = -1

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
