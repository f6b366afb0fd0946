"""Structured logging for the daemon.

The reference logs with bare ``log.Printf`` and two stray klog calls
(SURVEY.md §5 observability row). Here: one stdlib logging tree under
'kxdp' with a consistent format, level from config/env.
"""
from __future__ import annotations

import logging
import sys

_FORMAT = "%(asctime)s %(levelname).1s %(name)s: %(message)s"
_configured = False


def configure(level: str = "INFO") -> None:
    global _configured
    root = logging.getLogger("kxdp")
    root.setLevel(level.upper())
    if not _configured:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(_FORMAT))
        root.addHandler(h)
        root.propagate = False
        _configured = True


def get_logger(mod: str) -> logging.Logger:
    short = mod.replace("kata_xpu_device_plugin_amd", "kxdp")
    return logging.getLogger(short if short.startswith("kxdp") else f"kxdp.{short}")
