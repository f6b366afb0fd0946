"""Structured logging for the daemon.

The reference logs with bare ``log.Printf`` and two stray klog calls
(SURVEY.md §5 observability row). Here: one stdlib logging tree under
'kxdp' with a consistent format, level from config/env, and an optional
JSON format (``KXDP_LOG_FORMAT=json`` / configure(fmt="json")) for
cluster log pipelines.
"""
from __future__ import annotations

import json
import logging
import os
import sys

_FORMAT = "%(asctime)s %(levelname).1s %(name)s: %(message)s"
_configured = False


class _JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        doc = {
            "ts": self.formatTime(record, "%Y-%m-%dT%H:%M:%S%z"),
            "level": record.levelname,
            "logger": record.name,
            "msg": record.getMessage(),
        }
        if record.exc_info:
            doc["exc"] = self.formatException(record.exc_info)
        return json.dumps(doc)


def configure(level: str = "INFO", fmt: str = "") -> None:
    global _configured
    fmt = fmt or os.environ.get("KXDP_LOG_FORMAT", "text")
    root = logging.getLogger("kxdp")
    root.setLevel(level.upper())
    if not _configured:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(_JsonFormatter() if fmt == "json"
                       else logging.Formatter(_FORMAT))
        root.addHandler(h)
        root.propagate = False
        _configured = True


def get_logger(mod: str) -> logging.Logger:
    short = mod.replace("kata_xpu_device_plugin_amd", "kxdp")
    return logging.getLogger(short if short.startswith("kxdp") else f"kxdp.{short}")
