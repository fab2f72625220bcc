"""Structured logging with the `event=` operator contract.

Reference: internal/logger (zerolog + lumberjack rotation + optional OTLP).
Here: stdlib logging with a key=value line format whose first key is always
`event` — grep-able triage without stack traces — plus size-based rotation.
"""
from __future__ import annotations

import logging
import logging.handlers
import os
import sys
import threading
import time
from pathlib import Path
from typing import Any

_lock = threading.Lock()
_configured = False


class _KVFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        ts = time.strftime("%Y-%m-%dT%H:%M:%S", time.localtime(record.created))
        kv = getattr(record, "kv", None) or {}
        parts = [
            f"{ts}.{int(record.msecs):03d}",
            record.levelname.lower(),
            f"event={record.getMessage()}",
        ]
        for k, v in kv.items():
            s = str(v)
            if " " in s or '"' in s:
                s = '"' + s.replace('"', '\\"') + '"'
            parts.append(f"{k}={s}")
        if record.exc_info:
            parts.append(f'error="{record.exc_info[1]}"')
        return " ".join(parts)


class EventLogger:
    """Thin wrapper: log.info("sandbox_started", sandbox=name, pid=pid)."""

    def __init__(self, component: str):
        self._log = logging.getLogger(f"clawker.{component}")
        self.component = component

    def _emit(self, level: int, event: str, **kv: Any) -> None:
        kv.setdefault("component", self.component)
        self._log.log(level, event, extra={"kv": kv})

    def debug(self, event: str, **kv: Any) -> None:
        self._emit(logging.DEBUG, event, **kv)

    def info(self, event: str, **kv: Any) -> None:
        self._emit(logging.INFO, event, **kv)

    def warn(self, event: str, **kv: Any) -> None:
        self._emit(logging.WARNING, event, **kv)

    def error(self, event: str, **kv: Any) -> None:
        self._emit(logging.ERROR, event, **kv)


def setup(log_file: Path | None = None, level: str = "", stderr: bool = False,
          max_size_mb: int = 10, max_backups: int = 3) -> None:
    """Idempotent global setup: rotating file sink + optional stderr.
    Rotation knobs come from settings logging.* (reference: lumberjack
    rotation, schema.go logging :457)."""
    global _configured
    with _lock:
        if _configured:
            return
        root = logging.getLogger("clawker")
        root.setLevel(getattr(logging, (level or os.environ.get("CLAWKER_LOG_LEVEL", "INFO")).upper(), logging.INFO))
        fmt = _KVFormatter()
        if log_file is not None:
            log_file.parent.mkdir(parents=True, exist_ok=True)
            fh = logging.handlers.RotatingFileHandler(
                log_file, maxBytes=max(1, max_size_mb) * 1024 * 1024,
                backupCount=max(0, max_backups))
            fh.setFormatter(fmt)
            root.addHandler(fh)
        if stderr or os.environ.get("CLAWKER_LOG_STDERR"):
            sh = logging.StreamHandler(sys.stderr)
            sh.setFormatter(fmt)
            root.addHandler(sh)
        if not root.handlers:
            root.addHandler(logging.NullHandler())
        _configured = True


def get(component: str) -> EventLogger:
    return EventLogger(component)
