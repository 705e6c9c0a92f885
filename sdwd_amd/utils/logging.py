"""Three-sink logger: Rich console, rotating file, in-memory ring buffer.

Capability parity with the reference's shared.py:16-61 (Rich console handler
with branded level prefix, a 10 MB rotating ``distributed.log`` and a
16-entry ring buffer surfaced in the status UI), re-designed as a plain
library module with no webui host.
"""
from __future__ import annotations

import collections
import logging
import logging.handlers
import os
import threading
from typing import Deque, List

_LOG_NAME = "sdwd_amd"
_RING_CAPACITY = 64

_lock = threading.Lock()
_ring: Deque[str] = collections.deque(maxlen=_RING_CAPACITY)
_configured = False


class RingBufferHandler(logging.Handler):
    """Keeps the last N formatted records for status surfaces (UI/API)."""

    def emit(self, record: logging.LogRecord) -> None:
        try:
            msg = self.format(record)
        except Exception:  # pragma: no cover - formatting failure
            return
        with _lock:
            _ring.append(msg)


def ring_buffer() -> List[str]:
    """Snapshot of the most recent log lines (oldest first)."""
    with _lock:
        return list(_ring)


def configure(debug: bool | None = None, log_file: str | None = None) -> logging.Logger:
    """Configure the package logger once; later calls only adjust the level."""
    global _configured
    log = logging.getLogger(_LOG_NAME)
    if debug is None:
        debug = os.environ.get("SDWD_DEBUG", "0") not in ("", "0", "false")
    level = logging.DEBUG if debug else logging.INFO
    log.setLevel(level)
    if _configured:
        return log

    fmt = logging.Formatter(
        "[sdwd] %(asctime)s %(levelname)s %(name)s: %(message)s", "%H:%M:%S"
    )
    try:
        from rich.logging import RichHandler

        console: logging.Handler = RichHandler(
            show_path=False, markup=False, rich_tracebacks=False
        )
        console.setFormatter(logging.Formatter("%(message)s", "%H:%M:%S"))
    except Exception:  # pragma: no cover - rich is in the image
        console = logging.StreamHandler()
        console.setFormatter(fmt)
    log.addHandler(console)

    if log_file is None:
        log_file = os.environ.get("SDWD_LOG_FILE", "")
    if log_file:
        rotating = logging.handlers.RotatingFileHandler(
            log_file, maxBytes=10 * 1024 * 1024, backupCount=1
        )
        rotating.setFormatter(fmt)
        log.addHandler(rotating)

    ring = RingBufferHandler()
    ring.setFormatter(fmt)
    log.addHandler(ring)
    log.propagate = False
    _configured = True
    return log


def get_logger(name: str = "") -> logging.Logger:
    base = configure()
    return base.getChild(name) if name else base
