"""Small shared utilities: logging, ids, clocks."""

from __future__ import annotations

import logging
import os
import threading
import time
import uuid

_LOG_CONFIGURED = False
_LOCK = threading.Lock()


def get_logger(name: str) -> logging.Logger:
    """Structured-ish logger (the reference uses zap/logr; we use stdlib).

    Level comes from INSTASLICE_LOG_LEVEL (default INFO).
    """
    global _LOG_CONFIGURED
    with _LOCK:
        if not _LOG_CONFIGURED:
            level = os.environ.get("INSTASLICE_LOG_LEVEL", "INFO").upper()
            logging.basicConfig(
                level=getattr(logging, level, logging.INFO),
                format="%(asctime)s %(levelname)-5s %(name)s %(message)s",
            )
            _LOG_CONFIGURED = True
    return logging.getLogger(name)


def new_uid() -> str:
    return str(uuid.uuid4())


def now() -> float:
    """Wall-clock seconds (float)."""
    return time.time()


def monotonic_ms() -> float:
    return time.monotonic() * 1000.0
