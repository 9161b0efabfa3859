"""Small shared utilities: logging, ids, clocks."""

from __future__ import annotations

import logging
import os
import threading
import time
import uuid

_LOG_CONFIGURED = False
_LOCK = threading.Lock()


class _JsonFormatter(logging.Formatter):
    """One JSON object per line (the reference's zap production encoder
    analog) — for log aggregators. Enabled with INSTASLICE_LOG_JSON=1."""

    def format(self, record: logging.LogRecord) -> str:
        import json

        out = {
            "ts": round(record.created, 3),
            "level": record.levelname,
            "logger": record.name,
            "msg": record.getMessage(),
        }
        if record.exc_info:
            out["exc"] = self.formatException(record.exc_info)
        return json.dumps(out)


def get_logger(name: str) -> logging.Logger:
    """Structured logger (the reference uses zap/logr; we use stdlib).

    Level from INSTASLICE_LOG_LEVEL (default INFO); INSTASLICE_LOG_JSON=1
    switches to one-JSON-object-per-line output.
    """
    global _LOG_CONFIGURED
    with _LOCK:
        if not _LOG_CONFIGURED:
            level = os.environ.get("INSTASLICE_LOG_LEVEL", "INFO").upper()
            if os.environ.get("INSTASLICE_LOG_JSON", "") not in ("", "0"):
                handler = logging.StreamHandler()
                handler.setFormatter(_JsonFormatter())
                logging.basicConfig(
                    level=getattr(logging, level, logging.INFO),
                    handlers=[handler],
                )
            else:
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


def start_stack_sampler(path: str, interval_s: float = 0.005):
    """Wall-clock sampling profiler over ALL threads (sys._current_frames):
    cProfile only sees the calling thread, but the engines do their work on
    pump/worker threads. Returns a stop() that writes aggregated
    `count<TAB>thread<TAB>func@file:line` lines, innermost frame only —
    enough to see where time (including blocking) goes. Env-gated via
    INSTASLICE_PROFILE_DIR in the control plane and bench."""
    import collections
    import sys
    import threading
    import time as _time

    counts = collections.Counter()
    stop_flag = threading.Event()

    def sample():
        me = threading.get_ident()
        names = {}
        while not stop_flag.is_set():
            for t in threading.enumerate():
                names[t.ident] = t.name
            for tid, frame in sys._current_frames().items():
                if tid == me:
                    continue
                co = frame.f_code
                key = (names.get(tid, str(tid)),
                       f"{co.co_name}@{co.co_filename.rsplit('/', 1)[-1]}"
                       f":{frame.f_lineno}")
                counts[key] += 1
            _time.sleep(interval_s)

    t = threading.Thread(target=sample, daemon=True, name="stack-sampler")
    t.start()

    def stop():
        stop_flag.set()
        t.join(timeout=2.0)
        with open(path, "w") as f:
            for (tname, loc), n in counts.most_common():
                f.write(f"{n}\t{tname}\t{loc}\n")

    return stop
