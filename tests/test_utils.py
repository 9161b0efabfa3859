"""Utility coverage: stack sampler, JSON log formatter, uid/clock."""

import json
import logging
import threading
import time

from instaslice_amd.utils import _JsonFormatter, new_uid, start_stack_sampler


def test_stack_sampler_captures_all_threads(tmp_path):
    out = str(tmp_path / "s.samples")
    stop_evt = threading.Event()

    def busy():
        while not stop_evt.is_set():
            sum(range(500))

    t = threading.Thread(target=busy, name="busy-bee", daemon=True)
    t.start()
    stop = start_stack_sampler(out, interval_s=0.002)
    time.sleep(0.25)
    stop()
    stop_evt.set()
    t.join(timeout=2)
    text = open(out).read()
    assert "busy-bee" in text
    # line format: count<TAB>thread<TAB>func@file:line
    first = text.splitlines()[0].split("\t")
    assert len(first) == 3 and int(first[0]) > 0 and "@" in first[2]


def test_json_log_formatter():
    rec = logging.LogRecord("x", logging.WARNING, "f.py", 1,
                            "hello %s", ("world",), None)
    out = json.loads(_JsonFormatter().format(rec))
    assert out["msg"] == "hello world"
    assert out["level"] == "WARNING" and out["logger"] == "x"


def test_uids_unique():
    assert len({new_uid() for _ in range(100)}) == 100
