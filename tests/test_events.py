"""Event recorder internals: dedup counting, LRU retention, load shedding."""

import time

import pytest

from instaslice_amd.api import events as ev_mod
from instaslice_amd.api.events import _EventSink, emit
from instaslice_amd.store.memstore import MemStore


def _drain(sink, timeout=5.0):
    deadline = time.time() + timeout
    while not sink._q.empty() and time.time() < deadline:
        time.sleep(0.01)
    time.sleep(0.05)  # let the last item finish its write


def _fresh_sink():
    ev_mod._sink = _EventSink()
    return ev_mod._sink


@pytest.fixture(autouse=True)
def _reset_sink_after():
    # these tests tune the GLOBAL sink (shed depth, LRU cap); leave a clean
    # default for whatever runs next
    yield
    ev_mod._sink = None


def test_emit_dedup_bumps_count():
    sink = _fresh_sink()
    store = MemStore()
    ref = {"kind": "Pod", "namespace": "default", "name": "p"}
    for i in range(4):
        emit(store, ref, "Placed", f"msg{i}")
    _drain(sink)
    ev = store.get("Event", "p.Placed", "default")
    assert ev["count"] == 4
    assert ev["message"] == "msg3"
    assert ev["firstTimestamp"] <= ev["lastTimestamp"]


def test_lru_retention_deletes_oldest():
    sink = _fresh_sink()
    sink.MAX_LIVE_EVENTS = 5
    store = MemStore()
    for i in range(9):
        emit(store, {"kind": "Pod", "namespace": "default", "name": f"p{i}"},
             "Placed", "x")
    _drain(sink)
    live = [e["metadata"]["name"] for e in store.list("Event")]
    assert len(live) == 5, live
    assert "p8.Placed" in live and "p0.Placed" not in live


def test_normal_events_shed_under_backlog():
    sink = _fresh_sink()
    # deterministic backlog: depth -1 makes qsize() > depth always true, so
    # EVERY Normal sheds while Warnings always keep
    sink.NORMAL_SHED_DEPTH = -1
    store = MemStore()
    for i in range(10):
        emit(store, {"kind": "Pod", "namespace": "default", "name": f"n{i}"},
             "Placed", "x")
    emit(store, {"kind": "Pod", "namespace": "default", "name": "warn"},
         "Boom", "x", type_="Warning")
    _drain(sink)
    names = {e["metadata"]["name"] for e in store.list("Event")}
    assert names == {"warn.Boom"}, names
    assert sink.dropped == 10
