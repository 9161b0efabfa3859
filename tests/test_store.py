"""Store (API-server double) semantics tests — the envtest analog tier."""

import threading

import pytest

from instaslice_amd.api.types import new_pod
from instaslice_amd.store import AlreadyExists, Conflict, MemStore, NotFound


def _obj(name, kind="Thing", ns="", **extra):
    return {"apiVersion": "v1", "kind": kind,
            "metadata": {"name": name, "namespace": ns}, **extra}


def test_create_get_list_delete():
    s = MemStore()
    s.create(_obj("a"))
    s.create(_obj("b"))
    assert s.get("Thing", "a")["metadata"]["name"] == "a"
    assert [o["metadata"]["name"] for o in s.list("Thing")] == ["a", "b"]
    with pytest.raises(AlreadyExists):
        s.create(_obj("a"))
    s.delete("Thing", "a")
    with pytest.raises(NotFound):
        s.get("Thing", "a")


def test_update_conflict_on_stale_rv():
    s = MemStore()
    s.create(_obj("a"))
    o1 = s.get("Thing", "a")
    o2 = s.get("Thing", "a")
    o1["x"] = 1
    s.update(o1)
    o2["x"] = 2
    with pytest.raises(Conflict):
        s.update(o2)


def test_returned_objects_are_copies():
    s = MemStore()
    s.create(_obj("a", data={"k": "v"}))
    o = s.get("Thing", "a")
    o["data"]["k"] = "mutated"
    assert s.get("Thing", "a")["data"]["k"] == "v"


def test_finalizer_two_phase_delete():
    """k8s semantics the teardown machinery depends on
    (instaslice_controller.go:99-142 analog)."""
    s = MemStore()
    pod = new_pod("p1", profile="cpx-1x36")
    s.create(pod)
    s.delete("Pod", "p1", "default")
    # finalizer present: pod still exists, with deletionTimestamp
    p = s.get("Pod", "p1", "default")
    assert p["metadata"]["deletionTimestamp"]
    # strip the finalizer -> object actually goes away
    p["metadata"]["finalizers"] = []
    s.update(p)
    with pytest.raises(NotFound):
        s.get("Pod", "p1", "default")


def test_watch_replay_and_live_events():
    s = MemStore()
    s.create(_obj("a"))
    w = s.watch("Thing")
    et, obj = w.next(timeout=1)
    assert et == "ADDED" and obj["metadata"]["name"] == "a"
    s.create(_obj("b"))
    et, obj = w.next(timeout=1)
    assert et == "ADDED" and obj["metadata"]["name"] == "b"
    o = s.get("Thing", "b")
    o["x"] = 1
    s.update(o)
    et, obj = w.next(timeout=1)
    assert et == "MODIFIED" and obj["x"] == 1
    s.delete("Thing", "b")
    et, obj = w.next(timeout=1)
    assert et == "DELETED"
    w.stop()


def test_watch_kind_filter():
    s = MemStore()
    w = s.watch("Pod")
    s.create(_obj("a"))  # kind Thing: filtered out
    s.create(new_pod("p1"))
    et, obj = w.next(timeout=1)
    assert obj["kind"] == "Pod"


def test_update_with_retry_absorbs_conflicts():
    s = MemStore()
    s.create(_obj("a", n=0))
    n_threads, n_incr = 8, 25

    def bump():
        for _ in range(n_incr):
            def mut(o):
                o["n"] += 1
                return o
            s.update_with_retry("Thing", "a", "", mut, attempts=1000)

    ts = [threading.Thread(target=bump) for _ in range(n_threads)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert s.get("Thing", "a")["n"] == n_threads * n_incr


def test_persistence_roundtrip(tmp_path):
    """Checkpoint/resume: a store restart reloads all objects + rv (the
    durability the reference gets from etcd)."""
    path = str(tmp_path / "state.json")
    s1 = MemStore(persist_path=path)
    s1.create(_obj("a", n=1))
    s1.create(new_pod("p1", profile="cpx-1x36"))
    o = s1.get("Thing", "a")
    o["n"] = 2
    s1.update(o)
    s1.close()

    s2 = MemStore(persist_path=path)
    assert s2.get("Thing", "a")["n"] == 2
    pod = s2.get("Pod", "p1", "default")
    assert pod["metadata"]["finalizers"]
    # resourceVersion survives: no rv reuse after restart
    o2 = s2.get("Thing", "a")
    o2["n"] = 3
    s2.update(o2)
    assert int(s2.get("Thing", "a")["metadata"]["resourceVersion"]) > int(
        o["metadata"]["resourceVersion"])
    s2.close()


def test_persistence_debounced_write_behind(tmp_path):
    import json as _json
    import time as _time

    path = str(tmp_path / "state.json")
    s = MemStore(persist_path=path, persist_debounce_s=0.05)
    for i in range(20):
        s.create(_obj(f"x{i}"))
    deadline = _time.monotonic() + 5
    while _time.monotonic() < deadline:
        try:
            with open(path) as f:
                snap = _json.load(f)
            if len(snap["objects"]) == 20:
                break
        except (OSError, ValueError):
            pass
        _time.sleep(0.02)
    else:
        raise AssertionError("write-behind never flushed all objects")
    s.close()
