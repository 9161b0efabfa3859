"""Protocol-parity battery: the C++ store daemon (instaslice-stored) must be
indistinguishable from MemStore behind NetStoreClient. Every scenario runs
against BOTH backends via the parametrized fixture; the native daemon is the
control plane's scale-out path (store/csrc/stored_main.cpp), MemStore the
reference semantics (store/memstore.py)."""

import os
import threading
import time

import pytest

from instaslice_amd.api.types import new_pod
from instaslice_amd.store import AlreadyExists, Conflict, MemStore, NotFound
from instaslice_amd.store.native import NativeStoreServer, stored_available
from instaslice_amd.store.netstore import NetStoreClient, StoreServer

needs_stored = pytest.mark.skipif(
    not stored_available(),
    reason="instaslice-stored not built (python build_native.py)",
)


@pytest.fixture(params=["python", "native"])
def client(request):
    if request.param == "native":
        if not stored_available():
            pytest.skip("instaslice-stored not built")
        server = NativeStoreServer().start()
    else:
        server = StoreServer().start()
    c = NetStoreClient("127.0.0.1", server.port)
    yield c
    c.close()
    server.stop()


def _obj(name, kind="Thing", ns="", **extra):
    return {"apiVersion": "v1", "kind": kind,
            "metadata": {"name": name, "namespace": ns}, **extra}


def test_crud_roundtrip(client):
    client.create(_obj("a", x=1))
    got = client.get("Thing", "a")
    assert got["x"] == 1 and got["metadata"]["resourceVersion"] == "1"
    with pytest.raises(AlreadyExists):
        client.create(_obj("a"))
    got["x"] = 2
    updated = client.update(got)
    assert updated["x"] == 2
    assert updated["metadata"]["resourceVersion"] != "1"
    with pytest.raises(Conflict):
        client.update(got)  # stale rv
    assert [o["metadata"]["name"] for o in client.list("Thing")] == ["a"]
    client.delete("Thing", "a")
    with pytest.raises(NotFound):
        client.get("Thing", "a")
    with pytest.raises(NotFound):
        client.delete("Thing", "a")


def test_list_sorted_and_namespaced(client):
    client.create(_obj("b", ns="n2"))
    client.create(_obj("a", ns="n1"))
    client.create(_obj("c", kind="Other"))
    names = [o["metadata"]["name"] for o in client.list("Thing")]
    assert names == ["a", "b"]
    assert [o["metadata"]["name"] for o in client.list("Thing", "n2")] == ["b"]


def test_finalizer_two_phase_delete(client):
    obj = _obj("f")
    obj["metadata"]["finalizers"] = ["org.instaslice/accelarator"]
    client.create(obj)
    client.delete("Thing", "f")
    got = client.get("Thing", "f")  # still there, deletionTimestamp set
    assert got["metadata"]["deletionTimestamp"]
    got["metadata"]["finalizers"] = []
    client.update(got)  # finalizer removed -> object actually goes away
    with pytest.raises(NotFound):
        client.get("Thing", "f")


def test_patch_ops(client):
    client.create(_obj("p", spec={"allocations": {}, "n": 1}))
    res = client.patch("Thing", "p", "", [
        {"op": "set", "path": ["spec", "allocations", "u1", "allocationStatus"],
         "value": "creating"},
        {"op": "merge", "path": ["spec", "prepared"], "value": {"x": {"k": 1}}},
        {"op": "add_to_set", "path": ["spec", "used"], "value": 3},
        {"op": "add_to_set", "path": ["spec", "used"], "value": 1},
        {"op": "add_to_set", "path": ["spec", "used"], "value": 3},
    ])
    assert res["spec"]["allocations"]["u1"]["allocationStatus"] == "creating"
    assert res["spec"]["prepared"] == {"x": {"k": 1}}
    assert res["spec"]["used"] == [1, 3]
    res = client.patch("Thing", "p", "", [
        {"op": "test", "path": ["spec", "allocations", "u1", "allocationStatus"],
         "value": "creating"},
        {"op": "set", "path": ["spec", "allocations", "u1", "allocationStatus"],
         "value": "created"},
        {"op": "remove_from_set", "path": ["spec", "used"], "value": 3},
        {"op": "delete", "path": ["spec", "n"]},
    ])
    assert res["spec"]["allocations"]["u1"]["allocationStatus"] == "created"
    assert res["spec"]["used"] == [1]
    assert "n" not in res["spec"]
    # failed test leaves the object untouched
    before = client.get("Thing", "p")
    with pytest.raises(Conflict):
        client.patch("Thing", "p", "", [
            {"op": "test", "path": ["spec", "allocations", "u1", "allocationStatus"],
             "value": "creating"},
            {"op": "set", "path": ["spec", "boom"], "value": True},
        ])
    after = client.get("Thing", "p")
    assert after == before
    with pytest.raises(Conflict):
        client.patch("Thing", "p", "", [
            {"op": "test", "path": ["spec", "allocations", "u1"], "absent": True},
        ])
    with pytest.raises(NotFound):
        client.patch("Thing", "nope", "", [{"op": "set", "path": ["x"], "value": 1}])


def test_batch_and_quiet(client):
    res = client.batch([
        {"verb": "create", "obj": _obj("b1")},
        {"verb": "create", "obj": _obj("b1")},  # dup -> AlreadyExists
        {"verb": "patch", "kind": "Thing", "name": "b1", "namespace": "",
         "ops": [{"op": "set", "path": ["spec", "v"], "value": 7}]},
        {"verb": "delete", "kind": "Thing", "name": "missing", "namespace": ""},
    ])
    assert [r["ok"] for r in res] == [True, False, True, False]
    assert res[1]["error"]["type"] == "AlreadyExists"
    assert res[2]["result"]["spec"]["v"] == 7
    assert res[3]["error"]["type"] == "NotFound"
    res = client.batch([
        {"verb": "patch", "kind": "Thing", "name": "b1", "namespace": "",
         "ops": [{"op": "set", "path": ["spec", "v"], "value": 8}]},
    ], quiet=True)
    assert res[0]["ok"] and res[0]["result"] is None
    assert client.get("Thing", "b1")["spec"]["v"] == 8
    assert client.patch("Thing", "b1", "",
                        [{"op": "set", "path": ["spec", "v"], "value": 9}],
                        quiet=True) is None


def test_watch_kinds_replay_and_filters(client):
    client.create(new_pod("w1", labels={"grp": "a"}))
    client.create(new_pod("w2", labels={"grp": "b"}))
    w = client.watch("Pod")  # replay
    seen = {w.next(timeout=2)[1]["metadata"]["name"] for _ in range(2)}
    assert seen == {"w1", "w2"}
    wf = client.watch(replay=True, filters=[
        {"kind": "Pod", "labels": {"grp": "a"}},
        {"kind": "Thing", "name": "t9"},
    ])
    ev = wf.next(timeout=2)
    assert ev[1]["metadata"]["name"] == "w1"
    client.create(_obj("t8"))   # not matched
    client.create(_obj("t9"))   # matched by name filter
    ev = wf.next(timeout=2)
    assert ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "t9"
    client.create(new_pod("w3", labels={"grp": "a"}))
    ev = wf.next(timeout=2)
    assert ev[1]["metadata"]["name"] == "w3"
    w.stop()
    wf.stop()


def test_watch_modify_delete_events(client):
    w = client.watch("Thing", replay=False)
    client.create(_obj("m"))
    obj = client.get("Thing", "m")
    obj["x"] = 1
    client.update(obj)
    client.delete("Thing", "m")
    events = [w.next(timeout=2) for _ in range(3)]
    assert [e[0] for e in events] == ["ADDED", "MODIFIED", "DELETED"]
    w.stop()


def test_update_with_retry_conflict_absorption(client):
    client.create(_obj("r", n=0))

    def bump():
        for _ in range(20):
            client.update_with_retry(
                "Thing", "r", "",
                lambda o: {**o, "n": o["n"] + 1},
            )

    threads = [threading.Thread(target=bump) for _ in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert client.get("Thing", "r")["n"] == 80


@needs_stored
def test_native_concurrent_clients_and_patch_atomicity():
    """Hammer one object with concurrent add_to_set patches from several
    clients: every element must land exactly once (store-side atomicity)."""
    server = NativeStoreServer().start()
    clients = [NetStoreClient("127.0.0.1", server.port) for _ in range(4)]
    try:
        clients[0].create(_obj("hot", spec={"s": []}))

        def worker(ci, base):
            for k in range(25):
                clients[ci].patch("Thing", "hot", "", [
                    {"op": "add_to_set", "path": ["spec", "s"],
                     "value": base * 100 + k},
                ], quiet=True)

        threads = [threading.Thread(target=worker, args=(i, i)) for i in range(4)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        got = clients[0].get("Thing", "hot")["spec"]["s"]
        assert len(got) == 100 and got == sorted(got)
    finally:
        for c in clients:
            c.close()
        server.stop()


@needs_stored
def test_native_full_pod_lifecycle_stack():
    """controller + agent + fake SMI against the NATIVE store daemon: the
    complete creating -> created -> ungated -> deleted machine (the bench
    topology with the C++ data plane)."""
    from instaslice_amd.agent.daemonset import NodeAgent
    from instaslice_amd.controller.reconciler import PodController
    from instaslice_amd.smi.fake import FakeAmdSmi

    server = NativeStoreServer().start()
    cstore = NetStoreClient("127.0.0.1", server.port)
    astore = NetStoreClient("127.0.0.1", server.port)
    controller = PodController(cstore, teardown_grace_s=0.0, workers=2)
    controller.requeue_no_fit_s = 0.05
    agent = NodeAgent(astore, FakeAmdSmi(num_gpus=2, node_name="node-0"),
                      "node-0")
    agent.start()
    controller.start()
    bench = NetStoreClient("127.0.0.1", server.port)
    try:
        for i in range(10):
            bench.create(new_pod(f"n{i}", profile="cpx-1x36"))
        deadline = time.monotonic() + 20
        while time.monotonic() < deadline:
            pods = bench.list("Pod")
            if len(pods) == 10 and all(
                not p["spec"].get("schedulingGates") for p in pods
            ):
                break
            time.sleep(0.05)
        pods = bench.list("Pod")
        assert all(not p["spec"].get("schedulingGates") for p in pods)
        cr = bench.get("Instaslice", "node-0", "instaslice-system")
        assert len(cr["spec"]["allocations"]) == 10
        assert len(cr["spec"]["prepared"]) == 10
        assert bench.get("ConfigMap", "n0", "default")["data"][
            "ROCR_VISIBLE_DEVICES"]
        # drain one pod completely
        bench.delete("Pod", "n0", "default")
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            cr = bench.get("Instaslice", "node-0", "instaslice-system")
            if not any(a["podName"] == "n0"
                       for a in cr["spec"]["allocations"].values()):
                break
            time.sleep(0.05)
        assert not any(a["podName"] == "n0"
                       for a in cr["spec"]["allocations"].values())
        with pytest.raises(NotFound):
            bench.get("ConfigMap", "n0", "default")
    finally:
        controller.stop()
        agent.stop()
        for c in (cstore, astore, bench):
            c.close()
        server.stop()


@needs_stored
def test_client_reconnects_after_store_restart():
    """Kill the store daemon mid-session and restart it on the same port:
    a reconnect=True client must resume — calls work again and its watches
    are re-subscribed with replay (informer resync)."""
    server = NativeStoreServer().start()
    port = server.port
    c = NetStoreClient("127.0.0.1", port, reconnect=True)
    try:
        c.create(_obj("survivor"))
        w = c.watch("Thing", replay=False)
        server.stop()  # daemon killed; state is gone (no persistence)
        time.sleep(0.3)
        server = NativeStoreServer(port=port).start()
        # calls recover (engine error-backoff analog: retry until live)
        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            try:
                c.create(_obj("reborn"))
                break
            except (ConnectionError, TimeoutError, RuntimeError):
                time.sleep(0.2)
        else:
            pytest.fail("client never recovered")
        # the re-subscribed watch sees post-restart events
        ev = w.next(timeout=5)
        assert ev is not None and ev[1]["metadata"]["name"] == "reborn"
        w.stop()
    finally:
        c.close()
        server.stop()


@needs_stored
def test_native_persistence_checkpoint_resume(tmp_path):
    """SIGTERM flushes a msgpack snapshot; a fresh daemon resumes with the
    same objects AND resourceVersion monotonicity (the Python MemStore's
    persist_path contract, store/memstore.py)."""
    snap = str(tmp_path / "state.msgpack")
    server = NativeStoreServer(persist_path=snap).start()
    c = NetStoreClient("127.0.0.1", server.port)
    c.create(_obj("d1", spec={"v": 1}))
    c.create(_obj("d2"))
    c.patch("Thing", "d1", "", [{"op": "set", "path": ["spec", "v"], "value": 2}])
    rv_before = c.get("Thing", "d1")["metadata"]["resourceVersion"]
    c.close()
    server.stop()  # SIGTERM -> flush

    server = NativeStoreServer(persist_path=snap).start()
    c = NetStoreClient("127.0.0.1", server.port)
    try:
        got = c.get("Thing", "d1")
        assert got["spec"]["v"] == 2
        assert got["metadata"]["resourceVersion"] == rv_before
        assert len(c.list("Thing")) == 2
        # new writes continue the rv sequence past the resumed point
        c.patch("Thing", "d2", "", [{"op": "set", "path": ["x"], "value": 1}])
        assert int(c.get("Thing", "d2")["metadata"]["resourceVersion"]) > int(rv_before)
    finally:
        c.close()
        server.stop()


@needs_stored
def test_chaos_store_restart_mid_churn(tmp_path):
    """Kill (SIGTERM -> snapshot flush) and restart the persistent store
    daemon WHILE pods are churning: reconnect-enabled controller and agent
    resync their watches, state resumes from the snapshot, and the churn
    completes with consistent allocations. This is the etcd-restart drill
    the reference delegates to Kubernetes."""
    from instaslice_amd.agent.daemonset import NodeAgent
    from instaslice_amd.controller.reconciler import PodController
    from instaslice_amd.smi.fake import FakeAmdSmi

    snap = str(tmp_path / "chaos.msgpack")
    server = NativeStoreServer(persist_path=snap).start()
    port = server.port
    cstore = NetStoreClient("127.0.0.1", port, reconnect=True)
    astore = NetStoreClient("127.0.0.1", port, reconnect=True)
    controller = PodController(cstore, teardown_grace_s=0.0, workers=2)
    controller.requeue_no_fit_s = 0.05
    agent = NodeAgent(astore, FakeAmdSmi(num_gpus=2, node_name="node-0"),
                      "node-0", heartbeat_every_s=0.5)
    agent.start()
    controller.start()
    bench = NetStoreClient("127.0.0.1", port, reconnect=True)

    def wait_sched(name, timeout=30.0):
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            try:
                p = bench.get("Pod", name, "default")
                if not p["spec"].get("schedulingGates"):
                    return
            except (NotFound, ConnectionError, TimeoutError, RuntimeError):
                pass
            time.sleep(0.05)
        raise TimeoutError(name)

    try:
        for i in range(4):
            bench.create(new_pod(f"pre{i}", profile="cpx-1x36"))
            wait_sched(f"pre{i}")
        # hard restart: flush + new daemon on the same port/path
        server.stop()
        time.sleep(0.5)
        server = NativeStoreServer(port=port, persist_path=snap).start()
        # pre-restart state resumed?
        deadline = time.monotonic() + 20
        while time.monotonic() < deadline:
            try:
                cr = bench.get("Instaslice", "node-0", "instaslice-system")
                break
            except (NotFound, ConnectionError, TimeoutError, RuntimeError):
                time.sleep(0.2)
        else:
            pytest.fail("CR not resumed after restart")
        assert len(cr["spec"]["allocations"]) == 4
        # churn continues against the resumed store
        for i in range(6):
            created = False
            deadline = time.monotonic() + 20
            while time.monotonic() < deadline:
                try:
                    bench.create(new_pod(f"post{i}", profile="cpx-1x36"))
                    created = True
                    break
                except (ConnectionError, TimeoutError, RuntimeError):
                    time.sleep(0.2)
                except AlreadyExists:
                    created = True
                    break
            assert created
            wait_sched(f"post{i}")
        cr = bench.get("Instaslice", "node-0", "instaslice-system")
        allocs = cr["spec"]["allocations"]
        assert len(allocs) == 10
        slots = {(a["gpuUUID"], a["ordinal"]) for a in allocs.values()}
        assert len(slots) == 10, "double-booked slot after restart"
    finally:
        controller.stop()
        agent.stop()
        for c in (cstore, astore, bench):
            c.close()
        server.stop()


@needs_stored
def test_daemon_survives_protocol_garbage():
    """Malformed frames drop only the offending connection; the daemon keeps
    serving other clients (fuzz-resistance of the wire protocol)."""
    import socket as socklib
    import struct

    server = NativeStoreServer().start()
    good = NetStoreClient("127.0.0.1", server.port)
    try:
        good.create(_obj("keep"))
        for garbage in (
            b"\x00\x00\x00\x05\xc1\xc1\xc1\xc1\xc1",  # invalid msgpack tag
            b"\xff\xff\xff\xff",                       # absurd length prefix
            struct.pack(">I", 3) + b"\x93\x01",        # truncated payload
            b"GET / HTTP/1.1\r\n\r\n",                 # wrong protocol
        ):
            s = socklib.create_connection(("127.0.0.1", server.port), timeout=5)
            s.sendall(garbage)
            s.close()
        # oversized frame header (>64 MiB declared)
        s = socklib.create_connection(("127.0.0.1", server.port), timeout=5)
        s.sendall(struct.pack(">I", 200 << 20))
        s.close()
        assert good.get("Thing", "keep")["metadata"]["name"] == "keep"
        good.create(_obj("still-works"))
        assert len(good.list("Thing")) == 2
    finally:
        good.close()
        server.stop()


from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

_keys = st.sampled_from(["a", "b", "c", "list", "allocations", "x"])
_vals = st.one_of(st.integers(-5, 5), st.text(max_size=4), st.booleans(),
                  st.none(), st.dictionaries(_keys, st.integers(0, 3),
                                             max_size=2))
_paths = st.lists(_keys, min_size=1, max_size=3)
_ops = st.lists(st.one_of(
    st.fixed_dictionaries({"op": st.just("set"), "path": _paths,
                           "value": _vals}),
    st.fixed_dictionaries({"op": st.just("merge"), "path": _paths,
                           "value": st.dictionaries(_keys, _vals, max_size=2)}),
    st.fixed_dictionaries({"op": st.just("delete"), "path": _paths}),
    st.fixed_dictionaries({"op": st.just("add_to_set"), "path": _paths,
                           "value": st.integers(0, 5)}),
    st.fixed_dictionaries({"op": st.just("remove_from_set"), "path": _paths,
                           "value": st.integers(0, 5)}),
    st.fixed_dictionaries({"op": st.just("delete_where"), "path": _paths,
                           "field": _keys, "value": _vals}),
    st.fixed_dictionaries({"op": st.just("test"), "path": _paths,
                           "value": _vals}),
    st.fixed_dictionaries({"op": st.just("test"), "path": _paths,
                           "absent": st.just(True)}),
), min_size=1, max_size=6)


_fuzz_env = {}
_fuzz_n = [0]


def _fuzz_backends():
    """Shared servers for the differential fuzz (per-example server spawn
    exhausted fds/threads at high example counts)."""
    if not _fuzz_env:
        py_server = StoreServer().start()
        nat_server = NativeStoreServer().start()
        _fuzz_env["servers"] = (py_server, nat_server)
        _fuzz_env["py"] = NetStoreClient("127.0.0.1", py_server.port)
        _fuzz_env["nat"] = NetStoreClient("127.0.0.1", nat_server.port)
    return _fuzz_env["py"], _fuzz_env["nat"]


@needs_stored
@settings(max_examples=int(os.environ.get(
              "INSTASLICE_FUZZ_EXAMPLES", "150")),
          deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(batches=st.lists(_ops, min_size=1, max_size=5))
def test_patch_grammar_differential_fuzz(batches):
    """Differential fuzzing: the SAME random patch sequences applied to the
    Python MemStore and the C++ daemon must produce byte-identical objects
    and identical error types — the strongest form of the parity contract.
    (Found for real: list-traversing paths used to error on Python and
    silently clobber on C++.)"""
    py, nat = _fuzz_backends()
    _fuzz_n[0] += 1
    name = f"f{_fuzz_n[0]}"
    seed = {"apiVersion": "v1", "kind": "Thing",
            "metadata": {"name": name, "namespace": ""},
            "spec": {"allocations": {}, "list": [1, 2]}}
    py.create(seed)
    nat.create(seed)
    for ops in batches:
        ery = ern = None
        try:
            py.patch("Thing", name, "", ops)
        except Exception as e:  # noqa: BLE001
            ery = type(e).__name__
        try:
            nat.patch("Thing", name, "", ops)
        except Exception as e:  # noqa: BLE001
            ern = type(e).__name__
        assert ery == ern, f"error divergence {ery} vs {ern} on {ops}"
    a, b = py.get("Thing", name), nat.get("Thing", name)
    b = dict(b)
    assert a == b, f"state divergence:\n{a}\n{b}"


_verb_ops = st.lists(st.one_of(
    st.fixed_dictionaries({"verb": st.just("create"),
                           "fin": st.booleans(),
                           "spec": st.dictionaries(_keys, _vals, max_size=2)}),
    st.fixed_dictionaries({"verb": st.just("update"),
                           "stale": st.booleans(),
                           "spec": st.dictionaries(_keys, _vals, max_size=2)}),
    st.fixed_dictionaries({"verb": st.just("delete")}),
    st.fixed_dictionaries({"verb": st.just("strip_finalizers")}),
    st.fixed_dictionaries({"verb": st.just("patch"), "ops": _ops}),
), min_size=1, max_size=10)


@needs_stored
@settings(max_examples=int(os.environ.get(
              "INSTASLICE_FUZZ_EXAMPLES", "150")),
          deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(seq=_verb_ops)
def test_verb_lifecycle_differential_fuzz(seq):
    """Differential fuzzing over the FULL verb set, including finalizer
    two-phase deletion and stale-resourceVersion conflicts: both backends
    must expose identical state and identical error types after every
    step. (rv sequences are deterministic, so stale-rv updates conflict —
    or not — identically.)"""
    py, nat = _fuzz_backends()
    _fuzz_n[0] += 1
    name = f"v{_fuzz_n[0]}"

    def run(client, op):
        if op["verb"] == "create":
            obj = {"apiVersion": "v1", "kind": "VThing",
                   "metadata": {"name": name, "namespace": ""},
                   "spec": dict(op["spec"])}
            if op["fin"]:
                obj["metadata"]["finalizers"] = ["org.instaslice/accelarator"]
            return client.create(obj)
        if op["verb"] == "update":
            cur = client.get("VThing", name, "")
            cur["spec"] = dict(op["spec"])
            if op["stale"]:
                cur["metadata"]["resourceVersion"] = "0"
            return client.update(cur)
        if op["verb"] == "delete":
            return client.delete("VThing", name, "")
        if op["verb"] == "strip_finalizers":
            cur = client.get("VThing", name, "")
            cur["metadata"]["finalizers"] = []
            return client.update(cur)
        if op["verb"] == "patch":
            return client.patch("VThing", name, "", op["ops"])
        raise AssertionError(op)

    for op in seq:
        ery = ern = None
        try:
            run(py, op)
        except Exception as e:  # noqa: BLE001
            ery = type(e).__name__
        try:
            run(nat, op)
        except Exception as e:  # noqa: BLE001
            ern = type(e).__name__
        assert ery == ern, f"error divergence {ery} vs {ern} on {op}"
        sa = sb = None
        try:
            sa = py.get("VThing", name, "")
        except NotFound:
            pass
        try:
            sb = nat.get("VThing", name, "")
        except NotFound:
            pass
        # deletionTimestamp is wall-clock and may differ by nanoseconds;
        # compare it by presence, everything else exactly
        def norm(o):
            if o is None:
                return None
            o = dict(o)
            md = dict(o["metadata"])
            if md.get("deletionTimestamp"):
                md["deletionTimestamp"] = "SET"
            o["metadata"] = md
            return o
        assert norm(sa) == norm(sb), f"state divergence after {op}:\n{sa}\n{sb}"
