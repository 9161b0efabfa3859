"""TCP store tests: same semantics as MemStore, including watches, plus a
full controller+agent stack where the agent talks to the store over TCP
(the multi-rank benchmark topology)."""

import pytest

from instaslice_amd.api.types import new_pod
from instaslice_amd.store import Conflict, MemStore, NotFound
from instaslice_amd.store.netstore import NetStoreClient, StoreServer


@pytest.fixture
def served():
    server = StoreServer().start()
    client = NetStoreClient("127.0.0.1", server.port)
    yield server, client
    client.close()
    server.stop()


def _obj(name, kind="Thing", ns=""):
    return {"apiVersion": "v1", "kind": kind,
            "metadata": {"name": name, "namespace": ns}}


def test_verbs_roundtrip(served):
    _, c = served
    c.create(_obj("a"))
    assert c.get("Thing", "a")["metadata"]["name"] == "a"
    assert len(c.list("Thing")) == 1
    o = c.get("Thing", "a")
    o["x"] = 1
    c.update(o)
    assert c.get("Thing", "a")["x"] == 1
    with pytest.raises(Conflict):
        c.update(o)  # stale rv
    c.delete("Thing", "a")
    with pytest.raises(NotFound):
        c.get("Thing", "a")


def test_watch_over_tcp(served):
    server, c = served
    w = c.watch("Pod")
    server.store.create(new_pod("p1"))  # server-side write
    et, obj = w.next(timeout=2)
    assert et == "ADDED" and obj["metadata"]["name"] == "p1"
    c.create(new_pod("p2"))  # client-side write
    et, obj = w.next(timeout=2)
    assert obj["metadata"]["name"] == "p2"
    w.stop()


def test_update_with_retry_over_tcp(served):
    _, c = served
    c.create({**_obj("a"), "n": 0})

    def mut(o):
        o["n"] += 1
        return o

    for _ in range(5):
        c.update_with_retry("Thing", "a", "", mut)
    assert c.get("Thing", "a")["n"] == 5


def test_two_clients_conflict(served):
    server, c1 = served
    c2 = NetStoreClient("127.0.0.1", server.port)
    try:
        c1.create(_obj("a"))
        o1 = c1.get("Thing", "a")
        o2 = c2.get("Thing", "a")
        o1["x"] = 1
        c1.update(o1)
        o2["x"] = 2
        with pytest.raises(Conflict):
            c2.update(o2)
    finally:
        c2.close()


def test_full_stack_agent_over_tcp():
    """Controller on the server-side store; node agent connected via TCP —
    the exact bench.py multi-rank topology."""
    from instaslice_amd.agent.daemonset import NodeAgent
    from instaslice_amd.controller.reconciler import PodController
    from instaslice_amd.smi import FakeAmdSmi

    server = StoreServer().start()
    client = NetStoreClient("127.0.0.1", server.port)
    controller = PodController(server.store, teardown_grace_s=0.0)
    agent = NodeAgent(client, FakeAmdSmi(num_gpus=2, node_name="n0"), "n0")
    try:
        agent.start()
        controller.start()
        server.store.create(new_pod("p1", profile="cpx-1x36"))
        import time

        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            pod = server.store.get("Pod", "p1", "default")
            if not pod["spec"]["schedulingGates"]:
                break
            time.sleep(0.01)
        else:
            pytest.fail("pod never ungated through TCP store")
        cm = server.store.get("ConfigMap", "p1", "default")
        assert cm["data"]["ROCR_VISIBLE_DEVICES"]
    finally:
        controller.stop()
        agent.stop()
        client.close()
        server.stop()


def _server_factories():
    from instaslice_amd.store.native import NativeStoreServer, stored_available

    out = [("python", lambda: StoreServer().start())]
    if stored_available():
        out.append(("native", lambda: NativeStoreServer().start()))
    return out


@pytest.mark.parametrize("backend,factory", _server_factories(),
                         ids=lambda v: v if isinstance(v, str) else "")
def test_watch_resume_misses_nothing_no_relist(backend, factory):
    """Watch resume tokens: after a connection loss,
    a reconnecting watch replays ONLY the missed window from the server's
    bounded event history — zero lost events AND zero duplicate ADDED of
    pre-outage objects (the old full-relist resync re-delivered them)."""
    import socket as _socket
    import time

    server = factory()
    c1 = NetStoreClient("127.0.0.1", server.port, reconnect=True,
                        reconnect_backoff_s=0.05)
    c2 = NetStoreClient("127.0.0.1", server.port)
    try:
        w = c1.watch("Pod")
        c2.create(new_pod("pre-outage"))
        ev = w.next(timeout=5)
        assert ev and ev[0] == "ADDED" and ev[1]["metadata"]["name"] == "pre-outage"
        # sever c1's link; the server (and its history) stay up
        c1._sock.shutdown(_socket.SHUT_RDWR)
        c2.create(new_pod("during-outage"))
        c2.delete("Pod", "pre-outage", "default")  # MODIFIED (finalizer)
        # collect everything the watch delivers post-outage
        got = []
        deadline = time.monotonic() + 8.0
        while time.monotonic() < deadline:
            ev = w.next(timeout=0.25)
            if ev:
                got.append((ev[0], ev[1]["metadata"]["name"]))
            if ("ADDED", "during-outage") in got and any(
                    n == "pre-outage" and t == "MODIFIED" for t, n in got):
                break
        assert ("ADDED", "during-outage") in got, f"missed event: {got}"
        assert any(t == "MODIFIED" and n == "pre-outage" for t, n in got), (
            f"missed deletionTimestamp event: {got}")
        assert got.count(("ADDED", "during-outage")) == 1, f"duplicate: {got}"
        assert ("ADDED", "pre-outage") not in got, (
            f"full relist happened instead of a resume: {got}")
    finally:
        c1.close()
        c2.close()
        server.stop()


@pytest.mark.parametrize("backend,factory", _server_factories(),
                         ids=lambda v: v if isinstance(v, str) else "")
def test_watch_resume_falls_back_after_compaction(backend, factory):
    """A resume token older than the bounded history must fall back to the
    full ADDED relist (never silently lose the gap)."""
    import socket as _socket
    import time

    server = factory()
    c1 = NetStoreClient("127.0.0.1", server.port, reconnect=True,
                        reconnect_backoff_s=0.05)
    c2 = NetStoreClient("127.0.0.1", server.port)
    try:
        w = c1.watch("Pod")
        c2.create(new_pod("keeper"))
        assert w.next(timeout=5)
        # force the token far behind the compaction horizon
        w.last_rev = 0 if backend == "native" else None
        if backend == "python":
            w.last_rev = 0
        # age the history way past the window
        for i in range(4):
            c2.create(_obj(f"junk-{i}", kind="Junk"))
        server_hist = getattr(getattr(server, "store", None), "_history", None)
        if server_hist is not None:
            # simulate compaction: drop the front of the window
            while server_hist and server_hist[0][0] <= 2:
                server_hist.popleft()
        c1._sock.shutdown(_socket.SHUT_RDWR)
        got = []
        deadline = time.monotonic() + 8.0
        while time.monotonic() < deadline:
            ev = w.next(timeout=0.25)
            if ev:
                got.append((ev[0], ev[1]["metadata"]["name"]))
            if ("ADDED", "keeper") in got:
                break
        assert ("ADDED", "keeper") in got, (
            f"fallback relist never delivered the object: {got}")
    finally:
        c1.close()
        c2.close()
        server.stop()
