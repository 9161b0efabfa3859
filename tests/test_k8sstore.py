"""K8sStore behavioral tier: the Kubernetes adapter runs
the SAME lifecycle battery as the in-house stores, against a vendored fake
of the `kubernetes` client backed by an in-memory API-server double
(tests/vendor/kubernetes) that models the semantics a REAL cluster imposes:

  - string resourceVersions + 409 Conflict on stale replace
  - RFC3339 creation/deletionTimestamps (exercises timestamp_epoch)
  - finalizer two-phase deletion
  - status subresources (Instaslice heartbeat, Node capacity)
  - structural CRD pruning of unknown spec fields
  - watch streams with resourceVersion resume and 410 Gone

The envtest analog (reference: internal/controller/suite_test.go:52-84 —
which boots a real API server; none exists in this environment).
"""

import sys
import threading
import time
from pathlib import Path

import pytest

VENDOR = str(Path(__file__).resolve().parent / "vendor")
if VENDOR not in sys.path:
    sys.path.insert(0, VENDOR)

import kubernetes  # noqa: E402  (the vendored fake)

from instaslice_amd.api.types import AllocationStatus, new_instaslice, new_pod  # noqa: E402
from instaslice_amd.store import memstore  # noqa: E402
from instaslice_amd.store.k8sstore import K8sStore  # noqa: E402
from instaslice_amd.store.memstore import AlreadyExists, Conflict, NotFound  # noqa: E402

INSTASLICE_NS = "instaslice-system"


@pytest.fixture
def store():
    kubernetes.reset_server()
    return K8sStore()


def test_interface_matches_memstore():
    """Every verb the reconcilers use must exist with compatible signatures."""
    import inspect

    for verb in ("create", "get", "list", "update", "delete", "watch",
                 "update_with_retry", "patch", "batch"):
        assert hasattr(K8sStore, verb), f"K8sStore missing {verb}"
        mem_sig = inspect.signature(getattr(memstore.MemStore, verb))
        k8s_sig = inspect.signature(getattr(K8sStore, verb))
        assert list(mem_sig.parameters)[:3] == list(k8s_sig.parameters)[:3], verb


# -- verb semantics ----------------------------------------------------------


def test_crud_roundtrip_all_kinds(store):
    cr = new_instaslice("node-0")
    store.create(cr)
    got = store.get("Instaslice", "node-0", INSTASLICE_NS)
    assert got["spec"]["allocations"] == {}
    assert isinstance(got["metadata"]["resourceVersion"], str)

    pod = new_pod("p1", profile="cpx-1x36")
    store.create(pod)
    assert store.get("Pod", "p1", "default")["metadata"]["uid"]

    store.create({"apiVersion": "v1", "kind": "ConfigMap",
                  "metadata": {"name": "cm1", "namespace": "default"},
                  "data": {"K": "V"}})
    assert store.get("ConfigMap", "cm1", "default")["data"]["K"] == "V"

    store.create({"apiVersion": "v1", "kind": "Node",
                  "metadata": {"name": "node-0", "namespace": ""},
                  "status": {"capacity": {}}})
    assert store.get("Node", "node-0", "")["metadata"]["name"] == "node-0"

    assert [o["metadata"]["name"] for o in store.list("Instaslice")] == ["node-0"]
    assert len(store.list("Pod")) == 1


def test_error_mapping_404_409(store):
    with pytest.raises(NotFound):
        store.get("Pod", "missing", "default")
    with pytest.raises(NotFound):
        store.delete("Pod", "missing", "default")
    store.create(new_instaslice("node-0"))
    with pytest.raises(AlreadyExists):
        store.create(new_instaslice("node-0"))


def test_conflict_on_stale_resource_version(store):
    store.create(new_instaslice("node-0"))
    a = store.get("Instaslice", "node-0", INSTASLICE_NS)
    b = store.get("Instaslice", "node-0", INSTASLICE_NS)
    a["spec"]["cordoned"] = True
    store.update(a)
    b["spec"]["cordoned"] = False
    with pytest.raises(Conflict):
        store.update(b)  # stale rv must 409 -> Conflict


def test_finalizer_two_phase_delete_rfc3339(store):
    """delete() on a finalized pod sets an RFC3339 deletionTimestamp STRING
    (not an epoch float); removing the finalizer completes the delete."""
    from instaslice_amd import FINALIZER_NAME
    from instaslice_amd.api.types import timestamp_epoch

    store.create(new_pod("p1", profile="cpx-1x36"))
    store.delete("Pod", "p1", "default")
    pod = store.get("Pod", "p1", "default")
    dt = pod["metadata"]["deletionTimestamp"]
    assert isinstance(dt, str) and "T" in dt  # RFC3339, the real-server shape
    assert abs(timestamp_epoch(dt) - time.time()) < 5.0
    pod["metadata"]["finalizers"] = [
        f for f in pod["metadata"]["finalizers"] if f != FINALIZER_NAME]
    store.update(pod)
    with pytest.raises(NotFound):
        store.get("Pod", "p1", "default")


def test_patch_emulation_full_op_grammar(store):
    """The client-side patch emulation must honor the op grammar the
    reconcilers rely on: test (value + absent), set, delete, add_to_set,
    remove_from_set, merge — and raise Conflict through on a failed test."""
    store.create(new_instaslice("node-0"))
    store.patch("Instaslice", "node-0", INSTASLICE_NS, [
        {"op": "test", "path": ["spec", "allocations", "u1"], "absent": True},
        {"op": "set", "path": ["spec", "allocations", "u1"],
         "value": {"podUUID": "u1", "podName": "p", "namespace": "default",
                   "nodename": "node-0", "profile": "cpx-1x36",
                   "gpuUUID": "g", "ordinal": 0, "start": 0, "size": 1,
                   "allocationStatus": "creating"}},
        {"op": "merge", "path": ["spec", "prepared"], "value": {"pu": {
            "parentGpuUUID": "g", "ordinal": 0, "computeMode": "CPX",
            "memoryMode": "NPS1", "xcds": 1, "memoryGB": 36,
            "podUUID": "u1"}}},
        {"op": "add_to_set", "path": ["spec", "gpus", "g", "usedOrdinals"],
         "value": 0},
    ])
    cr = store.get("Instaslice", "node-0", INSTASLICE_NS)
    assert cr["spec"]["allocations"]["u1"]["allocationStatus"] == "creating"
    assert cr["spec"]["prepared"]["pu"]["xcds"] == 1
    assert cr["spec"]["gpus"]["g"]["usedOrdinals"] == [0]
    with pytest.raises(Conflict):
        store.patch("Instaslice", "node-0", INSTASLICE_NS, [
            {"op": "test",
             "path": ["spec", "allocations", "u1", "allocationStatus"],
             "value": "created"},  # actual: creating
            {"op": "set",
             "path": ["spec", "allocations", "u1", "allocationStatus"],
             "value": "ungated"},
        ])
    cr = store.get("Instaslice", "node-0", INSTASLICE_NS)
    assert cr["spec"]["allocations"]["u1"]["allocationStatus"] == "creating"
    store.patch("Instaslice", "node-0", INSTASLICE_NS, [
        {"op": "remove_from_set", "path": ["spec", "gpus", "g", "usedOrdinals"],
         "value": 0},
        {"op": "delete", "path": ["spec", "allocations", "u1"]},
    ])
    cr = store.get("Instaslice", "node-0", INSTASLICE_NS)
    assert cr["spec"]["allocations"] == {}
    assert cr["spec"]["gpus"]["g"]["usedOrdinals"] == []


def test_batch_partial_failure_shapes(store):
    """Reconcilers index batch results positionally and read
    error.type — the adapter must preserve both through failures."""
    store.create(new_instaslice("node-0"))
    res = store.batch([
        {"verb": "create", "obj": {"apiVersion": "v1", "kind": "ConfigMap",
                                   "metadata": {"name": "c1",
                                                "namespace": "default"},
                                   "data": {}}},
        {"verb": "patch", "kind": "Instaslice", "name": "node-0",
         "namespace": INSTASLICE_NS, "ops": [
             {"op": "test", "path": ["spec", "allocations", "nope"],
              "value": "x"}]},
        {"verb": "delete", "kind": "ConfigMap", "name": "missing",
         "namespace": "default"},
        {"verb": "patch", "kind": "Instaslice", "name": "node-0",
         "namespace": INSTASLICE_NS, "ops": [
             {"op": "set", "path": ["spec", "cordoned"], "value": True}]},
    ], quiet=True)
    assert res[0]["ok"]
    assert not res[1]["ok"] and res[1]["error"]["type"] == "Conflict"
    assert not res[2]["ok"] and res[2]["error"]["type"] == "NotFound"
    assert res[3]["ok"]  # later requests still ran (sequential, not aborted)
    assert store.get("Instaslice", "node-0", INSTASLICE_NS)["spec"]["cordoned"]


def test_status_subresource_heartbeat_survives(store):
    """status.heartbeat (agent liveness protocol) must land on a real API
    server where status is a subresource — a plain replace drops it."""
    store.create(new_instaslice("node-0"))
    store.patch("Instaslice", "node-0", INSTASLICE_NS, [
        {"op": "set", "path": ["status", "heartbeat"], "value": 1234.5},
    ])
    cr = store.get("Instaslice", "node-0", INSTASLICE_NS)
    assert cr["status"]["heartbeat"] == 1234.5
    # and spec writes don't clobber status
    store.patch("Instaslice", "node-0", INSTASLICE_NS, [
        {"op": "set", "path": ["spec", "cordoned"], "value": True},
    ])
    cr = store.get("Instaslice", "node-0", INSTASLICE_NS)
    assert cr["status"]["heartbeat"] == 1234.5 and cr["spec"]["cordoned"]


def test_node_capacity_add_and_remove(store):
    """Capacity pins go through the Node STATUS subresource, and removal
    must actually remove the key (merge-patches can't delete — replace
    semantics required)."""
    store.create({"apiVersion": "v1", "kind": "Node",
                  "metadata": {"name": "node-0", "namespace": ""},
                  "status": {"capacity": {}}})
    store.patch("Node", "node-0", "", [
        {"op": "set", "path": ["status", "capacity", "org.instaslice/p1"],
         "value": 1}])
    assert store.get("Node", "node-0", "")["status"]["capacity"][
        "org.instaslice/p1"] == 1
    store.patch("Node", "node-0", "", [
        {"op": "delete", "path": ["status", "capacity", "org.instaslice/p1"]}])
    assert "org.instaslice/p1" not in store.get(
        "Node", "node-0", "")["status"]["capacity"]


def test_structural_pruning_keeps_protocol_fields(store):
    """The API server prunes unknown spec fields per the structural schema.
    Every field the protocol writes must survive (the advisor-r1 failure
    mode: pruned cordoned/agentManagedTeardown/wholeGpu)."""
    cr = new_instaslice("node-0")
    cr["spec"]["cordoned"] = True
    cr["spec"]["agentManagedTeardown"] = True
    cr["spec"]["nominations"] = {
        "uid1": {"gpuUUID": "g", "wholeGpu": True, "ts": 1.0}}
    cr["spec"]["totally_bogus_field"] = {"x": 1}
    store.create(cr)
    got = store.get("Instaslice", "node-0", INSTASLICE_NS)
    assert got["spec"]["cordoned"] is True
    assert got["spec"]["agentManagedTeardown"] is True
    assert got["spec"]["nominations"]["uid1"]["wholeGpu"] is True
    assert "totally_bogus_field" not in got["spec"], (
        "fake server must prune unknown fields, or this tier proves nothing")


def test_watch_replay_then_stream(store):
    """Engine informers need LIST+WATCH: pre-existing objects replayed as
    ADDED, then live events; resourceVersion sequencing loses nothing."""
    store.create(new_instaslice("node-0"))
    w = store.watch("Instaslice", replay=True)
    ev = w.next(timeout=5.0)
    assert ev is not None and ev[0] == "ADDED"
    assert ev[1]["metadata"]["name"] == "node-0"
    store.patch("Instaslice", "node-0", INSTASLICE_NS, [
        {"op": "set", "path": ["spec", "cordoned"], "value": True}])
    for _ in range(10):
        ev = w.next(timeout=5.0)
        if ev and ev[0] == "MODIFIED" and ev[1]["spec"].get("cordoned"):
            break
    else:
        pytest.fail("MODIFIED event with the patch never arrived")
    w.stop()


def test_watch_filtered_merge(store):
    store.create(new_instaslice("node-0"))
    store.create(new_instaslice("node-1"))
    w = store.watch(filters=[
        {"kind": "Instaslice", "name": "node-1", "namespace": INSTASLICE_NS}])
    seen = set()
    deadline = time.monotonic() + 5.0
    while time.monotonic() < deadline and not seen:
        ev = w.next(timeout=0.5)
        if ev:
            seen.add(ev[1]["metadata"]["name"])
    assert seen == {"node-1"}
    w.stop()


def test_update_with_retry_under_contention(store):
    store.create(new_instaslice("node-0"))
    errors = []

    def bump(field):
        # spec.gpuUuids is additionalProperties:string in the CRD schema,
        # so concurrent counters stored as strings survive pruning
        def mut(obj):
            m = obj["spec"].setdefault("gpuUuids", {})
            m[field] = str(int(m.get(field, "0")) + 1)
            return obj
        for _ in range(20):
            try:
                store.update_with_retry(
                    "Instaslice", "node-0", INSTASLICE_NS, mut)
            except Exception as e:  # noqa: BLE001
                errors.append(e)

    threads = [threading.Thread(target=bump, args=(f"f{i}",)) for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors
    m = store.get("Instaslice", "node-0", INSTASLICE_NS)["spec"]["gpuUuids"]
    assert all(m[f"f{i}"] == "20" for i in range(4))


# -- the full operator lifecycle over the adapter ---------------------------


@pytest.fixture
def k8s_cluster():
    from instaslice_amd.runtime.cluster import Cluster
    from instaslice_amd.smi import FakeAmdSmi

    kubernetes.reset_server()
    c = Cluster(store=K8sStore(), teardown_grace_s=0.0)
    c.add_node("node-0", FakeAmdSmi(num_gpus=2, node_name="node-0"))
    c.start()
    yield c
    c.stop()


def test_k8s_happy_path_single_pod(k8s_cluster):
    """The test_e2e_fake.py happy path, verbatim semantics, over K8sStore:
    gated pod -> placement -> partition realized -> ConfigMap env ->
    ungated, with allocation/prepared/capacity all consistent."""
    c = k8s_cluster
    c.submit_pod("p1", "cpx-1x36")
    pod = c.wait_pod_scheduled("p1", timeout=20.0)
    assert not pod["spec"]["schedulingGates"]
    env = c.pod_env("p1")
    assert env["ROCR_VISIBLE_DEVICES"] == "0"
    assert env["INSTASLICE_PARTITION_UUID"]
    allocs = c.allocations("node-0")
    (alloc,) = allocs.values()
    assert alloc["allocationStatus"] == AllocationStatus.UNGATED
    assert alloc["computeMode"] == "CPX"
    (prep,) = c.prepared("node-0").values()
    assert prep["xcds"] == 1 and prep["memoryGB"] == 36
    node = c.store.get("Node", "node-0", "")
    assert node["status"]["capacity"].get("org.instaslice/p1") == 1


def test_k8s_teardown_cleans_everything(k8s_cluster):
    c = k8s_cluster
    c.submit_pod("p1", "cpx-1x36")
    c.wait_pod_scheduled("p1", timeout=20.0)
    c.delete_pod("p1")
    c.wait_pod_gone("p1", timeout=20.0)
    c.wait_allocations_empty("node-0", timeout=20.0)
    with pytest.raises(NotFound):
        c.pod_env("p1")
    node = c.store.get("Node", "node-0", "")
    assert "org.instaslice/p1" not in (node["status"].get("capacity") or {})


def test_k8s_contention_two_pods_race_for_slots(k8s_cluster):
    """Concurrent placements over the adapter: conflicts must resolve via
    the CAS/retry machinery, with no ordinal double-booked."""
    c = k8s_cluster
    for i in range(8):
        c.submit_pod(f"p{i}", "cpx-1x36")
    for i in range(8):
        c.wait_pod_scheduled(f"p{i}", timeout=30.0)
    allocs = c.allocations("node-0")
    assert len(allocs) == 8
    slots = {(a["gpuUUID"], a["ordinal"]) for a in allocs.values()}
    assert len(slots) == 8, "ordinal double-booked under contention"


def test_k8s_heartbeat_lands_in_status(k8s_cluster):
    c = k8s_cluster
    deadline = time.monotonic() + 10.0
    while time.monotonic() < deadline:
        cr = c.store.get("Instaslice", "node-0", INSTASLICE_NS)
        if (cr.get("status") or {}).get("heartbeat"):
            return
        time.sleep(0.1)
    pytest.fail("agent heartbeat never landed in CR status over K8sStore")


def test_k8s_randomized_lifecycle_invariants():
    """The randomized invariant sweep (tests/test_invariants_e2e.py) over
    the Kubernetes adapter: RFC3339 timestamps, status subresources and
    client-side patch emulation must preserve the same global consistency
    under interleaved submit/delete churn."""
    import random

    from instaslice_amd.runtime.cluster import Cluster
    from instaslice_amd.smi import FakeAmdSmi

    sys.path.insert(0, str(Path(__file__).resolve().parent))
    from test_invariants_e2e import _check_invariants, _quiesce

    kubernetes.reset_server()
    rng = random.Random(5)
    c = Cluster(store=K8sStore(), teardown_grace_s=0.0)
    c.add_node("node-0", FakeAmdSmi(num_gpus=2, node_name="node-0"))
    c.start()
    live, counter = set(), 0
    try:
        for _ in range(4):
            for _ in range(rng.randint(3, 8)):
                if live and rng.random() < 0.45:
                    victim = rng.choice(sorted(live))
                    live.discard(victim)
                    c.delete_pod(victim)
                else:
                    name = f"k8s-p{counter}"
                    counter += 1
                    c.submit_pod(name, rng.choice(
                        ["cpx-1x36", "qpx-2x72", "dpx-4x144"]))
                    live.add(name)
            _quiesce(c, live, timeout=40.0)
            _check_invariants(c, "node-0")
        for name in sorted(live):
            c.delete_pod(name)
        live.clear()
        _quiesce(c, live, timeout=40.0)
        _check_invariants(c, "node-0")
    finally:
        c.stop()
