"""Randomized lifecycle sweeps with quiesce-time invariant checks.

The r2 K8sStore tier exposed two teardown TOCTOU races that orphaned
prepared entries / allocations. This test makes that whole CLASS of bug a
regression: drive the full controller+agent stack through randomized
submit/delete interleavings (deterministic seeds), wait for quiesce, and
assert global consistency between every pair of stores the protocol
writes: allocations <-> prepared <-> usedOrdinals <-> ConfigMaps <->
node capacity pins <-> live pods.
"""

import random
import time

import pytest

from instaslice_amd.api.types import AllocationStatus
from instaslice_amd.controller.reconciler import INSTASLICE_NS
from instaslice_amd.runtime.cluster import Cluster
from instaslice_amd.smi.fake import FakeAmdSmi

PROFILES = ["cpx-1x36", "qpx-2x72", "dpx-4x144"]


def _quiesce(c: Cluster, live: set, timeout: float = 20.0) -> None:
    """Wait until every live pod is scheduled-or-unschedulable and every
    deleted pod has fully drained."""
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        settled = True
        pods = {p["metadata"]["name"] for p in c.store.list("Pod")}
        if pods != live:
            settled = False
        else:
            for name in live:
                pod = c.store.get("Pod", name, "default")
                gates = pod["spec"].get("schedulingGates")
                ann = pod["metadata"].get("annotations") or {}
                if gates and "org.instaslice/unschedulable" not in str(ann):
                    settled = False
                    break
        if settled:
            # two consecutive checks 50ms apart = stable
            time.sleep(0.05)
            pods2 = {p["metadata"]["name"] for p in c.store.list("Pod")}
            if pods2 == pods:
                return
        time.sleep(0.02)
    raise TimeoutError(f"cluster never quiesced; live={sorted(live)}")


def _check_invariants(c: Cluster, node: str) -> None:
    cr = c.store.get("Instaslice", node, INSTASLICE_NS)
    spec = cr["spec"]
    allocs = spec.get("allocations") or {}
    prepared = spec.get("prepared") or {}
    pods = {p["metadata"]["name"]: p for p in c.store.list("Pod")}
    cms = {m["metadata"]["name"] for m in c.store.list("ConfigMap")}
    node_obj = c.store.get("Node", node, "")
    capacity = node_obj["status"].get("capacity") or {}

    # 1. every allocation's pod exists and vice versa for realized pods
    for uid, a in allocs.items():
        assert a["podName"] in pods, f"allocation for dead pod {a['podName']}"
    # 2. prepared entries belong to a live allocation (the TOCTOU orphan)
    alloc_uids = set(allocs)
    for puid, prep in prepared.items():
        assert prep["podUUID"] in alloc_uids, (
            f"ORPHANED prepared entry {puid} (pod_uuid={prep['podUUID']})")
    # 3. realized allocations have exactly one prepared entry + ConfigMap
    #    + capacity pin
    for uid, a in allocs.items():
        if a["allocationStatus"] in (AllocationStatus.CREATED,
                                     AllocationStatus.UNGATED):
            n_prep = sum(1 for p in prepared.values()
                         if p["podUUID"] == uid)
            assert n_prep == 1, f"{a['podName']}: {n_prep} prepared entries"
            assert a["podName"] in cms, f"{a['podName']}: ConfigMap missing"
            assert capacity.get(f"org.instaslice/{a['podName']}") == 1, (
                f"{a['podName']}: capacity pin missing")
    # 4. no ConfigMap without its pod (ours carry the partition keys)
    for name in cms:
        cm = c.store.get("ConfigMap", name, "default")
        if "INSTASLICE_PARTITION_UUID" in (cm.get("data") or {}):
            assert name in pods, f"ORPHANED ConfigMap {name}"
    # 5. usedOrdinals mirror allocations per GPU
    by_gpu = {}
    for a in allocs.values():
        by_gpu.setdefault(a["gpuUUID"], set()).add(a["ordinal"])
    for uuid, gd in (spec.get("gpus") or {}).items():
        used = set(gd.get("usedOrdinals") or [])
        want = by_gpu.get(uuid, set())
        assert used == want, (
            f"gpu {uuid[:8]}: usedOrdinals {sorted(used)} != "
            f"allocations {sorted(want)}")
    # 6. no two allocations share a (gpu, ordinal)
    slots = [(a["gpuUUID"], a["ordinal"]) for a in allocs.values()]
    assert len(slots) == len(set(slots)), "double-booked ordinal"


@pytest.mark.parametrize("seed", [11, 23, 47])
def test_randomized_lifecycle_invariants(seed):
    rng = random.Random(seed)
    c = Cluster(teardown_grace_s=0.0, policy="packed-fit")
    c.add_node("node-0", FakeAmdSmi(num_gpus=4, node_name="node-0"))
    c.start()
    live: set = set()
    counter = 0
    try:
        for round_ in range(8):
            # burst of random ops without waiting in between (maximum
            # interleaving pressure on the controller/agent protocol)
            for _ in range(rng.randint(3, 10)):
                if live and rng.random() < 0.45:
                    victim = rng.choice(sorted(live))
                    live.discard(victim)
                    c.delete_pod(victim)
                else:
                    name = f"s{seed}-p{counter}"
                    counter += 1
                    c.submit_pod(name, rng.choice(PROFILES))
                    live.add(name)
            _quiesce(c, live)
            _check_invariants(c, "node-0")
        # full drain at the end: everything must clean to zero
        for name in sorted(live):
            c.delete_pod(name)
        live.clear()
        _quiesce(c, live)
        _check_invariants(c, "node-0")
        cr = c.store.get("Instaslice", "node-0", INSTASLICE_NS)
        assert not cr["spec"].get("allocations")
        assert not cr["spec"].get("prepared")
    finally:
        c.stop()
