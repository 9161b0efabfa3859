"""Property-based invariants: random submit/delete interleavings must never
corrupt the allocation state. Hypothesis drives op sequences against the full
controller+agent stack on FakeAmdSmi; after settling, the CR must satisfy the
structural invariants the whole design rests on (no double-booked ordinal,
mode coherence, prepared/allocation/ConfigMap/capacity 1:1)."""

import os
import time

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from instaslice_amd.controller.reconciler import INSTASLICE_NS
from instaslice_amd.runtime.cluster import Cluster
from instaslice_amd.smi.fake import FakeAmdSmi

PROFILES = ["cpx-1x36", "qpx-2x72", "dpx-4x144", "spx-8x288"]

# an op is (kind, argument): submit profile i, or delete the j-th live pod
ops_strategy = st.lists(
    st.one_of(
        st.tuples(st.just("submit"), st.integers(0, len(PROFILES) - 1)),
        st.tuples(st.just("delete"), st.integers(0, 30)),
    ),
    min_size=1, max_size=25,
)


def _settle(c: Cluster, submitted, deleted, timeout=45.0):
    """Wait until every live pod is scheduled or marked unschedulable and
    every deleted pod is fully drained."""
    deadline = time.monotonic() + timeout
    live = [p for p in submitted if p not in deleted]
    while time.monotonic() < deadline:
        pods = {p["metadata"]["name"]: p for p in c.store.list("Pod")}
        settled = True
        for name in live:
            p = pods.get(name)
            if p is None:
                settled = False
                break
            gated = bool(p["spec"].get("schedulingGates"))
            unsched = "org.instaslice/unschedulable" in (
                p["metadata"].get("annotations") or {})
            if gated and not unsched:
                settled = False
                break
        if settled:
            for name in deleted:
                if name in pods:
                    settled = False
                    break
        if settled:
            crs = c.store.list("Instaslice")
            names_alloc = {
                a["podName"]
                for cr in crs
                for a in (cr["spec"].get("allocations") or {}).values()
            }
            if not (names_alloc & set(deleted)):
                return
        time.sleep(0.02)
    pods = {p["metadata"]["name"]: (bool(p["spec"].get("schedulingGates")),
                                    p["metadata"].get("annotations"))
            for p in c.store.list("Pod")}
    raise TimeoutError(f"cluster never settled; pods={pods}")


def _check_invariants(c: Cluster):
    # 0. a pod is allocated on AT MOST one node (cross-CR uniqueness)
    owners = {}
    for cr in c.store.list("Instaslice"):
        for u in (cr["spec"].get("allocations") or {}):
            assert u not in owners, (
                f"pod {u} allocated on both {owners[u]} and "
                f"{cr['metadata']['name']}")
            owners[u] = cr["metadata"]["name"]
    for cr in c.store.list("Instaslice"):
        spec = cr["spec"]
        allocs = spec.get("allocations") or {}
        prepared = spec.get("prepared") or {}
        gpus = spec.get("gpus") or {}
        # 1. no (gpu, ordinal) double-booking
        slots = [(a["gpuUUID"], a["ordinal"]) for a in allocs.values()]
        assert len(slots) == len(set(slots)), f"double-booked slot: {slots}"
        # 2. allocation mode matches its GPU's live mode
        for a in allocs.values():
            gd = gpus[a["gpuUUID"]]
            assert a["computeMode"] == gd["computeMode"], (
                f"alloc mode {a['computeMode']} != gpu {gd['computeMode']}")
            n_parts = {"SPX": 1, "DPX": 2, "TPX": 3,
                       "QPX": 4, "CPX": 8}[gd["computeMode"]]
            assert 0 <= a["ordinal"] < n_parts
        # 3. prepared <-> created/ungated allocations 1:1
        realized = {u for u, a in allocs.items()
                    if a["allocationStatus"] in ("created", "ungated")}
        prepared_pods = {p["podUUID"] for p in prepared.values()}
        assert prepared_pods == realized, (
            f"prepared {prepared_pods} != realized {realized}")
        # 4. usedOrdinals == ordinals of realized allocations per GPU
        for uuid, gd in gpus.items():
            want = sorted(
                a["ordinal"] for u, a in allocs.items()
                if a["gpuUUID"] == uuid and u in realized
            )
            assert sorted(gd.get("usedOrdinals", [])) == want, (
                f"usedOrdinals {gd.get('usedOrdinals')} != {want}")
        # 5. ConfigMap + capacity pin exist exactly for realized pods
        cms = {m["metadata"]["name"] for m in c.store.list("ConfigMap")}
        realized_names = {allocs[u]["podName"] for u in realized}
        assert realized_names <= cms
        node = c.store.get("Node", cr["metadata"]["name"], "")
        caps = {k.split("/", 1)[1]
                for k in (node["status"].get("capacity") or {})}
        assert realized_names <= caps


@pytest.mark.parametrize("nodes", [1, 2])
@pytest.mark.parametrize("policy", ["packed-fit", "first-fit", "spread-fit"])
@settings(max_examples=int(os.environ.get("INSTASLICE_INVARIANT_EXAMPLES",
                                           "10")),
          deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(ops=ops_strategy)
def test_random_lifecycle_invariants(policy, nodes, ops):
    c = Cluster(teardown_grace_s=0.0, policy=policy)
    for n in range(nodes):
        c.add_node(f"node-{n}", FakeAmdSmi(num_gpus=2, node_name=f"node-{n}"))
    c.start()
    submitted, deleted = [], set()
    try:
        for kind, arg in ops:
            if kind == "submit":
                name = f"h{len(submitted)}"
                c.submit_pod(name, PROFILES[arg])
                submitted.append(name)
            else:
                live = [p for p in submitted if p not in deleted]
                if live:
                    victim = live[arg % len(live)]
                    c.delete_pod(victim)
                    deleted.add(victim)
        _settle(c, submitted, deleted)
        _check_invariants(c)
    finally:
        c.stop()
