"""Placer tests: first-fit + packed-fit over the MI355X mode matrix.

Exhaustive over profile mixes per SURVEY.md §7.2 item 6 ('unit: placer
exhaustive over profile mixes').
"""

import itertools

import pytest

from instaslice_amd.controller.policy import (
    FirstFitPolicy,
    GpuView,
    PackedFitPolicy,
    Placement,
)
from instaslice_amd.partition.profiles import ComputeMode, MemoryMode, mi355x_catalog

CAT = mi355x_catalog()
CPX = CAT.by_name("cpx-1x36")
QPX = CAT.by_name("qpx-2x72")
DPX = CAT.by_name("dpx-4x144")
SPX = CAT.by_name("spx-8x288")


def mk_view(idx, mode=ComputeMode.SPX, mem=MemoryMode.NPS1, occupied=()):
    return GpuView(
        node="n0", uuid=f"gpu-{idx}", index=idx, memory_gb=288,
        compute_mode=mode, memory_mode=mem, occupied=set(occupied),
    )


def test_firstfit_prefers_existing_mode():
    views = [
        mk_view(0, ComputeMode.SPX),
        mk_view(1, ComputeMode.CPX, MemoryMode.NPS4, occupied={0, 1}),
    ]
    p = FirstFitPolicy().place(CPX, views)
    assert p.gpu_uuid == "gpu-1" and p.ordinal == 2 and not p.needs_mode_change


def test_firstfit_reconfigures_idle_gpu():
    views = [mk_view(0, ComputeMode.SPX), mk_view(1, ComputeMode.SPX)]
    p = FirstFitPolicy().place(CPX, views)
    assert p.gpu_uuid == "gpu-0" and p.ordinal == 0 and p.needs_mode_change
    assert p.compute_mode == "CPX" and p.memory_mode == "NPS1"  # sticky memory


def test_mode_change_requires_idle_gpu():
    # gpu-0 in SPX but occupied (the SPX partition is allocated): cannot flip
    views = [mk_view(0, ComputeMode.SPX, occupied={0})]
    assert FirstFitPolicy().place(CPX, views) is None
    assert PackedFitPolicy().place(CPX, views) is None


def test_full_cpx_gpu_rejects():
    views = [mk_view(0, ComputeMode.CPX, occupied=set(range(8)))]
    assert FirstFitPolicy().place(CPX, views) is None


def test_memory_mode_planned_when_illegal_under_target():
    # NPS4 GPU asked to go SPX: NPS4 is illegal under SPX -> plan NPS1
    views = [mk_view(0, ComputeMode.CPX, MemoryMode.NPS4)]
    p = FirstFitPolicy().place(SPX, views)
    assert p.needs_mode_change and p.memory_mode == "NPS1"


def test_packedfit_packs_tightest_same_mode_gpu():
    views = [
        mk_view(0, ComputeMode.CPX, occupied={0}),
        mk_view(1, ComputeMode.CPX, occupied={0, 1, 2}),
        mk_view(2, ComputeMode.SPX),
    ]
    p = PackedFitPolicy().place(CPX, views)
    assert p.gpu_uuid == "gpu-1" and p.ordinal == 3  # most-occupied wins


def test_packedfit_prefers_matching_memory_mode_on_reconfig():
    views = [
        mk_view(0, ComputeMode.SPX, MemoryMode.NPS1),
        mk_view(1, ComputeMode.DPX, MemoryMode.NPS1),
    ]
    # both idle, both need a change; equal memory validity -> lower index
    p = PackedFitPolicy().place(CPX, views)
    assert p.gpu_uuid == "gpu-0"


@pytest.mark.parametrize("policy_cls", [FirstFitPolicy, PackedFitPolicy])
def test_exhaustive_mix_8gpu_node(policy_cls):
    """Feed every 3-profile request mix to an 8-GPU node, applying each
    placement; assert invariants: ordinals unique per GPU, mode changes only
    on idle GPUs, and the pod NEVER gets a smaller partition than requested
    (upsize/coarsening may legally hand out a bigger one)."""
    profiles = [CPX, QPX, DPX, SPX]
    for mix in itertools.product(profiles, repeat=3):
        views = [mk_view(i) for i in range(8)]
        policy = policy_cls()
        for prof in mix:
            p = policy.place(prof, views)
            assert p is not None, f"mix {[m.name for m in mix]} failed to place"
            v = next(v for v in views if v.uuid == p.gpu_uuid)
            if p.needs_mode_change:
                assert not v.occupied
                v.compute_mode = ComputeMode(p.compute_mode)
                v.memory_mode = MemoryMode(p.memory_mode)
            # the placement's mode is authoritative (upsize/coarsening may
            # differ from the profile's own mode) and must match the GPU
            assert v.compute_mode.value == p.compute_mode
            n_parts = v.compute_mode.num_partitions
            assert p.ordinal not in v.occupied
            assert p.ordinal < n_parts
            assert 8 // n_parts >= prof.xcds, "pod must never be undersized"
            v.occupied.add(p.ordinal)


@pytest.mark.parametrize("policy_cls", [FirstFitPolicy, PackedFitPolicy])
def test_capacity_64_cpx_partitions_per_node(policy_cls):
    """8 GPUs x CPX = 64 schedulable partitions (SURVEY.md §5 'long-context'
    analog note); the 65th request must not fit."""
    views = [mk_view(i) for i in range(8)]
    policy = policy_cls()
    placed = []
    for _ in range(64):
        p = policy.place(CPX, views)
        assert p is not None
        v = next(v for v in views if v.uuid == p.gpu_uuid)
        if p.needs_mode_change:
            v.compute_mode = ComputeMode.CPX
        v.occupied.add(p.ordinal)
        placed.append(p)
    assert policy.place(CPX, views) is None
    assert len({(p.gpu_uuid, p.ordinal) for p in placed}) == 64


def test_packedfit_leaves_room_for_spx():
    """Packing keeps whole GPUs free: 7 CPX pods should land on one GPU so an
    SPX (whole-GPU) job still fits on the other."""
    views = [mk_view(0), mk_view(1)]
    policy = PackedFitPolicy()
    for _ in range(7):
        p = policy.place(CPX, views)
        v = next(v for v in views if v.uuid == p.gpu_uuid)
        if p.needs_mode_change:
            v.compute_mode = ComputeMode.CPX
        v.occupied.add(p.ordinal)
    assert views[0].compute_mode is ComputeMode.CPX and len(views[0].occupied) == 7
    p = policy.place(SPX, views)
    assert p is not None and p.gpu_uuid == "gpu-1"


def test_spreadfit_balances_across_gpus():
    from instaslice_amd.controller.policy import SpreadFitPolicy

    views = [mk_view(0), mk_view(1)]
    policy = SpreadFitPolicy()
    parents = []
    for _ in range(4):
        p = policy.place(CPX, views)
        v = next(v for v in views if v.uuid == p.gpu_uuid)
        if p.needs_mode_change:
            v.compute_mode = ComputeMode.CPX
        v.occupied.add(p.ordinal)
        parents.append(p.gpu_uuid)
    # 4 pods over 2 GPUs: 2 each (packed-fit would put all 4 on gpu-0)
    assert parents.count("gpu-0") == 2 and parents.count("gpu-1") == 2


@pytest.mark.parametrize("policy_name", ["first-fit", "packed-fit", "spread-fit"])
def test_policy_registry(policy_name):
    from instaslice_amd.controller.policy import get_policy

    assert get_policy(policy_name).name == policy_name


def test_cross_node_packing_global_argmax():
    """packed-fit must pack onto the partially-filled GPU of a LATER node
    rather than flipping the idle GPU of the first node — cluster-wide
    argmax, not first-node-that-fits (the reference's findDeviceForASlice
    takes the first node, instaslice_controller.go:240-262)."""
    import time

    from instaslice_amd.runtime.cluster import Cluster
    from instaslice_amd.smi.fake import FakeAmdSmi

    c = Cluster(teardown_grace_s=0.0, policy="packed-fit")
    c.add_node("node-a", FakeAmdSmi(num_gpus=1, node_name="node-a"))
    c.add_node("node-b", FakeAmdSmi(num_gpus=1, node_name="node-b"))
    c.start()
    try:
        # seed node-b (second in name order) with one CPX pod
        c.submit_pod("seed", "cpx-1x36", node="node-b")
        c.wait_pod_scheduled("seed")
        # an unpinned CPX pod must join node-b's CPX GPU, not flip node-a
        c.submit_pod("join", "cpx-1x36")
        c.wait_pod_scheduled("join")
        allocs_b = c.allocations("node-b")
        assert any(a["podName"] == "join" for a in allocs_b.values()), (
            "packed-fit flipped a fresh GPU instead of packing cross-node")
        # spread-fit does the opposite: a third pod goes to the idle node-a
        time.sleep(0)  # (documentation beat: policies diverge here)
    finally:
        c.stop()


def test_cross_node_spreading():
    from instaslice_amd.runtime.cluster import Cluster
    from instaslice_amd.smi.fake import FakeAmdSmi

    c = Cluster(teardown_grace_s=0.0, policy="spread-fit")
    c.add_node("node-a", FakeAmdSmi(num_gpus=1, node_name="node-a"))
    c.add_node("node-b", FakeAmdSmi(num_gpus=1, node_name="node-b"))
    c.start()
    try:
        c.submit_pod("s0", "cpx-1x36", node="node-a")
        c.wait_pod_scheduled("s0")
        # spread-fit prefers the idle GPU on node-b over packing node-a
        c.submit_pod("s1", "cpx-1x36")
        c.wait_pod_scheduled("s1")
        assert any(a["podName"] == "s1"
                   for a in c.allocations("node-b").values()), (
            "spread-fit packed instead of spreading cross-node")
    finally:
        c.stop()


def test_upsize_serves_small_request_from_bigger_hole():
    """cpx request, CPX GPUs full, a QPX hole on an occupied GPU, no idle
    GPU: the request rides the 2-XCD slot instead of going unschedulable."""
    views = [
        mk_view(0, ComputeMode.CPX, occupied=set(range(8))),
        mk_view(1, ComputeMode.QPX, occupied={0, 1, 2}),
    ]
    p = PackedFitPolicy().place(CPX, views)
    assert p is not None
    assert p.gpu_uuid == "gpu-1" and p.compute_mode == "QPX"
    assert not p.needs_mode_change


def test_upsize_never_on_idle_gpu():
    """An idle GPU is flip territory: a cpx request must flip it (or, with
    QPX present and no big tenants, coarsen to QPX) — never 'upsize' into
    the idle GPU's current-mode slot."""
    views = [mk_view(0, ComputeMode.DPX)]  # idle, wrong mode
    p = PackedFitPolicy().place(CPX, views)
    assert p.needs_mode_change


def test_upsize_waste_capped():
    """cpx into a DPX hole wastes 3 XCDs > cap 2: refused (stranding beats
    the saving — churn study)."""
    views = [
        mk_view(0, ComputeMode.CPX, occupied=set(range(8))),
        mk_view(1, ComputeMode.DPX, occupied={0}),
    ]
    p = PackedFitPolicy().place(CPX, views)
    assert p is None


def test_coarsening_flips_idle_to_qpx_for_small_request():
    """QPX present + no big tenants: a cpx request flips the idle GPU to
    QPX, keeping small-pod capacity one fungible class."""
    views = [
        mk_view(0, ComputeMode.QPX, occupied={0, 1, 2, 3}),
        mk_view(1, ComputeMode.SPX),  # idle
    ]
    p = PackedFitPolicy().place(CPX, views)
    assert p.needs_mode_change and p.compute_mode == "QPX"


def test_coarsening_disabled_by_big_tenants():
    """A DPX tenant on the node disables coarsening: the cpx request flips
    the idle GPU to its own CPX mode (XCD efficiency wins)."""
    views = [
        mk_view(0, ComputeMode.QPX, occupied={0, 1, 2, 3}),
        mk_view(1, ComputeMode.DPX, occupied={0, 1}),
        mk_view(2, ComputeMode.SPX),  # idle
    ]
    p = PackedFitPolicy().place(CPX, views)
    assert p.needs_mode_change and p.compute_mode == "CPX"


def test_qpx_prefers_flip_over_upsize_when_idle_exists():
    """2-XCD profiles flip an idle GPU rather than stranding themselves in
    a DPX hole (their repacked arrangement wants dedicated GPUs)."""
    views = [
        mk_view(0, ComputeMode.DPX, occupied={0}),
        mk_view(1, ComputeMode.SPX),  # idle
    ]
    p = PackedFitPolicy().place(QPX, views)
    assert p.needs_mode_change and p.compute_mode == "QPX"
    # ...but with no idle GPU, the DPX hole beats unschedulable
    views = [mk_view(0, ComputeMode.DPX, occupied={0})]
    p = PackedFitPolicy().place(QPX, views)
    assert p is not None and p.compute_mode == "DPX" and not p.needs_mode_change


def test_gang_prefers_xgmi_neighbor_over_far_gpu():
    """2-GPU gang with a partial xGMI mesh: the second member must land on
    a 1-hop neighbor of the first member's GPU, not on a topologically
    distant GPU."""
    views = [
        mk_view(0, ComputeMode.CPX, occupied=set(range(8))),  # gang GPU full
        mk_view(1, ComputeMode.CPX, occupied={0}),  # far, same mode, free
        mk_view(2, ComputeMode.CPX, occupied={0}),  # neighbor, same mode, free
    ]
    p = PackedFitPolicy().place(
        CPX, views,
        prefer_gpus=frozenset({"gpu-0"}),
        xgmi_neighbors=frozenset({"gpu-2"}),
    )
    assert p.gpu_uuid == "gpu-2"


def test_gang_neighbor_flip_beats_far_same_mode():
    """An idle xGMI neighbor (needing a mode flip) outranks a free
    same-mode slot on a distant GPU — for RCCL gangs link locality
    dominates one flip's cost."""
    views = [
        mk_view(0, ComputeMode.CPX, occupied=set(range(8))),  # gang GPU full
        mk_view(1, ComputeMode.CPX, occupied={0}),  # far, same mode, free
        mk_view(2, ComputeMode.SPX),                # neighbor, idle
    ]
    p = PackedFitPolicy().place(
        CPX, views,
        prefer_gpus=frozenset({"gpu-0"}),
        xgmi_neighbors=frozenset({"gpu-2"}),
    )
    assert p.gpu_uuid == "gpu-2" and p.needs_mode_change


def test_gang_same_gpu_still_top():
    views = [
        mk_view(0, ComputeMode.CPX, occupied={0}),  # gang GPU, slots free
        mk_view(1, ComputeMode.CPX, occupied={0}),  # neighbor
    ]
    p = PackedFitPolicy().place(
        CPX, views,
        prefer_gpus=frozenset({"gpu-0"}),
        xgmi_neighbors=frozenset({"gpu-1"}),
    )
    assert p.gpu_uuid == "gpu-0"


def test_gang_lands_on_neighbor_clique_e2e():
    """End-to-end over the fake: 8-GPU node modeled as two 4-GPU xGMI
    cliques; a gang whose first member lands in clique A must keep its
    second member inside clique A (CR carries discovered topology)."""
    import time

    from instaslice_amd.runtime.cluster import Cluster
    from instaslice_amd.smi.fake import FakeAmdSmi
    from instaslice_amd.controller.reconciler import INSTASLICE_NS

    smi = FakeAmdSmi(num_gpus=8, node_name="node-0",
                     xgmi_cliques=[[0, 1, 2, 3], [4, 5, 6, 7]])
    c = Cluster(teardown_grace_s=0.0, policy="packed-fit")
    c.add_node("node-0", smi)
    c.start()
    try:
        cr = c.store.get("Instaslice", "node-0", INSTASLICE_NS)
        assert cr["spec"]["topology"], "CR must carry discovered topology"
        # occupy GPU0 fully with the first gang member's profile spx-8x288
        # (whole GPU) so the second member CANNOT share the GPU
        c.submit_pod("g1", "spx-8x288", group="team")
        c.wait_pod_scheduled("g1")
        g1 = next(iter(c.allocations("node-0").values()))
        first_gpu = g1["gpuUUID"]
        c.submit_pod("g2", "spx-8x288", group="team")
        c.wait_pod_scheduled("g2")
        allocs = c.allocations("node-0")
        g2 = next(a for a in allocs.values() if a["podName"] == "g2")
        # the second member must sit on a 1-hop neighbor of the first
        topo = cr["spec"]["topology"]
        assert g2["gpuUUID"] in topo.get(first_gpu, {}), (
            f"gang member landed outside the xGMI clique: "
            f"{g2['gpuUUID']} not neighbor of {first_gpu}")
    finally:
        c.stop()
