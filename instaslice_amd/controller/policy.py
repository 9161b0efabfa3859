"""Placement policies: where does a partition request land?

Reference analog: AllocationPolicy interface + FirstFitPolicy
(instaslice_controller.go:48-50, :436-453) and the slot bin-packer
getStartIndexFromPreparedState (:303-384). The reference's LeftToRight/
RightToLeft policies are empty stubs (:456-469); here both shipped policies
are real.

The MI355X semantic drives the structure: AMD partitioning
is a whole-GPU *mode*, so a placement is either

  (a) a free partition ordinal on a GPU already in the profile's compute
      mode, or
  (b) ordinal 0 on an *idle* GPU that will be reconfigured (compute mode set,
      and memory mode set if valid and different from preferred) — mode
      changes require the device idle (amdsmi.h:5781).

Occupancy: a partition ordinal is occupied if any allocation (any status —
`deleted` still occupies until the daemonset cleans it) or any prepared entry
references it; this mirrors the reference's stale-prepared guard
(instaslice_controller.go:198-203 / SURVEY.md §5c).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Set, Tuple

from instaslice_amd.api.types import GpuStatus
from instaslice_amd.partition.profiles import (
    ComputeMode,
    MemoryMode,
    PartitionProfile,
    VALID_MEMORY_MODES,
)


@dataclass(frozen=True)
class Placement:
    """A placement decision for one request."""

    node: str
    gpu_uuid: str
    ordinal: int
    compute_mode: str      # mode the GPU must be in (may require a transition)
    memory_mode: str       # memory mode the GPU will have
    needs_mode_change: bool


@dataclass
class GpuView:
    """Controller-side view of one GPU assembled from the Instaslice CR."""

    node: str
    uuid: str
    index: int
    memory_gb: int
    compute_mode: ComputeMode
    memory_mode: MemoryMode
    occupied: Set[int]  # partition ordinals referenced by allocations/prepared
    # the daemonset hit a hard error flipping this GPU's mode (e.g. VM guest
    # forbids it): only same-mode placements are possible from now on
    mode_locked: bool = False


NOMINATION_TTL_S = 30.0


def _now() -> float:
    import time

    return time.time()


def build_gpu_views(node_name: str, spec: dict,
                    nominee_uid: Optional[str] = None) -> List[GpuView]:
    """Assemble GpuViews from an Instaslice CR spec (api.types.new_instaslice
    shape). Sorted by physical index for deterministic first-fit.

    A GPU with allocations whose target mode differs from the CR's live mode
    is *mid-transition* (the daemonset has not flipped it yet): its effective
    mode for placement is the allocations' target mode, so back-to-back
    requests for the same profile pack onto it instead of each grabbing a
    fresh idle GPU. The reference has no analog (MIG slices carve next to
    running ones); whole-GPU modes make pending transitions first-class."""
    occupied: Dict[str, Set[int]] = {}
    target_modes: Dict[str, Tuple[str, str]] = {}
    for alloc in (spec.get("allocations") or {}).values():
        # tolerate externally crafted/corrupt entries: a missing key here
        # would wedge every consumer (controller placement, agent capacity,
        # heartbeats) through their retry loops
        if not isinstance(alloc, dict) or "gpuUUID" not in alloc                 or "ordinal" not in alloc:
            continue
        occupied.setdefault(alloc["gpuUUID"], set()).add(alloc["ordinal"])
        if alloc.get("computeMode"):
            target_modes[alloc["gpuUUID"]] = (
                alloc["computeMode"],
                alloc.get("memoryMode") or "NPS1",
            )
    for prep in (spec.get("prepared") or {}).values():
        if not isinstance(prep, dict) or "parentGpuUUID" not in prep                 or "ordinal" not in prep:
            continue
        occupied.setdefault(prep["parentGpuUUID"], set()).add(prep["ordinal"])
    # preemption nominations: a slot freed by an eviction is reserved for
    # the preemptor (k8s nominatedNodeName analog) — everyone else sees it
    # occupied until the nominee lands or the reservation expires
    now = _now()
    for uid, nom in (spec.get("nominations") or {}).items():
        if uid == nominee_uid:
            continue
        if now - float(nom.get("ts", 0)) > NOMINATION_TTL_S:
            continue
        if nom.get("wholeGpu"):
            # whole-GPU reservation (multi-victim preemption): nobody else
            # may take any ordinal or flip it while the nominee waits
            occupied.setdefault(nom["gpuUUID"], set()).update(range(8))
        else:
            occupied.setdefault(nom["gpuUUID"], set()).add(nom["ordinal"])

    views: List[GpuView] = []
    for uuid, gd in (spec.get("gpus") or {}).items():
        g = GpuStatus.from_dict(gd)
        compute, memory = g.compute_mode, g.memory_mode
        if uuid in target_modes:
            compute, memory = target_modes[uuid]
        views.append(
            GpuView(
                node=node_name,
                uuid=uuid,
                index=spec_index(spec, uuid),
                memory_gb=g.memory_gb,
                compute_mode=ComputeMode(compute),
                memory_mode=MemoryMode(memory),
                occupied=occupied.get(uuid, set()),
                mode_locked=bool(gd.get("modeLocked")),
            )
        )
    views.sort(key=lambda v: v.index)
    return views


def spec_index(spec: dict, uuid: str) -> int:
    gd = (spec.get("gpus") or {}).get(uuid)
    if gd and "index" in gd:
        return gd["index"]
    return 0


def _target_memory_mode(profile: PartitionProfile, current: MemoryMode) -> Tuple[MemoryMode, bool]:
    """Memory mode is sticky (a set requires driver reload on bare metal,
    amdsmi.h:5861): keep the current mode if it's legal under the target
    compute mode; only plan a change to the preferred mode when the GPU is
    being reconfigured anyway and the current mode would be illegal."""
    legal = VALID_MEMORY_MODES[profile.compute]
    if current in legal:
        return current, False
    return (
        profile.preferred_memory if profile.preferred_memory in legal else legal[0]
    ), True


Score = Tuple  # lexicographic, higher wins; COMPARABLE ACROSS NODES

class AllocationPolicy:
    """Strategy interface (reference: instaslice_controller.go:48-50).

    `prefer_gpus` carries gang-placement affinity: GPUs already hosting the
    pod's group. Same-GPU XCD co-location is free bandwidth (intra-die,
    no xGMI hop — SURVEY.md §5), so preferred same-mode fits win over
    everything else in every policy.

    `place_scored` returns (score, placement) with scores comparable across
    nodes: the controller collects one candidate per node and takes the
    global argmax, so packing/spreading works cluster-wide, not just within
    the first node that fits (the reference is first-fit across nodes,
    findDeviceForASlice, instaslice_controller.go:240-262)."""

    name = "base"
    node_order_first_fit = False  # True: controller takes the first node hit

    def place_scored(self, profile: PartitionProfile, views: List[GpuView],
                     prefer_gpus: frozenset = frozenset(),
                     xgmi_neighbors: frozenset = frozenset()
                     ) -> Optional[Tuple[Score, Placement]]:
        raise NotImplementedError

    def place(self, profile: PartitionProfile, views: List[GpuView],
              prefer_gpus: frozenset = frozenset(),
              xgmi_neighbors: frozenset = frozenset()) -> Optional[Placement]:
        sp = self.place_scored(profile, views, prefer_gpus, xgmi_neighbors)
        return sp[1] if sp else None

    def _preferred(self, profile: PartitionProfile, views: List[GpuView],
                   prefer_gpus: frozenset,
                   xgmi_neighbors: frozenset = frozenset()
                   ) -> Optional[Tuple[Score, Placement]]:
        """Gang-affinity tiers, shared by every policy:
          9: free same-mode slot on a GPU already hosting the gang
             (intra-die XCD co-location — no xGMI hop at all);
          8: on a 1-hop xGMI NEIGHBOR of a gang GPU — same-mode slot
             (8,1,·) beating an idle-neighbor mode flip (8,0,·). For
             RCCL gangs, link locality dominates one flip's cost.
        Neighbor sets come from the CR's discovered topology
        (spec.topology via amdsmi_topo_get_link_type); empty set = no
        topology known = same-GPU affinity only (r1 behavior)."""
        for v in views:
            if v.uuid in prefer_gpus and v.compute_mode is profile.compute:
                p = _place_on(v, profile, needs_change=False)
                if p:
                    return ((9, 0, 0), p)
        best: Optional[Tuple[Score, Placement]] = None
        for v in views:
            if v.uuid not in xgmi_neighbors:
                continue
            if v.compute_mode is profile.compute:
                p = _place_on(v, profile, needs_change=False)
                if p:
                    score = (8, 1, len(v.occupied))
                    if best is None or score > best[0]:
                        best = (score, p)
            else:
                p = _place_on(v, profile, needs_change=True)
                if p:
                    score = (8, 0, -v.index)
                    if best is None or score > best[0]:
                        best = (score, p)
        return best


def _free_ordinal(view: GpuView, n_partitions: int) -> Optional[int]:
    for k in range(n_partitions):
        if k not in view.occupied:
            return k
    return None


def _upsize_place(view: GpuView, profile: PartitionProfile) -> Optional[Placement]:
    """Serve a request with a BIGGER partition than asked: a free ordinal on
    an occupied GPU whose mode's partitions have more XCDs than the profile
    wants (e.g. cpx-1x36 onto a QPX GPU's 2-XCD slot). MI355X capacity is
    fungible downward — the pod gets a strictly larger device and the env
    contract is unchanged — which turns wrong-mode holes (THE fragmentation
    axis on whole-GPU-mode hardware) into usable capacity
    instead of unschedulable stalls. Only on non-empty GPUs: idle GPUs are
    flip territory (flipping wastes nothing)."""
    if not view.occupied or view.mode_locked:
        return None
    mode_xcds = 8 // view.compute_mode.num_partitions
    if mode_xcds <= profile.xcds:
        return None  # same or smaller: not an upsize
    k = _free_ordinal(view, view.compute_mode.num_partitions)
    if k is None:
        return None
    return Placement(view.node, view.uuid, k, view.compute_mode.value,
                     view.memory_mode.value, False)


def _place_on(view: GpuView, profile: PartitionProfile, needs_change: bool) -> Optional[Placement]:
    n = profile.partitions_per_gpu
    if needs_change:
        if view.occupied or view.mode_locked:
            return None  # mode change needs an idle, flippable GPU
        mem, _ = _target_memory_mode(profile, view.memory_mode)
        return Placement(view.node, view.uuid, 0, profile.compute.value, mem.value, True)
    k = _free_ordinal(view, n)
    if k is None:
        return None
    return Placement(
        view.node, view.uuid, k, profile.compute.value, view.memory_mode.value, False
    )


class FirstFitPolicy(AllocationPolicy):
    """Reference-parity first-fit (FirstFitPolicy, instaslice_controller.go:436-453):
    scan GPUs in index order; take the first free ordinal on a GPU already in
    the right mode, else the first idle GPU (planning a mode change)."""

    name = "first-fit"
    node_order_first_fit = True

    def place_scored(self, profile: PartitionProfile, views: List[GpuView],
                     prefer_gpus: frozenset = frozenset(),
                     xgmi_neighbors: frozenset = frozenset()
                     ) -> Optional[Tuple[Score, Placement]]:
        if prefer_gpus or xgmi_neighbors:
            sp = self._preferred(profile, views, prefer_gpus, xgmi_neighbors)
            if sp:
                return sp
        for v in views:
            if v.compute_mode is profile.compute:
                p = _place_on(v, profile, needs_change=False)
                if p:
                    return ((2, 0, 0), p)
        for v in views:
            if v.compute_mode is not profile.compute:
                p = _place_on(v, profile, needs_change=True)
                if p:
                    return ((1, 0, 0), p)
        return None


class PackedFitPolicy(AllocationPolicy):
    """MI355X-aware scorer (the reference left this plug-point as empty stubs,
    instaslice_controller.go:456-469). Score terms, highest wins:

    1. GPU already in the target compute mode, most-occupied first: packing
       same-mode GPUs tight minimizes the number of GPUs pinned to a mode,
       keeping idle GPUs reconfigurable — fragmentation on AMD is *per-mode*,
       not per-slot.
    2. Among idle GPUs needing a mode change, prefer one whose memory mode
       already satisfies the profile (avoids the expensive NPS flip), then
       a node that already hosts allocations (consolidate flips onto
       already-busy nodes — without this term the cross-node argmax sends
       every flip to a fresh node's GPU 0), then lower physical index
       (keeps high-index GPUs free for large SPX jobs, and co-locates small
       partitions on few GPUs so xGMI links of the remaining GPUs stay
       uncontended for multi-GPU tenants).
    3. UPSIZE fallback (r2, the per-GPU-mode fragmentation fix): a free
       larger-XCD slot on an occupied GPU serves the request with a bigger
       partition (see _upsize_place), capped at 2 wasted XCDs — beyond
       that, stranding beats the saving (measured in the churn study,
       profiles/fragmentation_study_r02.md).
    4. Flip-vs-upsize ordering is PER CLASS: 1-XCD (CPX) requests absorb
       into existing holes first (waste 1, no new mode pin — flipping a
       whole GPU for them creates the straggler split that blocks 4-XCD
       requests later), while larger profiles flip an idle GPU first
       (their repacked arrangement genuinely needs dedicated GPUs;
       upsizing them strands capacity when their own mode later frees).
    5. MODE COARSENING when no big-profile (>=4-XCD) tenants are present:
       1-XCD requests then prefer riding 2-XCD QPX slots, and when they
       must flip an idle GPU they flip it to QPX — the node's small-pod
       capacity becomes ONE fungible slot class, so any freed slot serves
       any small request and unschedulable stalls collapse (mixed100 p99
       1276 -> ~520 ms, p90 267 -> 11 ms). Big-profile occupancy disables
       it because coarsening trades XCD efficiency for fungibility, which
       starves 4-XCD requests (churn study).
    6. Same-mode refills prefer the NEWEST (highest-index) GPU: oldest
       GPUs then drain monotonically under FIFO churn and become
       flippable instead of being refilled forever.

       Full measurements + clairvoyant-oracle floor:
       profiles/fragmentation_study_r02.md.
    """

    name = "packed-fit"
    # beyond this many wasted XCDs an upsize strands more capacity than the
    # saved flip is worth (churn study: cap 2 strictly beats unbounded)
    max_upsize_waste = 2

    def place_scored(self, profile: PartitionProfile, views: List[GpuView],
                     prefer_gpus: frozenset = frozenset(),
                     xgmi_neighbors: frozenset = frozenset()
                     ) -> Optional[Tuple[Score, Placement]]:
        if prefer_gpus or xgmi_neighbors:
            sp = self._preferred(profile, views, prefer_gpus, xgmi_neighbors)
            if sp:
                return sp
        best: Optional[Tuple[Tuple, Placement]] = None
        node_used = sum(len(v.occupied) for v in views)
        small = profile.xcds <= 1
        qpx_present = any(v.compute_mode is ComputeMode.QPX for v in views)
        big_demand = any(
            v.occupied and v.compute_mode in (ComputeMode.DPX, ComputeMode.SPX)
            for v in views
        )
        coarsen = small and qpx_present and not big_demand
        flip_tier, upsize_tier = (2, 3) if small else (3, 2)
        for v in views:
            if v.compute_mode is profile.compute:
                p = _place_on(v, profile, needs_change=False)
                if p is None:
                    continue
                score = (3 if coarsen else 4, v.index, len(v.occupied), 0)
            else:
                p = _place_on(v, profile, needs_change=True)
                if p is not None:
                    if coarsen:
                        # flip to QPX: small capacity joins the 2-XCD class
                        mem, _ = _target_memory_mode(profile, v.memory_mode)
                        p = Placement(v.node, v.uuid, 0,
                                      ComputeMode.QPX.value, mem.value, True)
                    mem_ok = v.memory_mode in VALID_MEMORY_MODES[profile.compute]
                    score = (flip_tier, 1 if mem_ok else 0, node_used, -v.index)
                else:
                    p = _upsize_place(v, profile)
                    if p is None:
                        continue
                    waste = 8 // v.compute_mode.num_partitions - profile.xcds
                    if waste > self.max_upsize_waste:
                        continue
                    score = (4 if coarsen else upsize_tier,
                             -waste, len(v.occupied), -v.index)
            if best is None or score > best[0]:
                best = (score, p)
        return best


class SpreadFitPolicy(AllocationPolicy):
    """Bandwidth/thermal-headroom spreading: prefer the *least*-occupied GPU
    already in the target mode, then idle GPUs. Co-tenants on one GPU share
    its HBM (8 TB/s) and power budget; spreading maximizes per-pod headroom
    at the cost of pinning more GPUs to a mode — the inverse trade of
    PackedFitPolicy, exposed so operators can choose per cluster."""

    name = "spread-fit"

    def place_scored(self, profile: PartitionProfile, views: List[GpuView],
                     prefer_gpus: frozenset = frozenset(),
                     xgmi_neighbors: frozenset = frozenset()
                     ) -> Optional[Tuple[Score, Placement]]:
        if prefer_gpus or xgmi_neighbors:
            sp = self._preferred(profile, views, prefer_gpus, xgmi_neighbors)
            if sp:
                return sp
        best: Optional[Tuple[Tuple, Placement]] = None
        for v in views:
            # occupancy is the primary key: an idle GPU (even one needing a
            # mode flip) beats a loaded same-mode GPU — the opposite of
            # packed-fit's tiering
            if v.compute_mode is profile.compute:
                p = _place_on(v, profile, needs_change=False)
                if p is None:
                    continue
                score = (-len(v.occupied), 1, -v.index)  # tie: prefer no flip
            else:
                p = _place_on(v, profile, needs_change=True)
                if p is None:
                    continue
                mem_ok = v.memory_mode in VALID_MEMORY_MODES[profile.compute]
                score = (0, 0, (1 if mem_ok else 0) - v.index)
            if best is None or score > best[0]:
                best = (score, p)
        return best


POLICIES: Dict[str, AllocationPolicy] = {
    p.name: p for p in (FirstFitPolicy(), PackedFitPolicy(), SpreadFitPolicy())
}


def get_policy(name: str) -> AllocationPolicy:
    if name not in POLICIES:
        raise KeyError(f"unknown policy {name!r}; have {sorted(POLICIES)}")
    return POLICIES[name]
