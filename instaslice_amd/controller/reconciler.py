"""L1 cluster controller: gated-pod admission, placement, ungating, teardown.

Reference analog: InstasliceReconciler (instaslice_controller.go:64-238) with
the call stack documented in SURVEY.md §3.1/§3.4. State machine (unchanged):

    pod gated, no allocation   -> place -> Allocations[podUUID] = creating
    allocation created (agent) -> remove scheduling gate -> status = ungated
    pod deleting (gated)       -> remove finalizer, status = deleted
    pod deleting (was running) -> after grace period: finalizer off, deleted

Differences by design (not omissions):
  - placement is mode-aware (controller/policy.py) instead of MIG-slot
    first-fit;
  - CR mutations are single-round-trip guarded PATCHes (CAS on
    resourceVersion for placement, test-ops elsewhere); a failed guard
    falls back to the classic get-revalidate-update cycle instead of the
    reference's 1 s requeue-on-conflict (instaslice_controller.go:93);
  - reconciles read the engine's informer cache (watch-fed) rather than
    GET/LISTing per reconcile;
  - errors from the placer/store fail the reconcile loudly (the engine
    backs off and retries) instead of being logged and dropped.
"""

from __future__ import annotations

import time
import zlib
from typing import List, Optional, Tuple

from instaslice_amd import FINALIZER_NAME, POD_RESOURCE_PREFIX
from instaslice_amd.api.events import emit
from instaslice_amd.api.types import (
    AllocationDetails,
    AllocationStatus,
    pod_is_gated,
    pod_limits,
    ungate_pod,
)
from instaslice_amd.controller.policy import AllocationPolicy, build_gpu_views, get_policy
from instaslice_amd.partition.profiles import (
    PartitionProfile,
    ProfileCatalog,
    extract_profile_from_limits,
)
from instaslice_amd.runtime.engine import Engine, Key, Result, WatchSpec
from instaslice_amd.store.memstore import MemStore, NotFound
from instaslice_amd.utils import get_logger

INSTASLICE_NS = "instaslice-system"

# Requeue cadences (reference: instaslice_controller.go:225 no-capacity 5 s,
# :231 no-node-fits 2 s). Configurable because they bound allocation latency
# under contention.
REQUEUE_NO_FIT_S = 0.5


class PodController:
    def __init__(
        self,
        store: MemStore,
        policy: str = "packed-fit",
        teardown_grace_s: float = 30.0,
        requeue_no_fit_s: float = REQUEUE_NO_FIT_S,
        node_stale_after_s: float = 30.0,
        workers: int = 2,
        shard_index: int = 0,
        shard_count: int = 1,
    ) -> None:
        self.store = store
        self.policy: AllocationPolicy = get_policy(policy)
        # horizontal sharding: controller instance i of k owns the pods whose
        # crc32(ns/name) % k == i. Python reconciles are GIL-bound, so the
        # scale-out axis is processes, not worker threads; shards race only
        # on CR writes, where add_alloc's fresh-object re-validation +
        # resourceVersion conflicts already arbitrate (same mechanism as
        # concurrent workers). The reference runs exactly one controller
        # (leader election, cmd/controller/main.go:107-108) — sharding is a
        # deliberate scale-out improvement over that.
        self.shard_index = shard_index
        self.shard_count = max(1, shard_count)
        self.teardown_grace_s = teardown_grace_s
        self.requeue_no_fit_s = requeue_no_fit_s
        # failure detection: nodes whose agent heartbeat is older than this
        # receive no new placements, and their stuck `creating` allocations
        # are re-placed elsewhere. 0 disables. (CRs without a heartbeat field
        # — externally crafted — are treated as healthy.)
        self.node_stale_after_s = node_stale_after_s
        self.log = get_logger("controller")
        # latency observability: pod_uid -> submit time, and measured p50
        # feed (bounded: long-running controllers must not grow unboundedly)
        from collections import deque

        self.alloc_latency_s: deque = deque(maxlen=65536)
        self._pending_since: dict = {}
        # pods currently marked unschedulable: re-reconciled on ANY node-state
        # change so freed capacity is picked up event-driven, not by polling
        self._unschedulable_keys: set = set()
        # demand memory for drain-time mode planning: profile -> last time a
        # pod requesting it went unschedulable (drives desiredMode hints)
        self._starved_profiles: dict = {}
        # (ns, name) -> last event-driven retry (herd limit; see
        # _instaslice_to_pods). Entries are pruned with _unschedulable_keys.
        self._unsched_last_retry: dict = {}
        # node -> (placements_dict, ProfileCatalog): see _profile_for
        self._catalog_cache: dict = {}
        # pod uid -> (victim_name, t): preemption cooldown (_maybe_preempt)
        self._preempted_for: dict = {}
        # workers > 1 is safe: the engine guarantees a key is never
        # reconciled concurrently (runtime/engine.py in-flight guard), and
        # placement races between different pods are caught by the
        # re-validation inside add_alloc + resourceVersion conflicts
        self.engine = Engine(
            name=f"controller-{shard_index}" if shard_count > 1 else "controller",
            store=store,
            reconcile=self._reconcile,
            watches=[
                WatchSpec(kind="Pod", map_fn=self._own_pods_only),
                WatchSpec(kind="Instaslice", map_fn=self._instaslice_to_pods),
            ],
            workers=workers,
        )

    # -- sharding ----------------------------------------------------------

    def _owns(self, namespace: str, name: str) -> bool:
        if self.shard_count <= 1:
            return True
        h = zlib.crc32(f"{namespace}/{name}".encode())
        return h % self.shard_count == self.shard_index

    def _own_pods_only(self, event_type: str, obj: dict) -> List[Key]:
        md = obj["metadata"]
        ns, name = md.get("namespace", ""), md["name"]
        if not self._owns(ns, name):
            return []
        return [("Pod", ns, name)]

    # -- watch mapping ------------------------------------------------------
    # Reference: podMapFunc maps an Instaslice with "created" allocations back
    # to its pods (instaslice_controller.go:398-407).

    # waiting-pod retry herd limit: re-reconciling EVERY unschedulable pod
    # on EVERY node event is O(waiting x event-rate) — with thousands
    # waiting that's a reconcile storm (docs/FUTURE.md item). Each waiting
    # pod gets at most one event-driven retry per window; correctness is
    # unaffected because placement is re-validated per retry and the
    # periodic requeue fallback still exists.
    UNSCHED_RETRY_WINDOW_S = 0.05

    def _instaslice_to_pods(self, event_type: str, obj: dict) -> List[Key]:
        keys: List[Key] = []
        for alloc in (obj.get("spec", {}).get("allocations") or {}).values():
            if alloc.get("allocationStatus") in (
                AllocationStatus.CREATED,
                AllocationStatus.FAILED,
            ) and self._owns(alloc["namespace"], alloc["podName"]):
                keys.append(("Pod", alloc["namespace"], alloc["podName"]))
        # freed/changed capacity: give waiting pods another placement pass
        now = time.monotonic()
        for (ns, name) in list(self._unschedulable_keys):
            last = self._unsched_last_retry.get((ns, name), 0.0)
            if now - last >= self.UNSCHED_RETRY_WINDOW_S:
                self._unsched_last_retry[(ns, name)] = now
                keys.append(("Pod", ns, name))
        return keys

    # -- helpers ------------------------------------------------------------

    def _find_allocation(
        self, pod_uid: str, crs: List[dict]
    ) -> Optional[Tuple[dict, dict]]:
        """Return (instaslice_cr, allocation_dict) holding this pod's
        allocation, or None. `crs` is the reconcile's single Instaslice list
        (one store round-trip per reconcile, shared with placement)."""
        for cr in crs:
            alloc = (cr.get("spec", {}).get("allocations") or {}).get(pod_uid)
            if alloc:
                return cr, alloc
        return None

    def _profile_for(self, cr: dict, name: str) -> Optional[PartitionProfile]:
        placements = cr.get("spec", {}).get("placements") or {}
        if placements:
            # catalog parse is ~100 us and placements are static per node:
            # cache per node, validated by dict equality (cheap vs re-parse)
            node = cr["metadata"]["name"]
            cached = self._catalog_cache.get(node)
            if cached is not None and cached[0] == placements:
                cat = cached[1]
            else:
                cat = ProfileCatalog.from_dict(placements)
                self._catalog_cache[node] = (placements, cat)
            p = cat.by_name(name)
            if p:
                return p
        return None

    def _node_stale(self, cr: dict) -> bool:
        if not self.node_stale_after_s:
            return False
        hb = (cr.get("status") or {}).get("heartbeat")
        return hb is not None and (time.time() - float(hb)) > self.node_stale_after_s

    def _try_place(self, pod: dict, profile_name: str, crs: List[dict]) -> Optional[Result]:
        """Scan nodes, place, persist the allocation (status=creating).
        Reference: findDeviceForASlice loop (instaslice_controller.go:192-222)
        — but where the reference takes the FIRST node that fits, scored
        policies here collect one candidate per node and commit the global
        argmax, so packing/spreading is cluster-wide. FirstFit keeps the
        reference's take-the-first semantics."""
        md = pod["metadata"]
        node_selector = pod.get("spec", {}).get("nodeSelector") or {}
        want_node = node_selector.get("kubernetes.io/hostname")
        group = (md.get("annotations") or {}).get("org.instaslice/group", "")
        candidates = []  # (score, cr, placement, profile)
        for cr in crs:
            node_name = cr["metadata"]["name"]
            if want_node and node_name != want_node:
                continue
            if self._node_stale(cr):
                continue  # agent dead: no new placements here
            if (cr.get("spec") or {}).get("cordoned"):
                continue  # operator drained this node (CLI cordon)
            profile = self._profile_for(cr, profile_name)
            if profile is None:
                continue  # node does not offer this profile
            views = build_gpu_views(node_name, cr.get("spec", {}),
                                    nominee_uid=md["uid"])
            # gang affinity: GPUs already hosting this pod's group on this
            # node (same-GPU XCD co-location = intra-die bandwidth, no xGMI)
            prefer = frozenset(
                a["gpuUUID"]
                for a in (cr["spec"].get("allocations") or {}).values()
                if group and a.get("group") == group
            )
            # hop-aware gang spillover: 1-hop xGMI neighbors of gang GPUs
            # (discovered topology in the CR; empty when unknown)
            neighbors: frozenset = frozenset()
            if prefer:
                topo = cr["spec"].get("topology") or {}
                near = set()
                for g_uuid in prefer:
                    peers = topo.get(g_uuid) or {}
                    if not peers:
                        continue
                    # nearest tier = minimum hop count this platform reports
                    # (absolute values vary by driver: 1 on the fake, 2 on
                    # some bare-metal xGMI reports)
                    mh = min(int(h) for h in peers.values())
                    near.update(d for d, h in peers.items()
                                if int(h) == mh and d not in prefer)
                neighbors = frozenset(near)
            sp = self.policy.place_scored(profile, views, prefer_gpus=prefer,
                                          xgmi_neighbors=neighbors)
            if sp is None:
                continue
            candidates.append((sp[0], cr, sp[1], profile))
            if self.policy.node_order_first_fit:
                break
        if not candidates:
            # preemption (opt-in, org.instaslice/priority): evict the
            # lowest-priority same-profile allocation with strictly lower
            # priority; the drain frees an exact-fit slot and the existing
            # unschedulable-retry machinery places this pod on the event.
            # The reference has no preemption at all; k8s-style semantics
            # (victim pod is DELETED, finalizer-grace applies).
            if self._maybe_preempt(pod, profile_name, crs):
                # register for event-driven retry: the victim's drain emits
                # CR events that re-reconcile unschedulable pods, landing
                # the preemptor on its nominated slot within milliseconds
                # (the timer below is only the dead-agent fallback)
                self._mark_unschedulable(md, profile_name, unschedulable=True)
                return Result(requeue_after=self.requeue_no_fit_s)
        # best score first; on a lost CAS race, fall through to the next node
        candidates.sort(key=lambda c: c[0], reverse=True)
        for _, cr, placement, profile in candidates:
            node_name = cr["metadata"]["name"]
            # actual partition geometry comes from the placement's mode, not
            # the requested profile: an UPSIZE placement (policy fallback)
            # hands the pod a larger partition than it asked for
            from instaslice_amd.partition.profiles import ComputeMode as _CM

            actual_xcds = 8 // _CM(placement.compute_mode).num_partitions
            alloc = AllocationDetails(
                profile=profile_name,
                gpu_uuid=placement.gpu_uuid,
                ordinal=placement.ordinal,
                start=placement.ordinal * actual_xcds,
                size=actual_xcds,
                pod_uuid=md["uid"],
                pod_name=md["name"],
                namespace=md.get("namespace", "default"),
                nodename=node_name,
                allocation_status=AllocationStatus.CREATING,
                compute_mode=placement.compute_mode,
                memory_mode=placement.memory_mode,
                group=group,
                priority=self._pod_priority(pod),
            )
            updated = self._write_allocation(cr, node_name, alloc)
            if updated and md["uid"] in updated["spec"]["allocations"]:
                self.log.debug(
                    "placed pod %s profile %s on %s/%s ordinal %d",
                    md["name"], profile_name, node_name,
                    placement.gpu_uuid[:8], placement.ordinal,
                )
                upsized = actual_xcds > profile.xcds
                if upsized:
                    from instaslice_amd.metrics import get_metrics

                    get_metrics().allocation("upsized")
                emit(self.store, pod, "Placed",
                     f"allocated {profile_name} on {node_name}/"
                     f"{placement.gpu_uuid[:8]}#{placement.ordinal}"
                     + (" (mode change planned)"
                        if placement.needs_mode_change else "")
                     + (f" (upsized to a {actual_xcds}-XCD "
                        f"{placement.compute_mode} slot)" if upsized else ""))
                self._mark_unschedulable(md, profile_name, unschedulable=False)
                # normally the agent's "created" event advances this pod; the
                # timed recheck only notices a dead agent (no events then)
                return Result(
                    requeue_after=self.node_stale_after_s or None
                )
        emit(self.store, pod, "Unschedulable",
             f"no node can host profile {profile_name} right now",
             type_="Warning")
        self._mark_unschedulable(md, profile_name, unschedulable=True)
        return Result(requeue_after=self.requeue_no_fit_s)

    def _mark_unschedulable(self, md: dict, profile_name: str,
                            unschedulable: bool) -> None:
        from instaslice_amd import UNSCHEDULABLE_ANNOTATION

        key = (md.get("namespace", "default"), md["name"])
        if unschedulable:
            self._unschedulable_keys.add(key)
            self._starved_profiles[profile_name] = time.time()
        else:
            self._unschedulable_keys.discard(key)
            self._unsched_last_retry.pop(key, None)
            # common case: the pod was never marked — skip the
            # read-modify-write entirely (one store GET per placement
            # otherwise; stale-view miss just leaves a cosmetic annotation
            # that the next event-driven pass clears)
            if UNSCHEDULABLE_ANNOTATION not in (md.get("annotations") or {}):
                return

        def mut(p: dict) -> Optional[dict]:
            ann = p["metadata"].setdefault("annotations", {})
            if unschedulable:
                if ann.get(UNSCHEDULABLE_ANNOTATION) == profile_name:
                    return None
                ann[UNSCHEDULABLE_ANNOTATION] = profile_name
            else:
                if UNSCHEDULABLE_ANNOTATION not in ann:
                    return None
                del ann[UNSCHEDULABLE_ANNOTATION]
            return p

        self.store.update_with_retry(
            "Pod", md["name"], md.get("namespace", "default"), mut
        )

    def _write_allocation(self, cr: dict, node_name: str, alloc) -> Optional[dict]:
        """Persist a placement decision. Fast path: one CAS patch — the
        decision was taken on `cr`, so guard on its resourceVersion; any
        concurrent CR write (another shard, an agent commit) fails the guard
        and we fall back to the classic get-revalidate-update cycle, which
        re-checks the ordinal/mode against the fresh object (reference's
        placement race handling done atomically rather than by requeue,
        instaslice_controller.go:93)."""
        from instaslice_amd.store.memstore import Conflict

        from instaslice_amd.controller.policy import NOMINATION_TTL_S

        a = alloc
        # landing a placement consumes this pod's nomination (if any) and
        # garbage-collects expired ones — valid at this CR revision, and the
        # CAS guard makes the cleanup race-free
        now = time.time()
        nom_clears = [
            {"op": "delete", "path": ["spec", "nominations", nuid]}
            for nuid, nom in (cr["spec"].get("nominations") or {}).items()
            if nuid == a.pod_uuid or now - float(nom.get("ts", 0)) > NOMINATION_TTL_S
        ]
        try:
            return self.store.patch("Instaslice", node_name, INSTASLICE_NS, [
                {"op": "test", "path": ["metadata", "resourceVersion"],
                 "value": cr["metadata"]["resourceVersion"]},
                {"op": "test", "path": ["spec", "allocations", a.pod_uuid],
                 "absent": True},
                {"op": "set", "path": ["spec", "allocations", a.pod_uuid],
                 "value": a.to_dict()},
                *nom_clears,
            ])
        except NotFound:
            return None
        except Conflict:
            pass

        def add_alloc(cr_obj: dict) -> Optional[dict]:
            spec = cr_obj.setdefault("spec", {})
            allocs = spec.setdefault("allocations", {})
            if a.pod_uuid in allocs:
                return None  # raced with ourselves; done
            spec.get("nominations", {}).pop(a.pod_uuid, None)
            # re-validate the placement against the fresh CR (another
            # pod may have taken the ordinal between list() and now);
            # nominee view so our own reserved slot doesn't read occupied
            fresh_views = build_gpu_views(cr_obj["metadata"]["name"], spec,
                                          nominee_uid=a.pod_uuid)
            for v in fresh_views:
                if v.uuid == a.gpu_uuid:
                    if a.ordinal in v.occupied:
                        return None  # lost the race; requeue will re-place
                    if (
                        v.compute_mode.value != a.compute_mode
                        and v.occupied
                    ):
                        return None
            allocs[a.pod_uuid] = a.to_dict()
            return cr_obj

        return self.store.update_with_retry(
            "Instaslice", node_name, INSTASLICE_NS, add_alloc
        )

    def _gang_ready(self, pod: dict, alloc: dict) -> bool:
        from instaslice_amd import GROUP_SIZE_ANNOTATION

        ann = pod["metadata"].get("annotations") or {}
        group = alloc.get("group") or ""
        try:
            want = int(ann.get(GROUP_SIZE_ANNOTATION, 0))
        except (TypeError, ValueError):
            want = 0
        if not group or want <= 1:
            return True
        have = sum(
            1
            for cr in self._crs()
            for a in (cr["spec"].get("allocations") or {}).values()
            if a.get("group") == group
            and a.get("allocationStatus") in (AllocationStatus.CREATED,
                                              AllocationStatus.UNGATED)
        )
        if have < want:
            emit(self.store, pod, "GangWaiting",
                 f"partition ready; waiting for gang '{group}' "
                 f"({have}/{want} realized)")
            return False
        return True

    @staticmethod
    def _pod_priority(pod: dict) -> int:
        from instaslice_amd import PRIORITY_ANNOTATION

        try:
            return int((pod["metadata"].get("annotations") or {})
                       .get(PRIORITY_ANNOTATION, 0))
        except (TypeError, ValueError):
            return 0

    def _maybe_preempt(self, pod: dict, profile_name: str,
                       crs: List[dict]) -> bool:
        """Preemption planner, cheapest plan first:
        1. ONE lowest-priority same-profile victim (exact slot fit);
        2. else a WHOLE GPU whose every allocation is strictly lower
           priority — all of them are evicted and the GPU is nominated
           (mode-flip placement once it drains), which lets e.g. an SPX
           job displace a handful of low-priority CPX pods.
        Returns True if evictions were issued. A per-pod cooldown prevents
        cascading evictions while victims are still draining."""
        prio = self._pod_priority(pod)
        if prio <= 0:
            return False
        uid = pod["metadata"]["uid"]
        last = self._preempted_for.get(uid)
        if last is not None and time.monotonic() - last[1] < 10.0:
            # victim still draining (its DELETED event will retrigger us)
            return False
        victim = None  # (priority, cr_name, alloc_dict)
        for cr in crs:
            # skip nodes where the preemptor could never land (same filters
            # as _try_place): evicting victims on a stale/cordoned node
            # frees nothing the pod can use — it just destroys workloads
            if self._node_stale(cr) or (cr.get("spec") or {}).get("cordoned"):
                continue
            for a in (cr["spec"].get("allocations") or {}).values():
                if a.get("profile") != profile_name:
                    continue
                if a.get("allocationStatus") == AllocationStatus.DELETED:
                    return False  # a slot is already draining; just wait
                vprio = int(a.get("priority", 0))
                if vprio >= prio:
                    continue
                if victim is None or vprio < victim[0]:
                    victim = (vprio, cr["metadata"]["name"], a)
        if victim is None:
            return self._preempt_whole_gpu(pod, profile_name, prio, uid, crs)
        vprio, v_node, v_alloc = victim
        try:
            self.store.delete("Pod", v_alloc["podName"], v_alloc["namespace"])
        except NotFound:
            return False
        # reserve the freed slot for THIS pod (k8s nominatedNodeName analog):
        # without it, any waiting pod can steal the slot when it drains and
        # the preemptor evicts again — a cascade (caught by the e2e test)
        try:
            self.store.patch("Instaslice", v_node, INSTASLICE_NS, [
                {"op": "set", "path": ["spec", "nominations", uid],
                 "value": {"gpuUUID": v_alloc["gpuUUID"],
                           "ordinal": v_alloc["ordinal"],
                           "ts": time.time()}},
            ], quiet=True)
        except NotFound:
            pass
        emit(self.store, pod, "Preempting",
             f"evicting {v_alloc['namespace']}/{v_alloc['podName']} "
             f"(priority {vprio}) and nominating its slot", type_="Warning")
        emit(self.store,
             {"kind": "Pod", "namespace": v_alloc["namespace"],
              "name": v_alloc["podName"]},
             "Preempted",
             f"evicted for higher-priority pod {pod['metadata']['name']} "
             f"(priority {prio})", type_="Warning")
        self._preempted_for[uid] = (v_alloc["podName"], time.monotonic())
        self.log.warning(
            "preempting pod %s/%s (priority %d) for %s (priority %d); slot "
            "%s#%d nominated",
            v_alloc["namespace"], v_alloc["podName"], vprio,
            pod["metadata"]["name"], prio,
            v_alloc["gpuUUID"][:8], v_alloc["ordinal"],
        )
        from instaslice_amd.metrics import get_metrics

        get_metrics().allocation("preempted")
        return True

    def _preempt_whole_gpu(self, pod: dict, profile_name: str, prio: int,
                           uid: str, crs: List[dict]) -> bool:
        """Plan 2: evict every allocation on the GPU whose max victim
        priority is lowest (fewest victims as tie-break) and nominate the
        WHOLE GPU for this pod — once drained it is idle and flippable to
        the requested profile's mode."""
        best = None  # ((max_vprio, n), cr_name, gpu_uuid, victims)
        for cr in crs:
            if self._node_stale(cr) or (cr.get("spec") or {}).get("cordoned"):
                continue
            profile = self._profile_for(cr, profile_name)
            if profile is None:
                continue
            spec = cr["spec"]
            by_gpu: dict = {}
            for a in (spec.get("allocations") or {}).values():
                by_gpu.setdefault(a["gpuUUID"], []).append(a)
            for g_uuid, gd in (spec.get("gpus") or {}).items():
                victims = by_gpu.get(g_uuid)
                if not victims:
                    continue  # idle GPUs are normal-placement territory
                if gd.get("modeLocked") and gd.get(
                    "computeMode"
                ) != profile.compute.value:
                    continue  # cannot flip to the needed mode
                if any(a.get("allocationStatus") == AllocationStatus.DELETED
                       for a in victims):
                    return False  # capacity already draining; just wait
                vmax = max(int(a.get("priority", 0)) for a in victims)
                if vmax >= prio:
                    continue
                key = (vmax, len(victims))
                if best is None or key < best[0]:
                    best = (key, cr["metadata"]["name"], g_uuid, victims)
        if best is None:
            return False
        (vmax, n), cr_name, gpu_uuid, victims = best
        for a in victims:
            try:
                self.store.delete("Pod", a["podName"], a["namespace"])
            except NotFound:
                pass
            emit(self.store,
                 {"kind": "Pod", "namespace": a["namespace"],
                  "name": a["podName"]},
                 "Preempted",
                 f"evicted (whole-GPU preemption) for higher-priority pod "
                 f"{pod['metadata']['name']} (priority {prio})",
                 type_="Warning")
        try:
            self.store.patch("Instaslice", cr_name, INSTASLICE_NS, [
                {"op": "set", "path": ["spec", "nominations", uid],
                 "value": {"gpuUUID": gpu_uuid, "wholeGpu": True,
                           "ts": time.time()}},
            ], quiet=True)
        except NotFound:
            pass
        emit(self.store, pod, "Preempting",
             f"evicting {n} pod(s) (max priority {vmax}) from GPU "
             f"{gpu_uuid[:8]} and nominating the whole GPU", type_="Warning")
        self._preempted_for[uid] = (gpu_uuid, time.monotonic())
        self.log.warning(
            "whole-GPU preemption: %d pods off %s for %s (priority %d)",
            n, gpu_uuid[:8], pod["metadata"]["name"], prio,
        )
        from instaslice_amd.metrics import get_metrics

        get_metrics().allocation("preempted")
        return True

    STARVED_TTL_S = 30.0

    def _maybe_hint_mode(self, cr: dict, uid: str, alloc: dict) -> None:
        """DRAIN-TIME MODE PLANNING: when a teardown
        leaves a GPU empty while some profile recently went unschedulable,
        hint the agent to pre-flip the GPU to the starved mode
        (spec.gpus[uuid].desiredMode). On real hardware a compute-partition
        set costs O(100ms-1s); pre-flipping during idle time hides that
        from the next request's allocation latency. The agent only acts on
        the hint while the GPU is actually idle, so a racing placement
        simply wins and clears it."""
        if not self._starved_profiles:
            return
        now = time.time()
        for p, t in list(self._starved_profiles.items()):
            if now - t > self.STARVED_TTL_S:
                del self._starved_profiles[p]
        if not self._starved_profiles:
            return
        gpu_uuid = alloc["gpuUUID"]
        spec = cr["spec"]
        if any(isinstance(a, dict) and a.get("gpuUUID") == gpu_uuid
               for k, a in (spec.get("allocations") or {}).items() if k != uid):
            return  # GPU not drained empty
        gd = (spec.get("gpus") or {}).get(gpu_uuid) or {}
        if gd.get("modeLocked"):
            return
        prof_name = max(self._starved_profiles,
                        key=self._starved_profiles.get)
        profile = self._profile_for(cr, prof_name)
        if profile is None or profile.compute.value == gd.get("computeMode"):
            return
        from instaslice_amd.store.memstore import Conflict

        try:
            self.store.patch("Instaslice", cr["metadata"]["name"],
                             INSTASLICE_NS, [
                {"op": "set",
                 "path": ["spec", "gpus", gpu_uuid, "desiredMode"],
                 "value": profile.compute.value},
            ], quiet=True)
        except (Conflict, NotFound):
            pass

    def _set_allocation_status(self, cr_name: str, pod_uid: str, status: str,
                               expect: Optional[str] = None) -> None:
        from instaslice_amd.store.memstore import Conflict

        ops = []
        if expect is not None:
            ops.append({"op": "test",
                        "path": ["spec", "allocations", pod_uid, "allocationStatus"],
                        "value": expect})
        else:
            # require the allocation to exist (its podUUID key is itself)
            ops.append({"op": "test",
                        "path": ["spec", "allocations", pod_uid, "podUUID"],
                        "value": pod_uid})
        ops.append({"op": "set",
                    "path": ["spec", "allocations", pod_uid, "allocationStatus"],
                    "value": status})
        try:
            self.store.patch("Instaslice", cr_name, INSTASLICE_NS, ops)
        except (Conflict, NotFound):
            pass  # allocation gone or moved on; events re-reconcile us

    # -- reconcile ------------------------------------------------------------

    def _crs(self) -> List[dict]:
        """Current Instaslice CRs from the informer cache (kind-wide watch
        keeps it complete after replay; zero store round-trips, zero
        per-reconcile snapshot cost). Placement decisions taken on a cached
        CR are re-validated against the FRESH object inside add_alloc's
        update cycle, so staleness cannot double-book an ordinal."""
        crs = self.engine.cached_list("Instaslice")
        if crs:
            return crs
        return self.store.list("Instaslice")  # pre-replay startup window

    def _reconcile(self, key: Key) -> Result:
        _, namespace, name = key
        pod = self.engine.cached(("Pod", namespace, name))
        if pod is None:
            try:
                pod = self.store.get("Pod", name, namespace)
            except NotFound:
                self._unschedulable_keys.discard((namespace, name))
                self._unsched_last_retry.pop((namespace, name), None)
                return Result()
        md = pod["metadata"]
        uid = md.get("uid")
        if not uid:
            # malformed object (externally crafted): ignoring beats an
            # error-backoff retry loop — nothing we can key an allocation on
            self.log.warning("pod %s/%s has no uid; ignoring", namespace, name)
            return Result()

        # teardown path (reference: instaslice_controller.go:99-142)
        if md.get("deletionTimestamp"):
            self._unschedulable_keys.discard((namespace, name))
            self._unsched_last_retry.pop((namespace, name), None)
            self._pending_since.pop(uid, None)  # deleted before ungating
            self._preempted_for.pop(uid, None)  # bounded cooldown map
            gated = pod_is_gated(pod)
            if not gated:
                from instaslice_amd.api.types import timestamp_epoch

                elapsed = time.time() - timestamp_epoch(md["deletionTimestamp"])
                if elapsed < self.teardown_grace_s:
                    return Result(requeue_after=self.teardown_grace_s - elapsed)
            # ONE batched round-trip: full cleanup (or mark-deleted for
            # agent-managed teardown) AND strip our finalizer — the store
            # drops the pod once no finalizers remain
            reqs = []
            found = self._find_allocation(uid, self._crs())
            if found is None and FINALIZER_NAME in (md.get("finalizers") or []):
                # The informer cache can lag the allocation we ourselves
                # wrote (list+watch stores deliver events asynchronously;
                # K8sStore tier caught this): stripping the finalizer on a
                # stale miss orphans the allocation forever — the pod is
                # gone and nothing re-reconciles it. Confirm against the
                # source of truth before concluding there is no cleanup.
                found = self._find_allocation(uid, self.store.list("Instaslice"))
            if found:
                cr, alloc = found
                if not (cr["spec"].get("agentManagedTeardown")):
                    # FAST TEARDOWN: the agent's drain work is pure store
                    # bookkeeping unless it must reset the GPU mode (it
                    # advertises that via spec.agentManagedTeardown), so do
                    # the whole cleanup here in ONE batch — ConfigMap,
                    # capacity pin, allocation/prepared/ordinal — instead of
                    # the two-phase deleted-status handoff (saves one
                    # cross-process event hop + one round-trip per drain)
                    node = cr["metadata"]["name"]
                    reqs.append(
                        {"verb": "delete", "kind": "ConfigMap",
                         "name": alloc["podName"],
                         "namespace": alloc["namespace"]})
                    reqs.append(
                        {"verb": "patch", "kind": "Node", "name": node,
                         "namespace": "", "ops": [
                             {"op": "delete",
                              "path": ["status", "capacity",
                                       POD_RESOURCE_PREFIX + alloc["podName"]]},
                         ]})
                    reqs.append(
                        {"verb": "patch", "kind": "Instaslice", "name": node,
                         "namespace": INSTASLICE_NS, "ops": [
                             {"op": "test",
                              "path": ["spec", "allocations", uid, "podUUID"],
                              "value": uid},
                             {"op": "delete",
                              "path": ["spec", "allocations", uid]},
                             {"op": "remove_from_set",
                              "path": ["spec", "gpus", alloc["gpuUUID"],
                                       "usedOrdinals"],
                              "value": alloc["ordinal"]},
                             # predicate delete on the FRESH object: a
                             # key list computed from the cached CR can
                             # miss a prepared entry committed between
                             # our view and this patch (TOCTOU orphan
                             # caught by the K8sStore behavioral tier)
                             {"op": "delete_where",
                              "path": ["spec", "prepared"],
                              "field": "podUUID", "value": uid},
                         ]})
                elif alloc["allocationStatus"] != AllocationStatus.DELETED:
                    reqs.append(
                        {"verb": "patch", "kind": "Instaslice",
                         "name": cr["metadata"]["name"],
                         "namespace": INSTASLICE_NS, "ops": [
                             {"op": "test",
                              "path": ["spec", "allocations", uid, "podUUID"],
                              "value": uid},
                             {"op": "set",
                              "path": ["spec", "allocations", uid,
                                       "allocationStatus"],
                              "value": AllocationStatus.DELETED},
                         ]})
            fast_cleanup_idx = next(
                (i for i, r in enumerate(reqs)
                 if r.get("kind") == "Instaslice"
                 and any(op["op"] == "delete" and op["path"][:2] ==
                         ["spec", "allocations"] for op in r["ops"])),
                None,
            )
            if FINALIZER_NAME in (md.get("finalizers") or []):
                # guard on our view: an unconditional strip would bump the
                # rv (and emit MODIFIED) even when already stripped —
                # an event loop for pods carrying foreign finalizers
                reqs.append(
                    {"verb": "patch", "kind": "Pod", "name": name,
                     "namespace": namespace, "ops": [
                         {"op": "remove_from_set",
                          "path": ["metadata", "finalizers"],
                          "value": FINALIZER_NAME},
                     ]})
            if reqs:
                res = self.store.batch(reqs, quiet=True)
                if fast_cleanup_idx is not None and res[fast_cleanup_idx]["ok"]:
                    from instaslice_amd.metrics import get_metrics

                    get_metrics().allocation("deleted")
                    emit(self.store, pod, "PartitionReleased",
                         f"partition ordinal {alloc['ordinal']} on "
                         f"{alloc['gpuUUID'][:8]} released")
                    self._maybe_hint_mode(cr, uid, alloc)
            return Result()

        if not pod_is_gated(pod):
            # allocation-latency bookkeeping also lands here when the AGENT
            # fast-ungated the pod (data-plane admission completion) — the
            # ungate event reaches us via the pod watch either way
            self._preempted_for.pop(uid, None)  # placed; cooldown done
            t0 = self._pending_since.pop(uid, None)
            if t0 is not None:
                dt = time.monotonic() - t0
                self.alloc_latency_s.append(dt)
                from instaslice_amd.metrics import get_metrics

                get_metrics().allocation_latency(dt)
            return Result()

        # admission path
        crs = self._crs()
        found = self._find_allocation(uid, crs)
        if found is None:
            limits = pod_limits(pod)
            profile_name = extract_profile_from_limits(limits)
            if profile_name is None:
                return Result()  # not our pod
            if uid not in self._pending_since:
                self._pending_since[uid] = time.monotonic()
            return self._try_place(pod, profile_name, crs)

        cr, alloc = found
        status = alloc["allocationStatus"]
        if status == AllocationStatus.CREATING and self._node_stale(cr):
            # agent died before realizing it: reclaim and re-place elsewhere
            # (running pods can't migrate — their partition is on that node —
            # but pending ones should not wait for a dead agent)
            def reclaim(cr_obj: dict) -> Optional[dict]:
                allocs = cr_obj.get("spec", {}).get("allocations") or {}
                a = allocs.get(uid)
                if not a or a["allocationStatus"] != AllocationStatus.CREATING:
                    return None
                del allocs[uid]
                return cr_obj

            self.store.update_with_retry(
                "Instaslice", cr["metadata"]["name"], INSTASLICE_NS, reclaim
            )
            self.log.warning(
                "node %s stale; re-placing pending pod %s",
                cr["metadata"]["name"], name,
            )
            return Result(requeue_after=0.01)
        if status == AllocationStatus.CREATED:
            # gang ungate barrier (org.instaslice/group + group-size):
            # members keep their realized partitions but only ungate once
            # the WHOLE gang is realized — the RCCL-workload contract (all
            # ranks must start together; SURVEY.md §5 xGMI co-placement).
            # Event-driven: every member with a `created` allocation is
            # re-reconciled on each CR change, so the barrier lifts on the
            # event that realizes the last member.
            if not self._gang_ready(pod, alloc):
                return Result()
            # agent realized the partition: let the pod schedule. One batched
            # round-trip: ungate the pod AND flip the allocation to ungated.
            from instaslice_amd import GATE_NAME

            res = self.store.batch([
                {"verb": "patch", "kind": "Pod", "name": name,
                 "namespace": namespace, "ops": [
                     {"op": "test", "path": ["spec", "schedulingGates"],
                      "value": [{"name": GATE_NAME}]},
                     {"op": "set", "path": ["spec", "schedulingGates"],
                      "value": []},
                     # gate removal unblocks kube-scheduler; reflect in status
                     {"op": "set", "path": ["status", "phase"],
                      "value": "Pending"},
                     {"op": "set", "path": ["status", "conditions"],
                      "value": [{"type": "PodScheduled", "status": "True",
                                 "message": "ungated"}]},
                 ]},
                {"verb": "patch", "kind": "Instaslice",
                 "name": cr["metadata"]["name"], "namespace": INSTASLICE_NS,
                 "ops": [
                     {"op": "test",
                      "path": ["spec", "allocations", uid, "allocationStatus"],
                      "value": AllocationStatus.CREATED},
                     {"op": "set",
                      "path": ["spec", "allocations", uid, "allocationStatus"],
                      "value": AllocationStatus.UNGATED},
                 ]},
            ])
            if not res[0]["ok"] and res[0]["error"]["type"] == "Conflict":
                # gate list isn't exactly ours (extra gates / already ungated):
                # fall back to the precise read-modify-write
                def do_ungate(p: dict) -> Optional[dict]:
                    new = ungate_pod(p)
                    if new["spec"]["schedulingGates"] == p["spec"].get(
                        "schedulingGates"
                    ):
                        return None
                    new["status"]["phase"] = "Pending"
                    new["status"]["conditions"] = [
                        {"type": "PodScheduled", "status": "True",
                         "message": "ungated"}
                    ]
                    return new

                self.store.update_with_retry("Pod", name, namespace, do_ungate)
            t0 = self._pending_since.pop(uid, None)
            if t0 is not None:
                dt = time.monotonic() - t0
                self.alloc_latency_s.append(dt)
                from instaslice_amd.metrics import get_metrics

                get_metrics().allocation_latency(dt)
            return Result()
        if status == AllocationStatus.FAILED:
            # daemonset could not realize it: drop the allocation and
            # re-place on the next pass (the CR now carries modeLocked hints
            # so the placer avoids the failing transition)
            def drop(cr_obj: dict) -> Optional[dict]:
                allocs = cr_obj.get("spec", {}).get("allocations") or {}
                a = allocs.get(uid)
                if not a or a["allocationStatus"] != AllocationStatus.FAILED:
                    return None
                del allocs[uid]
                return cr_obj

            self.store.update_with_retry(
                "Instaslice", cr["metadata"]["name"], INSTASLICE_NS, drop
            )
            self.log.warning(
                "allocation for pod %s failed on %s; re-placing",
                name, alloc["gpuUUID"][:8],
            )
            return Result(requeue_after=0.01)
        if status == AllocationStatus.CREATING and self.node_stale_after_s:
            # agent normally advances this via events; the periodic recheck
            # only exists to notice a dead agent (no events ever arrive then)
            return Result(requeue_after=self.node_stale_after_s)
        # ungated / deleted: nothing to do here
        return Result()

    # -- lifecycle ------------------------------------------------------------

    def start(self) -> "PodController":
        self.engine.start()
        return self

    def stop(self) -> None:
        self.engine.stop()
