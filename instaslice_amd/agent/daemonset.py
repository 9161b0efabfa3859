"""L2 node agent: realizes partitions on the GPUs via the amd-smi layer.

Reference analog: InstaSliceDaemonsetReconciler
(instaslice_daemonset.go:95-275) with the call stacks of SURVEY.md §3.2/§3.3.
Responsibilities:

  boot      discover GPUs + profile catalog ONCE (cached — fixes the
            reference's per-reconcile nvml.Init hot spot, :112), publish the
            node's Instaslice CR, adopt pre-existing ("dangling") partitions
  creating  ensure the GPU is in the allocation's compute/memory mode
            (whole-GPU set; idle-checked), resolve the partition device at the
            allocated ordinal, publish the pod's visible-devices ConfigMap,
            patch node capacity, record Prepared, flip status -> created
  deleted   drop ConfigMap + capacity + Prepared, release the ordinal,
            remove the allocation entry

Every reconfigure captures amd-smi metrics before/after plus the mode-set
wall time (the north-star observability requirement; the reference registers
no custom metrics).
"""

from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional

from instaslice_amd import GATE_NAME, POD_RESOURCE_PREFIX, RESOURCE_PREFIX
from instaslice_amd.api.events import emit
from instaslice_amd.api.types import (
    AllocationStatus,
    GpuStatus,
    PreparedDetails,
    new_instaslice,
)
from instaslice_amd.partition.profiles import catalog_from_amdsmi_profiles
from instaslice_amd.runtime.engine import Engine, Key, Result, WatchSpec
from instaslice_amd.smi.base import (
    AmdSmi,
    PhysicalGpu,
    SmiBusy,
    SmiError,
    SmiNotSupported,
    SmiPermission,
)
from instaslice_amd.store.memstore import (AlreadyExists, Conflict,
                                            MemStore, NotFound)
from instaslice_amd.utils import get_logger

INSTASLICE_NS = "instaslice-system"
REQUEUE_BUSY_S = 0.5


class NodeAgent:
    def __init__(
        self,
        store: MemStore,
        smi: AmdSmi,
        node_name: str,
        reset_mode_on_empty: bool = False,
        heartbeat_every_s: float = 5.0,
        fast_ungate: bool = True,
    ) -> None:
        self.store = store
        self.smi = smi
        self.node_name = node_name
        # The reference destroys each slice on teardown; AMD mode flips are
        # whole-GPU and comparatively expensive, so by default the mode is
        # sticky when a GPU drains (the next same-profile pod lands with zero
        # reconfiguration). Set True for reference-parity teardown.
        self.reset_mode_on_empty = reset_mode_on_empty
        # data-plane admission completion (see _commit_prepare); turn off
        # for strict reference-parity flow (controller ungates everything)
        self.fast_ungate = fast_ungate
        self.log = get_logger(f"agent.{node_name}")
        # cached enumeration: gpu uuid -> PhysicalGpu (refreshed only after a
        # mode change of that GPU)
        self._gpus: Dict[str, PhysicalGpu] = {}
        # observability feed (metrics module scrapes this; bounded so a
        # long-running agent cannot grow without limit)
        from collections import deque

        self.reconfigure_events: deque = deque(maxlen=4096)
        # consecutive hard mode-set failures per GPU (3 strikes -> fail alloc)
        self._mode_set_failures: Dict[str, int] = {}
        self.prepare_failures = 0  # observability counter
        # failure-detection: periodic heartbeat into the CR status so the
        # controller stops placing onto nodes whose agent died (the reference
        # has no liveness signal at all)
        self.heartbeat_every_s = heartbeat_every_s
        self._hb_stop = threading.Event()
        self._hb_thread = None
        self._last_capacity: Dict[str, int] = {}
        # filtered watch: only THIS node's CR crosses the wire (a cluster
        # with N agents otherwise broadcasts every CR event to all N)
        self.engine = Engine(
            name=f"agent-{node_name}",
            store=store,
            reconcile=self._reconcile,
            watches=[WatchSpec(
                kind="Instaslice",
                map_fn=self._own_node_only,
                filters=[{"kind": "Instaslice", "name": node_name}],
            )],
        )

    def _own_node_only(self, event_type: str, obj: dict) -> List[Key]:
        if obj["metadata"]["name"] != self.node_name:
            return []
        return [("Instaslice", obj["metadata"].get("namespace", ""), self.node_name)]

    # -- discovery (boot, once) --------------------------------------------
    # Reference: discoverMigEnabledGpuWithSlices + discoverAvailableProfilesOnGpus
    # + discoverDanglingSlices (instaslice_daemonset.go:555-748).

    def discover(self) -> dict:
        self.smi.init()
        gpus = self.smi.list_gpus()
        self._gpus = {g.uuid: g for g in gpus}
        if not gpus:
            raise SmiError(f"node {self.node_name}: no AMD GPUs found")
        first = gpus[0]
        try:
            raw_profiles = self.smi.get_profile_config(first.uuid)
        except (SmiNotSupported, SmiError) as e:
            self.log.warning("profile discovery unsupported (%s); using MI355X model", e)
            raw_profiles = []
        catalog = catalog_from_amdsmi_profiles(
            first.model, first.memory_gb, raw_profiles
        )

        cr = new_instaslice(self.node_name)
        spec = cr["spec"]
        for g in gpus:
            spec["gpuUuids"][g.uuid] = g.model
            spec["gpus"][g.uuid] = {
                **GpuStatus(
                    uuid=g.uuid,
                    model=g.model,
                    memory_gb=g.memory_gb,
                    compute_mode=g.compute_mode,
                    memory_mode=g.memory_mode,
                ).to_dict(),
                "index": g.index,
            }
        spec["placements"] = catalog.to_dict()
        # xGMI link topology into the CR so the controller can score gang
        # placements hop-aware without talking to the device layer
        # (amdsmi_topo_get_link_type; empty dict = topology unknown)
        try:
            spec["topology"] = self.smi.get_topology()
        except SmiError as e:
            self.log.warning("topology discovery failed (%s); scoring "
                             "degrades to same-GPU affinity", e)
            spec["topology"] = {}
        # teardown ownership: when the agent must touch the device on drain
        # (mode reset), the controller leaves teardown to the agent's
        # deleted-status protocol; otherwise the controller fast-paths the
        # whole cleanup in its own batch (reconciler._reconcile teardown)
        spec["agentManagedTeardown"] = bool(self.reset_mode_on_empty)
        cr["status"]["processed"] = "true"

        try:
            existing = self.store.get("Instaslice", self.node_name, INSTASLICE_NS)
        except NotFound:
            existing = None

        if existing is None:
            try:
                self.store.create(cr)
            except AlreadyExists:
                existing = self.store.get("Instaslice", self.node_name, INSTASLICE_NS)
        if existing is not None:
            # agent restart: adopt durable state (allocations/prepared survive
            # in the CR — reference skips re-discovery when Processed=="true",
            # instaslice_daemonset.go:528-534 — but we still refresh live modes)
            def refresh(obj: dict) -> Optional[dict]:
                obj["spec"]["gpuUuids"] = spec["gpuUuids"]
                obj["spec"]["placements"] = spec["placements"]
                obj["spec"]["agentManagedTeardown"] = spec["agentManagedTeardown"]
                obj["spec"]["topology"] = spec["topology"]
                for uuid, gd in spec["gpus"].items():
                    old = obj["spec"].setdefault("gpus", {}).get(uuid)
                    if old:
                        gd = dict(gd, usedOrdinals=old.get("usedOrdinals", []))
                    obj["spec"]["gpus"][uuid] = gd
                obj["status"]["processed"] = "true"
                return obj

            self.store.update_with_retry("Instaslice", self.node_name, INSTASLICE_NS, refresh)

        # publish the Node object (capacity patches land on it)
        node = {
            "apiVersion": "v1",
            "kind": "Node",
            "metadata": {"name": self.node_name, "namespace": "", "labels": {}},
            "status": {"capacity": {}},
        }
        try:
            self.store.create(node)
        except AlreadyExists:
            pass
        return self.store.get("Instaslice", self.node_name, INSTASLICE_NS)

    # -- mode management ----------------------------------------------------

    def _ensure_gpu_mode(self, cr: dict, gpu_uuid: str, compute: str, memory: str) -> None:
        """Whole-GPU mode set with metrics capture + cached re-enumeration.
        Raises SmiBusy if the device refuses (caller requeues)."""
        g = self._gpus.get(gpu_uuid)
        if g is None:
            raise SmiError(f"unknown gpu {gpu_uuid}")
        changed = False
        event = {
            "gpu": gpu_uuid,
            "node": self.node_name,
            "from": f"{g.compute_mode}/{g.memory_mode}",
            "to": f"{compute}/{memory}",
        }
        try:
            event["metrics_before"] = self.smi.get_metrics(gpu_uuid)
        except SmiError:
            event["metrics_before"] = {}
        t0 = time.monotonic()
        if g.compute_mode != compute:
            self.smi.set_compute_partition(gpu_uuid, compute)
            changed = True
        if g.memory_mode != memory:
            try:
                self.smi.set_memory_partition(gpu_uuid, memory)
                changed = True
            except SmiNotSupported:
                # NPS flip may be impossible (VM guest / needs driver reload):
                # capacity math is unaffected, only locality — log and go on.
                self.log.warning(
                    "gpu %s: memory mode %s unsupported, staying on %s",
                    gpu_uuid[:8], memory, g.memory_mode,
                )
        event["set_wall_ms"] = (time.monotonic() - t0) * 1000.0
        if changed:
            # CPX re-enumeration: one handle becomes N
            for fresh in self.smi.list_gpus():
                self._gpus[fresh.uuid] = fresh
            try:
                event["metrics_after"] = self.smi.get_metrics(gpu_uuid)
            except SmiError:
                event["metrics_after"] = {}
            self.reconfigure_events.append(event)
            from instaslice_amd.metrics import get_metrics

            get_metrics().reconfigure(
                self.node_name, f"{compute}/{memory}", event["set_wall_ms"] / 1000.0
            )
            g = self._gpus[gpu_uuid]
            # device-plugin nudge (reference: label-toggle reload trick,
            # instaslice_daemonset.go:474-497). The AMD plugin re-enumerates
            # KFD on its health interval so no reload is required; this
            # label is the operator-visible signal (and a hook for plugins
            # that DO watch labels): bumped after every realized mode change
            try:
                self.store.patch("Node", self.node_name, "", [
                    {"op": "set",
                     "path": ["metadata", "labels",
                              "org.instaslice/last-reconfigure"],
                     "value": str(int(time.time()))},
                ], quiet=True)
            except Exception as e:  # noqa: BLE001 - label nudge is best-effort
                self.log.warning("device-plugin nudge failed: %s", e)
            self.log.info(
                "gpu %s reconfigured %s -> %s/%s in %.1f ms (%d partitions)",
                gpu_uuid[:8], event["from"], g.compute_mode, g.memory_mode,
                event["set_wall_ms"], len(g.partitions),
            )

    # -- create path ----------------------------------------------------------

    def _prepare_allocation(self, cr: dict, alloc: dict) -> Optional[dict]:
        """Realize one `creating` allocation on the DEVICE (mode set + ordinal
        resolution); returns the prepared-entry dict keyed by partition uuid,
        or None if it must be retried later. Store writes happen in one
        batched round-trip in the caller (_commit_prepare)."""
        gpu_uuid = alloc["gpuUUID"]
        self._ensure_gpu_mode(cr, gpu_uuid, alloc["computeMode"], alloc["memoryMode"])
        g = self._gpus[gpu_uuid]
        ordinal = alloc["ordinal"]
        part = next((p for p in g.partitions if p.ordinal == ordinal), None)
        if part is None:
            raise SmiError(
                f"gpu {gpu_uuid}: ordinal {ordinal} not present in mode {g.compute_mode}"
            )
        prep = PreparedDetails(
            parent_gpu_uuid=gpu_uuid,
            ordinal=ordinal,
            compute_mode=g.compute_mode,
            memory_mode=g.memory_mode,
            xcds=alloc["size"],
            memory_gb=part.memory_gb,
            pod_uuid=alloc["podUUID"],
            device_index=part.device_index,
        )
        return {part.uuid: prep.to_dict()}

    def _commit_prepare(self, alloc: dict, prepared_entry: dict) -> bool:
        """ONE batched store round-trip for the whole create-path commit:
        pod ConfigMap (visible-devices contract, reference createConfigMap
        instaslice_daemonset.go:796-818 -> ROCR_/HIP_ here) + node capacity
        pin (createInstaSliceResource, :277-300) + CR status flip + the
        FAST-UNGATE: for non-gang pods the same batch removes the pod's
        scheduling gate and marks the allocation `ungated` directly — the
        data plane completes the admission it just realized, saving the
        created-event -> controller -> ungate round (two cross-process event
        hops). Gang members stay `created` so the controller's group barrier
        decides when they ungate. If the pod's gate isn't exactly ours the
        guarded patch conflicts and the classic controller path takes over.
        Returns False if the allocation changed under us (raced teardown)."""
        part_uuid, prep = next(iter(prepared_entry.items()))
        gpu_uuid = alloc["gpuUUID"]
        g = self._gpus[gpu_uuid]
        # ROCR accepts device ordinals (or rocminfo-style GPU-<id> strings),
        # NOT amdsmi UUIDs — verified on MI355X: a UUID value yields "no
        # ROCm-capable device is detected". The partition's node-wide HIP
        # ordinal is the selector; the amdsmi UUID rides along for operators.
        cm = {
            "apiVersion": "v1",
            "kind": "ConfigMap",
            "metadata": {"name": alloc["podName"], "namespace": alloc["namespace"]},
            "data": {
                "ROCR_VISIBLE_DEVICES": str(prep["deviceIndex"]),
                "HIP_VISIBLE_DEVICES": str(prep["deviceIndex"]),
                "INSTASLICE_PARTITION_UUID": part_uuid,
                "INSTASLICE_PARTITION_ORDINAL": str(alloc["ordinal"]),
                "INSTASLICE_PARTITION_GPU": gpu_uuid,
            },
        }
        pu = alloc["podUUID"]
        # gang members must NOT self-ungate (controller barrier owns that)
        fast_ungate = self.fast_ungate and not alloc.get("group")
        target_status = (AllocationStatus.UNGATED if fast_ungate
                         else AllocationStatus.CREATED)
        res = self.store.batch([
            {"verb": "create", "obj": cm},
            {"verb": "patch", "kind": "Node", "name": self.node_name,
             "namespace": "", "ops": [
                 {"op": "set",
                  "path": ["status", "capacity", POD_RESOURCE_PREFIX + alloc["podName"]],
                  "value": 1},
             ]},
            {"verb": "patch", "kind": "Instaslice", "name": self.node_name,
             "namespace": INSTASLICE_NS, "ops": [
                 {"op": "test",
                  "path": ["spec", "allocations", pu, "allocationStatus"],
                  "value": AllocationStatus.CREATING},
                 {"op": "set",
                  "path": ["spec", "allocations", pu, "allocationStatus"],
                  "value": target_status},
                 {"op": "merge", "path": ["spec", "prepared"],
                  "value": prepared_entry},
                 {"op": "set", "path": ["spec", "gpus", gpu_uuid, "computeMode"],
                  "value": g.compute_mode},
                 {"op": "set", "path": ["spec", "gpus", gpu_uuid, "memoryMode"],
                  "value": g.memory_mode},
                 {"op": "add_to_set",
                  "path": ["spec", "gpus", gpu_uuid, "usedOrdinals"],
                  "value": alloc["ordinal"]},
             ]},
            *([{"verb": "patch", "kind": "Pod", "name": alloc["podName"],
                "namespace": alloc["namespace"], "ops": [
                    {"op": "test", "path": ["spec", "schedulingGates"],
                     "value": [{"name": GATE_NAME}]},
                    {"op": "set", "path": ["spec", "schedulingGates"],
                     "value": []},
                    {"op": "set", "path": ["status", "phase"],
                     "value": "Pending"},
                    {"op": "set", "path": ["status", "conditions"],
                     "value": [{"type": "PodScheduled", "status": "True",
                                "message": "ungated"}]},
                ]}] if fast_ungate else []),
        ], quiet=True)
        if not res[0]["ok"] and res[0]["error"]["type"] != "AlreadyExists":
            self.log.warning("configmap create failed: %s", res[0]["error"])
        if not res[1]["ok"]:
            self.log.warning("node capacity pin failed: %s", res[1]["error"])
        commit = res[2]
        if commit["ok"] and fast_ungate and not res[3]["ok"]:
            # OUR commit just landed but the pod's gate set wasn't exactly
            # ours (extra gates): hand the admission back to the controller
            # by downgrading to `created`. MUST be gated on commit["ok"] —
            # on a stale-cache rerun (commit conflicts because an earlier
            # pass already flipped the status) this downgrade would revert
            # a live ungated allocation (r1 flake).
            try:
                self.store.patch("Instaslice", self.node_name, INSTASLICE_NS, [
                    {"op": "test",
                     "path": ["spec", "allocations", pu, "allocationStatus"],
                     "value": AllocationStatus.UNGATED},
                    {"op": "set",
                     "path": ["spec", "allocations", pu, "allocationStatus"],
                     "value": AllocationStatus.CREATED},
                ], quiet=True)
            except (Conflict, NotFound):
                pass
        if not commit["ok"]:
            # Conflict has TWO causes with opposite remedies:
            #   (a) teardown raced us mid-create (allocation gone/deleted)
            #       -> undo our ConfigMap + capacity pin, nobody else will;
            #   (b) stale informer view re-ran an ALREADY-COMMITTED create
            #       (status is created/ungated in the store) -> the ConfigMap
            #       belongs to a live pod; undoing here deletes the env
            #       contract out from under an ungated pod (the r1
            #       "ungated pod without ConfigMap" flake). Check the fresh
            #       object to tell them apart.
            cur_status = None
            try:
                fresh = self.store.get("Instaslice", self.node_name, INSTASLICE_NS)
                cur = (fresh.get("spec", {}).get("allocations") or {}).get(pu)
                if isinstance(cur, dict):
                    cur_status = cur.get("allocationStatus")
            except NotFound:
                pass
            if cur_status in (AllocationStatus.CREATED, AllocationStatus.UNGATED):
                self.log.debug(
                    "prepare for pod %s already committed (stale view); "
                    "keeping ConfigMap", alloc["podName"],
                )
                return True
            self.log.debug("prepare commit superseded: %s", commit["error"]["msg"])
            self.store.batch([
                {"verb": "delete", "kind": "ConfigMap",
                 "name": alloc["podName"], "namespace": alloc["namespace"]},
                {"verb": "patch", "kind": "Node", "name": self.node_name,
                 "namespace": "", "ops": [
                     {"op": "delete",
                      "path": ["status", "capacity",
                               POD_RESOURCE_PREFIX + alloc["podName"]]},
                 ]},
            ], quiet=True)
            return False
        emit(self.store,
             {"kind": "Pod", "namespace": alloc["namespace"],
              "name": alloc["podName"]},
             "PartitionReady",
             f"partition {part_uuid[:8]} (ordinal {alloc['ordinal']} on "
             f"{gpu_uuid[:8]}, device {prep['deviceIndex']}) realized")
        return True

    def _needs_mode_change(self, alloc: dict) -> bool:
        g = self._gpus.get(alloc["gpuUUID"])
        return bool(g and g.compute_mode != alloc["computeMode"])

    def _batch_flips(self, cr: dict, allocations: dict) -> None:
        """FLIP BATCHING: when one reconcile pass holds
        `creating` allocations that need mode changes on SEVERAL distinct
        GPUs, run the device sets concurrently instead of paying the flip
        wall-time once per GPU in sequence (a real compute-partition set is
        O(100 ms-1 s); a burst of placements across 8 GPUs would otherwise
        serialize into seconds). Errors are swallowed here on purpose: the
        sequential per-allocation path immediately redoes the (now no-op or
        still-failing) flip and owns all error/strike accounting."""
        flips = {}
        for alloc in allocations.values():
            if not isinstance(alloc, dict):
                continue
            if alloc.get("allocationStatus") != AllocationStatus.CREATING:
                continue
            gpu = alloc.get("gpuUUID")
            if gpu and gpu not in flips and "computeMode" in alloc \
                    and self._needs_mode_change(alloc):
                flips[gpu] = (alloc["computeMode"],
                              alloc.get("memoryMode") or "NPS1")
        if len(flips) < 2:
            return  # nothing to overlap
        from concurrent.futures import ThreadPoolExecutor

        def one(gpu_uuid, compute, memory):
            try:
                self._ensure_gpu_mode(cr, gpu_uuid, compute, memory)
            except SmiError:
                pass  # sequential path re-attempts and does the accounting

        with ThreadPoolExecutor(max_workers=min(8, len(flips))) as ex:
            for gpu_uuid, (compute, memory) in flips.items():
                ex.submit(one, gpu_uuid, compute, memory)

    def _fail_allocation(self, pod_uuid: str, gpu_uuid: str, lock_mode: bool) -> None:
        """Flip the allocation to `failed` (controller will re-place) and
        optionally mark the GPU mode-locked in the CR."""
        reqs = [
            {"verb": "patch", "kind": "Instaslice", "name": self.node_name,
             "namespace": INSTASLICE_NS, "ops": [
                 {"op": "test",
                  "path": ["spec", "allocations", pod_uuid, "allocationStatus"],
                  "value": AllocationStatus.CREATING},
                 {"op": "set",
                  "path": ["spec", "allocations", pod_uuid, "allocationStatus"],
                  "value": AllocationStatus.FAILED},
             ]},
        ]
        if lock_mode:
            reqs.append(
                {"verb": "patch", "kind": "Instaslice", "name": self.node_name,
                 "namespace": INSTASLICE_NS, "ops": [
                     {"op": "set", "path": ["spec", "gpus", gpu_uuid, "modeLocked"],
                      "value": True},
                 ]})
        self.store.batch(reqs, quiet=True)
        from instaslice_amd.metrics import get_metrics

        get_metrics().allocation("failed")

    # -- reconcile ------------------------------------------------------------

    def _reconcile(self, key: Key) -> Result:
        # informer cache first (fed by our filtered watch): on the TCP store
        # this removes one GET round-trip from every reconcile
        cr = self.engine.cached(("Instaslice", INSTASLICE_NS, self.node_name))
        if cr is None:
            try:
                cr = self.store.get("Instaslice", self.node_name, INSTASLICE_NS)
            except NotFound:
                return Result()
        allocations = cr.get("spec", {}).get("allocations") or {}
        requeue: Optional[float] = None

        self._batch_flips(cr, allocations)

        for pod_uuid, alloc in sorted(allocations.items()):
            # one externally crafted/corrupt entry must not wedge the whole
            # node's reconcile loop (KeyError would abort every iteration
            # through engine error backoff, forever)
            if not isinstance(alloc, dict) or not all(
                k in alloc for k in
                ("allocationStatus", "gpuUUID", "ordinal", "podName",
                 "namespace", "podUUID")
            ):
                self.log.warning("skipping malformed allocation %r", pod_uuid)
                continue
            status = alloc["allocationStatus"]
            if status == AllocationStatus.CREATING:
                needs_flip = self._needs_mode_change(alloc)
                try:
                    prepared_entry = self._prepare_allocation(cr, alloc)
                except SmiBusy as e:
                    self.log.warning("gpu busy, requeueing: %s", e)
                    requeue = REQUEUE_BUSY_S
                    continue
                except (SmiNotSupported, SmiError) as e:
                    # Hard device failure. The reference logs-and-continues
                    # (instaslice_daemonset.go:173-189); we fail the
                    # allocation so the controller re-places it — and after
                    # repeated mode-set failures mark the GPU mode-locked so
                    # the placer stops planning flips on it (VM guests).
                    gpu = alloc["gpuUUID"]
                    n = self._mode_set_failures.get(gpu, 0) + 1
                    self._mode_set_failures[gpu] = n
                    self.prepare_failures += 1
                    # deterministic platform refusals lock the GPU's mode so
                    # the placer stops planning flips on it (VM guests);
                    # transient errors retry, then fail the allocation for
                    # re-placement WITHOUT locking (self-healing)
                    deterministic = isinstance(e, (SmiNotSupported, SmiPermission))
                    if not deterministic and n < 3:
                        self.log.warning(
                            "prepare failed (attempt %d) for pod %s: %s",
                            n, alloc["podName"], e,
                        )
                        requeue = REQUEUE_BUSY_S
                        continue
                    self.log.error(
                        "prepare failed for pod %s on gpu %s (%s): failing allocation",
                        alloc["podName"], gpu[:8], e,
                    )
                    self._mode_set_failures.pop(gpu, None)
                    emit(self.store,
                         {"kind": "Pod", "namespace": alloc["namespace"],
                          "name": alloc["podName"]},
                         "PartitionFailed",
                         f"device error realizing partition on "
                         f"{gpu[:8]}: {e}", type_="Warning")
                    self._fail_allocation(pod_uuid, gpu,
                                          lock_mode=deterministic and needs_flip)
                    continue
                self._mode_set_failures.pop(alloc["gpuUUID"], None)
                if prepared_entry is None:
                    requeue = REQUEUE_BUSY_S
                    continue
                part_uuid = next(iter(prepared_entry))
                if self._commit_prepare(alloc, prepared_entry):
                    from instaslice_amd.metrics import get_metrics

                    get_metrics().allocation("created")
                    self.log.debug(
                        "prepared partition %s for pod %s",
                        part_uuid[:8], alloc["podName"],
                    )
            elif status == AllocationStatus.DELETED:
                self._teardown_allocation(cr, pod_uuid, alloc)
                if self.reset_mode_on_empty:
                    self._maybe_reset_gpu(alloc["gpuUUID"])
        self._apply_mode_hints(cr, allocations)
        return Result(requeue_after=requeue)

    def _apply_mode_hints(self, cr: dict, allocations: dict) -> None:
        """Act on controller drain-time hints (spec.gpus[*].desiredMode):
        pre-flip an IDLE GPU to the demanded mode so the next request's
        latency excludes the flip wall time. The hint clears when satisfied,
        raced by a placement (GPU occupied again), refused by the platform,
        or targeting an unknown/locked GPU — it must never loop."""
        for gpu_uuid, gd in (cr.get("spec", {}).get("gpus") or {}).items():
            if not isinstance(gd, dict):
                continue
            want = gd.get("desiredMode")
            if not want:
                continue
            g = self._gpus.get(gpu_uuid)
            occupied = any(
                isinstance(a, dict) and a.get("gpuUUID") == gpu_uuid
                for a in allocations.values()
            )
            if g is None or gd.get("modeLocked") or occupied \
                    or g.compute_mode == want:
                pass  # clear below
            else:
                try:
                    self._ensure_gpu_mode(cr, gpu_uuid, want, g.memory_mode)
                    from instaslice_amd.metrics import get_metrics

                    get_metrics().allocation("preflip")
                    emit(self.store,
                         {"kind": "Instaslice",
                          "namespace": INSTASLICE_NS,
                          "name": self.node_name},
                         "PreFlipped",
                         f"idle GPU {gpu_uuid[:8]} pre-flipped to {want} "
                         "for starved demand (drain-time mode planning)")
                    self.log.info("pre-flipped idle gpu %s to %s (drain-time "
                                  "mode hint)", gpu_uuid[:8], want)
                except SmiBusy:
                    continue  # transient; retry on the next event
                except (SmiNotSupported, SmiError) as e:
                    self.log.warning("mode hint %s on %s refused: %s",
                                     want, gpu_uuid[:8], e)
            try:
                self.store.patch("Instaslice", self.node_name, INSTASLICE_NS, [
                    {"op": "delete",
                     "path": ["spec", "gpus", gpu_uuid, "desiredMode"]},
                ], quiet=True)
            except (Conflict, NotFound):
                pass

    # -- delete path ----------------------------------------------------------

    def _teardown_allocation(self, cr: dict, pod_uuid: str, alloc: dict) -> None:
        """ONE batched round-trip for the whole delete path: drop ConfigMap +
        capacity pin + allocation/prepared/ordinal from the CR (reference does
        these as separate API calls, instaslice_daemonset.go:415-470)."""
        cr_ops = [
            {"op": "test",
             "path": ["spec", "allocations", pod_uuid, "allocationStatus"],
             "value": AllocationStatus.DELETED},
            {"op": "delete", "path": ["spec", "allocations", pod_uuid]},
            {"op": "remove_from_set",
             "path": ["spec", "gpus", alloc["gpuUUID"], "usedOrdinals"],
             "value": alloc["ordinal"]},
            # predicate delete against the fresh object (see reconciler
            # teardown): stale-view key lists orphan prepared entries
            {"op": "delete_where", "path": ["spec", "prepared"],
             "field": "podUUID", "value": pod_uuid},
        ]
        res = self.store.batch([
            {"verb": "delete", "kind": "ConfigMap", "name": alloc["podName"],
             "namespace": alloc["namespace"]},
            {"verb": "patch", "kind": "Node", "name": self.node_name,
             "namespace": "", "ops": [
                 {"op": "delete",
                  "path": ["status", "capacity",
                           POD_RESOURCE_PREFIX + alloc["podName"]]},
             ]},
            {"verb": "patch", "kind": "Instaslice", "name": self.node_name,
             "namespace": INSTASLICE_NS, "ops": cr_ops},
        ], quiet=True)
        if res[2]["ok"]:
            from instaslice_amd.metrics import get_metrics

            get_metrics().allocation("deleted")
            emit(self.store,
                 {"kind": "Pod", "namespace": alloc["namespace"],
                  "name": alloc["podName"]},
                 "PartitionReleased",
                 f"partition ordinal {alloc['ordinal']} on "
                 f"{alloc['gpuUUID'][:8]} released")

    def _maybe_reset_gpu(self, gpu_uuid: str) -> None:
        """Reference-parity teardown (ci/gi Destroy analog,
        instaslice_daemonset.go:377-413): return a drained GPU to SPX/NPS1."""
        try:
            cr = self.store.get("Instaslice", self.node_name, INSTASLICE_NS)
        except NotFound:
            return
        spec = cr["spec"]
        in_use = any(
            a["gpuUUID"] == gpu_uuid for a in (spec.get("allocations") or {}).values()
        ) or any(
            p["parentGpuUUID"] == gpu_uuid for p in (spec.get("prepared") or {}).values()
        )
        if in_use:
            return
        try:
            self._ensure_gpu_mode(cr, gpu_uuid, "SPX", "NPS1")
        except (SmiBusy, SmiNotSupported) as e:
            self.log.warning("reset of %s skipped: %s", gpu_uuid[:8], e)
            return
        g = self._gpus[gpu_uuid]

        def mut(obj: dict) -> Optional[dict]:
            gd = obj["spec"]["gpus"].get(gpu_uuid)
            if not gd:
                return None
            gd.update(computeMode=g.compute_mode, memoryMode=g.memory_mode)
            return obj

        self.store.update_with_retry("Instaslice", self.node_name, INSTASLICE_NS, mut)

    # -- lifecycle ------------------------------------------------------------

    def _profile_capacity(self, cr: dict) -> Dict[str, int]:
        """Free slots per profile on this node right now: free ordinals on
        GPUs already in the profile's mode + idle flippable GPUs times the
        profile's partition count. This is what the AMD device plugin would
        advertise as extended resources; published on the heartbeat for
        schedulers/autoscalers (eventual consistency is fine there)."""
        from instaslice_amd.controller.policy import build_gpu_views
        from instaslice_amd.partition.profiles import ProfileCatalog

        spec = cr.get("spec", {})
        placements = spec.get("placements") or {}
        if not placements:
            return {}
        cat = ProfileCatalog.from_dict(placements)
        views = build_gpu_views(self.node_name, spec)
        out: Dict[str, int] = {}
        for prof in cat.profiles:
            n = 0
            for v in views:
                if v.compute_mode is prof.compute:
                    n += max(0, prof.partitions_per_gpu - len(v.occupied))
                elif not v.occupied and not v.mode_locked:
                    n += prof.partitions_per_gpu
            out[prof.name] = n
        return out

    def _heartbeat_loop(self) -> None:
        while not self._hb_stop.is_set():
            # liveness + GPU health telemetry in one patch: operators (and
            # the controller's staleness check) read status.heartbeat;
            # status.gpuMetrics carries the live amd-smi counters per GPU —
            # the north star's "amd-smi counters captured" made continuous,
            # not just per-reconfigure
            metrics = {}
            for uuid in list(self._gpus):
                try:
                    m = self.smi.get_metrics(uuid)
                except (SmiError, SmiNotSupported):
                    continue
                metrics[uuid] = {k: round(float(v), 3) for k, v in m.items()}
            ops = [{"op": "set", "path": ["status", "heartbeat"],
                    "value": time.time()}]
            if metrics:
                ops.append({"op": "set", "path": ["status", "gpuMetrics"],
                            "value": metrics})
            reqs = [{"verb": "patch", "kind": "Instaslice",
                     "name": self.node_name, "namespace": INSTASLICE_NS,
                     "ops": ops}]
            cr = self.engine.cached(
                ("Instaslice", INSTASLICE_NS, self.node_name))
            if cr is not None:
                cap = self._profile_capacity(cr)
                if cap and cap != self._last_capacity:
                    self._last_capacity = cap
                    reqs.append(
                        {"verb": "patch", "kind": "Node",
                         "name": self.node_name, "namespace": "", "ops": [
                             {"op": "set",
                              "path": ["status", "capacity",
                                       RESOURCE_PREFIX + prof],
                              "value": n}
                             for prof, n in sorted(cap.items())
                         ]})
            try:
                self.store.batch(reqs, quiet=True)
            except Exception as e:  # store outage: keep trying
                self.log.warning("heartbeat failed: %s", e)
            self._hb_stop.wait(self.heartbeat_every_s)

    def start(self) -> "NodeAgent":
        self.discover()
        self.engine.start()
        if self.heartbeat_every_s > 0:
            self._hb_stop.clear()
            self._hb_thread = threading.Thread(
                target=self._heartbeat_loop, daemon=True,
                name=f"heartbeat-{self.node_name}",
            )
            self._hb_thread.start()
        return self

    def stop(self) -> None:
        self._hb_stop.set()
        if self._hb_thread:
            self._hb_thread.join(timeout=2.0)
        self.engine.stop()
