"""Real Kubernetes API-server adapter with the MemStore verb set.

On a cluster the controller/daemonset run against the API server itself; the
reconcilers are store-agnostic, so this adapter is all that changes (the
in-memory and TCP stores cover tests and benches). Requires the `kubernetes`
pip package, which is intentionally NOT vendored — import is lazy and the
constructor raises a clear error when it's missing (this build environment
has no cluster to run against; tests/test_k8sstore.py covers the interface
contract and skips the live part without the package).

Mapping:
  Pod / ConfigMap / Node  -> core v1 API
  Instaslice              -> custom objects API (inference.codeflare.dev)
  Lease                   -> coordination.k8s.io v1
  resourceVersion conflicts -> kubernetes 409 -> Conflict
  watch                   -> kubernetes.watch.Watch streaming thread
"""

from __future__ import annotations

import queue
import threading
from typing import Callable, List, Optional

from instaslice_amd import API_GROUP, API_VERSION
from instaslice_amd.store.memstore import AlreadyExists, Conflict, NotFound
from instaslice_amd.utils import get_logger


def _require_k8s():
    try:
        import kubernetes  # noqa: F401

        return kubernetes
    except ImportError as e:
        raise ImportError(
            "store.k8sstore requires the `kubernetes` package "
            "(pip install kubernetes); use MemStore/NetStoreClient otherwise"
        ) from e


class _K8sWatch:
    def __init__(self) -> None:
        self._q: "queue.Queue[Optional[Tuple[str, dict]]]" = queue.Queue()
        self._stopped = False
        self.kind: Optional[str] = None

    def next(self, timeout: Optional[float] = None):
        try:
            return self._q.get(timeout=timeout)
        except queue.Empty:
            return None

    def stop(self) -> None:
        self._stopped = True
        self._q.put(None)


class K8sStore:
    """MemStore-compatible adapter over a live API server."""

    def __init__(self, namespace_default: str = "default") -> None:
        k8s = _require_k8s()
        try:
            k8s.config.load_incluster_config()
        except Exception:
            k8s.config.load_kube_config()
        self._k8s = k8s
        self._core = k8s.client.CoreV1Api()
        self._custom = k8s.client.CustomObjectsApi()
        self._coord = k8s.client.CoordinationV1Api()
        self.namespace_default = namespace_default
        self.log = get_logger("k8sstore")
        self._watch_threads: List[threading.Thread] = []

    # -- plumbing ------------------------------------------------------------

    def _raise(self, e, kind, name):
        status = getattr(e, "status", None)
        if status == 404:
            raise NotFound(f"{kind}/{name} not found") from None
        if status == 409:
            msg = str(getattr(e, "body", e))
            if "AlreadyExists" in msg or "already exists" in msg:
                raise AlreadyExists(f"{kind}/{name} exists") from None
            raise Conflict(f"{kind}/{name}: {msg}") from None
        raise

    def _route(self, kind: str):
        return kind  # dispatch happens in each verb

    @staticmethod
    def _to_dict(obj) -> dict:
        if isinstance(obj, dict):
            return obj
        import kubernetes.client as client

        return client.ApiClient().sanitize_for_serialization(obj)

    # -- verbs ----------------------------------------------------------------

    def create(self, obj: dict) -> dict:
        kind = obj["kind"]
        ns = obj["metadata"].get("namespace", self.namespace_default)
        name = obj["metadata"]["name"]
        ApiException = self._k8s.client.rest.ApiException
        try:
            if kind == "Instaslice":
                return self._custom.create_namespaced_custom_object(
                    API_GROUP, API_VERSION, ns, "instaslices", obj)
            if kind == "Pod":
                return self._to_dict(self._core.create_namespaced_pod(ns, obj))
            if kind == "ConfigMap":
                return self._to_dict(self._core.create_namespaced_config_map(ns, obj))
            if kind == "Node":
                return self._to_dict(self._core.create_node(obj))
            if kind == "Lease":
                return self._to_dict(self._coord.create_namespaced_lease(ns, obj))
            if kind == "Event":
                return self._to_dict(
                    self._core.create_namespaced_event(ns, obj))
            raise ValueError(f"unsupported kind {kind}")
        except ApiException as e:
            self._raise(e, kind, name)

    def get(self, kind: str, name: str, namespace: str = "") -> dict:
        ns = namespace or self.namespace_default
        ApiException = self._k8s.client.rest.ApiException
        try:
            if kind == "Instaslice":
                return self._custom.get_namespaced_custom_object(
                    API_GROUP, API_VERSION, ns, "instaslices", name)
            if kind == "Pod":
                return self._to_dict(self._core.read_namespaced_pod(name, ns))
            if kind == "ConfigMap":
                return self._to_dict(self._core.read_namespaced_config_map(name, ns))
            if kind == "Node":
                return self._to_dict(self._core.read_node(name))
            if kind == "Lease":
                return self._to_dict(self._coord.read_namespaced_lease(name, ns))
            if kind == "Event":
                return self._to_dict(
                    self._core.read_namespaced_event(name, ns))
            raise ValueError(f"unsupported kind {kind}")
        except ApiException as e:
            self._raise(e, kind, name)

    def list(self, kind: str, namespace: Optional[str] = None) -> List[dict]:
        ApiException = self._k8s.client.rest.ApiException
        try:
            if kind == "Instaslice":
                res = self._custom.list_cluster_custom_object(
                    API_GROUP, API_VERSION, "instaslices")
                return list(res.get("items", []))
            if kind == "Pod":
                res = (self._core.list_namespaced_pod(namespace)
                       if namespace else self._core.list_pod_for_all_namespaces())
                return [self._to_dict(i) for i in res.items]
            if kind == "ConfigMap":
                res = (self._core.list_namespaced_config_map(namespace)
                       if namespace
                       else self._core.list_config_map_for_all_namespaces())
                return [self._to_dict(i) for i in res.items]
            if kind == "Node":
                return [self._to_dict(i) for i in self._core.list_node().items]
            raise ValueError(f"unsupported kind {kind}")
        except ApiException as e:
            self._raise(e, kind, "*")

    def update(self, obj: dict) -> dict:
        """Full-object replace. Status is a SUBRESOURCE on Instaslice/Node
        (and Pod): a replace of the main resource silently keeps the old
        status on a real API server, so kinds whose status carries protocol
        (Instaslice heartbeat/processed, Node capacity) get a second
        status-subresource replace chained on the fresh resourceVersion.
        Without this, agent heartbeats and capacity pins are silently
        dropped on a real cluster (r2 behavioral-tier finding)."""
        kind = obj["kind"]
        ns = obj["metadata"].get("namespace", self.namespace_default)
        name = obj["metadata"]["name"]
        ApiException = self._k8s.client.rest.ApiException

        def with_rv(res: dict) -> dict:
            body = dict(obj)
            body["metadata"] = dict(obj["metadata"])
            body["metadata"]["resourceVersion"] = res["metadata"][
                "resourceVersion"]
            return body

        def status_phase(replace_status, res: dict) -> dict:
            """Chained status replace with ITS OWN conflict retry. The main
            replace has already committed, so a 409 here (another writer
            slipped between our two calls) must NOT surface as Conflict —
            the caller's update_with_retry would re-apply a mutation that
            already landed (double-applied counters caught by the
            contention test). Refresh the rv and retry the status write
            alone (last-writer-wins per subresource, the controller-runtime
            status().update pattern)."""
            body = with_rv(res)
            for _ in range(16):
                try:
                    return replace_status(body)
                except ApiException as e:
                    if getattr(e, "status", None) != 409:
                        raise
                    fresh = self.get(kind, name, ns)
                    body["metadata"]["resourceVersion"] = fresh["metadata"][
                        "resourceVersion"]
            raise Conflict(f"{kind}/{name}: status replace kept conflicting")

        try:
            if kind == "Instaslice":
                res = self._custom.replace_namespaced_custom_object(
                    API_GROUP, API_VERSION, ns, "instaslices", name, obj)
                if obj.get("status") is not None:
                    res = status_phase(
                        lambda b: self._custom
                        .replace_namespaced_custom_object_status(
                            API_GROUP, API_VERSION, ns, "instaslices",
                            name, b),
                        res)
                return res
            if kind == "Pod":
                return self._to_dict(self._core.replace_namespaced_pod(name, ns, obj))
            if kind == "ConfigMap":
                return self._to_dict(
                    self._core.replace_namespaced_config_map(name, ns, obj))
            if kind == "Node":
                # metadata/spec (labels nudge) via main replace, capacity via
                # the status subresource — replace (not merge-patch) so
                # removed capacity keys actually go away
                res = self._to_dict(self._core.replace_node(name, obj))
                if obj.get("status") is not None:
                    res = status_phase(
                        lambda b: self._to_dict(
                            self._core.replace_node_status(name, b)),
                        res)
                return res
            if kind == "Lease":
                return self._to_dict(
                    self._coord.replace_namespaced_lease(name, ns, obj))
            if kind == "Event":
                return self._to_dict(
                    self._core.replace_namespaced_event(name, ns, obj))
            raise ValueError(f"unsupported kind {kind}")
        except ApiException as e:
            self._raise(e, kind, name)

    def delete(self, kind: str, name: str, namespace: str = "", *, now: float = 0.0) -> None:
        ns = namespace or self.namespace_default
        ApiException = self._k8s.client.rest.ApiException
        try:
            if kind == "Instaslice":
                self._custom.delete_namespaced_custom_object(
                    API_GROUP, API_VERSION, ns, "instaslices", name)
            elif kind == "Pod":
                self._core.delete_namespaced_pod(name, ns)
            elif kind == "ConfigMap":
                self._core.delete_namespaced_config_map(name, ns)
            elif kind == "Node":
                self._core.delete_node(name)
            elif kind == "Event":
                self._core.delete_namespaced_event(name, ns)
            else:
                raise ValueError(f"unsupported kind {kind}")
        except ApiException as e:
            self._raise(e, kind, name)

    # -- watch ----------------------------------------------------------------

    _WATCHABLE = {
        "Pod": ("core", "list_pod_for_all_namespaces"),
        "ConfigMap": ("core", "list_config_map_for_all_namespaces"),
        "Node": ("core", "list_node"),
    }

    def watch(self, kind: Optional[str] = None, *, replay: bool = True,
              filters: Optional[list] = None):
        if kind is None and not filters:
            raise ValueError("K8sStore.watch requires an explicit kind")
        if filters:
            # one upstream watch per distinct kind named in the filters,
            # client-side filtered (k8s field selectors could narrow further;
            # correctness first)
            from instaslice_amd.store.memstore import _filter_matches

            kinds = {f.get("kind") for f in filters}
            if None in kinds:
                raise ValueError("K8sStore filters need explicit kinds")
            merged = _K8sWatch()
            merged.kind = None
            for knd in sorted(kinds):
                inner = self.watch(knd, replay=replay)

                def forward(inner=inner):
                    while not merged._stopped:
                        ev = inner.next(timeout=0.5)
                        if ev is None:
                            continue
                        if any(_filter_matches(f, ev[1]) for f in filters):
                            merged._q.put(ev)
                    inner.stop()

                t = threading.Thread(target=forward, daemon=True,
                                     name=f"k8swatch-filter-{knd}")
                t.start()
                self._watch_threads.append(t)
            return merged
        w = _K8sWatch()
        w.kind = kind

        def pump():
            # LIST+WATCH, the controller-runtime informer protocol: the
            # engine's cache needs the pre-existing objects (replay) — a
            # bare k8s watch only streams CHANGES, so without the initial
            # list, pods submitted before the controller started would
            # never be reconciled. The list's resourceVersion seeds the
            # stream so no event between list and watch is lost; a 410
            # Gone (compacted history) resets to a fresh relist.
            kwatch = self._k8s.watch.Watch()
            rv: Optional[int] = None
            while not w._stopped:
                try:
                    if rv is None and replay:
                        rv = 0
                        for o in self.list(kind):
                            if w._stopped:
                                return
                            rv = max(rv, int(
                                o["metadata"].get("resourceVersion", "0")))
                            w._q.put(("ADDED", o))
                    kwargs = {"timeout_seconds": 30}
                    if rv is not None:
                        kwargs["resource_version"] = str(rv)
                    if kind == "Instaslice":
                        stream = kwatch.stream(
                            self._custom.list_cluster_custom_object,
                            API_GROUP, API_VERSION, "instaslices", **kwargs)
                    else:
                        api, fn = self._WATCHABLE[kind]
                        target = getattr(
                            self._core if api == "core" else self._coord, fn)
                        stream = kwatch.stream(target, **kwargs)
                    for ev in stream:
                        if w._stopped:
                            return
                        obj = self._to_dict(ev["object"])
                        new_rv = obj.get("metadata", {}).get("resourceVersion")
                        if new_rv is not None:
                            try:
                                rv = max(rv or 0, int(new_rv))
                            except ValueError:
                                pass
                        w._q.put((ev["type"], obj))
                except Exception as e:  # reconnect loop (API server restarts)
                    if w._stopped:
                        return
                    status = getattr(e, "status", None)
                    if status == 410:  # history compacted: full relist
                        rv = None
                        self.log.warning("watch %s: 410 Gone, relisting", kind)
                    else:
                        self.log.warning("watch %s reconnecting: %s", kind, e)

        t = threading.Thread(target=pump, daemon=True, name=f"k8swatch-{kind}")
        t.start()
        self._watch_threads.append(t)
        return w

    def patch(self, kind: str, name: str, namespace: str = "",
              ops: Optional[list] = None, *, quiet: bool = False) -> dict:
        """PATCH emulation over get+update (real k8s rejects unknown-verb
        shortcuts; server-side-apply would be the native path). Failed test
        ops raise Conflict straight through — same contract as MemStore."""
        from instaslice_amd.store.memstore import apply_patch_ops

        def mut(obj: dict):
            apply_patch_ops(obj, ops or [])
            return obj

        res = self.update_with_retry(kind, name, namespace, mut)
        if res is None:
            raise NotFound(f"{kind}/{name} not found")
        return res

    def batch(self, requests: list, *, quiet: bool = False) -> list:
        out = []
        for req in requests:
            verb = req.get("verb")
            try:
                if verb == "create":
                    res = self.create(req["obj"])
                elif verb == "get":
                    res = self.get(req["kind"], req["name"], req.get("namespace", ""))
                elif verb == "update":
                    res = self.update(req["obj"])
                elif verb == "delete":
                    self.delete(req["kind"], req["name"], req.get("namespace", ""))
                    res = None
                elif verb == "patch":
                    res = self.patch(req["kind"], req["name"],
                                     req.get("namespace", ""), req.get("ops"))
                else:
                    raise ValueError(f"unknown batch verb {verb!r}")
                out.append({"ok": True, "result": res})
            except (Conflict, NotFound, AlreadyExists) as e:
                out.append({"ok": False,
                            "error": {"type": type(e).__name__, "msg": str(e)}})
        return out

    def update_with_retry(
        self, kind: str, name: str, namespace: str,
        mutate: Callable[[dict], Optional[dict]], attempts: int = 25,
    ):
        for attempt in range(attempts):
            try:
                obj = self.get(kind, name, namespace)
            except NotFound:
                return None
            new = mutate(obj)
            if new is None:
                return None
            try:
                return self.update(new)
            except Conflict:
                if attempt >= 2:  # hot object: jittered backoff breaks livelock
                    import random
                    import time as _time

                    _time.sleep(random.random() * 0.002 * attempt)
                continue
        raise Conflict(f"update_with_retry exhausted for {kind}/{name}")
