"""Single-process cluster harness: store + controller + node agents.

Used by the integration tests, the e2e fake runs, and bench.py. This is the
moral equivalent of the reference's KinD-based e2e environment
(test/e2e/e2e_test.go) but in-process, so allocation latency measures OUR
machinery, not container pulls.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

from instaslice_amd.agent.daemonset import NodeAgent
from instaslice_amd.api.types import new_pod
from instaslice_amd.controller.reconciler import INSTASLICE_NS, PodController
from instaslice_amd.smi.base import AmdSmi
from instaslice_amd.store.memstore import MemStore, NotFound
from instaslice_amd.utils import get_logger


class Cluster:
    def __init__(
        self,
        store: Optional[MemStore] = None,
        policy: str = "packed-fit",
        teardown_grace_s: float = 0.0,
        reset_mode_on_empty: bool = False,
        backend: str = "mem",
    ) -> None:
        # backend="native" serves state from the C++ store daemon and gives
        # each component its own TCP client — the production data plane
        # inside the single-process harness (requires instaslice-stored;
        # raises if it isn't built)
        self._native_server = None
        self._clients: List = []
        if store is None and backend == "native":
            from instaslice_amd.store.native import NativeStoreServer
            from instaslice_amd.store.netstore import NetStoreClient

            self._native_server = NativeStoreServer().start()

            def _client():
                c = NetStoreClient("127.0.0.1", self._native_server.port)
                self._clients.append(c)
                return c

            self._mk_store = _client
            self.store = _client()
        else:
            self.store = store or MemStore()
            self._mk_store = lambda: self.store
        self.controller = PodController(
            self._mk_store(), policy=policy, teardown_grace_s=teardown_grace_s
        )
        self.agents: Dict[str, NodeAgent] = {}
        self.reset_mode_on_empty = reset_mode_on_empty
        self.log = get_logger("cluster")
        self._started = False

    def add_node(self, node_name: str, smi: AmdSmi) -> NodeAgent:
        agent = NodeAgent(
            self._mk_store(), smi, node_name,
            reset_mode_on_empty=self.reset_mode_on_empty,
        )
        self.agents[node_name] = agent
        if self._started:
            agent.start()
        return agent

    def start(self) -> "Cluster":
        for agent in self.agents.values():
            agent.start()
        self.controller.start()
        self._started = True
        return self

    def stop(self) -> None:
        self.controller.stop()
        for agent in self.agents.values():
            agent.stop()
        for c in self._clients:
            c.close()
        if self._native_server is not None:
            self._native_server.stop()

    # -- workload helpers --------------------------------------------------

    def submit_pod(
        self, name: str, profile: str, namespace: str = "default",
        node: Optional[str] = None, group: Optional[str] = None,
    ) -> dict:
        sel = {"kubernetes.io/hostname": node} if node else None
        pod = new_pod(name, namespace=namespace, profile=profile,
                      node_selector=sel, group=group)
        return self.store.create(pod)

    def wait_pod_scheduled(self, name: str, namespace: str = "default",
                           timeout: float = 10.0) -> dict:
        """Block until the pod's scheduling gate is removed (the reference's
        'pod goes Running' moment, minus kubelet). Returns the pod."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            try:
                pod = self.store.get("Pod", name, namespace)
            except NotFound:
                time.sleep(0.002)
                continue
            if not pod["spec"].get("schedulingGates"):
                return pod
            time.sleep(0.002)
        raise TimeoutError(f"pod {namespace}/{name} not scheduled in {timeout}s")

    def wait_pod_outcome(self, name: str, namespace: str = "default",
                         timeout: float = 10.0) -> str:
        """Block until the pod is either scheduled ("scheduled") or marked
        unschedulable by the controller ("unschedulable")."""
        from instaslice_amd import UNSCHEDULABLE_ANNOTATION

        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            try:
                pod = self.store.get("Pod", name, namespace)
            except NotFound:
                time.sleep(0.002)
                continue
            if not pod["spec"].get("schedulingGates"):
                return "scheduled"
            ann = pod["metadata"].get("annotations") or {}
            if UNSCHEDULABLE_ANNOTATION in ann:
                return "unschedulable"
            time.sleep(0.002)
        raise TimeoutError(f"pod {namespace}/{name}: no outcome in {timeout}s")

    def delete_pod(self, name: str, namespace: str = "default") -> None:
        self.store.delete("Pod", name, namespace)

    def wait_pod_gone(self, name: str, namespace: str = "default",
                      timeout: float = 10.0) -> None:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            try:
                self.store.get("Pod", name, namespace)
            except NotFound:
                return
            time.sleep(0.002)
        raise TimeoutError(f"pod {namespace}/{name} still present after {timeout}s")

    def wait_pod_unallocated(self, pod_name: str, timeout: float = 10.0) -> None:
        """Block until no node CR carries an allocation for `pod_name`
        (capacity actually freed — the moment waiting pods can re-place)."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if not any(
                a.get("podName") == pod_name
                for cr in self.store.list("Instaslice")
                for a in (cr["spec"].get("allocations") or {}).values()
            ):
                return
            time.sleep(0.005)
        raise TimeoutError(f"allocation for {pod_name} never drained")

    def wait_allocations_empty(self, node: str, timeout: float = 10.0) -> None:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            cr = self.store.get("Instaslice", node, INSTASLICE_NS)
            if not cr["spec"].get("allocations") and not cr["spec"].get("prepared"):
                return
            time.sleep(0.002)
        raise TimeoutError(f"allocations on {node} not drained after {timeout}s")

    def pod_env(self, name: str, namespace: str = "default") -> Dict[str, str]:
        """The env the workload container receives (envFrom pod-named
        ConfigMap; reference contract samples/test-pod.yaml:17-20)."""
        cm = self.store.get("ConfigMap", name, namespace)
        return dict(cm["data"])

    def allocations(self, node: str) -> Dict[str, dict]:
        cr = self.store.get("Instaslice", node, INSTASLICE_NS)
        return cr["spec"].get("allocations") or {}

    def prepared(self, node: str) -> Dict[str, dict]:
        cr = self.store.get("Instaslice", node, INSTASLICE_NS)
        return cr["spec"].get("prepared") or {}
