"""Lease-based leader election over the store.

Reference analog: controller-runtime's lease leader election, enabled with
ids 7cbd68d5/7cbd68d6.codeflare.dev (cmd/controller/main.go:107-108,
cmd/daemonset/main.go:108). Gives single-writer semantics when several
controller replicas run against one store.
"""

from __future__ import annotations

import threading
import time
from typing import Optional

from instaslice_amd.store.memstore import AlreadyExists, Conflict, NotFound
from instaslice_amd.utils import get_logger, new_uid


class LeaderLease:
    def __init__(self, store, name: str, identity: Optional[str] = None,
                 ttl_s: float = 10.0, renew_every_s: float = 3.0) -> None:
        self.store = store
        self.name = name
        self.identity = identity or new_uid()
        self.ttl_s = ttl_s
        self.renew_every_s = renew_every_s
        self.log = get_logger(f"lease.{name}")
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.is_leader = threading.Event()

    def _lease_obj(self) -> dict:
        return {
            "apiVersion": "coordination.k8s.io/v1",
            "kind": "Lease",
            "metadata": {"name": self.name, "namespace": "instaslice-system"},
            "spec": {"holderIdentity": self.identity, "renewTime": time.time(),
                     "leaseDurationSeconds": self.ttl_s},
        }

    def _try_acquire(self) -> bool:
        try:
            cur = self.store.get("Lease", self.name, "instaslice-system")
        except NotFound:
            try:
                self.store.create(self._lease_obj())
                return True
            except AlreadyExists:
                return False
        spec = cur.get("spec", {})
        holder = spec.get("holderIdentity")
        expired = time.time() - float(spec.get("renewTime", 0)) > float(
            spec.get("leaseDurationSeconds", self.ttl_s)
        )
        if holder == self.identity or expired:
            cur["spec"] = self._lease_obj()["spec"]
            try:
                self.store.update(cur)
                return True
            except (Conflict, NotFound):
                return False
        return False

    def _loop(self) -> None:
        while not self._stop.is_set():
            got = self._try_acquire()
            if got and not self.is_leader.is_set():
                self.log.info("acquired leadership as %s", self.identity)
                self.is_leader.set()
            elif not got and self.is_leader.is_set():
                self.log.warning("LOST leadership")
                self.is_leader.clear()
            self._stop.wait(self.renew_every_s if got else self.renew_every_s / 2)

    def start(self) -> "LeaderLease":
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name=f"lease-{self.name}")
        self._thread.start()
        return self

    def wait_leader(self, timeout: Optional[float] = None) -> bool:
        return self.is_leader.wait(timeout)

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2.0)
        # release so a successor can take over immediately
        if self.is_leader.is_set():
            try:
                cur = self.store.get("Lease", self.name, "instaslice-system")
                if cur.get("spec", {}).get("holderIdentity") == self.identity:
                    cur["spec"]["renewTime"] = 0.0
                    self.store.update(cur)
            except Exception:
                pass
            self.is_leader.clear()
