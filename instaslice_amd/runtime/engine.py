"""Event-driven reconcile engine: the controller-runtime analog.

The reference gets watches, work queues, requeue-after and conflict-retry
from sigs.k8s.io/controller-runtime (SetupWithManager,
instaslice_controller.go:410-424). This module provides the same machinery
over our store: watch pumps feed a deduplicating keyed work queue; worker
threads call `reconcile(key)`; a `Result(requeue_after=...)` re-enqueues the
key later (the reference requeues 1 s/2 s/5 s in the same situations,
instaslice_controller.go:93,225,231).
"""

from __future__ import annotations

import heapq
import threading
import time
import traceback
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

from instaslice_amd.store.memstore import MemStore, Watch
from instaslice_amd.utils import get_logger

Key = Tuple[str, str, str]  # (kind, namespace, name)

# process-wide reconcile trace (bounded ring; see metrics.Tracer). Dumped via
# `python -m instaslice_amd status` debugging or read directly in tests.
_tracer = None


def get_tracer():
    global _tracer
    if _tracer is None:
        from instaslice_amd.metrics import Tracer

        _tracer = Tracer(capacity=8192)
    return _tracer


@dataclass
class Result:
    requeue_after: Optional[float] = None  # seconds; None = done


@dataclass
class WatchSpec:
    kind: str
    # maps an event object to the primary keys to enqueue; default: identity.
    # Reference analog: podMapFunc mapping Instaslice events to pods
    # (instaslice_controller.go:398-407).
    map_fn: Optional[Callable[[str, dict], List[Key]]] = None
    # server-side scoping (store watch filters): a node agent subscribes to
    # its OWN Instaslice CR only, so store event fan-out stays O(relevant)
    # instead of O(cluster) — the k8s field-selector analog
    filters: Optional[List[dict]] = None


class Engine:
    """One reconciler: N watch pumps -> keyed queue -> worker threads."""

    def __init__(
        self,
        name: str,
        store: MemStore,
        reconcile: Callable[[Key], Result],
        watches: List[WatchSpec],
        workers: int = 1,
        error_backoff_s: float = 0.2,
    ) -> None:
        self.name = name
        self.store = store
        self.reconcile = reconcile
        self.watch_specs = watches
        self.workers = workers
        self.error_backoff_s = error_backoff_s
        self.log = get_logger(f"engine.{name}")

        self._queue: List[Key] = []
        self._queued: set = set()
        # informer cache: latest object seen per key, fed by the watch pumps
        # (client-go shared-informer analog). Reconciles read it via
        # `cached()` instead of a store GET — on the TCP store that removes
        # one round-trip from every reconcile. Writers guard their patches
        # with test ops, so a stale cache surfaces as Conflict, after which
        # `invalidate()` forces the next reconcile back to a real GET.
        self._cache: Dict[Key, dict] = {}
        self._cache_lock = threading.Lock()
        # keys currently being reconciled (multi-worker safety: a key is
        # never processed concurrently; events arriving mid-reconcile mark it
        # dirty and it re-runs right after — controller-runtime semantics)
        self._in_flight: set = set()
        self._dirty: set = set()
        self._timers: List[Tuple[float, int, Key]] = []  # heap by deadline
        self._timer_seq = 0
        self._cv = threading.Condition()
        self._threads: List[threading.Thread] = []
        self._watches: List[Watch] = []
        self._stopping = False
        # observability: reconcile counters (metrics module hooks in here)
        self.reconcile_count = 0
        self.error_count = 0

    # -- queue ------------------------------------------------------------

    def enqueue(self, key: Key) -> None:
        with self._cv:
            if key in self._in_flight:
                self._dirty.add(key)
            elif key not in self._queued:
                self._queue.append(key)
                self._queued.add(key)
                self._cv.notify_all()

    def enqueue_after(self, key: Key, delay_s: float) -> None:
        with self._cv:
            self._timer_seq += 1
            heapq.heappush(self._timers, (time.monotonic() + delay_s, self._timer_seq, key))
            self._cv.notify_all()

    def _next_key(self) -> Optional[Key]:
        with self._cv:
            while not self._stopping:
                now = time.monotonic()
                while self._timers and self._timers[0][0] <= now:
                    _, _, k = heapq.heappop(self._timers)
                    if k not in self._queued:
                        self._queue.append(k)
                        self._queued.add(k)
                if self._queue:
                    k = self._queue.pop(0)
                    self._queued.discard(k)
                    self._in_flight.add(k)
                    return k
                timeout = None
                if self._timers:
                    timeout = max(0.0, self._timers[0][0] - now)
                self._cv.wait(timeout=timeout if timeout is not None else 0.5)
            return None

    # -- pumps & workers ----------------------------------------------------

    def _handle_event(self, spec: WatchSpec, event_type: str, obj: dict) -> None:
        """Cache + enqueue for one event. Runs either on a pump thread
        (queue-mode watches) or DIRECTLY on the store's notify/reader thread
        (push-mode set_callback) — one fewer handoff per event; must stay
        fast and never call the store."""
        md = obj.get("metadata", {})
        own_key = (obj["kind"], md.get("namespace", ""), md["name"])
        with self._cache_lock:
            if event_type == "DELETED":
                self._cache.pop(own_key, None)
            else:
                self._cache[own_key] = obj
        if spec.map_fn is None:
            keys = [own_key]
        else:
            keys = spec.map_fn(event_type, obj)
        for k in keys:
            self.enqueue(k)

    def _pump(self, spec: WatchSpec, watch: Watch) -> None:
        while not self._stopping:
            ev = watch.next(timeout=0.5)
            if ev is None:
                continue
            self._handle_event(spec, ev[0], ev[1])

    def cached(self, key: Key) -> Optional[dict]:
        """Latest watched object for `key`, or None. READ-ONLY — watch event
        objects are shared; mutate a copy."""
        with self._cache_lock:
            return self._cache.get(key)

    def cached_list(self, kind: str) -> List[dict]:
        """All cached objects of `kind`, name-sorted. READ-ONLY objects.
        Complete once the watch replay has been pumped; callers that can see
        an empty startup window must tolerate it (reconcile requeues cover
        it, same as a not-yet-created CR)."""
        with self._cache_lock:
            return [o for k, o in sorted(self._cache.items()) if k[0] == kind]

    def invalidate(self, key: Key) -> None:
        with self._cache_lock:
            self._cache.pop(key, None)

    def _work(self) -> None:
        from instaslice_amd.metrics import get_metrics

        metrics = get_metrics()
        while True:
            key = self._next_key()
            if key is None:
                return
            t0 = time.monotonic()
            try:
                self.reconcile_count += 1
                res = self.reconcile(key)
            except Exception:
                self.error_count += 1
                metrics.reconcile(self.name, "error", time.monotonic() - t0)
                self.log.error("reconcile %s failed:\n%s", key, traceback.format_exc())
                self._finish(key, dirty_requeue=True)
                self.enqueue_after(key, self.error_backoff_s)
                continue
            requeued = res and res.requeue_after is not None
            dt = time.monotonic() - t0
            metrics.reconcile(self.name, "requeue" if requeued else "ok", dt)
            get_tracer().event(
                "reconcile", engine=self.name, key=list(key),
                duration_s=round(dt, 6), requeued=bool(requeued),
            )
            self._finish(key)
            if requeued:
                self.enqueue_after(key, res.requeue_after)

    def _finish(self, key: Key, dirty_requeue: bool = False) -> None:
        with self._cv:
            self._in_flight.discard(key)
            if key in self._dirty and not dirty_requeue:
                self._dirty.discard(key)
                if key not in self._queued:
                    self._queue.append(key)
                    self._queued.add(key)
                    self._cv.notify_all()
            else:
                self._dirty.discard(key)

    # -- lifecycle ----------------------------------------------------------

    def start(self) -> "Engine":
        for spec in self.watch_specs:
            w = self.store.watch(spec.kind, replay=True, filters=spec.filters)
            self._watches.append(w)
            if hasattr(w, "set_callback"):
                # push mode: events invoke cache+enqueue straight from the
                # notify/reader thread — no pump thread, no queue handoff
                def make_cb(sp):
                    def cb(event_type, obj, _sp=sp):
                        self._handle_event(_sp, event_type, obj)
                    return cb

                w.set_callback(make_cb(spec))
            else:
                t = threading.Thread(
                    target=self._pump, args=(spec, w),
                    name=f"{self.name}-pump-{spec.kind}", daemon=True,
                )
                t.start()
                self._threads.append(t)
        for i in range(self.workers):
            t = threading.Thread(target=self._work, name=f"{self.name}-worker-{i}", daemon=True)
            t.start()
            self._threads.append(t)
        return self

    def stop(self) -> None:
        with self._cv:
            self._stopping = True
            self._cv.notify_all()
        for w in self._watches:
            w.stop()
        for t in self._threads:
            t.join(timeout=2.0)

    def wait_idle(self, timeout: float = 10.0, settle: float = 0.05) -> bool:
        """Test helper: block until the queue stays empty for `settle`
        seconds (ignoring pending timers). Returns False on timeout."""
        deadline = time.monotonic() + timeout
        quiet_since = None
        while time.monotonic() < deadline:
            with self._cv:
                empty = (not self._queue and not self._queued
                         and not self._in_flight)
            if empty:
                if quiet_since is None:
                    quiet_since = time.monotonic()
                elif time.monotonic() - quiet_since >= settle:
                    return True
            else:
                quiet_since = None
            time.sleep(0.01)
        return False
