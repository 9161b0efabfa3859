"""Observability: Prometheus metrics + reconcile-phase tracing.

The reference exposes only controller-runtime default metrics and registers
no custom ones (SURVEY.md §5 'Metrics'); the north star requires amd-smi
counters captured per reconfigure and reconcile-phase timing histograms
(they back the p50-allocation-latency objective). This module provides:

  - counters/histograms for reconciles, allocations, partition reconfigures
  - a lightweight phase tracer (ring buffer of timestamped events, the
    tracing subsystem the reference lacks)
  - an optional HTTP endpoint (/metrics via prometheus_client, /healthz,
    /readyz) mirroring the reference's probe wiring
    (cmd/controller/main.go:143-150)

prometheus_client is optional at runtime: without it the same API records
into in-process counters that tests and the CLI can read.
"""

from __future__ import annotations

import collections
import threading
import time
from typing import Deque, Dict, List, Optional, Tuple

try:
    import prometheus_client as prom

    PROM_AVAILABLE = True
except ImportError:  # pragma: no cover
    prom = None
    PROM_AVAILABLE = False

_LOCK = threading.Lock()
_REGISTRY = None


class Metrics:
    """Process-wide metrics registry (one per process; idempotent)."""

    def __init__(self) -> None:
        self._counts: Dict[Tuple[str, tuple], float] = {}
        self._hists: Dict[Tuple[str, tuple], List[float]] = {}
        if PROM_AVAILABLE:
            self.registry = prom.CollectorRegistry()
            self._p_reconciles = prom.Counter(
                "instaslice_reconciles_total",
                "Reconcile invocations",
                ["engine", "outcome"],
                registry=self.registry,
            )
            self._p_reconcile_s = prom.Histogram(
                "instaslice_reconcile_seconds",
                "Reconcile wall time",
                ["engine"],
                registry=self.registry,
                buckets=(0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1,
                         0.25, 0.5, 1.0, 2.5),
            )
            self._p_alloc_s = prom.Histogram(
                "instaslice_allocation_latency_seconds",
                "Gated pod submit -> ungated",
                registry=self.registry,
                buckets=(0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25,
                         0.5, 1.0, 2.5, 5.0, 10.0),
            )
            self._p_reconfig = prom.Counter(
                "instaslice_partition_reconfigures_total",
                "Whole-GPU partition mode changes",
                ["node", "to_mode"],
                registry=self.registry,
            )
            self._p_reconfig_s = prom.Histogram(
                "instaslice_partition_reconfigure_seconds",
                "amdsmi_set_gpu_*_partition wall time",
                registry=self.registry,
            )
            self._p_allocs = prom.Counter(
                "instaslice_allocations_total",
                "Allocation outcomes",
                ["outcome"],  # created | failed | deleted
                registry=self.registry,
            )

    # -- generic fallback recording -----------------------------------------

    def _count(self, name: str, labels: tuple, v: float = 1.0) -> None:
        with _LOCK:
            k = (name, labels)
            self._counts[k] = self._counts.get(k, 0.0) + v

    def _observe(self, name: str, labels: tuple, v: float) -> None:
        with _LOCK:
            self._hists.setdefault((name, labels), []).append(v)

    # -- typed API ------------------------------------------------------------

    def reconcile(self, engine: str, outcome: str, seconds: float) -> None:
        self._count("reconciles_total", (engine, outcome))
        self._observe("reconcile_seconds", (engine,), seconds)
        if PROM_AVAILABLE:
            self._p_reconciles.labels(engine, outcome).inc()
            self._p_reconcile_s.labels(engine).observe(seconds)

    def allocation_latency(self, seconds: float) -> None:
        self._observe("allocation_latency_seconds", (), seconds)
        if PROM_AVAILABLE:
            self._p_alloc_s.observe(seconds)

    def reconfigure(self, node: str, to_mode: str, seconds: float) -> None:
        self._count("partition_reconfigures_total", (node, to_mode))
        self._observe("partition_reconfigure_seconds", (), seconds)
        if PROM_AVAILABLE:
            self._p_reconfig.labels(node, to_mode).inc()
            self._p_reconfig_s.observe(seconds)

    def allocation(self, outcome: str) -> None:
        self._count("allocations_total", (outcome,))
        if PROM_AVAILABLE:
            self._p_allocs.labels(outcome).inc()

    # -- reads (tests / CLI) ---------------------------------------------------

    def count(self, name: str, labels: tuple = ()) -> float:
        with _LOCK:
            return self._counts.get((name, labels), 0.0)

    def samples(self, name: str, labels: tuple = ()) -> List[float]:
        with _LOCK:
            return list(self._hists.get((name, labels), []))

    def percentile(self, name: str, q: float, labels: tuple = ()) -> Optional[float]:
        xs = sorted(self.samples(name, labels))
        if not xs:
            return None
        idx = min(len(xs) - 1, int(q * len(xs)))
        return xs[idx]

    def export_text(self) -> str:
        if PROM_AVAILABLE:
            return prom.generate_latest(self.registry).decode()
        lines = []
        with _LOCK:
            for (name, labels), v in sorted(self._counts.items()):
                lines.append(f"{name}{list(labels)} {v}")
        return "\n".join(lines)


def get_metrics() -> Metrics:
    global _REGISTRY
    with _LOCK:
        if _REGISTRY is None:
            _REGISTRY = Metrics()
    return _REGISTRY


class Tracer:
    """Bounded ring of timestamped reconcile-phase events — the tracing
    subsystem the reference lacks (SURVEY.md §5 'Tracing: none')."""

    def __init__(self, capacity: int = 4096) -> None:
        self._events: Deque[dict] = collections.deque(maxlen=capacity)
        self._lock = threading.Lock()

    def event(self, kind: str, **fields) -> None:
        with self._lock:
            self._events.append({"ts": time.time(), "kind": kind, **fields})

    def span(self, kind: str, **fields):
        tracer = self

        class _Span:
            def __enter__(self):
                self.t0 = time.monotonic()
                return self

            def __exit__(self, *exc):
                tracer.event(kind, duration_s=time.monotonic() - self.t0,
                             error=bool(exc[0]), **fields)
                return False

        return _Span()

    def dump(self) -> List[dict]:
        with self._lock:
            return list(self._events)


def serve_http(metrics: Metrics, port: int, ready_fn=None) -> "object":
    """Serve /metrics, /healthz, /readyz (reference: probe endpoints at
    cmd/controller/main.go:143-150, metrics server :61). Returns the server;
    call .shutdown() to stop."""
    import http.server
    import socketserver

    class Handler(http.server.BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path.startswith("/metrics"):
                body = metrics.export_text().encode()
                self.send_response(200)
                self.send_header("Content-Type", "text/plain; version=0.0.4")
            elif self.path.startswith("/healthz"):
                body = b"ok"
                self.send_response(200)
                self.send_header("Content-Type", "text/plain")
            elif self.path.startswith("/readyz"):
                ready = ready_fn() if ready_fn else True
                body = b"ok" if ready else b"not ready"
                self.send_response(200 if ready else 503)
                self.send_header("Content-Type", "text/plain")
            else:
                body = b"not found"
                self.send_response(404)
                self.send_header("Content-Type", "text/plain")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):  # quiet
            pass

    srv = socketserver.ThreadingTCPServer(("0.0.0.0", port), Handler,
                                          bind_and_activate=True)
    srv.daemon_threads = True
    t = threading.Thread(target=srv.serve_forever, daemon=True,
                         name=f"metrics-http-{port}")
    t.start()
    return srv
