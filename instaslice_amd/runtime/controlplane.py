"""Control-plane process: TCP store server + cluster controller, co-located.

In production the controller is its own pod next to the API server; in the
benchmark and the single-node quick start the equivalent is one dedicated
process hosting the StoreServer with the PodController reconciling against
the backing MemStore in-process. Keeping it OUT of the node-agent processes
matters for latency: the reconcile chain is ~10 thread wakeups end-to-end,
and a separate process gives the control plane its own GIL instead of
competing with agent + workload threads.

Used via multiprocessing (bench.py rank 0) or `python -m instaslice_amd
controlplane`. Deliberately imports no torch."""

from __future__ import annotations

import sys
from typing import Optional


def run_control_plane(
    conn,
    policy: str = "packed-fit",
    teardown_grace_s: float = 0.0,
    requeue_no_fit_s: float = 0.05,
    workers: int = 4,
    port: int = 0,
    persist_path: Optional[str] = None,
) -> None:
    """multiprocessing target: start store+controller, report the port over
    `conn`, run until the parent sends anything (or closes the pipe)."""
    # short GIL switch interval: the reconcile path is wakeup-latency bound
    sys.setswitchinterval(0.001)
    from instaslice_amd.controller.reconciler import PodController
    from instaslice_amd.store.memstore import MemStore
    from instaslice_amd.store.netstore import StoreServer

    store = MemStore(persist_path=persist_path)
    server = StoreServer(store=store, port=port).start()
    controller = PodController(
        store,
        policy=policy,
        teardown_grace_s=teardown_grace_s,
        requeue_no_fit_s=requeue_no_fit_s,
        workers=workers,
    )
    controller.start()
    conn.send(server.port)
    try:
        conn.recv()  # blocks until shutdown request or EOF
    except (EOFError, OSError):
        pass
    controller.stop()
    server.stop()
    store.close()
