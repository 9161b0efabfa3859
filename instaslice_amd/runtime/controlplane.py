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


def _tune_gc() -> None:
    """Latency hygiene: gen-0 collections at the default threshold (700)
    pause the reconcile path every few thousand allocations; freeze the
    startup object graph and raise the thresholds so collections are rare
    and cheap. (Daemons keep GC ENABLED — only the cadence changes.)"""
    import gc

    gc.collect()
    gc.freeze()
    gc.set_threshold(50000, 50, 50)


def _run_controller_shard(
    port: int,
    policy: str,
    teardown_grace_s: float,
    requeue_no_fit_s: float,
    workers: int,
    shard_index: int,
    shard_count: int,
    conn,
) -> None:
    """multiprocessing target: one controller shard over TCP."""
    sys.setswitchinterval(0.001)
    _tune_gc()
    from instaslice_amd.controller.reconciler import PodController
    from instaslice_amd.store.netstore import NetStoreClient

    import os

    prof_dir = os.environ.get("INSTASLICE_PROFILE_DIR")
    prof_stop = None
    if prof_dir:
        from instaslice_amd.utils import start_stack_sampler

        os.makedirs(prof_dir, exist_ok=True)
        prof_stop = start_stack_sampler(
            os.path.join(prof_dir, f"shard-{shard_index}.samples"))
    store = NetStoreClient("127.0.0.1", port)
    controller = PodController(
        store,
        policy=policy,
        teardown_grace_s=teardown_grace_s,
        requeue_no_fit_s=requeue_no_fit_s,
        workers=workers,
        shard_index=shard_index,
        shard_count=shard_count,
    )
    controller.start()
    conn.send("ready")
    try:
        conn.recv()
    except (EOFError, OSError):
        pass
    controller.stop()
    store.close()
    if prof_stop is not None:
        prof_stop()


def run_control_plane(
    conn,
    policy: str = "packed-fit",
    teardown_grace_s: float = 0.0,
    requeue_no_fit_s: float = 0.05,
    workers: int = 4,
    port: int = 0,
    persist_path: Optional[str] = None,
    controller_shards: int = 1,
    native_store: Optional[bool] = None,
) -> None:
    """multiprocessing target: start store+controller(s), report the port
    over `conn`, run until the parent sends anything (or closes the pipe).

    controller_shards == 1 (default): the controller shares this process
    with the Python store (in-process MemStore — lowest single-stream
    latency). controller_shards > 1: the store serves alone and K controller
    shard PROCESSES connect over TCP, each owning pods by crc32 hash —
    Python reconciles are GIL-bound, so shards are the scale-out axis
    (one shard saturates near ~400 pods/s).

    native_store (None = auto): run the C++ store daemon
    (instaslice-stored) instead of the Python server whenever the binary is
    built — the Python server's GIL-bound wire handling caps throughput.
    Both support persist_path (checkpoint/resume)."""
    # short GIL switch interval: the reconcile path is wakeup-latency bound
    sys.setswitchinterval(0.001)
    _tune_gc()
    from instaslice_amd.controller.reconciler import PodController
    from instaslice_amd.store.memstore import MemStore
    from instaslice_amd.store.native import NativeStoreServer, stored_available
    from instaslice_amd.store.netstore import NetStoreClient, StoreServer

    if native_store is None:
        # measured: the daemon beats the Python server even for a single
        # rank WITH the controller paying TCP round-trips (p50 2.9 vs 4.9 ms
        # on the dev box) — the Python server's wire handling costs more
        # than the extra hops. Both back persistence (JSON for Python,
        # msgpack snapshots for the daemon).
        native_store = stored_available()
    store = None
    if native_store:
        server = NativeStoreServer(port=port, persist_path=persist_path).start()
    else:
        store = MemStore(persist_path=persist_path)
        server = StoreServer(store=store, port=port).start()
    controller = None
    shard_procs = []
    ctl_client = None
    if controller_shards <= 1:
        if store is None:
            ctl_client = NetStoreClient("127.0.0.1", server.port)
        controller = PodController(
            store if store is not None else ctl_client,
            policy=policy,
            teardown_grace_s=teardown_grace_s,
            requeue_no_fit_s=requeue_no_fit_s,
            workers=workers,
        )
        controller.start()
    else:
        import multiprocessing as mp

        ctx = mp.get_context("spawn")
        for i in range(controller_shards):
            parent_c, child_c = ctx.Pipe()
            p = ctx.Process(
                target=_run_controller_shard,
                args=(server.port, policy, teardown_grace_s, requeue_no_fit_s,
                      2, i, controller_shards, child_c),
                daemon=True, name=f"controller-shard-{i}",
            )
            p.start()
            shard_procs.append((p, parent_c))
        for _, c in shard_procs:
            c.recv()  # wait until every shard's watches are live
    conn.send(server.port)
    try:
        conn.recv()  # blocks until shutdown request or EOF
    except (EOFError, OSError):
        pass
    if controller is not None:
        controller.stop()
    for p, c in shard_procs:
        try:
            c.send("stop")
        except (BrokenPipeError, OSError):
            pass
    for p, _ in shard_procs:
        p.join(timeout=3.0)
        if p.is_alive():
            p.terminate()
    if ctl_client is not None:
        ctl_client.close()
    server.stop()
    if store is not None:
        store.close()
