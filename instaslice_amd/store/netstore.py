"""TCP-served store: MemStore behind a length-prefixed msgpack protocol.

The reference's coordination bus is the Kubernetes API server; ours is this
store. In-process callers use MemStore directly; multi-process deployments
(the N-rank benchmark: one controller, N node agents, one per GPU) talk to a
StoreServer over TCP with the same verb set, including watches.

Framing: 4-byte big-endian length + one msgpack map per message (msgpack
encodes/decodes the ~3 KB Instaslice CR ~3x faster than JSON — the control
plane is wire-handling bound under load, measured in profiles/).

  request:   {"id": 1, "verb": "get", "kind": "Pod", "name": ..., "namespace": ...}
  response:  {"id": 1, "ok": True, "result": {...}}
             {"id": 1, "ok": False, "error": {"type": "NotFound", "msg": ...}}
  watch:     request {"id": 2, "verb": "watch", "kind": "Pod",
             "filters": [...]} -> ok with {"watch_id": N}; events then stream
             as {"watch_id": N, "event": ["ADDED", {...}]}
  quiet:     a request with "quiet": True gets result=None on success —
             writers that ignore the echoed object (agents) skip the ~3 KB
             re-encode per mutation.

Exceptions round-trip by name so Conflict/NotFound semantics (and therefore
update_with_retry) behave identically to the in-memory store.
"""

from __future__ import annotations

import queue
import socket
import socketserver
import struct
import threading
from typing import Callable, List, Optional, Tuple

import msgpack

from instaslice_amd.store.memstore import (
    AlreadyExists,
    Conflict,
    MemStore,
    NotFound,
)
from instaslice_amd.utils import get_logger

_EXC_BY_NAME = {
    "Conflict": Conflict,
    "NotFound": NotFound,
    "AlreadyExists": AlreadyExists,
}

_LEN = struct.Struct(">I")


def _pack(obj: dict) -> bytes:
    payload = msgpack.packb(obj, use_bin_type=True)
    return _LEN.pack(len(payload)) + payload


def _read_msg(rfile) -> Optional[dict]:
    """One framed message from a buffered binary file; None on EOF."""
    hdr = rfile.read(4)
    if len(hdr) < 4:
        return None
    (n,) = _LEN.unpack(hdr)
    payload = rfile.read(n)
    if len(payload) < n:
        return None
    return msgpack.unpackb(payload, raw=False)


def _execute(store: MemStore, req: dict):
    """Dispatch one request dict onto the store (shared by the single-verb
    path and batch)."""
    verb = req.get("verb")
    if verb == "create":
        return store.create(req["obj"])
    if verb == "get":
        return store.get(req["kind"], req["name"], req.get("namespace", ""))
    if verb == "list":
        return store.list(req["kind"], req.get("namespace"))
    if verb == "update":
        return store.update(req["obj"])
    if verb == "delete":
        store.delete(req["kind"], req["name"], req.get("namespace", ""))
        return None
    if verb == "patch":
        return store.patch(req["kind"], req["name"], req.get("namespace", ""),
                           req.get("ops"))
    if verb == "ping":
        return "pong"
    raise ValueError(f"unknown verb {verb!r}")


class _Handler(socketserver.StreamRequestHandler):
    # small request/reply + async watch pushes on one socket: Nagle + delayed
    # ACK would add 40 ms stalls (observed as p99 spikes) — disable it
    disable_nagle_algorithm = True

    def handle(self) -> None:
        store: MemStore = self.server.store  # type: ignore[attr-defined]
        log = self.server.log  # type: ignore[attr-defined]
        wlock = threading.Lock()
        watches = []

        def send(obj: dict) -> None:
            data = _pack(obj)
            with wlock:
                self.wfile.write(data)
                self.wfile.flush()

        def pump_watch(watch_id: int, w) -> None:
            while True:
                ev = w.next(timeout=0.5)
                if ev is None:
                    if w._stopped or self.server.stopping:  # type: ignore[attr-defined]
                        return
                    continue
                try:
                    send({"watch_id": watch_id, "event": list(ev)})
                except (BrokenPipeError, OSError, ValueError):
                    # ValueError: wfile closed by the handler thread while a
                    # severed client tears down — same fate as a broken pipe
                    w.stop()
                    return

        next_watch_id = 0
        try:
            while True:
                req = _read_msg(self.rfile)
                if req is None:
                    break
                rid = req.get("id")
                verb = req.get("verb")
                quiet = req.get("quiet", False)
                try:
                    if verb == "batch":
                        # quiet: per-entry ok/error survives; payloads dropped
                        res = store.batch(req["requests"], quiet=quiet)
                    elif verb == "watch":
                        w = store.watch(req.get("kind"), replay=req.get("replay", True),
                                        filters=req.get("filters"),
                                        since=req.get("since"))
                        watches.append(w)
                        next_watch_id += 1
                        wid = next_watch_id
                        # respond BEFORE pumping so the id precedes any event
                        # on the wire (the client still buffers orphans in
                        # case its caller hasn't registered the id yet)
                        send({"id": rid, "ok": True, "result": {
                            "watch_id": wid,
                            "rev": getattr(w, "rev", None),
                            "resumed": getattr(w, "resumed", False)}})
                        threading.Thread(
                            target=pump_watch, args=(wid, w), daemon=True
                        ).start()
                        continue
                    else:
                        res = _execute(store, req)
                        if quiet:
                            res = None
                    send({"id": rid, "ok": True, "result": res})
                except (Conflict, NotFound, AlreadyExists) as e:
                    send({"id": rid, "ok": False,
                          "error": {"type": type(e).__name__, "msg": str(e)}})
                except Exception as e:  # malformed request: report, keep serving
                    log.warning("request error: %s", e)
                    send({"id": rid, "ok": False,
                          "error": {"type": "Error", "msg": str(e)}})
        except (ConnectionResetError, BrokenPipeError, OSError):
            pass
        finally:
            for w in watches:
                w.stop()


class StoreServer:
    """Serve a MemStore on host:port. port=0 picks a free port."""

    def __init__(self, store: Optional[MemStore] = None,
                 host: str = "127.0.0.1", port: int = 0) -> None:
        self.store = store or MemStore()
        self.log = get_logger("netstore.server")
        self._srv = socketserver.ThreadingTCPServer((host, port), _Handler,
                                                    bind_and_activate=True)
        self._srv.daemon_threads = True
        self._srv.allow_reuse_address = True
        self._srv.store = self.store  # type: ignore[attr-defined]
        self._srv.log = self.log  # type: ignore[attr-defined]
        self._srv.stopping = False  # type: ignore[attr-defined]
        self.host, self.port = self._srv.server_address
        self._thread = threading.Thread(target=self._srv.serve_forever,
                                        name="netstore-server", daemon=True)

    def start(self) -> "StoreServer":
        self._thread.start()
        return self

    def stop(self) -> None:
        self._srv.stopping = True  # type: ignore[attr-defined]
        self._srv.shutdown()
        self._srv.server_close()


class _ClientWatch:
    """Client-side watch mirroring memstore.Watch (incl. push-mode
    set_callback — see memstore.Watch.set_callback for the contract)."""

    def __init__(self) -> None:
        self._q: "queue.Queue[Optional[Tuple[str, dict]]]" = queue.Queue()
        self._stopped = False
        self.kind: Optional[str] = None
        self._cb = None
        self._cb_lock = threading.Lock()
        # highest event resourceVersion seen — the watch RESUME TOKEN a
        # reconnect passes as `since` so missed events replay without a
        # full relist (None until the subscribe response seeds it)
        self.last_rev: Optional[int] = None

    def set_callback(self, fn) -> None:
        with self._cb_lock:
            while True:
                try:
                    ev = self._q.get_nowait()
                except queue.Empty:
                    break
                if ev is not None:
                    fn(*ev)
            self._cb = fn

    def _deliver(self, event) -> None:
        if self._stopped:
            return
        try:
            rv = int(event[1]["metadata"]["resourceVersion"])
            if self.last_rev is None or rv > self.last_rev:
                self.last_rev = rv
        except (KeyError, TypeError, ValueError, IndexError):
            pass
        with self._cb_lock:
            if self._cb is not None:
                try:
                    self._cb(*event)
                except Exception:  # noqa: BLE001 - consumer bug must not kill reader
                    import traceback

                    traceback.print_exc()
                return
        self._q.put(event)

    def next(self, timeout: Optional[float] = None) -> Optional[Tuple[str, dict]]:
        try:
            return self._q.get(timeout=timeout)
        except queue.Empty:
            return None

    def stop(self) -> None:
        self._stopped = True
        self._q.put(None)


class NetStoreClient:
    """Store client with the MemStore interface (duck-typed).

    With reconnect=True (long-running daemons), a lost connection is
    re-established with exponential backoff and every live watch is
    re-subscribed WITH REPLAY — the k8s informer-resync analog. Consumers
    are level-triggered reconcilers, so the duplicate ADDED events a resync
    produces are harmless; in-flight calls during the outage fail with
    ConnectionError and the engines' error backoff retries them."""

    def __init__(self, host: str, port: int, timeout: float = 30.0,
                 reconnect: bool = False,
                 reconnect_backoff_s: float = 0.2) -> None:
        self.host, self.port = host, port
        self._reconnect = reconnect
        self._reconnect_backoff_s = reconnect_backoff_s
        self._sock = socket.create_connection((host, port), timeout=timeout)
        self._sock.settimeout(None)
        self._sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self._rfile = self._sock.makefile("rb")
        self._wlock = threading.Lock()
        self._pending: dict = {}
        self._watches: dict = {}
        # wid -> (watch, request_kwargs) for reconnect re-subscription
        self._watch_specs: dict = {}
        # events that arrived before watch() registered their id (the reader
        # thread outruns the caller between response and registration);
        # _watch_reg_lock makes register-vs-deliver atomic
        self._orphan_events: dict = {}
        self._watch_reg_lock = threading.Lock()
        self._next_id = 0
        self._idlock = threading.Lock()
        self._closed = False
        self.log = get_logger("netstore.client")
        self._reader = threading.Thread(target=self._read_loop,
                                        name="netstore-reader", daemon=True)
        self._reader.start()

    def _dispatch(self, msg: dict) -> None:
        if "watch_id" in msg and "event" in msg:
            wid = msg["watch_id"]
            et, obj = msg["event"]
            with self._watch_reg_lock:
                w = self._watches.get(wid)
                if w is None:
                    self._orphan_events.setdefault(wid, []).append((et, obj))
            if w is not None:
                w._deliver((et, obj))
            return
        ev = self._pending.pop(msg.get("id"), None)
        if ev is not None:
            ev[1] = msg
            ev[0].set()

    def _fail_pending(self, why: str) -> None:
        for rid in list(self._pending):
            ev = self._pending.pop(rid, None)
            if ev is not None:
                ev[1] = {"ok": False, "error": {"type": "Error", "msg": why}}
                ev[0].set()

    def _read_loop(self) -> None:
        while True:
            try:
                while True:
                    msg = _read_msg(self._rfile)
                    if msg is None:
                        break
                    self._dispatch(msg)
            except (OSError, ValueError):
                pass
            self._fail_pending("connection lost")
            if self._closed or not self._reconnect:
                break
            if not self._reconnect_once():
                break
        self._closed = True
        self._fail_pending("connection closed")
        for w in self._watches.values():
            w.stop()

    def _reconnect_once(self) -> bool:
        """Re-establish the connection and re-subscribe live watches.
        Runs ON the reader thread (reads its own responses inline)."""
        try:
            self._sock.close()
        except OSError:
            pass
        delay = self._reconnect_backoff_s
        while not self._closed:
            try:
                sock = socket.create_connection((self.host, self.port),
                                                timeout=5.0)
            except OSError:
                import time as _time

                _time.sleep(delay)
                delay = min(delay * 2, 5.0)
                continue
            sock.settimeout(None)
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            with self._wlock:
                self._sock = sock
                self._rfile = sock.makefile("rb")
            self._fail_pending("connection lost")  # sent into the dead socket
            # NOTE: _watch_specs must survive a FAILED resubscribe attempt
            # (e.g. connecting to the dying server's last accept) — it is
            # only replaced after every watch re-registered successfully
            with self._watch_reg_lock:
                live = [(w, spec) for (w, spec) in self._watch_specs.values()
                        if not w._stopped]
                self._watches = {}  # old wids are invalid on the new conn
            new_specs = {}
            ok = True
            for w, spec in live:
                with self._idlock:
                    self._next_id += 1
                    rid = self._next_id
                # resume from the last seen event when possible: the server
                # replays only the missed window from its bounded history,
                # falling back to a full ADDED relist on compaction
                resync = dict(spec, replay=True)
                if w.last_rev is not None:
                    resync["since"] = w.last_rev
                try:
                    with self._wlock:
                        self._sock.sendall(_pack({"id": rid, "verb": "watch",
                                                  **resync}))
                    while True:  # drain events of earlier re-subs inline
                        msg = _read_msg(self._rfile)
                        if msg is None:
                            ok = False
                            break
                        if msg.get("id") == rid:
                            new_wid = msg["result"]["watch_id"]
                            with self._watch_reg_lock:
                                self._watches[new_wid] = w
                            new_specs[new_wid] = (w, spec)
                            break
                        self._dispatch(msg)
                except (OSError, ValueError, KeyError):
                    ok = False
                if not ok:
                    break
            if ok:
                with self._watch_reg_lock:
                    self._watch_specs = new_specs
                self.log.warning(
                    "reconnected to store %s:%d (%d watches resynced)",
                    self.host, self.port, len(live))
                return True
            # failed attempt: drop partial registrations, keep specs, retry
            with self._watch_reg_lock:
                self._watches = {}
        return False

    def _call(self, verb: str, **kw):
        if self._closed:
            raise ConnectionError("netstore client closed")
        with self._idlock:
            self._next_id += 1
            rid = self._next_id
        ev = [threading.Event(), None]
        self._pending[rid] = ev
        req = _pack({"id": rid, "verb": verb, **kw})
        try:
            with self._wlock:
                self._sock.sendall(req)
        except OSError as e:
            self._pending.pop(rid, None)
            raise ConnectionError(f"netstore send failed: {e}") from e
        if not ev[0].wait(timeout=60.0):
            self._pending.pop(rid, None)
            raise TimeoutError(f"netstore call {verb} timed out")
        msg = ev[1]
        if msg["ok"]:
            return msg.get("result")
        err = msg["error"]
        if err["type"] == "Error" and "connection" in err["msg"]:
            raise ConnectionError(err["msg"])
        raise _EXC_BY_NAME.get(err["type"], RuntimeError)(err["msg"])

    # -- MemStore interface -------------------------------------------------

    def create(self, obj: dict) -> dict:
        return self._call("create", obj=obj)

    def get(self, kind: str, name: str, namespace: str = "") -> dict:
        return self._call("get", kind=kind, name=name, namespace=namespace)

    def list(self, kind: str, namespace: Optional[str] = None) -> List[dict]:
        return self._call("list", kind=kind, namespace=namespace)

    def update(self, obj: dict) -> dict:
        return self._call("update", obj=obj)

    def delete(self, kind: str, name: str, namespace: str = "", *, now: float = 0.0) -> None:
        self._call("delete", kind=kind, name=name, namespace=namespace)

    def patch(self, kind: str, name: str, namespace: str = "",
              ops: Optional[List[dict]] = None, *, quiet: bool = False):
        return self._call("patch", kind=kind, name=name, namespace=namespace,
                          ops=ops, quiet=quiet)

    def batch(self, requests: List[dict], *, quiet: bool = False) -> List[dict]:
        return self._call("batch", requests=requests, quiet=quiet)

    def watch(self, kind: Optional[str] = None, *, replay: bool = True,
              filters: Optional[List[dict]] = None):
        w = _ClientWatch()
        w.kind = kind
        spec = {"kind": kind, "replay": replay, "filters": filters}
        res = self._call("watch", kind=kind, replay=replay, filters=filters)
        wid = res["watch_id"]
        # resume-token baseline: the store revision at subscribe time (a
        # reconnect before any event then resumes from here, not relists)
        if isinstance(res, dict) and res.get("rev") is not None:
            w.last_rev = int(res["rev"])
        with self._watch_reg_lock:
            self._watches[wid] = w
            self._watch_specs[wid] = (w, spec)
            orphans = self._orphan_events.pop(wid, [])
        # deliver events that raced ahead of this registration, in order
        for ev in orphans:
            w._deliver(ev)
        return w

    def update_with_retry(
        self, kind: str, name: str, namespace: str,
        mutate: Callable[[dict], Optional[dict]], attempts: int = 25,
    ) -> Optional[dict]:
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
        raise Conflict(f"update_with_retry: {attempts} attempts exhausted for {kind}/{name}")

    def close(self) -> None:
        self._closed = True
        try:
            self._sock.close()
        except OSError:
            pass
