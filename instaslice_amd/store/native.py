"""Native store daemon wrapper: spawn/stop `instaslice-stored`.

The daemon (store/csrc/stored_main.cpp) speaks the exact netstore protocol,
so NetStoreClient works against it unchanged. Used by the control plane when
wire throughput matters (many agents / sharded controllers) and by the CPU
test tier for protocol-parity checks against MemStore."""

from __future__ import annotations

import os
import subprocess
from typing import Optional

from instaslice_amd.utils import get_logger

STORED_BIN = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                          "bin", "instaslice-stored")


def stored_available() -> bool:
    return os.path.exists(STORED_BIN) and os.access(STORED_BIN, os.X_OK)


class NativeStoreServer:
    """Runs the C++ store daemon as a child process.

    Mirrors StoreServer's start/stop/port surface (no .store attribute —
    state lives in the daemon; use a NetStoreClient)."""

    def __init__(self, port: int = 0, binary: Optional[str] = None,
                 persist_path: Optional[str] = None) -> None:
        self._binary = binary or STORED_BIN
        self._want_port = port
        self._persist_path = persist_path
        self._proc: Optional[subprocess.Popen] = None
        self.port: Optional[int] = None
        self.host = "127.0.0.1"
        self.log = get_logger("netstore.native")

    def start(self) -> "NativeStoreServer":
        cmd = [self._binary, str(self._want_port)]
        if self._persist_path:
            cmd.append(self._persist_path)
        self._proc = subprocess.Popen(
            cmd,
            stdout=subprocess.PIPE, stderr=subprocess.DEVNULL, text=True,
        )
        line = self._proc.stdout.readline().strip()
        if not line.startswith("LISTENING "):
            self.stop()
            raise RuntimeError(
                f"instaslice-stored failed to start (got {line!r})")
        self.port = int(line.split()[1])
        return self

    def stop(self) -> None:
        if self._proc is not None:
            self._proc.terminate()
            try:
                self._proc.wait(timeout=3.0)
            except subprocess.TimeoutExpired:
                self._proc.kill()
                self._proc.wait(timeout=3.0)
            self._proc = None
