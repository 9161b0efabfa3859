#!/usr/bin/env python3
"""Build every native component in-tree for gfx950.

Artifacts (all committed paths are source-only; .so/binaries are .gitignored
but DO travel to the GPU box with the gpurun snapshot):

  instaslice_amd/smi/_partitiond<EXT>   pybind11 module over libamd_smi.so
  instaslice_amd/bin/partitiond         standalone device daemon (stdio JSON)
  instaslice_amd/ops/_payload<EXT>      pybind11 module with gfx950 kernels
  instaslice_amd/bin/instaslice-payload standalone workload binary

Usage: python build_native.py [--force]
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
ROCM = Path(os.environ.get("ROCM_PATH", "/opt/rocm"))
HIPCC = str(ROCM / "bin" / "hipcc")
CLANGXX = str(ROCM / "bin" / "amdclang++")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
EXT = sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def pybind_includes() -> list:
    import pybind11

    return [f"-I{pybind11.get_include()}", f"-I{sysconfig.get_paths()['include']}"]


def _srchash(sources: list, cmd_sig: str) -> str:
    """Content hash of the sources + compile signature. mtime freshness was
    dropped (advisor r1): a stale artifact with a newer mtime could silently
    masquerade as a fresh build; a content hash is deterministic."""
    import hashlib

    h = hashlib.sha256()
    h.update(cmd_sig.encode())
    for s in sources:
        h.update(Path(s).read_bytes())
    return h.hexdigest()


def newer(target: Path, sources: list, cmd_sig: str = "") -> bool:
    """True iff `target` was built from exactly these source bytes."""
    if not target.exists():
        return False
    sidecar = target.with_suffix(target.suffix + ".srchash")
    if not sidecar.exists():
        return False
    return sidecar.read_text().strip() == _srchash(sources, cmd_sig)


def record_hash(target: Path, sources: list, cmd_sig: str = "") -> None:
    sidecar = target.with_suffix(target.suffix + ".srchash")
    sidecar.write_text(_srchash(sources, cmd_sig) + "\n")


def run(cmd: list) -> None:
    print("+", " ".join(str(c) for c in cmd), flush=True)
    subprocess.run([str(c) for c in cmd], check=True, cwd=ROOT)


def build(force: bool = False) -> None:
    smi_dir = ROOT / "instaslice_amd" / "smi"
    ops_dir = ROOT / "instaslice_amd" / "ops"
    bin_dir = ROOT / "instaslice_amd" / "bin"
    bin_dir.mkdir(exist_ok=True)
    (bin_dir / "__init__.py").touch()

    common = ["-O2", "-std=c++17", "-fPIC", f"-I{ROCM}/include",
              f"-L{ROCM}/lib", f"-Wl,-rpath,{ROCM}/lib"]

    # 1. _partitiond pybind module (host-only C++, links libamd_smi)
    src = smi_dir / "csrc" / "partitiond_pybind.cpp"
    hdr = smi_dir / "csrc" / "partitiond_core.hpp"
    out = smi_dir / f"_partitiond{EXT}"
    if force or not newer(out, [src, hdr], "pybind"):
        run([CLANGXX, "-shared", *common, *pybind_includes(), src, "-lamd_smi", "-o", out])
        record_hash(out, [src, hdr], "pybind")

    # 2. partitiond standalone daemon
    src = smi_dir / "csrc" / "partitiond_main.cpp"
    out = bin_dir / "partitiond"
    if force or not newer(out, [src, hdr], "daemon"):
        run([CLANGXX, *common, src, "-lamd_smi", "-o", out])
        record_hash(out, [src, hdr], "daemon")

    # 3. _payload pybind module (gfx950 device code)
    src = ops_dir / "csrc" / "payload_pybind.hip"
    hdr = ops_dir / "csrc" / "payload_core.hpp"
    out = ops_dir / f"_payload{EXT}"
    if force or not newer(out, [src, hdr], ARCH):
        run([HIPCC, f"--offload-arch={ARCH}", "-shared", *common,
             *pybind_includes(), src, "-o", out])
        record_hash(out, [src, hdr], ARCH)

    # 4. instaslice-payload standalone workload binary
    src = ops_dir / "csrc" / "payload_main.hip"
    out = bin_dir / "instaslice-payload"
    if force or not newer(out, [src, hdr], ARCH):
        run([HIPCC, f"--offload-arch={ARCH}", *common, src, "-o", out])
        record_hash(out, [src, hdr], ARCH)

    # 5. instaslice-stored: native store daemon (plain host C++, g++ —
    # no ROCm dependency; runs on any node incl. the CPU test tier)
    store_dir = ROOT / "instaslice_amd" / "store"
    src = store_dir / "csrc" / "stored_main.cpp"
    hdr2 = store_dir / "csrc" / "msgpack_value.hpp"
    out = bin_dir / "instaslice-stored"
    if force or not newer(out, [src, hdr2], "stored"):
        run(["g++", "-O2", "-std=c++17", "-pthread", src, "-o", out])
        record_hash(out, [src, hdr2], "stored")

    print("native build complete")


if __name__ == "__main__":
    build(force="--force" in sys.argv)
