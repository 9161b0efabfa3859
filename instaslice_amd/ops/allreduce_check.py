"""RCCL allreduce validation payload: the cross-partition e2e check.

SURVEY.md §5: "A RCCL allreduce micro-benchmark across two CPX partitions is
the right e2e validation payload for partition isolation + xGMI routing."
This script is that payload: launched with torchrun, one rank per partition,
each rank confined by the ROCR_VISIBLE_DEVICES its pod ConfigMap delivers.

    torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 \
        -m instaslice_amd.ops.allreduce_check [--numel N] [--iters K]

Each rank fills a tensor with (rank+1); after allreduce(SUM) every element
must equal world*(world+1)/2 exactly (integers in fp32: exact). On GPU the
process group is "nccl" (RCCL over xGMI); on CPU it falls back to gloo so the
same script is CI-testable here. Rank 0 prints one JSON line with the
verification verdict and measured bus bandwidth.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--numel", type=int, default=1 << 22)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--backend", choices=["auto", "gloo", "nccl"],
                    default="auto",
                    help="force a backend (the CPU test tier runs gloo "
                         "even on a machine that has a GPU)")
    args = ap.parse_args()

    import torch
    import torch.distributed as dist

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if args.backend == "auto":
        use_gpu = torch.cuda.is_available()
        backend = "nccl" if use_gpu else "gloo"
    else:
        backend = args.backend
        use_gpu = backend == "nccl"

    if world > 1:
        dist.init_process_group(backend=backend)
    if use_gpu:
        # the partition is whatever ROCR_VISIBLE_DEVICES exposes: device 0
        # of THIS process (a pod sees exactly one partition)
        torch.cuda.set_device(0)
        device = torch.device("cuda", 0)
    else:
        device = torch.device("cpu")

    x = torch.full((args.numel,), float(rank + 1), device=device)
    expect = world * (world + 1) / 2.0

    def allreduce():
        if world > 1:
            dist.all_reduce(x)
        return x

    # correctness first
    allreduce()
    if use_gpu:
        torch.cuda.synchronize()
    max_err = (x - expect).abs().max().item()
    ok = max_err == 0.0

    # bandwidth (ring allreduce moves 2*(world-1)/world * bytes per rank)
    x.fill_(float(rank + 1))
    if world > 1:
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        if world > 1:
            dist.all_reduce(x)
    if use_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    dt = time.perf_counter() - t0
    bytes_per = x.numel() * 4 * 2 * max(world - 1, 0) / max(world, 1)
    busbw_gbs = bytes_per * args.iters / dt / 1e9 if world > 1 and dt > 0 else 0.0

    if rank == 0:
        print(json.dumps({
            "payload": "allreduce_check",
            "ok": ok,
            "max_err": max_err,
            "world": world,
            "backend": backend,
            "device": str(device),
            "numel": args.numel,
            "busbw_gb_s": round(busbw_gbs, 2),
            "visible": os.environ.get("ROCR_VISIBLE_DEVICES", ""),
        }), flush=True)
    if world > 1:
        dist.destroy_process_group()
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
