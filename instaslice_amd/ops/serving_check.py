"""Serving validation payload: a real transformer decode loop inside the
allocated partition (BASELINE.json config 4 — the vLLM-pod analog).

The reference proves a slice works by running a CUDA vectoradd container
inside it (samples/test-pod.yaml); its vLLM sample is a deployment YAML it
never executes. This payload goes further: a random-init GPT-style decoder
(prefill + greedy decode, bf16 on GPU) runs as a CHILD PROCESS under the
pod's exact env contract (ROCR_VISIBLE_DEVICES from the pod ConfigMap), so
"a serving workload runs in the partition the operator carved" is measured,
not simulated. No network: weights are random-init, tokens synthetic —
stated in the output.

    python -m instaslice_amd.ops.serving_check [--layers 4] [--dmodel 512]
        [--prefill 128] [--decode 64]

Prints one JSON line: tokens/s decode, prefill latency, device, visible
devices. Exit 0 iff logits are finite and decode produced the requested
number of tokens.
"""

from __future__ import annotations

import argparse
import json
import math
import os
import sys
import time


def build_model(torch, vocab: int, d: int, n_layers: int, n_heads: int,
                device, dtype):
    """Minimal pre-LN GPT decoder in plain torch ops (eager ROCm path)."""
    import torch.nn as nn

    class Block(nn.Module):
        def __init__(self):
            super().__init__()
            self.ln1 = nn.LayerNorm(d)
            self.attn = nn.MultiheadAttention(d, n_heads, batch_first=True)
            self.ln2 = nn.LayerNorm(d)
            self.mlp = nn.Sequential(
                nn.Linear(d, 4 * d), nn.GELU(), nn.Linear(4 * d, d))

        def forward(self, x, mask):
            a, _ = self.attn(self.ln1(x), self.ln1(x), self.ln1(x),
                             attn_mask=mask, need_weights=False)
            x = x + a
            return x + self.mlp(self.ln2(x))

    class Tiny(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb = nn.Embedding(vocab, d)
            self.pos = nn.Embedding(4096, d)
            self.blocks = nn.ModuleList(Block() for _ in range(n_layers))
            self.ln = nn.LayerNorm(d)
            self.head = nn.Linear(d, vocab, bias=False)

        def forward(self, ids):
            t = ids.shape[1]
            x = self.emb(ids) + self.pos.weight[:t]
            mask = torch.full((t, t), float("-inf"), device=ids.device)
            mask = torch.triu(mask, diagonal=1)
            for b in self.blocks:
                x = b(x, mask)
            return self.head(self.ln(x))

    torch.manual_seed(0)
    return Tiny().to(device=device, dtype=dtype).eval()


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--layers", type=int, default=4)
    ap.add_argument("--dmodel", type=int, default=512)
    ap.add_argument("--heads", type=int, default=8)
    ap.add_argument("--vocab", type=int, default=32000)
    ap.add_argument("--prefill", type=int, default=128)
    ap.add_argument("--decode", type=int, default=64)
    args = ap.parse_args()

    import torch

    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", 0) if use_gpu else torch.device("cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32
    model = build_model(torch, args.vocab, args.dmodel, args.layers,
                        args.heads, device, dtype)

    g = torch.Generator().manual_seed(1)
    ids = torch.randint(0, args.vocab, (1, args.prefill), generator=g).to(device)

    with torch.no_grad():
        # warm + prefill timing
        logits = model(ids)
        if use_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        logits = model(ids)
        if use_gpu:
            torch.cuda.synchronize()
        prefill_ms = (time.perf_counter() - t0) * 1000.0

        # greedy decode (no KV cache: recompute — this is a validation
        # payload proving real compute in the partition, not a perf claim)
        t0 = time.perf_counter()
        produced = 0
        for _ in range(args.decode):
            nxt = logits[:, -1:].argmax(dim=-1)
            ids = torch.cat([ids, nxt], dim=1)
            logits = model(ids)
            produced += 1
        if use_gpu:
            torch.cuda.synchronize()
        decode_s = time.perf_counter() - t0

    finite = bool(torch.isfinite(logits.float()).all().item())
    ok = finite and produced == args.decode
    print(json.dumps({
        "payload": "serving_check",
        "ok": ok,
        "finite": finite,
        "device": str(device),
        "dtype": str(dtype).replace("torch.", ""),
        "layers": args.layers,
        "d_model": args.dmodel,
        "prefill_tokens": args.prefill,
        "prefill_ms": round(prefill_ms, 2),
        "decode_tokens": produced,
        "decode_tok_s": round(produced / decode_s, 2) if decode_s > 0 else 0.0,
        "weights": "random-init (no network)",
        "visible": os.environ.get("ROCR_VISIBLE_DEVICES", ""),
    }), flush=True)
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
