#!/usr/bin/env python3
"""Prefill attention kernel μbench: TF/s at the GQA shape the guide's ladder
quotes (B batch of equal-length seqs, H=32 q-heads, KVH=8, D=128).

Causal FLOPs = 2 (QK^T) + 2 (PV) MACs per (q, k<=q, h, d):
  flops = B * H * D * S*(S+1)/2 * 2 * 2

Usage (GPU box): python scripts/bench_attn_prefill.py [--seq 4096] [--batch 4]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402
import torch  # noqa: E402

from kllms_amd import ops  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seq", type=int, default=4096)
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--check", action="store_true")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    S, B, H, KVH, D = args.seq, args.batch, 32, 8, 128
    torch.manual_seed(0)
    T = S * B
    q = (torch.randn(T, H, D, dtype=torch.bfloat16, device="cuda") * 0.5)
    k = (torch.randn(T, KVH, D, dtype=torch.bfloat16, device="cuda") * 0.5)
    v = (torch.randn(T, KVH, D, dtype=torch.bfloat16, device="cuda") * 0.5)
    cu = torch.tensor([0] + list(np.cumsum([S] * B)), dtype=torch.int32, device="cuda")
    scale = D ** -0.5

    if args.check:
        out = ops.attn_prefill_varlen(q, k, v, cu, scale)
        ref = ops.torch_ref.attn_prefill_varlen(q, k, v, cu.cpu(), scale)
        d = (out.float() - ref.float().cuda()).abs()
        print(f"check: max abs diff {d.max().item():.4e}")

    for _ in range(3):
        ops.attn_prefill_varlen(q, k, v, cu, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        ops.attn_prefill_varlen(q, k, v, cu, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    flops = B * H * D * (S * (S + 1) / 2) * 4
    print(f"S={S} B={B}: {dt*1000:.3f} ms/call, {flops/dt/1e12:.1f} TF/s")


if __name__ == "__main__":
    main()
