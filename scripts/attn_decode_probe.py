"""Standalone perf probe for the paged decode attention kernel.

Sweeps stream count at Llama-3-8B geometry (H=32, KVH=8, D=128, BS=16) and
reports achieved HBM bandwidth against the K+V byte count (the memory-bound
roofline: each block streams its chunk once).
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from kllms_amd import ops  # noqa: E402


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    H, KVH, D, BS = 32, 8, 128, 16
    scale = D ** -0.5
    torch.manual_seed(0)
    for B, ctx in ((40, 576), (120, 576), (240, 576), (40, 2048), (120, 2048)):
        nblk = (ctx + BS - 1) // BS
        NB = B * nblk + 8
        kc = torch.randn(NB, KVH, BS, D, dtype=torch.bfloat16, device="cuda") * 0.3
        vc = torch.randn(NB, KVH, BS, D, dtype=torch.bfloat16, device="cuda") * 0.3
        perm = torch.randperm(NB - 8)[: B * nblk].to(torch.int32)
        bt = perm.view(B, nblk).cuda()
        lens = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
        q = torch.randn(B, H, D, dtype=torch.bfloat16, device="cuda") * 0.5
        t = bench(lambda: ops.attn_decode_paged(q, kc, vc, bt, lens, scale))
        kv_bytes = 2 * B * KVH * ctx * D * 2
        print(f"B={B} ctx={ctx}: {t * 1e6:.1f} us -> {kv_bytes / t / 1e12:.2f} TB/s", flush=True)


if __name__ == "__main__":
    main()
