#!/usr/bin/env python3
"""Decode-attention timing: bf16 cache vs fp8 cache with per-row scales.

The fp8 path halves KV bytes read per step; the per-row dequant scales add
one fp32 sidecar read per 128-element row (+3%) and two VALU multiplies per
key — this probe shows the net effect at the flagship bench decode shape
(120 streams = batch 24 x n=5, Llama-3-8B heads, ctx ~576).

Usage (GPU box): python scripts/bench_fp8_decode.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from kllms_amd import ops
from kllms_amd.ops import torch_ref

DEV = "cuda:0"


def bench(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    torch.manual_seed(0)
    B, H, KVH, D, BS = 120, 32, 8, 128, 16
    ctx = 576
    NB = B * ((ctx + BS - 1) // BS) + 8
    scale = D ** -0.5

    q = torch.randn(B, H, D, dtype=torch.bfloat16, device=DEV) * 0.5
    k = torch.randn(NB * BS, KVH, D, dtype=torch.bfloat16, device=DEV) * 0.3
    v = torch.randn(NB * BS, KVH, D, dtype=torch.bfloat16, device=DEV) * 0.3
    slots = torch.arange(NB * BS, device=DEV)
    lens = torch.full((B,), ctx, dtype=torch.int32, device=DEV)
    nblk = (ctx + BS - 1) // BS
    bt = torch.arange(B * nblk, dtype=torch.int32, device=DEV).reshape(B, nblk) % NB

    # bf16 cache
    kc = torch.zeros(NB, KVH, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    ops.store_kv(k, v, kc, vc, slots)
    t_bf16 = bench(lambda: ops.attn_decode_paged(q, kc, vc, bt, lens, scale))

    # fp8 cache + per-row scales
    kc8 = torch.zeros(NB, KVH, BS, D, dtype=torch.float8_e4m3fn, device=DEV)
    vc8 = torch.zeros_like(kc8)
    ks = torch.ones(NB, KVH, BS, device=DEV)
    vs = torch.ones_like(ks)
    ops.store_kv(k, v, kc8, vc8, slots, ks, vs)
    t_fp8 = bench(lambda: ops.attn_decode_paged(q, kc8, vc8, bt, lens, scale, ks, vs))

    # numerics sanity at this shape vs a bf16 oracle
    out8 = ops.attn_decode_paged(q, kc8, vc8, bt, lens, scale, ks, vs)
    oracle = torch_ref.attn_decode_paged(q, kc, vc, bt, lens, scale)
    err = (out8.float() - oracle.float()).abs().max().item()

    # store-kernel cost (per-row amax quantization vs plain bf16 scatter)
    t_store_bf = bench(lambda: ops.store_kv(k[:B], v[:B], kc, vc, slots[:B]), iters=500)
    t_store_f8 = bench(lambda: ops.store_kv(k[:B], v[:B], kc8, vc8, slots[:B], ks, vs), iters=500)

    print(f"decode attn B={B} H={H} KVH={KVH} ctx={ctx}:")
    print(f"  bf16 cache : {t_bf16:8.1f} us")
    print(f"  fp8 cache  : {t_fp8:8.1f} us  ({t_bf16 / t_fp8:.2f}x)  max|err| vs bf16 oracle {err:.4f}")
    print(f"store_kv (T={B} decode step):")
    print(f"  bf16       : {t_store_bf:8.1f} us")
    print(f"  fp8+scales : {t_store_f8:8.1f} us")


if __name__ == "__main__":
    main()
