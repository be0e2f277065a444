#!/usr/bin/env python3
"""Grouped-MoE kernel μbench: TF/s of moe_gateup / moe_down standalone at
Mixtral shapes (E=8, I=14336, H=4096), tokens spread uniformly.

Usage (GPU box): python scripts/bench_moe.py [--tokens 2048]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from kllms_amd import ops  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens", type=int, default=2048)   # sorted rows (T*k)
    ap.add_argument("--E", type=int, default=8)
    ap.add_argument("--I", type=int, default=14336)
    ap.add_argument("--H", type=int, default=4096)
    ap.add_argument("--iters", type=int, default=10)
    args = ap.parse_args()
    E, I, H, S = args.E, args.I, args.H, args.tokens
    BM = 128
    per = (S // E + BM - 1) // BM * BM
    S_pad = per * E
    torch.manual_seed(0)
    dev = "cuda"
    x = (torch.randn(S, H, dtype=torch.bfloat16, device=dev) * 0.1)
    w_gu = (torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device=dev) * 0.02)
    w_d = (torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) * 0.02)
    pad_off = torch.arange(E + 1, dtype=torch.int32, device=dev) * per
    sorted_ids = torch.full((S_pad,), -1, dtype=torch.int32, device=dev)
    # spread real rows across experts
    for e in range(E):
        n = S // E
        sorted_ids[e * per: e * per + n] = torch.arange(e * n, (e + 1) * n, dtype=torch.int32)
    zeros = torch.zeros(H, dtype=torch.bfloat16, device=dev)
    act = torch.empty(S_pad, I, dtype=torch.bfloat16, device=dev)
    y = torch.empty(S_pad, H, dtype=torch.bfloat16, device=dev)

    def gu():
        ops.moe_gateup(act, x, w_gu, sorted_ids, pad_off, zeros)

    def down():
        ops.moe_down(y, act, w_d, pad_off)

    for fn, name, flops in ((gu, "gateup", 2 * S_pad * 2 * I * H), (down, "down", 2 * S_pad * H * I)):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        print(f"{name}: S_pad={S_pad} {dt*1000:.3f} ms, {flops/dt/1e12:.1f} TF/s")

    # numerics spot-check vs torch for one expert slice
    e = 1
    rows = sorted_ids[e * per: e * per + 8].long()
    ref_gu = torch.nn.functional.linear(x[rows].float(), w_gu[e].float())
    g, u = ref_gu.split([I, I], dim=-1)
    ref_act = (g / (1 + torch.exp(-g))) * u
    d = (act[e * per: e * per + 8].float() - ref_act).abs()
    print("gateup check max abs:", d.max().item())


if __name__ == "__main__":
    main()
