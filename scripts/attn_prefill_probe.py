"""Standalone perf probe for the MFMA prefill attention kernel.

Times ops.attn_prefill_varlen on Llama-3-8B head geometry (H=32, KVH=8,
D=128) at several sequence lengths; reports achieved TFLOP/s against the
causal-attention FLOP count 4*H*D*S^2/2 (QK^T + PV). Random bf16 data
(guide §5.4 rule 25). Run on an MI355X box.
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from kllms_amd import ops  # noqa: E402


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    H, KVH, D = 32, 8, 128
    scale = D ** -0.5
    for S, nseq in ((512, 8), (1024, 4), (2048, 2), (4096, 1)):
        T = S * nseq
        torch.manual_seed(0)
        q = torch.randn(T, H, D, dtype=torch.bfloat16, device="cuda") * 0.5
        k = torch.randn(T, KVH, D, dtype=torch.bfloat16, device="cuda") * 0.5
        v = torch.randn(T, KVH, D, dtype=torch.bfloat16, device="cuda") * 0.5
        cu = torch.arange(0, T + 1, S, dtype=torch.int32, device="cuda")
        t = bench(lambda: ops.attn_prefill_varlen(q, k, v, cu, scale))
        flops = nseq * 4 * H * D * (S * S / 2)
        print(f"S={S} x{nseq}: {t * 1e3:.3f} ms  -> {flops / t / 1e12:.1f} TF/s", flush=True)


if __name__ == "__main__":
    main()
