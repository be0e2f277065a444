"""Probe LM-head GEMM algorithm behavior: [B,4096]x[4096,128256] bf16.

The bench profile showed torch F.linear landing on a stream-K Tensile kernel
at ~549 us (3.3x the ~167 us weight-streaming bound) with a paired bf16
workspace fill. This probe times the full GEMM vs column-chunked variants.
Run on an MI355X box: python scripts/lmhead_probe.py
"""

import sys
import time

import torch


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    V, H = 128256, 4096
    W = torch.randn(V, H, dtype=torch.bfloat16, device="cuda") * 0.02
    for B in (40, 80, 120, 160, 240):
        x = torch.randn(B, H, dtype=torch.bfloat16, device="cuda")
        t_full = bench(lambda: torch.nn.functional.linear(x, W))
        results = {"full": t_full}
        for nchunk in (2, 4):
            cs = V // nchunk
            chunks = [W[i * cs:(i + 1) * cs] for i in range(nchunk)]
            out = torch.empty(B, V, dtype=torch.bfloat16, device="cuda")

            def chunked():
                for i, c in enumerate(chunks):
                    out[:, i * cs:(i + 1) * cs] = torch.nn.functional.linear(x, c)

            results[f"chunk{nchunk}"] = bench(chunked)
        # row-split halves (each M/2 may dodge the stream-K pick)
        def rowsplit():
            torch.nn.functional.linear(x[: B // 2], W)
            torch.nn.functional.linear(x[B // 2:], W)
        results["rowsplit"] = bench(rowsplit)
        print(f"B={B}: " + ", ".join(f"{k}={v:.0f}us" for k, v in results.items()), flush=True)


if __name__ == "__main__":
    main()
