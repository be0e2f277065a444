#!/usr/bin/env python3
"""A/B: prefix caching on a shared-system-prompt serving workload.

Real consensus serving usually pins a long system prompt (and the chat
template header) across requests; the prefix cache should remove its
re-prefill entirely after the first request. Measures end-to-end req/s and
prefill token counts with caching on vs off on the same engine config.

Usage (GPU box): python scripts/bench_prefix.py [--model llama-3-8b]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

SYSTEM = ("You are a meticulous data-extraction assistant. Follow the output schema exactly. "
          "Rules: " + " ".join(f"rule {i}: always check field {i} twice." for i in range(60)))


def run(enable_cache: bool, model: str, steps: int, batch: int, n: int, on_gpu: bool):
    from kllms_amd import KLLMs

    kw = dict(model=model, device="cuda:0" if on_gpu else "cpu",
              use_hip_graphs=on_gpu, seed=0, default_max_new_tokens=32,
              max_seq_len=8192, enable_prefix_caching=enable_cache)
    if not on_gpu:
        kw["max_kv_blocks"] = 4096
    k = KLLMs(**kw)
    eng = k.client.engine
    t_total = 0.0
    done = 0
    for s in range(steps + 1):  # step 0 = warmup (cold cache)
        t0 = time.perf_counter()
        for i in range(batch):
            k.chat.completions.create(
                model=model,
                messages=[{"role": "system", "content": SYSTEM},
                          {"role": "user", "content": f"Extract entities from record {s}-{i}."}],
                n=n, temperature=0.8, max_tokens=32, seed=100 * s + i)
        if on_gpu:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        if s > 0:
            t_total += dt
            done += batch
    stats = eng.prefix_cache.stats if eng.prefix_cache else {}
    return {"cache": enable_cache, "req_per_s": round(done / t_total, 3),
            "s_per_req": round(t_total / done, 4), "prefix_stats": stats}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--n", type=int, default=5)
    args = ap.parse_args()
    on_gpu = torch.cuda.is_available()
    if not on_gpu:
        args.model = "tiny-llama"
        global SYSTEM
        SYSTEM = SYSTEM[:260]  # tiny preset's 512-token context window
    off = run(False, args.model, args.steps, args.batch, args.n, on_gpu)
    on = run(True, args.model, args.steps, args.batch, args.n, on_gpu)
    speedup = round(on["req_per_s"] / off["req_per_s"], 3)
    print(json.dumps({"model": args.model, "off": off, "on": on, "speedup": speedup}))


if __name__ == "__main__":
    main()
