"""Concurrent-client benchmark — parity with the reference's
"Concurrent (5 workers)" row (README_TESTS.md:214: ~2.4 req/s, ~2.1 s
latency against the OpenAI API).

W async workers each issue sequential n-way consensus requests against one
AsyncKLLMs client; the continuous-batching scheduler merges their streams
into shared decode batches. Reports aggregate req/s and mean latency.

    python scripts/bench_concurrent.py --model llama-3-8b --workers 5
"""

import argparse
import asyncio
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kllms_amd import AsyncKLLMs  # noqa: E402


async def worker(client, model, wid, n_requests, n, max_tokens, latencies):
    for i in range(n_requests):
        t0 = time.perf_counter()
        await client.chat.completions.create(
            messages=[{"role": "user", "content": f"worker {wid} request {i}: extract the entities. "
                       + "The quick brown fox jumps over the lazy dog. " * 8}],
            model=model, n=n, temperature=0.8, max_tokens=max_tokens, seed=1000 * wid + i,
        )
        latencies.append(time.perf_counter() - t0)


async def main_async(args):
    client = AsyncKLLMs(
        model=args.model, default_max_new_tokens=args.max_new,
        max_seq_len=1024, **({} if os.environ.get("KLLMS_GPU", "1") == "1" else {"max_kv_blocks": 2048}),
    )
    # warmup (engine build + graph capture)
    await client.chat.completions.create(
        messages=[{"role": "user", "content": "warmup"}], model=args.model, n=args.n, max_tokens=8,
    )

    latencies: list = []
    t0 = time.perf_counter()
    await asyncio.gather(*[
        worker(client, args.model, w, args.requests_per_worker, args.n, args.max_new, latencies)
        for w in range(args.workers)
    ])
    elapsed = time.perf_counter() - t0
    total = args.workers * args.requests_per_worker
    sched = client.client._scheduler
    print({
        "workers": args.workers,
        "total_requests": total,
        "elapsed_s": round(elapsed, 2),
        "req_per_s": round(total / elapsed, 3),
        "mean_latency_s": round(sum(latencies) / len(latencies), 4),
        "n": args.n,
        "scheduler_batches": getattr(sched, "admitted_batches", None),
        "scheduler_steps": getattr(sched, "steps", None),
    })


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--workers", type=int, default=5)
    ap.add_argument("--requests-per-worker", type=int, default=4)
    ap.add_argument("--n", type=int, default=3)
    ap.add_argument("--max-new", type=int, default=64)
    args = ap.parse_args()
    asyncio.run(main_async(args))


if __name__ == "__main__":
    main()
