"""Chunked-prefill tail-latency A/B.

Short consensus requests stream through the continuous-batching scheduler
while LONG prompts (default 2800 tokens) are periodically injected. Without
chunked prefill, each long prompt's packed prefill stalls every running
decode stream for its whole duration — short-request p95 latency spikes.
With `prefill_chunk_tokens`, the long prefill advances in slices between
decode steps.

    python scripts/bench_chunked.py --model llama-3-8b --chunk 0     # off
    python scripts/bench_chunked.py --model llama-3-8b --chunk 512   # on

Prints one JSON-ish dict: short-request p50/p95/max latency + long-request
latency + aggregate req/s.

Scale note: on MI355X an 8B packed prefill runs ~17 us/token (profiles/), so
a 2.8k-token prompt stalls decode for only ~50 ms — chunking pays at much
longer prompts (16k+) or bigger models, and costs a little extra arithmetic
per slice otherwise. On CPU (torch oracle path) the per-row decode forward is
slow, so CPU runs of this script exaggerate the chunked cost.
"""

import argparse
import asyncio
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kllms_amd import AsyncKLLMs  # noqa: E402


async def short_worker(client, model, wid, stop, latencies):
    i = 0
    while not stop.is_set():
        t0 = time.perf_counter()
        await client.chat.completions.create(
            messages=[{"role": "user", "content": f"w{wid} r{i}: summarize. "
                       + "The quick brown fox jumps over the lazy dog. " * 4}],
            model=model, n=3, temperature=0.8, max_tokens=24, seed=1000 * wid + i,
        )
        latencies.append(time.perf_counter() - t0)
        i += 1


async def long_injector(client, model, n_long, long_words, latencies):
    text = "entity extraction over a long document. " + ("lorem ipsum dolor sit amet " * (long_words // 5))
    for i in range(n_long):
        t0 = time.perf_counter()
        await client.chat.completions.create(
            messages=[{"role": "user", "content": text}],
            model=model, n=2, temperature=0.8, max_tokens=16, seed=77 + i,
        )
        latencies.append(time.perf_counter() - t0)
        await asyncio.sleep(0.3)


async def main_async(args):
    client = AsyncKLLMs(
        model=args.model,
        max_seq_len=4096,
        prefill_chunk_tokens=args.chunk or None,
        **({} if os.environ.get("KLLMS_GPU", "1") == "1" else {"max_kv_blocks": 4096}),
    )
    # warmup: engine build + graph capture on both batch shapes
    await client.chat.completions.create(
        messages=[{"role": "user", "content": "warmup"}], model=args.model, n=3, max_tokens=8)

    short_lat: list = []
    long_lat: list = []
    stop = asyncio.Event()
    t0 = time.perf_counter()
    workers = [asyncio.create_task(short_worker(client, args.model, w, stop, short_lat))
               for w in range(args.workers)]
    await long_injector(client, args.model, args.n_long, args.long_words, long_lat)
    stop.set()
    await asyncio.gather(*workers)
    elapsed = time.perf_counter() - t0

    short_lat.sort()
    p = lambda q: short_lat[min(len(short_lat) - 1, int(q * len(short_lat)))] if short_lat else None
    print({
        "chunk": args.chunk,
        "elapsed_s": round(elapsed, 2),
        "short_requests": len(short_lat),
        "short_p50_s": round(p(0.50), 3),
        "short_p95_s": round(p(0.95), 3),
        "short_max_s": round(short_lat[-1], 3) if short_lat else None,
        "long_requests": len(long_lat),
        "long_mean_s": round(statistics.mean(long_lat), 3) if long_lat else None,
        "req_per_s": round((len(short_lat) + len(long_lat)) / elapsed, 2),
    })


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--chunk", type=int, default=0, help="prefill_chunk_tokens (0 = off)")
    ap.add_argument("--workers", type=int, default=4)
    ap.add_argument("--n-long", type=int, default=6)
    ap.add_argument("--long-words", type=int, default=2100, help="~2800 tokens")
    args = ap.parse_args()
    asyncio.run(main_async(args))


if __name__ == "__main__":
    main()
