"""Structured-output benchmark — BASELINE config 3: parse() with a Pydantic
schema + likelihoods at n=16, Llama-3-8B TP=1.

Reference comparison row (README_TESTS.md:213, OpenAI API): structured output
~1.5 s avg latency, ~0.7 req/s.

    python scripts/bench_structured.py --model llama-3-8b --n 16
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from pydantic import BaseModel, Field  # noqa: E402

from kllms_amd import KLLMs  # noqa: E402


# Field bounds flow into the constrained-decode FSM (maxLength/maxItems), so
# even random-weight sampling terminates inside the token budget.
class LineItem(BaseModel):
    description: str = Field(max_length=20)
    quantity: int
    unit_price: float


class Invoice(BaseModel):
    vendor: str = Field(max_length=20)
    invoice_number: str = Field(max_length=10)
    total: float
    paid: bool
    items: list[LineItem] = Field(max_length=2)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--n", type=int, default=16)
    ap.add_argument("--requests", type=int, default=6)
    ap.add_argument("--max-new", type=int, default=200)
    args = ap.parse_args()

    extra = {} if os.environ.get("KLLMS_GPU", "1") == "1" else {"max_kv_blocks": 2048}
    client = KLLMs(model=args.model, default_max_new_tokens=args.max_new, max_seq_len=2048, **extra)

    # warmup (engine build, FSM table compile, graph capture)
    client.chat.completions.parse(
        messages=[{"role": "user", "content": "warmup"}],
        model=args.model, response_format=Invoice, n=args.n, max_tokens=args.max_new, seed=0,
    )

    latencies = []
    n_valid = 0
    n_total = 0
    qualities = []
    t0 = time.perf_counter()
    for i in range(args.requests):
        t1 = time.perf_counter()
        r = client.chat.completions.parse(
            messages=[{"role": "user", "content": f"Invoice #{i}: ACME Corp, 3 widgets at $9.99, paid."}],
            model=args.model, response_format=Invoice, n=args.n,
            max_tokens=args.max_new, temperature=0.9, seed=100 + i,
        )
        latencies.append(time.perf_counter() - t1)
        n_total += args.n
        n_valid += sum(1 for c in r.choices[1:] if c.message.parsed is not None)
        if r.likelihoods:
            leaves = []

            def walk(v):
                if isinstance(v, dict):
                    [walk(x) for x in v.values()]
                elif isinstance(v, (list, tuple)):
                    [walk(x) for x in v]
                elif isinstance(v, (int, float)):
                    leaves.append(float(v))

            walk(r.likelihoods)
            if leaves:
                qualities.append(sum(leaves) / len(leaves))
    elapsed = time.perf_counter() - t0

    print({
        "n": args.n,
        "requests": args.requests,
        "req_per_s": round(args.requests / elapsed, 3),
        "mean_latency_s": round(sum(latencies) / len(latencies), 4),
        "schema_valid_completions": f"{n_valid}/{n_total}",
        "mean_consensus_quality": round(sum(qualities) / len(qualities), 4) if qualities else None,
    })


if __name__ == "__main__":
    main()
