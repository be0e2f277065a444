"""kllms_amd demo: consensus completions, structured outputs, embeddings.

Runs on CPU with the tiny preset (seconds) or on an MI355X with
--model llama-3-8b. Mirrors the reference's example scenarios
(README_TESTS.md: basic completions, consensus, structured outputs,
temperature effects, error handling).
"""

import argparse
import asyncio
import os
import sys

from pydantic import BaseModel

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kllms_amd import AsyncKLLMs, KLLMs  # noqa: E402


class Invoice(BaseModel):
    vendor: str
    total: float
    paid: bool


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="tiny-llama")
    args = ap.parse_args()

    client = KLLMs(model=args.model, max_kv_blocks=512, default_max_new_tokens=32)

    print("== 1. single completion (n=1, plain wrap) ==")
    r = client.chat.completions.create(
        model=args.model, messages=[{"role": "user", "content": "Hello!"}], max_tokens=16,
    )
    print(" content:", repr(r.choices[0].message.content[:60]))

    print("== 2. consensus completion (n=5) ==")
    r = client.chat.completions.create(
        model=args.model,
        messages=[{"role": "user", "content": "What color is the sky?"}],
        n=5, temperature=1.0, max_tokens=16, seed=42,
    )
    print(" consensus:", repr(r.choices[0].message.content[:60]))
    print(" n choices:", len(r.choices), "likelihoods:", r.likelihoods)
    print(" usage:", r.usage)

    print("== 3. structured output with schema-constrained decoding (n=3) ==")
    r = client.chat.completions.parse(
        model=args.model,
        messages=[{"role": "user", "content": "Extract: ACME Corp, $1200.50, paid"}],
        response_format=Invoice, n=3, max_tokens=200, seed=7,
    )
    for c in r.choices:
        print(f"  choice[{c.index}] finish={c.finish_reason} content={c.message.content[:60]!r}")
    print(" parsed consensus:", r.choices[0].message.parsed)
    print(" likelihoods:", r.likelihoods)

    print("== 4. temperature effects ==")
    for temp in (0.0, 1.2):
        r = client.chat.completions.create(
            model=args.model, messages=[{"role": "user", "content": "counting: one two"}],
            n=3, temperature=temp, max_tokens=8, seed=1,
        )
        uniq = len({c.message.content for c in r.choices[1:]})
        print(f"  temperature={temp}: {uniq} distinct completions of 3")

    print("== 5. embeddings ==")
    embs = client.get_embeddings(["alpha beta", "alpha beta", "gamma"], "text-embedding-3-small", 2048, False)
    print(f"  3 embeddings of dim {len(embs[0])}; [0]==[1]: {embs[0] == embs[1]}")

    print("== 6. async consensus ==")

    async def run_async():
        ak = AsyncKLLMs(model=args.model, engine=client.client)
        return await ak.chat.completions.create(
            model=args.model, messages=[{"role": "user", "content": "hi"}], n=3, max_tokens=8,
        )

    r = asyncio.run(run_async())
    print("  async choices:", len(r.choices))

    print("== 7. error handling: empty messages ==")
    try:
        client.chat.completions.create(model=args.model, messages=[], n=2)
    except Exception as e:
        print("  raised:", type(e).__name__)

    print("== 8. per-token top_logprobs ==")
    r = client.chat.completions.create(
        model=args.model, messages=[{"role": "user", "content": "alternatives"}],
        n=1, max_tokens=3, temperature=0.0, logprobs=True, top_logprobs=3, seed=2,
    )
    tk = r.choices[0].logprobs.content[0]
    print(f"  token {tk.token!r} lp={tk.logprob:.3f}; "
          f"top-3: {[(t.token, round(t.logprob, 3)) for t in tk.top_logprobs]}")

    print("== 9. serving opt-ins (fp8 KV cache, whitespace-tolerant JSON, chunked prefill) ==")
    opted = KLLMs(model=args.model, device=client.client.config.device,
                  use_hip_graphs=False, max_kv_blocks=256,
                  kv_cache_dtype="fp8_e4m3", constrained_whitespace=True,
                  prefill_chunk_tokens=64)
    r = opted.chat.completions.parse(
        model=args.model, messages=[{"role": "user", "content": "extract"}],
        response_format=Invoice, n=2, max_tokens=200, seed=5,
    )
    print("  fp8-cache constrained parse choices:", len(r.choices),
          "parsed:", r.choices[0].message.parsed is not None)


if __name__ == "__main__":
    main()
