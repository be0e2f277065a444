"""Call the OpenAI-compatible server with plain HTTP (httpx).

Terminal 1:  python -m kllms_amd serve --model llama-3-8b --port 8000
             (CPU demo: --model tiny-llama --max-kv-blocks 512)
Terminal 2:  python examples/http_client.py [--port 8000]

Any stock OpenAI SDK works the same way — point base_url at the server.
The response is the k-LLMs consensus shape: choices[0] is the consensus,
choices[1..n] the originals, `likelihoods` the per-field confidence.
"""

import argparse
import json

import httpx


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--model", default=None, help="default: whatever the server loaded")
    args = ap.parse_args()
    base = f"http://127.0.0.1:{args.port}"

    model = args.model or httpx.get(f"{base}/v1/models").json()["data"][0]["id"]

    # consensus completion
    r = httpx.post(f"{base}/v1/chat/completions", timeout=120, json={
        "model": model,
        "messages": [{"role": "user", "content": "Name three prime numbers."}],
        "n": 5, "temperature": 0.8, "max_tokens": 48, "seed": 0,
    })
    r.raise_for_status()
    body = r.json()
    print("consensus:", body["choices"][0]["message"]["content"])
    print("originals:", [c["message"]["content"][:40] for c in body["choices"][1:]])
    print("likelihoods:", body["likelihoods"])

    # schema-constrained structured output
    schema = {"type": "object",
              "properties": {"name": {"type": "string", "maxLength": 16},
                             "age": {"type": "integer", "minimum": 0, "maximum": 120}},
              "required": ["name", "age"]}
    r = httpx.post(f"{base}/v1/chat/completions", timeout=120, json={
        "model": model,
        "messages": [{"role": "user", "content": "John is 30 years old."}],
        "n": 3, "max_tokens": 48, "seed": 1,
        "response_format": {"type": "json_schema",
                            "json_schema": {"name": "person", "schema": schema}},
    })
    r.raise_for_status()
    doc = json.loads(r.json()["choices"][0]["message"]["content"])
    print("parsed consensus:", doc)


if __name__ == "__main__":
    main()
