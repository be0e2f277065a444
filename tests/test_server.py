"""HTTP serving layer: OpenAI-compatible surface over the local engine.

fastapi TestClient (in-process ASGI) over a tiny CPU model — the same
consensus semantics the Python client exposes, reachable by any stock
OpenAI SDK."""

import json

import pytest

fastapi = pytest.importorskip("fastapi")

from fastapi.testclient import TestClient  # noqa: E402


@pytest.fixture(scope="module")
def http():
    from kllms_amd import AsyncKLLMs
    from kllms_amd.server import create_app

    client = AsyncKLLMs(model="tiny-llama", device="cpu", max_kv_blocks=512,
                        use_hip_graphs=False, seed=0, default_max_new_tokens=8)
    with TestClient(create_app(client)) as tc:
        yield tc


class TestRoutes:
    def test_health(self, http):
        r = http.get("/health")
        assert r.status_code == 200
        assert r.json()["status"] == "ok"
        assert r.json()["model"] == "tiny-llama"

    def test_models(self, http):
        r = http.get("/v1/models")
        assert r.status_code == 200
        body = r.json()
        assert body["object"] == "list"
        assert body["data"][0]["id"] == "tiny-llama"

    def test_create_consensus_shape(self, http):
        r = http.post("/v1/chat/completions", json={
            "model": "tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "n": 3, "max_tokens": 6, "temperature": 0.8, "seed": 7,
        })
        assert r.status_code == 200, r.text
        body = r.json()
        # k-LLMs contract: n+1 choices, consensus first, likelihoods present
        assert len(body["choices"]) == 4
        assert body["choices"][0]["index"] == 0
        assert [c["index"] for c in body["choices"][1:]] == [1, 2, 3]
        assert "likelihoods" in body
        assert body["usage"]["total_tokens"] == (
            body["usage"]["prompt_tokens"] + body["usage"]["completion_tokens"])

    def test_constrained_json_schema(self, http):
        schema = {"type": "object",
                  "properties": {"name": {"type": "string", "maxLength": 8},
                                 "age": {"type": "integer", "minimum": 0, "maximum": 99}},
                  "required": ["name", "age"]}
        r = http.post("/v1/chat/completions", json={
            "model": "tiny-llama",
            "messages": [{"role": "user", "content": "John is 30"}],
            "n": 2, "max_tokens": 32, "seed": 3,
            "response_format": {"type": "json_schema",
                                "json_schema": {"name": "person", "schema": schema}},
        })
        assert r.status_code == 200, r.text
        for choice in r.json()["choices"]:
            if choice["finish_reason"] == "stop":
                doc = json.loads(choice["message"]["content"])
                assert set(doc) == {"name", "age"} and 0 <= doc["age"] <= 99

    def test_seeded_reproducible(self, http):
        req = {"model": "tiny-llama",
               "messages": [{"role": "user", "content": "x"}],
               "n": 2, "max_tokens": 5, "temperature": 1.0, "seed": 11}
        a = http.post("/v1/chat/completions", json=req).json()
        b = http.post("/v1/chat/completions", json=req).json()
        assert [c["message"]["content"] for c in a["choices"]] == \
               [c["message"]["content"] for c in b["choices"]]


class TestErrors:
    def test_stream_rejected(self, http):
        r = http.post("/v1/chat/completions", json={
            "model": "tiny-llama", "stream": True,
            "messages": [{"role": "user", "content": "x"}]})
        assert r.status_code == 400
        assert "streaming" in r.json()["error"]["message"]

    def test_bad_params_400(self, http):
        r = http.post("/v1/chat/completions", json={
            "model": "tiny-llama", "temperature": 9.0,
            "messages": [{"role": "user", "content": "x"}]})
        assert r.status_code == 400
        assert r.json()["error"]["type"] == "invalid_request_error"

    def test_unknown_param_400(self, http):
        r = http.post("/v1/chat/completions", json={
            "model": "tiny-llama", "bogus_knob": 1,
            "messages": [{"role": "user", "content": "x"}]})
        assert r.status_code == 400
        assert "bogus_knob" in r.json()["error"]["message"]

    def test_empty_messages_400(self, http):
        r = http.post("/v1/chat/completions", json={"model": "tiny-llama", "messages": []})
        assert r.status_code == 400

    def test_context_length_400(self, http):
        r = http.post("/v1/chat/completions", json={
            "model": "tiny-llama",
            "messages": [{"role": "user", "content": "word " * 2000}]})
        assert r.status_code == 400
        assert r.json()["error"]["code"] == "context_length_exceeded"


class TestConcurrency:
    def test_parallel_requests_merge(self, http):
        """Concurrent HTTP requests ride the continuous-batching scheduler —
        all complete, each with its own seeded, reproducible output."""
        import concurrent.futures as cf

        def one(i):
            r = http.post("/v1/chat/completions", json={
                "model": "tiny-llama",
                "messages": [{"role": "user", "content": f"req {i}"}],
                "n": 2, "max_tokens": 6, "temperature": 0.9, "seed": i})
            assert r.status_code == 200, r.text
            return r.json()

        with cf.ThreadPoolExecutor(6) as ex:
            results = list(ex.map(one, range(6)))
        assert all(len(b["choices"]) == 3 for b in results)
        # seeded determinism survives concurrent admission
        again = one(3)
        assert [c["message"]["content"] for c in again["choices"]] == \
               [c["message"]["content"] for c in results[3]["choices"]]


class TestFuzz:
    def test_random_bodies_never_500(self, http):
        """Adversarial/garbage request bodies: always a clean 4xx/200,
        never an unhandled 500."""
        import random

        rng = random.Random(0)
        vals = [None, True, 0, -1, 1.5, "x", [], {}, {"role": "user"},
                [{"role": "user", "content": "hi"}], "유니코드", {"a": [1, {"b": None}]},
                float("1e308"), [[]], {"content": "y"}]
        keys = ["model", "messages", "n", "temperature", "top_p", "max_tokens",
                "stop", "seed", "logprobs", "top_logprobs", "logit_bias",
                "response_format", "tools", "tool_choice", "stream", "bogus"]
        for trial in range(60):
            body = {rng.choice(keys): rng.choice(vals)
                    for _ in range(rng.randint(0, 4))}
            if rng.random() < 0.5:
                body["messages"] = [{"role": "user", "content": "hi"}]
                body.setdefault("max_tokens", 4)
            r = http.post("/v1/chat/completions", json=body)
            assert r.status_code < 500, (trial, body, r.text[:300])


class TestEmbeddingsRoute:
    def test_embeddings_shape(self, http):
        r = http.post("/v1/embeddings", json={
            "input": ["hello world", "hello world!", "totally different topic here"],
            "model": "text-embedding-3-small"})
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["object"] == "list" and len(body["data"]) == 3
        assert body["data"][0]["index"] == 0
        v = body["data"][0]["embedding"]
        assert isinstance(v, list) and len(v) > 0
        assert body["usage"]["prompt_tokens"] > 0
        # near-duplicates more similar than unrelated text (ngram embedder)
        import math
        def cos(a, b):
            num = sum(x * y for x, y in zip(a, b))
            den = math.sqrt(sum(x * x for x in a)) * math.sqrt(sum(y * y for y in b))
            return num / den if den else 0.0
        e = [d["embedding"] for d in body["data"]]
        assert cos(e[0], e[1]) > cos(e[0], e[2])

    def test_embeddings_errors(self, http):
        assert http.post("/v1/embeddings", json={"input": 5}).status_code == 400
        r = http.post("/v1/embeddings", json={"input": "x", "model": "bogus-model"})
        assert r.status_code == 404
        assert r.json()["error"]["code"] == "model_not_found"
