"""KLLMs / AsyncKLLMs client-level tests: the public API surface and the §3.1
golden invariants through the REAL local engine (tiny preset, CPU)."""

import asyncio
import json

import pytest
from pydantic import BaseModel

from kllms_amd import AsyncKLLMs, KLLMs
from kllms_amd.types import KLLMsChatCompletion, KLLMsParsedChatCompletion

TINY = dict(model="tiny-llama", max_kv_blocks=512, use_hip_graphs=False, default_max_new_tokens=16, device="cpu")


@pytest.fixture(scope="module")
def client():
    return KLLMs(**TINY)


class TestCreate:
    def test_n5_golden_invariants(self, client):
        r = client.chat.completions.create(
            messages=[{"role": "user", "content": "Say something."}],
            model="tiny-llama", n=5, max_tokens=8, temperature=1.0, seed=11,
        )
        assert isinstance(r, KLLMsChatCompletion)
        assert len(r.choices) == 6  # consensus + 5 originals
        for i, c in enumerate(r.choices):
            assert c.index == i
        assert r.likelihoods is not None
        assert r.usage.prompt_tokens > 0
        # prompt charged once; completions summed over 5 streams
        assert r.usage.completion_tokens <= 5 * 8
        assert r.usage.total_tokens == r.usage.prompt_tokens + r.usage.completion_tokens

    def test_n1_plain_wrap(self, client):
        r = client.chat.completions.create(
            messages=[{"role": "user", "content": "hi"}], model="tiny-llama", n=1, max_tokens=4,
        )
        assert len(r.choices) == 1
        assert r.likelihoods is None

    def test_seed_reproducible(self, client):
        kw = dict(messages=[{"role": "user", "content": "x"}], model="tiny-llama", n=3, max_tokens=6, seed=5, temperature=0.8)
        r1 = client.chat.completions.create(**kw)
        r2 = client.chat.completions.create(**kw)
        assert [c.message.content for c in r1.choices] == [c.message.content for c in r2.choices]

    def test_logprobs_surface(self, client):
        r = client.chat.completions.create(
            messages=[{"role": "user", "content": "x"}], model="tiny-llama", n=2,
            max_tokens=4, logprobs=True, seed=1,
        )
        orig = r.choices[1]
        assert orig.logprobs is not None
        assert len(orig.logprobs.content) == len(orig.message.content.encode()) or orig.logprobs.content
        for t in orig.logprobs.content:
            assert t.logprob <= 0.0

    def test_top_logprobs(self, client):
        r = client.chat.completions.create(
            messages=[{"role": "user", "content": "x"}], model="tiny-llama", n=2,
            max_tokens=4, logprobs=True, top_logprobs=3, seed=1, temperature=0.0,
        )
        for orig in r.choices[1:]:
            assert orig.logprobs is not None and orig.logprobs.content
            for t in orig.logprobs.content:
                assert len(t.top_logprobs) == 3
                lps = [a.logprob for a in t.top_logprobs]
                assert lps == sorted(lps, reverse=True)
                # greedy decode: the chosen token IS the top alternative
                assert t.top_logprobs[0].token == t.token
                assert t.top_logprobs[0].logprob == pytest.approx(t.logprob, abs=1e-4)
        # consensus choice carries the FIRST original's logprobs (reference
        # consolidation.py:129 copies completion.choices[0].logprobs)
        assert r.choices[0].logprobs == r.choices[1].logprobs

    def test_top_logprobs_requires_logprobs(self, client):
        with pytest.raises(ValueError):
            client.chat.completions.create(
                messages=[{"role": "user", "content": "x"}], model="tiny-llama",
                n=1, max_tokens=2, top_logprobs=3,
            )


class Extraction(BaseModel):
    label: str
    score: int


class TestParse:
    def test_parse_constrained_valid_json(self, client):
        r = client.chat.completions.parse(
            messages=[{"role": "user", "content": "extract"}],
            model="tiny-llama", response_format=Extraction, n=3, max_tokens=200, seed=2,
        )
        assert isinstance(r, KLLMsParsedChatCompletion)
        assert len(r.choices) == 4
        # every completed original is schema-valid JSON
        for c in r.choices[1:]:
            if c.finish_reason == "stop":
                Extraction.model_validate(json.loads(c.message.content))
        assert r.likelihoods is not None

    def test_get_embeddings(self, client):
        embs = client.get_embeddings(["hello world", "hello world", "different"], "text-embedding-3-small", 2048, False)
        assert len(embs) == 3
        assert embs[0] == embs[1]
        assert embs[0] != embs[2]


class TestAsync:
    def test_async_create(self):
        ak = AsyncKLLMs(**TINY)

        async def run():
            return await ak.chat.completions.create(
                messages=[{"role": "user", "content": "hello"}], model="tiny-llama", n=3, max_tokens=6, seed=9,
            )

        r = asyncio.run(run())
        assert len(r.choices) == 4
        assert r.likelihoods is not None

    def test_async_parse(self):
        ak = AsyncKLLMs(**TINY)

        async def run():
            return await ak.chat.completions.parse(
                messages=[{"role": "user", "content": "x"}], model="tiny-llama",
                response_format=Extraction, n=2, max_tokens=150, seed=4,
            )

        r = asyncio.run(run())
        assert len(r.choices) == 3


class TestReferenceScenarios:
    """Mirrors the reference's example/comprehensive test themes
    (README_TESTS.md: nested models, multi-turn, error handling,
    temperature effects)."""

    def test_deep_nested_parse(self, client):
        class Employee(BaseModel):
            name: str
            role: str

        class Department(BaseModel):
            dept: str
            staff: list[Employee]

        class Company(BaseModel):
            company: str
            departments: list[Department]

        r = client.chat.completions.parse(
            messages=[{"role": "user", "content": "extract the org chart"}],
            model="tiny-llama", response_format=Company, n=3, max_tokens=400, seed=7,
        )
        assert len(r.choices) == 4
        for c in r.choices[1:]:
            if c.finish_reason == "stop":
                Company.model_validate(json.loads(c.message.content))
        # consensus likelihoods mirror the nested structure when parseable
        if r.choices[0].message.parsed is not None:
            assert isinstance(r.choices[0].message.parsed, Company)
            assert "company" in r.likelihoods

    def test_multi_turn_conversation(self, client):
        r = client.chat.completions.create(
            messages=[
                {"role": "system", "content": "You are terse."},
                {"role": "user", "content": "What is 2+2?"},
                {"role": "assistant", "content": "4"},
                {"role": "user", "content": "And doubled?"},
            ],
            model="tiny-llama", n=3, max_tokens=6, seed=3,
        )
        assert len(r.choices) == 4
        # multi-turn prompt is longer than the last message alone
        solo = client.chat.completions.create(
            messages=[{"role": "user", "content": "And doubled?"}],
            model="tiny-llama", n=1, max_tokens=2,
        )
        assert r.usage.prompt_tokens > solo.usage.prompt_tokens

    def test_empty_messages_raises(self, client):
        with pytest.raises(ValueError):
            client.chat.completions.create(messages=[], model="tiny-llama", n=1)

    def test_unknown_model_raises(self):
        from kllms_amd import KLLMs
        with pytest.raises(Exception):
            KLLMs(model="no-such-model-xyz").chat.completions.create(
                messages=[{"role": "user", "content": "x"}], model="no-such-model-xyz",
            )

    def test_greedy_consensus_is_unanimous(self, client):
        """temperature=0: all n streams identical -> consensus == the unique
        completion and every leaf likelihood is 1.0."""
        r = client.chat.completions.create(
            messages=[{"role": "user", "content": "deterministic"}],
            model="tiny-llama", n=4, temperature=0.0, max_tokens=6, seed=9,
        )
        texts = {c.message.content for c in r.choices[1:]}
        assert len(texts) == 1
        assert r.choices[0].message.content == texts.pop()

        def leaves(x):
            if isinstance(x, dict):
                for v in x.values():
                    yield from leaves(v)
            elif isinstance(x, list):
                for v in x:
                    yield from leaves(v)
            elif isinstance(x, (int, float)):
                yield x

        lv = list(leaves(r.likelihoods))
        assert lv and all(v == pytest.approx(1.0) for v in lv)

    def test_close_and_context_manager(self):
        from kllms_amd import KLLMs

        with KLLMs(**TINY) as c:
            r = c.chat.completions.create(
                messages=[{"role": "user", "content": "x"}], model="tiny-llama", n=1, max_tokens=2)
            assert r.choices
            # force a scheduler into existence, then ensure close() stops it
            sched = c.client.scheduler
            assert sched is not None
        assert c.client._scheduler is None  # shut down on exit

    def test_async_context_manager(self):
        async def run():
            async with AsyncKLLMs(**TINY) as ac:
                r = await ac.chat.completions.create(
                    messages=[{"role": "user", "content": "x"}], model="tiny-llama", n=2, max_tokens=2)
                return r
        r = asyncio.run(run())
        assert len(r.choices) == 3

    def test_shared_engine_not_closed_by_borrower(self):
        from kllms_amd import KLLMs

        owner = KLLMs(**TINY)
        _ = owner.client.scheduler
        with KLLMs(engine=owner.client) as borrower:
            pass
        assert owner.client._scheduler is not None  # borrower must not kill it
        owner.close()

    def test_embeddings_namespace(self, client):
        """OpenAI-shaped embeddings namespace (client.client.embeddings.create)."""
        r = client.client.embeddings.create(input=["alpha", "beta"], model="text-embedding-3-small")
        assert len(r.data) == 2
        assert r.data[0].index == 0 and r.data[1].index == 1
        assert len(r.data[0].embedding) == len(r.data[1].embedding) > 0
        assert r.usage.prompt_tokens > 0

    def test_async_get_embeddings(self):
        ak = AsyncKLLMs(**TINY)

        async def run():
            return await ak.get_embeddings(["one", "one", "two"], "text-embedding-3-small", 2048, False)

        embs = asyncio.run(run())
        assert len(embs) == 3
        assert embs[0] == embs[1] != embs[2]


def test_invalid_model_raises_model_not_found():
    from kllms_amd import KLLMs

    with pytest.raises(ValueError, match="model_not_found"):
        KLLMs(model="no-such-model", device="cpu", max_kv_blocks=32).chat.completions.create(
            model="no-such-model", messages=[{"role": "user", "content": "hi"}])


def test_logit_bias_served_natively():
    """OpenAI `logit_bias` (the reference forwards it to the remote API):
    +100 on a token forces it; -100 bans it."""
    from kllms_amd import KLLMs

    k = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=256,
              use_hip_graphs=False, default_max_new_tokens=6)
    force = 77
    r = k.chat.completions.create(
        model="tiny-llama", messages=[{"role": "user", "content": "hello"}],
        n=2, temperature=1.0, max_tokens=5, seed=3,
        logit_bias={str(force): 100.0})
    eng = k.client.engine
    # +100 dominates every random-init logit: all sampled tokens == force
    ids = eng.tokenizer.encode(r.choices[1].message.content)
    assert all(i == force for i in ids), ids

    # ban: the greedy top token (raw stream ids, not a decode round-trip)
    from kllms_amd.engine.engine import GenRequest
    from kllms_amd.engine.sampling import SamplingParams

    ids = eng.tokenizer.encode("hello")
    base = eng.generate([GenRequest(prompt_ids=ids, n=1,
                                    sampling=SamplingParams(temperature=0.0, max_tokens=3))])[0]
    top = base.streams[0].token_ids[0]
    banned = eng.generate([GenRequest(prompt_ids=ids, n=1,
                                      sampling=SamplingParams(temperature=0.0, max_tokens=3,
                                                              logit_bias={top: -100.0}))])[0]
    assert banned.streams[0].token_ids[0] != top


class TestToolCalls:
    """FORCED function calls served natively: the function's parameters
    schema compiles into the decoding DFA; the consensus layer treats the
    result exactly as it treats the reference's remote tool-call responses
    (consensus choice borrows tool_calls from choice 0)."""

    TOOLS = [{"type": "function", "function": {
        "name": "get_weather",
        "parameters": {"type": "object",
                       "properties": {"city": {"type": "string", "maxLength": 10},
                                      "days": {"type": "integer", "minimum": 0, "maximum": 9}},
                       "required": ["city", "days"]}}}]

    def _client(self):
        from kllms_amd.engine.api import LocalEngineClient

        return LocalEngineClient(model="tiny-llama", device="cpu", max_kv_blocks=256,
                                 use_hip_graphs=False, default_max_new_tokens=48)

    def test_forced_tool_call_schema_valid(self):
        import json as _json

        client = self._client()
        r = client.chat.completions.create(
            model="tiny-llama", messages=[{"role": "user", "content": "weather in Paris?"}],
            tools=self.TOOLS, tool_choice={"type": "function", "function": {"name": "get_weather"}},
            n=2, temperature=0.8, max_tokens=48, seed=4)
        for ch in r.choices:
            assert ch.message.content is None
            tc = ch.message.tool_calls
            assert tc and tc[0].function.name == "get_weather"
            if ch.finish_reason == "tool_calls":
                args = _json.loads(tc[0].function.arguments)
                assert set(args) == {"city", "days"} and 0 <= args["days"] <= 9

    def test_required_single_tool(self):
        client = self._client()
        r = client.chat.completions.create(
            model="tiny-llama", messages=[{"role": "user", "content": "go"}],
            tools=self.TOOLS, tool_choice="required", n=1, temperature=0.0, max_tokens=48)
        assert r.choices[0].message.tool_calls

    def test_unknown_function_raises(self):
        client = self._client()
        with pytest.raises(ValueError, match="unknown function"):
            client.chat.completions.create(
                model="tiny-llama", messages=[{"role": "user", "content": "x"}],
                tools=self.TOOLS, tool_choice={"type": "function", "function": {"name": "nope"}})

    def test_auto_falls_back_to_content(self):
        client = self._client()
        r = client.chat.completions.create(
            model="tiny-llama", messages=[{"role": "user", "content": "hi"}],
            tools=self.TOOLS, tool_choice="auto", n=1, temperature=0.0, max_tokens=6)
        assert r.choices[0].message.tool_calls is None
        assert r.choices[0].message.content is not None

    def test_consensus_layer_over_tool_calls(self):
        """KLLMs().create with forced tools: n+1 choices, consensus choice 0
        borrows tool_calls from original choice 0 (reference semantics)."""
        from kllms_amd import KLLMs

        k = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=256,
                  use_hip_graphs=False, default_max_new_tokens=48)
        r = k.chat.completions.create(
            model="tiny-llama", messages=[{"role": "user", "content": "weather?"}],
            tools=self.TOOLS, tool_choice={"type": "function", "function": {"name": "get_weather"}},
            n=3, temperature=0.9, max_tokens=48, seed=9)
        assert len(r.choices) == 4
        assert r.choices[0].message.tool_calls is not None
        assert r.choices[0].message.tool_calls[0].function.name == "get_weather"


def test_async_tool_call_consensus():
    import asyncio

    from kllms_amd import AsyncKLLMs

    async def run():
        k = AsyncKLLMs(model="tiny-llama", device="cpu", max_kv_blocks=256,
                       use_hip_graphs=False, default_max_new_tokens=48)
        r = await k.chat.completions.create(
            model="tiny-llama", messages=[{"role": "user", "content": "weather?"}],
            tools=TestToolCalls.TOOLS,
            tool_choice={"type": "function", "function": {"name": "get_weather"}},
            n=3, temperature=0.9, max_tokens=48, seed=2)
        assert len(r.choices) == 4
        assert r.choices[0].message.tool_calls[0].function.name == "get_weather"
        k.close()

    asyncio.run(run())


class TestParamValidation:
    """OpenAI-documented sampling-param ranges raise locally (the reference
    lets the remote API 400 these; a switched-over client sees the same)."""

    def _client(self):
        from kllms_amd import KLLMs
        return KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=128,
                     use_hip_graphs=False, seed=0)

    def test_out_of_range_params_raise(self):
        c = self._client()
        msgs = [{"role": "user", "content": "hi"}]
        import pytest as _pytest
        for bad in (dict(temperature=2.5), dict(temperature=-0.1),
                    dict(top_p=1.5), dict(top_p=-0.01),
                    dict(n=129), dict(max_tokens=0),
                    dict(frequency_penalty=3.0), dict(presence_penalty=-2.5)):
            kw = dict(max_tokens=4)
            kw.update(bad)
            with _pytest.raises(ValueError):
                c.chat.completions.create(messages=msgs, model="tiny-llama", **kw)
        # n=0 is NOT an error: the wrapper only forwards n>1 (reference
        # completions.py:72 shape), so it behaves as the default n=1
        out = c.chat.completions.create(messages=msgs, model="tiny-llama",
                                        max_tokens=2, n=0)
        assert len(out.choices) == 1

    def test_boundary_values_accepted(self):
        c = self._client()
        msgs = [{"role": "user", "content": "hi"}]
        out = c.chat.completions.create(messages=msgs, model="tiny-llama",
                                        temperature=2.0, top_p=1.0, n=1,
                                        max_tokens=2, frequency_penalty=2.0,
                                        presence_penalty=-2.0)
        assert out.choices


def test_create_many_honors_penalties():
    """Regression: the packed-batch path (chat_completions_create_many, the
    bench/serving entry) silently dropped frequency/presence penalties."""
    from kllms_amd import KLLMs

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=256,
              use_hip_graphs=False, seed=0).client
    base = dict(messages=[{"role": "user", "content": "repeat"}],
                model="tiny-llama", n=1, temperature=0.0, max_tokens=12, seed=1)
    plain = c.chat_completions_create_many([dict(base)])[0]
    penal = c.chat_completions_create_many([dict(base, frequency_penalty=1.9,
                                                 presence_penalty=1.9)])[0]
    # greedy with strong penalties must diverge from unpenalized greedy
    assert plain.choices[0].message.content != penal.choices[0].message.content
    # and must match the single-request path with the same penalties
    single = c.chat_completions_create(**dict(base, frequency_penalty=1.9,
                                              presence_penalty=1.9))
    assert penal.choices[0].message.content == single.choices[0].message.content


def test_cli_entry_help():
    import subprocess
    import sys
    r = subprocess.run([sys.executable, "-m", "kllms_amd", "help"],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0 and "serve" in r.stdout and "bench" in r.stdout
    r2 = subprocess.run([sys.executable, "-m", "kllms_amd", "nonsense"],
                        capture_output=True, text=True, timeout=120)
    assert r2.returncode == 2


def test_key_aligner_through_client():
    """consensus_aligner="key" (the reference's commented-in L1b engine)
    must serve end to end through the public client."""
    from kllms_amd import KLLMs

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=256,
              use_hip_graphs=False, seed=0, consensus_aligner="key")
    out = c.chat.completions.create(
        messages=[{"role": "user", "content": "list items"}],
        model="tiny-llama", n=3, temperature=0.9, max_tokens=10, seed=5)
    assert len(out.choices) == 4 and out.choices[0].index == 0
    assert out.likelihoods is not None


def test_content_part_arrays_supported():
    """OpenAI content-part arrays (text parts) are flattened for the local
    engine; non-text parts get a clean error."""
    from kllms_amd import KLLMs

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=256,
              use_hip_graphs=False, seed=0)
    parts = [{"type": "text", "text": "Hello "}, {"type": "text", "text": "world"}]
    a = c.chat.completions.create(messages=[{"role": "user", "content": parts}],
                                  model="tiny-llama", max_tokens=4, temperature=0.0)
    b = c.chat.completions.create(messages=[{"role": "user", "content": "Hello world"}],
                                  model="tiny-llama", max_tokens=4, temperature=0.0)
    assert a.choices[0].message.content == b.choices[0].message.content

    import pytest as _pytest
    with _pytest.raises(ValueError, match="text content parts"):
        c.chat.completions.create(
            messages=[{"role": "user", "content": [{"type": "image_url",
                                                    "image_url": {"url": "x"}}]}],
            model="tiny-llama", max_tokens=4)


def test_request_timeout_caps_generation():
    """Client-level `timeout` bounds generation wall time: the stream
    finishes with reason "length" at the next step boundary and returns
    what it has, instead of running out max_tokens."""
    import time as _time

    from kllms_amd import KLLMs

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=512,
              use_hip_graphs=False, seed=0, timeout=1e-6)
    t0 = _time.monotonic()
    out = c.chat.completions.create(messages=[{"role": "user", "content": "go"}],
                                    model="tiny-llama", max_tokens=400,
                                    temperature=0.0)
    assert _time.monotonic() - t0 < 30
    assert out.choices[0].finish_reason == "length"
    assert len(out.choices[0].message.content or "") < 400

    # per-call override beats the client default
    c2 = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=512,
               use_hip_graphs=False, seed=0)
    out2 = c2.chat.completions.create(messages=[{"role": "user", "content": "go"}],
                                      model="tiny-llama", max_tokens=6,
                                      temperature=0.0, timeout=1e-6)
    assert out2.choices[0].finish_reason == "length"
