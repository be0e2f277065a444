"""Coverage for the remaining consensus components: llm-consensus mode (C34),
usage summation (C39), async mirrors (C19/C21/C26/C38), TTL cache."""

import asyncio

import pytest

from kllms_amd.consensus import (
    ConsensusSettings,
    async_consensus_values,
    async_recursive_list_alignments,
    consensus_as_primitive,
    consolidate_consensus_usage,
    string_consensus_llm,
)
from kllms_amd.consensus.aio import async_consolidate_chat_completions
from kllms_amd.types.openai_compat import (
    ChatCompletion,
    ChatCompletionMessage,
    Choice,
    CompletionTokensDetails,
    CompletionUsage,
)
from kllms_amd.utils.ttl_cache import TTLCache


class StubLLMClient:
    """Chat-completions-shaped stub standing in for the local engine client."""

    consensus_model = "stub"

    def __init__(self, reply: str):
        self.reply = reply
        self.calls = []

        outer = self

        class _C:
            class completions:
                @staticmethod
                def create(**kw):
                    outer.calls.append(kw)
                    return ChatCompletion(
                        id="x", created=0, model=kw["model"],
                        choices=[Choice(finish_reason="stop", index=0,
                                        message=ChatCompletionMessage(role="assistant", content=outer.reply))],
                    )

        self.chat = _C()


def no_embed(texts):
    raise AssertionError("embeddings must not be called")


class TestLLMConsensus:
    def test_string_consensus_llm_calls_client(self):
        client = StubLLMClient("The sky is blue")
        result = string_consensus_llm(["The sky is blue today", "The sky is blue"], client)
        assert result == "The sky is blue"
        assert client.calls[0]["model"] == "stub"
        assert "consensus string" in client.calls[0]["messages"][0]["content"]

    def test_primitive_llm_mode_dispatch(self, fake_embed):
        settings = ConsensusSettings(
            string_consensus_method="llm-consensus", string_similarity_method="embeddings"
        )
        client = StubLLMClient("merged answer")
        vals = ["a long candidate string value " * 3, "another long candidate string " * 3]
        result, conf = consensus_as_primitive(vals, settings, fake_embed, client=client)
        assert result == "merged answer"
        assert 0 < conf <= 1


class TestUsageConsolidation:
    def test_sums_usage_and_details(self):
        u1 = CompletionUsage(prompt_tokens=10, completion_tokens=5, total_tokens=15,
                             completion_tokens_details=CompletionTokensDetails(reasoning_tokens=3))
        u2 = CompletionUsage(prompt_tokens=1, completion_tokens=2, total_tokens=3,
                             completion_tokens_details=CompletionTokensDetails(reasoning_tokens=4))
        out = consolidate_consensus_usage([u1, None, u2])
        assert out.prompt_tokens == 11
        assert out.completion_tokens == 7
        assert out.total_tokens == 18
        assert out.completion_tokens_details.reasoning_tokens == 7

    def test_empty(self):
        assert consolidate_consensus_usage([]) is None


class TestAsyncMirrors:
    def test_async_consensus_values(self):
        async def embed(texts):
            raise AssertionError

        async def run():
            return await async_consensus_values(["yes", "yes", "no"],
                                                ConsensusSettings(string_similarity_method="levenshtein"),
                                                embed)

        val, conf = asyncio.run(run())
        assert val == "yes"
        assert conf == pytest.approx(2 / 3, abs=1e-4)

    def test_async_alignment_with_async_embed_bridge(self, fake_embed):
        calls = {"n": 0}

        async def aembed(texts):
            calls["n"] += 1
            return fake_embed(texts)

        long_a = "x" * 60 + " alpha"
        long_b = "x" * 60 + " beta"

        async def run():
            return await async_recursive_list_alignments(
                [{"items": [long_a]}, {"items": [long_b]}], "embeddings", aembed, None, 0.51,
            )

        aligned, km = asyncio.run(run())
        assert len(aligned) == 2
        assert calls["n"] >= 1  # the async embed fn was actually bridged

    def test_async_consolidate(self):
        comp = ChatCompletion(
            id="c", created=0, model="m",
            choices=[
                Choice(finish_reason="stop", index=i,
                       message=ChatCompletionMessage(role="assistant", content=c))
                for i, c in enumerate(["blue", "blue", "red"])
            ],
        )

        async def aembed(texts):
            raise AssertionError

        async def run():
            return await async_consolidate_chat_completions(comp, aembed)

        r = asyncio.run(run())
        assert r.choices[0].message.content == "blue"
        assert len(r.choices) == 4


class TestTTLCache:
    def test_expiry_and_lru(self):
        t = {"now": 0.0}
        c = TTLCache(maxsize=2, ttl=10, timer=lambda: t["now"])
        c["a"] = 1
        c["b"] = 2
        c["c"] = 3  # evicts oldest
        assert "a" not in c
        assert c["b"] == 2
        t["now"] = 11.0
        assert "b" not in c  # expired
