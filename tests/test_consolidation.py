"""Consolidation-layer tests: §3.1 choice-layout invariants
(behavioral contract of reference consolidation.py:63-493)."""

import json

import pytest
from pydantic import BaseModel

from kllms_amd.consensus import consolidate_chat_completions, consolidate_parsed_chat_completions
from kllms_amd.consensus.consolidation import _format_consensus_content, _safe_parse_content
from kllms_amd.types import KLLMsChatCompletion, KLLMsParsedChatCompletion
from kllms_amd.types.openai_compat import (
    ChatCompletion,
    ChatCompletionMessage,
    Choice,
    CompletionUsage,
    ParsedChatCompletion,
    ParsedChatCompletionMessage,
    ParsedChoice,
)


def no_embed(texts):
    raise AssertionError("embeddings must not be called")


def make_completion(contents, model="llama-3-8b"):
    choices = [
        Choice(
            finish_reason="stop",
            index=i,
            message=ChatCompletionMessage(role="assistant", content=c),
        )
        for i, c in enumerate(contents)
    ]
    return ChatCompletion(
        id="chatcmpl-test",
        choices=choices,
        created=1700000000,
        model=model,
        usage=CompletionUsage(prompt_tokens=10, completion_tokens=20, total_tokens=30),
    )


class TestSafeParse:
    def test_json_roundtrip(self):
        assert _safe_parse_content('{"a": 1}') == {"a": 1}

    def test_free_text_wrapped(self):
        assert _safe_parse_content("hello world") == {"text": "hello world"}

    def test_format_unwraps_text(self):
        assert _format_consensus_content({"text": "hi"}) == "hi"
        assert _format_consensus_content({"a": 1}) == '{"a": 1}'
        assert _format_consensus_content(None) == ""


class TestConsolidateInvariants:
    def test_choice_layout(self):
        contents = ['{"city": "Paris"}', '{"city": "Paris"}', '{"city": "Lyon"}']
        comp = make_completion(contents)
        result = consolidate_chat_completions(comp, no_embed)
        assert isinstance(result, KLLMsChatCompletion)
        # n+1 choices, consensus at index 0, originals re-indexed
        assert len(result.choices) == 4
        for i, c in enumerate(result.choices):
            assert c.index == i
        assert json.loads(result.choices[0].message.content) == {"city": "Paris"}
        for i, c in enumerate(result.choices[1:]):
            assert c.message.content == contents[i]
        assert result.likelihoods["city"] == pytest.approx(2 / 3, abs=1e-4)
        assert result.usage.total_tokens == 30

    def test_single_choice_plain_wrap(self):
        comp = make_completion(["hello"])
        result = consolidate_chat_completions(comp, no_embed)
        assert len(result.choices) == 1
        assert result.likelihoods is None

    def test_free_text_consensus(self):
        comp = make_completion(["blue", "blue", "green"])
        result = consolidate_chat_completions(comp, no_embed)
        assert result.choices[0].message.content == "blue"
        assert result.likelihoods["text"] == pytest.approx(2 / 3, abs=1e-4)

    def test_list_of_completions_shape(self):
        comps = [make_completion(['{"x": 1}']) for _ in range(3)]
        result = consolidate_chat_completions(comps, no_embed)
        assert len(result.choices) == 4
        assert json.loads(result.choices[0].message.content) == {"x": 1}

    def test_empty_choices_asserts(self):
        comp = make_completion(["a"])
        comp.choices = []
        with pytest.raises(AssertionError):
            consolidate_chat_completions(comp, no_embed)


class Person(BaseModel):
    name: str
    age: int


def make_parsed_completion(contents):
    choices = [
        ParsedChoice(
            finish_reason="stop",
            index=i,
            message=ParsedChatCompletionMessage(
                role="assistant", content=c, parsed=Person.model_validate_json(c)
            ),
        )
        for i, c in enumerate(contents)
    ]
    return ParsedChatCompletion(
        id="chatcmpl-parsed",
        choices=choices,
        created=1700000000,
        model="llama-3-8b",
        usage=CompletionUsage(prompt_tokens=5, completion_tokens=6, total_tokens=11),
    )


class TestConsolidateParsed:
    def test_parsed_consensus(self):
        contents = [
            '{"name": "Alice", "age": 30}',
            '{"name": "Alice", "age": 30}',
            '{"name": "Bob", "age": 31}',
        ]
        comp = make_parsed_completion(contents)
        result = consolidate_parsed_chat_completions(comp, no_embed, response_format=Person)
        assert isinstance(result, KLLMsParsedChatCompletion)
        assert len(result.choices) == 4
        parsed = result.choices[0].message.parsed
        assert isinstance(parsed, Person)
        assert parsed.name == "Alice"
        assert parsed.age == 30
        assert set(result.likelihoods.keys()) == {"name", "age"}

    def test_validation_failure_silent_none(self):
        # consensus age becomes a float mean that still validates; force a
        # failure with incompatible values -> medoid string under int field
        class Strict(BaseModel):
            age: int

        contents = ['{"age": "not a number at all"}', '{"age": "not a number at all"}', '{"age": "still not numeric here"}']
        comp_choices = [
            ParsedChoice(
                finish_reason="stop",
                index=i,
                message=ParsedChatCompletionMessage(role="assistant", content=c, parsed=None),
            )
            for i, c in enumerate(contents)
        ]
        comp = ParsedChatCompletion(
            id="x", choices=comp_choices, created=0, model="m"
        )
        result = consolidate_parsed_chat_completions(comp, no_embed, response_format=Strict)
        # consensus age is a non-numeric string -> Strict validation fails -> None
        assert result.choices[0].message.parsed is None
