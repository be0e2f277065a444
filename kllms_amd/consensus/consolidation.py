"""Consolidation layer: n ChatCompletion choices -> one consensus completion.

Behavioral re-implementation of k_llms/utils/consolidation.py:25-493.
Invariants (ref consolidation.py:116-144):
- ``len(choices) == n+1``; ``choices[0]`` is the consensus;
- ``choices[i].index == i``; originals keep their own finish_reason/logprobs;
- consensus choice borrows finish_reason/logprobs/function_call/tool_calls/
  refusal from original choice 0;
- ``likelihoods`` mirrors the extraction structure;
- usage passed through from the source completion;
- n == 1 short-circuits to a plain wrap.
"""

from __future__ import annotations

import json
from typing import Any, List, Optional, Type, Union

from pydantic import BaseModel

from ..types.completions import KLLMsChatCompletion
from ..types.openai_compat import (
    ChatCompletion,
    ChatCompletionMessage,
    Choice,
    ParsedChatCompletion,
    ParsedChatCompletionMessage,
    ParsedChoice,
)
from ..types.parsed import KLLMsParsedChatCompletion
from .alignment import recursive_list_alignments
from .settings import SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE, ConsensusSettings
from .values import consensus_values


def _safe_parse_content(content: str) -> dict[str, Any]:
    """json.loads, else wrap as {"text": content} (ref consolidation.py:25-38).
    This is how free-text answers enter the JSON consensus machinery."""
    try:
        return json.loads(content)
    except (json.JSONDecodeError, TypeError):
        return {"text": content}


def _format_consensus_content(consensus_content: Any) -> str:
    """Inverse of _safe_parse_content (ref consolidation.py:41-60)."""
    if consensus_content is None:
        return ""
    if (
        isinstance(consensus_content, dict)
        and len(consensus_content) == 1
        and "text" in consensus_content
        and isinstance(consensus_content["text"], str)
    ):
        return consensus_content["text"]
    return json.dumps(consensus_content)


def _consensus_over_contents(
    contents: list[dict[str, Any]],
    get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any,
    consensus_settings: ConsensusSettings,
    aligner: str = "similarity",
) -> tuple[Any, Any]:
    """Pre-align then run consensus (ref consolidation.py:97-111).

    ``aligner`` selects the alignment engine: "similarity" (default,
    pairwise-similarity Hungarian alignment) or "key" (the L1b key-based
    engine — activated in the reference by the commented import swap at
    consolidation.py:22)."""
    if len(contents) >= 2:
        if consensus_settings.string_similarity_method == "embeddings" and get_embeddings_from_text is not None:
            # M5: one batched on-device embedding pass + one cosine GEMM
            # replaces every per-pair embedding call in the alignment hot
            # loop; short-string pairs batch through the Myers Levenshtein
            # kernel (the reference's <=50-char fallback path)
            from .accel import precompute_levenshtein_cache, precompute_similarity_cache

            precompute_similarity_cache(contents, get_embeddings_from_text, client=client)
            precompute_levenshtein_cache(contents, client=client)
        if aligner == "key":
            # the reference's key-based aligner takes no embed fn / client
            # (key matching is exact; ref key_based_alignment.py:350-359)
            from .key_based_alignment import recursive_align

            aligned_seq, _ = recursive_align(
                contents,
                consensus_settings.string_similarity_method,
                consensus_settings.min_support_ratio,
            )
        else:
            aligned_seq, _ = recursive_list_alignments(
                contents,
                consensus_settings.string_similarity_method,
                get_embeddings_from_text,
                client,
                consensus_settings.min_support_ratio,
            )
        contents = [(d if isinstance(d, dict) else {}) for d in aligned_seq]
    return consensus_values(
        contents,
        consensus_settings,
        get_embeddings_from_text,
        client=client,
    )


def _build_consensus_choice(base_choice: Choice, content_str: str) -> Choice:
    message = ChatCompletionMessage(
        role="assistant",
        content=content_str,
        function_call=base_choice.message.function_call,
        tool_calls=base_choice.message.tool_calls,
        refusal=base_choice.message.refusal,
    )
    return Choice(
        finish_reason=base_choice.finish_reason,
        index=0,
        message=message,
        logprobs=base_choice.logprobs,
    )


def consolidate_tool_call_completions(
    completion: ChatCompletion,
    get_openai_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    consensus_settings: ConsensusSettings = None,  # type: ignore[assignment]
    aligner: str = "similarity",
) -> KLLMsChatCompletion:
    """Consensus over FORCED tool-call responses (beyond-reference: the
    reference forwards `tools` to the remote API and its consolidator then
    crashes on the all-None contents — tool arguments are JSON, which is
    exactly what the consensus machinery consumes). The n choices' argument
    strings run through the standard consolidation; every resulting choice
    is re-wrapped as a tool call of the same function, with the CONSENSUS
    arguments on choice 0 and `likelihoods` mirroring the argument schema."""
    import uuid as _uuid

    from ..types.openai_compat import ChatCompletionMessageToolCall, Function

    name = completion.choices[0].message.tool_calls[0].function.name
    shadow = completion.model_copy(deep=True)
    for c in shadow.choices:
        c.message.content = c.message.tool_calls[0].function.arguments
        c.message.tool_calls = None
    out = consolidate_chat_completions(
        shadow, get_openai_embeddings_from_text, client, consensus_settings, aligner
    )
    orig_calls = [c.message.tool_calls[0] for c in completion.choices]
    for i, c in enumerate(out.choices):
        call_id = f"call_{_uuid.uuid4().hex[:24]}" if i == 0 else orig_calls[i - 1].id
        c.message.tool_calls = [ChatCompletionMessageToolCall(
            id=call_id, function=Function(name=name, arguments=c.message.content or ""))]
        c.message.content = None
    return out


def consolidate_chat_completions(
    completions: Union[List[ChatCompletion], ChatCompletion],
    get_openai_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    consensus_settings: ConsensusSettings = None,  # type: ignore[assignment]
    aligner: str = "similarity",
) -> KLLMsChatCompletion:
    """Consolidate one completion with n choices, or a list of completions,
    into a KLLMsChatCompletion with consensus (ref consolidation.py:63-216)."""
    if consensus_settings is None:
        consensus_settings = ConsensusSettings()

    if isinstance(completions, ChatCompletion):
        completion = completions
        assert len(completion.choices) > 0, "Cannot consolidate empty list of choices"

        if len(completion.choices) == 1:
            return KLLMsChatCompletion.model_validate(completion.model_dump())

        choice_contents = [
            _safe_parse_content(c.message.content) for c in completion.choices if c.message.content
        ]
        consensus_content, likelihoods = _consensus_over_contents(
            choice_contents, get_openai_embeddings_from_text, client, consensus_settings, aligner
        )

        content_str = _format_consensus_content(consensus_content)
        consolidated_choice = _build_consensus_choice(completion.choices[0], content_str)

        individual_choices = [
            Choice(finish_reason=c.finish_reason, index=i + 1, message=c.message, logprobs=c.logprobs)
            for i, c in enumerate(completion.choices)
        ]

        return KLLMsChatCompletion.model_validate(
            {
                **completion.model_dump(),
                "choices": [c.model_dump() for c in [consolidated_choice] + individual_choices],
                "likelihoods": likelihoods,
                "usage": completion.usage.model_dump() if completion.usage else None,
            }
        )

    # list-of-completions shape (ref consolidation.py:147-216)
    completion_list = completions
    assert len(completion_list) > 0, "Cannot consolidate empty list of completions"

    if len(completion_list) == 1:
        return KLLMsChatCompletion.model_validate(completion_list[0].model_dump())

    completion_contents = [
        _safe_parse_content(comp.choices[0].message.content)
        for comp in completion_list
        if comp.choices and comp.choices[0].message.content
    ]
    consensus_content, likelihoods = _consensus_over_contents(
        completion_contents, get_openai_embeddings_from_text, client, consensus_settings, aligner
    )

    base_completion = completion_list[0]
    content_str = _format_consensus_content(consensus_content)
    consolidated_choice = _build_consensus_choice(base_completion.choices[0], content_str)

    individual_choices = [
        Choice(
            finish_reason=comp.choices[0].finish_reason,
            index=i + 1,
            message=comp.choices[0].message,
            logprobs=comp.choices[0].logprobs,
        )
        for i, comp in enumerate(completion_list)
        if comp.choices
    ]

    return KLLMsChatCompletion.model_validate(
        {
            **base_completion.model_dump(),
            "choices": [c.model_dump() for c in [consolidated_choice] + individual_choices],
            "likelihoods": likelihoods,
            "usage": base_completion.usage.model_dump() if base_completion.usage else None,
        }
    )


def consolidate_parsed_chat_completions(
    completion: ParsedChatCompletion,
    get_openai_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    consensus_settings: ConsensusSettings = None,  # type: ignore[assignment]
    response_format: Optional[Type[BaseModel]] = None,
    aligner: str = "similarity",
) -> KLLMsParsedChatCompletion:
    """As consolidate_chat_completions plus re-validation of the consensus dict
    into the response_format model -> message.parsed; silent None on failure
    (ref consolidation.py:306-399)."""
    if consensus_settings is None:
        consensus_settings = ConsensusSettings()

    assert len(completion.choices) > 0, "Cannot consolidate empty list of choices"

    if len(completion.choices) == 1:
        return KLLMsParsedChatCompletion.model_validate(completion.model_dump())

    parsed_choice_contents = [
        _safe_parse_content(c.message.content) for c in completion.choices if c.message.content
    ]
    consensus_content, likelihoods = _consensus_over_contents(
        parsed_choice_contents, get_openai_embeddings_from_text, client, consensus_settings, aligner
    )

    parsed_consensus = None
    if response_format and consensus_content is not None:
        try:
            if isinstance(response_format, type) and issubclass(response_format, BaseModel):
                parsed_consensus = response_format.model_validate(consensus_content)
        except Exception:
            parsed_consensus = None

    content_str = _format_consensus_content(consensus_content)
    base_choice = completion.choices[0]
    consolidated_message = ParsedChatCompletionMessage(
        role="assistant",
        content=content_str,
        function_call=base_choice.message.function_call,
        tool_calls=base_choice.message.tool_calls,
        refusal=base_choice.message.refusal,
        parsed=parsed_consensus,
    )
    consolidated_choice = ParsedChoice(
        finish_reason=base_choice.finish_reason,
        index=0,
        message=consolidated_message,
        logprobs=base_choice.logprobs,
    )

    individual_choices = [
        ParsedChoice(finish_reason=c.finish_reason, index=i + 1, message=c.message, logprobs=c.logprobs)
        for i, c in enumerate(completion.choices)
    ]

    dumped = {
        **completion.model_dump(),
        "likelihoods": likelihoods,
        "usage": completion.usage.model_dump() if completion.usage else None,
    }
    result = KLLMsParsedChatCompletion.model_validate(
        {**dumped, "choices": [c.model_dump() for c in [consolidated_choice] + individual_choices]}
    )
    # model_dump round-trips `parsed` through a plain dict; restore the live
    # pydantic instances on the validated result.
    result.choices[0].message.parsed = parsed_consensus
    for i, c in enumerate(completion.choices):
        result.choices[i + 1].message.parsed = getattr(c.message, "parsed", None)
    return result
