"""Async consensus API.

The reference duplicates the entire consensus engine as line-for-line async
mirrors (ref consensus_utils.py:1523-2274, consolidation.py:219-303,402-493).
Here the async API is a thin bridge over the one shared sync implementation:
the consensus math runs in a worker thread (it is pure CPU/GPU compute with no
awaits in the reference either — its async mirror is sequential awaits, SURVEY
§3.3), and the caller's *async* embedding callable is bridged back onto the
running event loop via run_coroutine_threadsafe. Observable semantics are
identical to the reference mirrors.
"""

from __future__ import annotations

import asyncio
from typing import Any, Optional, Type

from pydantic import BaseModel

from ..types.completions import KLLMsChatCompletion
from ..types.parsed import KLLMsParsedChatCompletion
from .alignment import recursive_list_alignments
from .consolidation import (
    consolidate_chat_completions,
    consolidate_parsed_chat_completions,
)
from .primitive import string_consensus_llm
from .settings import (
    ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    ConsensusSettings,
    StringSimilarityMethod,
)
from .values import consensus_values


def _bridge_async_embeddings(async_fn: ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE, loop: asyncio.AbstractEventLoop):
    """Wrap an async embeddings callable into a sync callable usable from a
    worker thread, scheduling the coroutine on the caller's event loop."""
    if async_fn is None:
        return None

    def sync_fn(texts: list[str]) -> list[list[float]]:
        future = asyncio.run_coroutine_threadsafe(async_fn(texts), loop)
        return future.result()

    return sync_fn


async def async_string_consensus_llm(values: list[str], client: Any) -> str:
    """Async mirror of string_consensus_llm (ref consensus_utils.py:1051-1073)."""
    return await asyncio.to_thread(string_consensus_llm, values, client)


async def async_recursive_list_alignments(
    values: list[Any],
    string_similarity_method: StringSimilarityMethod,
    async_get_openai_embeddings_from_text: ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    min_support_ratio: float = 0.5,
    max_novelty_ratio: float = 0.25,
    current_path: str = "",
    reference_idx: Optional[int] = None,
):
    """Async mirror of recursive_list_alignments (ref consensus_utils.py:1916-2059)."""
    loop = asyncio.get_running_loop()
    sync_embed = _bridge_async_embeddings(async_get_openai_embeddings_from_text, loop)
    return await asyncio.to_thread(
        recursive_list_alignments,
        values,
        string_similarity_method,
        sync_embed,
        client,
        min_support_ratio,
        max_novelty_ratio,
        current_path,
        reference_idx,
    )


async def async_consensus_values(
    values: list[Any],
    consensus_settings: ConsensusSettings,
    async_get_openai_embeddings_from_text: ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    parent_valid_frac: float = 1.0,
):
    """Async mirror of consensus_values (ref consensus_utils.py:1779-1858)."""
    loop = asyncio.get_running_loop()
    sync_embed = _bridge_async_embeddings(async_get_openai_embeddings_from_text, loop)
    return await asyncio.to_thread(
        consensus_values,
        values,
        consensus_settings,
        sync_embed,
        client,
        parent_valid_frac,
    )


async def async_consolidate_chat_completions(
    completion,
    async_get_openai_embeddings_from_text: ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    consensus_settings: ConsensusSettings = None,  # type: ignore[assignment]
    aligner: str = "similarity",
) -> KLLMsChatCompletion:
    """Async mirror of consolidate_chat_completions (ref consolidation.py:219-303)."""
    loop = asyncio.get_running_loop()
    sync_embed = _bridge_async_embeddings(async_get_openai_embeddings_from_text, loop)
    return await asyncio.to_thread(
        consolidate_chat_completions,
        completion,
        sync_embed,
        client,
        consensus_settings or ConsensusSettings(),
        aligner,
    )


async def async_consolidate_parsed_chat_completions(
    completion,
    async_get_openai_embeddings_from_text: ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    consensus_settings: ConsensusSettings = None,  # type: ignore[assignment]
    response_format: Optional[Type[BaseModel]] = None,
    aligner: str = "similarity",
) -> KLLMsParsedChatCompletion:
    """Async mirror of consolidate_parsed_chat_completions (ref consolidation.py:402-493)."""
    loop = asyncio.get_running_loop()
    sync_embed = _bridge_async_embeddings(async_get_openai_embeddings_from_text, loop)
    return await asyncio.to_thread(
        consolidate_parsed_chat_completions,
        completion,
        sync_embed,
        client,
        consensus_settings or ConsensusSettings(),
        response_format,
        aligner,
    )
