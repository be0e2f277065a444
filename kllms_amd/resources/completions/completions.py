"""Completions.create / Completions.parse and async twins.

Behavioral re-implementation of the reference resource layer
(k_llms/resources/completions/completions.py:19-294): builds call params,
forces ``stream=False`` (streaming unsupported, reference README.md:125-127),
sets ``n``, makes ONE engine call — the shared-prefill + n-way fanned decode
batch — then consolidates via the consensus layer. n <= 1 is also routed
through consolidation (plain wrap).
"""

from __future__ import annotations

from typing import TYPE_CHECKING, Any, Dict, List, Optional, Type, Union

from pydantic import BaseModel

from ...consensus.aio import (
    async_consolidate_chat_completions,
    async_consolidate_parsed_chat_completions,
)
from ...consensus.consolidation import (
    consolidate_chat_completions,
    consolidate_parsed_chat_completions,
)
from ...types.completions import KLLMsChatCompletion
from ...types.parsed import KLLMsParsedChatCompletion

if TYPE_CHECKING:
    from ...client import AsyncKLLMs, KLLMs


def _build_call_params(
    messages: List[Dict[str, Any]],
    model: str,
    temperature: Optional[float],
    max_tokens: Optional[int],
    top_p: Optional[float],
    frequency_penalty: Optional[float],
    presence_penalty: Optional[float],
    stop: Optional[Union[str, List[str]]],
    seed: Optional[int],
    response_format: Optional[Any],
    extra: Dict[str, Any],
    force_stream_false: bool,
) -> Dict[str, Any]:
    """Assemble engine call params (ref completions.py:42-70)."""
    call_params: Dict[str, Any] = {"messages": messages, "model": model}
    if force_stream_false:
        call_params["stream"] = False
    if temperature is not None:
        call_params["temperature"] = temperature
    if max_tokens is not None:
        call_params["max_tokens"] = max_tokens
    if top_p is not None:
        call_params["top_p"] = top_p
    if frequency_penalty is not None:
        call_params["frequency_penalty"] = frequency_penalty
    if presence_penalty is not None:
        call_params["presence_penalty"] = presence_penalty
    if stop is not None:
        call_params["stop"] = stop
    if seed is not None:
        call_params["seed"] = seed
    if response_format is not None:
        call_params["response_format"] = response_format
    call_params.update(extra)
    return call_params


def _apply_wrapper_timeout(call_params: Dict[str, Any], wrapper) -> None:
    """Client-level `timeout` (reference C1 holds it for the HTTP client;
    locally it bounds generation wall-time — streams past the deadline
    finish with reason "length")."""
    if "timeout" not in call_params and getattr(wrapper, "timeout", None):
        call_params["timeout"] = wrapper.timeout


class Completions:
    def __init__(self, wrapper: "KLLMs"):
        self._wrapper = wrapper

    def create(
        self,
        *,
        messages: List[Dict[str, Any]],
        model: str,
        n: Optional[int] = None,
        temperature: Optional[float] = None,
        max_tokens: Optional[int] = None,
        top_p: Optional[float] = None,
        frequency_penalty: Optional[float] = None,
        presence_penalty: Optional[float] = None,
        stop: Optional[Union[str, List[str]]] = None,
        seed: Optional[int] = None,
        response_format: Optional[Any] = None,
        **kwargs: Any,
    ) -> KLLMsChatCompletion:
        """n-way sampled completion + consensus (ref completions.py:19-87)."""
        kwargs.pop("stream", None)  # streaming unsupported; always forced off
        call_params = _build_call_params(
            messages, model, temperature, max_tokens, top_p, frequency_penalty,
            presence_penalty, stop, seed, response_format, kwargs, force_stream_false=True,
        )
        _apply_wrapper_timeout(call_params, self._wrapper)

        def embeddings_wrapper(texts: List[str]) -> List[List[float]]:
            return self._wrapper.get_embeddings(texts, "text-embedding-3-small", 2048, False)

        if n and n > 1:
            call_params["n"] = n
        # ONE engine call: shared prefill, n decode streams fanned out.
        completion = self._wrapper.client.chat.completions.create(**call_params)
        if completion.choices and all(c.message.tool_calls for c in completion.choices):
            from ...consensus.consolidation import consolidate_tool_call_completions

            return consolidate_tool_call_completions(
                completion,
                embeddings_wrapper,
                client=self._wrapper.client,
                aligner=getattr(self._wrapper, "consensus_aligner", "similarity"),
            )
        return consolidate_chat_completions(
            completion,
            embeddings_wrapper,
            client=self._wrapper.client,
            aligner=getattr(self._wrapper, "consensus_aligner", "similarity"),
        )

    def parse(
        self,
        *,
        messages: List[Dict[str, Any]],
        model: str,
        response_format: Type[BaseModel],
        n: Optional[int] = None,
        temperature: Optional[float] = None,
        max_tokens: Optional[int] = None,
        top_p: Optional[float] = None,
        frequency_penalty: Optional[float] = None,
        presence_penalty: Optional[float] = None,
        stop: Optional[Union[str, List[str]]] = None,
        seed: Optional[int] = None,
        **kwargs: Any,
    ) -> KLLMsParsedChatCompletion:
        """Structured-output completion: JSON-schema-constrained decoding runs
        locally (Pydantic -> JSON schema -> token-mask automaton fused into the
        sampling kernel), then consensus re-validation (ref completions.py:89-150)."""
        call_params = _build_call_params(
            messages, model, temperature, max_tokens, top_p, frequency_penalty,
            presence_penalty, stop, seed, response_format, kwargs, force_stream_false=False,
        )
        _apply_wrapper_timeout(call_params, self._wrapper)

        def embeddings_wrapper(texts: List[str]) -> List[List[float]]:
            return self._wrapper.get_embeddings(texts, "text-embedding-3-small", 2048, False)

        if n and n > 1:
            call_params["n"] = n
        completion = self._wrapper.client.beta.chat.completions.parse(**call_params)
        return consolidate_parsed_chat_completions(
            completion,
            embeddings_wrapper,
            response_format=response_format,
            client=self._wrapper.client,
            aligner=getattr(self._wrapper, "consensus_aligner", "similarity"),
        )


class AsyncCompletions:
    def __init__(self, wrapper: "AsyncKLLMs"):
        self._wrapper = wrapper

    async def create(
        self,
        *,
        messages: List[Dict[str, Any]],
        model: str,
        response_format: Optional[Any] = None,
        n: Optional[int] = None,
        temperature: Optional[float] = None,
        max_tokens: Optional[int] = None,
        top_p: Optional[float] = None,
        frequency_penalty: Optional[float] = None,
        presence_penalty: Optional[float] = None,
        stop: Optional[Union[str, List[str]]] = None,
        seed: Optional[int] = None,
        **kwargs: Any,
    ) -> KLLMsChatCompletion:
        """Async mirror of create (ref completions.py:157-228)."""
        kwargs.pop("stream", None)
        call_params = _build_call_params(
            messages, model, temperature, max_tokens, top_p, frequency_penalty,
            presence_penalty, stop, seed, response_format, kwargs, force_stream_false=True,
        )
        _apply_wrapper_timeout(call_params, self._wrapper)

        async def embeddings_wrapper(texts: List[str]) -> List[List[float]]:
            return await self._wrapper.get_embeddings(texts, "text-embedding-3-small", 2048, False)

        if n and n > 1:
            call_params["n"] = n
        completion = await self._wrapper.client.chat.completions.acreate(**call_params)
        if completion.choices and all(c.message.tool_calls for c in completion.choices):
            import asyncio as _asyncio

            from ...consensus.consolidation import consolidate_tool_call_completions
            from ...consensus.aio import _bridge_async_embeddings

            loop = _asyncio.get_running_loop()
            return await _asyncio.to_thread(
                consolidate_tool_call_completions,
                completion,
                _bridge_async_embeddings(embeddings_wrapper, loop),
                self._wrapper.client,
                None,
                getattr(self._wrapper, "consensus_aligner", "similarity"),
            )
        return await async_consolidate_chat_completions(
            completion,
            embeddings_wrapper,
            client=self._wrapper.client,
            aligner=getattr(self._wrapper, "consensus_aligner", "similarity"),
        )

    async def parse(
        self,
        *,
        messages: List[Dict[str, Any]],
        model: str,
        response_format: Type[BaseModel],
        n: Optional[int] = None,
        temperature: Optional[float] = None,
        max_tokens: Optional[int] = None,
        top_p: Optional[float] = None,
        frequency_penalty: Optional[float] = None,
        presence_penalty: Optional[float] = None,
        stop: Optional[Union[str, List[str]]] = None,
        seed: Optional[int] = None,
        **kwargs: Any,
    ) -> KLLMsParsedChatCompletion:
        """Async mirror of parse (ref completions.py:230-294)."""
        call_params = _build_call_params(
            messages, model, temperature, max_tokens, top_p, frequency_penalty,
            presence_penalty, stop, seed, response_format, kwargs, force_stream_false=False,
        )
        _apply_wrapper_timeout(call_params, self._wrapper)

        async def embeddings_wrapper(texts: List[str]) -> List[List[float]]:
            return await self._wrapper.get_embeddings(texts, "text-embedding-3-small", 2048, False)

        if n and n > 1:
            call_params["n"] = n
        completion = await self._wrapper.client.beta.chat.completions.aparse(**call_params)
        return await async_consolidate_parsed_chat_completions(
            completion,
            embeddings_wrapper,
            response_format=response_format,
            client=self._wrapper.client,
            aligner=getattr(self._wrapper, "consensus_aligner", "similarity"),
        )
