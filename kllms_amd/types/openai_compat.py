"""OpenAI-SDK-compatible response types, defined locally.

The reference (k_llms/types/completions.py, k_llms/types/parsed.py) inherits
from ``openai.types.chat.*`` pydantic models. This framework is standalone —
no OpenAI SDK, no network — so the same shapes are defined here with identical
field names/semantics, allowing drop-in use of code written against the
OpenAI SDK response objects.
"""

from __future__ import annotations

from typing import Any, List, Literal, Optional

from pydantic import BaseModel, ConfigDict

FinishReason = Literal["stop", "length", "tool_calls", "content_filter", "function_call", "error"]


class _Compat(BaseModel):
    model_config = ConfigDict(extra="allow")


class FunctionCall(_Compat):
    arguments: str
    name: str


class Function(_Compat):
    arguments: str
    name: str


class ChatCompletionMessageToolCall(_Compat):
    id: str
    function: Function
    type: Literal["function"] = "function"


class ChatCompletionTokenLogprob(_Compat):
    token: str
    bytes: Optional[List[int]] = None
    logprob: float
    top_logprobs: List["TopLogprob"] = []


class TopLogprob(_Compat):
    token: str
    bytes: Optional[List[int]] = None
    logprob: float


class ChoiceLogprobs(_Compat):
    content: Optional[List[ChatCompletionTokenLogprob]] = None
    refusal: Optional[List[ChatCompletionTokenLogprob]] = None


class ChatCompletionMessage(_Compat):
    content: Optional[str] = None
    refusal: Optional[str] = None
    role: Literal["assistant"] = "assistant"
    function_call: Optional[FunctionCall] = None
    tool_calls: Optional[List[ChatCompletionMessageToolCall]] = None


class Choice(_Compat):
    finish_reason: FinishReason
    index: int
    logprobs: Optional[ChoiceLogprobs] = None
    message: ChatCompletionMessage


class PromptTokensDetails(_Compat):
    audio_tokens: Optional[int] = None
    cached_tokens: Optional[int] = None


class CompletionTokensDetails(_Compat):
    accepted_prediction_tokens: Optional[int] = None
    audio_tokens: Optional[int] = None
    reasoning_tokens: Optional[int] = None
    rejected_prediction_tokens: Optional[int] = None


class CompletionUsage(_Compat):
    completion_tokens: int
    prompt_tokens: int
    total_tokens: int
    completion_tokens_details: Optional[CompletionTokensDetails] = None
    prompt_tokens_details: Optional[PromptTokensDetails] = None


class ChatCompletion(_Compat):
    id: str
    choices: List[Choice]
    created: int
    model: str
    object: Literal["chat.completion"] = "chat.completion"
    service_tier: Optional[str] = None
    system_fingerprint: Optional[str] = None
    usage: Optional[CompletionUsage] = None


class ParsedChatCompletionMessage(ChatCompletionMessage):
    parsed: Optional[Any] = None


class ParsedChoice(Choice):
    message: ParsedChatCompletionMessage


class ParsedChatCompletion(ChatCompletion):
    choices: List[ParsedChoice]  # type: ignore[assignment]


class Embedding(_Compat):
    embedding: List[float]
    index: int
    object: Literal["embedding"] = "embedding"


class EmbeddingUsage(_Compat):
    prompt_tokens: int
    total_tokens: int


class CreateEmbeddingResponse(_Compat):
    data: List[Embedding]
    model: str
    object: Literal["list"] = "list"
    usage: Optional[EmbeddingUsage] = None


__all__ = [
    "ChatCompletion",
    "ChatCompletionMessage",
    "ChatCompletionMessageToolCall",
    "ChatCompletionTokenLogprob",
    "Choice",
    "ChoiceLogprobs",
    "CompletionTokensDetails",
    "CompletionUsage",
    "CreateEmbeddingResponse",
    "Embedding",
    "EmbeddingUsage",
    "FinishReason",
    "Function",
    "FunctionCall",
    "ParsedChatCompletion",
    "ParsedChatCompletionMessage",
    "ParsedChoice",
    "PromptTokensDetails",
    "TopLogprob",
]
