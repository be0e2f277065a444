"""KLLMs / AsyncKLLMs client facade.

Behavioral re-implementation of the reference facade (k_llms/client.py:15-72):
same class names, constructor shape, ``.chat.completions`` namespace,
``.client`` property and ``.get_embeddings`` closure — but the underlying
client is a local MI355X engine handle instead of ``openai.OpenAI``.
Inference (tokenize, shared prefill, n-way fanned decode, sampling,
constrained decoding, embeddings) runs locally; there is no network.

Embedding price accounting (k_llms/client.py:12-13,116) is kept
shape-compatible with price 0 (local compute is free).
"""

from __future__ import annotations

import asyncio
import os
from typing import Any, Awaitable, Callable, Optional

from .resources.completions import AsyncCompletions, Completions

# Embedding-model constants (reference: k_llms/client.py:12-13). Locally both
# names resolve to the engine's embedder; pricing is zero.
MAX_TOKENS_PER_MODEL = {"text-embedding-3-small": 8191, "text-embedding-3-large": 8191}
PRICING = {"text-embedding-3-small": 0.0, "text-embedding-3-large": 0.0}


class BaseEngineWrapper:
    """Holds engine configuration (reference C1: k_llms/client.py:15-28).

    ``api_key`` / ``base_url`` / ``timeout`` / ``max_retries`` are accepted for
    drop-in compatibility with the reference constructor; they are unused by
    the local engine. Everything else in ``**kwargs`` is forwarded to the
    engine client (model path/preset, dtype, tp degree, kv block size, ...).
    """

    def __init__(
        self,
        api_key: Optional[str] = None,
        base_url: Optional[str] = None,
        timeout: Optional[float] = None,
        max_retries: int = 2,
        **kwargs: Any,
    ):
        self.api_key = api_key or os.environ.get("OPENAI_API_KEY")
        self.base_url = base_url
        self.timeout = timeout
        self.max_retries = max_retries
        # "similarity" (default) or "key" — selects the alignment engine
        # (reference: the commented import swap at consolidation.py:22)
        self.consensus_aligner = kwargs.pop("consensus_aligner", "similarity")
        self._extra_kwargs = kwargs

    def _make_engine_client(self):
        from .engine.api import LocalEngineClient

        engine = self._extra_kwargs.pop("engine", None)
        if engine is not None and isinstance(engine, LocalEngineClient):
            self._owns_engine = False
            return engine
        self._owns_engine = True
        return LocalEngineClient(**self._extra_kwargs)

    def close(self) -> None:
        """Stop the continuous-batching scheduler thread, if one was started
        (OpenAI clients expose close(); engines shared via ``engine=`` are
        left running for their owner)."""
        client = getattr(self, "_client", None)
        if client is not None and getattr(self, "_owns_engine", False):
            sched = getattr(client, "_scheduler", None)
            if sched is not None:
                sched.shutdown()
                client._scheduler = None

    def __enter__(self):
        return self

    def __exit__(self, *exc) -> None:
        self.close()


class KLLMs(BaseEngineWrapper):
    """Sync facade (reference: k_llms/client.py:31-44)."""

    def __init__(self, **kwargs: Any):
        super().__init__(**kwargs)
        self._client = self._make_engine_client()
        self.chat = Chat(self)
        self.get_embeddings: Callable[[list[str], str, int, bool], list[list[float]]] = (
            lambda texts, model, batch_size, verbose: get_embeddings(self._client, texts, model, batch_size, verbose)
        )

    @property
    def client(self):
        return self._client


class AsyncKLLMs(BaseEngineWrapper):
    """Async facade (reference: k_llms/client.py:47-60). Shares the same local
    engine; requests are submitted through worker threads into the engine's
    batch scheduler."""

    def __init__(self, **kwargs: Any):
        super().__init__(**kwargs)
        self._client = self._make_engine_client()
        self.chat = AsyncChat(self)
        self.get_embeddings: Callable[[list[str], str, int, bool], Awaitable[list[list[float]]]] = (
            lambda texts, model, batch_size, verbose: async_get_embeddings(
                self._client, texts, model, batch_size, verbose
            )
        )

    @property
    def client(self):
        return self._client

    async def aclose(self) -> None:
        self.close()

    async def __aenter__(self):
        return self

    async def __aexit__(self, *exc) -> None:
        self.close()


class Chat:
    """Namespace shim: ``.completions`` (reference: k_llms/client.py:63-66)."""

    def __init__(self, wrapper: "KLLMs"):
        self._wrapper = wrapper
        self.completions = Completions(wrapper)


class AsyncChat:
    def __init__(self, wrapper: "AsyncKLLMs"):
        self._wrapper = wrapper
        self.completions = AsyncCompletions(wrapper)


def get_embeddings(
    engine_client,
    texts: list[str],
    model: str = "text-embedding-3-small",
    batch_size: int = 2048,
    verbose: bool = False,
) -> list[list[float]]:
    """Batched local embeddings (reference: k_llms/client.py:75-122).

    Same signature and batching behavior; token cropping uses the engine's own
    tokenizer instead of tiktoken; price is always zero locally.
    """
    if model not in MAX_TOKENS_PER_MODEL:
        raise ValueError(f"Model {model} not supported. Available models: {list(MAX_TOKENS_PER_MODEL.keys())}")

    max_tokens = MAX_TOKENS_PER_MODEL[model]
    processed_texts = [engine_client.crop_to_tokens(text, max_tokens) for text in texts]

    embeddings: list[list[float]] = []
    total_price = 0.0
    for idx in range(0, len(processed_texts), batch_size):
        batch = processed_texts[idx : idx + batch_size]
        response = engine_client.embeddings.create(input=batch, model=model)
        if response.usage is not None:
            total_price += response.usage.prompt_tokens * PRICING[model] / 1_000_000.0
        embeddings.extend([x.embedding for x in response.data])

    if verbose:
        print(f"TOTAL PRICE: ${total_price:.6f}")
    return embeddings


async def async_get_embeddings(
    engine_client,
    texts: list[str],
    model: str = "text-embedding-3-small",
    batch_size: int = 2048,
    verbose: bool = False,
) -> list[list[float]]:
    """Async twin (reference: k_llms/client.py:125-196). The reference's
    selective-crop heuristic (crop only when len(text)*3 > max_tokens,
    ref :149-152) and its retry-with-full-crop fallback (ref :177-191) are
    kept, with the crop work offloaded to a thread."""
    if model not in MAX_TOKENS_PER_MODEL:
        raise ValueError(f"Model {model} not supported. Available models: {list(MAX_TOKENS_PER_MODEL.keys())}")

    max_tokens = MAX_TOKENS_PER_MODEL[model]

    def selective_crop(ts: list[str]) -> list[str]:
        return [
            engine_client.crop_to_tokens(t, max_tokens) if len(t) * 3 > max_tokens else t for t in ts
        ]

    def encode_and_crop(ts: list[str]) -> list[str]:
        return [engine_client.crop_to_tokens(t, max_tokens) for t in ts]

    async def run_batches(processed: list[str]) -> tuple[list[list[float]], float]:
        embs: list[list[float]] = []
        price = 0.0
        for idx in range(0, len(processed), batch_size):
            batch = processed[idx : idx + batch_size]
            response = await asyncio.to_thread(engine_client.embeddings.create, input=batch, model=model)
            if response.usage is not None:
                price += response.usage.prompt_tokens * PRICING[model] / 1_000_000.0
            embs.extend([x.embedding for x in response.data])
        return embs, price

    processed_texts = await asyncio.to_thread(selective_crop, texts)
    try:
        embeddings, total_price = await run_batches(processed_texts)
    except Exception as e:
        if verbose:
            print(f"Embedding request failed with error: {e}. Retrying with all strings cropped.")
        processed_texts = await asyncio.to_thread(encode_and_crop, texts)
        embeddings, total_price = await run_batches(processed_texts)

    if verbose:
        print(f"TOTAL PRICE: ${total_price:.6f}")
    return embeddings
