"""kllms_amd — MI355X-native self-consistency inference framework.

A from-scratch AMD Instinct MI355X (gfx950 / CDNA4) implementation of the
k-LLMs self-consistency client (reference: retab-dev/k-LLMs, k_llms/__init__.py:1-3):
the same ``KLLMs`` / ``AsyncKLLMs`` API and return types, but the n completions
are produced by a local inference engine (shared prefill + fanned decode,
hand-written HIP/CDNA4 kernels, paged KV cache in HBM3E, RCCL tensor
parallelism over xGMI) instead of a remote OpenAI endpoint, and the
consolidation/consensus math runs locally (on-device for the batched
similarity paths).
"""

from .client import KLLMs, AsyncKLLMs

__version__ = "0.1.0"

__all__ = ["KLLMs", "AsyncKLLMs"]
