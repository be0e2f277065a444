"""Local MI355X inference engine: tokenizer, model runner, paged KV cache,
shared-prefill/fanned-decode scheduler, sampling, constrained decoding and
embeddings — the native replacement for every remote OpenAI capability the
reference invokes."""

from .config import EngineConfig, ModelArchConfig, MODEL_PRESETS
from .sampling import SamplingParams

__all__ = ["EngineConfig", "ModelArchConfig", "MODEL_PRESETS", "SamplingParams"]
