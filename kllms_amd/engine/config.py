"""Engine and model-architecture configuration.

The reference has a default-only consensus config; the native
engine adds EngineConfig — model preset/path, dtype, TP degree, KV block
size, hipGraph toggle — exposed through ``KLLMs(**kwargs)``.

Presets cover the BASELINE.json configs: Llama-3-8B (TP=1 headline),
Llama-3-70B (TP=8), Mixtral-8x7B (MoE), plus tiny CPU-testable variants.
"""

from __future__ import annotations

from typing import List, Literal, Optional, Union

from pydantic import BaseModel, Field


class ModelArchConfig(BaseModel):
    """Transformer architecture hyperparameters (Llama/Mixtral family)."""

    arch: Literal["llama", "mixtral"] = "llama"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: Optional[int] = None  # defaults to hidden_size // num_heads
    rope_theta: float = 500000.0
    rms_norm_eps: float = 1e-5
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = False
    # Qwen2-family: bias on the fused QKV projection (o_proj stays bias-free)
    attention_qkv_bias: bool = False
    # HF rope_scaling dict (Llama-3.1 "llama3" banded NTK, "linear");
    # None = plain RoPE
    rope_scaling: Optional[dict] = None
    # MoE (mixtral only)
    num_experts: int = 0
    num_experts_per_tok: int = 2

    @property
    def head_dim_(self) -> int:
        return self.head_dim or (self.hidden_size // self.num_heads)


MODEL_PRESETS: dict[str, ModelArchConfig] = {
    # Llama-3-8B — the BASELINE headline config
    "llama-3-8b": ModelArchConfig(
        arch="llama", vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_layers=32, num_heads=32, num_kv_heads=8, rope_theta=500000.0,
    ),
    # Llama-3.1: same dims, banded-NTK rope scaling to 128k context
    "llama-3.1-8b": ModelArchConfig(
        arch="llama", vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_layers=32, num_heads=32, num_kv_heads=8, rope_theta=500000.0,
        max_position_embeddings=131072,
        rope_scaling={"rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
                      "high_freq_factor": 4.0, "original_max_position_embeddings": 8192},
    ),
    "llama-3-70b": ModelArchConfig(
        arch="llama", vocab_size=128256, hidden_size=8192, intermediate_size=28672,
        num_layers=80, num_heads=64, num_kv_heads=8, rope_theta=500000.0,
    ),
    "mixtral-8x7b": ModelArchConfig(
        arch="mixtral", vocab_size=32000, hidden_size=4096, intermediate_size=14336,
        num_layers=32, num_heads=32, num_kv_heads=8, rope_theta=1e6,
        num_experts=8, num_experts_per_tok=2, rms_norm_eps=1e-5,
    ),
    # Mid preset for GPU kernel/engine tests: real head_dim (128), small depth
    "mid-llama": ModelArchConfig(
        arch="llama", vocab_size=2048, hidden_size=512, intermediate_size=1024,
        num_layers=2, num_heads=4, num_kv_heads=2, rope_theta=10000.0,
        max_position_embeddings=2048,
    ),
    "mid-qwen": ModelArchConfig(
        arch="llama", vocab_size=2048, hidden_size=512, intermediate_size=1024,
        num_layers=2, num_heads=4, num_kv_heads=2, rope_theta=10000.0,
        max_position_embeddings=2048, attention_qkv_bias=True,
        tie_word_embeddings=True,
    ),
    "mid-mixtral": ModelArchConfig(
        arch="mixtral", vocab_size=2048, hidden_size=512, intermediate_size=1024,
        num_layers=2, num_heads=4, num_kv_heads=2, rope_theta=10000.0,
        num_experts=4, num_experts_per_tok=2, max_position_embeddings=2048,
    ),
    # Tiny CPU-testable presets (same code paths, toy sizes)
    "tiny-llama": ModelArchConfig(
        arch="llama", vocab_size=512, hidden_size=64, intermediate_size=128,
        num_layers=2, num_heads=4, num_kv_heads=2, rope_theta=10000.0,
        max_position_embeddings=512,
    ),
    # tiny variant with 4 KV heads so gloo TP=4 tests divide evenly
    "tiny-llama-kv4": ModelArchConfig(
        arch="llama", vocab_size=512, hidden_size=128, intermediate_size=256,
        num_layers=2, num_heads=8, num_kv_heads=4, rope_theta=10000.0,
        max_position_embeddings=512,
    ),
    "tiny-mixtral": ModelArchConfig(
        arch="mixtral", vocab_size=512, hidden_size=64, intermediate_size=128,
        num_layers=2, num_heads=4, num_kv_heads=2, rope_theta=10000.0,
        num_experts=4, num_experts_per_tok=2, max_position_embeddings=512,
    ),
    # Qwen2 family: Llama compute graph + QKV bias (+ tied embeddings on
    # the small members); head_dim 128 so every HIP kernel applies as-is
    "qwen2-7b": ModelArchConfig(
        arch="llama", vocab_size=152064, hidden_size=3584, intermediate_size=18944,
        num_layers=28, num_heads=28, num_kv_heads=4, rope_theta=1e6,
        rms_norm_eps=1e-6, max_position_embeddings=32768, attention_qkv_bias=True,
    ),
    "tiny-qwen": ModelArchConfig(
        arch="llama", vocab_size=512, hidden_size=64, intermediate_size=128,
        num_layers=2, num_heads=4, num_kv_heads=2, rope_theta=10000.0,
        max_position_embeddings=512, attention_qkv_bias=True,
        tie_word_embeddings=True,
    ),
}


class EngineConfig(BaseModel):
    """Runtime configuration for the local engine."""

    model: str = "llama-3-8b"          # preset name or path to a weights dir
    weights_path: Optional[str] = None  # safetensors dir; None = random init
    dtype: Literal["bfloat16", "float32", "float16"] = "bfloat16"
    seed: int = 0

    # Parallelism: TP over RCCL/xGMI (one process per GPU)
    tp_size: int = 1

    # KV cache
    kv_block_size: int = 16
    # "bf16" (default, matches compute dtype) or "fp8_e4m3" (half the KV
    # traffic/footprint; OCP e4m3 with PER-ROW dequant scales — each
    # (token, head) row is stored as x/s with s = amax(|row|)/448 kept in a
    # parallel fp32 [NB, KVH, BS] tensor (~3% overhead), so outlier-heavy
    # real checkpoints don't saturate e4m3's +-448 range)
    kv_cache_dtype: Literal["bf16", "fp8_e4m3"] = "bf16"
    # Fraction of free HBM given to the KV cache after weights are resident
    kv_memory_fraction: float = 0.70
    max_kv_blocks: Optional[int] = None  # explicit cap (used on CPU/tests)

    max_seq_len: int = 8192
    max_batch_size: int = 256

    # embeddings.create backend: "token_mean" mean-pools the model's token
    # embeddings (semantically meaningful once real weights are loaded);
    # "ngram" is a deterministic signed char-3-gram hashing embedder whose
    # cosine tracks STRING similarity with no weights at all; "auto" picks
    # ngram under random init and token_mean when a checkpoint is loaded
    embedding_mode: Literal["auto", "token_mean", "ngram"] = "auto"

    # Decode-step hipGraph capture
    use_hip_graphs: bool = True
    # constrained decoding: allow optional JSON whitespace between tokens
    # (default emits compact JSON — smaller DFA, fewer wasted tokens)
    constrained_whitespace: bool = False
    # cross-request prefix caching: full prompt blocks are published to a
    # digest-keyed cache; later prompts sharing the prefix skip re-prefilling
    # it (LRU + eviction under allocator pressure)
    enable_prefix_caching: bool = True
    prefix_cache_fraction: float = 0.5  # max fraction of KV blocks held
    # ignore hits shorter than this: a short cached head saves less prefill
    # than the tail's decode-mode forward costs over the packed kernel
    prefix_cache_min_tokens: int = 128

    # scheduler: prefill prompts longer than this in slices interleaved with
    # decode steps, so a long prompt doesn't stall running streams
    # (None = whole-prompt prefill at admission)
    prefill_chunk_tokens: Optional[int] = None
    hip_graph_batch_sizes: list[int] = Field(default_factory=lambda: [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 160, 192, 224, 256])

    # Generation defaults
    default_max_new_tokens: int = 512
    # default: the checkpoint's generation_config.json eos_token_id (which
    # may be a LIST — Llama-3 stops on <|eot_id|> OR <|end_of_text|>),
    # else the tokenizer's
    eos_token_id: Optional[Union[int, List[int]]] = None

    # Device ("cuda" is ROCm/HIP under torch-rocm; "cpu" for tests)
    device: Optional[str] = None

    def effective_weights_dir(self) -> Optional[str]:
        """The directory to load safetensors from: explicit weights_path, or
        the model path itself when it is a directory holding safetensors
        (README contract: model = preset name OR path to a weights dir).
        None = deterministic random init."""
        import glob as _glob
        import os as _os

        if self.weights_path:
            return self.weights_path
        if _os.path.isdir(str(self.model)) and _glob.glob(_os.path.join(self.model, "*.safetensors")):
            return self.model
        return None

    def resolve_arch(self) -> ModelArchConfig:
        if self.model in MODEL_PRESETS:
            return MODEL_PRESETS[self.model]
        # A path: read config.json (HF-style) next to the weights
        import json
        import os

        cfg_path = os.path.join(self.weights_path or self.model, "config.json")
        if not os.path.exists(cfg_path):
            # OpenAI-style model-not-found error instead of a raw ENOENT
            raise ValueError(
                f"model {self.model!r} is neither a preset "
                f"({', '.join(sorted(MODEL_PRESETS))}) nor a directory with a "
                f"config.json (model_not_found)"
            )
        with open(cfg_path) as f:
            hf = json.load(f)
        arch = "mixtral" if "mixtral" in hf.get("model_type", "").lower() or hf.get("num_local_experts") else "llama"
        return ModelArchConfig(
            arch=arch,
            vocab_size=hf["vocab_size"],
            hidden_size=hf["hidden_size"],
            intermediate_size=hf["intermediate_size"],
            num_layers=hf["num_hidden_layers"],
            num_heads=hf["num_attention_heads"],
            num_kv_heads=hf.get("num_key_value_heads", hf["num_attention_heads"]),
            rope_theta=hf.get("rope_theta", 10000.0),
            rms_norm_eps=hf.get("rms_norm_eps", 1e-5),
            max_position_embeddings=hf.get("max_position_embeddings", 8192),
            tie_word_embeddings=hf.get("tie_word_embeddings", False),
            num_experts=hf.get("num_local_experts", 0),
            num_experts_per_tok=hf.get("num_experts_per_tok", 2),
            # Qwen2 configs carry no attention_bias key but always use QKV
            # bias; Llama-style configs say so explicitly
            attention_qkv_bias=hf.get(
                "attention_bias", hf.get("model_type", "").lower() == "qwen2"),
            rope_scaling=hf.get("rope_scaling"),
        )
