"""Llama-3 dense decoder for MI355X.

The model runner behind engine.generate() — the native replacement for the
reference's remote `chat.completions.create` model execution
(k_llms/resources/completions/completions.py:73). Torch modules orchestrate
hipBLASLt/rocBLAS GEMMs (via F.linear) while every non-GEMM hot op
(RMSNorm, RoPE, varlen prefill attention, paged decode attention, SwiGLU,
KV scatter) dispatches to the hand-written CDNA4 HIP kernels in kllms_amd.ops.

Two forward modes driven by ForwardBatch:
- prefill: packed varlen batch (shared prefill across requests), causal
  attention over the fresh contiguous K/V, K/V scattered into paged cache;
- decode: one token per stream against the paged KV cache (the hipGraph-
  captured hot loop).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Literal, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelArchConfig
from ..ops.torch_ref import build_cos_sin_cache
from ..parallel.tp import ColumnParallelLinear, ParallelContext, RowParallelLinear


@dataclass
class ForwardBatch:
    mode: Literal["prefill", "decode"]
    positions: torch.Tensor              # [T] long
    slot_mapping: torch.Tensor           # [T] long — flat KV slots to write
    kv_caches: List[Tuple]  # per layer (k, v, k_scale|None, v_scale|None)
    cu_seqlens: Optional[torch.Tensor] = None   # [n_seq+1] int32 (prefill)
    block_tables: Optional[torch.Tensor] = None  # [B, max_blocks] int32 (decode)
    context_lens: Optional[torch.Tensor] = None  # [B] int32 (decode)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelArchConfig, ctx: ParallelContext, layer_idx: int, dtype):
        super().__init__()
        self.layer_idx = layer_idx
        tp = ctx.world_size
        assert cfg.num_heads % tp == 0 and cfg.num_kv_heads % tp == 0
        self.num_heads = cfg.num_heads // tp
        self.num_kv_heads = cfg.num_kv_heads // tp
        self.head_dim = cfg.head_dim_
        self.scale = 1.0 / math.sqrt(self.head_dim)
        q_size = cfg.num_heads * self.head_dim
        kv_size = cfg.num_kv_heads * self.head_dim
        # Fused QKV, head-sharded across ranks (each rank holds its q, k and
        # v head shards — partition-aware slicing)
        self.qkv_proj = ColumnParallelLinear(
            cfg.hidden_size, q_size + 2 * kv_size, ctx, dtype=dtype,
            partition_sizes=[q_size, kv_size, kv_size],
            bias=cfg.attention_qkv_bias,  # Qwen2-style
        )
        self.o_proj = RowParallelLinear(q_size, cfg.hidden_size, ctx, dtype=dtype)
        self._q = self.num_heads * self.head_dim
        self._kv = self.num_kv_heads * self.head_dim

    def forward(self, hidden: torch.Tensor, batch: ForwardBatch, cos_sin: torch.Tensor) -> torch.Tensor:
        T = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        # strided views into the fused QKV row — the HIP kernels take the
        # token stride, so no contiguity copies on the hot path
        q, k, v = qkv.split([self._q, self._kv, self._kv], dim=-1)
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_kv_heads, self.head_dim)
        v = v.view(T, self.num_kv_heads, self.head_dim)
        ops.rope_inplace(q, k, batch.positions, cos_sin)

        k_cache, v_cache, k_scale, v_scale = batch.kv_caches[self.layer_idx]
        ops.store_kv(k, v, k_cache, v_cache, batch.slot_mapping, k_scale, v_scale)

        if batch.mode == "prefill":
            out = ops.attn_prefill_varlen(q, k, v, batch.cu_seqlens, self.scale)
        else:
            out = ops.attn_decode_paged(
                q, k_cache, v_cache, batch.block_tables, batch.context_lens, self.scale,
                k_scale, v_scale,
            )
        return self.o_proj(out.reshape(T, -1))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelArchConfig, ctx: ParallelContext, dtype):
        super().__init__()
        self.gate_up_proj = ColumnParallelLinear(
            cfg.hidden_size, 2 * cfg.intermediate_size, ctx, dtype=dtype,
            partition_sizes=[cfg.intermediate_size, cfg.intermediate_size],
        )
        self.down_proj = RowParallelLinear(cfg.intermediate_size, cfg.hidden_size, ctx, dtype=dtype)
        self._i = cfg.intermediate_size // ctx.world_size

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        gate_up = self.gate_up_proj(x)
        gate, up = gate_up.split([self._i, self._i], dim=-1)
        return self.down_proj(ops.silu_mul(gate, up))


class RMSNormModule(nn.Module):
    def __init__(self, hidden_size: int, eps: float, dtype):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(hidden_size, dtype=dtype), requires_grad=False)
        self.eps = eps


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelArchConfig, ctx: ParallelContext, layer_idx: int, dtype):
        super().__init__()
        self.input_layernorm = RMSNormModule(cfg.hidden_size, cfg.rms_norm_eps, dtype)
        self.self_attn = LlamaAttention(cfg, ctx, layer_idx, dtype)
        self.post_attention_layernorm = RMSNormModule(cfg.hidden_size, cfg.rms_norm_eps, dtype)
        self.mlp = self._make_mlp(cfg, ctx, dtype)

    def _make_mlp(self, cfg: ModelArchConfig, ctx: ParallelContext, dtype):
        return LlamaMLP(cfg, ctx, dtype)

    def forward(
        self, hidden: torch.Tensor, residual: Optional[torch.Tensor], batch: ForwardBatch, cos_sin: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        if residual is None:
            residual = hidden
            hidden = ops.rmsnorm(hidden, self.input_layernorm.weight, self.input_layernorm.eps)
        else:
            hidden, residual = ops.fused_add_rmsnorm(
                hidden, residual, self.input_layernorm.weight, self.input_layernorm.eps
            )
        hidden = self.self_attn(hidden, batch, cos_sin)
        hidden, residual = ops.fused_add_rmsnorm(
            hidden, residual, self.post_attention_layernorm.weight, self.post_attention_layernorm.eps
        )
        hidden = self.mlp(hidden)
        return hidden, residual


class LlamaForCausalLM(nn.Module):
    layer_cls = LlamaDecoderLayer

    def __init__(self, cfg: ModelArchConfig, ctx: Optional[ParallelContext] = None, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.ctx = ctx or ParallelContext()
        self.dtype = dtype
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        self.embed_tokens.weight.requires_grad_(False)
        self.layers = nn.ModuleList(
            [self.layer_cls(cfg, self.ctx, i, dtype) for i in range(cfg.num_layers)]
        )
        self.norm = RMSNormModule(cfg.hidden_size, cfg.rms_norm_eps, dtype)
        # LM head replicated across TP ranks: hidden states are replicated
        # after each layer's all-reduce, so full logits come out identical on
        # every rank — no logits gather needed (SURVEY §2.3 point 3).
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False, dtype=dtype)
        self.lm_head.weight.requires_grad_(False)
        self.register_buffer(
            "cos_sin",
            build_cos_sin_cache(cfg.head_dim_, cfg.max_position_embeddings, cfg.rope_theta, "cpu",
                                rope_scaling=cfg.rope_scaling),
            persistent=False,
        )

    # --- init ---------------------------------------------------------------
    @torch.no_grad()
    def random_init_(self, seed: int = 0) -> None:
        """Deterministic, TP-degree-invariant random init: every parameter is
        generated as its FULL tensor from a per-name seed, then the local
        shard is sliced out — so TP=1/2/4/8 hold numerically identical models.
        """
        import hashlib

        tp = self.ctx.world_size
        rank = self.ctx.rank
        for name, p in self.named_parameters():
            h = int(hashlib.md5(f"{seed}/{name}".encode()).hexdigest()[:15], 16)
            g = torch.Generator(device=p.device)
            g.manual_seed(h)
            shard_dim = None
            full_shape = list(p.shape)
            mod = self._owner_module(name)
            leaf = name.split(".")[-1]
            spec = getattr(mod, "shard_spec", {})
            if isinstance(mod, ColumnParallelLinear):
                shard_dim = 0
                full_shape[0] *= tp
                full = torch.ones(full_shape, dtype=torch.float32, device=p.device) if (
                    "layernorm" in name or name.endswith("norm.weight")
                ) else torch.randn(full_shape, generator=g, dtype=torch.float32, device=p.device) * 0.02
                p.copy_(mod.shard_full_tensor(full).to(p.dtype))
                continue
            elif isinstance(mod, RowParallelLinear):
                shard_dim = 1
                full_shape[1] *= tp
            n_parts = 1
            if leaf in spec:
                entry = spec[leaf]
                # spec value: dim, or (dim, n_parts) for FUSED dims (e.g. the
                # MoE [gate;up] stack) where each equal part shards separately
                shard_dim, n_parts = entry if isinstance(entry, tuple) else (entry, 1)
                full_shape[shard_dim] *= tp
            if "layernorm" in name or name.endswith("norm.weight"):
                full = torch.ones(full_shape, dtype=torch.float32, device=p.device)
            else:
                std = 0.02
                full = torch.randn(full_shape, generator=g, dtype=torch.float32, device=p.device) * std
            if shard_dim is not None and tp > 1:
                size = p.shape[shard_dim]
                if n_parts > 1:
                    # local layout is the concat of this rank's shard of EACH
                    # part — a contiguous narrow would give rank 0 all of the
                    # first part (matches weights.py's per-part sharding)
                    part_full = full.shape[shard_dim] // n_parts
                    part_loc = size // n_parts
                    full = torch.cat(
                        [full.narrow(shard_dim, j * part_full + rank * part_loc, part_loc)
                         for j in range(n_parts)],
                        dim=shard_dim,
                    )
                else:
                    full = full.narrow(shard_dim, rank * size, size)
            p.copy_(full.to(p.dtype))
        if self.cfg.tie_word_embeddings:
            # mirror the weights loader: tied checkpoints share one matrix
            self.lm_head.weight.copy_(self.embed_tokens.weight)

    def _owner_module(self, param_name: str):
        parts = param_name.split(".")[:-1]
        mod = self
        for part in parts:
            mod = getattr(mod, part) if not part.isdigit() else mod[int(part)]
        return mod

    # --- forward ------------------------------------------------------------
    def forward_hidden(self, input_ids: torch.Tensor, batch: ForwardBatch) -> torch.Tensor:
        hidden = self.embed_tokens(input_ids)
        residual = None
        cos_sin = self.cos_sin
        for layer in self.layers:
            hidden, residual = layer(hidden, residual, batch, cos_sin)
        hidden, _ = ops.fused_add_rmsnorm(hidden, residual, self.norm.weight, self.norm.eps)
        return hidden

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        return self.lm_head(hidden).float()

    def forward_prefill(self, input_ids: torch.Tensor, batch: ForwardBatch) -> torch.Tensor:
        """Returns logits for the LAST token of each packed sequence: [n_seq, V]."""
        hidden = self.forward_hidden(input_ids, batch)
        last_idx = (batch.cu_seqlens[1:] - 1).long()
        return self.compute_logits(hidden[last_idx])

    def forward_decode(self, input_ids: torch.Tensor, batch: ForwardBatch) -> torch.Tensor:
        """One token per stream: [B] ids -> [B, V] logits."""
        hidden = self.forward_hidden(input_ids, batch)
        return self.compute_logits(hidden)

    def to_device(self, device) -> "LlamaForCausalLM":
        self.to(device)
        self.cos_sin = self.cos_sin.to(device)
        return self
