"""Mixtral MoE decoder for MI355X.

Same attention/backbone as Llama (GQA, RoPE, RMSNorm); the MLP is a
top-k-routed mixture of experts. Experts are TP-SHARDED, not
expert-parallel — every rank holds a 1/tp slice of every expert, so routing
needs no all-to-all and the layer keeps the dense model's two all-reduces
("experts TP-sharded so no all-to-all").

Expert compute has two regimes: decode shapes run a DENSE batched GEMM over
all experts (same HBM traffic as routed compute when nearly every expert is
touched; hipGraph-friendly because shapes are data-independent), prefill
shapes bucket tokens per expert and run grouped GEMMs.

Router semantics match Mixtral: softmax over ALL expert logits (fp32), top-k
(default 2), renormalize the selected weights.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelArchConfig
from ..parallel.tp import ParallelContext
from .llama import LlamaDecoderLayer, LlamaForCausalLM


class MixtralMoE(nn.Module):
    # TP sharding of the stacked expert tensors (dim scaled by tp at full
    # size); w_gate_up's dim 1 is the FUSED [gate; up] stack — 2 parts, each
    # sharded separately (matches weights.py's per-part expert sharding)
    shard_spec = {"w_gate_up": (1, 2), "w_down": 2}

    def __init__(self, cfg: ModelArchConfig, ctx: ParallelContext, dtype):
        super().__init__()
        assert cfg.num_experts > 0
        self.ctx = ctx
        tp = ctx.world_size
        assert cfg.intermediate_size % tp == 0
        self.E = cfg.num_experts
        self.k = cfg.num_experts_per_tok
        self.I = cfg.intermediate_size // tp
        H = cfg.hidden_size
        self.gate = nn.Linear(H, self.E, bias=False, dtype=dtype)
        self.gate.weight.requires_grad_(False)
        self.w_gate_up = nn.Parameter(torch.empty(self.E, 2 * self.I, H, dtype=dtype), requires_grad=False)
        self.w_down = nn.Parameter(torch.empty(self.E, H, self.I, dtype=dtype), requires_grad=False)

    def load_expert_(self, e: int, which: str, t: torch.Tensor) -> None:
        if which == "gate_up":
            self.w_gate_up.data[e].copy_(t.to(self.w_gate_up.dtype))
        else:
            self.w_down.data[e].copy_(t.to(self.w_down.dtype))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        router_logits = self.gate(x).float()                       # [T, E]
        probs = torch.softmax(router_logits, dim=-1)
        topw, topi = torch.topk(probs, self.k, dim=-1)             # [T, k]
        topw = topw / topw.sum(dim=-1, keepdim=True)

        if T <= 8 * self.E:
            out = self._forward_dense(x, topw, topi)
        elif x.is_cuda:
            out = self._forward_grouped_hip(x, topw, topi)
        else:
            out = self._forward_grouped(x, topw, topi)
        return self.ctx.all_reduce(out)

    def _forward_dense(self, x: torch.Tensor, topw: torch.Tensor, topi: torch.Tensor) -> torch.Tensor:
        """Decode-shape path: with few tokens and top-k routing, (nearly)
        every expert's weights stream from HBM anyway, so computing ALL
        experts with batched GEMMs costs the same memory traffic as exact
        routing, needs no token bucketing, and replays cleanly in a hipGraph
        (no data-dependent shapes)."""
        T = x.shape[0]
        xb = x.unsqueeze(0).expand(self.E, T, -1)                     # [E, T, H]
        gu2 = torch.bmm(xb, self.w_gate_up.transpose(1, 2)).view(self.E * T, 2 * self.I)
        act = ops.silu_mul(gu2[:, : self.I], gu2[:, self.I:]).view(self.E, T, self.I)
        ye = torch.bmm(act, self.w_down.transpose(1, 2))              # [E, T, H]
        # combine: weight[e, t] = topw where topi == e else 0
        w = torch.zeros(self.E, T, device=x.device, dtype=torch.float32)
        w.scatter_(0, topi.t().long(), topw.t().float())
        return (ye.float() * w.unsqueeze(-1)).sum(dim=0).to(x.dtype)

    def _forward_grouped_hip(self, x: torch.Tensor, topw: torch.Tensor, topi: torch.Tensor) -> torch.Tensor:
        """Prefill-shape path on GPU: TWO grouped MFMA kernel launches
        (ops/hip/moe.hip) over tokens sorted by expert — gate/up GEMM with
        fused silu*mul, then the down GEMM — plus one index_add to scatter
        the routing-weighted rows back to token order. Replaces the
        per-expert loop (<=2 GEMM launches per layer)."""
        from .. import ops

        BM = 128
        T, H = x.shape
        dev = x.device
        flat_exp = topi.reshape(-1).to(torch.int32)               # [T*k]
        flat_tok = torch.arange(T, device=dev, dtype=torch.int32).repeat_interleave(self.k)
        flat_w = topw.reshape(-1)
        order = torch.argsort(flat_exp, stable=True)
        sorted_exp = flat_exp[order].long()
        sorted_tok = flat_tok[order]
        sorted_w = flat_w[order]

        counts = torch.bincount(sorted_exp, minlength=self.E)
        padded = (counts + BM - 1) // BM * BM                      # 0 stays 0
        pad_off = torch.zeros(self.E + 1, device=dev, dtype=torch.long)
        torch.cumsum(padded, 0, out=pad_off[1:])
        starts = torch.zeros(self.E, device=dev, dtype=torch.long)
        torch.cumsum(counts[:-1], 0, out=starts[1:])
        N = sorted_exp.numel()
        dest = (pad_off[sorted_exp] + torch.arange(N, device=dev) - starts[sorted_exp])

        S = int(pad_off[-1].item())                                # prefill-only host sync
        sorted_ids = torch.full((S,), -1, device=dev, dtype=torch.int32)
        sorted_ids[dest] = sorted_tok
        w_padded = torch.zeros(S, device=dev, dtype=flat_w.dtype)
        w_padded[dest] = sorted_w

        if getattr(self, "_zeros_page", None) is None or self._zeros_page.numel() < H:
            self._zeros_page = torch.zeros(H, device=dev, dtype=x.dtype)
        act = torch.empty(S, self.I, device=dev, dtype=x.dtype)
        y = torch.empty(S, H, device=dev, dtype=x.dtype)
        pad_off32 = pad_off.to(torch.int32)
        ops.moe_gateup(act, x, self.w_gate_up, sorted_ids, pad_off32, self._zeros_page)
        ops.moe_down(y, act, self.w_down, pad_off32)

        valid = sorted_ids >= 0
        out = torch.zeros(T, H, device=dev, dtype=torch.float32)
        out.index_add_(0, sorted_ids[valid].long(),
                       y[valid].float() * w_padded[valid].unsqueeze(1).float())
        return out.to(x.dtype)

    def _forward_grouped(self, x: torch.Tensor, topw: torch.Tensor, topi: torch.Tensor) -> torch.Tensor:
        """Prefill-shape path: tokens bucketed per expert, one GEMM group per
        expert (rocBLAS); avoids the dense path's E/k x FLOP overhead when
        T is large enough that compute, not weight streaming, dominates."""
        T = x.shape[0]
        out = torch.zeros_like(x)
        flat_exp = topi.reshape(-1)
        flat_tok = torch.arange(T, device=x.device).repeat_interleave(self.k)
        flat_w = topw.reshape(-1)
        for e in range(self.E):
            sel = (flat_exp == e).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                continue
            toks = flat_tok[sel]
            xe = x[toks]
            gu = torch.nn.functional.linear(xe, self.w_gate_up[e])
            g, u = gu.split([self.I, self.I], dim=-1)
            ye = torch.nn.functional.linear(ops.silu_mul(g, u), self.w_down[e])
            out.index_add_(0, toks, ye * flat_w[sel].unsqueeze(-1).to(ye.dtype))
        return out


class MixtralDecoderLayer(LlamaDecoderLayer):
    def _make_mlp(self, cfg: ModelArchConfig, ctx: ParallelContext, dtype):
        return MixtralMoE(cfg, ctx, dtype)


class MixtralForCausalLM(LlamaForCausalLM):
    layer_cls = MixtralDecoderLayer

    def load_expert_(self, layer: int, e: int, which: str, t: torch.Tensor) -> None:
        self.layers[layer].mlp.load_expert_(e, which, t)
