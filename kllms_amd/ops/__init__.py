"""Op dispatch: HIP/CDNA4 kernels on GPU, torch reference on CPU.

Policy (BASELINE north star): on a GPU (`tensor.is_cuda`, which on
PyTorch-ROCm means an AMD GPU) the hand-written gfx950 HIP kernels in
``kllms_amd._C`` are the ONLY path — if the extension is missing the op
RAISES instead of silently falling back to eager PyTorch. On CPU the pure
torch reference implementations (torch_ref.py) run, which is what the
GPU-less CI environment uses.

Set ``KLLMS_AMD_FORCE_TORCH_OPS=1`` to force the torch path on GPU (used only
by numerics tests to produce the oracle on-device).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from . import torch_ref

_C = None
_C_err: Optional[str] = None
try:
    from kllms_amd import _C  # type: ignore[no-redef]  # built by setup.py build_ext --inplace
except Exception as e:  # pragma: no cover - import-time probe
    _C = None
    _C_err = str(e)


def hip_available() -> bool:
    return _C is not None


def _force_torch() -> bool:
    return os.environ.get("KLLMS_AMD_FORCE_TORCH_OPS", "0") == "1"


def _hip_or_raise():
    if _C is None:
        raise RuntimeError(
            "kllms_amd HIP extension (kllms_amd/_C*.so) is not built but a GPU "
            "tensor was passed. Build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_C_err}"
        )
    return _C


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda and not _force_torch():
        out = torch.empty_like(x)
        _hip_or_raise().rmsnorm(out, x, weight, eps)
        return out
    return torch_ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    if x.is_cuda and not _force_torch():
        # in-place: residual += x; x_out = rmsnorm(residual)
        out = torch.empty_like(x)
        _hip_or_raise().fused_add_rmsnorm(out, x, residual, weight, eps)
        return out, residual
    return torch_ref.fused_add_rmsnorm(x, residual, weight, eps)


def rope_inplace(q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor, cos_sin: torch.Tensor) -> None:
    if q.is_cuda and not _force_torch():
        _hip_or_raise().rope_inplace(q, k, positions, cos_sin)
        return
    torch_ref.rope_inplace(q, k, positions, cos_sin)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if gate.is_cuda and not _force_torch():
        out = torch.empty(gate.shape, dtype=gate.dtype, device=gate.device)
        _hip_or_raise().silu_mul(out, gate, up)
        return out
    return torch_ref.silu_mul(gate, up)


def store_kv(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> None:
    """k_scale/v_scale: [NB, KVH, BS] fp32 per-row dequant scales, REQUIRED
    when the caches are fp8 (written here: s = amax(|row|)/448), None for bf16."""
    if k.is_cuda and not _force_torch():
        none = torch.empty(0, dtype=torch.float32, device=k.device)
        _hip_or_raise().store_kv(
            k, v, k_cache, v_cache, slot_mapping,
            k_scale if k_scale is not None else none,
            v_scale if v_scale is not None else none)
        return
    torch_ref.store_kv(k, v, k_cache, v_cache, slot_mapping, k_scale, v_scale)


def _prefill_tiles(cu_seqlens: torch.Tensor, device) -> tuple:
    """Grouped q-tile metadata for the MFMA prefill kernel: 32-row q-tiles of
    each sequence packed 8 per workgroup (the 8 waves share the K/V LDS
    staging); idle wave slots padded with -1. One host round-trip per prefill
    batch (once per generate() call)."""
    NW = 8
    cu = cu_seqlens.cpu().tolist()
    grp_start, grp_len, grp_qpos0 = [], [], []
    for i in range(len(cu) - 1):
        s, e = cu[i], cu[i + 1]
        L = e - s
        tiles = list(range(0, L, 32))
        for g0 in range(0, len(tiles), NW):
            grp_start.append(s)
            grp_len.append(L)
            grp = tiles[g0 : g0 + NW]
            grp_qpos0.extend(grp + [-1] * (NW - len(grp)))
    t = lambda x: torch.tensor(x, dtype=torch.int32, device=device)
    return t(grp_start), t(grp_qpos0), t(grp_len)


def attn_prefill_varlen(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, cu_seqlens: torch.Tensor, scale: float
) -> torch.Tensor:
    if q.is_cuda and not _force_torch():
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        gs, gq, gl = _prefill_tiles(cu_seqlens, q.device)
        _hip_or_raise().attn_prefill(out, q, k, v, gs, gq, gl, scale)
        return out
    return torch_ref.attn_prefill_varlen(q, k, v, cu_seqlens, scale)


def attn_decode_paged(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    if q.is_cuda and not _force_torch():
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        none = torch.empty(0, dtype=torch.float32, device=q.device)
        _hip_or_raise().attn_decode_paged(
            out, q, k_cache, v_cache, block_tables, context_lens, scale,
            k_scale if k_scale is not None else none,
            v_scale if v_scale is not None else none)
        return out
    return torch_ref.attn_decode_paged(q, k_cache, v_cache, block_tables, context_lens, scale,
                                       k_scale, v_scale)


def sample(
    logits: torch.Tensor,
    temperatures: torch.Tensor,
    top_ps: torch.Tensor,
    top_ks: torch.Tensor,
    seeds: torch.Tensor,
    steps: torch.Tensor,
    mask: Optional[torch.Tensor] = None,
    dbg: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    if logits.is_cuda and not _force_torch():
        B = logits.shape[0]
        tokens = torch.empty(B, dtype=torch.long, device=logits.device)
        logprobs = torch.empty(B, dtype=torch.float32, device=logits.device)
        _hip_or_raise().sample(
            tokens, logprobs, logits, temperatures, top_ps, top_ks, seeds, steps,
            mask if mask is not None else torch.empty(0, dtype=torch.int32, device=logits.device),
            dbg if dbg is not None else torch.empty(0, dtype=torch.float32, device=logits.device),
        )
        return tokens, logprobs
    return torch_ref.sample(logits, temperatures, top_ps, top_ks, seeds, steps, mask)


def moe_gateup(
    act: torch.Tensor,
    x: torch.Tensor,
    w_gate_up: torch.Tensor,
    sorted_ids: torch.Tensor,
    pad_offsets: torch.Tensor,
    zeros: torch.Tensor,
) -> None:
    """Grouped MoE gate/up GEMM + fused silu*mul into `act` (sorted-by-expert
    row space). GPU-only (Mixtral prefill hot path); the CPU path keeps the
    per-expert torch loop in models/mixtral.py."""
    _hip_or_raise().moe_gateup(act, x, w_gate_up, sorted_ids, pad_offsets, zeros)


def moe_down(
    y: torch.Tensor,
    act: torch.Tensor,
    w_down: torch.Tensor,
    pad_offsets: torch.Tensor,
) -> None:
    """Grouped MoE down GEMM in sorted-row space (see moe_gateup)."""
    _hip_or_raise().moe_down(y, act, w_down, pad_offsets)


def levenshtein_pairs(
    chars: torch.Tensor,
    lens: torch.Tensor,
    pair_i: torch.Tensor,
    pair_j: torch.Tensor,
) -> torch.Tensor:
    """Batched Myers bit-parallel edit distance over packed [N, 64] uint8
    normalized strings; returns int32 distances per (i, j) pair. GPU-only —
    the CPU consensus path keeps the pure-python DP (utils/text.py)."""
    return _hip_or_raise().levenshtein_pairs(chars, lens, pair_i, pair_j)
