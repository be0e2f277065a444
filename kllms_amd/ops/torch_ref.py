"""Pure-PyTorch reference implementations of every engine op.

These are the numerics oracles for the HIP/CDNA4 kernels (tests compare the
kernels against these in fp32) and the CPU execution path for GPU-less test
environments. On a GPU box the dispatch layer (ops/__init__.py) routes to the
HIP kernels and refuses to fall back silently.

Conventions:
- q/k/v activations: [tokens, heads, head_dim]
- paged KV cache: [num_blocks, kv_heads, block_size, head_dim]
- cos/sin cache: [max_pos, head_dim] with cos in [:, :D/2], sin in [:, D/2:]
- logits: [batch, vocab]
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


# --- normalization ------------------------------------------------------------

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    norm = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (norm * weight.float()).to(x.dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    """residual' = x + residual; y = rmsnorm(residual')."""
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps), new_residual


# --- rotary embedding ---------------------------------------------------------

def build_cos_sin_cache(head_dim: int, max_pos: int, theta: float, device,
                        dtype=torch.float32, rope_scaling=None) -> torch.Tensor:
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64, device=device) / half))
    if rope_scaling:
        rt = rope_scaling.get("rope_type") or rope_scaling.get("type")
        if rt == "llama3":
            # Llama-3.1 frequency-banded NTK scaling: low-frequency bands
            # (wavelength > old context) are divided by `factor`; high-
            # frequency bands are untouched; the band between interpolates.
            import math as _math

            factor = float(rope_scaling.get("factor", 8.0))
            lo_f = float(rope_scaling.get("low_freq_factor", 1.0))
            hi_f = float(rope_scaling.get("high_freq_factor", 4.0))
            old_ctx = float(rope_scaling.get("original_max_position_embeddings", 8192))
            wavelen = 2 * _math.pi / inv_freq
            lo_wl = old_ctx / lo_f
            hi_wl = old_ctx / hi_f
            smooth = ((old_ctx / wavelen - lo_f) / (hi_f - lo_f)).clamp(0.0, 1.0)
            scaled = (1 - smooth) * inv_freq / factor + smooth * inv_freq
            inv_freq = torch.where(wavelen > lo_wl, inv_freq / factor,
                                   torch.where(wavelen < hi_wl, inv_freq, scaled))
        elif rt == "linear":
            inv_freq = inv_freq / float(rope_scaling.get("factor", 1.0))
        # unknown types: serve unscaled rather than failing (documented)
    t = torch.arange(max_pos, dtype=torch.float64, device=device)
    freqs = torch.outer(t, inv_freq)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(dtype)


def rope_inplace(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor, cos_sin: torch.Tensor
) -> None:
    """Llama-style (rotate-half over contiguous halves) rotary embedding,
    applied in place to q [T, H, D] and k [T, KVH, D]."""
    D = q.shape[-1]
    half = D // 2
    cs = cos_sin[positions]  # [T, D]
    cos = cs[:, :half].unsqueeze(1).float()  # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1).float()
    for t in (q, k):
        tf = t.float()
        x1 = tf[..., :half]
        x2 = tf[..., half:]
        t[..., :half] = (x1 * cos - x2 * sin).to(t.dtype)
        t[..., half:] = (x2 * cos + x1 * sin).to(t.dtype)


# --- activations --------------------------------------------------------------

def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


# --- paged KV cache -----------------------------------------------------------

def store_kv(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
    k_scale=None,
    v_scale=None,
) -> None:
    """Scatter new K/V ([T, KVH, D]) into the paged caches at flat slots.

    A flat slot s addresses (block = s // block_size, offset = s % block_size).
    fp8 caches with k_scale/v_scale ([NB, KVH, BS] fp32): each (token, head)
    row is quantized with its OWN scale s = max(amax(|row|)/448, 1e-8), the
    scale written alongside — outlier rows no longer saturate e4m3.
    """
    num_blocks, kv_heads, block_size, head_dim = k_cache.shape
    blk = slot_mapping // block_size
    off = slot_mapping % block_size
    if k_scale is not None:
        sk = (k.float().abs().amax(-1) / 448.0).clamp_(min=1e-8)  # [T, KVH]
        sv = (v.float().abs().amax(-1) / 448.0).clamp_(min=1e-8)
        k_scale[blk, :, off] = sk
        v_scale[blk, :, off] = sv
        k_cache[blk, :, off, :] = (k.float() / sk.unsqueeze(-1)).to(k_cache.dtype)
        v_cache[blk, :, off, :] = (v.float() / sv.unsqueeze(-1)).to(v_cache.dtype)
        return
    k_cache[blk, :, off, :] = k.to(k_cache.dtype)
    v_cache[blk, :, off, :] = v.to(v_cache.dtype)


# --- attention ----------------------------------------------------------------

def attn_prefill_varlen(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    scale: float,
) -> torch.Tensor:
    """Causal varlen attention over packed sequences.

    q: [T, H, D]; k/v: [T, KVH, D] (GQA: H % KVH == 0). Returns [T, H, D].
    """
    T, H, D = q.shape
    KVH = k.shape[1]
    rep = H // KVH
    out = torch.empty_like(q)
    cu = cu_seqlens.tolist()
    for i in range(len(cu) - 1):
        s, e = cu[i], cu[i + 1]
        qi = q[s:e].float().permute(1, 0, 2)  # [H, L, D]
        ki = k[s:e].float().repeat_interleave(rep, dim=1).permute(1, 0, 2)
        vi = v[s:e].float().repeat_interleave(rep, dim=1).permute(1, 0, 2)
        o = torch.nn.functional.scaled_dot_product_attention(
            qi, ki, vi, is_causal=True, scale=scale
        )
        out[s:e] = o.permute(1, 0, 2).to(q.dtype)
    return out


def attn_decode_paged(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
    k_scale=None,
    v_scale=None,
) -> torch.Tensor:
    """Single-token decode attention against the paged KV cache.

    q: [B, H, D]; caches [num_blocks, KVH, block_size, D];
    block_tables: [B, max_blocks] int32; context_lens: [B] (length INCLUDING
    the token being decoded, whose K/V are already stored). k_scale/v_scale
    ([NB, KVH, BS] fp32, fp8 caches only): per-row dequant multipliers.
    """
    B, H, D = q.shape
    _, KVH, BS, _ = k_cache.shape
    rep = H // KVH
    out = torch.empty_like(q)
    for b in range(B):
        L = int(context_lens[b])
        nblk = (L + BS - 1) // BS
        blocks = block_tables[b, :nblk].long()
        kk = k_cache[blocks].permute(1, 0, 2, 3).reshape(KVH, nblk * BS, D)[:, :L].float()
        vv = v_cache[blocks].permute(1, 0, 2, 3).reshape(KVH, nblk * BS, D)[:, :L].float()
        if k_scale is not None:
            kk = kk * k_scale[blocks].permute(1, 0, 2).reshape(KVH, nblk * BS)[:, :L, None]
            vv = vv * v_scale[blocks].permute(1, 0, 2).reshape(KVH, nblk * BS)[:, :L, None]
        qb = q[b].float()  # [H, D]
        # per-head GQA mapping without materializing repeats
        for h in range(H):
            g = h // rep
            s = (qb[h] @ kk[g].T) * scale  # [L]
            p = torch.softmax(s, dim=-1)
            out[b, h] = (p @ vv[g]).to(q.dtype)
    return out


# --- sampling -----------------------------------------------------------------

def _stream_generator(seed: int, step: int, device) -> torch.Generator:
    g = torch.Generator(device="cpu")
    g.manual_seed((seed * 0x9E3779B97F4A7C15 + step * 0xBF58476D1CE4E5B9) % (2**63))
    return g


def sample(
    logits: torch.Tensor,
    temperatures: torch.Tensor,
    top_ps: torch.Tensor,
    top_ks: torch.Tensor,
    seeds: torch.Tensor,
    steps: torch.Tensor,
    mask: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fused sampling reference.

    - mask: optional [B, ceil(V/32)] uint32 bitmask of ALLOWED tokens
      (constrained decoding); disallowed logits become -inf.
    - temperature == 0 -> greedy.
    - top_k <= 0 means no top-k; top_p >= 1 means no top-p.
    - Returns (token_ids [B], logprobs [B]) where logprob is the
      log-softmax of the MASKED, UNtempered distribution at the chosen token
      (OpenAI-style model logprob).
    """
    B, V = logits.shape
    lf = logits.float().clone()
    if mask is not None:
        bit = torch.arange(V, device=logits.device)
        allowed = (mask[:, bit // 32] >> (bit % 32).to(torch.int64)) & 1
        lf = torch.where(allowed.bool(), lf, torch.full_like(lf, float("-inf")))
    base_logprobs = torch.log_softmax(lf, dim=-1)

    tokens = torch.empty(B, dtype=torch.long, device=logits.device)
    for b in range(B):
        temp = float(temperatures[b])
        row = lf[b]
        if temp == 0.0:
            tokens[b] = torch.argmax(row)
            continue
        scaled = row / temp
        k = int(top_ks[b])
        p = float(top_ps[b])
        probs = torch.softmax(scaled, dim=-1)
        if k > 0 and k < V:
            kth = torch.topk(probs, k).values[-1]
            probs = torch.where(probs >= kth, probs, torch.zeros_like(probs))
        if p < 1.0:
            sorted_probs, sorted_idx = torch.sort(probs, descending=True)
            cumsum = torch.cumsum(sorted_probs, dim=-1)
            # keep the smallest prefix with mass >= top_p (first token always kept)
            cut = (cumsum - sorted_probs) >= p
            sorted_probs = torch.where(cut, torch.zeros_like(sorted_probs), sorted_probs)
            probs = torch.zeros_like(probs).scatter(0, sorted_idx, sorted_probs)
        probs = probs / probs.sum()
        g = _stream_generator(int(seeds[b]), int(steps[b]), logits.device)
        u = torch.rand(V, generator=g).to(logits.device)
        # Gumbel-max over the truncated renormalized distribution
        e = (-torch.log(u.clamp_min(1e-20))).clamp_min(1e-20)  # Exp(1)
        gumbel = -torch.log(e)
        masked_logp = torch.where(probs > 0, torch.log(probs), torch.full_like(probs, float("-inf")))
        tokens[b] = torch.argmax(masked_logp + gumbel)

    logprobs = base_logprobs[torch.arange(B, device=logits.device), tokens]
    return tokens, logprobs
