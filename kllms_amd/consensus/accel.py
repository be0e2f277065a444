"""On-device consensus acceleration (M5).

The alignment hot loop evaluates O(L^2 * m^2) pairwise
similarities; with the "embeddings" method each NEW long-string pair costs an
embedding computation in the reference — one remote API roundtrip per string
(consensus_utils.py:652-657). Locally the same contract is served in ONE
batched pass: walk the candidate contents, collect every embeddable string
(the > 50-char gate from string_similarity, consensus_utils.py:813), embed
them all in a single engine call (one gather + mean-pool + normalize on the
GPU), compute the full cosine matrix with one GEMM, and pre-populate the
similarity TTL cache that ``string_similarity`` consults first — so the
entire alignment/consensus pass then runs without a single per-pair
embedding call.
"""

from __future__ import annotations

from typing import Any, Iterable, List

import numpy as np

from .settings import SIMILARITY_SCORE_LOWER_BOUND
from .similarity import _set_cached_similarity

EMBED_MIN_CHARS = 51  # strings strictly longer than 50 chars use embeddings


def collect_embeddable_strings(objs: Iterable[Any]) -> List[str]:
    """All unique strings (> 50 chars) anywhere in the nested contents."""
    seen: set = set()
    out: List[str] = []

    def walk(v: Any) -> None:
        if isinstance(v, str):
            if len(v) >= EMBED_MIN_CHARS and v not in seen:
                seen.add(v)
                out.append(v)
        elif isinstance(v, dict):
            for x in v.values():
                walk(x)
        elif isinstance(v, (list, tuple)):
            for x in v:
                walk(x)

    for o in objs:
        walk(o)
    return out


def precompute_similarity_cache(contents: Iterable[Any], embed_fn, client=None) -> int:
    """Batch-embed every embeddable string and cache all pairwise cosine
    similarities (0.5*(cos+1) rescale, matching similarity.cosine_similarity).
    With a local engine client the embeddings STAY on the GPU and the cosine
    matrix is one device GEMM (engine.embed_dev); otherwise the injected
    embed_fn runs and the GEMM is numpy. Returns the number of cached pairs;
    never raises (acceleration is best-effort — on any failure the per-pair
    fallback path still works)."""
    try:
        strings = collect_embeddable_strings(contents)
        if len(strings) < 2:
            return 0
        eng = getattr(client, "engine", None) if client is not None else None
        sim = None
        if eng is not None and hasattr(eng, "embed_dev"):
            try:
                import torch

                with getattr(client, "_engine_lock", _NullCtx()):
                    unit_t, _ = eng.embed_dev(strings)     # [N, H] on device, unit rows
                cos_t = unit_t @ unit_t.T                  # one device GEMM
                sim_t = (0.5 * (cos_t + 1.0)).clamp(SIMILARITY_SCORE_LOWER_BOUND, 1.0)
                zero = (unit_t.abs().sum(dim=1) == 0)
                sim_t[zero, :] = SIMILARITY_SCORE_LOWER_BOUND
                sim_t[:, zero] = SIMILARITY_SCORE_LOWER_BOUND
                sim = sim_t.cpu().numpy()                  # ONE transfer for the matrix
            except Exception:
                sim = None
        if sim is None:
            vecs = np.asarray(embed_fn(strings), dtype=np.float64)
            norms = np.linalg.norm(vecs, axis=1, keepdims=True)
            safe = np.where(norms == 0, 1.0, norms)
            unit = vecs / safe
            cos = unit @ unit.T
            sim = 0.5 * (cos + 1.0)
            sim = np.clip(sim, SIMILARITY_SCORE_LOWER_BOUND, 1.0)
            zero_rows = (norms[:, 0] == 0)
            sim[zero_rows, :] = SIMILARITY_SCORE_LOWER_BOUND
            sim[:, zero_rows] = SIMILARITY_SCORE_LOWER_BOUND

        n_cached = 0
        for i in range(len(strings)):
            for j in range(i + 1, len(strings)):
                _set_cached_similarity(strings[i], strings[j], "embeddings", float(sim[i, j]))
                n_cached += 1
        return n_cached
    except Exception:
        return 0


class _NullCtx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


def collect_short_strings(objs: Iterable[Any]) -> List[str]:
    """Unique strings that the reference scores with normalized Levenshtein
    under the 'embeddings' method (either side <= 50 chars, ref :813-820)."""
    seen: set = set()
    out: List[str] = []

    def walk(v: Any) -> None:
        if isinstance(v, str):
            if len(v) < EMBED_MIN_CHARS and v not in seen:
                seen.add(v)
                out.append(v)
        elif isinstance(v, dict):
            for x in v.values():
                walk(x)
        elif isinstance(v, (list, tuple)):
            for x in v:
                walk(x)

    for o in objs:
        walk(o)
    return out


def precompute_levenshtein_cache(contents: Iterable[Any], client=None,
                                 method: str = "embeddings") -> int:
    """Batch every short-string pair's normalized Levenshtein similarity on
    the GPU (ops/hip/levenshtein.hip: Myers bit-parallel, one thread per
    pair) and pre-populate the TTL similarity cache. No-op without a local
    GPU engine; never raises."""
    try:
        import torch

        from .. import ops
        from .similarity import normalize_string

        eng = getattr(client, "engine", None) if client is not None else None
        if eng is None or eng.device.type != "cuda" or not ops.hip_available():
            return 0
        strings = collect_short_strings(contents)
        if len(strings) < 2:
            return 0
        norm = [normalize_string(s) for s in strings]
        if any(len(n) > 64 for n in norm):
            # normalized forms longer than the 64-bit Myers vector: leave
            # those to the CPU path (cannot happen under the 50-char gate)
            keep = [i for i, n in enumerate(norm) if len(n) <= 64]
            strings = [strings[i] for i in keep]
            norm = [norm[i] for i in keep]
            if len(strings) < 2:
                return 0
        N = len(strings)
        dev = eng.device
        chars = torch.zeros(N, 64, dtype=torch.uint8)
        lens = torch.zeros(N, dtype=torch.int32)
        for i, nstr in enumerate(norm):
            b = nstr.encode("ascii", errors="replace")
            chars[i, : len(b)] = torch.tensor(list(b), dtype=torch.uint8)
            lens[i] = len(b)
        ii, jj = torch.triu_indices(N, N, offset=1)
        dist = ops.levenshtein_pairs(chars.to(dev), lens.to(dev),
                                     ii.to(torch.int32).to(dev), jj.to(torch.int32).to(dev))
        dist = dist.cpu().numpy()
        lens_np = lens.numpy()
        ii_np, jj_np = ii.numpy(), jj.numpy()
        maxlen = np.maximum(lens_np[ii_np], lens_np[jj_np])
        sim = np.where(maxlen == 0, 1.0,
                       np.maximum(SIMILARITY_SCORE_LOWER_BOUND,
                                  1.0 - dist / np.maximum(maxlen, 1)))
        for k in range(len(ii_np)):
            _set_cached_similarity(strings[ii_np[k]], strings[jj_np[k]], method, float(sim[k]))
        return len(ii_np)
    except Exception:
        return 0
