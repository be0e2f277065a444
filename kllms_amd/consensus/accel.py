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


def precompute_similarity_cache(contents: Iterable[Any], embed_fn) -> int:
    """Batch-embed every embeddable string and cache all pairwise cosine
    similarities (0.5*(cos+1) rescale, matching similarity.cosine_similarity).
    Returns the number of cached pairs; never raises (acceleration is
    best-effort — on any failure the per-pair fallback path still works)."""
    try:
        strings = collect_embeddable_strings(contents)
        if len(strings) < 2:
            return 0
        vecs = np.asarray(embed_fn(strings), dtype=np.float64)
        norms = np.linalg.norm(vecs, axis=1, keepdims=True)
        safe = np.where(norms == 0, 1.0, norms)
        unit = vecs / safe
        cos = unit @ unit.T  # one GEMM for the full matrix
        sim = 0.5 * (cos + 1.0)
        sim = np.clip(sim, SIMILARITY_SCORE_LOWER_BOUND, 1.0)
        zero_rows = (norms[:, 0] == 0)

        n_cached = 0
        for i in range(len(strings)):
            for j in range(i + 1, len(strings)):
                value = SIMILARITY_SCORE_LOWER_BOUND if (zero_rows[i] or zero_rows[j]) else float(sim[i, j])
                _set_cached_similarity(strings[i], strings[j], "embeddings", value)
                n_cached += 1
        return n_cached
    except Exception:
        return 0
