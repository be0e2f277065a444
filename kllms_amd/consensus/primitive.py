"""Primitive-value consensus: LLM merge, hybrid numeric clustering, medoid.

Behavioral re-implementation of consensus_as_primitive
(ref consensus_utils.py:1075-1237) and the LLM string-consensus path
(ref :989-1073). Three branches:

(a) llm-consensus (only when string_consensus_method == "llm-consensus" AND
    string_similarity_method == "embeddings"): merge candidate strings with a
    chat model using the fixed system prompt below; confidence = mean
    similarity of the result to the candidates (unrounded, not scaled by
    parent_valid_frac — ref :1090-1096).
(b) hybrid numeric: sort -> 1D closeness clustering (rel_eps/abs_eps) ->
    majority / unique-max / tie-break by cross-cluster support where clusters
    also match via abs/rel, signless, and power-of-10 equivalence; None can
    win if it is the plurality; winner = cluster mean (ref :1098-1219).
(c) fallback similarity medoid: full pairwise similarity matrix, NaN
    diagonal, argmax of row means (ref :1221-1237).
"""

from __future__ import annotations

import json
import math
from typing import Any

import numpy as np

from .settings import (
    SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    ConsensusSettings,
    logger,
)
from .similarity import generic_similarity

# Fixed system prompt for the LLM string-consensus mode (ref :989-1024).
SYSTEM_PROMPT_STRING_CONSENSUS_LLM = """
You are a helpful assistant that builds a consensus string from a list of strings.
## Context
- We are doing a voting-like document extraction task, this is just a small part of the task.
- We generate multiple response candidates (strings) for a given field, and we need to define the consensus string.

## Instructions
- You will be given a list of strings.
- You need to build a consensus string from the list of strings.
- The consensus string should be a string that is most similar to the majority of the strings in the list.
- On general, the consensus string is meant to capture the "general idea/information" of the list, not the exact wording.
- If the list is too diverse and you cannot elect a consensus string, return "Uncertain" -- But avoid this answer whenever possible.
- If the list is empty, return "Unknown".

## Output
- The output should be a raw string, not a JSON. Not enclosed in quotes.

## Examples
### Example 1
- Input: ["The sky is blue", "The sky is blue", "The sky is blue"]
- Output: The sky is blue

### Example 2
- Input: ["The sky is blue", "The sky is green", "The sky is red"]
- Output: Uncertain

### Example 3
- Input: []
- Output: Unknown

### Example 4
- Input: ["The sky is blue tonight", "The sky is blue today", "The sky is blue"]
- Output: The sky is blue

I think you got the point.
"""


def _consensus_llm_model(client: Any) -> str:
    """The reference hard-codes "gpt-5-mini" (ref :1038); locally the merge
    runs through the local engine, using its configured consensus model."""
    return getattr(client, "consensus_model", None) or "local"


def string_consensus_llm(values: list[str], client: Any) -> str:
    """Merge candidate strings with a chat-completions client (ref :1026-1048).

    ``client`` is anything exposing ``.chat.completions.create(model=...,
    messages=[...])`` and returning an object with ``.choices[0].message.content``
    — the local ``KLLMs`` client satisfies this.
    """
    assert len(values) > 0, "Cannot build consensus string from empty list"
    values_json_dumped = [json.dumps(v) for v in values]
    response = client.chat.completions.create(
        model=_consensus_llm_model(client),
        messages=[
            {"role": "system", "content": SYSTEM_PROMPT_STRING_CONSENSUS_LLM},
            {"role": "user", "content": f"Input: {values_json_dumped}\nOutput:"},
        ],
    )
    content = response.choices[0].message.content
    if content is None:
        logger.warning("No content returned from LLM string consensus")
        return values[0]
    return str(content).strip()


# --- hybrid numeric helpers (ref :1127-1164) ---------------------------------

def _is_close_absrel(a: float, b: float, rel_eps: float, abs_eps: float) -> bool:
    denom = max(abs(a), abs(b), 1.0)
    return abs(a - b) <= max(abs_eps, rel_eps * denom)


def _is_close_signless(a: float, b: float, rel_eps: float, abs_eps: float) -> bool:
    return _is_close_absrel(abs(a), abs(b), rel_eps, abs_eps)


def _is_close_power10(a: float, b: float, rel_eps: float, abs_eps: float, k_range: tuple[int, int] = (-6, 6)) -> bool:
    if a == 0.0 or b == 0.0:
        return _is_close_absrel(a, b, rel_eps, abs_eps)
    for k in range(k_range[0], k_range[1] + 1):
        if _is_close_absrel(a, b * (10.0 ** k), rel_eps, abs_eps):
            return True
    return False


def _cluster_1d(xs_sorted: list[float], rel_eps: float, abs_eps: float) -> list[list[float]]:
    """Greedy 1D clustering of sorted values by adjacent closeness (ref :1127-1144)."""
    if not xs_sorted:
        return []
    clusters: list[list[float]] = []
    current = [xs_sorted[0]]
    for i in range(len(xs_sorted) - 1):
        a, b = xs_sorted[i], xs_sorted[i + 1]
        denom = max(abs(a), abs(b), 1.0)
        if abs(b - a) <= max(abs_eps, rel_eps * denom):
            current.append(b)
        else:
            clusters.append(current)
            current = [b]
    clusters.append(current)
    return clusters


def _numeric_consensus(
    values: list[Any],
    settings: ConsensusSettings,
    parent_valid_frac: float,
) -> tuple[Any, float]:
    """Branch (b): the hybrid vote-or-mean numeric consensus (ref :1098-1219)."""
    total = len(values)
    none_count = sum(1 for v in values if v is None)
    frac_none = none_count / total if total else 0.0

    xs: list[float] = []
    for v in values:
        if isinstance(v, bool):
            continue
        if isinstance(v, (int, float)):
            try:
                vf = float(v)
                if math.isfinite(vf):
                    xs.append(vf)
            except Exception:
                pass
    if not xs:
        return (None, parent_valid_frac)

    xs.sort()
    clusters = _cluster_1d(xs, settings.rel_eps, settings.abs_eps)
    sizes_num = [len(c) for c in clusters]
    max_size_num = max(sizes_num, default=0)
    sizes_all = sizes_num + ([none_count] if none_count > 0 else [])
    max_size_all = max(sizes_all) if sizes_all else 0

    if none_count > max_size_num:
        return (None, round(frac_none, 5))

    if max_size_all > total / 2 or sizes_all.count(max_size_all) == 1:
        # clear majority, or a unique largest bucket
        if none_count > 0 and none_count == max_size_all:
            return (None, round(none_count / total, 5))
        max_idx = int(np.argmax(sizes_num))
        rep = float(np.mean(clusters[max_idx]))
        return (rep, round(max_size_all / total, 5))

    # Tie between equally-sized buckets: break by cross-cluster support.
    candidate_indices = [i for i, c in enumerate(clusters) if len(c) == max_size_all]
    include_none_candidate = none_count > 0 and none_count == max_size_all
    centers = [float(np.median(c)) if c else float("nan") for c in clusters]
    spreads = [float(np.std(c)) if len(c) > 1 else 0.0 for c in clusters]
    supports: list[tuple[str, int, int]] = []
    for ci in candidate_indices:
        support = len(clusters[ci])
        c_center = centers[ci]
        for oi, other in enumerate(clusters):
            if oi == ci or len(other) >= len(clusters[ci]):
                continue
            o_center = centers[oi]
            if (
                _is_close_absrel(c_center, o_center, settings.rel_eps, settings.abs_eps)
                or _is_close_signless(c_center, o_center, settings.rel_eps, settings.abs_eps)
                or _is_close_power10(c_center, o_center, settings.rel_eps, settings.abs_eps)
            ):
                support += len(other)
        supports.append(("numeric", ci, support))
    if include_none_candidate:
        supports.append(("none", -1, none_count))
    supports.sort(
        key=lambda t: (
            -t[2],
            1 if t[0] != "numeric" else 0,
            spreads[t[1]] if t[1] >= 0 else float("inf"),
            -abs(centers[t[1]]) if t[1] >= 0 else 0.0,
        )
    )
    best_kind, best_idx, best_support = supports[0]
    if best_kind == "none":
        return (None, round(best_support / total, 5))
    rep = float(np.mean(clusters[best_idx]))
    return (rep, round(best_support / total, 5))


def consensus_as_primitive(
    values: list[Any],
    consensus_settings: ConsensusSettings,
    sync_get_openai_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    parent_valid_frac: float = 1.0,
) -> tuple[Any, float]:
    """Elect a consensus primitive (ref consensus_utils.py:1075-1237)."""
    non_none_values = [v for v in values if v is not None]
    if len(non_none_values) == 0:
        return (None, parent_valid_frac)
    if len(non_none_values) == 1:
        return (non_none_values[0], parent_valid_frac * (len(non_none_values) / len(values)))

    first_val_type = type(non_none_values[0])

    # (a) LLM string merge
    if (
        first_val_type is str
        and consensus_settings.string_consensus_method == "llm-consensus"
        and consensus_settings.string_similarity_method == "embeddings"
    ):
        consensus_string = string_consensus_llm(non_none_values, client=client)
        similarities = [
            generic_similarity(
                consensus_string, v, consensus_settings.string_similarity_method, sync_get_openai_embeddings_from_text
            )
            for v in non_none_values
        ]
        return consensus_string, float(np.nanmean(similarities))

    # (b) hybrid numeric (bool is an int subclass, so bool-typed values route
    # here and, with every bool skipped from xs, resolve to (None, pvf) —
    # observable reference behavior, ref :1102,1110-1117)
    if first_val_type in (int, float, bool) or all(isinstance(v, (int, float)) for v in non_none_values):
        return _numeric_consensus(values, consensus_settings, parent_valid_frac)

    # (c) similarity medoid
    n = len(values)
    if n == 0:
        return (None, 0.0)
    if n == 1:
        return (values[0], parent_valid_frac)
    sim_matrix = np.zeros((n, n), dtype=float)
    for i in range(n):
        for j in range(i + 1, n):
            sim = generic_similarity(
                values[i], values[j], consensus_settings.string_similarity_method, sync_get_openai_embeddings_from_text
            )
            sim_matrix[i, j] = sim_matrix[j, i] = sim
        sim_matrix[i, i] = np.nan
    avg_sims = np.nanmean(sim_matrix, axis=1)
    best_idx = int(np.argmax(avg_sims))
    confidence = parent_valid_frac * float(avg_sims[best_idx])
    return (values[best_idx], round(confidence, 5))
