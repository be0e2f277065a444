"""Key selection for record-list alignment.

Behavioral re-implementation of the reference's cascade key selection
(k_llms/utils/key_selection.py:24-445): for lists of JSON records, discover
scalar dot-paths, score each as an alignment key (coverage / uniqueness /
pairwise-Jaccard stability / support histogram, lexicographic score tuple),
funnel through a 4-stage cascade, and optionally grow greedy + brute-force
composite keys up to ``max_k``.

One structural difference from the reference: the cascade is parameterized by
an optional value canonicalizer, so the fuzzy variant
(fuzzy_key_selection.py) reuses this implementation instead of duplicating
the funnel.
"""

from __future__ import annotations

import math
import re
from collections import Counter
from itertools import combinations
from typing import Any, Callable, Dict, List, Optional, Set, Tuple

from pydantic import BaseModel, ConfigDict

JSONPath = str
Canonicalizer = Optional[Callable[[Any], Any]]

# Configurable record-list keys (reference: key_selection.py:36)
RECORD_LIST_KEYS: List[str] = ["products"]


def normalize_scalar(value: Any) -> Any:
    """Lowercase + collapse whitespace for strings (ref :24-30)."""
    if isinstance(value, str):
        return re.sub(r"\s+", " ", value.strip().lower())
    return value


def iter_records(extraction: Dict[str, Any], list_key: Optional[str] = None) -> List[Dict[str, Any]]:
    """Record dicts from a named list key, RECORD_LIST_KEYS, or the first
    list-of-dicts found (ref :38-76)."""
    records: List[Dict[str, Any]] = []
    if list_key is not None:
        seq = extraction.get(list_key)
        if isinstance(seq, list):
            records = [x for x in seq if isinstance(x, dict)]
        return records
    for candidate in RECORD_LIST_KEYS:
        seq = extraction.get(candidate)
        if isinstance(seq, list):
            records.extend(x for x in seq if isinstance(x, dict))
    if records:
        return records
    for value in extraction.values():
        if isinstance(value, list):
            records.extend(x for x in value if isinstance(x, dict))
    return records


def _resolve_path(record: Any, parts: List[str]) -> Tuple[bool, Any]:
    cur = record
    for token in parts:
        if isinstance(cur, dict) and token in cur:
            cur = cur[token]
        else:
            return False, None
    return True, cur


def values_for_path(extraction: Dict[str, Any], path: JSONPath, list_key: Optional[str] = None) -> List[Any]:
    """Scalar values for a dot path across all records (ref :78-96)."""
    parts = path.split(".")
    out: List[Any] = []
    for record in iter_records(extraction, list_key=list_key):
        ok, cur = _resolve_path(record, parts)
        if ok and cur is not None and not isinstance(cur, (dict, list)):
            out.append(normalize_scalar(cur))
    return out


def tuple_values_for_paths(
    extraction: Dict[str, Any], paths: List[JSONPath], list_key: Optional[str] = None
) -> List[Tuple[Any, ...]]:
    """Composite key tuples per record; records missing any component are
    skipped (ref :213-236)."""
    parts_list = [p.split(".") for p in paths]
    out: List[Tuple[Any, ...]] = []
    for record in iter_records(extraction, list_key=list_key):
        components: List[Any] = []
        for parts in parts_list:
            ok, cur = _resolve_path(record, parts)
            if not ok or cur is None or isinstance(cur, (dict, list)):
                components = []
                break
            components.append(normalize_scalar(cur))
        if components:
            out.append(tuple(components))
    return out


def discover_scalar_paths(extractions: List[Dict[str, Any]], list_key: Optional[str] = None) -> List[JSONPath]:
    """All dot paths resolving to scalars; list-valued paths excluded (ref :99-121)."""
    found: Set[str] = set()
    for extraction in extractions:
        for record in iter_records(extraction, list_key=list_key):
            stack: List[Tuple[str, Any]] = [("", record)]
            while stack:
                base, node = stack.pop()
                if not isinstance(node, dict):
                    continue
                for key, value in node.items():
                    path = f"{base}.{key}" if base else key
                    if isinstance(value, dict):
                        stack.append((path, value))
                    elif isinstance(value, list):
                        continue
                    else:
                        found.add(path)
    return sorted(found)


def jaccard(a: Set[Any], b: Set[Any]) -> float:
    if not a and not b:
        return 1.0
    if not a or not b:
        return 0.0
    union = len(a | b)
    return len(a & b) / union if union else 1.0


class KeyMetrics(BaseModel):
    """Per-key quality metrics + the lexicographic ranking tuple (ref :131-151)."""

    model_config = ConfigDict(frozen=True)

    path: Tuple[str, ...]
    coverage_min: float
    coverage_mean: float
    uniqueness_min: float
    uniqueness_mean: float
    jaccard_min: float
    jaccard_mean: float
    I_E: int             # values present in all extractions
    I_E_minus_1: int     # present in E-1 extractions
    I_ge_2: int          # present in >= 2 extractions
    union_size: int
    score_tuple: Tuple


def _evaluate_per_vals(
    extractions: List[Dict[str, Any]],
    per_vals: List[List[Any]],
    depth_hint: int,
    n_paths: int,
    path: Tuple[str, ...],
    list_key: Optional[str] = None,
) -> KeyMetrics:
    """Score one candidate key given its per-extraction value lists (ref :154-210)."""
    E = len(extractions)
    per_sets = [set(vs) for vs in per_vals]

    coverage: List[float] = []
    uniqueness: List[float] = []
    for vs, e in zip(per_vals, extractions):
        total = len(iter_records(e, list_key=list_key))
        non_null = len(vs)
        coverage.append(non_null / max(1, total))
        cnt = Counter(vs)
        uniq = sum(1 for _v, c in cnt.items() if c == 1)
        uniqueness.append(uniq / max(1, non_null) if non_null else 0.0)

    j_scores = [jaccard(per_sets[i], per_sets[j]) for i in range(E) for j in range(i + 1, E)]
    j_mean = sum(j_scores) / len(j_scores) if j_scores else 1.0
    j_min = min(j_scores) if j_scores else 1.0

    support: Counter = Counter()
    for s in per_sets:
        for v in s:
            support[v] += 1
    counts_by_sup = Counter(support.values())
    I_E = counts_by_sup.get(E, 0)
    I_Em1 = counts_by_sup.get(E - 1, 0) if E >= 2 else 0
    I_2p = sum(c for sup, c in counts_by_sup.items() if sup >= 2)
    U = len(set().union(*per_sets)) if per_sets else 0

    # stability-first lexicographic score, higher is better (ref :189-199)
    score_tuple = (
        round(j_min, 6),
        I_E,
        I_Em1,
        round(j_mean, 6),
        round(min(uniqueness), 6) if uniqueness else 0.0,
        round(min(coverage), 6) if coverage else 0.0,
        -U,
        depth_hint,
        -n_paths,
    )
    return KeyMetrics(
        path=path,
        coverage_min=min(coverage) if coverage else 0.0,
        coverage_mean=sum(coverage) / len(coverage) if coverage else 0.0,
        uniqueness_min=min(uniqueness) if uniqueness else 0.0,
        uniqueness_mean=sum(uniqueness) / len(uniqueness) if uniqueness else 0.0,
        jaccard_min=j_min,
        jaccard_mean=j_mean,
        I_E=I_E,
        I_E_minus_1=I_Em1,
        I_ge_2=I_2p,
        union_size=U,
        score_tuple=score_tuple,
    )


def _maybe_canon(vals: List[Any], canon: Canonicalizer) -> List[Any]:
    return [canon(v) for v in vals] if canon is not None else vals


def evaluate_single_key(
    extractions: List[Dict[str, Any]],
    path: JSONPath,
    list_key: Optional[str] = None,
    canon: Canonicalizer = None,
) -> KeyMetrics:
    per_vals = [_maybe_canon(values_for_path(e, path, list_key=list_key), canon) for e in extractions]
    return _evaluate_per_vals(extractions, per_vals, path.count("."), 1, (path,), list_key=list_key)


def evaluate_composite_key(
    extractions: List[Dict[str, Any]],
    paths: List[JSONPath],
    list_key: Optional[str] = None,
    canon: Canonicalizer = None,
) -> KeyMetrics:
    per_vals = []
    for e in extractions:
        tuples = tuple_values_for_paths(e, paths, list_key=list_key)
        if canon is not None:
            tuples = [tuple(canon(c) for c in t) for t in tuples]
        per_vals.append(tuples)
    return _evaluate_per_vals(
        extractions, per_vals, sum(p.count(".") for p in paths), len(paths), tuple(paths), list_key=list_key
    )


class CascadeConfig(BaseModel):
    model_config = ConfigDict(frozen=True)

    min_coverage: float = 0.0
    min_uniqueness: float = 0.0
    topk_stage1: int = 30
    topk_stage2: int = 12
    topk_stage3: int = 6


class CascadeReport(BaseModel):
    model_config = ConfigDict(frozen=True)

    stage0_kept: List[KeyMetrics]
    stage1_kept: List[KeyMetrics]
    stage2_kept: List[KeyMetrics]
    stage3_kept: List[KeyMetrics]
    final_best: KeyMetrics


def cascade_select_keys(
    extractions: List[Dict[str, Any]],
    candidates: List[str],
    config: CascadeConfig = None,  # type: ignore[assignment]
    list_key: Optional[str] = None,
    canon: Canonicalizer = None,
) -> CascadeReport:
    """The 4-stage funnel (ref :310-367): gate, stability sort, intra-JSON
    quality sort, union-size filter, depth tie-break."""
    if config is None:
        config = CascadeConfig()
    singles = [evaluate_single_key(extractions, p, list_key=list_key, canon=canon) for p in candidates]

    pool0 = [
        m for m in singles
        if m.I_ge_2 > 0
        and m.jaccard_min > 0.0
        and m.coverage_min >= config.min_coverage
        and m.uniqueness_min >= config.min_uniqueness
    ]
    if not pool0:
        raise ValueError("No keys pass Stage 0 (require I_ge_2>0, jaccard_min>0, and coverage).")

    pool1 = sorted(
        pool0,
        key=lambda m: (m.I_E, m.I_E_minus_1, round(m.jaccard_min, 6), round(m.jaccard_mean, 6)),
        reverse=True,
    )[: config.topk_stage1]
    pool2 = sorted(
        pool1,
        key=lambda m: (round(m.uniqueness_min, 6), round(m.coverage_min, 6)),
        reverse=True,
    )[: config.topk_stage2]
    pool3 = sorted(pool2, key=lambda m: (m.union_size,))[: config.topk_stage3]
    final_sorted = sorted(
        pool3,
        key=lambda m: (sum(p.count(".") for p in m.path), -len(m.path)),
        reverse=True,
    )
    return CascadeReport(
        stage0_kept=pool0, stage1_kept=pool1, stage2_kept=pool2, stage3_kept=pool3,
        final_best=final_sorted[0],
    )


class KeySelectionResult(BaseModel):
    model_config = ConfigDict(frozen=True)

    best_single: KeyMetrics
    best_composite: Optional[KeyMetrics]
    candidate_table: List[KeyMetrics]
    min_support_for_autolock: int
    cascade_report: CascadeReport


def stability_tuple(m: KeyMetrics) -> Tuple:
    return (round(m.jaccard_min, 6), m.I_E, m.I_E_minus_1, round(m.jaccard_mean, 6))


def select_best_keys(
    extractions: List[Dict[str, Any]],
    max_candidates_for_composite: int = 20,
    max_k: int = 3,
    min_support_ratio_for_autolock: float = 0.75,
    cascade_cfg: CascadeConfig = None,  # type: ignore[assignment]
    list_key: Optional[str] = None,
) -> KeySelectionResult:
    """Full selection: cascade on singles, then greedy + brute-force composite
    growth accepted only when stability improves (ref :396-445)."""
    if cascade_cfg is None:
        cascade_cfg = CascadeConfig()
    if not extractions:
        raise ValueError("No extractions provided.")

    E = len(extractions)
    autolock_t = max(2, math.ceil(min_support_ratio_for_autolock * E))

    candidates = discover_scalar_paths(extractions, list_key=list_key)
    if not candidates:
        raise ValueError("No scalar candidate paths discovered.")

    report = cascade_select_keys(extractions, candidates, cascade_cfg, list_key=list_key)
    best_single = report.final_best

    singles_all = [evaluate_single_key(extractions, p, list_key=list_key) for p in candidates]
    singles_all = [m for m in singles_all if m.I_ge_2 > 0 and m.jaccard_min > 0.0]
    singles_all.sort(
        key=lambda m: (
            round(m.jaccard_min, 6), m.I_E, m.I_E_minus_1, round(m.jaccard_mean, 6),
            round(m.uniqueness_min, 6), round(m.coverage_min, 6), -m.union_size,
        ),
        reverse=True,
    )

    topN = [m.path[0] for m in report.stage3_kept][:max_candidates_for_composite]
    best_combo: Optional[KeyMetrics] = None
    if topN:
        current = [topN[0]]
        best_combo = evaluate_composite_key(extractions, current, list_key=list_key)
        improved = True
        while improved and len(current) < max_k:
            improved = False
            for cand in (p for p in topN if p not in current):
                trial = evaluate_composite_key(extractions, current + [cand], list_key=list_key)
                if trial.score_tuple > best_combo.score_tuple and stability_tuple(trial) > stability_tuple(best_combo):
                    best_combo = trial
                    current.append(cand)
                    improved = True
        # brute-force sweep over 2..max_k subsets of the stage-3 pool
        for r in range(2, min(max_k, len(topN)) + 1):
            for combo in combinations(topN, r):
                trial = evaluate_composite_key(extractions, list(combo), list_key=list_key)
                if stability_tuple(trial) > stability_tuple(best_combo) or trial.score_tuple > best_combo.score_tuple:
                    best_combo = trial

    return KeySelectionResult(
        best_single=best_single,
        best_composite=best_combo,
        candidate_table=singles_all,
        min_support_for_autolock=autolock_t,
        cascade_report=report,
    )
