"""Key-based alignment engine — the alternative aligner (L1b).

Behavioral re-implementation of k_llms/utils/key_based_alignment.py:47-516:
``recursive_align`` has the same signature/return shape as the
similarity-based ``recursive_list_alignments``, but aligns lists-of-records
by SELECTED KEY TUPLES (key_selection + fuzzy fallback cascade) instead of
pairwise similarity; scalar lists fall back to positional zip. Per-source
views are re-materialized through the path mappings.

Activation: the reference swaps an import at consolidation.py:22; here the
aligner is selectable via the ``aligner`` argument of the consolidation entry
points ("similarity" | "key").
"""

from __future__ import annotations

import logging
from copy import deepcopy
from typing import Any, Dict, List, Optional, Sequence, Tuple

from .fuzzy_key_selection import select_best_keys_with_fuzzy_fallback
from .key_selection import CascadeConfig, select_best_keys

logger = logging.getLogger("kllms_amd.consensus.key_align")

# CLI-style verbosity knobs kept for parity (ref :24-43); routed to logging.
VERBOSE: bool = False
LOG_FILE: Optional[str] = None


def _log(msg: str) -> None:
    if VERBOSE:
        print(msg)
    if LOG_FILE:
        try:
            with open(LOG_FILE, "a", encoding="utf-8") as lf:
                lf.write(msg + "\n")
        except Exception:
            pass
    logger.debug(msg)


def _get_key_tuple(obj: Dict[str, Any], paths: Tuple[str, ...]) -> Optional[Tuple[Any, ...]]:
    """Resolve a (composite) key tuple from a record; None when any path is
    missing or non-scalar (ref :50-72)."""
    values = []
    for path in paths:
        cur: Any = obj
        for part in path.split("."):
            if isinstance(cur, dict) and part in cur:
                cur = cur[part]
            else:
                return None
        if cur is None or isinstance(cur, (dict, list)):
            return None
        values.append(cur)
    return tuple(values)


def _align_lists_by_key(
    lists_to_align: Sequence[Optional[List[Dict[str, Any]]]],
    key_paths: Tuple[str, ...],
) -> Tuple[List[List[Optional[Dict[str, Any]]]], List[List[Optional[int]]]]:
    """Row-align records across sources by key tuple; row order follows the
    longest source, then leftover keys sorted (ref :75-151)."""
    if not any(lists_to_align):
        return [], []

    all_key_tuples: set = set()
    indexes: List[Dict[Tuple[Any, ...], int]] = []
    for source_list in lists_to_align:
        mapping: Dict[Tuple[Any, ...], int] = {}
        if isinstance(source_list, list):
            for i, item in enumerate(source_list):
                if isinstance(item, dict):
                    kt = _get_key_tuple(item, key_paths)
                    if kt is not None and kt not in mapping:
                        mapping[kt] = i
                        all_key_tuples.add(kt)
        indexes.append(mapping)

    def _safe_len(sl) -> int:
        return len(sl) if isinstance(sl, list) else 0

    best_source_idx = max(range(len(lists_to_align)), key=lambda i: _safe_len(lists_to_align[i]))
    ordered_keys: List[Tuple[Any, ...]] = []
    seen: set = set()
    best_list = lists_to_align[best_source_idx]
    if isinstance(best_list, list):
        for item in best_list:
            if isinstance(item, dict):
                kt = _get_key_tuple(item, key_paths)
                if kt is not None and kt not in seen:
                    ordered_keys.append(kt)
                    seen.add(kt)
    ordered_keys.extend(sorted(all_key_tuples - seen))

    aligned_rows: List[List[Optional[Dict[str, Any]]]] = []
    original_indices: List[List[Optional[int]]] = []
    for kt in ordered_keys:
        row: List[Optional[Dict[str, Any]]] = []
        idx_row: List[Optional[int]] = []
        for source_idx, source_list in enumerate(lists_to_align):
            oi = indexes[source_idx].get(kt)
            if oi is not None and isinstance(source_list, list):
                row.append(source_list[oi])
                idx_row.append(oi)
            else:
                row.append(None)
                idx_row.append(None)
        aligned_rows.append(row)
        original_indices.append(idx_row)
    return aligned_rows, original_indices


def _choose_key_paths(lists: List[list], cascade_cfg: CascadeConfig) -> Optional[Tuple[str, ...]]:
    """Key-selection failure cascade: standard (with composite) -> fuzzy ->
    None (ref :219-301)."""
    dummy_extractions = [{"items": lst} for lst in lists]
    try:
        result = select_best_keys(dummy_extractions, list_key="items", cascade_cfg=cascade_cfg)
        use_composite = (
            result.best_composite is not None
            and result.best_composite.score_tuple > result.best_single.score_tuple
        )
        std_paths = result.best_composite.path if use_composite else result.best_single.path
        std_metrics = result.best_composite if use_composite else result.best_single
        _log(f"[KEY-SELECT standard] path={list(std_paths)} jaccard_min={round(std_metrics.jaccard_min, 6)}")
        try:
            comp = select_best_keys_with_fuzzy_fallback(
                dummy_extractions, cascade_cfg=cascade_cfg, list_key="items",
                fuzzy_numeric_round_decimals=2, enable_fuzzy_fallback=True, prefer_fuzzy_if_better=True,
            )
            if comp.chosen == "fuzzy" and comp.fuzzy_best is not None:
                _log(f"[KEY-SELECT fuzzy] chosen=fuzzy path={list(comp.fuzzy_best.path)}")
                return comp.fuzzy_best.path
            return std_paths
        except Exception:
            return std_paths
    except ValueError:
        try:
            comp = select_best_keys_with_fuzzy_fallback(
                dummy_extractions, cascade_cfg=cascade_cfg, list_key="items",
                fuzzy_numeric_round_decimals=2, enable_fuzzy_fallback=True, prefer_fuzzy_if_better=True,
            )
            chosen = comp.fuzzy_best if comp.chosen == "fuzzy" else comp.normal_best
            if chosen is not None:
                _log(f"[KEY-SELECT fallback] chosen={comp.chosen} path={list(chosen.path)}")
                return chosen.path
        except Exception:
            pass
        _log("[KEY-SELECT] no key found (standard failed, fuzzy failed)")
        return None


def _compute_key_aligned_structure(
    values: Sequence[Any],
    original_paths: Sequence[Optional[str]],
    cascade_cfg: CascadeConfig,
) -> Tuple[Any, Dict[str, List[Optional[str]]]]:
    """Recursive core: one ALIGNED representative structure + per-leaf path
    mappings back to every source (ref :170-345)."""
    if not values or all(v is None for v in values):
        return None, {}
    non_nulls = [v for v in values if v is not None]
    if not non_nulls:
        return None, {}

    first_type = type(non_nulls[0])
    same_type = all(isinstance(v, first_type) for v in non_nulls)
    key_mappings: Dict[str, List[Optional[str]]] = {}

    # scalars / mixed types: first non-null is the representative
    if not same_type or first_type not in (dict, list):
        return deepcopy(non_nulls[0]), {"": list(original_paths)}

    if first_type is dict:
        dicts = [v if isinstance(v, dict) else {} for v in values]
        aligned_dict: Dict[str, Any] = {}
        for key in sorted({k for d in dicts for k in d.keys()}):
            sub_vals = [d.get(key) for d in dicts]
            sub_paths = [(f"{p}.{key}" if p else key) if p is not None else None for p in original_paths]
            aligned_value, sub_mapping = _compute_key_aligned_structure(sub_vals, sub_paths, cascade_cfg)
            aligned_dict[key] = aligned_value
            for sub_key, paths in sub_mapping.items():
                key_mappings[f"{key}.{sub_key}" if sub_key else key] = paths
        return aligned_dict, key_mappings

    # lists
    lists = [v if isinstance(v, list) else [] for v in values]
    is_list_of_dicts = all(all(isinstance(item, dict) for item in lst) for lst in lists if lst)

    if is_list_of_dicts:
        key_paths = _choose_key_paths(lists, cascade_cfg)
        if key_paths:
            aligned_rows, original_indices = _align_lists_by_key(lists, key_paths)
            aligned_list = []
            for i, row in enumerate(aligned_rows):
                row_paths = [
                    (f"{p}.{original_indices[i][j]}" if p else str(original_indices[i][j]))
                    if (p is not None and original_indices[i][j] is not None)
                    else None
                    for j, p in enumerate(original_paths)
                ]
                aligned_item, sub_mapping = _compute_key_aligned_structure(row, row_paths, cascade_cfg)
                aligned_list.append(aligned_item)
                for sub_key, paths in sub_mapping.items():
                    key_mappings[f"{i}.{sub_key}" if sub_key else str(i)] = paths
            return aligned_list, key_mappings

    # zip fallback for scalar lists / failed key selection (ref :325-345)
    _log("[ALIGN] Fallback zip alignment for lists (scalars or no key)")
    aligned_list = []
    max_len = max((len(lst) for lst in lists), default=0)
    for i in range(max_len):
        row = [lst[i] if i < len(lst) else None for lst in lists]
        row_paths = [
            ((f"{p}.{i}" if p else str(i)) if (isinstance(values[j], list) and i < len(values[j])) else None)
            if p is not None else None
            for j, p in enumerate(original_paths)
        ]
        aligned_item, sub_mapping = _compute_key_aligned_structure(row, row_paths, cascade_cfg)
        aligned_list.append(aligned_item)
        for sub_key, paths in sub_mapping.items():
            key_mappings[f"{i}.{sub_key}" if sub_key else str(i)] = paths
    return aligned_list, key_mappings


def _get_value_by_path(obj: Any, path: Optional[str]) -> Any:
    """Dot-path lookup with integer list indices (ref :355-388)."""
    if path is None:
        return None
    if path == "":
        return obj
    cur = obj
    for token in path.split("."):
        if token == "":
            continue
        # int-parse attempt, exactly as the reference: tokens int() accepts
        # (including unicode decimal digits) are list indices; tokens it
        # rejects (e.g. superscript '¹', which str.isdigit() wrongly admits)
        # fall through to dict lookup
        try:
            idx = int(token)
        except ValueError:
            idx = None
        if idx is not None:
            if isinstance(cur, list) and 0 <= idx < len(cur):
                cur = cur[idx]
                continue
            return None
        if isinstance(cur, dict) and token in cur:
            cur = cur[token]
        else:
            return None
    return cur


def _materialize_source_view(
    aligned_node: Any,
    key_mappings: Dict[str, List[Optional[str]]],
    source_idx: int,
    current_path: str = "",
    source_root: Optional[Dict[str, Any]] = None,
) -> Any:
    """Project the aligned structure back onto one source via the path
    mappings (ref :391-440)."""
    if source_root is None:
        raise ValueError("source_root must be provided at the top-level call.")
    if isinstance(aligned_node, dict):
        return {
            k: _materialize_source_view(
                v, key_mappings, source_idx, f"{current_path}.{k}" if current_path else k, source_root
            )
            for k, v in aligned_node.items()
        }
    if isinstance(aligned_node, list):
        return [
            _materialize_source_view(
                v, key_mappings, source_idx, f"{current_path}.{i}" if current_path else str(i), source_root
            )
            for i, v in enumerate(aligned_node)
        ]
    mapped = key_mappings.get(current_path)
    if mapped is not None and 0 <= source_idx < len(mapped):
        return _get_value_by_path(source_root, mapped[source_idx])
    return deepcopy(aligned_node)


def recursive_align(
    values: Sequence[Any],
    string_similarity_method: str,
    min_support_ratio: float = 0.5,
    max_novelty_ratio: float = 0.25,
    current_path: str = "",
    reference_idx: Optional[int] = None,
    min_uniqueness: Optional[float] = None,
    min_coverage: Optional[float] = None,
) -> Tuple[Sequence[Any], Dict[str, List[Optional[str]]]]:
    """Key-based recursive alignment (ref key_based_alignment.py:350-474).

    Signature matches the reference's EXACTLY — note it differs from the
    similarity aligner's (no embeddings fn / client params: key matching is
    exact, ``string_similarity_method`` is accepted and unused)."""
    if not values:
        return list(values), {}
    if all(v is None for v in values):
        return list(values), {current_path: [current_path for _ in values]}
    non_nulls = [v for v in values if v is not None]
    if not non_nulls:
        return list(values), {}

    eff_min_coverage = min_coverage if min_coverage is not None else min_support_ratio
    eff_min_uniqueness = min_uniqueness if min_uniqueness is not None else 0.5
    cascade_cfg = CascadeConfig(min_coverage=eff_min_coverage, min_uniqueness=eff_min_uniqueness)

    original_paths: List[Optional[str]] = [current_path for _ in values]
    aligned_data, raw_key_mappings = _compute_key_aligned_structure(values, original_paths, cascade_cfg)

    per_source_outputs: List[Any] = []
    for i, src_root in enumerate(values):
        if isinstance(src_root, dict):
            materialized_root: Dict[str, Any] = src_root
        elif isinstance(src_root, list):
            materialized_root = {"items": src_root}
            if raw_key_mappings:
                raw_key_mappings = {
                    (f"items.{k}" if k else "items"): v for k, v in raw_key_mappings.items()
                }
        else:
            materialized_root = {}
        per_source_outputs.append(
            _materialize_source_view(
                aligned_node=aligned_data,
                key_mappings=raw_key_mappings,
                source_idx=i,
                current_path="",
                source_root=materialized_root,
            )
        )

    if current_path:
        prefixed: Dict[str, List[Optional[str]]] = {}
        for key, paths in raw_key_mappings.items():
            pref_key = f"{current_path}.{key}" if key else current_path
            pref_paths: List[Optional[str]] = []
            for p in paths:
                if p is None or p == "":
                    pref_paths.append(current_path if current_path else None)
                else:
                    pref_paths.append(f"{current_path}.{p}" if current_path else p)
            prefixed[pref_key] = pref_paths
        return per_source_outputs, prefixed
    return per_source_outputs, raw_key_mappings
