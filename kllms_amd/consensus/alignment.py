"""List alignment: dynamic threshold, reference building, Hungarian matching,
support pruning and the recursive structural walk.

Behavioral re-implementation of the alignment core
(ref consensus_utils.py:81-613). The pairwise-similarity evaluations are the
hot loop — they flow through SimilarityCache so the batched on-device
embedding path (ops/consensus) can back them.

Key reference quirks kept on purpose (observable behavior):
- dynamic threshold = max(0.5, 0.95 * min(inlier best-match sims)), best
  matches searched only in lists j > i with per-list "used" sets
  (ref :185-252);
- reference built by greedy support grouping with representative re-election
  via consensus_as_primitive over *(list_idx, pos)* INDEX TUPLES with a dummy
  embedding fn — the tuples route to the similarity-medoid branch
  (ref :282-321);
- Hungarian alignment accepts pairs with sim >= 0.95 * dynamic threshold
  (ref :336-379,410);
- columns pruned below min_support_ratio; if every column is below it, the
  threshold drops to the max support (ref :109-149);
- final column order by Condorcet pairwise majority (majority_order).
"""

from __future__ import annotations

from collections import defaultdict
from copy import deepcopy
from typing import Any, Callable, Optional

import numpy as np
from scipy.optimize import linear_sum_assignment

from .majority_order import _original_positions, sort_by_original_majority
from .primitive import consensus_as_primitive
from .settings import (
    SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    ConsensusSettings,
    StringSimilarityMethod,
    logger,
)
from .similarity import generic_similarity

Index = tuple[int, int]  # (list_idx, element_idx)


class SimilarityCache:
    """Memoizes pairwise similarity keyed by symmetric index pairs (ref :81-106)."""

    def __init__(self, sim_fn: Callable[[Any, Any], float], list_of_lists: list[list[Any]]):
        self.sim_fn = sim_fn
        self.cache: dict[tuple[Index, Index], float] = {}
        self.list_of_lists = list_of_lists

    def get(self, a_idx: Index, b_idx: Index) -> float:
        key = (a_idx, b_idx)
        reverse_key = (b_idx, a_idx)
        if key in self.cache:
            return self.cache[key]
        if reverse_key in self.cache:
            return self.cache[reverse_key]
        sim = self.sim_fn(
            self.list_of_lists[a_idx[0]][a_idx[1]],
            self.list_of_lists[b_idx[0]][b_idx[1]],
        )
        self.cache[key] = sim
        self.cache[reverse_key] = sim
        return sim


def _prune_low_support_elements(aligned_lists: list[list[Any]], min_support_ratio: float) -> list[list[Any]]:
    """Drop columns whose non-None support is below threshold (ref :109-149)."""
    if not aligned_lists:
        return aligned_lists
    n_lists = len(aligned_lists)
    n_cols_set = {len(lst) for lst in aligned_lists}
    if len(n_cols_set) > 1:
        logger.warning("All lists must have the same number of columns")
        return aligned_lists
    if not n_cols_set:
        return aligned_lists
    n_cols = n_cols_set.pop()
    if n_cols == 0:
        return aligned_lists

    support = []
    for col_idx in range(n_cols):
        non_none = sum(1 for lst in aligned_lists if lst[col_idx] is not None)
        support.append(non_none / n_lists)

    max_support = max(support)
    if max_support < min_support_ratio:
        logger.debug(
            "All columns below threshold; keeping columns at the highest support %s", max_support
        )
        min_support_ratio = max_support

    keep_cols = [i for i, s in enumerate(support) if s >= min_support_ratio]
    return [[lst[i] if i < len(lst) else None for i in keep_cols] for lst in aligned_lists]


def low_cutoff_bound(scores) -> float:
    """Detect a significant low-end jump in sorted scores (ref :152-175)."""
    if len(scores) == 0:
        return 0.0
    eps = 0.0001
    scores = np.sort(scores)
    low_cutoff = scores[0]
    # Look for a "jump" near the low end; only cut if the jump is significant.
    diffs = np.diff(scores[: int(0.2 * len(scores))])
    if len(diffs) > 0:
        jump_threshold = np.median(diffs) * 3
        jump_idx = np.argmax(diffs > jump_threshold)
        if diffs[jump_idx] > jump_threshold:
            low_cutoff = scores[jump_idx + 1] + eps
    return float(low_cutoff)


def remove_outliers(data: list[float]) -> list[float]:
    lower = low_cutoff_bound(data)
    return [el for el in data if el >= lower]


def _compute_dynamic_threshold(sim_cache: SimilarityCache) -> float:
    """Distribution-based threshold from best-match similarities (ref :185-252)."""
    list_of_lists = sim_cache.list_of_lists
    BASE_THRESHOLD = 0.5
    if not list_of_lists or len(list_of_lists) < 2:
        return BASE_THRESHOLD

    similarity_scores: list[float] = []
    total_lists = len(list_of_lists)

    for i in range(total_lists):
        list_i = list_of_lists[i]
        if not list_i:
            continue
        used_elements = {j: set() for j in range(total_lists) if j != i}
        for k_i in range(len(list_i)):
            best_match_score = BASE_THRESHOLD
            best_match: Optional[Index] = None
            for j in range(i + 1, total_lists):
                list_j = list_of_lists[j]
                if not list_j:
                    continue
                for k_j in range(len(list_j)):
                    if k_j in used_elements[j]:
                        continue
                    sim = sim_cache.get((i, k_i), (j, k_j))
                    if sim > best_match_score:
                        best_match_score = sim
                        best_match = (j, k_j)
            if best_match is not None and best_match_score > 0:
                similarity_scores.append(best_match_score)
                used_elements[best_match[0]].add(best_match[1])

    similarity_scores.sort()
    similarity_scores = remove_outliers(similarity_scores)
    if not similarity_scores:
        return BASE_THRESHOLD
    return max(BASE_THRESHOLD, 0.95 * similarity_scores[0])


def _build_reference_list(
    sim_cache: SimilarityCache,
    client: Any,
    min_support_ratio: float = 0.5,
    max_novelty_ratio: float = 0.5,
    threshold: float = 0.4,
) -> list[Index]:
    """Greedy support-grouping of every (list, pos) element (ref :255-333)."""
    list_of_lists = sim_cache.list_of_lists

    candidate_elements: list[Index] = [
        (list_idx, obj_pos) for list_idx, lst in enumerate(list_of_lists) for obj_pos in range(len(lst))
    ]

    support_groups: dict[Index, list[Index]] = defaultdict(list)
    support_groups_used_lists: dict[Index, set[int]] = defaultdict(set)

    for obj_index1 in candidate_elements:
        list_idx1 = obj_index1[0]
        best_sim = -1.0
        best_group_repr: Optional[Index] = None
        for group_repr, group_used_lists in support_groups_used_lists.items():
            if list_idx1 in group_used_lists:
                continue  # one element per source list per group
            sim = sim_cache.get(obj_index1, group_repr)
            if sim >= threshold and sim > best_sim:
                best_sim = sim
                best_group_repr = group_repr

        if best_group_repr is not None:
            support_groups[best_group_repr].append(obj_index1)
            support_groups_used_lists[best_group_repr].add(list_idx1)
            # Re-elect the representative: consensus over the INDEX TUPLES,
            # which routes to the similarity-medoid branch (ref :308-318).
            def dummy_embeddings_fn(strings):
                return [[0.0] * 10 for _ in strings]

            new_repr, _ = consensus_as_primitive(
                support_groups[best_group_repr],
                ConsensusSettings(),
                sync_get_openai_embeddings_from_text=dummy_embeddings_fn,
                client=client,
            )
            if new_repr != best_group_repr:
                support_groups[new_repr] = support_groups[best_group_repr]
                support_groups_used_lists[new_repr] = support_groups_used_lists[best_group_repr]
                del support_groups[best_group_repr]
                del support_groups_used_lists[best_group_repr]
        else:
            support_groups[obj_index1] = [obj_index1]
            support_groups_used_lists[obj_index1] = {list_idx1}

    support_ratios: dict[Index, float] = {
        k: len(v) / len(list_of_lists) for k, v in support_groups.items()
    }
    support_ratios = {k: v for k, v in support_ratios.items() if v >= min_support_ratio}
    support_ratios = dict(sorted(support_ratios.items(), key=lambda x: (-x[1], x[0])))
    return list(support_ratios.keys())


def _align_lists_to_reference_hungarian(
    sim_cache: SimilarityCache,
    reference_indices: list[Index],
    threshold: float = 0.4,
) -> list[list[Any]]:
    """Per-list Hungarian assignment against the reference (ref :336-379)."""
    list_of_lists = sim_cache.list_of_lists
    n_lists = len(list_of_lists)
    n_refs = len(reference_indices)

    aligned_lists: list[list[Any]] = [[None for _ in range(n_refs)] for _ in range(n_lists)]
    if not reference_indices:
        return aligned_lists

    for list_idx, lst in enumerate(list_of_lists):
        n_objs = len(lst)
        if n_objs == 0:
            continue
        sim_matrix = np.full((n_refs, n_objs), -np.inf)
        for ref_pos, ref_index in enumerate(reference_indices):
            for obj_pos in range(n_objs):
                obj_index = (list_idx, obj_pos)
                if obj_index == ref_index:
                    sim_matrix[ref_pos, obj_pos] = 1.0
                    continue
                sim_matrix[ref_pos, obj_pos] = sim_cache.get(obj_index, ref_index)

        cost_matrix = 1.0 - sim_matrix
        row_ind, col_ind = linear_sum_assignment(cost_matrix)
        for ref_pos, obj_pos in zip(row_ind, col_ind):
            sim = sim_matrix[ref_pos, obj_pos]
            if sim >= threshold and aligned_lists[list_idx][ref_pos] is None:
                aligned_lists[list_idx][ref_pos] = lst[obj_pos]

    return aligned_lists


def lists_alignment(
    list_of_lists: list[list[Any]],
    sim_fn: Callable[[Any, Any], float],
    client: Any = None,
    min_support_ratio: float = 0.5,
    max_novelty_ratio: float = 0.25,
    reference_list_idx: Optional[int] = None,
) -> tuple[list[list[Any]], list[list[int | None]]]:
    """Master alignment entry (ref :382-430). Returns (aligned, original_positions)."""
    if not list_of_lists or all(not lst for lst in list_of_lists):
        return [[] for _ in list_of_lists], [[None for _ in lst] for lst in list_of_lists]

    sim_cache = SimilarityCache(sim_fn, list_of_lists)

    if reference_list_idx is None:
        dynamic_threshold = _compute_dynamic_threshold(sim_cache)
        reference_list = _build_reference_list(
            sim_cache, client, min_support_ratio, max_novelty_ratio, threshold=dynamic_threshold
        )
        aligned = _align_lists_to_reference_hungarian(
            sim_cache, reference_list, threshold=0.95 * dynamic_threshold
        )
        aligned = _prune_low_support_elements(aligned, min_support_ratio)
        aligned, original_list_reference_indices = sort_by_original_majority(aligned, list_of_lists)
    else:
        # Known ground-truth reference: align everything to it, no pruning.
        reference_list = [(reference_list_idx, i) for i in range(len(list_of_lists[reference_list_idx]))]
        aligned = _align_lists_to_reference_hungarian(sim_cache, reference_list, threshold=0.0)
        original_list_reference_indices = _original_positions(aligned, list_of_lists)

    return aligned, original_list_reference_indices


def exists_nested_lists(values: list[Any]) -> bool:
    """True if any value nests a list, directly or under dicts (ref :433-455)."""
    if not values:
        return False
    for v in values:
        if isinstance(v, list):
            return True
        if isinstance(v, dict) and exists_nested_lists(list(v.values())):
            return True
    return False


def recursive_list_alignments(
    values: list[Any],
    string_similarity_method: StringSimilarityMethod,
    sync_get_openai_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    min_support_ratio: float = 0.5,
    max_novelty_ratio: float = 0.25,
    current_path: str = "",
    reference_idx: Optional[int] = None,
) -> tuple[list[Any], dict[str, list[str | None]]]:
    """Recursive structural walk aligning nested lists (ref :458-613).

    dicts -> per-key recursion (keys sorted); lists -> lists_alignment then
    per-column recursion. Returns (aligned values, key_mappings) where
    key_mappings maps each aligned dot-path to the original per-source paths.
    """
    if not values:
        return values, {}

    if all(v is None for v in values):
        return values, {current_path: [current_path for _ in values]}

    non_nulls = [v for v in values if v is not None]

    # Defensive copy: alignment mutates the structures in place below.
    values = deepcopy(values)

    first_type = type(non_nulls[0])
    same_type = all(isinstance(x, first_type) for x in non_nulls)
    key_mappings: dict[str, list[str | None]] = {}

    if not same_type or first_type not in (dict, list):
        key_mappings[current_path] = [
            current_path if (v is not None or idx == reference_idx) else None for idx, v in enumerate(values)
        ]
        return values, key_mappings

    if first_type is dict:
        dicts_only = [(d if isinstance(d, dict) else {}) for d in values]
        all_keys = sorted({k for d in dicts_only for k in d.keys()})

        for key in all_keys:
            values_for_key = [d.get(key) for d in dicts_only]
            _current_path = f"{current_path}.{key}" if current_path else key
            aligned_values_for_key, sub_key_mapping = recursive_list_alignments(
                values_for_key,
                string_similarity_method,
                sync_get_openai_embeddings_from_text,
                client,
                min_support_ratio,
                max_novelty_ratio=max_novelty_ratio,
                current_path=_current_path,
                reference_idx=reference_idx,
            )
            for _d, aligned_value in zip(dicts_only, aligned_values_for_key):
                _d[key] = aligned_value
            key_mappings.update(sub_key_mapping)

        values = [{k: _d.get(k) for k in all_keys} for _d in dicts_only]

    if first_type is list:
        lists_only = [(lst if isinstance(lst, list) else []) for lst in values]
        original_list_reference_indices: list[list[int | None]] = [[None for _ in lst] for lst in lists_only]

        if any(lst for lst in lists_only):
            def sim_fn(a, b):
                return generic_similarity(a, b, string_similarity_method, sync_get_openai_embeddings_from_text)

            aligned_lists_only, original_list_reference_indices = lists_alignment(
                lists_only,
                sim_fn,
                client,
                min_support_ratio=min_support_ratio,
                max_novelty_ratio=max_novelty_ratio,
                reference_list_idx=reference_idx,
            )
            for l_idx, new_lst in enumerate(aligned_lists_only):
                values[l_idx] = new_lst
        else:
            for i in range(len(values)):
                values[i] = []

        if len(values) > 0:
            list_length = len(values[0])
            if list_length > 0:
                for i in range(list_length):
                    values_i = [lst[i] for lst in values]
                    values_i, sub_key_mapping = recursive_list_alignments(
                        values_i,
                        string_similarity_method,
                        sync_get_openai_embeddings_from_text,
                        client,
                        min_support_ratio,
                        max_novelty_ratio=max_novelty_ratio,
                        current_path="",
                        reference_idx=reference_idx,
                    )
                    for l_idx, new_v in enumerate(values_i):
                        values[l_idx][i] = new_v

                    # Rewrite the key mapping through the original positions.
                    for key, sub_values in sub_key_mapping.items():
                        _key_path = f"{current_path}.{i}" if current_path else str(i)
                        _key_path = f"{_key_path}.{key}" if key else _key_path
                        current_values: list[str | None] = []
                        for l_idx, v in enumerate(sub_values):
                            _original_position = original_list_reference_indices[l_idx][i]
                            if _original_position is None or v is None:
                                current_values.append(None)
                            else:
                                _orig = (
                                    f"{current_path}.{_original_position}" if current_path else _original_position
                                )
                                _orig = f"{_orig}.{v}" if v else _orig
                                current_values.append(_orig)
                        key_mappings[_key_path] = current_values
            elif current_path:  # empty root paths unsupported
                key_mappings[current_path] = [current_path] * len(values)

    return values, key_mappings
