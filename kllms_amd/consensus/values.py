"""Consensus dispatcher and structural (dict/list) recursion.

Behavioral re-implementation of consensus_values / consensus_dict /
consensus_list (ref consensus_utils.py:1269-1454). The dispatcher defines the
``likelihoods`` schema: nested dicts for objects, lists for arrays, floats at
leaves. ``parent_valid_frac`` multiplies down the tree at each level by the
fraction of non-null / correctly-typed values (ref :1414-1444).
"""

from __future__ import annotations

from typing import Any

from .primitive import consensus_as_primitive
from .settings import (
    SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    ConsensusSettings,
)
from .voting import voting_consensus

# consensus_dict skips keys CONTAINING these prefixes anywhere
# (substring containment, ref :1287-1294 — contrast with dict_similarity's
# match-at-start behavior).
_SPECIAL_FIELD_PREFIXES = ["reasoning___", "source___"]


def consensus_dict(
    dict_values: list[dict],
    consensus_settings: ConsensusSettings,
    sync_get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    parent_valid_frac: float = 1.0,
) -> tuple[dict, dict[str, Any]]:
    """Field-wise merge; keys in first-seen order across dicts (ref :1269-1306)."""
    seen: set = set()
    all_keys = [k for d in dict_values for k in d.keys() if k not in seen and not seen.add(k)]

    result: dict = {}
    confs: dict = {}
    for key in all_keys:
        if any(prefix in key for prefix in _SPECIAL_FIELD_PREFIXES):
            continue  # reasoning/source fields are excluded from consensus
        sub_vals = [d.get(key, None) for d in dict_values]
        val, conf = consensus_values(
            sub_vals,
            consensus_settings,
            sync_get_embeddings_from_text,
            parent_valid_frac=parent_valid_frac,
            client=client,
        )
        result[key] = val
        confs[key] = conf
    return (result, confs)


def consensus_list(
    list_values: list[list[Any]],
    consensus_settings: ConsensusSettings,
    sync_get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    parent_valid_frac: float = 1.0,
) -> tuple[list[Any], list[Any]]:
    """Element-wise merge over the max length (ref :1309-1352)."""
    if not list_values:
        return ([], [])
    non_empty = [lst for lst in list_values if lst]
    if not non_empty:
        return ([], [])

    maximum_len = max(len(lst) for lst in list_values)
    if maximum_len == 0:
        return ([], [])

    final_list: list[Any] = []
    confidences: list[Any] = []
    for i in range(maximum_len):
        items = [(lst[i] if i < len(lst) else None) for lst in list_values]
        val_i, conf_i = consensus_values(
            items,
            consensus_settings,
            sync_get_embeddings_from_text,
            parent_valid_frac=parent_valid_frac,
            client=client,
        )
        final_list.append(val_i)
        confidences.append(conf_i)
    return final_list, confidences


def intermediary_consensus_cleanup(obj: Any) -> Any:
    """Strip empty strings / empty containers recursively (ref :1355-1370)."""
    if isinstance(obj, dict):
        new_obj = {k: w for k, v in obj.items() if (w := intermediary_consensus_cleanup(v)) is not None}
        return new_obj if new_obj else None
    if isinstance(obj, (list, tuple)):
        new_list = [w for v in obj if (w := intermediary_consensus_cleanup(v)) is not None]
        return new_list if new_list else None
    if isinstance(obj, str):
        stripped = obj.strip()
        return stripped if stripped else None
    return obj


def consensus_values(
    values: list[Any],
    consensus_settings: ConsensusSettings,
    sync_get_openai_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    client: Any = None,
    parent_valid_frac: float = 1.0,
) -> tuple[Any, Any]:
    """Type-routing dispatcher (ref consensus_utils.py:1376-1454).

    - str/bool where every value has < 3 whitespace-separated words
      ("enum-like") -> voting_consensus;
    - dict -> consensus_dict (parent_valid_frac scaled by dict-typed fraction);
    - list -> consensus_list (same scaling);
    - else -> consensus_as_primitive over the non-None values.
    """
    if not values:
        return (None, parent_valid_frac)

    non_none_values = [v for v in values if v is not None]
    if not non_none_values:
        return (None, 0.0)

    first = non_none_values[0]

    if isinstance(first, (str, bool)):
        values_as_strings = [str(v).strip() for v in non_none_values]
        is_enum_like = all(len(v.split()) < 3 for v in values_as_strings)
        if is_enum_like:
            return voting_consensus(values, consensus_settings, parent_valid_frac=parent_valid_frac)

    if isinstance(first, dict):
        dicts_only = [v for v in values if isinstance(v, dict)]
        parent_valid_frac *= len(dicts_only) / len(values)
        return consensus_dict(
            dicts_only,
            consensus_settings,
            sync_get_openai_embeddings_from_text,
            parent_valid_frac=parent_valid_frac,
            client=client,
        )

    if isinstance(first, list):
        lists_only = [v for v in values if isinstance(v, list)]
        parent_valid_frac *= len(lists_only) / len(values)
        return consensus_list(
            lists_only,
            consensus_settings,
            sync_get_openai_embeddings_from_text,
            parent_valid_frac=parent_valid_frac,
            client=client,
        )

    parent_valid_frac *= len(non_none_values) / len(values)
    if sync_get_openai_embeddings_from_text is None:
        raise ValueError("an embeddings function is required for primitive consensus")
    return consensus_as_primitive(
        non_none_values,
        consensus_settings,
        sync_get_openai_embeddings_from_text,
        parent_valid_frac=parent_valid_frac,
        client=client,
    )
