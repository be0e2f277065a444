"""Fuzzy key selection: canonicalized-value fallback.

Behavioral re-implementation of k_llms/utils/fuzzy_key_selection.py:29-232:
the same cascade run on canonicalized scalars (numbers rounded to N decimals,
strings lowercased / whitespace-collapsed); the fuzzy result is chosen only
when it strictly improves the stability tuple. Unlike the reference (which
duplicates the cascade), this reuses key_selection.cascade_select_keys with a
canonicalizer.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from pydantic import BaseModel, ConfigDict

from .key_selection import (
    CascadeConfig,
    KeyMetrics,
    cascade_select_keys,
    discover_scalar_paths,
    select_best_keys,
    stability_tuple,
)


def _normalize_string(value: str) -> str:
    return " ".join(value.strip().lower().split())


def _canonicalize_scalar(value: Any, numeric_round_decimals: int) -> Any:
    """Numbers rounded to N decimals; strings normalized; bool/None as-is
    (ref :40-53)."""
    if isinstance(value, (int, float)) and not isinstance(value, bool):
        try:
            return round(float(value), numeric_round_decimals)
        except Exception:
            return value
    if isinstance(value, str):
        return _normalize_string(value)
    return value


class SelectionComparison(BaseModel):
    model_config = ConfigDict(frozen=True)

    normal_best: Optional[KeyMetrics]
    fuzzy_best: Optional[KeyMetrics]
    chosen: str  # "normal" | "fuzzy"


def select_best_keys_with_fuzzy_fallback(
    extractions: List[Dict[str, Any]],
    cascade_cfg: CascadeConfig = None,  # type: ignore[assignment]
    list_key: Optional[str] = None,
    fuzzy_numeric_round_decimals: int = 2,
    enable_fuzzy_fallback: bool = True,
    prefer_fuzzy_if_better: bool = True,
) -> SelectionComparison:
    """Standard selection, then fuzzy; fuzzy wins only on a strictly better
    stability tuple (ref :175-232)."""
    if cascade_cfg is None:
        cascade_cfg = CascadeConfig()

    normal_best: Optional[KeyMetrics] = None
    try:
        normal_best = select_best_keys(extractions, cascade_cfg=cascade_cfg, list_key=list_key).best_single
    except ValueError:
        normal_best = None

    fuzzy_best: Optional[KeyMetrics] = None
    if enable_fuzzy_fallback:
        candidates = discover_scalar_paths(extractions, list_key=list_key)
        if candidates:
            try:
                fuzzy_best = cascade_select_keys(
                    extractions,
                    candidates,
                    cascade_cfg,
                    list_key=list_key,
                    canon=lambda v: _canonicalize_scalar(v, fuzzy_numeric_round_decimals),
                ).final_best
            except ValueError:
                fuzzy_best = None

    if normal_best is None and fuzzy_best is None:
        raise ValueError("No keys pass Stage 0 (normal or fuzzy)")
    if normal_best is not None and (not enable_fuzzy_fallback or fuzzy_best is None):
        return SelectionComparison(normal_best=normal_best, fuzzy_best=None, chosen="normal")
    if normal_best is None:
        return SelectionComparison(normal_best=None, fuzzy_best=fuzzy_best, chosen="fuzzy")
    if prefer_fuzzy_if_better and stability_tuple(fuzzy_best) > stability_tuple(normal_best):
        return SelectionComparison(normal_best=normal_best, fuzzy_best=fuzzy_best, chosen="fuzzy")
    return SelectionComparison(normal_best=normal_best, fuzzy_best=fuzzy_best, chosen="normal")
