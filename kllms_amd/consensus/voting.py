"""Vote-based (enum-like) consensus (ref consensus_utils.py:936-982)."""

from __future__ import annotations

from collections import Counter

from .settings import ConsensusSettings, logger
from .similarity import sanitize_value


def voting_consensus(
    values: list[str | bool | None],
    consensus_settings: ConsensusSettings,
    verbose: bool = False,
    parent_valid_frac: float = 1.0,
) -> tuple[str | bool | None, float]:
    """Pick the most common value.

    Behavioral contract (ref :936-982):
    - all-None input -> (None, parent_valid_frac) unrounded;
    - booleans: None counts as False;
    - strings: vote on sanitized forms (None excluded unless
      allow_none_as_candidate), winner mapped back to the FIRST original
      surface form with that sanitized key;
    - confidence = parent_valid_frac * best_count / total, rounded to 5dp.
    """
    total_values = len(values)

    if not any(v is not None for v in values):
        return (None, parent_valid_frac)

    first_non_none = next((v for v in values if v is not None), None)
    is_boolean = isinstance(first_non_none, bool)

    if is_boolean:
        processed_values = []
        for v in values:
            v2 = v or False  # None counts as False (ref :954-958)
            try:
                hash(v2)
            except TypeError:
                # hardening over the reference (which crashes in Counter):
                # a truthy unhashable candidate (list/dict mixed in with
                # bools) votes as True
                v2 = True
            processed_values.append(v2)
        counts = Counter(processed_values)
        best_val, best_count = counts.most_common(1)[0]
    else:
        if consensus_settings.allow_none_as_candidate:
            valid_values = values
        else:
            valid_values = [v for v in values if v is not None]
        processed = [(sanitize_value(v) if v is not None else None) for v in valid_values]
        counts = Counter(processed)
        best_normalized, best_count = counts.most_common(1)[0]
        best_val = valid_values[processed.index(best_normalized)]

    confidence = parent_valid_frac * (best_count / total_values)

    if verbose:
        logger.debug(
            "voting_consensus: values=%r best=%r count=%d conf=%f",
            values, best_val, best_count, confidence,
        )

    return (best_val, round(confidence, 5))
