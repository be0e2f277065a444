"""Similarity suite: string / numeric / dict / list / generic.

Behavioral parity notes (reference = k_llms/utils/consensus_utils.py):

- ``cosine_similarity`` rescales to [0,1] via 0.5*(cos+1) and clips to
  [1e-8, 1] (ref :626-649) — the rescale shifts every embedding-based
  threshold, so it is kept.
- ``string_similarity`` (ref :797-824): the "embeddings" method only fires
  when BOTH strings are longer than 50 chars; otherwise (and on any
  embedding failure) it falls back to Levenshtein. Results are memoized in a
  lock-guarded TTL cache keyed (min(s1,s2), max(s1,s2), method).
- ``generic_similarity`` (ref :895-917): the falsy quirk — two falsy values
  (``0``, ``""``, ``[]``, ``None``, ``False``) compare as identical (1.0) —
  is observable behavior and is kept.
- ``numerical_similarity`` (ref :827-841): bools exact; numbers equal iff
  isclose(rel_tol=0.01); floor 1e-8.
- ``dict_similarity`` (ref :844-870): key-union average, ignoring keys that
  MATCH (at string start) the reasoning___/source___ patterns.
- ``list_similarity`` (ref :873-892): positional average over max length,
  missing positions compared as None.
"""

from __future__ import annotations

import re
from itertools import zip_longest
from math import isclose
from threading import Lock
from typing import Any, Callable

import numpy as np

from ..utils.text import ascii_transliterate, levenshtein_distance
from ..utils.ttl_cache import TTLCache
from .settings import (
    IGNORED_KEY_PATTERNS,
    SIMILARITY_SCORE_LOWER_BOUND,
    SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    ConsensusSettings,
    NumericalPrimitive,
    StringSimilarityMethod,
    logger,
)

# --- caching (ref :620-623, :780-794) ---------------------------------------
embeddings_cache = TTLCache(maxsize=1024, ttl=300)
similarity_cache = TTLCache(maxsize=1024, ttl=300)
embeddings_cache_lock = Lock()
similarity_cache_lock = Lock()


def _get_cached_similarity(s1: str, s2: str, method: str) -> float | None:
    key = (min(s1, s2), max(s1, s2), method)
    with similarity_cache_lock:
        return similarity_cache.get(key)


def _set_cached_similarity(s1: str, s2: str, method: str, value: float) -> None:
    key = (min(s1, s2), max(s1, s2), method)
    with similarity_cache_lock:
        similarity_cache[key] = value


def get_embeddings(s: str, sync_get_embeddings_from_text: Callable[[list[str]], list[list[float]]]) -> list[float]:
    """Single-string wrapper over the injected batch embed fn (ref :652-657)."""
    logger.debug("Embedding computed for %r", s[:40])
    return sync_get_embeddings_from_text([s])[0]


# --- basic string metrics ----------------------------------------------------

def normalize_string(text: str) -> str:
    """Lowercase and strip everything non-alphanumeric (ref :660-673)."""
    if not text:
        return ""
    return re.sub(r"[^a-zA-Z0-9]", "", text).lower()


def hamming_distance_padded(s: str, t: str) -> int:
    """Hamming distance on normalized strings, padding with spaces (ref :676-694)."""
    s = normalize_string(s)
    t = normalize_string(t)
    return sum(a != b for a, b in zip_longest(s, t, fillvalue=" "))


def hamming_similarity(str_1: str, str_2: str) -> float:
    str_1 = normalize_string(str_1)
    str_2 = normalize_string(str_2)
    max_length = max(len(str_1), len(str_2))
    if max_length == 0:
        return 1.0
    dist = hamming_distance_padded(str_1, str_2)
    return max(SIMILARITY_SCORE_LOWER_BOUND, 1 - (dist / max_length))


def jaccard_similarity(str_1: str, str_2: str) -> float:
    """Character-set Jaccard on normalized strings (ref :720-742)."""
    str_1 = normalize_string(str_1)
    str_2 = normalize_string(str_2)
    set_a, set_b = set(str_1), set(str_2)
    union = set_a | set_b
    if not union:
        return 1.0
    return max(SIMILARITY_SCORE_LOWER_BOUND, len(set_a & set_b) / len(union))


def levenshtein_similarity(str_1: str, str_2: str) -> float:
    """1 - normalized edit distance on normalized strings (ref :745-761)."""
    str_1 = normalize_string(str_1)
    str_2 = normalize_string(str_2)
    max_length = max(len(str_1), len(str_2))
    if max_length == 0:
        return 1.0
    dist = levenshtein_distance(str_1, str_2)
    return max(SIMILARITY_SCORE_LOWER_BOUND, 1 - (dist / max_length))


def cosine_similarity(vec1: list[float], vec2: list[float]) -> float:
    """Cosine similarity rescaled to [0,1] via 0.5*(cos+1) (ref :626-649)."""
    arr1 = np.asarray(vec1, dtype=float)
    arr2 = np.asarray(vec2, dtype=float)
    if arr1.shape != arr2.shape:
        raise ValueError("Vectors must have the same shape for cosine similarity")
    norm1 = np.linalg.norm(arr1)
    norm2 = np.linalg.norm(arr2)
    if norm1 == 0 or norm2 == 0:
        return SIMILARITY_SCORE_LOWER_BOUND
    similarity = float(np.dot(arr1, arr2) / (norm1 * norm2))
    similarity = 0.5 * (similarity + 1.0)
    return float(np.clip(similarity, SIMILARITY_SCORE_LOWER_BOUND, 1.0))


def key_normalization(key: str) -> str:
    """Replace numeric path segments with '*' (ref :764-774)."""
    return ".".join("*" if part.isdigit() else part for part in key.split("."))


def sanitize_value(v: str | bool) -> str:
    """Voting equivalence key: lowercase, de-space, transliterate, alnum-only
    (ref :925-933)."""
    s = str(v).lower()
    s = s.replace(" ", "")
    s = ascii_transliterate(s)
    return re.sub(r"[^a-zA-Z0-9]", "", s)


# --- dispatchers --------------------------------------------------------------

def string_similarity(
    s1: str,
    s2: str,
    method: StringSimilarityMethod,
    sync_get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
) -> float:
    cached = _get_cached_similarity(s1, s2, method)
    if cached is not None:
        return cached
    result: float | None = None
    if method == "jaccard":
        result = jaccard_similarity(s1, s2)
    elif method == "hamming":
        result = hamming_similarity(s1, s2)
    elif method == "embeddings" and len(s1) > 50 and len(s2) > 50:
        # Embeddings only pay off for long strings (ref :813)
        try:
            result = cosine_similarity(
                get_embeddings(s1, sync_get_embeddings_from_text),
                get_embeddings(s2, sync_get_embeddings_from_text),
            )
        except Exception:
            logger.exception("Error getting embeddings for %r / %r", s1[:40], s2[:40])
    if result is None:
        result = levenshtein_similarity(s1, s2)
    _set_cached_similarity(s1, s2, method, result)
    return result


def numerical_similarity(val1: NumericalPrimitive, val2: NumericalPrimitive) -> float:
    if isinstance(val1, bool) and isinstance(val2, bool):
        return 1.0 if val1 == val2 else SIMILARITY_SCORE_LOWER_BOUND
    if isinstance(val1, (int, float)) and isinstance(val2, (int, float)) and isclose(val1, val2, rel_tol=0.01):
        return 1.0
    return 1.0 if val1 == val2 else SIMILARITY_SCORE_LOWER_BOUND


def dict_similarity(
    d1: dict,
    d2: dict,
    string_similarity_method: StringSimilarityMethod,
    sync_get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
) -> float:
    all_keys = set(d1.keys()) | set(d2.keys())
    keys = [k for k in all_keys if not any(re.match(p, k) for p in IGNORED_KEY_PATTERNS)]
    if not keys:
        return 1.0
    total = 0.0
    for k in keys:
        total += generic_similarity(d1.get(k), d2.get(k), string_similarity_method, sync_get_embeddings_from_text)
    return total / len(keys)


def list_similarity(
    l1: list | tuple,
    l2: list | tuple,
    string_similarity_method: StringSimilarityMethod,
    sync_get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
) -> float:
    max_len = max(len(l1), len(l2))
    if max_len == 0:
        return 1.0
    total = 0.0
    for i in range(max_len):
        v1 = l1[i] if i < len(l1) else None
        v2 = l2[i] if i < len(l2) else None
        total += generic_similarity(v1, v2, string_similarity_method, sync_get_embeddings_from_text)
    return total / max_len


def generic_similarity(
    v1: Any,
    v2: Any,
    string_similarity_method: StringSimilarityMethod,
    sync_get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
) -> float:
    # Falsy quirk (ref :903-904): two falsy values are "identical".
    if not bool(v1) and not bool(v2):
        return 1.0
    if v1 is None or v2 is None:
        return SIMILARITY_SCORE_LOWER_BOUND
    if isinstance(v1, str) and isinstance(v2, str):
        return string_similarity(v1, v2, string_similarity_method, sync_get_embeddings_from_text)
    if isinstance(v1, NumericalPrimitive) and isinstance(v2, NumericalPrimitive):
        return numerical_similarity(v1, v2)
    if isinstance(v1, dict) and isinstance(v2, dict):
        return dict_similarity(v1, v2, string_similarity_method, sync_get_embeddings_from_text)
    if isinstance(v1, (list, tuple)) and isinstance(v2, (list, tuple)):
        return list_similarity(v1, v2, string_similarity_method, sync_get_embeddings_from_text)
    return SIMILARITY_SCORE_LOWER_BOUND


def compute_similarity_scores(
    values: list[Any],
    consensus_settings: ConsensusSettings,
    sync_get_embeddings_from_text: SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
) -> list[float]:
    """Mean pairwise similarity per candidate (ref :1243-1263); useful as the
    per-choice score vector."""
    n = len(values)
    if n == 0:
        return []
    if n == 1:
        return [1.0]
    sim_matrix = np.zeros((n, n), dtype=float)
    for i in range(n):
        for j in range(i + 1, n):
            sim = generic_similarity(
                values[i], values[j], consensus_settings.string_similarity_method, sync_get_embeddings_from_text
            )
            sim_matrix[i, j] = sim_matrix[j, i] = sim
        sim_matrix[i, i] = 1.0
    return [float(round(score, 5)) for score in sim_matrix.mean(axis=1)]
