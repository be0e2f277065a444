"""Consensus configuration and shared type aliases.

Reference parity: ConsensusSettings and every default value match
k_llms/utils/consensus_utils.py:53-69 — these constants define the observable
output behavior (cluster widths, vote thresholds, trim fractions) and must not
drift.
"""

from __future__ import annotations

import logging
import os
from typing import Awaitable, Callable, Literal

from pydantic import BaseModel

NumericalPrimitive = int | float
EnumLikeType = str | bool

StringSimilarityMethod = Literal["levenshtein", "jaccard", "hamming", "embeddings"]
StringConsensusMethod = Literal["centroid", "llm-consensus"]

# Injectable embedding providers (reference: consensus_utils.py:29-30). The
# reference names carry "OPENAI"; locally the embeddings come from the MI355X
# engine's embedder, but the callable contract is identical, and the original
# names are kept as aliases for drop-in compatibility.
SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE = Callable[[list[str]], list[list[float]]]
ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE = Callable[[list[str]], Awaitable[list[list[float]]]]
SYNC_GET_OPENAI_EMBEDDINGS_FROM_TEXT_TYPE = SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE
ASYNC_GET_OPENAI_EMBEDDINGS_FROM_TEXT_TYPE = ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE

SIMILARITY_SCORE_LOWER_BOUND = 1e-8

# Keys matching these patterns are ignored by dict similarity (match at key
# start — consensus_utils.py:38-43,858) and skipped entirely by consensus_dict
# (substring containment — consensus_utils.py:1287-1294).
IGNORED_KEY_PATTERNS = [r"reasoning___", r"source___"]

logger = logging.getLogger("kllms_amd.consensus")
if os.getenv("ENV_NAME") == "dev":
    logger.setLevel(logging.DEBUG)
else:
    logger.setLevel(logging.INFO)


class ConsensusSettings(BaseModel):
    """All consensus behavior knobs (defaults = reference defaults)."""

    allow_none_as_candidate: bool = False
    # String-specific settings
    string_similarity_method: StringSimilarityMethod = "embeddings"
    string_consensus_method: StringConsensusMethod = "centroid"
    # Alignment thresholds
    minimum_voters_threshold: float = 0.75
    min_support_ratio: float = 0.51  # at least 51% of voters must agree
    # Numeric consensus (hybrid vote-or-mean) clustering tolerances
    rel_eps: float = 0.03
    abs_eps: float = 1e-6
    # Majority threshold for voting
    base_maj_thresh: float = 0.6
    maj_loosen_k: float = 0.1
    # Robust trimmed mean fraction (n >= 5)
    trim_frac: float = 0.2
