"""Consensus engine: alignment, similarity, voting, numeric clustering,
medoid election and confidence propagation.

This subpackage is the behavioral re-implementation of the reference's L1/L2
layers (k_llms/utils/consensus_utils.py, consolidation.py, majority_sorting.py)
with one shared sync implementation plus thin async bridges (the reference
duplicates ~800 LoC of line-for-line async mirrors; here the async API wraps
the shared implementation — same observable semantics).
"""

from .settings import (
    ConsensusSettings,
    SIMILARITY_SCORE_LOWER_BOUND,
    SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE,
    # reference-compatible aliases
    SYNC_GET_OPENAI_EMBEDDINGS_FROM_TEXT_TYPE,
    ASYNC_GET_OPENAI_EMBEDDINGS_FROM_TEXT_TYPE,
)
from .similarity import (
    cosine_similarity,
    generic_similarity,
    string_similarity,
    numerical_similarity,
    dict_similarity,
    list_similarity,
    normalize_string,
    sanitize_value,
    hamming_similarity,
    hamming_distance_padded,
    jaccard_similarity,
    levenshtein_similarity,
    key_normalization,
    compute_similarity_scores,
)
from .alignment import (
    SimilarityCache,
    lists_alignment,
    recursive_list_alignments,
    exists_nested_lists,
    low_cutoff_bound,
    remove_outliers,
)
from .majority_order import sort_by_original_majority
from .voting import voting_consensus
from .primitive import consensus_as_primitive, string_consensus_llm
from .values import consensus_values, consensus_dict, consensus_list, intermediary_consensus_cleanup
from .usage import consolidate_consensus_usage
from .consolidation import (
    consolidate_chat_completions,
    consolidate_parsed_chat_completions,
)
from .aio import (
    async_consensus_values,
    async_recursive_list_alignments,
    async_consolidate_chat_completions,
    async_consolidate_parsed_chat_completions,
    async_string_consensus_llm,
)

__all__ = [
    "ConsensusSettings",
    "SIMILARITY_SCORE_LOWER_BOUND",
    "SYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE",
    "ASYNC_GET_EMBEDDINGS_FROM_TEXT_TYPE",
    "SYNC_GET_OPENAI_EMBEDDINGS_FROM_TEXT_TYPE",
    "ASYNC_GET_OPENAI_EMBEDDINGS_FROM_TEXT_TYPE",
    "SimilarityCache",
    "cosine_similarity",
    "generic_similarity",
    "string_similarity",
    "numerical_similarity",
    "dict_similarity",
    "list_similarity",
    "normalize_string",
    "sanitize_value",
    "hamming_similarity",
    "hamming_distance_padded",
    "jaccard_similarity",
    "levenshtein_similarity",
    "key_normalization",
    "compute_similarity_scores",
    "lists_alignment",
    "recursive_list_alignments",
    "exists_nested_lists",
    "low_cutoff_bound",
    "remove_outliers",
    "sort_by_original_majority",
    "voting_consensus",
    "consensus_as_primitive",
    "string_consensus_llm",
    "consensus_values",
    "consensus_dict",
    "consensus_list",
    "intermediary_consensus_cleanup",
    "consolidate_consensus_usage",
    "consolidate_chat_completions",
    "consolidate_parsed_chat_completions",
    "async_consensus_values",
    "async_recursive_list_alignments",
    "async_consolidate_chat_completions",
    "async_consolidate_parsed_chat_completions",
    "async_string_consensus_llm",
]
