"""Similarity suite unit tests (behavioral contract of reference
consensus_utils.py:620-917)."""

import math

import pytest

from kllms_amd.consensus import (
    SIMILARITY_SCORE_LOWER_BOUND,
    cosine_similarity,
    dict_similarity,
    generic_similarity,
    hamming_similarity,
    jaccard_similarity,
    levenshtein_similarity,
    normalize_string,
    numerical_similarity,
    sanitize_value,
    string_similarity,
    key_normalization,
    compute_similarity_scores,
)
from kllms_amd.consensus.settings import ConsensusSettings


def no_embed(texts):
    raise AssertionError("embeddings must not be called")


class TestNormalize:
    def test_normalize_string(self):
        assert normalize_string("Hello, World! 123") == "helloworld123"
        assert normalize_string("") == ""

    def test_sanitize_value(self):
        assert sanitize_value("Héllo Wörld") == "helloworld"
        assert sanitize_value(True) == "true"
        assert sanitize_value("A-B_c") == "abc"

    def test_key_normalization(self):
        assert key_normalization("a.0.b.12.c") == "a.*.b.*.c"


class TestStringMetrics:
    def test_levenshtein(self):
        assert levenshtein_similarity("abc", "abc") == 1.0
        assert levenshtein_similarity("", "") == 1.0
        assert levenshtein_similarity("abc", "abd") == pytest.approx(2 / 3)
        # normalized before comparison
        assert levenshtein_similarity("A B C", "abc") == 1.0

    def test_jaccard(self):
        assert jaccard_similarity("abc", "bca") == 1.0
        assert jaccard_similarity("", "") == 1.0
        assert jaccard_similarity("ab", "bc") == pytest.approx(1 / 3)

    def test_hamming(self):
        assert hamming_similarity("abc", "abc") == 1.0
        assert hamming_similarity("abc", "abd") == pytest.approx(2 / 3)
        # shorter string padded with spaces
        assert hamming_similarity("ab", "abcd") == pytest.approx(0.5)

    def test_cosine_rescaled(self):
        # orthogonal vectors -> 0.5 after the 0.5*(cos+1) rescale
        assert cosine_similarity([1, 0], [0, 1]) == pytest.approx(0.5)
        assert cosine_similarity([1, 0], [1, 0]) == pytest.approx(1.0)
        assert cosine_similarity([1, 0], [-1, 0]) == pytest.approx(SIMILARITY_SCORE_LOWER_BOUND)
        assert cosine_similarity([0, 0], [1, 0]) == SIMILARITY_SCORE_LOWER_BOUND


class TestStringSimilarityDispatch:
    def test_short_strings_fall_back_to_levenshtein(self):
        # embeddings method with <=50-char strings must NOT call the embed fn
        r = string_similarity("short string", "short strong", "embeddings", no_embed)
        assert r == levenshtein_similarity("short string", "short strong")

    def test_long_strings_use_embeddings(self, fake_embed):
        s1 = "a" * 60
        s2 = "a" * 61
        r = string_similarity(s1, s2, "embeddings", fake_embed)
        assert 0.0 < r <= 1.0

    def test_embed_failure_falls_back(self):
        s1 = "x" * 60
        s2 = "y" * 60
        r = string_similarity(s1, s2, "embeddings", no_embed)
        assert r == levenshtein_similarity(s1, s2)


class TestNumericalSimilarity:
    def test_bool_exact(self):
        assert numerical_similarity(True, True) == 1.0
        assert numerical_similarity(True, False) == SIMILARITY_SCORE_LOWER_BOUND

    def test_relative_tolerance(self):
        assert numerical_similarity(100.0, 100.9) == 1.0  # within 1%
        assert numerical_similarity(100.0, 105.0) == SIMILARITY_SCORE_LOWER_BOUND


class TestGenericSimilarity:
    def test_falsy_quirk(self):
        # 0, "", [], None, False are all mutually identical (ref :903-904)
        for a in (0, "", [], None, False):
            for b in (0, "", [], None, False):
                assert generic_similarity(a, b, "levenshtein", no_embed) == 1.0

    def test_one_none(self):
        assert generic_similarity(None, "x", "levenshtein", no_embed) == SIMILARITY_SCORE_LOWER_BOUND

    def test_dict_ignores_reasoning_keys(self):
        d1 = {"a": "x", "reasoning___a": "foo"}
        d2 = {"a": "x", "reasoning___a": "completely different"}
        assert dict_similarity(d1, d2, "levenshtein", no_embed) == 1.0

    def test_list_positional(self):
        assert generic_similarity([1, 2], [1, 2], "levenshtein", no_embed) == 1.0
        # missing position compared as None -> lower bound contribution
        r = generic_similarity([1, 2], [1], "levenshtein", no_embed)
        assert r == pytest.approx((1.0 + SIMILARITY_SCORE_LOWER_BOUND) / 2)

    def test_tuple_as_list(self):
        assert generic_similarity((0, 1), (0, 1), "levenshtein", no_embed) == 1.0
        assert generic_similarity((0, 1), (0, 5), "levenshtein", no_embed) == pytest.approx(
            (1.0 + SIMILARITY_SCORE_LOWER_BOUND) / 2
        )


def test_compute_similarity_scores():
    scores = compute_similarity_scores(["abc", "abc", "xyz"], ConsensusSettings(), no_embed)
    assert len(scores) == 3
    assert scores[0] == scores[1] > scores[2]
