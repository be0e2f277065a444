"""Consensus dispatcher / voting / numeric clustering / medoid tests
(behavioral contract of reference consensus_utils.py:936-1454)."""

import pytest

from kllms_amd.consensus import (
    ConsensusSettings,
    consensus_as_primitive,
    consensus_values,
    voting_consensus,
)


def no_embed(texts):
    raise AssertionError("embeddings must not be called")


def lev_settings(**kw):
    return ConsensusSettings(string_similarity_method="levenshtein", **kw)


class TestVoting:
    def test_string_majority(self):
        val, conf = voting_consensus(["Paris", "paris", "London"], ConsensusSettings())
        assert val == "Paris"  # winner mapped back to first original surface form
        assert conf == pytest.approx(2 / 3, abs=1e-5)

    def test_bool_none_counts_false(self):
        val, conf = voting_consensus([True, None, None], ConsensusSettings())
        assert val is False
        assert conf == pytest.approx(2 / 3, abs=1e-5)

    def test_all_none(self):
        val, conf = voting_consensus([None, None], ConsensusSettings(), parent_valid_frac=0.7)
        assert val is None
        assert conf == 0.7

    def test_none_excluded_for_strings(self):
        val, conf = voting_consensus(["a", "a", None, None, None], ConsensusSettings())
        assert val == "a"
        assert conf == pytest.approx(2 / 5, abs=1e-5)

    def test_sanitized_equivalence(self):
        val, _ = voting_consensus(["été", "ete!", "x"], ConsensusSettings())
        assert val == "été"


class TestEnumDispatch:
    def test_short_strings_vote(self):
        val, conf = consensus_values(["yes", "yes", "no"], lev_settings(), no_embed)
        assert val == "yes"
        assert conf == pytest.approx(2 / 3, abs=1e-5)

    def test_long_strings_medoid(self):
        vals = ["the quick brown fox jumps", "the quick brown fox jumped", "something else entirely here"]
        val, conf = consensus_values(vals, lev_settings(), no_embed)
        assert val in vals[:2]
        assert 0 < conf <= 1

    def test_three_word_strings_not_enum(self):
        # >= 3 words -> primitive consensus (medoid), not voting
        vals = ["a b c", "a b c", "a b d"]
        val, conf = consensus_values(vals, lev_settings(), no_embed)
        assert val == "a b c"


class TestNumericConsensus:
    def test_majority_cluster_mean(self):
        val, conf = consensus_as_primitive([10.0, 10.1, 50.0], lev_settings(), no_embed)
        # 10.0 and 10.1 cluster (within 3%), mean = 10.05
        assert val == pytest.approx(10.05)
        assert conf == pytest.approx(2 / 3, abs=1e-4)

    def test_exact_majority(self):
        val, conf = consensus_as_primitive([5, 5, 5, 7], lev_settings(), no_embed)
        assert val == 5.0
        assert conf == pytest.approx(0.75)

    def test_power10_tiebreak(self):
        # two singleton clusters; 100 vs 1000 match via power-of-10 support
        val, conf = consensus_as_primitive([100.0, 1000.0, 333.0], lev_settings(), no_embed)
        assert val in (100.0, 1000.0, 333.0)

    def test_int_passthrough(self):
        val, conf = consensus_as_primitive([3, 3], lev_settings(), no_embed)
        assert val == 3.0
        assert conf == 1.0

    def test_single_non_none_early_return(self):
        # one non-None value short-circuits before the numeric branch
        val, conf = consensus_as_primitive([None, None, None, 4.0], lev_settings(), no_embed)
        assert val == 4.0
        assert conf == pytest.approx(0.25)

    def test_none_plurality_wins(self):
        # Nones outnumber the largest numeric cluster -> None wins
        val, conf = consensus_as_primitive([None, None, None, 4.0, 9.0], lev_settings(), no_embed)
        assert val is None
        assert conf == pytest.approx(0.6)

    def test_bools_route_numeric_and_yield_none(self):
        # bool-typed non-enum path: xs stays empty -> (None, pvf). Observable
        # reference behavior (consensus_utils.py:1102,1110-1117).
        val, conf = consensus_as_primitive([True, False, True], lev_settings(), no_embed, parent_valid_frac=0.9)
        assert val is None
        assert conf == 0.9


class TestStructuralDispatch:
    def test_dict_recursion_and_likelihood_shape(self):
        dicts = [
            {"name": "Alice", "age": 30},
            {"name": "Alice", "age": 30},
            {"name": "Bob", "age": 31},
        ]
        val, conf = consensus_values(dicts, lev_settings(), no_embed)
        assert val["name"] == "Alice"
        assert val["age"] == pytest.approx(30.0)
        assert set(conf.keys()) == {"name", "age"}

    def test_dict_skips_reasoning_fields(self):
        dicts = [{"a": "x", "reasoning___a": "r1"}, {"a": "x", "reasoning___a": "r2"}]
        val, conf = consensus_values(dicts, lev_settings(), no_embed)
        assert "reasoning___a" not in val
        assert "reasoning___a" not in conf

    def test_parent_valid_frac_scaling(self):
        # one of three values is not a dict -> frac scaled by 2/3
        vals = [{"a": "x"}, {"a": "x"}, "oops"]
        val, conf = consensus_values(vals, lev_settings(), no_embed)
        assert val == {"a": "x"}
        assert conf["a"] == pytest.approx(2 / 3, abs=1e-4)

    def test_list_elementwise(self):
        lists = [["a", "b"], ["a", "b"], ["a", "c"]]
        val, conf = consensus_values(lists, lev_settings(), no_embed)
        assert val == ["a", "b"]
        assert conf[0] == pytest.approx(1.0)
        assert conf[1] == pytest.approx(2 / 3, abs=1e-4)

    def test_empty_values(self):
        assert consensus_values([], lev_settings(), no_embed) == (None, 1.0)
        assert consensus_values([None, None], lev_settings(), no_embed) == (None, 0.0)

    def test_confidence_rounding_5dp(self):
        _, conf = consensus_values(["a", "a", "b", "c", "d", "e", "f"], lev_settings(), no_embed)
        assert conf == round(2 / 7, 5)
