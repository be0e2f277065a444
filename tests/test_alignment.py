"""List alignment + recursive alignment tests (behavioral contract of
reference consensus_utils.py:81-613, majority_sorting.py)."""

import pytest

from kllms_amd.consensus import (
    lists_alignment,
    recursive_list_alignments,
    exists_nested_lists,
    low_cutoff_bound,
    sort_by_original_majority,
)
from kllms_amd.consensus.similarity import generic_similarity


def no_embed(texts):
    raise AssertionError("embeddings must not be called")


def sim_fn(a, b):
    return generic_similarity(a, b, "levenshtein", no_embed)


class TestListsAlignment:
    def test_identical_lists(self):
        lists = [["apple", "banana"], ["apple", "banana"], ["apple", "banana"]]
        aligned, orig_idx = lists_alignment(lists, sim_fn, min_support_ratio=0.51)
        assert all(len(lst) == 2 for lst in aligned)
        for lst in aligned:
            assert set(lst) == {"apple", "banana"}
        # column order follows majority original order
        assert aligned[0] == ["apple", "banana"]
        assert orig_idx[0] == [0, 1]

    def test_permuted_lists(self):
        lists = [["apple pie", "banana split"], ["banana split", "apple pie"], ["apple pie", "banana split"]]
        aligned, _ = lists_alignment(lists, sim_fn, min_support_ratio=0.51)
        # every row contains both items, aligned per column
        cols = list(zip(*aligned))
        for col in cols:
            vals = {v for v in col if v is not None}
            assert len(vals) == 1

    def test_missing_element_pruned(self):
        lists = [["aaaa"], ["aaaa"], ["aaaa", "zzzz"]]
        aligned, _ = lists_alignment(lists, sim_fn, min_support_ratio=0.51)
        # "zzzz" supported by 1/3 lists -> pruned
        assert all(len(lst) == 1 for lst in aligned)
        assert aligned[2] == ["aaaa"]

    def test_empty_lists(self):
        aligned, orig = lists_alignment([[], []], sim_fn)
        assert aligned == [[], []]

    def test_known_reference_idx(self):
        lists = [["b", "a"], ["a", "b"]]
        aligned, _ = lists_alignment(lists, sim_fn, reference_list_idx=0)
        assert aligned[0] == ["b", "a"]
        assert aligned[1] == ["b", "a"]


class TestRecursiveAlignment:
    def test_scalar_passthrough(self):
        values = ["x", "y", None]
        aligned, km = recursive_list_alignments(values, "levenshtein", no_embed, None, 0.51)
        assert aligned == values
        assert km[""] == ["", "", None]

    def test_dict_of_lists(self):
        values = [
            {"items": ["alpha", "beta"]},
            {"items": ["beta", "alpha"]},
            {"items": ["alpha", "beta"]},
        ]
        aligned, km = recursive_list_alignments(values, "levenshtein", no_embed, None, 0.51)
        for d in aligned:
            assert len(d["items"]) == 2
        cols = list(zip(*[d["items"] for d in aligned]))
        for col in cols:
            assert len({v for v in col if v is not None}) == 1
        # key_mappings records aligned path -> original path per source
        assert "items.0" in km
        assert len(km["items.0"]) == 3

    def test_does_not_mutate_input(self):
        values = [{"a": ["x"]}, {"a": ["x"]}]
        snapshot = [{"a": ["x"]}, {"a": ["x"]}]
        recursive_list_alignments(values, "levenshtein", no_embed, None, 0.51)
        assert values == snapshot

    def test_all_none(self):
        aligned, km = recursive_list_alignments([None, None], "levenshtein", no_embed, None, 0.51, current_path="p")
        assert aligned == [None, None]
        assert km == {"p": ["p", "p"]}

    def test_mixed_types_passthrough(self):
        values = [{"a": 1}, ["list"], "str"]
        aligned, km = recursive_list_alignments(values, "levenshtein", no_embed, None, 0.51)
        assert aligned == values


def test_exists_nested_lists():
    assert exists_nested_lists([[1]])
    assert exists_nested_lists([{"a": [1]}])
    assert not exists_nested_lists(["x", 1, {"a": "b"}])
    assert not exists_nested_lists([])


def test_low_cutoff_bound_empty():
    assert low_cutoff_bound([]) == 0.0


def test_sort_by_original_majority_basic():
    originals = [["a", "b", "c"], ["a", "b", "c"]]
    # aligned columns scrambled: (c, a, b) by identity
    aligned = [
        [originals[0][2], originals[0][0], originals[0][1]],
        [originals[1][2], originals[1][0], originals[1][1]],
    ]
    sorted_lists, idx = sort_by_original_majority(aligned, originals)
    assert sorted_lists[0] == ["a", "b", "c"]
    assert idx[0] == [0, 1, 2]


class TestAlignmentInternals:
    """Targeted tests for the alignment internals' observable quirks
    (reference consensus_utils.py:185-333)."""

    def test_dynamic_threshold_single_list(self):
        from kllms_amd.consensus.alignment import SimilarityCache, _compute_dynamic_threshold

        cache = SimilarityCache(sim_fn, [["a"]])
        assert _compute_dynamic_threshold(cache) == 0.5  # <2 lists -> base

    def test_dynamic_threshold_tracks_best_matches(self):
        from kllms_amd.consensus.alignment import SimilarityCache, _compute_dynamic_threshold

        # identical elements across lists -> best-match sims all 1.0 ->
        # threshold = max(0.5, 0.95 * 1.0)
        lists = [["alpha", "beta"], ["alpha", "beta"], ["alpha", "beta"]]
        thr = _compute_dynamic_threshold(SimilarityCache(sim_fn, lists))
        assert thr == pytest.approx(0.95)

    def test_reference_build_groups_by_support(self):
        from kllms_amd.consensus.alignment import SimilarityCache, _build_reference_list

        lists = [["apple"], ["apple"], ["apple", "zebra"]]
        cache = SimilarityCache(sim_fn, lists)
        refs = _build_reference_list(cache, None, min_support_ratio=0.5, threshold=0.8)
        # "apple" group has support 3/3 >= 0.5; "zebra" 1/3 < 0.5 pruned
        assert len(refs) == 1
        li, pos = refs[0]
        assert lists[li][pos] == "apple"

    def test_reference_reelection_uses_index_medoid(self):
        from kllms_amd.consensus.alignment import SimilarityCache, _build_reference_list

        # three near-identical strings: the re-elected representative is the
        # medoid of the (list_idx, pos) INDEX TUPLES (reference quirk :308-318)
        lists = [["colour"], ["colour"], ["colour"]]
        cache = SimilarityCache(sim_fn, lists)
        refs = _build_reference_list(cache, None, min_support_ratio=0.5, threshold=0.8)
        assert len(refs) == 1
        assert refs[0] in [(0, 0), (1, 0), (2, 0)]

    def test_hungarian_respects_threshold(self):
        from kllms_amd.consensus.alignment import (
            SimilarityCache,
            _align_lists_to_reference_hungarian,
        )

        lists = [["aaaa"], ["zzzz"]]
        cache = SimilarityCache(sim_fn, lists)
        aligned = _align_lists_to_reference_hungarian(cache, [(0, 0)], threshold=0.9)
        assert aligned[0] == ["aaaa"]      # the reference element itself (sim 1.0)
        assert aligned[1] == [None]        # "zzzz" below threshold

    def test_prune_keeps_highest_when_all_below(self):
        from kllms_amd.consensus.alignment import _prune_low_support_elements

        # both columns below 0.9 -> threshold relaxes to the max support
        aligned = [["a", None], [None, None], ["a", None]]
        out = _prune_low_support_elements(aligned, 0.9)
        assert out == [["a"], [None], ["a"]]
