"""L1b tests: key selection cascade, fuzzy fallback, key-based alignment
(behavioral contract of reference key_selection.py / fuzzy_key_selection.py /
key_based_alignment.py)."""

import pytest

from kllms_amd.consensus.fuzzy_key_selection import select_best_keys_with_fuzzy_fallback
from kllms_amd.consensus.key_based_alignment import recursive_align, _align_lists_by_key
from kllms_amd.consensus.key_selection import (
    CascadeConfig,
    discover_scalar_paths,
    evaluate_single_key,
    iter_records,
    select_best_keys,
    values_for_path,
)


def make_extractions():
    """Three extractions of the same product list, shuffled, with one noisy
    price value and a low-quality 'category' constant field."""
    base = [
        {"sku": "A1", "name": "Widget", "price": 9.99, "category": "tools"},
        {"sku": "B2", "name": "Gadget", "price": 19.99, "category": "tools"},
        {"sku": "C3", "name": "Gizmo", "price": 29.99, "category": "tools"},
    ]
    e1 = {"products": [dict(r) for r in base]}
    e2 = {"products": [dict(base[2]), dict(base[0]), dict(base[1])]}
    e3 = {"products": [dict(base[1]), dict(base[2]), dict(base[0])]}
    e3["products"][0]["price"] = 19.992  # tiny numeric noise
    return [e1, e2, e3]


class TestDiscovery:
    def test_iter_records_named_key(self):
        e = {"items": [{"a": 1}, {"a": 2}], "other": 3}
        assert len(iter_records(e, list_key="items")) == 2

    def test_iter_records_autodetect(self):
        e = {"whatever": [{"a": 1}], "scalar": "x"}
        assert len(iter_records(e)) == 1

    def test_discover_scalar_paths(self):
        ex = make_extractions()
        paths = discover_scalar_paths(ex)
        assert "sku" in paths and "name" in paths and "price" in paths
        # nested
        e = {"products": [{"meta": {"id": 7}, "tags": ["x"]}]}
        paths = discover_scalar_paths([e])
        assert "meta.id" in paths
        assert all("tags" not in p for p in paths)  # list-valued paths excluded

    def test_values_for_path_normalized(self):
        e = {"products": [{"name": "  Foo   Bar "}]}
        assert values_for_path(e, "name") == ["foo bar"]


class TestSelection:
    def test_stable_unique_key_wins(self):
        ex = make_extractions()
        # the aligner path gates constant keys via min_uniqueness=0.5
        # (reference key_based_alignment.py:380-382); with the default config
        # the smallest-union stage would pick the constant 'category'.
        result = select_best_keys(ex, cascade_cfg=CascadeConfig(min_uniqueness=0.5))
        assert result.best_single.path[0] in ("sku", "name")
        assert result.best_single.jaccard_min == 1.0
        assert result.best_single.uniqueness_min == 1.0

    def test_metrics_shape(self):
        ex = make_extractions()
        m = evaluate_single_key(ex, "sku")
        assert m.I_E == 3       # all 3 sku values present in every extraction
        assert m.coverage_min == 1.0
        assert m.union_size == 3

    def test_no_candidates_raises(self):
        with pytest.raises(ValueError):
            select_best_keys([{"products": []}])

    def test_fuzzy_beats_noisy_numeric(self):
        # price differs only in the 3rd decimal in one extraction: fuzzy
        # canonicalization (round 2dp) makes it stable
        ex = make_extractions()
        m_std = evaluate_single_key(ex, "price")
        assert m_std.jaccard_min < 1.0
        comp = select_best_keys_with_fuzzy_fallback(ex)
        assert comp.chosen in ("normal", "fuzzy")
        # the overall winner is sku/name either way (already perfectly stable)


class TestKeyAlignment:
    def test_align_lists_by_key(self):
        l1 = [{"sku": "A", "v": 1}, {"sku": "B", "v": 2}]
        l2 = [{"sku": "B", "v": 20}, {"sku": "A", "v": 10}]
        rows, idx = _align_lists_by_key([l1, l2], ("sku",))
        assert len(rows) == 2
        for row in rows:
            skus = {r["sku"] for r in row if r}
            assert len(skus) == 1
        # original indices map back
        assert idx[0] == [0, 1] or idx[0] == [1, 0] or all(i is not None for i in idx[0])

    def test_recursive_align_reorders_records(self):
        ex = make_extractions()
        values = list(ex)
        aligned, mappings = recursive_align(values, "levenshtein", min_support_ratio=0.5)
        assert len(aligned) == 3
        # every aligned source now lists products in the SAME sku order
        orders = [[p["sku"] if p else None for p in a["products"]] for a in aligned]
        assert orders[0] == orders[1] == orders[2]
        assert mappings  # path traceability present

    def test_scalar_zip_fallback(self):
        values = [{"tags": ["a", "b"]}, {"tags": ["a", "b", "c"]}]
        aligned, _ = recursive_align(values, "levenshtein")
        assert len(aligned[0]["tags"]) == 3  # zipped to max length

    def test_all_none(self):
        aligned, km = recursive_align([None, None], "levenshtein", current_path="p")
        assert aligned == [None, None]
        assert km == {"p": ["p", "p"]}


class TestAlignerSelection:
    def test_consolidation_with_key_aligner(self):
        import json

        from kllms_amd.consensus import consolidate_chat_completions
        from kllms_amd.types.openai_compat import ChatCompletion, ChatCompletionMessage, Choice

        recs1 = {"products": [{"sku": "A", "price": 1.0}, {"sku": "B", "price": 2.0}]}
        recs2 = {"products": [{"sku": "B", "price": 2.0}, {"sku": "A", "price": 1.0}]}
        comp = ChatCompletion(
            id="x", created=0, model="m",
            choices=[
                Choice(finish_reason="stop", index=i,
                       message=ChatCompletionMessage(role="assistant", content=json.dumps(c)))
                for i, c in enumerate([recs1, recs2, recs1])
            ],
        )

        def no_embed(texts):
            raise AssertionError

        r = consolidate_chat_completions(comp, no_embed, aligner="key")
        out = json.loads(r.choices[0].message.content)
        assert {p["sku"] for p in out["products"]} == {"A", "B"}
        assert r.likelihoods is not None
