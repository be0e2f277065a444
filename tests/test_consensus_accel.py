"""M5 batched-similarity acceleration tests."""

import pytest

from kllms_amd.consensus.accel import collect_embeddable_strings, precompute_similarity_cache
from kllms_amd.consensus.similarity import _get_cached_similarity, cosine_similarity, string_similarity


LONG_A = "the quick brown fox jumps over the lazy dog and keeps on running far away"
LONG_B = "the quick brown fox jumps over the lazy dog and keeps running far away!!"
LONG_C = "a completely different sentence about databases, indexes and storage engines"


def test_collect_embeddable_strings():
    contents = [
        {"a": LONG_A, "b": "short", "nested": {"c": [LONG_B, 42, None]}},
        [LONG_A, LONG_C],
    ]
    strings = collect_embeddable_strings(contents)
    assert set(strings) == {LONG_A, LONG_B, LONG_C}  # unique, >50 chars only


def test_precompute_populates_cache(fake_embed):
    n = precompute_similarity_cache([{"x": LONG_A, "y": LONG_B, "z": LONG_C}], fake_embed)
    assert n == 3
    cached = _get_cached_similarity(LONG_A, LONG_B, "embeddings")
    assert cached is not None
    # must equal the per-pair path's value exactly
    direct = cosine_similarity(fake_embed([LONG_A])[0], fake_embed([LONG_B])[0])
    assert cached == pytest.approx(direct, abs=1e-12)


def test_alignment_uses_cache_without_embedding_calls(fake_embed):
    calls = {"n": 0}

    def counting_embed(texts):
        calls["n"] += 1
        return fake_embed(texts)

    precompute_similarity_cache([[LONG_A], [LONG_B]], counting_embed)
    assert calls["n"] == 1  # ONE batched call

    def exploding_embed(texts):
        raise AssertionError("per-pair embedding was called despite precompute")

    r = string_similarity(LONG_A, LONG_B, "embeddings", exploding_embed)
    assert 0 < r <= 1


def test_precompute_never_raises():
    def bad_embed(texts):
        raise RuntimeError("boom")

    assert precompute_similarity_cache([{"x": LONG_A, "y": LONG_B}], bad_embed) == 0
