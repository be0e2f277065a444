"""Property-based fuzzing of the consensus engine (hypothesis).

Invariants checked on arbitrary JSON-like candidate sets:
- consensus_values and recursive_list_alignments never raise;
- likelihood leaves are finite floats in [0, 1];
- the likelihoods tree mirrors the consensus value tree's structure;
- consolidation preserves the §3.1 choice-layout invariants for any contents.
"""

import json
import math

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings, strategies as st  # noqa: E402

from kllms_amd.consensus import ConsensusSettings, consensus_values, recursive_list_alignments
from kllms_amd.consensus.consolidation import consolidate_chat_completions
from kllms_amd.types.openai_compat import ChatCompletion, ChatCompletionMessage, Choice


def no_embed(texts):
    # deterministic cheap vectors so the embeddings method also fuzzes safely
    return [[float(len(t) % 7), 1.0] for t in texts]


SETTINGS = ConsensusSettings(string_similarity_method="levenshtein")

scalars = st.one_of(
    st.none(),
    st.booleans(),
    st.integers(min_value=-10**6, max_value=10**6),
    st.floats(allow_nan=False, allow_infinity=False, width=32),
    st.text(max_size=12),
)
json_values = st.recursive(
    scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(min_size=1, max_size=6), children, max_size=4),
    ),
    max_leaves=12,
)


def check_likelihood_tree(conf):
    if isinstance(conf, dict):
        for v in conf.values():
            check_likelihood_tree(v)
    elif isinstance(conf, (list, tuple)):
        for v in conf:
            check_likelihood_tree(v)
    else:
        assert isinstance(conf, (int, float)), f"non-numeric confidence leaf: {conf!r}"
        assert math.isfinite(conf)
        assert -1e-9 <= conf <= 1.0 + 1e-9, f"confidence out of range: {conf}"


@settings(max_examples=150, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(json_values, min_size=1, max_size=5))
def test_consensus_values_total(values):
    val, conf = consensus_values(values, SETTINGS, no_embed)
    check_likelihood_tree(conf)


@settings(max_examples=80, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(json_values, min_size=2, max_size=4))
def test_recursive_alignment_total(values):
    aligned, km = recursive_list_alignments(values, "levenshtein", no_embed, None, 0.51)
    assert len(aligned) == len(values)
    for paths in km.values():
        assert len(paths) == len(values)


@settings(max_examples=60, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(st.one_of(st.text(max_size=30),
                          json_values.map(lambda v: json.dumps(v))),
                min_size=2, max_size=5))
def test_consolidation_invariants_total(contents):
    comp = ChatCompletion(
        id="fuzz", created=0, model="m",
        choices=[
            Choice(finish_reason="stop", index=i,
                   message=ChatCompletionMessage(role="assistant", content=c or " "))
            for i, c in enumerate(contents)
        ],
    )
    r = consolidate_chat_completions(comp, no_embed)
    assert len(r.choices) == len(contents) + 1
    assert [c.index for c in r.choices] == list(range(len(contents) + 1))
    assert isinstance(r.choices[0].message.content, str)
    if r.likelihoods is not None:
        check_likelihood_tree(r.likelihoods)


@settings(max_examples=120, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(json_values, min_size=1, max_size=5))
def test_key_aligner_total_and_shape_preserving(values):
    """The key-based aligner (L1b) must be total over arbitrary JSON and
    return one aligned value per input plus a path-mapping table whose
    per-path lists have one entry per source."""
    from kllms_amd.consensus.key_based_alignment import recursive_align

    aligned, km = recursive_align(values, "levenshtein", 0.5)
    assert len(aligned) == len(values)
    for path, sources in km.items():
        assert len(sources) == len(values), (path, sources)


@settings(max_examples=100, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(json_values, min_size=2, max_size=4))
def test_key_aligner_consensus_pipeline(values):
    """consensus over key-aligned values never raises and produces a
    range-valid likelihood tree (the 'key' aligner end-to-end)."""
    from kllms_amd.consensus.key_based_alignment import recursive_align

    aligned, _ = recursive_align(values, "levenshtein", 0.5)
    consensus, conf = consensus_values(list(aligned), SETTINGS, no_embed)
    check_likelihood_tree(conf)
