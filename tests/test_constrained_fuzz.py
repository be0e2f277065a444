"""Property-based tests for the constrained-decoding FSM.

For ANY schema the generator produces and ANY instance conforming to it,
the compiled byte DFA must accept the canonical compact serialization —
and in whitespace mode, any re-indented serialization. This is the
soundness direction (valid JSON is never rejected); the completeness
direction (the FSM never lets an invalid byte through) is covered by the
deterministic tests in test_constrained.py.
"""

import json

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings, strategies as st  # noqa: E402

from kllms_amd.engine.constrained import compile_dfa, schema_to_ir  # noqa: E402


# ---------------------------------------------------------------------------
# (schema, conforming instance) strategy
# ---------------------------------------------------------------------------

# printable-ASCII strings excluding chars the FSM expresses via escapes
_safe_text = st.text(
    alphabet=st.characters(min_codepoint=0x20, max_codepoint=0x7E, exclude_characters='"\\'),
    max_size=8,
)

_keys = st.sampled_from(["a", "b", "c", "name", "value", "k1", "k2"])


def _leaf():
    return st.one_of(
        st.tuples(st.just({"type": "boolean"}), st.booleans()),
        st.tuples(st.just({"type": "null"}), st.none()),
        st.tuples(st.just({"type": "integer"}), st.integers(min_value=-10**9, max_value=10**9)),
        st.tuples(st.just({"type": "string"}), _safe_text),
        _safe_text.flatmap(
            lambda v: st.just(({"enum": [v, v + "x"]}, v))
        ),
        st.tuples(
            st.just({"type": "number"}),
            st.floats(allow_nan=False, allow_infinity=False, width=32),
        ),
    )


def _extend(children):
    def arr(pair_strategy):
        return pair_strategy.flatmap(
            lambda pair: st.lists(st.just(pair[1]), min_size=0, max_size=3).map(
                lambda items: ({"type": "array", "items": pair[0]}, items)
            )
        )

    def obj(pair_strategy):
        return st.lists(
            st.tuples(_keys, pair_strategy), min_size=1, max_size=3,
            unique_by=lambda kv: kv[0],
        ).map(
            lambda kvs: (
                {
                    "type": "object",
                    "properties": {k: s for k, (s, _) in kvs},
                    "required": [k for k, _ in kvs],
                },
                {k: v for k, (_, v) in kvs},
            )
        )

    return st.one_of(arr(children), obj(children))


schema_and_instance = st.recursive(_leaf(), _extend, max_leaves=6)


def _accepts(schema: dict, text: str, ws: bool) -> bool:
    ir = schema_to_ir(schema, schema.get("$defs", {}), ws=ws)
    trans, accepting, s = compile_dfa(ir)
    for b in text.encode():
        nxt = int(trans[s, b])
        if nxt == 0xFFFF:
            return False
        s = nxt
    return bool(accepting[s])


def _dumps_compact(value) -> str:
    return json.dumps(value, separators=(",", ":"))


@settings(max_examples=120, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(schema_and_instance)
def test_compact_serialization_accepted(pair):
    schema, value = pair
    text = _dumps_compact(value)
    # the FSM's number grammar is JSON's; python float repr is JSON-compatible
    assert _accepts(schema, text, ws=False), f"{schema} rejected {text!r}"


@settings(max_examples=60, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(schema_and_instance, st.sampled_from([1, 2, 4]))
def test_ws_mode_accepts_pretty(pair, indent):
    schema, value = pair
    text = json.dumps(value, indent=indent)
    assert _accepts(schema, text, ws=True), f"{schema} (ws) rejected {text!r}"


@settings(max_examples=60, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(schema_and_instance)
def test_mutated_byte_never_reaches_accept_silently(pair):
    """Flipping a structural byte must either be rejected mid-walk or leave
    the DFA in a non-accepting state (never 'accepted garbage')."""
    schema, value = pair
    text = _dumps_compact(value)
    if len(text) < 2:
        return
    # corrupt one byte to an impossible structural character
    mid = len(text) // 2
    corrupted = text[:mid] + "\x01" + text[mid + 1:]
    assert not _accepts(schema, corrupted, ws=False)


@settings(max_examples=300, deadline=None, suppress_health_check=[HealthCheck.too_slow])
@given(st.integers(-10**7, 10**7), st.integers(0, 10**7), st.integers(-10**7 - 1000, 10**7 + 1000))
def test_integer_range_is_exact(lo, span, probe):
    """Two-sided integer bounds compile to an EXACT range DFA: str(x) is
    accepted iff minimum <= x <= maximum (no digit-count slack)."""
    hi = lo + span
    sch = {"type": "integer", "minimum": lo, "maximum": hi}
    for x in {probe, lo, hi, lo - 1, hi + 1}:
        assert _accepts(sch, str(x), ws=False) == (lo <= x <= hi), (lo, hi, x)


def test_integer_range_edges():
    sch = {"type": "integer", "minimum": 0, "maximum": 120}
    assert _accepts(sch, "120", ws=False) and _accepts(sch, "0", ws=False)
    for bad in ("121", "323", "999", "-1", "-0", "05", "00"):
        assert not _accepts(sch, bad, ws=False), bad
    # exclusive bounds (pydantic gt/lt)
    sch = {"type": "integer", "exclusiveMinimum": 0, "exclusiveMaximum": 10}
    assert _accepts(sch, "1", ws=False) and _accepts(sch, "9", ws=False)
    assert not _accepts(sch, "0", ws=False) and not _accepts(sch, "10", ws=False)
    # negative-only range
    sch = {"type": "integer", "minimum": -250, "maximum": -3}
    assert _accepts(sch, "-3", ws=False) and _accepts(sch, "-250", ws=False)
    for bad in ("-2", "-251", "0", "3"):
        assert not _accepts(sch, bad, ws=False), bad


def test_number_bounds_cap_magnitude():
    """number fields with bounds cap the integer-part digits and drop the
    exponent form (magnitude guard; exact float ranges aren't a DFA)."""
    sch = {"type": "number", "minimum": 0, "maximum": 100}
    for ok in ("0", "100", "99.75", "3.14159", "0.001"):
        assert _accepts(sch, ok, ws=False), ok
    for bad in ("12345", "1234.5", "-1.0", "1e9", "2E2"):
        assert not _accepts(sch, bad, ws=False), bad
    # unbounded numbers keep the full JSON grammar incl. exponent
    sch = {"type": "number"}
    for ok in ("-12345.67", "1e9", "6.02E23"):
        assert _accepts(sch, ok, ws=False), ok


def test_format_strings_accept_valid_reject_garbage():
    """uuid/date/time/date-time formats compile to fixed-shape DFAs so
    pydantic UUID/date/datetime fields validate from the sampled surface."""
    cases = {
        "uuid": (["01234567-89ab-CDEF-0123-456789abcdef"],
                 ["0123456789ab-CDEF-0123-456789abcdef", "zz234567-89ab-cdef-0123-456789abcdef",
                  "01234567-89ab-cdef-0123-456789abcde"]),
        "date": (["2024-02-29", "0001-01-01", "9999-12-31"],
                 ["2024-13-01", "2024-00-10", "2024-01-32", "24-01-01", "2024/01/01"]),
        "time": (["00:00:00", "23:59:59.123456"],
                 ["24:00:00", "12:60:00", "12:00:61", "1:00:00"]),
        "date-time": (["2024-06-15T08:30:00Z", "2024-06-15T08:30:00.5+05:30",
                       "2024-06-15T23:59:59"],
                      ["2024-06-15 08:30:00", "2024-06-15T25:00:00Z"]),
    }
    for fmt, (goods, bads) in cases.items():
        sch = {"type": "string", "format": fmt}
        for g in goods:
            assert _accepts(sch, json.dumps(g), ws=False), (fmt, g)
        for b in bads:
            assert not _accepts(sch, json.dumps(b), ws=False), (fmt, b)
    # unknown format falls back to the plain string grammar
    sch = {"type": "string", "format": "hostname"}
    assert _accepts(sch, json.dumps("anything at all"), ws=False)


def test_pydantic_uuid_datetime_fields_parse_end_to_end():
    import datetime
    import uuid as _uuid

    from pydantic import BaseModel

    from kllms_amd import KLLMs

    class Event(BaseModel):
        id: _uuid.UUID
        day: datetime.date
        ts: datetime.datetime

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=512,
              use_hip_graphs=False, seed=0)
    r = c.chat.completions.parse(
        messages=[{"role": "user", "content": "emit an event"}],
        model="tiny-llama", response_format=Event, n=6,
        max_tokens=128, temperature=1.0, seed=4)
    finished = [ch for ch in r.choices[1:] if ch.finish_reason == "stop"]
    assert finished, "no stream finished within budget"
    parsed = [ch for ch in finished if ch.message.parsed is not None]
    # calendar-impossible draws (Feb 31) are the only allowed misses
    assert len(parsed) >= max(1, len(finished) - 1), \
        [ch.message.content for ch in finished]


def test_additional_properties_value_type_enforced():
    """pydantic Dict[str, T]: free keys but TYPED values by construction."""
    sch = {"type": "object", "additionalProperties": {"type": "integer",
                                                      "minimum": 0, "maximum": 99}}
    for ok in ('{}', '{"a":1}', '{"a":1,"bc":42}'):
        assert _accepts(sch, ok, ws=False), ok
    for bad in ('{"a":"x"}', '{"a":100}', '{"a":[1]}', '{"a":1,}'):
        assert not _accepts(sch, bad, ws=False), bad

    # end to end: Dict[str, int] parses from the sampled surface
    from typing import Dict as _Dict

    from pydantic import BaseModel

    from kllms_amd import KLLMs

    class Scores(BaseModel):
        scores: _Dict[str, int]

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=512,
              use_hip_graphs=False, seed=0)
    r = c.chat.completions.parse(
        messages=[{"role": "user", "content": "score things"}],
        model="tiny-llama", response_format=Scores, n=6,
        max_tokens=64, temperature=1.0, seed=9)
    finished = [ch for ch in r.choices[1:] if ch.finish_reason == "stop"]
    assert finished
    assert all(ch.message.parsed is not None for ch in finished), \
        [ch.message.content for ch in finished]


def test_prefix_items_positional_types_enforced():
    """pydantic Tuple[int, str]: each position's subschema is enforced."""
    from typing import Tuple as _Tuple

    from pydantic import BaseModel

    class M(BaseModel):
        pair: _Tuple[int, str]

    sch = M.model_json_schema()
    assert _accepts(sch, '{"pair":[3,"x"]}', ws=False)
    for bad in ('{"pair":["x",3]}', '{"pair":[3,"x",1]}', '{"pair":[3]}', '{"pair":[]}'):
        assert not _accepts(sch, bad, ws=False), bad
    # variadic tail: Tuple[int, ...]-style (prefix + typed items)
    sch2 = {"type": "array", "minItems": 1, "maxItems": 3,
            "prefixItems": [{"type": "string"}], "items": {"type": "integer"}}
    for ok in ('["a"]', '["a",1]', '["a",1,2]'):
        assert _accepts(sch2, ok, ws=False), ok
    for bad in ('["a",1,2,3]', '["a","b"]', '[1]'):
        assert not _accepts(sch2, bad, ws=False), bad


def test_kitchen_sink_model_streams_all_validate():
    """Every field type the constrained compiler supports, in ONE model:
    all stop-finished streams must pydantic-validate (calendar-impossible
    dates are the single tolerated miss)."""
    import datetime
    import enum
    import uuid as _uuid
    from typing import Dict as _Dict
    from typing import List as _List
    from typing import Literal as _Literal
    from typing import Optional as _Optional
    from typing import Tuple as _Tuple

    from pydantic import BaseModel, Field

    from kllms_amd import KLLMs

    class Color(str, enum.Enum):
        red = "red"
        blue = "blue"

    class Inner(BaseModel):
        name: str = Field(max_length=6)
        score: int = Field(ge=-5, le=120)

    class Sink(BaseModel):
        id: _uuid.UUID
        day: datetime.date
        kind: _Literal["a", "b"]
        color: Color
        ratio: float = Field(ge=0, le=10)
        flag: bool
        note: _Optional[str] = Field(default=None, max_length=5)
        pair: _Tuple[int, str]
        items: _List[Inner] = Field(max_length=2)
        counts: _Dict[str, int]

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=1024,
              use_hip_graphs=False, seed=0, max_seq_len=512)
    r = c.chat.completions.parse(
        messages=[{"role": "user", "content": "emit"}],
        model="tiny-llama", response_format=Sink, n=8,
        max_tokens=400, temperature=1.0, seed=13)
    finished = [ch for ch in r.choices[1:] if ch.finish_reason == "stop"]
    assert finished, "no stream finished"
    misses = [ch.message.content for ch in finished if ch.message.parsed is None]
    # tolerated: impossible calendar dates (e.g. Feb 31) and floats beyond
    # le=10 (digit-cap only); everything else must validate
    for m in misses:
        import json as _json
        doc = _json.loads(m)
        day_ok = True
        try:
            datetime.date.fromisoformat(doc["day"])
        except ValueError:
            day_ok = False
        assert (not day_ok) or not (0 <= doc["ratio"] <= 10), m


def test_pattern_subset_matches_python_re():
    """JSON-Schema `pattern` (pydantic Field(pattern=...)): the compiled
    DFA agrees with python re.fullmatch on the supported subset; anything
    unsupported falls back to the plain string grammar (never over-
    constrains into an unsatisfiable schema)."""
    import random
    import re
    import string as strmod

    cases = [
        (r"^[A-Z]{2}-\d{4}$", ["AB-1234"], ["ab-1234", "AB-123", "AB_1234"]),
        (r"^\d+(\.\d+)?$", ["1", "3.14"], ["", ".5", "1."]),
        (r"cat|dog|bird", ["cat", "dog"], ["cats", "catdog"]),
        (r"a{2,4}b?", ["aa", "aaaab"], ["a", "aaaaa", "ab"]),
        (r"[^0-9]+", ["abc"], ["a1", ""]),
        (r"(?:ab)+c", ["abc", "ababc"], ["ac", "abab"]),
    ]
    rng = random.Random(0)
    alpha = strmod.ascii_letters + strmod.digits + "-_. "
    for pat, goods, bads in cases:
        sch = {"type": "string", "pattern": pat}
        for g in goods:
            assert _accepts(sch, f'"{g}"', ws=False), (pat, g)
        for b in bads:
            assert not _accepts(sch, f'"{b}"', ws=False), (pat, b)
        rx = re.compile(pat.lstrip("^").rstrip("$"))
        for _ in range(200):
            s = "".join(rng.choice(alpha) for _ in range(rng.randint(0, 8)))
            assert _accepts(sch, f'"{s}"', ws=False) == (rx.fullmatch(s) is not None), (pat, s)
    for pat in (r"(?=x)a", r"\bword\b", r"(a)\1", "héllo"):
        assert _accepts({"type": "string", "pattern": pat}, '"anything"', ws=False)


def test_pattern_field_end_to_end():
    from pydantic import BaseModel, Field

    from kllms_amd import KLLMs

    class Order(BaseModel):
        code: str = Field(pattern=r"^[A-Z]{2}-\d{4}$")

    c = KLLMs(model="tiny-llama", device="cpu", max_kv_blocks=512,
              use_hip_graphs=False, seed=0)
    r = c.chat.completions.parse(
        messages=[{"role": "user", "content": "make an order code"}],
        model="tiny-llama", response_format=Order, n=6,
        max_tokens=48, temperature=1.0, seed=21)
    finished = [ch for ch in r.choices[1:] if ch.finish_reason == "stop"]
    assert finished
    assert all(ch.message.parsed is not None for ch in finished), \
        [ch.message.content for ch in finished]


def test_dict_min_max_properties():
    sch = {"type": "object", "additionalProperties": {"type": "integer"},
           "minProperties": 1, "maxProperties": 2}
    for ok in ('{"a":1}', '{"a":1,"b":2}'):
        assert _accepts(sch, ok, ws=False), ok
    for bad in ('{}', '{"a":1,"b":2,"c":3}'):
        assert not _accepts(sch, bad, ws=False), bad


def test_single_element_allof_unwrapped():
    """pydantic-v1-style {allOf: [{$ref}], description} wrappers compile to
    the wrapped schema; multi-element allOf rejects loudly."""
    import pytest as _pytest

    from kllms_amd.engine.constrained import SchemaCompileError, schema_to_ir

    defs = {"Inner": {"type": "object", "properties": {"x": {"type": "integer"}},
                      "required": ["x"]}}
    sch = {"type": "object",
           "properties": {"inner": {"allOf": [{"$ref": "#/$defs/Inner"}],
                                    "description": "d"}},
           "required": ["inner"], "$defs": defs}
    assert _accepts(sch, '{"inner":{"x":3}}', ws=False)
    assert not _accepts(sch, '{"inner":{"x":"s"}}', ws=False)
    with _pytest.raises(SchemaCompileError):
        schema_to_ir({"allOf": [{"type": "integer"}, {"minimum": 3}]}, {})
