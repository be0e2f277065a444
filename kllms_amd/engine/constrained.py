"""JSON-schema constrained decoding: schema -> byte DFA -> per-state token masks.

Native replacement for OpenAI's server-side structured-output decoding
(reference boundary: k_llms/resources/completions/completions.py:134 `beta.
chat.completions.parse`). Pipeline:

1. Pydantic model -> JSON schema (done by the caller).
2. Schema -> regex-like IR -> Thompson NFA -> subset-constructed DFA over the
   byte alphabet (compact JSON only: no inter-token whitespace, fixed
   property order — standard guided-generation canonicalization).
3. DFA x tokenizer vocab -> dense token transition table next_state[S, V]
   (vectorized numpy walk over token byte columns) and per-state ALLOWED
   bitmasks (uint32 words), which the fused sampling kernel consumes as an
   additive -inf vocab mask.

Per decode step the engine does an O(1) state lookup for the mask and an O(1)
advance — all the heavy work is one-time per (schema, vocab) and cached.
"""

from __future__ import annotations

import json
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

# ---------------------------------------------------------------------------
# Regex IR
# ---------------------------------------------------------------------------

class _Node:
    pass


class Lit(_Node):
    def __init__(self, s: bytes):
        self.s = s


class Cls(_Node):
    def __init__(self, bytes_set: set):
        self.set = bytes_set


class Seq(_Node):
    def __init__(self, parts: List[_Node]):
        self.parts = parts


class Alt(_Node):
    def __init__(self, parts: List[_Node]):
        self.parts = parts


class Star(_Node):
    def __init__(self, inner: _Node):
        self.inner = inner


class Opt(_Node):
    def __init__(self, inner: _Node):
        self.inner = inner


_DIGITS = set(b"0123456789")
_HEX = set(b"0123456789abcdefABCDEF")
_WS_SET = set(b" \t\n\r")
# JSON string body chars: printable ASCII except '"' and '\\'. Non-ASCII
# content is expressible via \uXXXX escapes, which keeps every constrained
# output valid UTF-8 regardless of how the tokenizer splits bytes.
_STR_CHAR = {b for b in range(0x20, 0x7F) if b not in (0x22, 0x5C)}


def _lead_ws(node: _Node, ws: bool) -> _Node:
    """Optional JSON whitespace before a token (whitespace-tolerant mode)."""
    return Seq([Star(Cls(_WS_SET)), node]) if ws else node


def _bounded(inner_factory, min_n: int, max_n: Optional[int]) -> _Node:
    """min_n required occurrences then (max_n - min_n) optional ones, or a
    Star tail when max_n is None."""
    parts: List[_Node] = [inner_factory() for _ in range(min_n)]
    if max_n is None:
        parts.append(Star(inner_factory()))
    else:
        for _ in range(max_n - min_n):
            parts.append(Opt(inner_factory()))
    return Seq(parts)


def _json_string_ir(min_len: int = 0, max_len: Optional[int] = None) -> _Node:
    def char() -> _Node:
        return Alt([
            Cls(_STR_CHAR),
            Seq([
                Lit(b"\\"),
                Alt([
                    Cls(set(b'"\\/bfnrt')),
                    Seq([Lit(b"u"), Cls(_HEX), Cls(_HEX), Cls(_HEX), Cls(_HEX)]),
                ]),
            ]),
        ])

    return Seq([Lit(b'"'), _bounded(char, min_len, max_len), Lit(b'"')])


_RE_CLASSES = {
    ord("d"): set(b"0123456789"),
    ord("w"): set(b"abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789_"),
    ord("s"): set(b" \t"),  # JSON strings can't hold raw \n\r; escaped ws only
}
# the printable-ASCII universe of the constrained string alphabet (no '"'
# or '\\' — those need JSON escapes, which patterns rarely intend)
_RE_ANY = {c for c in range(0x20, 0x7F)} - {0x22, 0x5C}


class _PatternUnsupported(Exception):
    pass


def _compile_pattern_ir(pat: str) -> _Node:
    """A safe regex SUBSET -> IR, for JSON-Schema `pattern` (pydantic
    Field(pattern=...)): literals, escapes (\\d \\w \\s \\. etc.),
    [...] classes with ranges and negation, groups, alternation `|`,
    quantifiers `* + ? {m} {m,} {m,n}`, and `^`/`$` anchors (stripped —
    compilation is full-match, which is strictly stronger than the spec's
    re.search, so every emitted string still validates). Unsupported
    constructs (backrefs, lookaround, non-ASCII, nested quantifier edge
    cases) raise _PatternUnsupported and the caller falls back to the
    plain string grammar."""
    b = pat.encode("ascii", errors="strict").lstrip(b"^")
    if b.endswith(b"$") and not b.endswith(b"\\$"):
        b = b[:-1]
    pos = 0

    def peek():
        return b[pos] if pos < len(b) else None

    def parse_class() -> set:
        nonlocal pos
        assert b[pos] == 0x5B  # [
        pos += 1
        neg = peek() == 0x5E  # ^
        if neg:
            pos += 1
        chars: set = set()
        first = True
        while True:
            c = peek()
            if c is None:
                raise _PatternUnsupported("unterminated class")
            if c == 0x5D and not first:  # ]
                pos += 1
                break
            first = False
            if c == 0x5C:  # escape
                pos += 1
                e = peek()
                if e is None:
                    raise _PatternUnsupported("trailing backslash")
                pos += 1
                low = e | 0x20
                if low in _RE_CLASSES:
                    cls = _RE_CLASSES[low]
                    chars |= (_RE_ANY - cls) if (e < 0x61) else cls
                    continue
                chars.add(e)
                continue
            pos += 1
            if peek() == 0x2D and pos + 1 < len(b) and b[pos + 1] != 0x5D:  # range
                pos += 1
                hi = b[pos]
                pos += 1
                chars |= set(range(c, hi + 1))
            else:
                chars.add(c)
        out = (_RE_ANY - chars) if neg else (chars & _RE_ANY)
        if not out:
            raise _PatternUnsupported("empty class")
        return out

    def parse_atom() -> _Node:
        nonlocal pos
        c = peek()
        if c is None:
            raise _PatternUnsupported("dangling quantifier")
        if c == 0x28:  # (
            pos += 1
            if peek() == 0x3F:  # (?: or lookaround
                if pos + 1 < len(b) and b[pos + 1] == 0x3A:
                    pos += 2
                else:
                    raise _PatternUnsupported("lookaround/named group")
            node = parse_alt()
            if peek() != 0x29:
                raise _PatternUnsupported("unbalanced group")
            pos += 1
            return node
        if c == 0x5B:  # [
            return Cls(parse_class())
        if c == 0x2E:  # .
            pos += 1
            return Cls(set(_RE_ANY))
        if c == 0x5C:  # escape
            pos += 1
            e = peek()
            if e is None:
                raise _PatternUnsupported("trailing backslash")
            pos += 1
            low = e | 0x20
            if low in _RE_CLASSES:
                cls = _RE_CLASSES[low]
                return Cls(set(cls) if e >= 0x61 else set(_RE_ANY - cls))
            if e in (0x62, 0x42) or 0x30 <= e <= 0x39:  # \b, backrefs
                raise _PatternUnsupported("anchor/backref escape")
            return Lit(bytes([e]))
        if c in (0x2A, 0x2B, 0x3F, 0x7B, 0x29, 0x7C):  # * + ? { ) |
            raise _PatternUnsupported("misplaced metachar")
        if c in (0x22, 0x5C) or not 0x20 <= c < 0x7F:
            raise _PatternUnsupported("char needs JSON escaping")
        pos += 1
        return Lit(bytes([c]))

    def parse_quant(node: _Node) -> _Node:
        nonlocal pos
        c = peek()
        if c == 0x2A:  # *
            pos += 1
            return Star(node)
        if c == 0x2B:  # +
            pos += 1
            return Seq([node, Star(node)])
        if c == 0x3F:  # ?
            pos += 1
            return Opt(node)
        if c == 0x7B:  # {m,n}
            end = b.find(b"}", pos)
            if end < 0:
                raise _PatternUnsupported("unterminated {}")
            body = b[pos + 1:end].decode()
            pos = end + 1
            try:
                if "," in body:
                    lo_s, hi_s = body.split(",", 1)
                    m = int(lo_s)
                    n = int(hi_s) if hi_s else None
                else:
                    m = n = int(body)
            except ValueError:
                raise _PatternUnsupported("bad {} bound")
            if m > 64 or (n is not None and n > 64):
                raise _PatternUnsupported("repetition too large")
            head = [node] * m
            if n is None:
                return Seq(head + [Star(node)])
            return Seq(head + [Opt(node) for _ in range(n - m)])
        return node

    def parse_seq() -> _Node:
        parts: List[_Node] = []
        while peek() is not None and peek() not in (0x7C, 0x29):
            parts.append(parse_quant(parse_atom()))
        return Seq(parts) if parts else Lit(b"")

    def parse_alt() -> _Node:
        nonlocal pos
        opts = [parse_seq()]
        while peek() == 0x7C:
            pos += 1
            opts.append(parse_seq())
        return opts[0] if len(opts) == 1 else Alt(opts)

    node = parse_alt()
    if pos != len(b):
        raise _PatternUnsupported("trailing garbage")
    return node


def _format_string_ir(fmt: str) -> Optional[_Node]:
    """Fixed-shape DFAs for the JSON-Schema string formats pydantic emits
    (uuid.UUID, datetime.date/time/datetime fields). Without these a
    formatted field samples an arbitrary string and pydantic re-validation
    rejects essentially every stream; with them the sampled surface parses.
    Date shapes are calendar-plausible (day 01-31 regardless of month —
    pydantic still rejects e.g. Feb 31, a rare draw), times are 24h."""
    H = _HEX
    hexs = lambda n: [Cls(H) for _ in range(n)]
    d = lambda: Cls(_DIGITS)
    if fmt == "uuid":
        return Seq(hexs(8) + [Lit(b"-")] + hexs(4) + [Lit(b"-")] + hexs(4)
                   + [Lit(b"-")] + hexs(4) + [Lit(b"-")] + hexs(12))
    month = Alt([Seq([Lit(b"0"), Cls(set(b"123456789"))]), Seq([Lit(b"1"), Cls(set(b"012"))])])
    day = Alt([Seq([Lit(b"0"), Cls(set(b"123456789"))]),
               Seq([Cls(set(b"12")), d()]), Seq([Lit(b"3"), Cls(set(b"01"))])])
    date = Seq([d(), d(), d(), d(), Lit(b"-"), month, Lit(b"-"), day])
    hh = Alt([Seq([Cls(set(b"01")), d()]), Seq([Lit(b"2"), Cls(set(b"0123"))])])
    m60 = Seq([Cls(set(b"012345")), d()])
    time = Seq([hh, Lit(b":"), m60, Lit(b":"), m60,
                Opt(Seq([Lit(b"."), Cls(_DIGITS), Star(Cls(_DIGITS))]))])
    if fmt == "date":
        return date
    if fmt == "time":
        return time
    if fmt == "date-time":
        tz = Alt([Lit(b"Z"), Seq([Cls(set(b"+-")), hh, Lit(b":"), m60])])
        return Seq([date, Lit(b"T"), time, Opt(tz)])
    return None


def _digits_any(n: int) -> _Node:
    return Seq([Cls(_DIGITS) for _ in range(n)]) if n else Lit(b"")


def _ge_str(s: bytes) -> _Node:
    """len(s)-digit strings (leading zeros allowed) numerically >= s."""
    if not s:
        return Lit(b"")
    d = s[0]
    opts: List[_Node] = [Seq([Lit(bytes([d])), _ge_str(s[1:])])]
    hi = {c for c in _DIGITS if c > d}
    if hi:
        opts.append(Seq([Cls(hi), _digits_any(len(s) - 1)]))
    return Alt(opts)


def _le_str(s: bytes, min_first: int = 0x30) -> _Node:
    """len(s)-digit strings numerically <= s, first digit >= min_first
    (min_first=ord('1') forbids a leading zero)."""
    if not s:
        return Lit(b"")
    d = s[0]
    opts: List[_Node] = []
    if d >= min_first:
        opts.append(Seq([Lit(bytes([d])), _le_str(s[1:])]))
    lo = {c for c in _DIGITS if min_first <= c < d}
    if lo:
        opts.append(Seq([Cls(lo), _digits_any(len(s) - 1)]))
    return Alt(opts) if opts else Alt([])


def _between_str(lo_s: bytes, hi_s: bytes) -> _Node:
    """Equal-length digit strings numerically in [lo_s, hi_s]."""
    if not lo_s:
        return Lit(b"")
    l0, h0 = lo_s[0], hi_s[0]
    if l0 == h0:
        return Seq([Lit(bytes([l0])), _between_str(lo_s[1:], hi_s[1:])])
    opts: List[_Node] = [
        Seq([Lit(bytes([l0])), _ge_str(lo_s[1:])]),
        Seq([Lit(bytes([h0])), _le_str(hi_s[1:])]),
    ]
    mid = {c for c in _DIGITS if l0 < c < h0}
    if mid:
        opts.append(Seq([Cls(mid), _digits_any(len(lo_s) - 1)]))
    return Alt(opts)


def _nonneg_range_ir(a: int, b: int) -> _Node:
    """Decimal integers in [a, b] (0 <= a <= b), no leading zeros."""
    sa, sb = str(a).encode(), str(b).encode()
    opts: List[_Node] = []
    for L in range(len(sa), len(sb) + 1):
        if L == len(sa) == len(sb):
            opts.append(_between_str(sa, sb))
        elif L == len(sa):
            opts.append(_ge_str(sa))  # [a, 10^L - 1]; sa has no leading zero
        elif L == len(sb):
            opts.append(_le_str(sb, min_first=0x31))  # [10^(L-1), b]
        else:
            opts.append(Seq([Cls(set(b"123456789")), _digits_any(L - 1)]))
    return Alt(opts)


def _int_range_ir(lo: int, hi: int) -> _Node:
    """EXACT JSON-integer range [lo, hi]: the DFA admits precisely the
    integers in range (tight-prefix construction, O(digits^2) IR) — a
    schema's minimum/maximum are enforced by construction, not by the
    digit-count approximation."""
    assert lo <= hi
    if hi < 0:
        return Seq([Lit(b"-"), _nonneg_range_ir(-hi, -lo)])
    opts: List[_Node] = []
    if lo < 0:
        opts.append(Seq([Lit(b"-"), _nonneg_range_ir(1, -lo)]))  # JSON has no -0
        lo = 0
    opts.append(_nonneg_range_ir(lo, hi))
    return Alt(opts)


def _integer_ir(max_digits: Optional[int] = None, allow_negative: bool = True) -> _Node:
    """JSON integer; with one-sided bounds, the digit COUNT is capped from
    the schema's minimum/maximum (two-sided bounds take the exact
    _int_range_ir path instead)."""
    sign = Opt(Lit(b"-")) if allow_negative else Lit(b"")
    if max_digits is None:
        body = Alt([Lit(b"0"), Seq([Cls(set(b"123456789")), Star(Cls(_DIGITS))])])
    else:
        body = Alt([Lit(b"0"), Seq([Cls(set(b"123456789")),
                                    _bounded(lambda: Cls(_DIGITS), 0, max(0, max_digits - 1))])])
    return Seq([sign, body])


def _number_ir(max_int_digits: Optional[int] = None, allow_negative: bool = True) -> _Node:
    """JSON number. With schema bounds the INTEGER PART's digit count is
    capped (exact float ranges are not DFA-expressible; the cap bounds the
    magnitude so maximum=100 can't emit 12345.0) and the exponent is
    dropped (an exponent reintroduces unbounded magnitude)."""
    frac = Seq([Lit(b"."), Cls(_DIGITS), Star(Cls(_DIGITS))])
    exp = Seq([Cls(set(b"eE")), Opt(Cls(set(b"+-"))), Cls(_DIGITS), Star(Cls(_DIGITS))])
    if max_int_digits is None:
        return Seq([_integer_ir(None, allow_negative), Opt(frac), Opt(exp)])
    return Seq([_integer_ir(max_int_digits, allow_negative), Opt(frac)])


class SchemaCompileError(ValueError):
    pass


def schema_to_ir(schema: Dict[str, Any], defs: Dict[str, Any], depth: int = 0, ws: bool = False) -> _Node:
    """Schema -> IR. With ws=True every JSON token (value start, key, comma,
    colon, closer) admits optional leading whitespace; ws inside atoms
    (strings, numbers) stays forbidden, as in the JSON grammar."""
    return _lead_ws(_schema_ir_body(schema, defs, depth, ws), ws)


def _schema_ir_body(schema: Dict[str, Any], defs: Dict[str, Any], depth: int, ws: bool) -> _Node:
    if depth > 32:
        raise SchemaCompileError("schema nesting too deep (recursive $ref?)")
    if "$ref" in schema:
        ref = schema["$ref"]
        name = ref.split("/")[-1]
        if name not in defs:
            raise SchemaCompileError(f"unresolved $ref {ref}")
        return _schema_ir_body(defs[name], defs, depth + 1, ws)
    if "enum" in schema:
        return Alt([Lit(json.dumps(v).encode()) for v in schema["enum"]])
    if "const" in schema:
        return Lit(json.dumps(schema["const"]).encode())
    if "anyOf" in schema or "oneOf" in schema:
        opts = schema.get("anyOf") or schema.get("oneOf")
        return Alt([_schema_ir_body(s, defs, depth + 1, ws) for s in opts])
    if "allOf" in schema:
        # pydantic v1-style wrapper: a single-element allOf is just the
        # element (plus sibling annotations like description). True
        # multi-constraint intersections aren't DFA-composable here; merge
        # sibling keys into a single-element body, else reject loudly.
        parts = schema["allOf"]
        if len(parts) == 1:
            merged = {**{k: v for k, v in schema.items() if k != "allOf"}, **parts[0]}
            return _schema_ir_body(merged, defs, depth + 1, ws)
        raise SchemaCompileError("multi-element allOf is not supported")

    t = schema.get("type")
    if isinstance(t, list):
        return Alt([_schema_ir_body({**schema, "type": ti}, defs, depth + 1, ws) for ti in t])
    if t == "string":
        if schema.get("format"):
            body = _format_string_ir(schema["format"])
            if body is not None:
                return Seq([Lit(b'"'), body, Lit(b'"')])
            # unknown formats fall through to the plain string grammar
        if schema.get("pattern"):
            try:
                body = _compile_pattern_ir(schema["pattern"])
                return Seq([Lit(b'"'), body, Lit(b'"')])
            except (_PatternUnsupported, UnicodeEncodeError):
                pass  # unsupported construct: plain string grammar
        return _json_string_ir(schema.get("minLength", 0), schema.get("maxLength"))
    if t == "integer":
        mn, mx = schema.get("minimum"), schema.get("maximum")
        # pydantic gt/lt emit exclusive bounds (JSON Schema draft 2020)
        if schema.get("exclusiveMinimum") is not None:
            emn = int(schema["exclusiveMinimum"]) + 1
            mn = emn if mn is None else max(int(mn), emn)
        if schema.get("exclusiveMaximum") is not None:
            emx = int(schema["exclusiveMaximum"]) - 1
            mx = emx if mx is None else min(int(mx), emx)
        if mn is not None and mx is not None:
            if int(mn) > int(mx):
                raise SchemaCompileError(f"empty integer range [{mn}, {mx}]")
            return _int_range_ir(int(mn), int(mx))
        max_digits = None
        if mx is not None or mn is not None:
            bound = max(abs(int(mx)) if mx is not None else 0,
                        abs(int(mn)) if mn is not None else 0)
            max_digits = max(1, len(str(bound)))
        allow_neg = mn is None or mn < 0
        return _integer_ir(max_digits, allow_neg)
    if t == "number":
        mn, mx = schema.get("minimum"), schema.get("maximum")
        if schema.get("exclusiveMinimum") is not None:
            mn = schema["exclusiveMinimum"] if mn is None else max(mn, schema["exclusiveMinimum"])
        if schema.get("exclusiveMaximum") is not None:
            mx = schema["exclusiveMaximum"] if mx is None else min(mx, schema["exclusiveMaximum"])
        max_int_digits = None
        if mn is not None or mx is not None:
            bound = max(abs(float(mx)) if mx is not None else 0.0,
                        abs(float(mn)) if mn is not None else 0.0)
            max_int_digits = max(1, len(str(int(bound))))
        allow_neg = mn is None or mn < 0
        return _number_ir(max_int_digits, allow_neg)
    if t == "boolean":
        return Alt([Lit(b"true"), Lit(b"false")])
    if t == "null":
        return Lit(b"null")
    if t == "array":
        pref = schema.get("prefixItems")
        if pref and schema.get("minItems", 0) >= len(pref):
            # pydantic Tuple[...]: POSITIONAL subschemas enforced (the
            # any-value fallback admitted ["x", 3] for Tuple[int, str]);
            # items beyond the prefix follow the `items` schema
            parts: List[_Node] = [Lit(b"[")]
            for i, sub in enumerate(pref):
                if i:
                    parts.append(_lead_ws(Lit(b","), ws))
                parts.append(_lead_ws(schema_to_ir(sub, defs, depth + 1, ws), ws))
            n = len(pref)
            max_items = schema.get("maxItems")
            tail_sch = schema.get("items", {})
            if tail_sch is not False and (max_items is None or max_items > n):
                tail = lambda: Seq([
                    _lead_ws(Lit(b","), ws),
                    _lead_ws(schema_to_ir(tail_sch, defs, depth + 1, ws) if tail_sch
                             else _any_value_ir(defs, depth + 1, ws=ws), ws),
                ])
                parts.append(_bounded(tail, max(0, schema.get("minItems", n) - n),
                                      None if max_items is None else max_items - n))
            parts.append(_lead_ws(Lit(b"]"), ws))
            return Seq(parts)
        item = schema.get("items", {})

        def item_ir() -> _Node:
            return schema_to_ir(item, defs, depth + 1, ws) if item else _any_value_ir(defs, depth + 1, ws=ws)

        min_items = schema.get("minItems", 0)
        max_items = schema.get("maxItems")
        comma = lambda: _lead_ws(Lit(b","), ws)
        if min_items == 0:
            more = _bounded(lambda: Seq([comma(), item_ir()]), 0,
                            None if max_items is None else max(0, max_items - 1))
            return Seq([Lit(b"["), Opt(Seq([item_ir(), more])), _lead_ws(Lit(b"]"), ws)])
        head = [item_ir()] + [Seq([comma(), item_ir()]) for _ in range(min_items - 1)]
        more = _bounded(lambda: Seq([comma(), item_ir()]), 0,
                        None if max_items is None else max(0, max_items - min_items))
        return Seq([Lit(b"[")] + head + [more, _lead_ws(Lit(b"]"), ws)])
    if t == "object" or "properties" in schema:
        props = schema.get("properties", {})
        if not props:
            ap = schema.get("additionalProperties")
            if isinstance(ap, dict):
                # pydantic Dict[str, T]: free-form keys, TYPED values — the
                # value subschema is enforced (an untyped fallback here let
                # Dict[str, int] sample string values that pydantic rejects)
                def member() -> _Node:
                    return Seq([_lead_ws(_json_string_ir(), ws), _lead_ws(Lit(b":"), ws),
                                _lead_ws(schema_to_ir(ap, defs, depth + 1, ws), ws)])
                min_p = schema.get("minProperties", 0)
                max_p = schema.get("maxProperties")
                more = lambda: Seq([_lead_ws(Lit(b","), ws), member()])
                if min_p == 0:
                    rest = _bounded(more, 0, None if max_p is None else max(0, max_p - 1))
                    body: _Node = Opt(Seq([member(), rest]))
                else:
                    head = [member()] + [more() for _ in range(min_p - 1)]
                    rest = _bounded(more, 0, None if max_p is None else max(0, max_p - min_p))
                    body = Seq(head + [rest])
                return Seq([Lit(b"{"), body, _lead_ws(Lit(b"}"), ws)])
            return _any_object_ir(defs, depth + 1, ws)
        parts: List[_Node] = [Lit(b"{")]
        required = set(schema.get("required", list(props.keys())))
        first = True
        for key, sub in props.items():
            field = Seq([
                Lit(b"") if first else _lead_ws(Lit(b","), ws),
                _lead_ws(Lit(json.dumps(key).encode()), ws),
                _lead_ws(Lit(b":"), ws),
                schema_to_ir(sub, defs, depth + 1, ws),
            ])
            if key in required or first:
                # fields emitted in schema order; the first field is always
                # emitted so comma placement stays regular
                parts.append(field)
            else:
                parts.append(Opt(field))
            first = False
        parts.append(_lead_ws(Lit(b"}"), ws))
        return Seq(parts)
    # untyped: any JSON value (bounded nesting)
    return _any_value_ir(defs, depth + 1, ws=ws)


def _any_value_ir(defs, depth: int, max_depth: int = 3, ws: bool = False) -> _Node:
    """'Any JSON value' with nesting bounded at max_depth levels.

    max_depth=3 keeps the free-form DFA ~7k states (measured; depth 4
    exceeds the 20k compile cap in both compact and ws modes). Typed
    schemas never hit this path and stay small."""
    scalar = Alt([_json_string_ir(), _number_ir(), Alt([Lit(b"true"), Lit(b"false")]), Lit(b"null")])
    node = scalar
    for _ in range(max_depth):
        val = _lead_ws(node, ws)
        key = _lead_ws(_json_string_ir(), ws)
        colon = _lead_ws(Lit(b":"), ws)
        arr = Seq([Lit(b"["), Opt(Seq([val, Star(Seq([_lead_ws(Lit(b","), ws), val]))])), _lead_ws(Lit(b"]"), ws)])
        obj = Seq([
            Lit(b"{"),
            Opt(Seq([
                key, colon, val,
                Star(Seq([_lead_ws(Lit(b","), ws), key, colon, val])),
            ])),
            _lead_ws(Lit(b"}"), ws),
        ])
        node = Alt([scalar, arr, obj])
    return node


def _any_object_ir(defs, depth: int, ws: bool = False) -> _Node:
    v = _any_value_ir(defs, depth, ws=ws)
    member = Seq([_lead_ws(_json_string_ir(), ws), _lead_ws(Lit(b":"), ws), _lead_ws(v, ws)])
    return Seq([Lit(b"{"), Opt(Seq([member, Star(Seq([_lead_ws(Lit(b","), ws), member]))])), _lead_ws(Lit(b"}"), ws)])


# ---------------------------------------------------------------------------
# Thompson NFA -> DFA
# ---------------------------------------------------------------------------

class _NFABuilder:
    def __init__(self):
        self.eps: List[List[int]] = []
        self.edges: List[Dict[int, List[int]]] = []  # state -> byte -> [targets]

    def new_state(self) -> int:
        self.eps.append([])
        self.edges.append({})
        return len(self.eps) - 1

    def add_eps(self, a: int, b: int):
        self.eps[a].append(b)

    def add_edge(self, a: int, byte: int, b: int):
        self.edges[a].setdefault(byte, []).append(b)

    def build(self, node: _Node) -> Tuple[int, int]:
        """Returns (start, accept) fragment."""
        if isinstance(node, Lit):
            s = self.new_state()
            cur = s
            for by in node.s:
                nxt = self.new_state()
                self.add_edge(cur, by, nxt)
                cur = nxt
            return s, cur
        if isinstance(node, Cls):
            s, e = self.new_state(), self.new_state()
            for by in node.set:
                self.add_edge(s, by, e)
            return s, e
        if isinstance(node, Seq):
            if not node.parts:
                s = self.new_state()
                return s, s
            s0, e0 = self.build(node.parts[0])
            for p in node.parts[1:]:
                s1, e1 = self.build(p)
                self.add_eps(e0, s1)
                e0 = e1
            return s0, e0
        if isinstance(node, Alt):
            s, e = self.new_state(), self.new_state()
            for p in node.parts:
                ps, pe = self.build(p)
                self.add_eps(s, ps)
                self.add_eps(pe, e)
            return s, e
        if isinstance(node, Star):
            s, e = self.new_state(), self.new_state()
            ps, pe = self.build(node.inner)
            self.add_eps(s, ps)
            self.add_eps(pe, ps)
            self.add_eps(s, e)
            self.add_eps(pe, e)
            return s, e
        if isinstance(node, Opt):
            s, e = self.new_state(), self.new_state()
            ps, pe = self.build(node.inner)
            self.add_eps(s, ps)
            self.add_eps(pe, e)
            self.add_eps(s, e)
            return s, e
        raise TypeError(node)


def _eps_closure(nfa: _NFABuilder, states: frozenset) -> frozenset:
    stack = list(states)
    seen = set(states)
    while stack:
        s = stack.pop()
        for t in nfa.eps[s]:
            if t not in seen:
                seen.add(t)
                stack.append(t)
    return frozenset(seen)


DEAD = np.uint16(0xFFFF)


def compile_dfa(node: _Node) -> Tuple[np.ndarray, np.ndarray, int]:
    """Returns (trans[S,256] uint16 with 0xFFFF=dead, accepting[S] bool, start)."""
    nfa = _NFABuilder()
    start, accept = nfa.build(node)
    d0 = _eps_closure(nfa, frozenset([start]))
    state_ids: Dict[frozenset, int] = {d0: 0}
    work = [d0]
    rows: List[np.ndarray] = []
    accepting: List[bool] = []
    while work:
        cur = work.pop(0)
        row = np.full(256, DEAD, dtype=np.uint16)
        # gather byte transitions
        by_byte: Dict[int, set] = {}
        for s in cur:
            for by, targets in nfa.edges[s].items():
                by_byte.setdefault(by, set()).update(targets)
        for by, targets in by_byte.items():
            nxt = _eps_closure(nfa, frozenset(targets))
            if nxt not in state_ids:
                state_ids[nxt] = len(state_ids)
                work.append(nxt)
            row[by] = state_ids[nxt]
        rows.append(row)
        accepting.append(accept in cur)
        if len(state_ids) > 20000:
            raise SchemaCompileError("DFA too large")
    # rows were appended in BFS pop order == id order
    trans = np.stack(rows)
    return trans, np.array(accepting, dtype=bool), 0


# ---------------------------------------------------------------------------
# Token-level tables
# ---------------------------------------------------------------------------

_TABLE_CACHE: Dict[Tuple[str, int], "JsonSchemaConstraint"] = {}


def _dfa_vocab_product(trans: np.ndarray, tok_bytes: List[Optional[bytes]],
                       S: int, V: int) -> np.ndarray:
    """next_state[S, V]: for every DFA state, where each token's byte string
    lands (DEAD if any byte dies).

    SPARSE level-vectorized trie walk: tokens are organized into a byte trie
    (shared prefixes walked once), and the live (trie node, source state)
    pairs are kept as flat arrays grouped by node — after one byte most of
    the S source states are dead for most subtrees, so work scales with the
    LIVE pairs, not S x nodes. All per-level bookkeeping is bincount/cumsum
    vectorized. (A dense per-byte-position walk over all V columns took 67 s
    on a 128k vocab; this takes ~2 s.)
    """
    next_state = np.full((S, V), DEAD, dtype=np.uint16)

    # frontier entry: (node_id_at_level, [(token, bytes) longer than depth])
    frontier = [(0, [(i, b) for i, b in enumerate(tok_bytes) if b])]
    pair_row = np.arange(S, dtype=np.int64)       # source DFA state
    pair_val = np.arange(S, dtype=np.int64)       # state after node's prefix
    node_start = {0: 0}
    node_count = {0: S}
    depth = 0
    while frontier:
        child_parent: List[int] = []
        child_byte: List[int] = []
        child_tok1: List[int] = []                # single finishing token or -1
        child_multi: dict = {}                    # child -> [tokens] (dup strings)
        child_rest: List[list] = []
        for node, members in frontier:
            groups: dict = {}
            for tb in members:
                groups.setdefault(tb[1][depth], []).append(tb)
            for byte, sub in groups.items():
                c = len(child_parent)
                child_parent.append(node)
                child_byte.append(byte)
                done = [t for t, b in sub if len(b) == depth + 1]
                if len(done) == 1:
                    child_tok1.append(done[0])
                else:
                    child_tok1.append(-1)
                    if done:
                        child_multi[c] = done
                child_rest.append([tb for tb in sub if len(tb[1]) > depth + 1])
        if not child_parent:
            break
        n_children = len(child_parent)
        cb = np.asarray(child_byte, dtype=np.int64)
        par_start = np.asarray([node_start[p] for p in child_parent], dtype=np.int64)
        par_count = np.asarray([node_count[p] for p in child_parent], dtype=np.int64)
        total = int(par_count.sum())
        if total == 0:
            break
        coff = np.zeros(n_children + 1, dtype=np.int64)
        np.cumsum(par_count, out=coff[1:])
        which_child = np.repeat(np.arange(n_children, dtype=np.int64), par_count)
        src_idx = par_start[which_child] + (np.arange(total, dtype=np.int64) - coff[which_child])
        new_row = pair_row[src_idx]
        new_val = trans[pair_val[src_idx], cb[which_child]].astype(np.int64)
        alive = new_val != DEAD

        # finish: pairs of children where a token ends -> one fancy write
        tok1 = np.asarray(child_tok1, dtype=np.int64)
        fin = alive & (tok1[which_child] >= 0)
        if fin.any():
            next_state[new_row[fin], tok1[which_child[fin]]] = new_val[fin].astype(np.uint16)
        for c, toks in child_multi.items():       # duplicate byte strings: rare
            sel = slice(int(coff[c]), int(coff[c + 1]))
            a = alive[sel]
            if not a.any():
                continue
            r, v = new_row[sel][a], new_val[sel][a].astype(np.uint16)
            for t in toks:
                next_state[r, t] = v

        # next level: alive pairs of children that have continuations
        has_rest = np.asarray([bool(r) for r in child_rest])
        keep = alive & has_rest[which_child]
        if not keep.any():
            break
        kept_child = which_child[keep]
        pair_row = new_row[keep]
        pair_val = new_val[keep]
        counts = np.bincount(kept_child, minlength=n_children)
        starts = np.zeros(n_children + 1, dtype=np.int64)
        np.cumsum(counts, out=starts[1:])
        node_start = {}
        node_count = {}
        frontier = []
        for c in np.nonzero(counts)[0].tolist():
            if child_rest[c]:
                node_start[c] = int(starts[c])
                node_count[c] = int(counts[c])
                frontier.append((c, child_rest[c]))
        depth += 1
    return next_state


class JsonSchemaConstraint:
    """Per-request constraint handle. State is an int DFA state; tables are
    cached per (schema, tokenizer vocab)."""

    def __init__(self, schema: Dict[str, Any], tokenizer, whitespace: bool = False):
        self.schema = schema
        self.whitespace = whitespace
        tok_key = getattr(tokenizer, "cache_key", None) or (id(type(tokenizer)), tokenizer.vocab_size)
        key = (json.dumps(schema, sort_keys=True), tok_key, whitespace)
        cached = _TABLE_CACHE.get(key)  # type: ignore[arg-type]
        if cached is not None:
            self.__dict__.update(cached.__dict__)
            return

        defs = schema.get("$defs", schema.get("definitions", {}))
        ir = schema_to_ir(schema, defs, ws=whitespace)
        trans, accepting, start = compile_dfa(ir)
        S = trans.shape[0]
        V = tokenizer.vocab_size
        self.start_state = start
        self.accepting = accepting
        self.eos_id = tokenizer.eos_id

        # token byte strings
        tok_bytes: List[Optional[bytes]] = [tokenizer.token_bytes(i) for i in range(V)]
        # next_state[S, V] via a LEVEL-VECTORIZED TRIE walk: tokens sharing a
        # byte prefix (BPE merge families, the byte tokenizer's <tkN> filler
        # block) are walked once per unique prefix instead of once per token —
        # a naive per-byte-position walk over all V columns was 30x slower on
        # a 128k vocab (67 s -> ~2 s for the same product).
        next_state = _dfa_vocab_product(trans, tok_bytes, S, V)
        self.next_state = next_state

        # allowed bitmask per state
        W = (V + 31) // 32
        allowed = next_state != DEAD  # [S, V]
        bits = np.zeros((S, W), dtype=np.uint32)
        idx = np.arange(V)
        for w in range(W):
            sel = (idx // 32) == w
            sub = allowed[:, sel]
            shifts = (idx[sel] % 32).astype(np.uint32)
            bits[:, w] = (sub.astype(np.uint32) << shifts[None, :]).sum(axis=1, dtype=np.uint32)
        # accepting states may also emit EOS
        if self.eos_id is not None:
            bits[accepting, self.eos_id // 32] |= np.uint32(1 << (self.eos_id % 32))
        self._mask_t = torch.from_numpy(bits.view(np.int32))
        # dead-end guard: states with no allowed token at all -> EOS-only
        empty = ~allowed.any(axis=1) & ~accepting
        if empty.any() and self.eos_id is not None:
            # build the bit in uint32 and view as int32: 1 << 31 overflows
            # np.int32 directly (NumPy >= 1.24 raises OverflowError)
            eos_bit = int(np.uint32(1 << (self.eos_id % 32)).view(np.int32))
            self._mask_t[np.nonzero(empty)[0], self.eos_id // 32] |= eos_bit
        # terminal = accepting with no possible continuation (e.g. the closing
        # brace of the top-level object): generation stops there immediately.
        # Accepting-but-extendable states (e.g. "12" under an integer schema)
        # continue until the model emits EOS (allowed by the mask above).
        self._terminal = accepting & ~allowed.any(axis=1)

        _TABLE_CACHE[key] = self  # type: ignore[index]

    # --- engine interface -----------------------------------------------------
    def init_state(self) -> int:
        return self.start_state

    def allowed_mask(self, state: int) -> torch.Tensor:
        return self._mask_t[state]

    def advance(self, state: int, token: int) -> int:
        if token == self.eos_id:
            return state
        nxt = int(self.next_state[state, token])
        if nxt == int(DEAD):
            return state  # should not happen under the mask; stay put
        return nxt

    def is_final(self, state: int) -> bool:
        return bool(self._terminal[state])
