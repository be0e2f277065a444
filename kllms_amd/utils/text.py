"""Text utilities: Levenshtein distance and ASCII transliteration.

Standalone replacements for the ``python-Levenshtein`` and ``Unidecode``
dependencies of the reference (requirements.txt:1-8). Only the behavior the
consensus engine observes is implemented:

- ``levenshtein_distance`` — classic edit distance (insert/delete/substitute,
  unit costs), equal to ``Levenshtein.distance`` (used at
  consensus_utils.py:745-761 on normalize_string() outputs, i.e. short
  ASCII-ish strings).
- ``ascii_transliterate`` — accent-stripping transliteration used by
  ``sanitize_value`` (consensus_utils.py:925-933). The reference pipes the
  value through ``unidecode`` and then strips all non-alphanumerics, so only
  the alphanumeric portion of the transliteration is observable; NFKD
  decomposition plus a small latin special-case table covers that.
"""

from __future__ import annotations

import unicodedata

# Latin letters whose NFKD decomposition does not yield an ASCII base letter
# but which unidecode maps to one (observable through sanitize_value).
_SPECIAL = {
    "æ": "ae", "Æ": "AE", "œ": "oe", "Œ": "OE",
    "ø": "o", "Ø": "O", "đ": "d", "Đ": "D",
    "ð": "d", "Ð": "D", "þ": "th", "Þ": "Th",
    "ß": "ss", "ł": "l", "Ł": "L", "ħ": "h", "Ħ": "H",
    "ı": "i", "İ": "I", "ŋ": "ng", "Ŋ": "NG",
    "ĸ": "k", "ſ": "s", "ƒ": "f", "Ƒ": "F",
}


def ascii_transliterate(text: str) -> str:
    out = []
    for ch in text:
        if ord(ch) < 128:
            out.append(ch)
            continue
        if ch in _SPECIAL:
            out.append(_SPECIAL[ch])
            continue
        decomp = unicodedata.normalize("NFKD", ch)
        kept = "".join(c for c in decomp if ord(c) < 128 and not unicodedata.combining(c))
        out.append(kept)
    return "".join(out)


def levenshtein_distance(s: str, t: str) -> int:
    if s == t:
        return 0
    if not s:
        return len(t)
    if not t:
        return len(s)
    # keep the shorter string as the DP row
    if len(s) < len(t):
        s, t = t, s
    prev = list(range(len(t) + 1))
    for i, cs in enumerate(s, 1):
        cur = [i]
        for j, ct in enumerate(t, 1):
            cur.append(min(
                prev[j] + 1,          # deletion
                cur[j - 1] + 1,       # insertion
                prev[j - 1] + (cs != ct),  # substitution
            ))
        prev = cur
    return prev[-1]
