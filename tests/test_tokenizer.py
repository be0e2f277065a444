"""Tokenizer tests: byte-level round trips, chat templating, crop semantics
(the local equivalents of the reference's tiktoken usage)."""

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from kllms_amd.engine.tokenizer import ByteTokenizer


@pytest.fixture(scope="module")
def tok():
    return ByteTokenizer(2048)


class TestRoundTrip:
    @settings(max_examples=200, deadline=None)
    @given(st.text(max_size=60))
    def test_encode_decode_identity(self, tok, s):
        assert tok.decode(tok.encode(s)) == s

    @settings(max_examples=100, deadline=None)
    @given(st.binary(max_size=40))
    def test_arbitrary_bytes_survive(self, tok, b):
        s = b.decode("utf-8", errors="replace")
        assert tok.decode(tok.encode(s)) == s

    def test_ids_in_vocab(self, tok):
        ids = tok.encode("hello é世界")
        assert all(0 <= i < tok.vocab_size for i in ids)


class TestChatTemplate:
    def test_roles_present_and_deterministic(self, tok):
        msgs = [
            {"role": "system", "content": "be brief"},
            {"role": "user", "content": "hi"},
            {"role": "assistant", "content": "hello"},
            {"role": "user", "content": "bye"},
        ]
        p1 = tok.apply_chat_template(msgs)
        p2 = tok.apply_chat_template(msgs)
        assert p1 == p2
        for m in msgs:
            assert m["content"] in p1
        # longer history -> longer prompt
        assert len(p1) > len(tok.apply_chat_template(msgs[:1]))


class TestCrop:
    def test_crop_to_tokens_bounds(self, tok):
        text = "word " * 300
        cropped = tok.crop_to_tokens(text, 50)
        assert len(tok.encode(cropped)) <= 50
        assert text.startswith(cropped)

    def test_crop_noop_when_short(self, tok):
        assert tok.crop_to_tokens("short", 100) == "short"

    def test_crop_zero(self, tok):
        assert tok.encode(tok.crop_to_tokens("abc", 0)) == []
