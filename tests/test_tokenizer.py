"""Tokenizer tests: byte-level round trips, chat templating, crop semantics
(the local equivalents of the reference's tiktoken usage)."""

import os
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


class TestHFChatTemplate:
    """HF tokenizer_config.json chat_template (Jinja2) drives the prompt
    format for real checkpoints; the Llama-3 format stays the fallback."""

    CHATML = (
        "{% for message in messages %}"
        "{{ '<|im_start|>' + message['role'] + '\n' + message['content'] + '<|im_end|>' + '\n' }}"
        "{% endfor %}"
        "{% if add_generation_prompt %}{{ '<|im_start|>assistant\n' }}{% endif %}"
    )

    def _fixture_dir(self, tmp_path):
        import shutil
        src = os.path.join(os.path.dirname(__file__), "data", "bpe_tokenizer.json")
        shutil.copy(src, tmp_path / "tokenizer.json")
        return tmp_path

    def test_template_rendered(self, tmp_path):
        import json
        d = self._fixture_dir(tmp_path)
        (d / "tokenizer_config.json").write_text(json.dumps({
            "chat_template": self.CHATML,
            "eos_token": "<|eot_id|>",
        }))
        from kllms_amd.engine.tokenizer import HFTokenizer
        tok = HFTokenizer(str(d / "tokenizer.json"))
        out = tok.apply_chat_template([
            {"role": "system", "content": "be brief"},
            {"role": "user", "content": "hi"},
        ])
        assert out == ("<|im_start|>system\nbe brief<|im_end|>\n"
                       "<|im_start|>user\nhi<|im_end|>\n"
                       "<|im_start|>assistant\n")

    def test_no_config_falls_back_to_llama3(self, tmp_path):
        d = self._fixture_dir(tmp_path)
        from kllms_amd.engine.tokenizer import HFTokenizer
        tok = HFTokenizer(str(d / "tokenizer.json"))
        out = tok.apply_chat_template([{"role": "user", "content": "x"}])
        assert "<|start_header_id|>user<|end_header_id|>" in out

    def test_broken_template_falls_back(self, tmp_path):
        import json
        d = self._fixture_dir(tmp_path)
        (d / "tokenizer_config.json").write_text(json.dumps({
            "chat_template": "{{ raise_exception('nope') }}",
        }))
        from kllms_amd.engine.tokenizer import HFTokenizer
        tok = HFTokenizer(str(d / "tokenizer.json"))
        out = tok.apply_chat_template([{"role": "user", "content": "x"}])
        assert "<|start_header_id|>" in out  # fell back, did not crash

    def test_declared_eos_overrides_name_guess(self, tmp_path):
        import json
        d = self._fixture_dir(tmp_path)
        from kllms_amd.engine.tokenizer import HFTokenizer
        base = HFTokenizer(str(d / "tokenizer.json"))
        vocab_eos = base._id_to_token.get(base.eos_id)
        # declare a DIFFERENT token from the vocab as eos
        other = next(t for t, i in base._tok.get_vocab().items()
                     if t.startswith("<|") and t != vocab_eos and t.endswith("|>"))
        (d / "tokenizer_config.json").write_text(json.dumps({"eos_token": other}))
        tok = HFTokenizer(str(d / "tokenizer.json"))
        assert tok._id_to_token[tok.eos_id] == other
