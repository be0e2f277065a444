"""Tokenization (CPU).

Replaces the reference's delegation of tokenization to the OpenAI server and
to tiktoken (k_llms/client.py:98-102). Two backends:

- ``HFTokenizer``: loads a ``tokenizer.json`` (HuggingFace ``tokenizers``
  runtime, available offline) for real model vocabularies.
- ``ByteTokenizer``: self-contained byte-level tokenizer for synthetic /
  random-weight runs (no network, no vocab files): ids 0..255 are raw bytes,
  then BOS/EOS/PAD specials; the remaining ids up to ``vocab_size`` decode to
  ``<tk{id}>`` filler strings. Fully reversible for byte-range ids, which is
  what JSON-constrained decoding exercises.

Chat templating: a minimal Llama-3-style header template — the engine only
needs a deterministic prompt rendering for synthetic workloads.
"""

from __future__ import annotations

import os
from typing import Optional, Sequence


class BaseTokenizer:
    vocab_size: int
    bos_id: Optional[int]
    eos_id: Optional[int]
    pad_id: int

    def encode(self, text: str, add_bos: bool = False) -> list[int]:
        raise NotImplementedError

    def decode(self, ids: Sequence[int]) -> str:
        raise NotImplementedError

    def crop_to_tokens(self, text: str, max_tokens: int) -> str:
        return self.decode(self.encode(text)[:max_tokens])

    def apply_chat_template(self, messages: list[dict]) -> str:
        parts = ["<|begin_of_text|>"]
        for m in messages:
            role = m.get("role", "user")
            content = m.get("content", "")
            parts.append(f"<|start_header_id|>{role}<|end_header_id|>\n\n{content}<|eot_id|>")
        parts.append("<|start_header_id|>assistant<|end_header_id|>\n\n")
        return "".join(parts)


class ByteTokenizer(BaseTokenizer):
    """Byte-level tokenizer over a (possibly much larger) model vocab."""

    N_BYTES = 256

    def __init__(self, vocab_size: int = 128256):
        assert vocab_size >= self.N_BYTES + 3
        self.vocab_size = vocab_size
        self.bos_id = self.N_BYTES
        self.eos_id = self.N_BYTES + 1
        self.pad_id = self.N_BYTES + 2
        self.cache_key = ("byte", vocab_size)

    def encode(self, text: str, add_bos: bool = False) -> list[int]:
        ids = list(text.encode("utf-8", errors="replace"))
        if add_bos:
            ids = [self.bos_id] + ids
        return ids

    def decode(self, ids: Sequence[int]) -> str:
        out = bytearray()
        pieces: list[str] = []
        for i in ids:
            if 0 <= i < self.N_BYTES:
                out.append(i)
            else:
                if out:
                    pieces.append(out.decode("utf-8", errors="replace"))
                    out = bytearray()
                if i == self.bos_id:
                    pieces.append("<|begin_of_text|>")
                elif i == self.eos_id:
                    pieces.append("<|eot_id|>")
                elif i == self.pad_id:
                    pass
                else:
                    pieces.append(f"<tk{i}>")
        if out:
            pieces.append(out.decode("utf-8", errors="replace"))
        return "".join(pieces)

    def token_str(self, i: int) -> Optional[str]:
        """The exact string a single token contributes to decoded output, or
        None for non-text control tokens. Used by the constrained-decode FSM."""
        if 0 <= i < self.N_BYTES:
            try:
                return bytes([i]).decode("utf-8")
            except UnicodeDecodeError:
                return None  # lone continuation bytes are not valid text alone
        if i == self.eos_id or i == self.bos_id or i == self.pad_id:
            return None
        return f"<tk{i}>"

    def token_bytes(self, i: int) -> Optional[bytes]:
        if 0 <= i < self.N_BYTES:
            return bytes([i])
        if i in (self.bos_id, self.eos_id, self.pad_id):
            return None
        return f"<tk{i}>".encode()


def _bytelevel_unicode_to_byte() -> dict:
    """Inverse of the GPT-2/Llama-3 ByteLevel byte->unicode alphabet: every
    raw byte is represented in vocab strings by a printable unicode char
    (0x20 -> 'Ġ', etc.). Standard table from the GPT-2 tokenizer."""
    bs = list(range(ord("!"), ord("~") + 1)) + list(range(0xA1, 0xAD)) + list(range(0xAE, 0x100))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return {chr(c): b for b, c in zip(bs, cs)}


class HFTokenizer(BaseTokenizer):
    """Wraps a HuggingFace ``tokenizers`` fast tokenizer file.

    ``token_bytes`` derives each token's TRUE byte string from the vocab
    surface form — ByteLevel alphabet chars (Ġ = space, Ċ = newline, …)
    mapped through the inverse GPT-2 table, SentencePiece ▁ markers mapped
    to spaces and ``<0xNN>`` byte-fallback tokens to their raw byte — NOT
    from per-token ``decode([i])``, which mangles byte-fallback and marker
    tokens (reference boundary this feeds:
    /root/reference/k_llms/resources/completions/completions.py:134 — the
    server-side parse() must constrain over the real vocab)."""

    def __init__(self, tokenizer_json: str):
        import json as _json

        from tokenizers import Tokenizer  # offline wheelhouse package

        self._tok = Tokenizer.from_file(tokenizer_json)
        self.vocab_size = self._tok.get_vocab_size()
        self.cache_key = ("hf", os.path.abspath(tokenizer_json), os.path.getmtime(tokenizer_json))
        vocab = self._tok.get_vocab()
        def _first_id(*names):
            # `or`-chaining is wrong here: a valid token id of 0 is falsy
            for nm in names:
                if nm in vocab:
                    return vocab[nm]
            return None

        self.bos_id = _first_id("<|begin_of_text|>", "<s>")
        self.eos_id = _first_id("<|eot_id|>", "<|im_end|>", "<|end_of_text|>", "</s>", "<|endoftext|>")
        self.pad_id = self.eos_id if self.eos_id is not None else 0

        # Real-checkpoint chat fidelity: HF checkpoints ship the model's own
        # prompt format as a Jinja2 `chat_template` in tokenizer_config.json
        # (Qwen ChatML, Llama-3 headers, ...). Load it when present; the
        # hard-coded Llama-3 format stays the fallback.
        self._chat_template = None
        self._cfg_bos = self._cfg_eos = None
        tc_path = os.path.join(os.path.dirname(os.path.abspath(tokenizer_json)),
                               "tokenizer_config.json")
        if os.path.exists(tc_path):
            try:
                with open(tc_path, "r", encoding="utf-8") as f:
                    tc = _json.load(f)
                ct = tc.get("chat_template")
                if isinstance(ct, list):  # newer multi-template form
                    ct = next((e.get("template") for e in ct
                               if e.get("name") == "default"), None) or (
                        ct[0].get("template") if ct else None)
                if isinstance(ct, str) and ct.strip():
                    self._chat_template = ct
                def _tok_str(v):
                    return v.get("content") if isinstance(v, dict) else v
                self._cfg_bos = _tok_str(tc.get("bos_token"))
                self._cfg_eos = _tok_str(tc.get("eos_token"))
                # prefer the checkpoint's declared eos over the name guess
                if self._cfg_eos in vocab:
                    self.eos_id = vocab[self._cfg_eos]
                    self.pad_id = self.eos_id
            except Exception:
                self._chat_template = None

        # id -> vocab surface string, and the set of special/added ids
        self._id_to_token = {i: t for t, i in vocab.items()}
        with open(tokenizer_json, "r", encoding="utf-8") as f:
            spec = _json.load(f)
        self._special_ids = {a["id"] for a in spec.get("added_tokens", []) if a.get("special")}
        self._u2b = _bytelevel_unicode_to_byte()
        # scheme detection: ByteLevel decoder/pretokenizer => GPT-2 alphabet;
        # otherwise SentencePiece-style (metaspace + <0xNN> byte fallback)
        def _has_bytelevel(node) -> bool:
            if isinstance(node, dict):
                if node.get("type") == "ByteLevel":
                    return True
                return any(_has_bytelevel(v) for v in node.values())
            if isinstance(node, list):
                return any(_has_bytelevel(v) for v in node)
            return False

        self._byte_level = _has_bytelevel(spec.get("decoder")) or _has_bytelevel(spec.get("pre_tokenizer"))
        self._tb_cache: dict = {}

    def apply_chat_template(self, messages: list[dict]) -> str:
        if not self._chat_template:
            return super().apply_chat_template(messages)
        try:
            import json as _json
            from datetime import datetime

            import jinja2
            from jinja2.sandbox import ImmutableSandboxedEnvironment

            if getattr(self, "_compiled_template", None) is None:
                # sandboxed + trim/lstrip to match transformers' rendering
                env = ImmutableSandboxedEnvironment(trim_blocks=True, lstrip_blocks=True)
                env.filters["tojson"] = lambda v, **kw: _json.dumps(v, **kw)
                self._compiled_template = env.from_string(self._chat_template)

            def raise_exception(msg):  # HF templates call this on bad input
                raise jinja2.TemplateError(msg)

            out = self._compiled_template.render(
                messages=messages,
                add_generation_prompt=True,
                bos_token=self._cfg_bos or "",
                eos_token=self._cfg_eos or "",
                raise_exception=raise_exception,
                strftime_now=lambda fmt: datetime.now().strftime(fmt),
                tools=None,
            )
            return out
        except Exception:
            # template requires features we don't model (tool schemas, ...)
            return super().apply_chat_template(messages)

    def encode(self, text: str, add_bos: bool = False) -> list[int]:
        ids = self._tok.encode(text, add_special_tokens=False).ids
        if add_bos and self.bos_id is not None:
            ids = [self.bos_id] + ids
        return ids

    def decode(self, ids: Sequence[int]) -> str:
        return self._tok.decode(list(ids), skip_special_tokens=True)

    def token_bytes(self, i: int) -> Optional[bytes]:
        if i in self._tb_cache:
            return self._tb_cache[i]
        b = self._token_bytes_uncached(i)
        self._tb_cache[i] = b
        return b

    def _token_bytes_uncached(self, i: int) -> Optional[bytes]:
        if i in self._special_ids:
            return None
        s = self._id_to_token.get(i)
        if s is None:
            return None
        if self._byte_level:
            try:
                return bytes(self._u2b[ch] for ch in s)
            except KeyError:
                # not in the ByteLevel alphabet (e.g. an added non-special
                # token stored verbatim): take its UTF-8 bytes
                return s.encode("utf-8")
        # SentencePiece-style vocab
        if len(s) == 6 and s.startswith("<0x") and s.endswith(">"):
            try:
                return bytes([int(s[3:5], 16)])
            except ValueError:
                pass
        return s.replace("▁", " ").encode("utf-8")


def load_tokenizer(model: str, weights_path: Optional[str], vocab_size: int) -> BaseTokenizer:
    path = weights_path or model
    if os.path.isdir(path):
        tj = os.path.join(path, "tokenizer.json")
        if os.path.exists(tj):
            return HFTokenizer(tj)
    return ByteTokenizer(vocab_size)
