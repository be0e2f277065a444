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


class HFTokenizer(BaseTokenizer):
    """Wraps a HuggingFace ``tokenizers`` fast tokenizer file."""

    def __init__(self, tokenizer_json: str):
        from tokenizers import Tokenizer  # offline wheelhouse package

        self._tok = Tokenizer.from_file(tokenizer_json)
        self.vocab_size = self._tok.get_vocab_size()
        vocab = self._tok.get_vocab()
        self.bos_id = vocab.get("<|begin_of_text|>") or vocab.get("<s>")
        self.eos_id = vocab.get("<|eot_id|>") or vocab.get("<|end_of_text|>") or vocab.get("</s>")
        self.pad_id = self.eos_id or 0

    def encode(self, text: str, add_bos: bool = False) -> list[int]:
        ids = self._tok.encode(text, add_special_tokens=False).ids
        if add_bos and self.bos_id is not None:
            ids = [self.bos_id] + ids
        return ids

    def decode(self, ids: Sequence[int]) -> str:
        return self._tok.decode(list(ids), skip_special_tokens=True)

    def token_bytes(self, i: int) -> Optional[bytes]:
        s = self._tok.decode([i], skip_special_tokens=True)
        return s.encode() if s else None


def load_tokenizer(model: str, weights_path: Optional[str], vocab_size: int) -> BaseTokenizer:
    path = weights_path or model
    if os.path.isdir(path):
        tj = os.path.join(path, "tokenizer.json")
        if os.path.exists(tj):
            return HFTokenizer(tj)
    return ByteTokenizer(vocab_size)
