"""Train a REAL byte-level BPE tokenizer (HuggingFace `tokenizers` runtime)
for offline real-vocab testing.

There is no network in this environment, so no published checkpoint vocab is
fetchable — but the vocab *format* is what matters for correctness of
`HFTokenizer.token_bytes` and DFA×vocab constrained decoding: this produces a
genuine `tokenizer.json` with the GPT-2/Llama-3 ByteLevel alphabet (Ġ space
markers, multi-byte merges, added special tokens), the same shape as
Llama-3's own tokenizer.json (reference boundary:
/root/reference/k_llms/client.py:98 tiktoken usage).

Run: python scripts/make_test_tokenizer.py [out_path]
Writes tests/data/bpe_tokenizer.json by default (committed fixture).
"""

import json
import os
import sys

from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers, processors

CORPUS = [
    # English prose + JSON-shaped text so merges cover both
    "The quick brown fox jumps over the lazy dog. ",
    "Paris is the capital of France. Berlin is the capital of Germany. ",
    '{"name": "Alice", "age": 30, "city": "Paris", "active": true}',
    '{"items": [{"id": 1, "price": 9.99}, {"id": 2, "price": 19.5}], "total": 29.49}',
    '{"company": {"departments": [{"name": "R&D", "employees": 42}]}}',
    "What is the sum of 2 and 3? The answer is 5. ",
    "self-consistency inference engines sample n completions and vote. ",
    "The temperature today is 23.5 degrees; tomorrow it will be 19. ",
    '{"status": "ok", "count": 123, "ratio": 0.51, "tags": ["a", "b"]}',
    "données précises, naïve café, 東京 and Ελλάδα exercise multi-byte UTF-8. ",
    "system user assistant header tokens appear in chat templates. ",
    '{"answer": "yes", "confidence": 0.95, "reasons": ["fast", "cheap"]}',
] * 50

SPECIALS = [
    "<|begin_of_text|>",
    "<|end_of_text|>",
    "<|start_header_id|>",
    "<|end_header_id|>",
    "<|eot_id|>",
]


def make_sp(out_path: str) -> None:
    """SentencePiece-style fixture: Unigram + Metaspace (real ▁ markers) +
    the full <0xNN> byte-fallback special block — the OTHER vocab format
    HFTokenizer.token_bytes must decode (Llama-2/Mistral tokenizer.json
    shape)."""
    from tokenizers import decoders, models, pre_tokenizers, trainers

    tok = Tokenizer(models.Unigram())
    tok.pre_tokenizer = pre_tokenizers.Metaspace()
    tok.decoder = decoders.Metaspace()
    byte_fb = [f"<0x{i:02X}>" for i in range(256)]
    trainer = trainers.UnigramTrainer(
        vocab_size=1200, special_tokens=["<unk>", "<s>", "</s>"],
        unk_token="<unk>", show_progress=False,
        initial_alphabet=list("abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789{}[]:,.\"'+-_ "),
    )
    tok.train_from_iterator(CORPUS, trainer=trainer)
    # byte-fallback pieces are ORDINARY vocab entries in real SP tokenizers
    # (Llama-2 shape), not special added tokens
    tok.add_tokens(byte_fb)
    tok.save(out_path)
    t2 = Tokenizer.from_file(out_path)
    v = t2.get_vocab()
    assert any(k.startswith("\u2581") or k.startswith("▁") for k in v), "no metaspace markers"
    assert "<0x41>" in v
    print(f"wrote {out_path}: vocab={t2.get_vocab_size()}")


def main(out_path: str) -> None:
    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    tok.post_processor = processors.ByteLevel(trim_offsets=False)
    trainer = trainers.BpeTrainer(
        vocab_size=2048,
        special_tokens=SPECIALS,
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
        show_progress=False,
    )
    tok.train_from_iterator(CORPUS, trainer=trainer)
    os.makedirs(os.path.dirname(out_path), exist_ok=True)
    tok.save(out_path)
    # sanity: round-trip + Ġ markers present
    t2 = Tokenizer.from_file(out_path)
    s = 'Hello world {"key": "value", "n": 42}'
    ids = t2.encode(s, add_special_tokens=False).ids
    assert t2.decode(ids) == s, (t2.decode(ids), s)
    vocab = t2.get_vocab()
    assert any(tk.startswith("Ġ") for tk in vocab), "no ByteLevel space markers?"
    print(f"wrote {out_path}: vocab={t2.get_vocab_size()}, sample ids={ids[:8]}")


if __name__ == "__main__":
    data = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests", "data")
    main(sys.argv[1] if len(sys.argv) > 1 else os.path.join(data, "bpe_tokenizer.json"))
    make_sp(os.path.join(data, "sp_tokenizer.json"))
