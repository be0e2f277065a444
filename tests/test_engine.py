"""Engine integration tests on the tiny CPU preset: greedy determinism,
paged-decode vs full-prefill numerics, KV fork/copy-on-write, stop handling,
usage accounting (SURVEY §4 'implication' items 3)."""

import pytest
import torch

from kllms_amd.engine.config import EngineConfig
from kllms_amd.engine.engine import GenRequest, LLMEngine
from kllms_amd.engine.kvcache import PagedKVCache
from kllms_amd.engine.sampling import SamplingParams
from kllms_amd.models.llama import ForwardBatch


@pytest.fixture(scope="module")
def engine():
    cfg = EngineConfig(
        model="tiny-llama", max_kv_blocks=512, use_hip_graphs=False,
        default_max_new_tokens=16, device="cpu", seed=0,
    )
    return LLMEngine(cfg)


def greedy(max_tokens=8, seed=0):
    return SamplingParams(temperature=0.0, max_tokens=max_tokens, seed=seed)


class TestGreedyDeterminism:
    def test_same_prompt_same_output(self, engine):
        req = lambda: GenRequest(prompt_ids=[1, 2, 3, 4, 5], n=2, sampling=greedy())
        out1 = engine.generate([req()])[0]
        out2 = engine.generate([req()])[0]
        assert out1.streams[0].token_ids == out2.streams[0].token_ids
        # greedy: all n streams identical
        assert out1.streams[0].token_ids == out1.streams[1].token_ids

    def test_stochastic_streams_differ(self, engine):
        req = GenRequest(
            prompt_ids=list(range(1, 30)), n=4,
            sampling=SamplingParams(temperature=1.0, max_tokens=12, seed=7),
        )
        out = engine.generate([req])[0]
        seqs = {tuple(s.token_ids) for s in out.streams}
        assert len(seqs) > 1  # overwhelmingly likely with vocab 512

    def test_seeded_reproducible(self, engine):
        mk = lambda: GenRequest(
            prompt_ids=[5, 6, 7], n=3,
            sampling=SamplingParams(temperature=0.9, max_tokens=10, seed=123),
        )
        o1 = engine.generate([mk()])[0]
        o2 = engine.generate([mk()])[0]
        for s1, s2 in zip(o1.streams, o2.streams):
            assert s1.token_ids == s2.token_ids


class TestPagedDecodeNumerics:
    def test_decode_matches_prefill_logits(self, engine):
        """Greedy-decode k tokens, then re-run one big prefill over
        prompt+decoded prefix: the next-token logits must match."""
        prompt = list(range(1, 20))
        out = engine.generate([GenRequest(prompt_ids=prompt, n=1, sampling=greedy(6))])[0]
        toks = out.streams[0].token_ids
        assert len(toks) >= 2

        # big prefill over prompt + all but last decoded token
        full = prompt + toks[:-1]
        seq = engine.kv.alloc_sequence(len(full))
        batch = ForwardBatch(
            mode="prefill",
            positions=torch.arange(len(full)),
            slot_mapping=torch.tensor(engine.kv.prefill_slot_mapping(seq), dtype=torch.long),
            kv_caches=engine.kv.layer_caches(),
            cu_seqlens=torch.tensor([0, len(full)], dtype=torch.int32),
        )
        logits = engine.model.forward_prefill(torch.tensor(full), batch)
        engine.kv.free_sequence(seq)
        assert int(logits[0].argmax()) == toks[-1]


class TestKVFork:
    def test_fork_refcounts_and_eager_tail_copy(self, engine):
        kv: PagedKVCache = engine.kv
        if engine.prefix_cache is not None:
            engine.prefix_cache.evict_all()
        free0 = kv.allocator.num_free
        parent = kv.alloc_sequence(kv.block_size + 3)  # 2 blocks, second partial
        child1 = kv.fork(parent)
        child2 = kv.fork(parent)
        # full blocks shared by refcount; partial tail block copied eagerly
        assert kv.allocator.refcount(parent.blocks[0]) == 3
        assert child1.blocks[0] == parent.blocks[0]
        assert child1.blocks[1] != parent.blocks[1]
        assert child2.blocks[1] != parent.blocks[1]
        assert kv.allocator.refcount(parent.blocks[1]) == 1
        assert kv.allocator.refcount(child1.blocks[1]) == 1
        # appends never trigger COW after an eager fork
        kv.append_slot(child1)
        assert kv.allocator.refcount(child1.blocks[1]) == 1
        kv.free_sequence(parent)
        kv.free_sequence(child1)
        kv.free_sequence(child2)
        if engine.prefix_cache is not None:
            engine.prefix_cache.evict_all()
        assert kv.allocator.num_free == free0

    def test_fork_at_block_boundary_shares_all(self, engine):
        kv: PagedKVCache = engine.kv
        parent = kv.alloc_sequence(kv.block_size)  # exactly one full block
        child = kv.fork(parent)
        assert child.blocks == parent.blocks
        assert kv.allocator.refcount(parent.blocks[0]) == 2
        # child's first append opens a fresh block
        kv.append_slot(child)
        assert len(child.blocks) == 2
        assert kv.allocator.refcount(child.blocks[1]) == 1
        kv.free_sequence(parent)
        kv.free_sequence(child)

    def test_no_block_leak_after_generate(self, engine):
        if engine.prefix_cache is not None:
            engine.prefix_cache.evict_all()
        free0 = engine.kv.allocator.num_free
        engine.generate([GenRequest(prompt_ids=list(range(1, 40)), n=5, sampling=greedy(9))])
        if engine.prefix_cache is not None:
            engine.prefix_cache.evict_all()
        assert engine.kv.allocator.num_free == free0


class TestStops:
    def test_eos_finishes(self, engine):
        # with max_tokens=1 the stream terminates after at most one token
        out = engine.generate([GenRequest(prompt_ids=[1, 2], n=1, sampling=greedy(1))])[0]
        assert out.streams[0].finish_reason in ("stop", "length")
        assert len(out.streams[0].token_ids) <= 1

    def test_usage_accounting(self, engine):
        out = engine.generate([GenRequest(prompt_ids=list(range(10)), n=3, sampling=greedy(4))])[0]
        assert out.prompt_tokens == 10
        assert sum(len(s.token_ids) for s in out.streams) <= 3 * 4


class TestBatchedRequests:
    def test_multi_request_batch_matches_single(self, engine):
        r1 = lambda: GenRequest(prompt_ids=[3, 1, 4, 1, 5], n=1, sampling=greedy(6))
        r2 = lambda: GenRequest(prompt_ids=[2, 7, 1, 8], n=1, sampling=greedy(6))
        batch_out = engine.generate([r1(), r2()])
        solo1 = engine.generate([r1()])[0]
        solo2 = engine.generate([r2()])[0]
        assert batch_out[0].streams[0].token_ids == solo1.streams[0].token_ids
        assert batch_out[1].streams[0].token_ids == solo2.streams[0].token_ids


class TestRobustness:
    def test_kv_exhaustion_frees_blocks(self):
        eng = LLMEngine(EngineConfig(
            model="tiny-llama", max_kv_blocks=8, use_hip_graphs=False, device="cpu", seed=0,
        ))
        if eng.prefix_cache is not None:
            eng.prefix_cache.evict_all()
        free0 = eng.kv.allocator.num_free
        with pytest.raises(RuntimeError):
            # 8 blocks * 16 = 128 slots; this needs far more
            eng.generate([GenRequest(prompt_ids=list(range(1, 100)), n=8,
                                     sampling=greedy(64))])
        if eng.prefix_cache is not None:
            eng.prefix_cache.evict_all()
        assert eng.kv.allocator.num_free == free0, "blocks leaked after failure"
        # engine still serves afterwards
        out = eng.generate([GenRequest(prompt_ids=[1, 2, 3], n=1, sampling=greedy(4))])[0]
        assert len(out.streams) == 1

    def test_admission_splits_oversized_batches(self, engine):
        small = EngineConfig(
            model="tiny-llama", max_kv_blocks=512, use_hip_graphs=False,
            device="cpu", seed=0, max_batch_size=4,
        )
        eng = LLMEngine(small)
        reqs = [GenRequest(prompt_ids=[1 + i, 2, 3], n=3, sampling=greedy(4)) for i in range(4)]
        outs = eng.generate(reqs)  # 12 streams > max 4 -> split into sub-batches
        assert len(outs) == 4
        assert all(len(o.streams) == 3 for o in outs)
        # equal to serving them individually
        solo = [eng.generate([GenRequest(prompt_ids=[1 + i, 2, 3], n=3, sampling=greedy(4))])[0]
                for i in range(4)]
        for a, b in zip(outs, solo):
            assert [s.token_ids for s in a.streams] == [s.token_ids for s in b.streams]


class TestFp8KVCache:
    def test_fp8_cache_close_to_bf16(self):
        """Opt-in fp8 KV cache: generation runs and greedy outputs stay close
        to the bf16-cache engine on the same weights (identical for short
        decodes at these magnitudes)."""
        mk = lambda dt: LLMEngine(EngineConfig(
            model="tiny-llama", max_kv_blocks=256, use_hip_graphs=False,
            device="cpu", seed=0, kv_cache_dtype=dt,
        ))
        req = lambda: GenRequest(prompt_ids=list(range(1, 30)), n=1, sampling=greedy(8))
        out_bf16 = mk("bf16").generate([req()])[0]
        eng8 = mk("fp8_e4m3")
        assert eng8.kv.k_all.dtype == __import__("torch").float8_e4m3fn
        out_fp8 = eng8.generate([req()])[0]
        a, b = out_bf16.streams[0].token_ids, out_fp8.streams[0].token_ids
        agree = sum(x == y for x, y in zip(a, b))
        assert agree >= len(a) - 2, f"fp8 cache diverged early: {a} vs {b}"

    def test_per_row_scales_survive_outliers(self):
        """An outlier K/V row beyond e4m3's +-448 would be clamped under a
        static scale; the per-row amax/448 scale must preserve it to e4m3's
        RELATIVE precision, and decode must dequantize with it."""
        import torch
        from kllms_amd.ops import torch_ref

        KVH, D, BS, NB = 2, 128, 16, 8
        kc = torch.zeros(NB, KVH, BS, D, dtype=torch.float8_e4m3fn)
        vc = torch.zeros_like(kc)
        ks = torch.ones(NB, KVH, BS)
        vs = torch.ones_like(ks)
        T = 20
        k = torch.randn(T, KVH, D, dtype=torch.bfloat16) * 0.3
        v = torch.randn(T, KVH, D, dtype=torch.bfloat16) * 0.3
        k[5, 0, 3] = 1200.0
        v[9, 1, 100] = -777.0
        slots = torch.arange(T)
        torch_ref.store_kv(k, v, kc, vc, slots, ks, vs)
        # round-trip of the outliers (e4m3: 3 mantissa bits -> rel err <= 2^-4)
        deq_k = kc.float()[0, 0, 5] * ks[0, 0, 5]
        assert abs(deq_k[3].item() - 1200.0) <= 1200.0 * 0.0625
        deq_v = vc.float()[0, 1, 9] * vs[0, 1, 9]
        assert abs(deq_v[100].item() + 777.0) <= 777.0 * 0.0625
        # non-outlier rows keep a tight absolute error too (own small scale)
        err = (kc.float()[0, 1] * ks[0, 1, :, None] - k[:BS, 1].float()).abs()
        assert err.max().item() <= 0.3 * 4 * 0.0625
        # decode consumes the scales: matches a bf16-cache oracle
        q = torch.randn(2, KVH * 2, D, dtype=torch.bfloat16) * 0.5
        bt = torch.zeros(2, 2, dtype=torch.int32)
        bt[:, 1] = 1
        lens = torch.tensor([T, T - 6], dtype=torch.int32)
        out = torch_ref.attn_decode_paged(q, kc, vc, bt, lens, D ** -0.5, ks, vs)
        kb = torch.zeros(NB, KVH, BS, D, dtype=torch.bfloat16)
        vb = torch.zeros_like(kb)
        torch_ref.store_kv(k, v, kb, vb, slots)
        oracle = torch_ref.attn_decode_paged(q, kb, vb, bt, lens, D ** -0.5)
        assert torch.allclose(out.float(), oracle.float(), rtol=8e-2, atol=8e-2)


class TestPenalties:
    def test_apply_penalties_math(self):
        """OpenAI penalty semantics: logit -= freq_pen * count + pres_pen * present."""
        from kllms_amd.engine.engine import _Stream
        from kllms_amd.engine.kvcache import SequenceKV

        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu"))
        V = eng.arch.vocab_size
        seq = eng.kv.alloc_sequence(4)
        s = _Stream(0, 0, seq, SamplingParams(frequency_penalty=0.5, presence_penalty=0.25), seed=0)
        s.out.token_ids = [7, 7, 9]
        logits = torch.zeros(1, V)
        out = eng._apply_penalties(logits.clone(), [s])
        assert out[0, 7].item() == pytest.approx(-(0.5 * 2 + 0.25))
        assert out[0, 9].item() == pytest.approx(-(0.5 * 1 + 0.25))
        assert out[0, 3].item() == 0.0
        # no-penalty stream short-circuits untouched
        s2 = _Stream(0, 0, seq, SamplingParams(), seed=0)
        s2.out.token_ids = [7]
        out2 = eng._apply_penalties(logits.clone(), [s2])
        assert torch.equal(out2, logits)
        eng.kv.free_sequence(seq)

    def test_frequency_penalty_discourages_repeats(self):
        """Greedy decode with a huge frequency penalty cannot emit the same
        token twice in a row (its logit drops immediately)."""
        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=256,
                                     use_hip_graphs=False, device="cpu", seed=0))
        out = eng.generate([GenRequest(
            prompt_ids=[5, 6, 7], n=1,
            sampling=SamplingParams(temperature=0.0, max_tokens=12, frequency_penalty=100.0))])[0]
        toks = out.streams[0].token_ids
        assert len(toks) == len(set(toks)), f"repeated token under huge penalty: {toks}"


class TestStopStrings:
    def test_stop_string_trims_text(self):
        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=256,
                                     use_hip_graphs=False, device="cpu", seed=0))
        base = eng.generate([GenRequest(
            prompt_ids=[9, 9, 9], n=1,
            sampling=SamplingParams(temperature=0.0, max_tokens=10))])[0]
        text = base.streams[0].text
        assert len(text) > 8
        # pick a substring the greedy continuation definitely contains
        stop = text[4:8]
        out = eng.generate([GenRequest(
            prompt_ids=[9, 9, 9], n=1,
            sampling=SamplingParams(temperature=0.0, max_tokens=10, stop=stop))])[0]
        s = out.streams[0]
        assert s.finish_reason == "stop"
        assert stop not in s.text
        assert text.startswith(s.text)

    def test_stop_list(self):
        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=256,
                                     use_hip_graphs=False, device="cpu", seed=0))
        base = eng.generate([GenRequest(
            prompt_ids=[2, 4, 6], n=1,
            sampling=SamplingParams(temperature=0.0, max_tokens=10))])[0]
        stop2 = base.streams[0].text[3:6]
        out = eng.generate([GenRequest(
            prompt_ids=[2, 4, 6], n=1,
            sampling=SamplingParams(temperature=0.0, max_tokens=10,
                                    stop=["@@never@@", stop2]))])[0]
        assert out.streams[0].finish_reason == "stop"
        assert stop2 not in out.streams[0].text


class TestServingMetadata:
    def test_timings_and_create_many(self):
        from kllms_amd.engine.api import LocalEngineClient

        client = LocalEngineClient(model="tiny-llama", max_kv_blocks=512,
                                   use_hip_graphs=False, device="cpu", seed=0)
        r = client.chat_completions_create(
            messages=[{"role": "user", "content": "hello"}], model="tiny-llama",
            n=2, max_tokens=4, temperature=0.0, seed=1)
        tm = r.timings
        assert tm["completion_tokens"] == r.usage.completion_tokens
        for key in ("prefill_ms", "decode_ms", "decode_steps", "decode_tokens_per_s_batch"):
            assert key in tm, f"missing timing {key}: {tm}"
        assert tm["prefill_ms"] > 0 and tm["decode_ms"] > 0

        # create_many == individual create for seeded greedy requests
        params = [dict(messages=[{"role": "user", "content": f"p{i}"}],
                       model="tiny-llama", n=2, max_tokens=5, temperature=0.0, seed=10 + i)
                  for i in range(3)]
        batch = client.chat_completions_create_many([dict(p) for p in params])
        singles = [client.chat_completions_create(**dict(p)) for p in params]
        for b, s in zip(batch, singles):
            assert [c.message.content for c in b.choices] == [c.message.content for c in s.choices]
            assert b.usage.prompt_tokens == s.usage.prompt_tokens


class TestFp8ScaleProperties:
    def test_round_trip_error_bound_randomized(self):
        """Property: for ANY row magnitudes (including extreme outliers and
        all-zero rows), per-row-scaled fp8 round-trip keeps every element's
        error within e4m3's relative precision of the row amax."""
        import torch
        from kllms_amd.ops import torch_ref

        g = torch.Generator().manual_seed(123)
        KVH, D, BS, NB = 2, 128, 16, 4
        for trial in range(25):
            T = int(torch.randint(1, NB * BS + 1, (1,), generator=g))
            mag = 10.0 ** float(torch.empty(1).uniform_(-3, 5, generator=g))
            k = (torch.randn(T, KVH, D, generator=g) * mag).bfloat16()
            v = (torch.randn(T, KVH, D, generator=g) * mag).bfloat16()
            if trial % 5 == 0:
                k[0, 0].zero_()  # all-zero row: scale floors at 1e-8, no NaN
            kc = torch.zeros(NB, KVH, BS, D, dtype=torch.float8_e4m3fn)
            vc = torch.zeros_like(kc)
            ks = torch.ones(NB, KVH, BS)
            vs = torch.ones_like(ks)
            slots = torch.randperm(NB * BS, generator=g)[:T]
            torch_ref.store_kv(k, v, kc, vc, slots, ks, vs)
            blk, off = slots // BS, slots % BS
            deq = kc.float()[blk, :, off] * ks[blk, :, off, None]
            err = (deq - k.float()).abs()
            # bound: half-ulp at the top of the scaled range = amax * 2^-4,
            # plus the bf16 input's own rounding
            amax = k.float().abs().amax(-1, keepdim=True)
            assert torch.isfinite(deq).all()
            assert (err <= amax * 0.0625 + 1e-6).all(), (trial, err.max(), amax.max())


class TestEmbeddingModes:
    def test_ngram_similarity_is_semantic(self):
        """VERDICT r1 'missing' item 5: under random init the embedder must
        still produce STRING-similarity-meaningful cosines. Signed char-3-gram
        hashing: near-duplicate long strings ≫ unrelated strings."""
        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu", seed=0))
        assert eng._embedding_mode() == "ngram"  # auto + random init
        texts = [
            "The quarterly revenue increased by fifteen percent compared to last year",
            "The quarterly revenue increased by fourteen percent compared to last year",
            "Bananas are an excellent source of potassium and vitamin B6 for athletes",
        ]
        v, total = eng.embed_dev(texts)
        sim = v @ v.T
        assert sim[0, 1] > 0.8 > sim[0, 2], sim
        assert total > 0
        # deterministic; unit rows; zero row for empty text
        v2, _ = eng.embed_dev(texts)
        assert torch.equal(v, v2)
        assert torch.allclose(v.norm(dim=1), torch.ones(3), atol=1e-5)
        ve, _ = eng.embed_dev(["", "abc"])
        assert ve[0].abs().sum() == 0

    def test_mode_override_and_auto_with_weights(self, tmp_path):
        pytest.importorskip("safetensors")
        import sys
        sys.path.insert(0, "tests")
        from test_weights_io import _make_hf_llama_checkpoint

        from kllms_amd.engine.config import MODEL_PRESETS

        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu", seed=0,
                                     embedding_mode="token_mean"))
        assert eng._embedding_mode() == "token_mean"
        v, _ = eng.embed_dev(["hello world"])
        assert v.shape[1] == eng.model.embed_tokens.weight.shape[1]
        # with a real checkpoint on disk, auto resolves to token_mean
        cfg = MODEL_PRESETS["tiny-llama"]
        _make_hf_llama_checkpoint(tmp_path, cfg)
        import json as _json
        (tmp_path / "config.json").write_text(_json.dumps({
            "model_type": "llama", "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size, "intermediate_size": cfg.intermediate_size,
            "num_hidden_layers": cfg.num_layers, "num_attention_heads": cfg.num_heads,
            "num_key_value_heads": cfg.num_kv_heads, "rope_theta": cfg.rope_theta,
            "rms_norm_eps": cfg.rms_norm_eps, "max_position_embeddings": 512,
        }))
        eng2 = LLMEngine(EngineConfig(model=str(tmp_path), max_kv_blocks=64,
                                      use_hip_graphs=False, device="cpu", seed=0))
        assert eng2._embedding_mode() == "token_mean"

    def test_model_dir_loads_weights_without_explicit_path(self, tmp_path):
        """README contract: model may be a PATH to a weights dir — weights
        must actually load (previously random-init unless weights_path set)."""
        pytest.importorskip("safetensors")
        import sys
        sys.path.insert(0, "tests")
        from test_weights_io import _make_hf_llama_checkpoint

        from kllms_amd.engine.config import MODEL_PRESETS

        cfg = MODEL_PRESETS["tiny-llama"]
        tensors = _make_hf_llama_checkpoint(tmp_path, cfg)
        import json as _json
        (tmp_path / "config.json").write_text(_json.dumps({
            "model_type": "llama", "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size, "intermediate_size": cfg.intermediate_size,
            "num_hidden_layers": cfg.num_layers, "num_attention_heads": cfg.num_heads,
            "num_key_value_heads": cfg.num_kv_heads, "rope_theta": cfg.rope_theta,
            "rms_norm_eps": cfg.rms_norm_eps, "max_position_embeddings": 512,
        }))
        eng = LLMEngine(EngineConfig(model=str(tmp_path), max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu", seed=0))
        got = eng.model.embed_tokens.weight.float()
        assert torch.allclose(got, tensors["model.embed_tokens.weight"].to(got.dtype).float())


class TestMultiEos:
    def test_config_list_and_generation_config(self, tmp_path):
        """A checkpoint's generation_config.json eos_token_id list (Llama-3
        style) makes the engine stop on ANY of the ids."""
        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu", seed=0,
                                     eos_token_id=[7, 9]))
        assert eng.eos_token_id == 7 and eng.eos_token_ids == frozenset({7, 9})

        # generation_config.json next to a weights dir is honored
        pytest.importorskip("safetensors")
        import json as _json
        import sys
        sys.path.insert(0, "tests")
        from test_weights_io import _make_hf_llama_checkpoint

        from kllms_amd.engine.config import MODEL_PRESETS
        cfg = MODEL_PRESETS["tiny-llama"]
        _make_hf_llama_checkpoint(tmp_path, cfg)
        (tmp_path / "config.json").write_text(_json.dumps({
            "model_type": "llama", "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size, "intermediate_size": cfg.intermediate_size,
            "num_hidden_layers": cfg.num_layers, "num_attention_heads": cfg.num_heads,
            "num_key_value_heads": cfg.num_kv_heads, "max_position_embeddings": 512,
        }))
        (tmp_path / "generation_config.json").write_text(_json.dumps({
            "eos_token_id": [11, 23]}))
        eng2 = LLMEngine(EngineConfig(model=str(tmp_path), max_kv_blocks=64,
                                      use_hip_graphs=False, device="cpu", seed=0))
        assert eng2.eos_token_ids == frozenset({11, 23})

    def test_secondary_eos_stops_stream(self):
        """Force the model's greedy next token to be a SECONDARY eos id and
        check the stream finishes with reason 'stop'."""
        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu", seed=0))
        probe = eng.generate([GenRequest(prompt_ids=[1, 2, 3], n=1, sampling=greedy(1))])[0]
        tok = probe.streams[0].token_ids
        if not tok:
            pytest.skip("first greedy token was already eos")
        eng2 = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=64,
                                      use_hip_graphs=False, device="cpu", seed=0,
                                      eos_token_id=[eng.eos_token_id, tok[0]]))
        out = eng2.generate([GenRequest(prompt_ids=[1, 2, 3], n=1, sampling=greedy(8))])[0]
        assert out.streams[0].finish_reason == "stop"
        assert out.streams[0].token_ids == []  # eos content excluded


class TestRopeScaling:
    def test_llama3_banded_scaling_formula(self):
        """Llama-3.1 rope_scaling: low-frequency bands /factor, high-frequency
        untouched, smooth interpolation between — checked against a direct
        transcription of the published formula."""
        import math

        from kllms_amd.ops.torch_ref import build_cos_sin_cache

        D, P, theta = 128, 64, 500000.0
        rs = {"rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
              "high_freq_factor": 4.0, "original_max_position_embeddings": 8192}
        got = build_cos_sin_cache(D, P, theta, "cpu", rope_scaling=rs)
        plain = build_cos_sin_cache(D, P, theta, "cpu")
        assert not torch.equal(got, plain)

        half = D // 2
        inv = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64) / half))
        out = []
        for f in inv.tolist():
            wl = 2 * math.pi / f
            if wl > 8192 / 1.0:
                out.append(f / 8.0)
            elif wl < 8192 / 4.0:
                out.append(f)
            else:
                s = (8192 / wl - 1.0) / (4.0 - 1.0)
                out.append((1 - s) * f / 8.0 + s * f)
        t = torch.arange(P, dtype=torch.float64)
        freqs = torch.outer(t, torch.tensor(out, dtype=torch.float64))
        want = torch.cat([freqs.cos(), freqs.sin()], dim=-1).float()
        assert torch.allclose(got, want, atol=1e-6)

        # high-frequency (early) bands must be IDENTICAL to unscaled
        hi_cols = [i for i, f in enumerate(inv.tolist()) if 2 * math.pi / f < 8192 / 4.0]
        assert hi_cols, "expected some high-frequency bands"
        assert torch.allclose(got[:, hi_cols], plain[:, hi_cols], atol=1e-7)

    def test_linear_scaling_and_engine_wiring(self, tmp_path):
        from kllms_amd.ops.torch_ref import build_cos_sin_cache

        lin = build_cos_sin_cache(64, 32, 10000.0, "cpu",
                                  rope_scaling={"type": "linear", "factor": 2.0})
        plain = build_cos_sin_cache(64, 64, 10000.0, "cpu")
        # position 2p under factor-2 linear == position p unscaled
        assert torch.allclose(lin[2], plain[1], atol=1e-6)

        # arch config threads rope_scaling into the model's cache
        from kllms_amd.engine.config import MODEL_PRESETS
        from kllms_amd.models.llama import LlamaForCausalLM
        from kllms_amd.parallel.tp import ParallelContext
        cfg = MODEL_PRESETS["tiny-llama"].model_copy(update={
            "rope_scaling": {"type": "linear", "factor": 2.0}})
        m = LlamaForCausalLM(cfg, ParallelContext(), dtype=torch.float32)
        m2 = LlamaForCausalLM(MODEL_PRESETS["tiny-llama"], ParallelContext(), dtype=torch.float32)
        assert not torch.equal(m.cos_sin, m2.cos_sin)
