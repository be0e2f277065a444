"""Cross-request prefix caching (kvcache.PrefixCache).

Correctness bar: a prefix-cache hit must be OUTPUT-INVISIBLE — identical
greedy tokens and logprobs vs an engine with caching disabled — while
skipping the re-prefill of shared full blocks."""

import pytest
import torch

from kllms_amd.engine.config import EngineConfig
from kllms_amd.engine.engine import GenRequest, LLMEngine
from kllms_amd.engine.kvcache import PagedKVCache, PrefixCache
from kllms_amd.engine.sampling import SamplingParams


def _cfg(**kw):
    base = dict(model="tiny-llama", max_kv_blocks=256, use_hip_graphs=False,
                default_max_new_tokens=8, device="cpu", seed=0,
                prefix_cache_min_tokens=16)
    base.update(kw)
    return EngineConfig(**base)


def greedy(n=8):
    return SamplingParams(temperature=0.0, max_tokens=n)


SHARED = list(range(1, 67))  # 66 tokens = 4 full blocks + tail at bs=16


class TestPrefixCacheUnit:
    def _kv(self, blocks=64):
        return PagedKVCache(2, 2, 32, 16, blocks, torch.device("cpu"), torch.float32)

    def test_match_register_roundtrip(self):
        kv = self._kv()
        pc = PrefixCache(kv, max_blocks=16)
        ids = list(range(40))  # 2 full blocks + partial
        assert pc.match(ids) == []
        seq = kv.alloc_sequence(len(ids))
        pc.register(ids, seq)
        m = pc.match(ids)
        assert m == seq.blocks[:2]
        # a longer prompt with the same head matches the same chain
        m2 = pc.match(ids + [99] * 30)
        assert m2 == seq.blocks[:2]
        # a diverging second block matches only the first
        div = ids[:16] + [777] * 24
        assert pc.match(div) == seq.blocks[:1]
        # blocks survive the sequence being freed (cache holds a ref)
        kv.free_sequence(seq)
        assert kv.allocator.refcount(m[0]) == 1

    def test_never_matches_whole_prompt(self):
        kv = self._kv()
        pc = PrefixCache(kv, max_blocks=16)
        ids = list(range(32))  # exactly 2 blocks
        seq = kv.alloc_sequence(len(ids))
        pc.register(ids, seq)
        # at most 1 block may match: >=1 token must remain for prefill logits
        assert len(pc.match(ids)) == 1

    def test_lru_eviction_bound(self):
        kv = self._kv()
        pc = PrefixCache(kv, max_blocks=3)
        for base in range(5):
            ids = [1000 * base + i for i in range(33)]
            seq = kv.alloc_sequence(len(ids))
            pc.register(ids, seq)
            kv.free_sequence(seq)
        assert len(pc._map) == 3

    def test_pressure_eviction_frees_blocks(self):
        kv = self._kv(blocks=8)
        pc = PrefixCache(kv, max_blocks=8)
        ids = list(range(64))  # 4 blocks
        seq = kv.alloc_sequence(len(ids))
        pc.register(ids, seq)
        kv.free_sequence(seq)  # cache now sole owner of 4 blocks
        assert kv.allocator.num_free == 4
        # allocating past the free pool triggers on_pressure -> cache drains
        got = [kv.allocator.alloc() for _ in range(8)]
        assert len(got) == 8
        assert len(pc._map) == 0


class TestPrefixCacheEngine:
    def test_hit_outputs_match_uncached(self):
        eng_c = LLMEngine(_cfg(enable_prefix_caching=True))
        eng_u = LLMEngine(_cfg(enable_prefix_caching=False))

        reqs = lambda suffix: [GenRequest(prompt_ids=SHARED + suffix, n=2, sampling=greedy())]
        # warm the cache with one request, then serve a same-prefix request
        eng_c.generate(reqs([70, 71, 72]))
        assert eng_c.prefix_cache.stats["entries"] > 0
        out_c = eng_c.generate(reqs([80, 81, 82, 83]))[0]
        hits_before = eng_c.prefix_cache.hits
        assert hits_before >= 1

        eng_u.generate(reqs([70, 71, 72]))
        out_u = eng_u.generate(reqs([80, 81, 82, 83]))[0]

        for sc, su in zip(out_c.streams, out_u.streams):
            assert sc.token_ids == su.token_ids
            assert sc.logprobs == pytest.approx(su.logprobs, abs=1e-4)

    def test_hit_and_miss_mixed_batch(self):
        eng = LLMEngine(_cfg())
        eng.generate([GenRequest(prompt_ids=SHARED + [7], n=1, sampling=greedy())])
        outs = eng.generate([
            GenRequest(prompt_ids=SHARED + [8, 9], n=2, sampling=greedy()),     # hit
            GenRequest(prompt_ids=[500 - i for i in range(50)], n=1, sampling=greedy()),  # miss
        ])
        ref = LLMEngine(_cfg(enable_prefix_caching=False)).generate([
            GenRequest(prompt_ids=SHARED + [8, 9], n=2, sampling=greedy()),
            GenRequest(prompt_ids=[500 - i for i in range(50)], n=1, sampling=greedy()),
        ])
        for o, r in zip(outs, ref):
            for so, sr in zip(o.streams, r.streams):
                assert so.token_ids == sr.token_ids

    def test_decode_never_corrupts_cached_blocks(self):
        """A stream decoding PAST a cached block boundary must CoW, leaving
        the cached KV intact for later matches."""
        eng = LLMEngine(_cfg())
        # prompt ends exactly at a block boundary: first decode token would
        # write into a fresh block (not the cached one)
        ids = list(range(2, 50))  # 48 = 3 full blocks
        eng.generate([GenRequest(prompt_ids=ids, n=1, sampling=greedy(12))])
        out_a = eng.generate([GenRequest(prompt_ids=ids + [55], n=1, sampling=greedy(12))])[0]
        # same request again — now fully against cached prefix KV
        out_b = eng.generate([GenRequest(prompt_ids=ids + [55], n=1, sampling=greedy(12))])[0]
        assert out_a.streams[0].token_ids == out_b.streams[0].token_ids

    def test_scheduler_chunked_prefill_uses_prefix(self):
        from kllms_amd.engine.scheduler import BatchScheduler

        eng = LLMEngine(_cfg(prefill_chunk_tokens=24, max_kv_blocks=512))
        sched = BatchScheduler(eng)
        long_ids = SHARED + [200 + i for i in range(60)]   # > chunk => chunked
        f1 = sched.submit(GenRequest(prompt_ids=long_ids, n=1, sampling=greedy(4)))
        f1.result(60)
        f2 = sched.submit(GenRequest(prompt_ids=long_ids[:100] + [499, 498], n=1, sampling=greedy(4)))
        out2 = f2.result(60)
        sched.shutdown()
        assert eng.prefix_cache.hits >= 1
        # parity with uncached engine
        eng_u = LLMEngine(_cfg(enable_prefix_caching=False, prefill_chunk_tokens=24, max_kv_blocks=512))
        ref = eng_u.generate([GenRequest(prompt_ids=long_ids[:100] + [499, 498], n=1, sampling=greedy(4))])[0]
        assert out2.streams[0].token_ids == ref.streams[0].token_ids

    def test_tokens_saved_accounting(self):
        eng = LLMEngine(_cfg())
        eng.generate([GenRequest(prompt_ids=SHARED, n=1, sampling=greedy())])
        eng.generate([GenRequest(prompt_ids=SHARED + [5], n=1, sampling=greedy())])
        assert eng.prefix_cache.stats["tokens_saved"] >= 64
