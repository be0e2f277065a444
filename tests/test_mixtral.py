"""Mixtral MoE tests: router semantics, TP-sharded experts, engine e2e
(CPU tiny preset; the GPU e2e lives in test_gpu_kernels-style gpu tests)."""

import pytest
import torch

from kllms_amd.engine.config import MODEL_PRESETS, EngineConfig
from kllms_amd.engine.engine import GenRequest, LLMEngine
from kllms_amd.engine.sampling import SamplingParams
from kllms_amd.models.mixtral import MixtralForCausalLM, MixtralMoE
from kllms_amd.parallel.tp import ParallelContext


@pytest.fixture(scope="module")
def engine():
    return LLMEngine(EngineConfig(model="tiny-mixtral", max_kv_blocks=256, use_hip_graphs=False, device="cpu", seed=1))


class TestRouter:
    def test_topk_renormalized_weights(self):
        cfg = MODEL_PRESETS["tiny-mixtral"]
        moe = MixtralMoE(cfg, ParallelContext(), dtype=torch.float32)
        torch.manual_seed(0)
        moe.gate.weight.copy_(torch.randn_like(moe.gate.weight) * 0.1)
        moe.w_gate_up.copy_(torch.randn_like(moe.w_gate_up) * 0.05)
        moe.w_down.copy_(torch.randn_like(moe.w_down) * 0.05)
        x = torch.randn(5, cfg.hidden_size)
        out = moe(x)
        assert out.shape == x.shape
        assert torch.isfinite(out).all()

    def test_moe_matches_dense_expert_math(self):
        """With all experts identical, MoE output == single expert MLP output
        (weights renormalize to 1)."""
        cfg = MODEL_PRESETS["tiny-mixtral"]
        moe = MixtralMoE(cfg, ParallelContext(), dtype=torch.float32)
        torch.manual_seed(1)
        moe.gate.weight.copy_(torch.randn_like(moe.gate.weight))
        w_gu = torch.randn(2 * moe.I, cfg.hidden_size) * 0.1
        w_d = torch.randn(cfg.hidden_size, moe.I) * 0.1
        for e in range(moe.E):
            moe.w_gate_up.data[e].copy_(w_gu)
            moe.w_down.data[e].copy_(w_d)
        x = torch.randn(7, cfg.hidden_size)
        out = moe(x)

        from kllms_amd import ops

        gu = torch.nn.functional.linear(x, w_gu)
        g, u = gu.split([moe.I, moe.I], dim=-1)
        ref = torch.nn.functional.linear(ops.silu_mul(g.contiguous(), u.contiguous()), w_d)
        assert torch.allclose(out, ref, atol=1e-4)


class TestMixtralEngine:
    def test_generate_end_to_end(self, engine):
        out = engine.generate([
            GenRequest(prompt_ids=list(range(1, 40)), n=3,
                       sampling=SamplingParams(temperature=1.0, max_tokens=10, seed=3))
        ])[0]
        assert len(out.streams) == 3
        assert all(len(s.token_ids) > 0 for s in out.streams)

    def test_greedy_deterministic(self, engine):
        mk = lambda: GenRequest(prompt_ids=[1, 2, 3], n=1, sampling=SamplingParams(temperature=0.0, max_tokens=6))
        o1 = engine.generate([mk()])[0]
        o2 = engine.generate([mk()])[0]
        assert o1.streams[0].token_ids == o2.streams[0].token_ids


def test_embed_batched_matches_per_text(engine):
    texts = ["hello world", "", "a much longer text with more tokens in it"]
    vecs, total = engine.embed(texts)
    assert len(vecs) == 3
    assert total == sum(len(engine.tokenizer.encode(t)) for t in texts)
    solo, _ = engine.embed([texts[2]])
    assert vecs[2] == pytest.approx(solo[0], abs=1e-6)
    assert vecs[1] == [0.0] * len(vecs[1])


def test_dense_path_matches_grouped():
    """The decode-shape dense-all-experts path and the bucketed grouped path
    must agree numerically."""
    cfg = MODEL_PRESETS["tiny-mixtral"]
    moe = MixtralMoE(cfg, ParallelContext(), dtype=torch.float32)
    torch.manual_seed(3)
    moe.gate.weight.copy_(torch.randn_like(moe.gate.weight) * 0.2)
    moe.w_gate_up.copy_(torch.randn_like(moe.w_gate_up) * 0.1)
    moe.w_down.copy_(torch.randn_like(moe.w_down) * 0.1)
    x = torch.randn(11, cfg.hidden_size)
    logits = moe.gate(x).float()
    probs = torch.softmax(logits, dim=-1)
    topw, topi = torch.topk(probs, moe.k, dim=-1)
    topw = topw / topw.sum(dim=-1, keepdim=True)
    dense = moe._forward_dense(x, topw, topi)
    grouped = moe._forward_grouped(x, topw, topi)
    assert torch.allclose(dense, grouped, atol=1e-4), (dense - grouped).abs().max()


def test_mixtral_fp8_cache_and_prefix_reuse():
    """Mixtral through the fp8 KV cache (per-row scales) AND the prefix
    cache: a second request sharing the prompt head must hit the cache and
    still produce the same greedy tokens as a cold engine."""
    from kllms_amd.engine.config import EngineConfig
    from kllms_amd.engine.engine import GenRequest, LLMEngine
    from kllms_amd.engine.sampling import SamplingParams

    import torch

    mk = lambda: LLMEngine(EngineConfig(
        model="tiny-mixtral", max_kv_blocks=256, use_hip_graphs=False,
        device="cpu", seed=0, kv_cache_dtype="fp8_e4m3",
        prefix_cache_min_tokens=16))
    eng = mk()
    assert eng.kv.fp8 and eng.kv.k_scale_all is not None
    head = list(range(1, 40))  # > 2 full blocks
    g = lambda tail: GenRequest(prompt_ids=head + tail, n=1,
                                sampling=SamplingParams(temperature=0.0, max_tokens=6))
    out1 = eng.generate([g([41, 42])])[0]
    h0 = eng.prefix_cache.hits
    out2 = eng.generate([g([43, 44])])[0]
    assert eng.prefix_cache.hits > h0, "second request must hit the prefix cache"
    # cold engine equivalence for the second prompt
    cold = mk().generate([g([43, 44])])[0]
    assert out2.streams[0].token_ids == cold.streams[0].token_ids
