"""Qwen2-family support: Llama compute graph + fused-QKV bias (column-
sharded with the weight, no extra communication) + tied embeddings.

The reference serves whatever model the API names; widening the local
engine's architecture coverage is the corresponding parity axis."""

import multiprocessing as mp
import os

import pytest
import torch

from kllms_amd.engine.config import MODEL_PRESETS, EngineConfig
from kllms_amd.engine.engine import GenRequest, LLMEngine
from kllms_amd.engine.sampling import SamplingParams
from kllms_amd.models.llama import LlamaForCausalLM
from kllms_amd.parallel.tp import ParallelContext

PORT = 29817


def greedy(max_tokens=8):
    return SamplingParams(temperature=0.0, max_tokens=max_tokens, seed=0)


class TestQwenArch:
    def test_bias_and_tie_in_random_init(self):
        cfg = MODEL_PRESETS["tiny-qwen"]
        model = LlamaForCausalLM(cfg, ParallelContext(), dtype=torch.float32)
        model.random_init_(0)
        b = model.layers[0].self_attn.qkv_proj.bias
        assert b is not None and b.abs().max() > 0
        assert torch.equal(model.lm_head.weight, model.embed_tokens.weight)
        # plain Llama has no bias parameter at all
        plain = LlamaForCausalLM(MODEL_PRESETS["tiny-llama"], ParallelContext(), dtype=torch.float32)
        assert plain.layers[0].self_attn.qkv_proj.bias is None

    def test_bias_changes_logits(self):
        """The bias must actually reach the computation."""
        from kllms_amd.models.llama import ForwardBatch

        eng = LLMEngine(EngineConfig(model="tiny-qwen", max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu", seed=0))
        ids = list(range(1, 12))

        def logits_once():
            seq = eng.kv.alloc_sequence(len(ids))
            batch = ForwardBatch(
                mode="prefill",
                positions=torch.arange(len(ids)),
                slot_mapping=torch.tensor(eng.kv.prefill_slot_mapping(seq)),
                kv_caches=eng.kv.layer_caches(),
                cu_seqlens=torch.tensor([0, len(ids)], dtype=torch.int32),
            )
            out = eng.model.forward_prefill(torch.tensor(ids), batch)
            eng.kv.free_sequence(seq)
            return out

        base = logits_once()
        with torch.no_grad():
            for layer in eng.model.layers:
                layer.self_attn.qkv_proj.bias.zero_()
        zeroed = logits_once()
        assert not torch.allclose(base, zeroed)

    def test_engine_generates(self):
        eng = LLMEngine(EngineConfig(model="tiny-qwen", max_kv_blocks=128,
                                     use_hip_graphs=False, device="cpu", seed=0))
        out = eng.generate([GenRequest(prompt_ids=list(range(1, 20)), n=3, sampling=greedy())])[0]
        assert len(out.streams) == 3
        assert out.streams[0].token_ids == out.streams[1].token_ids  # greedy


class TestQwenCheckpoint:
    def test_load_fused_bias_and_tied_head(self, tmp_path):
        pytest.importorskip("safetensors")
        from kllms_amd.engine.weights import load_safetensors_weights

        from test_weights_io import _make_hf_llama_checkpoint

        cfg = MODEL_PRESETS["tiny-qwen"]
        tensors = _make_hf_llama_checkpoint(tmp_path, cfg)
        model = LlamaForCausalLM(cfg, ParallelContext(), dtype=torch.float32)
        load_safetensors_weights(model, str(tmp_path), ParallelContext())
        want = torch.cat([
            tensors["model.layers.0.self_attn.q_proj.bias"],
            tensors["model.layers.0.self_attn.k_proj.bias"],
            tensors["model.layers.0.self_attn.v_proj.bias"],
        ], dim=0)
        assert torch.equal(model.layers[0].self_attn.qkv_proj.bias, want)
        # tied: lm_head mirrors embed_tokens (no lm_head.weight in files)
        assert torch.equal(model.lm_head.weight, tensors["model.embed_tokens.weight"])

    def test_bias_checkpoint_into_biasless_model_raises(self, tmp_path):
        pytest.importorskip("safetensors")
        from kllms_amd.engine.weights import load_safetensors_weights

        from test_weights_io import _make_hf_llama_checkpoint

        qcfg = MODEL_PRESETS["tiny-qwen"]
        _make_hf_llama_checkpoint(tmp_path, qcfg)
        plain = MODEL_PRESETS["tiny-llama"].model_copy(
            update={"tie_word_embeddings": True})  # same shapes, no bias
        model = LlamaForCausalLM(plain, ParallelContext(), dtype=torch.float32)
        with pytest.raises(RuntimeError, match="attention_qkv_bias"):
            load_safetensors_weights(model, str(tmp_path), ParallelContext())


def _qwen_tp_worker(rank: int, world_size: int, q):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(PORT)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        ctx = ParallelContext(world_size=world_size, rank=rank)
        eng = LLMEngine(
            EngineConfig(model="tiny-qwen", tp_size=world_size, max_kv_blocks=128,
                         use_hip_graphs=False, device="cpu", seed=0),
            parallel_ctx=ctx,
        )
        out = eng.generate([GenRequest(prompt_ids=list(range(1, 25)), n=2,
                                       sampling=greedy())])[0]
        q.put((rank, out.streams[0].token_ids))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_qwen_tp2_matches_tp1():
    """The sharded QKV bias must keep TP=2 numerically identical to TP=1
    (TP-degree-invariant init generates the FULL bias then slices)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_qwen_tp_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, tokens = q.get(timeout=240)
        results[rank] = tokens
    for p in procs:
        p.join(timeout=60)
    assert results[0] == results[1], "qwen TP ranks diverged"

    eng = LLMEngine(EngineConfig(model="tiny-qwen", max_kv_blocks=128,
                                 use_hip_graphs=False, device="cpu", seed=0))
    out = eng.generate([GenRequest(prompt_ids=list(range(1, 25)), n=2, sampling=greedy())])[0]
    assert out.streams[0].token_ids == results[0], "qwen TP=2 diverged from TP=1"
