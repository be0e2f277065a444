"""Weight-loading tests: HF-style safetensors -> fused/sharded layout, and
the HF tokenizers-file path."""

import json
import os

import pytest
import torch

from kllms_amd.engine.config import MODEL_PRESETS
from kllms_amd.engine.weights import load_safetensors_weights
from kllms_amd.models.llama import LlamaForCausalLM
from kllms_amd.parallel.tp import ParallelContext


def _make_hf_llama_checkpoint(tmp_path, cfg, seed=0):
    """Write a random HF-style (unfused q/k/v, gate/up) checkpoint.
    Honors cfg.attention_qkv_bias (Qwen2-style q/k/v bias tensors) and
    cfg.tie_word_embeddings (no lm_head.weight tensor in the files)."""
    from safetensors.torch import save_file

    g = torch.Generator().manual_seed(seed)
    H = cfg.hidden_size
    D = cfg.head_dim_
    tensors = {
        "model.embed_tokens.weight": torch.randn(cfg.vocab_size, H, generator=g),
        "model.norm.weight": torch.randn(H, generator=g),
    }
    if not cfg.tie_word_embeddings:
        tensors["lm_head.weight"] = torch.randn(cfg.vocab_size, H, generator=g)
    for L in range(cfg.num_layers):
        p = f"model.layers.{L}"
        tensors[f"{p}.self_attn.q_proj.weight"] = torch.randn(cfg.num_heads * D, H, generator=g)
        tensors[f"{p}.self_attn.k_proj.weight"] = torch.randn(cfg.num_kv_heads * D, H, generator=g)
        tensors[f"{p}.self_attn.v_proj.weight"] = torch.randn(cfg.num_kv_heads * D, H, generator=g)
        if cfg.attention_qkv_bias:
            tensors[f"{p}.self_attn.q_proj.bias"] = torch.randn(cfg.num_heads * D, generator=g)
            tensors[f"{p}.self_attn.k_proj.bias"] = torch.randn(cfg.num_kv_heads * D, generator=g)
            tensors[f"{p}.self_attn.v_proj.bias"] = torch.randn(cfg.num_kv_heads * D, generator=g)
        tensors[f"{p}.self_attn.o_proj.weight"] = torch.randn(H, cfg.num_heads * D, generator=g)
        tensors[f"{p}.mlp.gate_proj.weight"] = torch.randn(cfg.intermediate_size, H, generator=g)
        tensors[f"{p}.mlp.up_proj.weight"] = torch.randn(cfg.intermediate_size, H, generator=g)
        tensors[f"{p}.mlp.down_proj.weight"] = torch.randn(H, cfg.intermediate_size, generator=g)
        tensors[f"{p}.input_layernorm.weight"] = torch.randn(H, generator=g)
        tensors[f"{p}.post_attention_layernorm.weight"] = torch.randn(H, generator=g)
    save_file({k: v.contiguous() for k, v in tensors.items()}, str(tmp_path / "model.safetensors"))
    return tensors


def test_safetensors_load_fuses_correctly(tmp_path):
    pytest.importorskip("safetensors")
    cfg = MODEL_PRESETS["tiny-llama"]
    tensors = _make_hf_llama_checkpoint(tmp_path, cfg)

    model = LlamaForCausalLM(cfg, ParallelContext(), dtype=torch.float32)
    load_safetensors_weights(model, str(tmp_path), ParallelContext())

    # fused qkv = [q; k; v]
    got = model.layers[0].self_attn.qkv_proj.weight
    want = torch.cat([
        tensors["model.layers.0.self_attn.q_proj.weight"],
        tensors["model.layers.0.self_attn.k_proj.weight"],
        tensors["model.layers.0.self_attn.v_proj.weight"],
    ], dim=0)
    assert torch.equal(got, want)
    # fused gate_up = [gate; up]
    got = model.layers[1].mlp.gate_up_proj.weight
    want = torch.cat([
        tensors["model.layers.1.mlp.gate_proj.weight"],
        tensors["model.layers.1.mlp.up_proj.weight"],
    ], dim=0)
    assert torch.equal(got, want)
    assert torch.equal(model.lm_head.weight, tensors["lm_head.weight"])
    assert torch.equal(model.norm.weight, tensors["model.norm.weight"])


def test_engine_boots_from_weights_dir(tmp_path):
    pytest.importorskip("safetensors")
    cfg = MODEL_PRESETS["tiny-llama"]
    _make_hf_llama_checkpoint(tmp_path, cfg)
    # HF-style config.json so resolve_arch() reads the architecture
    (tmp_path / "config.json").write_text(json.dumps({
        "model_type": "llama",
        "vocab_size": cfg.vocab_size,
        "hidden_size": cfg.hidden_size,
        "intermediate_size": cfg.intermediate_size,
        "num_hidden_layers": cfg.num_layers,
        "num_attention_heads": cfg.num_heads,
        "num_key_value_heads": cfg.num_kv_heads,
        "rope_theta": cfg.rope_theta,
        "rms_norm_eps": cfg.rms_norm_eps,
        "max_position_embeddings": cfg.max_position_embeddings,
    }))

    from kllms_amd.engine.config import EngineConfig
    from kllms_amd.engine.engine import GenRequest, LLMEngine
    from kllms_amd.engine.sampling import SamplingParams

    eng = LLMEngine(EngineConfig(
        model=str(tmp_path), weights_path=str(tmp_path),
        max_kv_blocks=128, use_hip_graphs=False, device="cpu",
    ))
    out = eng.generate([GenRequest(prompt_ids=[1, 2, 3], n=1,
                                   sampling=SamplingParams(temperature=0.0, max_tokens=4))])[0]
    assert len(out.streams[0].token_ids) > 0


def test_hf_tokenizer_roundtrip(tmp_path):
    tokenizers = pytest.importorskip("tokenizers")
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    trainer = trainers.BpeTrainer(special_tokens=["<unk>", "<|begin_of_text|>", "<|eot_id|>"], vocab_size=200)
    tok.train_from_iterator(["hello world", "the quick brown fox", "hello fox"] * 20, trainer)
    path = tmp_path / "tokenizer.json"
    tok.save(str(path))

    from kllms_amd.engine.tokenizer import HFTokenizer, load_tokenizer

    ht = HFTokenizer(str(path))
    ids = ht.encode("hello world")
    assert ids
    assert "hello" in ht.decode(ids)
    assert ht.crop_to_tokens("hello world", 1)  # crops without error
    assert ht.token_bytes(ids[0])

    # load_tokenizer picks the file up from a weights dir
    lt = load_tokenizer(str(tmp_path), None, 512)
    assert isinstance(lt, HFTokenizer)


def test_safetensors_tp2_shards(tmp_path):
    """TP=2 loading: each rank's fused qkv holds its q/k/v HEAD shards (not a
    contiguous slice of the fused dim)."""
    pytest.importorskip("safetensors")
    cfg = MODEL_PRESETS["tiny-llama"]
    tensors = _make_hf_llama_checkpoint(tmp_path, cfg, seed=5)
    D = cfg.head_dim_
    q_full = tensors["model.layers.0.self_attn.q_proj.weight"]
    k_full = tensors["model.layers.0.self_attn.k_proj.weight"]
    v_full = tensors["model.layers.0.self_attn.v_proj.weight"]

    for rank in (0, 1):
        ctx = ParallelContext(world_size=2, rank=rank)
        model = LlamaForCausalLM(cfg, ctx, dtype=torch.float32)
        load_safetensors_weights(model, str(tmp_path), ctx)
        got = model.layers[0].self_attn.qkv_proj.weight
        qh = cfg.num_heads // 2 * D
        kh = cfg.num_kv_heads // 2 * D
        want = torch.cat([
            q_full[rank * qh:(rank + 1) * qh],
            k_full[rank * kh:(rank + 1) * kh],
            v_full[rank * kh:(rank + 1) * kh],
        ], dim=0)
        assert torch.equal(got, want), f"rank {rank} qkv shard mismatch"
        # row-parallel down_proj: columns sharded
        down_full = tensors["model.layers.0.mlp.down_proj.weight"]
        got_down = model.layers[0].mlp.down_proj.weight
        half = cfg.intermediate_size // 2
        assert torch.equal(got_down, down_full[:, rank * half:(rank + 1) * half])


def _make_hf_mixtral_checkpoint(tmp_path, cfg, seed=0):
    from safetensors.torch import save_file

    g = torch.Generator().manual_seed(seed)
    H, D, I = cfg.hidden_size, cfg.head_dim_, cfg.intermediate_size
    tensors = {
        "model.embed_tokens.weight": torch.randn(cfg.vocab_size, H, generator=g),
        "model.norm.weight": torch.randn(H, generator=g),
        "lm_head.weight": torch.randn(cfg.vocab_size, H, generator=g),
    }
    for L in range(cfg.num_layers):
        p = f"model.layers.{L}"
        tensors[f"{p}.self_attn.q_proj.weight"] = torch.randn(cfg.num_heads * D, H, generator=g)
        tensors[f"{p}.self_attn.k_proj.weight"] = torch.randn(cfg.num_kv_heads * D, H, generator=g)
        tensors[f"{p}.self_attn.v_proj.weight"] = torch.randn(cfg.num_kv_heads * D, H, generator=g)
        tensors[f"{p}.self_attn.o_proj.weight"] = torch.randn(H, cfg.num_heads * D, generator=g)
        tensors[f"{p}.block_sparse_moe.gate.weight"] = torch.randn(cfg.num_experts, H, generator=g)
        for e in range(cfg.num_experts):
            q = f"{p}.block_sparse_moe.experts.{e}"
            tensors[f"{q}.w1.weight"] = torch.randn(I, H, generator=g)   # gate
            tensors[f"{q}.w2.weight"] = torch.randn(H, I, generator=g)   # down
            tensors[f"{q}.w3.weight"] = torch.randn(I, H, generator=g)   # up
        tensors[f"{p}.input_layernorm.weight"] = torch.randn(H, generator=g)
        tensors[f"{p}.post_attention_layernorm.weight"] = torch.randn(H, generator=g)
    save_file({k: v.contiguous() for k, v in tensors.items()}, str(tmp_path / "model.safetensors"))
    return tensors


def test_mixtral_expert_tp2_shards(tmp_path):
    """TP=2 Mixtral: each rank's stacked w_gate_up holds [gate_shard; up_shard]
    per expert (per-part sharding, matching the TP-invariant random init)."""
    pytest.importorskip("safetensors")
    from kllms_amd.models.mixtral import MixtralForCausalLM

    cfg = MODEL_PRESETS["mid-mixtral"]
    tensors = _make_hf_mixtral_checkpoint(tmp_path, cfg, seed=3)
    I = cfg.intermediate_size
    half = I // 2
    for rank in (0, 1):
        ctx = ParallelContext(world_size=2, rank=rank)
        model = MixtralForCausalLM(cfg, ctx, dtype=torch.float32)
        load_safetensors_weights(model, str(tmp_path), ctx)
        moe = model.layers[0].mlp
        for e in range(cfg.num_experts):
            gate_full = tensors[f"model.layers.0.block_sparse_moe.experts.{e}.w1.weight"]
            up_full = tensors[f"model.layers.0.block_sparse_moe.experts.{e}.w3.weight"]
            down_full = tensors[f"model.layers.0.block_sparse_moe.experts.{e}.w2.weight"]
            want_gu = torch.cat([
                gate_full[rank * half:(rank + 1) * half],
                up_full[rank * half:(rank + 1) * half],
            ], dim=0)
            assert torch.equal(moe.w_gate_up[e], want_gu), f"rank {rank} expert {e} gate_up"
            assert torch.equal(moe.w_down[e], down_full[:, rank * half:(rank + 1) * half])


def test_mixtral_random_init_matches_loader_layout(tmp_path):
    """random_init_ under TP must produce the SAME local layout convention as
    the loader: reassembling both ranks' w_gate_up recovers a tensor whose
    gate half equals ranks' first parts (regression for the fused-stack
    contiguous-narrow bug)."""
    from kllms_amd.models.mixtral import MixtralForCausalLM

    cfg = MODEL_PRESETS["mid-mixtral"]
    m1 = MixtralForCausalLM(cfg, ParallelContext(), dtype=torch.float32)
    m1.random_init_(seed=0)
    full = m1.layers[0].mlp.w_gate_up  # [E, 2I, H] at TP=1
    I = cfg.intermediate_size
    half = I // 2
    for rank in (0, 1):
        m2 = MixtralForCausalLM(cfg, ParallelContext(world_size=2, rank=rank), dtype=torch.float32)
        m2.random_init_(seed=0)
        local = m2.layers[0].mlp.w_gate_up  # [E, I, H]
        want = torch.cat([
            full[:, rank * half:(rank + 1) * half],            # gate shard
            full[:, I + rank * half:I + (rank + 1) * half],    # up shard
        ], dim=1)
        assert torch.equal(local, want), f"rank {rank} init layout mismatch"
