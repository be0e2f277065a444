"""Safetensors weight loading with per-TP-rank sharding.

The load-only "checkpoint" subsystem: maps HuggingFace-style
Llama/Mixtral parameter names onto this model's fused/sharded layout and
slices each rank's shard while streaming tensors from the safetensors files
(no full-model host copy).
"""

from __future__ import annotations

import glob
import os
from typing import Dict

import torch

from ..parallel.tp import ParallelContext


def _shard(t: torch.Tensor, dim: int, rank: int, world: int) -> torch.Tensor:
    if world == 1:
        return t
    size = t.shape[dim] // world
    return t.narrow(dim, rank * size, size)


@torch.no_grad()
def load_safetensors_weights(model, weights_dir: str, ctx: ParallelContext) -> None:
    from safetensors import safe_open

    rank, world = ctx.rank, ctx.world_size
    files = sorted(glob.glob(os.path.join(weights_dir, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no .safetensors files under {weights_dir}")

    params: Dict[str, torch.Tensor] = dict(model.named_parameters())
    cfg = model.cfg
    H, KVH, D = cfg.num_heads, cfg.num_kv_heads, cfg.head_dim_

    # staging for fused tensors
    staged: Dict[str, dict] = {}
    assigned: set = set()
    expert_loads: Dict[str, int] = {}   # "layers.{L}.mlp.w_gate_up" -> #experts loaded

    def assign(our_name: str, tensor: torch.Tensor):
        p = params[our_name]
        assert p.shape == tensor.shape, (our_name, p.shape, tensor.shape)
        p.copy_(tensor.to(p.dtype))
        assigned.add(our_name)

    for f in files:
        with safe_open(f, framework="pt") as sf:
            for name in sf.keys():
                t = sf.get_tensor(name)
                n = name.replace("model.", "")
                if n == "embed_tokens.weight":
                    assign("embed_tokens.weight", t)
                elif n == "norm.weight":
                    assign("norm.weight", t)
                elif name == "lm_head.weight":
                    assign("lm_head.weight", t)
                elif ".self_attn.q_proj." in n or ".self_attn.k_proj." in n or ".self_attn.v_proj." in n:
                    layer = n.split(".")[1]
                    leaf = "bias" if n.endswith(".bias") else "weight"
                    key = f"layers.{layer}.self_attn.qkv_proj.{leaf}"
                    if leaf == "bias" and key not in params:
                        raise RuntimeError(
                            f"checkpoint has {name} but the model was built without "
                            f"attention_qkv_bias (set it in the arch config)")
                    st = staged.setdefault(key, {})
                    which = "q" if ".q_proj." in n else ("k" if ".k_proj." in n else "v")
                    # shard heads per rank, then fuse (bias shards dim 0 too)
                    st[which] = _shard(t, 0, rank, world)
                    if len(st) == 3:
                        assign(key, torch.cat([st["q"], st["k"], st["v"]], dim=0))
                        del staged[key]
                elif ".self_attn.o_proj." in n:
                    layer = n.split(".")[1]
                    assign(f"layers.{layer}.self_attn.o_proj.weight", _shard(t, 1, rank, world))
                elif ".mlp.gate_proj." in n or ".mlp.up_proj." in n:
                    layer = n.split(".")[1]
                    key = f"layers.{layer}.mlp.gate_up_proj.weight"
                    st = staged.setdefault(key, {})
                    st["gate" if ".gate_proj." in n else "up"] = _shard(t, 0, rank, world)
                    if len(st) == 2:
                        assign(key, torch.cat([st["gate"], st["up"]], dim=0))
                        del staged[key]
                elif ".mlp.down_proj." in n:
                    layer = n.split(".")[1]
                    assign(f"layers.{layer}.mlp.down_proj.weight", _shard(t, 1, rank, world))
                elif ".input_layernorm." in n or ".post_attention_layernorm." in n:
                    assign(n, t)
                elif ".block_sparse_moe.gate." in n:
                    layer = n.split(".")[1]
                    assign(f"layers.{layer}.mlp.gate.weight", t)
                elif ".block_sparse_moe.experts." in n:
                    # experts.{e}.w1/w2/w3 -> fused gate_up (w1,w3) / down (w2)
                    parts = n.split(".")
                    layer, e, w = parts[1], int(parts[4]), parts[5]
                    if w in ("w1", "w3"):
                        key = f"layers.{layer}.mlp.experts.{e}.gate_up"
                        st = staged.setdefault(key, {})
                        st["gate" if w == "w1" else "up"] = _shard(t, 0, rank, world)
                        if len(st) == 2:
                            model.load_expert_(int(layer), e, "gate_up", torch.cat([st["gate"], st["up"]], dim=0))
                            del staged[key]
                            k = f"layers.{layer}.mlp.w_gate_up"
                            expert_loads[k] = expert_loads.get(k, 0) + 1
                    else:  # w2
                        model.load_expert_(int(layer), e, "down", _shard(t, 1, rank, world))
                        k = f"layers.{layer}.mlp.w_down"
                        expert_loads[k] = expert_loads.get(k, 0) + 1
    if staged:
        raise RuntimeError(f"incomplete fused groups after load: {list(staged)[:4]}")

    # tied-embedding checkpoints (e.g. Llama-3.2) ship no lm_head.weight
    if cfg.tie_word_embeddings and "lm_head.weight" not in assigned:
        params["lm_head.weight"].copy_(params["embed_tokens.weight"])
        assigned.add("lm_head.weight")

    # completeness: a parameter the checkpoint never touched keeps its random
    # init and silently produces garbage logits — fail loudly instead.
    missing = []
    for name in params:
        if name in assigned:
            continue
        if name in expert_loads:
            if expert_loads[name] == cfg.num_experts:
                continue
            missing.append(f"{name} ({expert_loads[name]}/{cfg.num_experts} experts)")
            continue
        missing.append(name)
    if missing:
        raise RuntimeError(f"checkpoint left {len(missing)} parameters unassigned: {missing[:6]}")
