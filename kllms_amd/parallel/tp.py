"""Tensor parallelism over RCCL / xGMI.

The reference has zero distributed code (its backend is a remote API); this is
the native sharding layer the BASELINE configs require:
column/row-parallel linear layers with an all-reduce after the attention
output projection and the MLP down projection — exactly 2 all-reduces per
transformer layer at TP > 1.

Design for the MI355X node: one process per GPU over torch.distributed
(backend "nccl" IS RCCL on ROCm); xGMI is point-to-point (7 links x ~153 GB/s
per GPU), so decode-step tensors ([n_streams, hidden] bf16, tens of KB) are
latency-bound — they go through a single fused all_reduce call per boundary,
not bucketed chunks — while prefill activations (MBs) saturate the ring.
CPU tests run the same code over the gloo backend.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn


@dataclass
class ParallelContext:
    """Process-group handle for one tensor-parallel engine instance."""

    world_size: int = 1
    rank: int = 0
    group: Optional[object] = None  # dist.ProcessGroup
    # size-switched custom collective (parallel/collective.py); None = RCCL only
    custom_ar: Optional[object] = None

    @classmethod
    def from_env_or_single(cls) -> "ParallelContext":
        if dist.is_available() and dist.is_initialized():
            return cls(world_size=dist.get_world_size(), rank=dist.get_rank(), group=None)
        return cls()

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.world_size > 1:
            ca = self.custom_ar
            if ca is not None and ca.should_use(t):
                return ca.all_reduce_(t)
            dist.all_reduce(t, group=self.group)
        return t

    def all_gather_cat(self, t: torch.Tensor, dim: int = -1) -> torch.Tensor:
        if self.world_size == 1:
            return t
        parts = [torch.empty_like(t) for _ in range(self.world_size)]
        dist.all_gather(parts, t.contiguous(), group=self.group)
        return torch.cat(parts, dim=dim)


class ColumnParallelLinear(nn.Module):
    """Y = X W^T with W row-sharded over ranks (output features split).

    No communication on forward; the sharded output feeds a row-parallel
    layer (attention QKV -> O, MLP gate/up -> down).
    """

    def __init__(self, in_features: int, out_features: int, ctx: ParallelContext, dtype=None,
                 partition_sizes=None, bias: bool = False):
        super().__init__()
        assert out_features % ctx.world_size == 0, (out_features, ctx.world_size)
        self.ctx = ctx
        self.in_features = in_features
        self.out_features_per_rank = out_features // ctx.world_size
        # For FUSED projections (qkv, gate_up): the full output dim is a
        # concatenation of logical parts and each rank must hold the
        # concatenation of its SHARD OF EACH PART — a contiguous slice of the
        # fused dim would give rank 0 all of q and none of k/v.
        self.partition_sizes = list(partition_sizes) if partition_sizes else [out_features]
        assert sum(self.partition_sizes) == out_features
        assert all(p % ctx.world_size == 0 for p in self.partition_sizes)
        self.weight = nn.Parameter(
            torch.empty(self.out_features_per_rank, in_features, dtype=dtype), requires_grad=False
        )
        if bias:
            # output-sharded, so the bias shards with the weight's dim 0 and
            # needs no communication (Qwen2-style attention qkv bias)
            self.bias = nn.Parameter(
                torch.empty(self.out_features_per_rank, dtype=dtype), requires_grad=False
            )
        else:
            self.register_parameter("bias", None)

    def shard_full_tensor(self, full: torch.Tensor) -> torch.Tensor:
        """Slice this rank's shard out of the FULL fused weight (dim 0)."""
        tp, rank = self.ctx.world_size, self.ctx.rank
        if tp == 1:
            return full
        pieces = []
        off = 0
        for part in self.partition_sizes:
            sz = part // tp
            pieces.append(full.narrow(0, off + rank * sz, sz))
            off += part
        return torch.cat(pieces, dim=0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.nn.functional.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """Y = X W^T with W column-sharded over ranks (input features split);
    forward ends with one RCCL all-reduce (the per-layer TP boundary)."""

    def __init__(self, in_features: int, out_features: int, ctx: ParallelContext, dtype=None):
        super().__init__()
        assert in_features % ctx.world_size == 0, (in_features, ctx.world_size)
        self.ctx = ctx
        self.in_features_per_rank = in_features // ctx.world_size
        self.out_features = out_features
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_features_per_rank, dtype=dtype), requires_grad=False
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = torch.nn.functional.linear(x, self.weight)
        return self.ctx.all_reduce(y)


def init_distributed_from_env(device: str = "cuda") -> ParallelContext:
    """torchrun-style init: one rank per GPU over RCCL (or gloo on CPU)."""
    import os

    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return ParallelContext()
    backend = "nccl" if device.startswith("cuda") and torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    if backend == "nccl":
        torch.cuda.set_device(rank % torch.cuda.device_count())
    return ParallelContext(world_size=dist.get_world_size(), rank=rank)
