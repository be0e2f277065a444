"""Size-switched TP collective: custom one-shot xGMI all-reduce + RCCL ring.

SURVEY §5.8: decode-step tensors ([n_streams, hidden] bf16, tens of KB) are
LATENCY-bound on the fully-connected xGMI mesh — every rank reads all peers
directly (one hop) instead of paying the ring's 2(N-1) hops. Prefill
activations (MBs) are BANDWIDTH-bound — the RCCL ring saturates the per-link
153 GB/s. The crossover is size-thresholded here.

The device side is ops/hip/allreduce.hip (hipIpc-mapped peer staging buffers,
two-phase epoch handshake in uncached memory, hipGraph-replayable); handle
exchange rides torch.distributed's existing rendezvous (all_gather_object).

Env knobs:
  KLLMS_CUSTOM_AR=0          disable (RCCL for every size)
  KLLMS_CUSTOM_AR_MAX=bytes  one-shot threshold (default 1 MiB)
"""

from __future__ import annotations

import logging
import os
from typing import Optional

import torch
import torch.distributed as dist

log = logging.getLogger(__name__)

# Crossover math (xGMI fully-connected, 7 links x ~153 GB/s): one-shot moves
# ~S per link while the ring moves ~2S/7 per link but pays ~2(N-1) hop
# latencies, so one-shot wins until S is several MB. 4 MiB also covers the
# largest captured decode-step tensor (max_batch 256 x hidden 8192 bf16), so
# hipGraph-captured decode never falls back to RCCL mid-graph.
_DEFAULT_MAX_BYTES = 4 << 20


class CustomAllReduce:
    """Per-rank handle over the IPC one-shot all-reduce context."""

    def __init__(self, rank: int, world: int, device: torch.device,
                 max_bytes: Optional[int] = None, group=None):
        from .. import ops

        if not ops.hip_available():  # pragma: no cover - GPU-only path
            raise RuntimeError("custom all-reduce needs the HIP extension")
        self._C = ops._C
        self.rank = rank
        self.world = world
        self.device = device
        self.max_bytes = max_bytes or int(os.environ.get("KLLMS_CUSTOM_AR_MAX", _DEFAULT_MAX_BYTES))
        self._ctx, handles = self._C.ipc_allreduce_create(rank, world, self.max_bytes)
        gathered = [None] * world
        dist.all_gather_object(gathered, bytes(handles), group=group)
        blob = b"".join(gathered)  # type: ignore[arg-type]
        self._C.ipc_allreduce_connect(self._ctx, blob)
        # all ranks must finish opening peers before anyone launches
        dist.barrier(group=group)
        self.calls = 0

    def should_use(self, t: torch.Tensor) -> bool:
        return (
            t.dtype == torch.bfloat16
            and t.is_contiguous()
            and t.numel() % 8 == 0
            and t.numel() * 2 <= self.max_bytes
        )

    def all_reduce_(self, t: torch.Tensor) -> torch.Tensor:
        """In-place one-shot sum across all ranks (bf16, fp32-accumulated)."""
        self._C.ipc_allreduce_run(self._ctx, t, t)
        self.calls += 1
        return t

    def close(self) -> None:
        if getattr(self, "_ctx", None):
            self._C.ipc_allreduce_destroy(self._ctx)
            self._ctx = 0


def maybe_init_custom_allreduce(ctx, device: torch.device) -> Optional[CustomAllReduce]:
    """Attach a CustomAllReduce to a ParallelContext when the platform
    supports it; silently fall back to RCCL-only otherwise."""
    if ctx.world_size <= 1 or device.type != "cuda":
        return None
    if os.environ.get("KLLMS_CUSTOM_AR", "1") == "0":
        return None
    if not (dist.is_available() and dist.is_initialized()):
        return None
    try:
        car = CustomAllReduce(ctx.rank, ctx.world_size, device, group=ctx.group)
        ctx.custom_ar = car
        log.info("custom one-shot all-reduce enabled (<=%d bytes)", car.max_bytes)
        return car
    except Exception as e:  # pragma: no cover - platform-dependent
        log.warning("custom all-reduce unavailable, using RCCL only: %s", e)
        return None
