"""hipGraph capture of the decode step.

The decode inner loop is launch-bound (dozens of small kernels x 32+ layers
per token); capturing it as a hipGraph (torch.cuda.CUDAGraph on ROCm IS
hipGraph) replays the whole step as one graph launch. Paged-KV pointer
indirection is handled the standard way: all graph inputs (token ids,
positions, slot mapping, block tables, context lens) live in fixed
pre-allocated buffers that the engine writes into before replay; the KV cache
tensors themselves are allocated once and never move.

Batch sizes are bucketed (config.hip_graph_batch_sizes); a decode batch of
size B runs the smallest captured bucket >= B with tail padding. Padded lanes
replay against a scratch KV block (slot 0 writes are masked by pointing
padded slots at a dedicated scratch block) and their outputs are discarded.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from ..models.llama import ForwardBatch


class DecodeGraphRunner:
    def __init__(self, engine, batch_sizes: List[int]):
        self.engine = engine
        self.batch_sizes = sorted(batch_sizes)
        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.buffers: Dict[int, dict] = {}
        self.max_blocks_per_seq = (engine.config.max_seq_len + engine.config.kv_block_size - 1) // engine.config.kv_block_size
        # one scratch block for padded lanes' KV writes
        self._scratch_block = engine.kv.allocator.alloc()
        self._enabled = True

    def _capture(self, bs: int) -> None:
        eng = self.engine
        dev = eng.device
        mb = self.max_blocks_per_seq
        buf = {
            "ids": torch.zeros(bs, dtype=torch.long, device=dev),
            "positions": torch.zeros(bs, dtype=torch.long, device=dev),
            "slots": torch.full((bs,), self._scratch_block * eng.config.kv_block_size, dtype=torch.long, device=dev),
            "block_tables": torch.full((bs, mb), self._scratch_block, dtype=torch.int32, device=dev),
            "context_lens": torch.ones(bs, dtype=torch.int32, device=dev),
        }
        batch = ForwardBatch(
            mode="decode",
            positions=buf["positions"],
            slot_mapping=buf["slots"],
            kv_caches=eng.kv.layer_caches(),
            block_tables=buf["block_tables"],
            context_lens=buf["context_lens"],
        )
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                out = eng.model.forward_hidden(buf["ids"], batch)
        torch.cuda.current_stream().wait_stream(s)

        # Capture the LAYERS only; the LM-head GEMM runs eagerly per step —
        # under capture rocBLAS picks a stream-K algorithm (~3x slower, plus a
        # workspace fill) for the [B,H]x[H,V] shape; eager picks the
        # weight-streaming-bound one (measured in scripts/lmhead_probe.py).
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            out = eng.model.forward_hidden(buf["ids"], batch)
        buf["out"] = out
        self.graphs[bs] = g
        self.buffers[bs] = buf

    def _recover_generator_state(self) -> None:
        """Un-stick the default Philox generator after a failed capture.

        hipGraph capture registers the device's default RNG generator at
        capture_begin; only a successful capture_end runs the epilogue that
        clears the generator's 'capturing' flag. A capture that dies between
        the two (e.g. cudaStreamEndCapture on an invalidated capture) leaves
        the flag set, and the next RNG op ANYWHERE in the process then raises
        "Offset increment outside graph capture encountered unexpectedly".
        Swapping in a fresh clone of the state (same seed/offset, flag clear)
        repairs it.
        """
        try:
            idx = self.engine.device.index
            if idx is None:
                idx = torch.cuda.current_device()
            gen = torch.cuda.default_generators[idx]
            gen.graphsafe_set_state(gen.clone_state())
        except Exception:
            pass  # best effort: eager fallback works regardless for non-RNG ops

    def _bucket(self, b: int) -> Optional[int]:
        for bs in self.batch_sizes:
            if bs >= b:
                return bs
        return None

    def run(self, ids: torch.Tensor, batch: ForwardBatch) -> torch.Tensor:
        if not self._enabled:
            return self.engine.model.forward_decode(ids, batch)
        B = ids.shape[0]
        bs = self._bucket(B)
        if bs is None or batch.block_tables.shape[1] > self.max_blocks_per_seq:
            return self.engine.model.forward_decode(ids, batch)
        if bs not in self.graphs:
            try:
                self._capture(bs)
            except Exception as e:  # capture failure: loud, then eager
                import logging

                logging.getLogger("kllms_amd.engine").error(
                    "hipGraph capture failed for batch %d (%s); falling back to eager",
                    bs, e, exc_info=True,
                )
                self._enabled = False
                self._recover_generator_state()
                return self.engine.model.forward_decode(ids, batch)
        buf = self.buffers[bs]
        scratch_slot = self._scratch_block * self.engine.config.kv_block_size
        buf["ids"][:B].copy_(ids)
        buf["ids"][B:].fill_(0)
        buf["positions"][:B].copy_(batch.positions)
        buf["positions"][B:].fill_(0)
        buf["slots"][:B].copy_(batch.slot_mapping)
        buf["slots"][B:].fill_(scratch_slot)
        nb = batch.block_tables.shape[1]
        buf["block_tables"][:B, :nb].copy_(batch.block_tables)
        buf["block_tables"][:B, nb:].fill_(self._scratch_block)
        buf["block_tables"][B:].fill_(self._scratch_block)
        buf["context_lens"][:B].copy_(batch.context_lens)
        buf["context_lens"][B:].fill_(1)
        self.graphs[bs].replay()
        return self.engine.model.compute_logits(buf["out"][:B])
