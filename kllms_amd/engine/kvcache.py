"""Paged KV cache with block reference counting and copy-on-write forking.

The memory backbone of shared-prefill/fanned-decode (BASELINE north star):
the prompt's KV blocks are written once at prefill, then the n decode streams
FORK the block table — full blocks are shared by reference count; a stream
copies a block only when it must write into a shared one (its last, partial
block). Sized for 288 GB of HBM3E per MI355X: with an 8B bf16 model resident
(~16 GB) the cache can hold ~2M tokens of KV.

Layout per layer: k_cache/v_cache = [num_blocks, kv_heads, block_size, head_dim].
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Tuple

import torch


class BlockAllocator:
    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free: list[int] = list(range(num_blocks - 1, -1, -1))
        self._refcount = [0] * num_blocks

    @property
    def num_free(self) -> int:
        return len(self._free)

    def alloc(self) -> int:
        if not self._free:
            raise RuntimeError("KV cache out of blocks")
        b = self._free.pop()
        assert self._refcount[b] == 0
        self._refcount[b] = 1
        return b

    def incref(self, block: int) -> None:
        assert self._refcount[block] > 0
        self._refcount[block] += 1

    def free(self, block: int) -> None:
        assert self._refcount[block] > 0
        self._refcount[block] -= 1
        if self._refcount[block] == 0:
            self._free.append(block)

    def refcount(self, block: int) -> int:
        return self._refcount[block]


@dataclass
class SequenceKV:
    """A decode stream's view of the cache: its block table and length."""

    blocks: List[int] = field(default_factory=list)
    num_tokens: int = 0


class PagedKVCache:
    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        block_size: int,
        num_blocks: int,
        device,
        dtype=torch.bfloat16,
    ):
        self.num_layers = num_layers
        self.block_size = block_size
        self.num_blocks = num_blocks
        self.device = device
        # One contiguous allocation for all layers keeps pointers stable for
        # hipGraph capture and makes block copies (COW) two device ops.
        self.k_all = torch.zeros(num_layers, num_blocks, num_kv_heads, block_size, head_dim,
                                 device=device, dtype=dtype)
        self.v_all = torch.zeros_like(self.k_all)
        self.k_caches = [self.k_all[i] for i in range(num_layers)]
        self.v_caches = [self.v_all[i] for i in range(num_layers)]
        self.allocator = BlockAllocator(num_blocks)

    def layer_caches(self) -> List[Tuple[torch.Tensor, torch.Tensor]]:
        return list(zip(self.k_caches, self.v_caches))

    # --- sequence-level operations ------------------------------------------
    def alloc_sequence(self, num_tokens: int) -> SequenceKV:
        n_blocks = (num_tokens + self.block_size - 1) // self.block_size
        blocks: List[int] = []
        try:
            for _ in range(n_blocks):
                blocks.append(self.allocator.alloc())
        except Exception:
            for b in blocks:
                self.allocator.free(b)
            raise
        return SequenceKV(blocks=blocks, num_tokens=num_tokens)

    def fork(self, parent: SequenceKV) -> SequenceKV:
        """Share the parent's blocks by refcount. The block the child will
        write into next (a partially-filled tail block) is copied EAGERLY so
        the decode hot loop never needs copy-on-write bookkeeping."""
        child = SequenceKV(blocks=[], num_tokens=parent.num_tokens)
        tail = parent.num_tokens % self.block_size
        try:
            for i, b in enumerate(parent.blocks):
                if i == len(parent.blocks) - 1 and tail != 0:
                    child.blocks.append(self._copy_block(b))
                else:
                    self.allocator.incref(b)
                    child.blocks.append(b)
        except Exception:
            for b in child.blocks:
                self.allocator.free(b)
            child.blocks = []
            raise
        return child

    def _copy_block(self, src: int) -> int:
        dst = self.allocator.alloc()
        self.k_all[:, dst].copy_(self.k_all[:, src])
        self.v_all[:, dst].copy_(self.v_all[:, src])
        return dst

    def append_slot(self, seq: SequenceKV) -> int:
        """Reserve the slot for one new token; returns the FLAT slot index.
        Handles block growth and copy-on-write of a shared last block."""
        pos = seq.num_tokens
        off = pos % self.block_size
        blk_idx = pos // self.block_size
        if blk_idx == len(seq.blocks):
            seq.blocks.append(self.allocator.alloc())
        else:
            last = seq.blocks[blk_idx]
            if self.allocator.refcount(last) > 1:
                # copy-on-write: detach from the shared block
                new = self._copy_block(last)
                self.allocator.free(last)
                seq.blocks[blk_idx] = new
        seq.num_tokens += 1
        return seq.blocks[blk_idx] * self.block_size + off

    def prefill_slot_mapping(self, seq: SequenceKV, start: int = 0) -> list[int]:
        """Flat slots for tokens [start, seq.num_tokens) of a fresh sequence."""
        slots = []
        for pos in range(start, seq.num_tokens):
            slots.append(seq.blocks[pos // self.block_size] * self.block_size + pos % self.block_size)
        return slots

    def free_sequence(self, seq: SequenceKV) -> None:
        for b in seq.blocks:
            self.allocator.free(b)
        seq.blocks = []
        seq.num_tokens = 0
