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

import hashlib
from array import array
from collections import OrderedDict
from dataclasses import dataclass, field
from typing import Callable, List, Optional, Tuple

import torch


class BlockAllocator:
    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free: list[int] = list(range(num_blocks - 1, -1, -1))
        self._refcount = [0] * num_blocks
        # called when the pool runs dry before raising — the prefix cache
        # hooks this to drop its cached (refcounted) blocks under pressure
        self.on_pressure: Optional[Callable[[], None]] = None

    @property
    def num_free(self) -> int:
        return len(self._free)

    def alloc(self) -> int:
        if not self._free and self.on_pressure is not None:
            self.on_pressure()
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
        # fp8 caches carry per-(token, head)-ROW dequant scales (s = amax/448
        # at store time): outlier rows in real checkpoints would otherwise
        # saturate e4m3's +-448 under a static scale. fp32 [NB, KVH, BS] per
        # layer per cache = 1/32 of the fp8 payload; init 1.0 (neutral for
        # never-written slots). bf16 caches carry no scales (None).
        self.fp8 = dtype == torch.float8_e4m3fn
        if self.fp8:
            self.k_scale_all = torch.ones(num_layers, num_blocks, num_kv_heads, block_size,
                                          device=device, dtype=torch.float32)
            self.v_scale_all = torch.ones_like(self.k_scale_all)
        else:
            self.k_scale_all = self.v_scale_all = None
        self.allocator = BlockAllocator(num_blocks)

    def layer_caches(self) -> List[Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor], Optional[torch.Tensor]]]:
        """Per layer (k, v, k_scale, v_scale); scales are None for bf16."""
        if self.fp8:
            return [(self.k_caches[i], self.v_caches[i], self.k_scale_all[i], self.v_scale_all[i])
                    for i in range(self.num_layers)]
        return [(self.k_caches[i], self.v_caches[i], None, None) for i in range(self.num_layers)]

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
        if self.fp8:
            self.k_scale_all[:, dst].copy_(self.k_scale_all[:, src])
            self.v_scale_all[:, dst].copy_(self.v_scale_all[:, src])
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


class PrefixCache:
    """Shared-prefix KV reuse across requests.

    Consensus batches share the chat-template header, and real serving
    usually shares a long system prompt; this maps a chain digest of each
    FULL prompt block (BLAKE2b over (parent digest, block token ids) — chain
    keying makes a block's identity include its whole prefix) to a cache
    block id. Cached blocks hold one extra refcount owned by the cache; a
    matching request increfs and SKIPS prefilling those tokens (its tail
    attends to the cached KV through the paged cache, engine.prefill_chunk).
    Full blocks are never written after prefill (decode appends go to new /
    CoW-copied tail blocks), so sharing is safe by the same refcount
    machinery the n-way fork uses.

    Eviction: LRU above ``max_blocks``, and everything on allocator
    pressure (BlockAllocator.on_pressure) — a cached block whose only ref is
    the cache returns to the free pool immediately."""

    def __init__(self, kv: "PagedKVCache", max_blocks: int):
        self.kv = kv
        self.max_blocks = max_blocks
        self._map: "OrderedDict[bytes, int]" = OrderedDict()
        self.hits = 0
        self.misses = 0
        self.tokens_saved = 0
        kv.allocator.on_pressure = self.evict_all

    def _digests(self, prompt_ids: List[int], n_blocks: int) -> List[bytes]:
        bs = self.kv.block_size
        out: List[bytes] = []
        h = b""
        for k in range(n_blocks):
            m = hashlib.blake2b(h, digest_size=16)
            m.update(array("q", prompt_ids[k * bs: (k + 1) * bs]).tobytes())
            h = m.digest()
            out.append(h)
        return out

    def match(self, prompt_ids: List[int], min_tokens: int = 0) -> List[int]:
        """Longest cached chain of full prompt blocks (always leaves >= 1
        prompt token un-matched so prefill still produces logits). A chain
        shorter than ``min_tokens`` counts as a miss and returns [] — a short
        cached head saves less prefill than the decode-mode tail forward
        costs over the packed kernel. Returns block ids WITHOUT increfing —
        the caller owns that step."""
        bs = self.kv.block_size
        limit = (len(prompt_ids) - 1) // bs
        if limit <= 0:
            return []
        blocks: List[int] = []
        for d in self._digests(prompt_ids, limit):
            b = self._map.get(d)
            if b is None:
                break
            self._map.move_to_end(d)
            blocks.append(b)
        if blocks and len(blocks) * bs < min_tokens:
            blocks = []
        if blocks:
            self.hits += 1
            self.tokens_saved += len(blocks) * bs
        else:
            self.misses += 1
        return blocks

    def register(self, prompt_ids: List[int], seq: SequenceKV) -> None:
        """Publish a prefilled prompt's full blocks."""
        n_full = min(len(seq.blocks), len(prompt_ids) // self.kv.block_size)
        if n_full <= 0:
            return
        for k, d in enumerate(self._digests(prompt_ids, n_full)):
            if d in self._map:
                self._map.move_to_end(d)
                continue
            self.kv.allocator.incref(seq.blocks[k])
            self._map[d] = seq.blocks[k]
        while len(self._map) > self.max_blocks:
            _, b = self._map.popitem(last=False)
            self.kv.allocator.free(b)

    def evict_all(self) -> None:
        while self._map:
            _, b = self._map.popitem(last=False)
            self.kv.allocator.free(b)

    @property
    def stats(self) -> dict:
        return {"entries": len(self._map), "hits": self.hits,
                "misses": self.misses, "tokens_saved": self.tokens_saved}
