"""LLMEngine: shared prefill + n-way fanned decode.

The device-side hot section of the framework — the native replacement for the
reference's single remote API call (k_llms/resources/completions/completions.py:73).
One ``generate()`` call takes a batch of requests, prefills every prompt ONCE
in a packed varlen batch, forks each prompt's KV blocks n ways (copy-on-write
paged cache), and steps all decode streams together until completion.

Usage semantics mirror OpenAI's n>1 accounting: prompt tokens
counted once per request (the prefill is genuinely shared), completion tokens
summed across the n streams.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, List, Optional, Tuple

import torch

from ..models.llama import ForwardBatch
from .config import EngineConfig, ModelArchConfig
from .kvcache import PagedKVCache, SequenceKV
from .sampling import SamplingParams
from .tokenizer import BaseTokenizer, load_tokenizer
from .. import ops


@dataclass
class GenRequest:
    prompt_ids: List[int]
    n: int = 1
    sampling: SamplingParams = field(default_factory=SamplingParams)
    constraint: Optional[Any] = None  # kllms_amd.engine.constrained.JsonSchemaConstraint
    # absolute time.monotonic() cutoff (client `timeout`): streams past it
    # finish with reason "length" at the next step boundary. TP>1 ignores
    # deadlines — a wall-clock stop is rank-divergent and would break the
    # followers' lockstep replay (serve.py)
    deadline: Optional[float] = None


@dataclass
class StreamOutput:
    token_ids: List[int] = field(default_factory=list)
    logprobs: List[float] = field(default_factory=list)
    # per emitted token: top-K (token_id, logprob) alternatives — only
    # populated for streams whose sampling.top_logprobs > 0
    top_logprobs: List[List[Tuple[int, float]]] = field(default_factory=list)
    text: str = ""
    finish_reason: str = "length"
    stream_idx: int = 0  # position among the request's n streams


@dataclass
class RequestOutput:
    prompt_tokens: int
    streams: List[StreamOutput] = field(default_factory=list)
    # wall-clock phase timings (observability)
    prefill_ms: float = 0.0
    decode_ms: float = 0.0


class _Stream:
    """One active decode stream (one of a request's n samples)."""

    def __init__(self, req_idx: int, stream_idx: int, seq: SequenceKV, sampling: SamplingParams,
                 seed: int, constraint_state=None, constraint=None):
        self.req_idx = req_idx
        self.stream_idx = stream_idx
        self.seq = seq
        self.sampling = sampling
        self.seed = seed
        self.out = StreamOutput(stream_idx=stream_idx)
        self.deadline: Optional[float] = None
        self.done = False
        self.step = 0
        self.constraint = constraint
        self.constraint_state = constraint_state
        self.last_token: int = -1


class _DecodeBatchState:
    """Device-resident decode batch: token ids, positions, context lengths,
    block tables and sampling parameters live on the GPU and are updated in
    place each step; rebuilt only when stream membership changes."""

    def __init__(self, engine: "LLMEngine", streams: List[_Stream]):
        dev = engine.device
        bs = engine.kv.block_size
        self.streams = streams
        self.ids = torch.tensor([s.last_token for s in streams], dtype=torch.long, device=dev)
        self.positions = torch.tensor([s.seq.num_tokens for s in streams], dtype=torch.long, device=dev)
        self.ctx = torch.tensor([s.seq.num_tokens + 1 for s in streams], dtype=torch.int32, device=dev)
        # block-table width with headroom so per-step growth is rare
        cur_max = max(len(s.seq.blocks) for s in streams)
        horizon = max(s.sampling.max_tokens or engine.config.default_max_new_tokens for s in streams)
        self.width = cur_max + (horizon + bs - 1) // bs + 1
        bt = torch.zeros((len(streams), self.width), dtype=torch.int32)
        for i, s in enumerate(streams):
            bt[i, : len(s.seq.blocks)] = torch.tensor(s.seq.blocks, dtype=torch.int32)
        self.bt = bt.to(dev)
        self.temps = torch.tensor([s.sampling.temperature for s in streams], dtype=torch.float32, device=dev)
        self.top_ps = torch.tensor([s.sampling.top_p for s in streams], dtype=torch.float32, device=dev)
        self.top_ks = torch.tensor([s.sampling.top_k for s in streams], dtype=torch.int32, device=dev)
        self.seeds = torch.tensor([s.seed for s in streams], dtype=torch.int64, device=dev)
        self.steps = torch.tensor([s.step for s in streams], dtype=torch.int64, device=dev)

    def apply_bt_updates(self, updates: List[tuple]) -> None:
        need = max(j for _, j, _ in updates) + 1
        if need > self.width:
            pad = torch.zeros((self.bt.shape[0], need + 8 - self.width), dtype=torch.int32, device=self.bt.device)
            self.bt = torch.cat([self.bt, pad], dim=1)
            self.width = self.bt.shape[1]
        rows = torch.tensor([u[0] for u in updates], dtype=torch.long, device=self.bt.device)
        cols = torch.tensor([u[1] for u in updates], dtype=torch.long, device=self.bt.device)
        vals = torch.tensor([u[2] for u in updates], dtype=torch.int32, device=self.bt.device)
        self.bt[rows, cols] = vals


class LLMEngine:
    def __init__(self, config: EngineConfig, parallel_ctx=None):
        from ..parallel.tp import ParallelContext

        self.config = config
        self.arch: ModelArchConfig = config.resolve_arch()
        if config.device is not None:
            self.device = torch.device(config.device)
        else:
            self.device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
        # Parallel context is tied to config.tp_size, NOT to whether
        # torch.distributed happens to be initialized: under a data-parallel
        # launch (torchrun, one independent engine per rank — the bench's
        # weak-scaling mode) a tp_size=1 engine must NOT adopt the world as a
        # TP group, or its layers would all-reduce across ranks that are
        # processing different batches.
        if parallel_ctx is not None:
            self.ctx = parallel_ctx
        elif config.tp_size > 1:
            self.ctx = ParallelContext.from_env_or_single()
            assert self.ctx.world_size == config.tp_size, (
                f"tp_size={config.tp_size} but torch.distributed world is {self.ctx.world_size}"
            )
        else:
            self.ctx = ParallelContext()

        # size-switched custom collective: one-shot xGMI all-reduce for small
        # (decode-step) tensors, RCCL ring above threshold
        if self.ctx.world_size > 1 and self.ctx.custom_ar is None:
            from ..parallel.collective import maybe_init_custom_allreduce

            dev_for_ar = self.device if self.device.type == "cuda" else torch.device("cpu")
            maybe_init_custom_allreduce(self.ctx, dev_for_ar)

        # monotonic per-request counter for unseeded-request RNG derivation
        self._seed_counter = 0
        # device-tensor cache for per-request logit_bias dicts (keyed by id)
        self._logit_bias_cache: dict = {}
        self.dtype = {"bfloat16": torch.bfloat16, "float16": torch.float16, "float32": torch.float32}[config.dtype]
        # CPU bf16 matmuls are slow and torch CPU attention paths prefer f32
        if self.device.type == "cpu" and self.dtype == torch.bfloat16:
            self.dtype = torch.float32

        # the RoPE cos/sin table (and thus every position) is bounded by the
        # architecture's max_position_embeddings
        if config.max_seq_len > self.arch.max_position_embeddings:
            config.max_seq_len = self.arch.max_position_embeddings

        self.tokenizer: BaseTokenizer = load_tokenizer(config.model, config.weights_path, self.arch.vocab_size)
        eos_cfg = config.eos_token_id
        if eos_cfg is None:
            eos_cfg = self._generation_config_eos()
        if eos_cfg is None:
            eos_cfg = self.tokenizer.eos_id
        # a checkpoint may declare SEVERAL stop ids (generation_config.json
        # eos_token_id list, e.g. Llama-3's [<|end_of_text|>, <|eot_id|>]);
        # eos_token_id stays the primary (used where ONE id is needed, e.g.
        # the constrained-decode DFA's accepting-state bit)
        ids = list(eos_cfg) if isinstance(eos_cfg, (list, tuple)) else [eos_cfg]
        ids = [i for i in ids if i is not None]
        if not ids:  # e.g. an empty list in generation_config.json
            ids = [self.tokenizer.eos_id]
        self.eos_token_id = ids[0]
        self.eos_token_ids = frozenset(ids)

        self.model = self._build_model()
        self.kv = self._build_kv_cache()
        # cross-request shared-prefix KV reuse (kvcache.PrefixCache)
        self.prefix_cache = None
        if config.enable_prefix_caching:
            from .kvcache import PrefixCache

            max_pfx = max(8, int(self.kv.num_blocks * config.prefix_cache_fraction))
            self.prefix_cache = PrefixCache(self.kv, max_pfx)
        self._graph_runner = None
        if config.use_hip_graphs and self.device.type == "cuda":
            from .graph_runner import DecodeGraphRunner

            self._graph_runner = DecodeGraphRunner(self, config.hip_graph_batch_sizes)

    # --- construction --------------------------------------------------------
    def _generation_config_eos(self):
        """eos_token_id from the checkpoint's generation_config.json (int or
        list), or None when absent / random init."""
        import json as _json
        import os as _os

        wdir = self.config.effective_weights_dir() or (
            self.config.model if _os.path.isdir(str(self.config.model)) else None)
        if not wdir:
            return None
        p = _os.path.join(wdir, "generation_config.json")
        if not _os.path.exists(p):
            return None
        try:
            with open(p) as f:
                return _json.load(f).get("eos_token_id")
        except Exception:
            return None

    def _build_model(self):
        from ..models.llama import LlamaForCausalLM
        from ..models.mixtral import MixtralForCausalLM

        cls = MixtralForCausalLM if self.arch.arch == "mixtral" else LlamaForCausalLM
        model = cls(self.arch, self.ctx, dtype=self.dtype)
        wdir = self.config.effective_weights_dir()
        if wdir:
            from .weights import load_safetensors_weights

            load_safetensors_weights(model, wdir, self.ctx)
            model.to_device(self.device)
        else:
            model.to_device(self.device)
            model.random_init_(self.config.seed)
        model.eval()
        return model

    def _build_kv_cache(self) -> PagedKVCache:
        a = self.arch
        tp = self.ctx.world_size
        kv_heads_local = max(1, a.num_kv_heads // tp)
        elem = torch.tensor([], dtype=self.dtype).element_size()
        scale_bytes = 0
        if self.config.kv_cache_dtype == "fp8_e4m3":
            elem = 1
            # per-row fp32 dequant scale sidecar: 4 B per (head_dim) row
            scale_bytes = 2 * a.num_layers * kv_heads_local * self.config.kv_block_size * 4
        block_bytes = (2 * a.num_layers * kv_heads_local * self.config.kv_block_size *
                       a.head_dim_ * elem + scale_bytes)
        if self.config.max_kv_blocks is not None:
            num_blocks = self.config.max_kv_blocks
        elif self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            budget = int(free * self.config.kv_memory_fraction)
            num_blocks = max(64, budget // block_bytes)
            # cap the decode block-table width implied by max_seq_len anyway
        else:
            num_blocks = 512
        cache_dtype = self.dtype
        if self.config.kv_cache_dtype == "fp8_e4m3":
            cache_dtype = torch.float8_e4m3fn
        return PagedKVCache(
            a.num_layers, kv_heads_local, a.head_dim_, self.config.kv_block_size,
            int(num_blocks), self.device, cache_dtype,
        )

    # --- helpers --------------------------------------------------------------
    def _sampling_tensors(self, streams: List[_Stream]):
        B = len(streams)
        dev = self.device
        temps = torch.tensor([s.sampling.temperature for s in streams], dtype=torch.float32, device=dev)
        top_ps = torch.tensor([s.sampling.top_p for s in streams], dtype=torch.float32, device=dev)
        top_ks = torch.tensor([s.sampling.top_k for s in streams], dtype=torch.int32, device=dev)
        seeds = torch.tensor([s.seed for s in streams], dtype=torch.int64, device=dev)
        steps = torch.tensor([s.step for s in streams], dtype=torch.int64, device=dev)
        return temps, top_ps, top_ks, seeds, steps

    def _constraint_mask(self, streams: List[_Stream]) -> Optional[torch.Tensor]:
        if not any(s.constraint is not None for s in streams):
            return None
        V = self.arch.vocab_size
        W = (V + 31) // 32
        mask = torch.empty((len(streams), W), dtype=torch.int32)
        full = torch.full((W,), -1, dtype=torch.int32)
        for i, s in enumerate(streams):
            if s.constraint is None:
                mask[i] = full
            else:
                mask[i] = s.constraint.allowed_mask(s.constraint_state)
        return mask.to(self.device)

    def _apply_penalties(self, logits: torch.Tensor, streams: List[_Stream]) -> torch.Tensor:
        """frequency/presence penalties and logit_bias (OpenAI semantics),
        applied only when a stream requests them (rare; host-side
        composition)."""
        if not any(s.sampling.frequency_penalty or s.sampling.presence_penalty
                   or s.sampling.logit_bias for s in streams):
            return logits
        for i, s in enumerate(streams):
            fp, pp = s.sampling.frequency_penalty, s.sampling.presence_penalty
            if (fp or pp) and s.out.token_ids:
                ids = torch.tensor(s.out.token_ids, device=logits.device)
                counts = torch.bincount(ids, minlength=logits.shape[1]).to(logits.dtype)
                logits[i] -= fp * counts + pp * (counts > 0).to(logits.dtype)
            lb = s.sampling.logit_bias
            if lb:
                key = id(lb)
                cached = self._logit_bias_cache.get(key)
                if cached is None:
                    idx = torch.tensor([int(t) for t in lb], dtype=torch.long,
                                       device=logits.device)
                    val = torch.tensor([float(v) for v in lb.values()],
                                       dtype=logits.dtype, device=logits.device)
                    cached = (idx, val)
                    self._logit_bias_cache[key] = cached
                    if len(self._logit_bias_cache) > 256:
                        self._logit_bias_cache.pop(next(iter(self._logit_bias_cache)))
                logits[i, cached[0]] += cached[1]
        return logits

    # --- main entry -----------------------------------------------------------
    @torch.inference_mode()
    def generate(self, requests: List[GenRequest]) -> List[RequestOutput]:
        if not requests:
            return []
        # admission control: bound concurrent decode streams; excess requests
        # are served in sub-batches (same results, sequential batches)
        total_streams = sum(max(1, r.n) for r in requests)
        if total_streams > self.config.max_batch_size and len(requests) > 1:
            outputs: List[RequestOutput] = []
            sub: List[GenRequest] = []
            sub_streams = 0
            for r in requests:
                rn = max(1, r.n)
                if sub and sub_streams + rn > self.config.max_batch_size:
                    outputs.extend(self.generate(sub))
                    sub, sub_streams = [], 0
                sub.append(r)
                sub_streams += rn
            if sub:
                outputs.extend(self.generate(sub))
            return outputs

        parent_seqs: List[SequenceKV] = []
        streams: List[_Stream] = []
        try:
            return self._generate_batch(requests, parent_seqs, streams)
        except Exception:
            # free every block this batch still holds (KV exhaustion mid-run,
            # kernel failure, ...) so the engine stays usable
            for seq in parent_seqs:
                if seq.blocks:
                    self.kv.free_sequence(seq)
            for s in streams:
                if s.seq.blocks:
                    self.kv.free_sequence(s.seq)
            raise

    def begin_requests(
        self, requests: List[GenRequest], parent_seqs: List[SequenceKV], streams: List[_Stream]
    ) -> List[RequestOutput]:
        """Phase 1 of serving: shared prefill over all prompts, KV fork into n
        streams per request, first-token sampling. Appends the new streams to
        ``streams`` and returns one RequestOutput per request (streams are
        attached as they finish). Used by generate() and by the continuous-
        batching scheduler (which interleaves this with decode steps)."""
        dev = self.device

        # ---- prefix-cache match + sequence allocation ----------------------
        # a request whose prompt head is cached prefills only its TAIL (the
        # tail attends to the cached KV through the paged cache); misses go
        # through the packed varlen prefill as before
        matched: List[int] = []
        min_hit = self.config.prefix_cache_min_tokens
        for req in requests:
            ids = req.prompt_ids
            assert len(ids) > 0, "empty prompt"
            pb = (self.prefix_cache.match(ids, min_hit)
                  if self.prefix_cache is not None else [])
            seq = self._alloc_with_prefix(ids, pb)
            parent_seqs.append(seq)
            matched.append(len(pb) * self.kv.block_size)

        # ---- shared prefill: one packed varlen batch over the MISSES --------
        all_ids: List[int] = []
        all_pos: List[int] = []
        all_slots: List[int] = []
        cu = [0]
        pack_rows: List[int] = []
        for i, req in enumerate(requests):
            if matched[i] > 0:
                continue
            ids = req.prompt_ids
            seq = parent_seqs[i]
            all_ids.extend(ids)
            all_pos.extend(range(len(ids)))
            all_slots.extend(self.kv.prefill_slot_mapping(seq))
            cu.append(cu[-1] + len(ids))
            pack_rows.append(i)

        logits_by_req: List[Optional[torch.Tensor]] = [None] * len(requests)
        if pack_rows:
            batch = ForwardBatch(
                mode="prefill",
                positions=torch.tensor(all_pos, dtype=torch.long, device=dev),
                slot_mapping=torch.tensor(all_slots, dtype=torch.long, device=dev),
                kv_caches=self.kv.layer_caches(),
                cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            )
            input_ids = torch.tensor(all_ids, dtype=torch.long, device=dev)
            packed_logits = self.model.forward_prefill(input_ids, batch)  # [n_miss, V]
            for row, i in enumerate(pack_rows):
                logits_by_req[i] = packed_logits[row]
        hit_idx = [i for i in range(len(requests)) if matched[i] > 0]
        if hit_idx:
            # ALL cached-prefix tails go through ONE decode-mode forward
            # (rows from different sequences, per-row block tables)
            tail_logits = self._prefill_tails(
                [parent_seqs[i] for i in hit_idx],
                [requests[i].prompt_ids for i in hit_idx],
                [matched[i] for i in hit_idx],
            )
            for row, i in enumerate(hit_idx):
                logits_by_req[i] = tail_logits[row]
        prefill_logits = torch.stack([l for l in logits_by_req])  # [n_req, V]

        if self.prefix_cache is not None:
            for req, seq in zip(requests, parent_seqs):
                self.prefix_cache.register(req.prompt_ids, seq)

        # ---- fork + first-token sampling ------------------------------------
        outputs = [RequestOutput(prompt_tokens=len(r.prompt_ids)) for r in requests]
        self._fork_and_sample(requests, parent_seqs, prefill_logits, streams)
        return outputs

    def _prefill_tails(self, seqs: List[SequenceKV], ids_list: List[List[int]],
                       starts: List[int]) -> torch.Tensor:
        """One decode-mode forward over the concatenated TAILS of several
        prefix-cache-hit prompts (row i attends to its own sequence's KV,
        cached head included). Returns last-token logits per sequence
        [len(seqs), V]."""
        dev = self.device
        all_ids: List[int] = []
        all_pos: List[int] = []
        all_slots: List[int] = []
        ctx_lens: List[int] = []
        bt_blocks: List[List[int]] = []
        last_rows: List[int] = []
        for seq, ids, start in zip(seqs, ids_list, starts):
            slots_all = self.kv.prefill_slot_mapping(seq)
            for p in range(start, len(ids)):
                all_ids.append(ids[p])
                all_pos.append(p)
                all_slots.append(slots_all[p])
                ctx_lens.append(p + 1)
                bt_blocks.append(seq.blocks)
            last_rows.append(len(all_ids) - 1)
        max_nb = max(len(b) for b in bt_blocks)
        bt = torch.zeros((len(bt_blocks), max_nb), dtype=torch.int32)
        for r, blocks in enumerate(bt_blocks):
            bt[r, : len(blocks)] = torch.tensor(blocks, dtype=torch.int32)
        batch = ForwardBatch(
            mode="decode",
            positions=torch.tensor(all_pos, dtype=torch.long, device=dev),
            slot_mapping=torch.tensor(all_slots, dtype=torch.long, device=dev),
            kv_caches=self.kv.layer_caches(),
            block_tables=bt.to(dev),
            context_lens=torch.tensor(ctx_lens, dtype=torch.int32, device=dev),
        )
        hidden = self.model.forward_hidden(torch.tensor(all_ids, dtype=torch.long, device=dev), batch)
        return self.model.compute_logits(hidden[torch.tensor(last_rows, device=dev)])

    def _alloc_with_prefix(self, ids: List[int], prefix_blocks: List[int]) -> SequenceKV:
        """Sequence backed by cached prefix blocks (increfed) + freshly
        allocated tail blocks."""
        if not prefix_blocks:
            return self.kv.alloc_sequence(len(ids))
        bs = self.kv.block_size
        n_total = (len(ids) + bs - 1) // bs
        seq = SequenceKV(blocks=[], num_tokens=len(ids))
        try:
            for b in prefix_blocks:
                self.kv.allocator.incref(b)
                seq.blocks.append(b)
            for _ in range(n_total - len(prefix_blocks)):
                seq.blocks.append(self.kv.allocator.alloc())
        except Exception:
            for b in seq.blocks:
                self.kv.allocator.free(b)
            seq.blocks = []
            raise
        return seq

    def _fork_and_sample(
        self,
        requests: List[GenRequest],
        parent_seqs: List[SequenceKV],
        prefill_logits: torch.Tensor,
        streams: List[_Stream],
    ) -> List[_Stream]:
        """Fork each request's prompt KV into n streams (copy-on-write block
        refs) and sample the first token of every stream from that request's
        prefill logits row."""
        new_streams: List[_Stream] = []
        for ri, req in enumerate(requests):
            # unseeded requests draw from a per-engine monotonic counter, NOT
            # the batch-local index: sampled output must not depend on which
            # other requests were admitted together (the scheduler's
            # result-transparent merging contract)
            rid = self._seed_counter
            self._seed_counter += 1
            base_seed = req.sampling.seed if req.sampling.seed is not None else (self.config.seed * 1000003 + rid)
            for si in range(max(1, req.n)):
                seq = self.kv.fork(parent_seqs[ri])
                cstate = req.constraint.init_state() if req.constraint is not None else None
                st = _Stream(ri, si, seq, req.sampling, seed=base_seed + 7919 * si,
                             constraint=req.constraint, constraint_state=cstate)
                st.deadline = req.deadline if self.ctx.world_size == 1 else None
                streams.append(st)
                new_streams.append(st)
            self.kv.free_sequence(parent_seqs[ri])  # streams hold their own refs

        rep_logits = torch.cat([
            prefill_logits[ri].unsqueeze(0).expand(max(1, requests[ri].n), -1)
            for ri in range(len(requests))
        ])
        self._sample_and_append(rep_logits.contiguous(), new_streams)
        return new_streams

    def prefill_chunk(
        self, seq: SequenceKV, prompt_ids: List[int], start: int, end: int, want_logits: bool = False
    ) -> Optional[torch.Tensor]:
        """Prefill positions [start, end) of a prompt into an already-allocated
        sequence (chunked prefill: the scheduler slices long prompts between
        decode steps so running streams aren't stalled for a whole prefill).

        Chunks after the first must attend to KV written by earlier chunks,
        which lives in the paged cache — so this runs the DECODE forward path
        with one row per chunk token (row i: position start+i, context
        start+i+1). Identical math to the packed prefill (RoPE positions and
        causal structure preserved); the LM head runs only on the last token
        of the final chunk (want_logits=True)."""
        dev = self.device
        ids = torch.tensor(prompt_ids[start:end], dtype=torch.long, device=dev)
        pos = torch.arange(start, end, dtype=torch.long, device=dev)
        slots_all = self.kv.prefill_slot_mapping(seq)
        slots = torch.tensor(slots_all[start:end], dtype=torch.long, device=dev)
        nb = len(seq.blocks)
        bt = (
            torch.tensor(seq.blocks, dtype=torch.int32, device=dev)
            .unsqueeze(0).expand(end - start, nb).contiguous()
        )
        ctx = torch.arange(start + 1, end + 1, dtype=torch.int32, device=dev)
        batch = ForwardBatch(
            mode="decode",
            positions=pos,
            slot_mapping=slots,
            kv_caches=self.kv.layer_caches(),
            block_tables=bt,
            context_lens=ctx,
        )
        hidden = self.model.forward_hidden(ids, batch)
        if want_logits:
            return self.model.compute_logits(hidden[-1:])
        return None

    def finish_stream(self, s: _Stream) -> StreamOutput:
        """Phase 3: detokenize (unless a stop-string already trimmed the
        text) and release the stream's KV blocks."""
        if not s.out.text:
            s.out.text = self.tokenizer.decode(s.out.token_ids)
        self.kv.free_sequence(s.seq)
        return s.out

    def _generate_batch(
        self, requests: List[GenRequest], parent_seqs: List[SequenceKV], streams: List[_Stream]
    ) -> List[RequestOutput]:
        dev = self.device
        t0 = time.perf_counter()
        outputs = self.begin_requests(requests, parent_seqs, streams)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        t1 = time.perf_counter()

        # ---- decode loop (device-resident batch state) -----------------------
        active = [s for s in streams if not s.done]
        n_decode_steps = 0
        t_forward = 0.0
        t_sample = 0.0
        state: Optional[_DecodeBatchState] = None
        while active:
            ts = time.perf_counter()
            if state is None:
                state = _DecodeBatchState(self, active)
            logits = self._decode_step_state(state)
            tf = time.perf_counter()
            any_done = self._sample_state(state, logits)
            if any_done:
                active = [s for s in active if not s.done]
                state = None
            t_forward += tf - ts
            t_sample += time.perf_counter() - tf
            n_decode_steps += 1

        if dev.type == "cuda":
            torch.cuda.synchronize()
        t2 = time.perf_counter()
        self.last_timings = {
            "prefill_ms": (t1 - t0) * 1000,
            "decode_ms": (t2 - t1) * 1000,
            "decode_steps": n_decode_steps,
            "decode_forward_ms": t_forward * 1000,
            "decode_sample_ms": t_sample * 1000,
            "n_streams": len(streams),
        }

        # ---- collect ---------------------------------------------------------
        for s in streams:
            self.finish_stream(s)
        for s in streams:
            outputs[s.req_idx].streams.append(s.out)
        for o in outputs:
            o.prefill_ms = (t1 - t0) * 1000
            o.decode_ms = (t2 - t1) * 1000
        return outputs

    # --- decode internals -----------------------------------------------------
    def _decode_step_state(self, state: "_DecodeBatchState") -> torch.Tensor:
        """One forward over the persistent device-resident batch state: the
        only host work per step is block-allocator bookkeeping (real work only
        every block_size tokens)."""
        bs = self.kv.block_size
        bt_updates = []
        for i, s in enumerate(state.streams):
            p = s.seq.num_tokens
            if p % bs == 0:
                s.seq.blocks.append(self.kv.allocator.alloc())
                bt_updates.append((i, len(s.seq.blocks) - 1, s.seq.blocks[-1]))
            s.seq.num_tokens += 1
        if bt_updates:
            state.apply_bt_updates(bt_updates)

        slots = (
            state.bt.gather(1, (state.positions // bs).unsqueeze(1).to(torch.int64)).squeeze(1).to(torch.int64) * bs
            + state.positions % bs
        )
        batch = ForwardBatch(
            mode="decode",
            positions=state.positions,
            slot_mapping=slots,
            kv_caches=self.kv.layer_caches(),
            block_tables=state.bt,
            context_lens=state.ctx,
        )
        if self._graph_runner is not None:
            return self._graph_runner.run(state.ids, batch)
        return self.model.forward_decode(state.ids, batch)

    def _topk_logprobs(self, logits: torch.Tensor, mask: Optional[torch.Tensor], K: int):
        """Top-K alternatives from the same distribution the sampler scores:
        log-softmax of the MASKED, UNtempered logits (OpenAI `top_logprobs`
        semantics, matching ops.sample's chosen-token logprob)."""
        lf = logits.float()
        if mask is not None:
            V = lf.shape[1]
            bit = torch.arange(V, device=lf.device)
            allowed = (mask[:, torch.div(bit, 32, rounding_mode="floor")] >> (bit % 32).to(torch.int64)) & 1
            lf = torch.where(allowed.bool(), lf, torch.full_like(lf, float("-inf")))
        lp = torch.log_softmax(lf, dim=-1)
        vals, idx = lp.topk(K, dim=-1)
        return idx, vals

    def _append_topk(self, streams: List["_Stream"], K: int,
                     ids_flat: List[float], lps_flat: List[float]) -> None:
        for i, s in enumerate(streams):
            ks = s.sampling.top_logprobs
            if ks > 0:
                s.out.top_logprobs.append(
                    [(int(ids_flat[i * K + j]), lps_flat[i * K + j]) for j in range(min(ks, K))]
                )

    def _sample_state(self, state: "_DecodeBatchState", logits: torch.Tensor) -> bool:
        streams = state.streams
        logits = self._apply_penalties(logits, streams)
        mask = self._constraint_mask(streams)
        tokens, logprobs = ops.sample(
            logits, state.temps, state.top_ps, state.top_ks, state.seeds, state.steps, mask
        )
        K = max((s.sampling.top_logprobs for s in streams), default=0)
        parts = [tokens.to(torch.float64), logprobs.to(torch.float64)]
        if K > 0:
            tl_idx, tl_vals = self._topk_logprobs(logits, mask, K)
            parts += [tl_idx.reshape(-1).to(torch.float64), tl_vals.reshape(-1).to(torch.float64)]
        state.ids = tokens
        state.steps += 1
        state.positions += 1
        state.ctx += 1
        # ONE device->host transfer per step (tokens + logprobs [+ topk] packed)
        packed = torch.cat(parts).cpu()
        B = tokens.shape[0]
        tokens_l = [int(x) for x in packed[:B].tolist()]
        logprobs_l = packed[B : 2 * B].tolist()
        if K > 0:
            flat = packed[2 * B :].tolist()
            self._append_topk(streams, K, flat[: B * K], flat[B * K :])
        any_done = False
        for i, s in enumerate(streams):
            tok = tokens_l[i]
            s.step += 1
            s.last_token = tok
            if s.constraint is not None:
                s.constraint_state = s.constraint.advance(s.constraint_state, tok)
            s.out.token_ids.append(tok)
            s.out.logprobs.append(logprobs_l[i])
            self._check_stop(s)
            any_done |= s.done
        return any_done

    def _sample_and_append(self, logits: torch.Tensor, streams: List[_Stream]) -> None:
        """Sampling for the FIRST token (from the shared prefill logits);
        subsequent steps go through the device-resident _sample_state path."""
        logits = self._apply_penalties(logits, streams)
        temps, top_ps, top_ks, seeds, steps = self._sampling_tensors(streams)
        mask = self._constraint_mask(streams)
        tokens, logprobs = ops.sample(logits, temps, top_ps, top_ks, seeds, steps, mask)
        K = max((s.sampling.top_logprobs for s in streams), default=0)
        if K > 0:
            tl_idx, tl_vals = self._topk_logprobs(logits, mask, K)
            self._append_topk(streams, K, tl_idx.reshape(-1).tolist(), tl_vals.reshape(-1).tolist())
        tokens_l = tokens.tolist()
        logprobs_l = logprobs.tolist()
        for i, s in enumerate(streams):
            tok = tokens_l[i]
            s.step += 1
            s.last_token = tok
            if s.constraint is not None:
                s.constraint_state = s.constraint.advance(s.constraint_state, tok)
            s.out.token_ids.append(tok)
            s.out.logprobs.append(logprobs_l[i])
            self._check_stop(s)

    def _check_stop(self, s: _Stream) -> None:
        max_new = s.sampling.max_tokens or self.config.default_max_new_tokens
        if s.last_token in self.eos_token_ids:
            s.out.token_ids.pop()  # EOS itself is not part of the content
            s.out.logprobs.pop()
            if len(s.out.top_logprobs) > len(s.out.token_ids):
                s.out.top_logprobs.pop()
            s.out.finish_reason = "stop"
            s.done = True
            return
        if s.constraint is not None and s.constraint.is_final(s.constraint_state):
            s.out.finish_reason = "stop"
            s.done = True
            return
        if len(s.out.token_ids) >= max_new or s.seq.num_tokens >= self.config.max_seq_len - 1:
            s.out.finish_reason = "length"
            s.done = True
            return
        if s.deadline is not None and time.monotonic() >= s.deadline:
            s.out.finish_reason = "length"
            s.done = True
            return
        stops = s.sampling.stop_list
        if stops:
            text = self.tokenizer.decode(s.out.token_ids)
            for st in stops:
                idx = text.find(st)
                if idx >= 0:
                    s.out.text = text[:idx]
                    s.out.finish_reason = "stop"
                    s.done = True
                    return

    # --- embeddings ------------------------------------------------------------
    @torch.inference_mode()
    def embed(self, texts: List[str]) -> Tuple[List[List[float]], int]:
        """Local embedding path: token-embedding mean-pool, L2-normalized.
        The whole batch runs as ONE flat gather + segment-mean on device.
        Returns (vectors, total_tokens)."""
        unit, total = self.embed_dev(texts)
        out = unit.cpu().tolist()
        return out, total

    NGRAM_DIM = 512

    def _embedding_mode(self) -> str:
        mode = getattr(self.config, "embedding_mode", "auto")
        if mode != "auto":
            return mode
        # random-init token embeddings are semantic noise;
        # the signed n-gram hash tracks string similarity with no weights
        return "token_mean" if self.config.effective_weights_dir() else "ngram"

    def _embed_ngram_dev(self, texts: List[str]) -> Tuple[torch.Tensor, int]:
        """Deterministic signed char-3-gram hashing embedder: each lowercased
        text's padded 3-grams (utf-8 byte triples) hash (splitmix64) into one
        of NGRAM_DIM buckets with a ±1 sign bit; the L2-normalized count
        vector's cosine is a Jaccard-like string similarity — semantically
        meaningful for the consensus "embeddings" method without any model
        weights. Fully vectorized (one numpy pass per text): ~1000x the
        per-gram torch-indexing loop it replaced."""
        import numpy as np

        M64 = np.uint64(0xFFFFFFFFFFFFFFFF)
        out = np.zeros((len(texts), self.NGRAM_DIM), dtype=np.float32)
        total_tokens = 0
        for i, t in enumerate(texts):
            total_tokens += len(self.tokenizer.encode(t))
            if not t.strip():
                continue  # zero row for empty texts, matching token_mean
            b = f"  {t.lower()} ".encode("utf-8", errors="replace")
            a = np.frombuffer(b, dtype=np.uint8).astype(np.uint64)
            g = (a[:-2] << np.uint64(16)) | (a[1:-1] << np.uint64(8)) | a[2:]
            h = (g + np.uint64(0x9E3779B97F4A7C15)) & M64
            h ^= h >> np.uint64(30)
            h = (h * np.uint64(0xBF58476D1CE4E5B9)) & M64
            h ^= h >> np.uint64(27)
            h = (h * np.uint64(0x94D049BB133111EB)) & M64
            h ^= h >> np.uint64(31)
            idx = (h % np.uint64(self.NGRAM_DIM)).astype(np.int64)
            sign = (((h >> np.uint64(20)) & np.uint64(1)).astype(np.float64) * 2.0 - 1.0)
            row = np.bincount(idx, weights=sign, minlength=self.NGRAM_DIM)
            n = float(np.linalg.norm(row))
            if n > 1e-12:
                out[i] = row / n
        return torch.from_numpy(out).to(self.device), total_tokens

    def embed_dev(self, texts: List[str]) -> Tuple[torch.Tensor, int]:
        """Device-resident variant of embed(): returns the [N, H] float32
        unit-vector tensor ON the engine device (zero rows for empty texts),
        so the consensus accel can run its cosine GEMM without a host round
        trip (consolidation similarity math on-device)."""
        if self._embedding_mode() == "ngram":
            return self._embed_ngram_dev(texts)
        emb = self.model.embed_tokens.weight
        H = emb.shape[1]
        ids_per_text = [self.tokenizer.encode(t) for t in texts]
        total_tokens = sum(len(ids) for ids in ids_per_text)
        flat = [i for ids in ids_per_text for i in ids]
        if not flat:
            return torch.zeros(len(texts), H, device=self.device), 0
        flat_t = torch.tensor(flat, dtype=torch.long, device=self.device)
        seg = torch.tensor(
            [s for s, ids in enumerate(ids_per_text) for _ in ids],
            dtype=torch.long, device=self.device,
        )
        gathered = emb[flat_t].float()
        sums = torch.zeros(len(texts), H, device=self.device)
        sums.index_add_(0, seg, gathered)
        counts = torch.tensor(
            [max(1, len(ids)) for ids in ids_per_text], dtype=torch.float32, device=self.device
        ).unsqueeze(1)
        means = sums / counts
        norms = means.norm(dim=1, keepdim=True).clamp_min(1e-12)
        unit = torch.where(norms > 1e-11, means / norms, means)
        empties = [i for i, ids in enumerate(ids_per_text) if not ids]
        if empties:
            unit[torch.tensor(empties, device=self.device)] = 0.0
        return unit, total_tokens
