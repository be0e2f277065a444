"""Continuous-batching scheduler.

The production serving layer over LLMEngine: concurrent callers submit
requests that MERGE into the running decode batch instead of waiting for the
whole previous batch to finish. One worker thread owns the engine:

    loop:
      - drain newly-submitted requests (respecting the stream budget)
      - prefill + fork them (engine.begin_requests) and add their streams
        to the active set
      - run ONE decode step over all active streams
      - retire finished streams; resolve a request's future when its last
        stream completes

Merging is RESULT-TRANSPARENT: sampling is counter-based per (seed, step),
so a stream draws the same random sequence regardless of which other
requests share its batch (greedy decodes are bitwise-insensitive to batch
composition up to GEMM reduction-order noise).

The async client (AsyncKLLMs) routes through this scheduler, which is what
makes many concurrent ``await client.chat.completions.create(...)`` calls
share prefill batches and decode steps.
"""

from __future__ import annotations

import queue
import threading
from concurrent.futures import Future
from dataclasses import dataclass
from typing import Dict, List, Optional

from .engine import GenRequest, LLMEngine, RequestOutput, _DecodeBatchState, _Stream


class _WorkerContext:
    """Mutable state owned by the scheduler worker thread."""

    def __init__(self):
        self.active: List[_Stream] = []
        self.stream_ticket: Dict[int, "_Ticket"] = {}   # id(stream) -> ticket
        self.state: Optional[_DecodeBatchState] = None
        self.pending: List["_PendingPrefill"] = []       # chunked prefills in flight


@dataclass
class _Ticket:
    request: GenRequest
    future: Future
    output: Optional[RequestOutput] = None
    remaining: int = 0


@dataclass
class _PendingPrefill:
    """A long prompt being prefilled in slices (config.prefill_chunk_tokens):
    one chunk advances per scheduler loop iteration, between decode steps."""
    ticket: _Ticket
    seq: object          # SequenceKV holding the whole prompt's blocks
    next_pos: int = 0


class BatchScheduler:
    def __init__(self, engine: LLMEngine, admit_wait_s: float = 0.002, engine_lock=None):
        self.engine = engine
        self.admit_wait_s = admit_wait_s
        # serializes engine access against direct (non-scheduled) callers
        self.engine_lock = engine_lock or threading.Lock()
        self._queue: "queue.Queue[_Ticket]" = queue.Queue()
        # TP>1 serving: rank 0 attaches a parallel.serve.TPCoordinator that
        # broadcasts each engine-touching phase to the follower ranks
        self.coordinator = None
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()
        self._stop = threading.Event()
        # observability
        self.steps = 0
        self.admitted_batches = 0
        # cumulative phase timers (seconds) — where the worker thread spends
        # its time; read via .stats for bench diagnostics
        self.t_drain = 0.0
        self.t_admit = 0.0
        self.t_prefill = 0.0
        self.t_step = 0.0

    @property
    def stats(self) -> dict:
        return {
            "steps": self.steps,
            "admitted_batches": self.admitted_batches,
            "drain_s": round(self.t_drain, 4),
            "admit_s": round(self.t_admit, 4),
            "chunked_prefill_s": round(self.t_prefill, 4),
            "decode_step_s": round(self.t_step, 4),
        }

    # --- public -----------------------------------------------------------
    def submit(self, request: GenRequest) -> Future:
        fut: Future = Future()
        self._queue.put(_Ticket(request=request, future=fut))
        self._ensure_thread()
        return fut

    def shutdown(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=30)

    # --- worker -----------------------------------------------------------
    def _ensure_thread(self) -> None:
        with self._lock:
            if self._thread is None or not self._thread.is_alive():
                self._stop.clear()
                self._thread = threading.Thread(target=self._loop, name="kllms-scheduler", daemon=True)
                self._thread.start()

    def _drain(self, active_streams: int, block: bool) -> List[_Ticket]:
        """Admit pending tickets up to the engine's stream budget.

        When the engine is idle (``block``), concurrent submissions arrive as
        a burst staggered over a few ms; a short rolling admission window
        coalesces the whole burst into ONE packed prefill batch instead of
        fragmenting it (a fragmented admission costs an extra prefill plus
        the late wave's decode tail — ~100 ms/step on the n=5 batch-24
        bench). While decode streams are running, drains stay non-blocking
        so steps are never stalled."""
        budget = self.engine.config.max_batch_size - active_streams
        tickets: List[_Ticket] = []
        try:
            while budget > 0:
                if not tickets:
                    t = self._queue.get(timeout=self.admit_wait_s) if block else self._queue.get_nowait()
                elif block:
                    # rolling window: a 1 ms silent gap ends the admission
                    t = self._queue.get(timeout=0.001)
                else:
                    t = self._queue.get_nowait()
                need = max(1, t.request.n)
                if tickets and need > budget:
                    # keep order: put it back and stop admitting this round
                    self._queue.put(t)
                    break
                tickets.append(t)
                budget -= need
        except queue.Empty:
            pass
        return tickets

    def _loop(self) -> None:
        import time as _time

        import torch

        ctx = _WorkerContext()
        with torch.inference_mode():
            while not self._stop.is_set():
                idle = not ctx.active and not ctx.pending
                # pending chunked prefills reserve their future stream slots
                reserved = sum(max(1, p.ticket.request.n) for p in ctx.pending)
                t0 = _time.perf_counter()
                tickets = self._drain(len(ctx.active) + reserved, block=idle)
                t1 = _time.perf_counter()
                self.t_drain += t1 - t0
                coord = self.coordinator
                with self.engine_lock:
                    if tickets:
                        if coord is not None:
                            coord.admit([t.request for t in tickets])
                        self._admit(ctx, tickets)
                        t2 = _time.perf_counter()
                        self.t_admit += t2 - t1
                        t1 = t2
                    if ctx.pending:
                        if coord is not None:
                            coord.advance_prefill()
                        self._advance_prefill(ctx)
                        t2 = _time.perf_counter()
                        self.t_prefill += t2 - t1
                        t1 = t2
                    if ctx.active:
                        if coord is not None:
                            coord.step()
                        self._step(ctx)
                        self.t_step += _time.perf_counter() - t1

    def _admit(self, ctx: "_WorkerContext", tickets: List[_Ticket]) -> None:
        chunk = self.engine.config.prefill_chunk_tokens
        if chunk:
            long_tickets = [t for t in tickets if len(t.request.prompt_ids) > chunk]
            tickets = [t for t in tickets if len(t.request.prompt_ids) <= chunk]
            for t in long_tickets:
                self._start_chunked(ctx, t)
        if tickets:
            self._admit_whole(ctx, tickets)

    def _start_chunked(self, ctx: "_WorkerContext", t: _Ticket) -> None:
        eng = self.engine
        ids = t.request.prompt_ids
        try:
            pb = (eng.prefix_cache.match(ids, eng.config.prefix_cache_min_tokens)
                  if eng.prefix_cache is not None else [])
            seq = eng._alloc_with_prefix(ids, pb)
        except Exception as e:
            t.future.set_exception(e)
            return
        t.output = RequestOutput(prompt_tokens=len(ids))
        t.remaining = max(1, t.request.n)
        # cached prefix blocks are already filled: start chunking after them
        ctx.pending.append(_PendingPrefill(ticket=t, seq=seq,
                                           next_pos=len(pb) * eng.kv.block_size))
        self.admitted_batches += 1

    def _advance_prefill(self, ctx: "_WorkerContext") -> None:
        """Run ONE chunk of the head-of-line pending prefill; on the final
        chunk, fork the n streams and sample their first token."""
        eng = self.engine
        p = ctx.pending[0]
        t = p.ticket
        ids = t.request.prompt_ids
        chunk = eng.config.prefill_chunk_tokens or len(ids)
        end = min(p.next_pos + chunk, len(ids))
        final = end == len(ids)
        popped = False
        new_streams: List[_Stream] = []
        try:
            logits = eng.prefill_chunk(p.seq, ids, p.next_pos, end, want_logits=final)
            p.next_pos = end
            if not final:
                return
            ctx.pending.pop(0)
            popped = True
            if eng.prefix_cache is not None:
                eng.prefix_cache.register(ids, p.seq)
            eng._fork_and_sample([t.request], [p.seq], logits, new_streams)
        except Exception as e:
            # pop exactly once: only if the try block didn't get there
            if not popped:
                ctx.pending.pop(0)
            if p.seq.blocks:
                eng.kv.free_sequence(p.seq)
            for st in new_streams:
                if st.seq.blocks:
                    eng.kv.free_sequence(st.seq)
            if not t.future.done():
                t.future.set_exception(e)
            return
        for st in new_streams:
            ctx.stream_ticket[id(st)] = t
        self._retire(eng, new_streams, ctx.stream_ticket)
        ctx.active.extend(st for st in new_streams if not st.done)
        ctx.state = None

    def _admit_whole(self, ctx: "_WorkerContext", tickets: List[_Ticket]) -> None:
        eng = self.engine
        reqs = [t.request for t in tickets]
        parent_seqs: List = []
        new_streams: List[_Stream] = []
        try:
            outputs = eng.begin_requests(reqs, parent_seqs, new_streams)
        except Exception as e:
            for seq in parent_seqs:
                if seq.blocks:
                    eng.kv.free_sequence(seq)
            for st in new_streams:
                if st.seq.blocks:
                    eng.kv.free_sequence(st.seq)
            for t in tickets:
                t.future.set_exception(e)
            return
        for t, out, req in zip(tickets, outputs, reqs):
            t.output = out
            t.remaining = max(1, req.n)
        expanded = [t for t in tickets for _ in range(max(1, t.request.n))]
        for st, t in zip(new_streams, expanded):
            ctx.stream_ticket[id(st)] = t
        # first-token sampling may already have finished streams
        self._retire(eng, new_streams, ctx.stream_ticket)
        ctx.active.extend(st for st in new_streams if not st.done)
        ctx.state = None
        self.admitted_batches += 1

    def _step(self, ctx: "_WorkerContext") -> None:
        eng = self.engine
        try:
            if ctx.state is None:
                ctx.state = _DecodeBatchState(eng, ctx.active)
            logits = eng._decode_step_state(ctx.state)
            any_done = eng._sample_state(ctx.state, logits)
            self.steps += 1
        except Exception as e:
            # poisoned batch: fail every in-flight request, free KV
            for st in ctx.active:
                if st.seq.blocks:
                    eng.kv.free_sequence(st.seq)
                t = ctx.stream_ticket.pop(id(st), None)
                if t is not None and not t.future.done():
                    t.future.set_exception(e)
            ctx.active = []
            ctx.state = None
            return
        if any_done:
            self._retire(eng, ctx.active, ctx.stream_ticket)
            ctx.active = [st for st in ctx.active if not st.done]
            ctx.state = None

    def _retire(self, eng: LLMEngine, streams: List[_Stream], stream_ticket: Dict[int, "_Ticket"]) -> None:
        for s in streams:
            if not s.done:
                continue
            t = stream_ticket.pop(id(s), None)
            if t is None:
                continue
            eng.finish_stream(s)
            t.output.streams.append(s.out)
            t.remaining -= 1
            if t.remaining == 0:
                # order by stream index for API parity with generate()
                t.output.streams.sort(key=lambda o: o.stream_idx)
                t.future.set_result(t.output)
