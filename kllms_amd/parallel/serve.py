"""TP>1 serving: rank-0 request broadcast + lockstep follower replay.

The reference scales invisibly behind its remote API
(/root/reference/k_llms/resources/completions/completions.py:73); locally,
a tensor-parallel engine spans ALL ranks, so every rank must execute the
SAME sequence of collective-bearing engine calls (prefill batches, decode
steps) with identical arguments or the RCCL all-reduces deadlock.

Design (one decision maker, deterministic replay):

- Rank 0 runs the public client + continuous-batching scheduler unchanged.
  A ``TPCoordinator`` attached to the scheduler broadcasts a compact action
  record over a dedicated gloo control group BEFORE each engine-touching
  scheduler phase: ``admit`` (with the serialized requests), ``advance``
  (one chunk of a pending chunked prefill), ``step`` (one decode step).
- Ranks 1..N-1 run ``TPFollower.run()``: receive actions and replay them
  through THE SAME ``BatchScheduler`` methods over a private worker context
  (synthesized tickets, ignored futures). All follower-side state evolution
  (stream retirement, KV growth, chunk boundaries, RNG) is derived from
  engine results that are bitwise-identical across ranks: TP all-reduce
  output is identical on every rank, the LM head is replicated, sampling is
  counter-based on (seed, step), and unseeded requests draw seeds from a
  per-engine monotonic counter that advances in admission order — so the
  followers never diverge from rank 0's decisions.

The control plane is gloo (host memory) so tiny action records never touch
the GPU or interleave with RCCL compute collectives.

CONSTRAINT: at TP>1 every engine-touching call must go through the
scheduler (the async client's default path). A direct rank-0-only
``generate()`` — or the opt-in llm string-consensus mode, which issues its
own create() from inside consolidation — would run TP collectives on rank 0
alone and deadlock. The default (centroid) consensus mode never does this;
the embeddings path (engine.embed/embed_dev) is collective-free by design.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import torch
import torch.distributed as dist


def _serialize_request(req) -> Dict[str, Any]:
    c = req.constraint
    return {
        "prompt_ids": list(req.prompt_ids),
        "n": req.n,
        "sampling": req.sampling.model_dump(),
        "constraint": None if c is None else {"schema": c.schema, "whitespace": getattr(c, "whitespace", False)},
    }


def _deserialize_request(d: Dict[str, Any], engine):
    from ..engine.constrained import JsonSchemaConstraint
    from ..engine.engine import GenRequest
    from ..engine.sampling import SamplingParams

    constraint = None
    if d["constraint"] is not None:
        constraint = JsonSchemaConstraint(
            d["constraint"]["schema"], engine.tokenizer, whitespace=d["constraint"]["whitespace"]
        )
    return GenRequest(
        prompt_ids=d["prompt_ids"], n=d["n"],
        sampling=SamplingParams(**d["sampling"]), constraint=constraint,
    )


class _ControlGroup:
    """Gloo subgroup for action broadcasts (collective: every rank constructs
    one, in the same order, exactly once)."""

    def __init__(self):
        assert dist.is_initialized(), "TP serving needs torch.distributed initialized"
        self.world = dist.get_world_size()
        self.rank = dist.get_rank()
        self.group = dist.new_group(ranks=list(range(self.world)), backend="gloo")

    def bcast(self, obj: Optional[Any]) -> Any:
        lst = [obj]
        dist.broadcast_object_list(lst, src=0, group=self.group)
        return lst[0]


class TPCoordinator:
    """Attached to rank 0's BatchScheduler as ``scheduler.coordinator``; the
    scheduler calls these hooks (under its engine lock) right before the
    corresponding engine-touching phase."""

    def __init__(self, engine):
        self.engine = engine
        self.ctrl = _ControlGroup()
        assert self.ctrl.rank == 0, "TPCoordinator belongs on rank 0"
        self._stopped = False

    def admit(self, requests: List) -> None:
        self.ctrl.bcast(("admit", [_serialize_request(r) for r in requests]))

    def advance_prefill(self) -> None:
        self.ctrl.bcast(("advance",))

    def step(self) -> None:
        self.ctrl.bcast(("step",))

    def stop(self) -> None:
        if not self._stopped:
            self._stopped = True
            self.ctrl.bcast(("stop",))


class TPFollower:
    """Ranks 1..N-1: replay rank 0's scheduler actions in lockstep."""

    def __init__(self, engine):
        self.engine = engine
        self.ctrl = _ControlGroup()
        assert self.ctrl.rank != 0, "rank 0 drives the scheduler, not a follower"

    def run(self) -> None:
        from concurrent.futures import Future

        from ..engine.scheduler import BatchScheduler, _Ticket, _WorkerContext

        sched = BatchScheduler(self.engine)  # worker thread never started
        ctx = _WorkerContext()
        with torch.inference_mode():
            while True:
                action = self.ctrl.bcast(None)
                kind = action[0]
                if kind == "stop":
                    break
                with sched.engine_lock:
                    if kind == "admit":
                        reqs = [_deserialize_request(d, self.engine) for d in action[1]]
                        tickets = [_Ticket(request=r, future=Future()) for r in reqs]
                        sched._admit(ctx, tickets)
                    elif kind == "advance":
                        sched._advance_prefill(ctx)
                    elif kind == "step":
                        sched._step(ctx)
                    else:  # pragma: no cover - protocol error
                        raise RuntimeError(f"unknown TP action {kind!r}")
