"""TP>1 serving over the continuous-batching scheduler (gloo, world_size=2).

VERDICT round-1 item 2: the scheduler must serve TP>1 — rank 0 broadcasts
every engine-touching phase (admit / chunked-prefill advance / decode step)
and the follower ranks replay them in lockstep (kllms_amd/parallel/serve.py).
Covers: plain create, n>1 fan-out, constrained (parse) requests crossing the
broadcast boundary, chunked prefill, and greedy parity with a TP=1 engine.
"""

import multiprocessing as mp
import os

import pytest

PORT = 29791


def _mk_engine_kwargs(tp, world=None):
    model = "tiny-llama-kv4" if (world or tp) >= 4 else "tiny-llama"
    return dict(model=model, tp_size=tp, max_kv_blocks=512,
                use_hip_graphs=False, device="cpu", seed=0,
                default_max_new_tokens=8, max_seq_len=512,
                prefill_chunk_tokens=16)


def _serving_worker(rank: int, world_size: int, q):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(PORT + world_size)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from kllms_amd.engine.api import LocalEngineClient
        from kllms_amd.parallel.serve import TPCoordinator, TPFollower

        client = LocalEngineClient(**_mk_engine_kwargs(world_size))
        eng = client.engine

        if rank != 0:
            TPFollower(eng).run()
            q.put((rank, "follower-done"))
            return

        sched = client.scheduler
        coord = TPCoordinator(eng)
        sched.coordinator = coord

        from pydantic import BaseModel

        class Answer(BaseModel):
            value: int

        import concurrent.futures as cf

        with cf.ThreadPoolExecutor(4) as pool:
            f1 = pool.submit(client.chat_completions_create, True,
                             messages=[{"role": "user", "content": "count to three"}],
                             n=3, temperature=0.0, max_tokens=6, seed=5)
            f2 = pool.submit(client.chat_completions_parse, True,
                             messages=[{"role": "user", "content": "give a number"}],
                             response_format=Answer, n=2, temperature=0.0, max_tokens=12, seed=7)
            # long prompt -> chunked prefill path ('advance' actions)
            f3 = pool.submit(client.chat_completions_create, True,
                             messages=[{"role": "user", "content": "z" * 80}],
                             n=1, temperature=0.0, max_tokens=4, seed=9)
            r1, r2, r3 = f1.result(120), f2.result(120), f3.result(120)

        sched.shutdown()
        coord.stop()
        q.put((rank, {
            "r1": [c.message.content for c in r1.choices],
            "r2": [c.message.content for c in r2.choices],
            "r3": [c.message.content for c in r3.choices],
        }))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
@pytest.mark.parametrize("world", [2, 4])
def test_tp_scheduler_serving_matches_tp1(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_serving_worker, args=(r, world, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, payload = q.get(timeout=480)
        results[rank] = payload
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    for r in range(1, world):
        assert results[r] == "follower-done"
    tp2 = results[0]
    assert len(tp2["r1"]) == 3 and len(tp2["r2"]) == 2 and len(tp2["r3"]) == 1
    # greedy n=3: all streams identical
    assert tp2["r1"][0] == tp2["r1"][1] == tp2["r1"][2]

    # TP=1 reference (same seeds, same engine config, scheduler path)
    from pydantic import BaseModel

    from kllms_amd.engine.api import LocalEngineClient

    class Answer(BaseModel):
        value: int

    client = LocalEngineClient(**_mk_engine_kwargs(1, world=world))
    r1 = client.chat_completions_create(True,
        messages=[{"role": "user", "content": "count to three"}],
        n=3, temperature=0.0, max_tokens=6, seed=5)
    r2 = client.chat_completions_parse(True,
        messages=[{"role": "user", "content": "give a number"}],
        response_format=Answer, n=2, temperature=0.0, max_tokens=12, seed=7)
    r3 = client.chat_completions_create(True,
        messages=[{"role": "user", "content": "z" * 80}],
        n=1, temperature=0.0, max_tokens=4, seed=9)
    client.scheduler.shutdown()

    assert tp2["r1"] == [c.message.content for c in r1.choices]
    assert tp2["r2"] == [c.message.content for c in r2.choices]
    assert tp2["r3"] == [c.message.content for c in r3.choices]


class TestCollectiveDispatch:
    """CPU-testable logic of the size-switched custom collective."""

    def test_should_use_gates(self):
        import torch

        from kllms_amd.parallel.collective import CustomAllReduce

        car = CustomAllReduce.__new__(CustomAllReduce)   # no IPC init on CPU
        car.max_bytes = 4 << 20
        ok = torch.zeros(4096, dtype=torch.bfloat16)
        assert car.should_use(ok)
        assert not car.should_use(ok.float())                      # dtype
        assert not car.should_use(torch.zeros(4097, dtype=torch.bfloat16)[:4095])  # numel%8
        assert not car.should_use(torch.zeros(3 << 20, dtype=torch.bfloat16))      # too big
        t = torch.zeros(16, 16, dtype=torch.bfloat16).t()
        assert not car.should_use(t)                               # non-contiguous

    def test_env_disable(self, monkeypatch):
        import torch

        from kllms_amd.parallel.collective import maybe_init_custom_allreduce
        from kllms_amd.parallel.tp import ParallelContext

        monkeypatch.setenv("KLLMS_CUSTOM_AR", "0")
        ctx = ParallelContext(world_size=2, rank=0)
        assert maybe_init_custom_allreduce(ctx, torch.device("cpu")) is None

    def test_serialize_roundtrip_constraint(self):
        """GenRequest (de)serialization across the TP control plane keeps
        constrained-decoding state (schema + whitespace flag)."""
        from kllms_amd.engine.config import EngineConfig
        from kllms_amd.engine.constrained import JsonSchemaConstraint
        from kllms_amd.engine.engine import GenRequest, LLMEngine
        from kllms_amd.engine.sampling import SamplingParams
        from kllms_amd.parallel.serve import _deserialize_request, _serialize_request

        eng = LLMEngine(EngineConfig(model="tiny-llama", max_kv_blocks=64,
                                     use_hip_graphs=False, device="cpu"))
        schema = {"type": "object", "properties": {"v": {"type": "integer"}}, "required": ["v"]}
        c = JsonSchemaConstraint(schema, eng.tokenizer, whitespace=True)
        req = GenRequest(prompt_ids=[1, 2, 3], n=2,
                         sampling=SamplingParams(temperature=0.5, max_tokens=7, seed=3),
                         constraint=c)
        d = _serialize_request(req)
        import pickle
        d = pickle.loads(pickle.dumps(d))    # the broadcast pickles it
        req2 = _deserialize_request(d, eng)
        assert req2.prompt_ids == req.prompt_ids and req2.n == 2
        assert req2.sampling.model_dump() == req.sampling.model_dump()
        assert req2.constraint.whitespace is True
        assert (req2.constraint.next_state == c.next_state).all()
