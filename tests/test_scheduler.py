"""Continuous-batching scheduler tests: result parity with generate(),
concurrent-submission merging, failure isolation."""

import asyncio
import threading
import time

import pytest

from kllms_amd.engine.config import EngineConfig
from kllms_amd.engine.engine import GenRequest, LLMEngine
from kllms_amd.engine.sampling import SamplingParams
from kllms_amd.engine.scheduler import BatchScheduler


@pytest.fixture(scope="module")
def engine():
    return LLMEngine(EngineConfig(
        model="tiny-llama", max_kv_blocks=512, use_hip_graphs=False,
        device="cpu", seed=0, max_batch_size=64,
    ))


def greedy_req(prompt, max_tokens=8, n=1):
    return GenRequest(prompt_ids=prompt, n=n,
                      sampling=SamplingParams(temperature=0.0, max_tokens=max_tokens))


class TestScheduler:
    def test_single_request_matches_generate(self, engine):
        direct = engine.generate([greedy_req([1, 2, 3, 4], n=3)])[0]
        sched = BatchScheduler(engine)
        out = sched.submit(greedy_req([1, 2, 3, 4], n=3)).result(timeout=120)
        sched.shutdown()
        assert [s.token_ids for s in out.streams] == [s.token_ids for s in direct.streams]
        assert [s.stream_idx for s in out.streams] == [0, 1, 2]

    def test_concurrent_submissions_merge_and_match(self, engine):
        prompts = [[1 + i, 7, 3, 9] for i in range(6)]
        direct = [engine.generate([greedy_req(p, 10, n=2)])[0] for p in prompts]

        sched = BatchScheduler(engine, admit_wait_s=0.05)
        futs = [sched.submit(greedy_req(p, 10, n=2)) for p in prompts]
        outs = [f.result(timeout=120) for f in futs]
        merged_batches = sched.admitted_batches
        sched.shutdown()

        for o, d in zip(outs, direct):
            assert [s.token_ids for s in o.streams] == [s.token_ids for s in d.streams]
        # 6 requests submitted back-to-back must NOT need 6 separate batches
        assert merged_batches < 6, f"no merging happened ({merged_batches} batches)"

    def test_mid_decode_admission(self, engine):
        """A request submitted while another decodes must still finish and
        match its solo output (merged into the running batch)."""
        sched = BatchScheduler(engine)
        f1 = sched.submit(greedy_req([5, 5, 5], max_tokens=40))
        time.sleep(0.2)  # let decode start
        f2 = sched.submit(greedy_req([9, 8, 7], max_tokens=6))
        o2 = f2.result(timeout=120)
        o1 = f1.result(timeout=120)
        sched.shutdown()
        d1 = engine.generate([greedy_req([5, 5, 5], max_tokens=40)])[0]
        d2 = engine.generate([greedy_req([9, 8, 7], max_tokens=6)])[0]
        assert o1.streams[0].token_ids == d1.streams[0].token_ids
        assert o2.streams[0].token_ids == d2.streams[0].token_ids

    def test_failure_isolated(self, engine):
        sched = BatchScheduler(engine)
        bad = GenRequest(prompt_ids=[], n=1, sampling=SamplingParams(max_tokens=4))
        with pytest.raises(Exception):
            sched.submit(bad).result(timeout=60)
        # scheduler still serves afterwards
        ok = sched.submit(greedy_req([1, 2], 4)).result(timeout=120)
        sched.shutdown()
        assert len(ok.streams) == 1


class TestAsyncClientScheduling:
    def test_concurrent_async_calls(self):
        from kllms_amd import AsyncKLLMs

        ak = AsyncKLLMs(model="tiny-llama", max_kv_blocks=512, use_hip_graphs=False,
                        device="cpu", default_max_new_tokens=8)

        async def one(i):
            return await ak.chat.completions.create(
                messages=[{"role": "user", "content": f"req {i}"}],
                model="tiny-llama", n=2, max_tokens=6, temperature=0.0,
            )

        async def run():
            return await asyncio.gather(*[one(i) for i in range(4)])

        results = asyncio.run(run())
        assert len(results) == 4
        for r in results:
            assert len(r.choices) == 3
        # scheduler was engaged and merged work
        sched = ak.client._scheduler
        assert sched is not None and sched.admitted_batches >= 1
