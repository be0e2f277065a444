"""Driver-contract test for bench.py: one JSON line on stdout with the
exact fields and semantics the round driver parses (BASELINE.json metric,
whole-job aggregate value, ms_per_step consistency)."""

import json
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(600)
def test_bench_json_contract(tmp_path):
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "0",
         "--model", "tiny-llama", "--batch", "2", "--max-new", "8",
         "--prompt-len", "64"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        capture_output=True, text=True, timeout=500, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.strip()]
    assert len(lines) == 1, f"exactly ONE json line expected, got {len(lines)}"
    d = json.loads(lines[0])

    assert d["metric"] == "end_to_end_consensus_requests_per_s"
    assert d["unit"] == "req/s"
    assert d["higher_is_better"] is True
    assert d["scaling"] in ("weak", "strong")
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 0
    assert d["value"] > 0
    assert d["vs_baseline"] == pytest.approx(d["value"] / 1.2, rel=1e-3)
    # value is the WHOLE-JOB rate; ms_per_step must be consistent with it
    assert d["ms_per_step"] == pytest.approx(
        1000.0 * d["config"]["global_batch"] / d["value"], rel=1e-2)
    assert d["dtype"] in ("bf16", "float32")
    assert "synthetic" in d["data"]
    cfg = d["config"]
    for k in ("model", "n", "global_batch", "seq_len", "parallelism",
              "completions_per_s", "consensus_latency_s_per_request",
              "batch_wall_s"):
        assert k in cfg, k
    assert cfg["parallelism"] == "dp1 tp1"


@pytest.mark.timeout(600)
def test_bench_driver_multiproc_invocation(tmp_path):
    """The EXACT shape the driver uses for SCALE_rNN: torch.distributed.run
    with N ranks — on CPU this exercises the gloo/dp path end to end and
    must still print exactly ONE JSON line (rank 0) with the whole-job
    aggregate value."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29841", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "0", "--model", "tiny-llama",
         "--batch", "2", "--max-new", "6", "--prompt-len", "48"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        capture_output=True, text=True, timeout=500, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.strip().startswith("{")]
    assert len(lines) == 1, f"exactly ONE json line expected: {out.stdout[-800:]}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["scaling"] == "weak"
    assert d["config"]["parallelism"].startswith("dp2")
    # whole-job aggregate: 2 ranks x per-rank batch
    assert d["config"]["global_batch"] == 4
    assert d["value"] > 0


@pytest.mark.timeout(600)
def test_bench_tp_serving_invocation(tmp_path):
    """bench.py --parallel tp: rank 0 drives the scheduler, rank 1 replays
    (TPCoordinator/TPFollower) — the full TP serving stack under the bench
    entry, on CPU/gloo."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29843", "bench.py", "--gpus", "2",
         "--parallel", "tp", "--steps", "2", "--warmup", "0",
         "--model", "tiny-llama", "--batch", "2", "--max-new", "6",
         "--prompt-len", "48"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        capture_output=True, text=True, timeout=500, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.strip().startswith("{")]
    assert len(lines) == 1, f"exactly ONE json line expected: {out.stdout[-800:]}"
    d = json.loads(lines[0])
    assert d["config"]["parallelism"] == "tp2"
    assert d["value"] > 0
