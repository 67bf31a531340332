"""The bench.py driver contract, exercised end-to-end on CPU:
`python bench.py --cpu --tiny --steps 1 --warmup 1` must run the real
SPMD bench loop (gloo world of 1) and print exactly one JSON line with
every field the driver parses. Guards the round-end BENCH/SCALE runs
against contract drift."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_bench_torchrun_two_rank_contract():
    """The driver's exact N>1 launcher (torch.distributed.run, one rank
    per GPU) on CPU/gloo: one JSON line from rank 0 with n_gpus=2 and
    weak-scaled global batch."""
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(29600 + os.getpid() % 300),
         os.path.join(REPO, "bench.py"), "--cpu", "--tiny",
         "--gpus", "2", "--steps", "1", "--warmup", "1"],
        capture_output=True, text=True, timeout=540, env=env, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, out.stdout  # rank 0 only
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["global_batch"] == 20  # 10 prompts/rank, weak
    assert rec["value"] > 0


@pytest.mark.timeout(600)
def test_bench_split_pool_roles():
    """--actors selects split actor/learner pools (bench role
    arithmetic): 2 ranks as 1 actor + 1 learner."""
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(29100 + os.getpid() % 300),
         os.path.join(REPO, "bench.py"), "--cpu", "--tiny",
         "--gpus", "2", "--steps", "1", "--warmup", "0", "--actors", "1"],
        capture_output=True, text=True, timeout=540, env=env, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = json.loads([l for l in out.stdout.splitlines()
                      if l.startswith("{")][0])
    assert "1 actors + 1 learners" in rec["config"]["parallelism"]
    assert rec["value"] > 0


@pytest.mark.timeout(600)
def test_bench_cpu_tiny_json_contract():
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    env["MASTER_PORT"] = str(26000 + os.getpid() % 500)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--cpu", "--tiny",
         "--steps", "1", "--warmup", "1"],
        capture_output=True, text=True, timeout=540, env=env, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, out.stdout
    rec = json.loads(json_lines[0])

    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in rec, field
    assert rec["metric"] == "rl_samples_per_sec"
    assert rec["unit"] == "samples/s"
    assert rec["n_gpus"] == 1 and rec["steps"] == 1 and rec["warmup"] == 1
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    # value consistency: samples/s x s/step == samples/step
    samples_per_step = rec["value"] * rec["ms_per_step"] / 1e3
    assert samples_per_step == pytest.approx(
        rec["config"]["samples_per_step"], rel=0.05)
    # tiny/CPU mode must never claim a baseline comparison
    assert rec["vs_baseline"] is None
    for k in ("model", "global_batch", "num_candidates", "samples_per_step",
              "max_new_tokens", "parallelism"):
        assert k in rec["config"], k
