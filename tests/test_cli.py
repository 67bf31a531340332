"""End-to-end CLI test: `python train_distributed.py` (the reference's
entry point, flag for flag) on CPU with the synthetic dataset, including
a `--resume` continuation — the exact surface a user drives."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

BASE_ARGS = [
    "--model", "tiny-qwen2", "--synthetic_dataset", "10",
    "--batch_size", "4", "--num_candidates", "2", "--topk", "2",
    "--max_new_tokens", "12", "--max_prompt_tokens", "48",
    "--number_of_actors", "1", "--number_of_learners", "1",
    "--learner_chunk_size", "2", "--train_batch_size", "2",
    "--max_lora_rank", "4", "--lora_alpha", "8",
    "--eval_every", "0", "--save_every", "100",
    "--backend_device", "cpu", "--run_name", "cli",
    "--lora_save_path", "cli_adapter",
]


def _run(tmpdir, extra):
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "train_distributed.py"),
         *BASE_ARGS, *extra],
        capture_output=True, text=True, timeout=540, env=env, cwd=tmpdir)
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-3000:])
    return out


@pytest.mark.timeout(600)
def test_cli_eval_only(tmp_path):
    """--eval_only scores the test split (pass@1 / BoN@8) and exits
    without training artifacts."""
    tmpdir = str(tmp_path)
    _run(tmpdir, ["--episodes", "5", "--eval_only", "--run_name", "ev"])
    recs = [json.loads(l) for l in
            open(os.path.join(tmpdir, "metrics_ev.jsonl"))]
    assert any("eval/pass@1(mean8)" in r and "eval/BoN(8)" in r
               for r in recs)
    assert not any("loss" in r for r in recs)
    assert not os.path.exists(os.path.join(tmpdir, "run_ev"))


@pytest.mark.timeout(1200)
def test_cli_train_and_resume(tmp_path):
    tmpdir = str(tmp_path)
    out = _run(tmpdir, ["--episodes", "1"])
    assert "Number of train samples: 9" in out.stdout
    assert "Sample problem:" in out.stdout  # per-round dump (reference parity)

    # artifacts: per-round adapter + end-of-episode full checkpoint
    assert os.path.exists(os.path.join(tmpdir, "cli_adapter",
                                       "adapter_model.safetensors"))
    run_dir = os.path.join(tmpdir, "run_cli")
    ckpts = [d for d in os.listdir(run_dir) if d.startswith("model_")]
    assert ckpts
    ckpt = os.path.join(run_dir, sorted(ckpts)[-1])
    for f in ("trainer_state.pt", "optimizer_state.pt",
              "engine_state_rank0.pt", "engine_state_rank1.pt",
              "adapter_config.json"):
        assert os.path.exists(os.path.join(ckpt, f)), f
    with open(os.path.join(ckpt, "adapter_config.json")) as fh:
        assert json.load(fh)["r"] == 4

    # metrics written with the reference key set
    mfile = os.path.join(tmpdir, "metrics_cli.jsonl")
    recs = [json.loads(l) for l in open(mfile)]
    assert any("loss" in r for r in recs)

    # resume from the checkpoint and run one more episode
    _run(tmpdir, ["--episodes", "2", "--resume", ckpt,
                  "--run_name", "cli2"])
    run2 = os.path.join(tmpdir, "run_cli2")
    assert any(d.startswith("model_") for d in os.listdir(run2))
    recs2 = [json.loads(l) for l in
             open(os.path.join(tmpdir, "metrics_cli2.jsonl"))]
    train2 = [r for r in recs2 if "loss" in r]
    # resumed counters continue, not restart: episode 1, steps past ep-0's
    assert train2 and train2[0]["episode"] == 1
    assert train2[0]["total_batch_steps"] >= 4  # 3 rounds in ep 0 + 1
