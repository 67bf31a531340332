"""End-to-end control-plane test: 1 actor + 1 learner over gloo on CPU —
BASELINE.json config 1 (plumbing, no GPU). Exercises chunking, generate
fan-out, rewards, advantage, top-k, learner update, weight broadcast,
adapter save, eval and metrics."""

import json
import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world_size, tmpdir, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(tmpdir)

    from distrl_llm_amd.parallel.worker import build_worker
    from distrl_llm_amd.rl.data import (ListDataset, process_dataset,
                                        r1_preprompt, synthetic_math_dataset)
    from distrl_llm_amd.rl.rewards import reward_function

    config = {
        "run_name": "gloo_test",
        "project_name": "test",
        "lora_save_path": os.path.join(tmpdir, "lora_adapter"),
        "lr": 1e-3,
        "max_prompt_tokens": 48,
        "max_new_tokens": 16,
        "episodes": 1,
        "num_candidates": 2,
        "batch_size": 4,
        "train_batch_size": 2,
        "temperature": 1.0,
        "save_every": 100,
        "eval_every": 1,
        "model": "tiny-qwen2",
        "dataset": "synthetic",
        "number_of_actors": 1,
        "number_of_learners": 1,
        "learner": "grpo",
        "use_vllm": True,
        "max_lora_rank": 4,
        "topk": 2,
        "learner_chunk_size": 2,
        "actor_gpu_usage": 0.91,
        "learner_gpu_usage": 0.35,
        "lora_alpha": 8,
        "lora_dropout": 0.0,
        "seed": 11,
        "use_8bit_adam": True,
    }
    rows = synthetic_math_dataset(8, seed=1)
    rows = process_dataset(None, rows, r1_preprompt)
    train = ListDataset(rows[:6], seed=1)
    test = ListDataset(rows[6:], seed=1)
    trainer = build_worker(
        rank, world_size, config, train_dataset=train, test_dataset=test,
        reward_function=reward_function, device=torch.device("cpu"),
        engine_overrides={"num_kv_blocks": 512, "kv_block_size": 8,
                          "max_seq_length": 512})
    trainer.train()
    trainer.fabric.close()


@pytest.mark.timeout(600)
def test_two_rank_gloo_end_to_end(tmp_path):
    tmpdir = str(tmp_path)
    port = 29000 + os.getpid() % 500
    mp.spawn(_worker, nprocs=2, args=(2, tmpdir, port), join=True)

    # adapter (PEFT format) written by rank 0 every round
    adir = os.path.join(tmpdir, "lora_adapter")
    assert os.path.exists(os.path.join(adir, "adapter_config.json"))
    assert os.path.exists(os.path.join(adir, "adapter_model.safetensors"))
    with open(os.path.join(adir, "adapter_config.json")) as f:
        cfg = json.load(f)
    assert cfg["r"] == 4 and cfg["peft_type"] == "LORA"

    # final checkpoint directory
    assert any(d.startswith("model_") for d in
               os.listdir(os.path.join(tmpdir, "run_gloo_test")))

    # metrics with the reference key set
    metrics_file = os.path.join(tmpdir, "metrics_gloo_test.jsonl")
    assert os.path.exists(metrics_file)
    records = [json.loads(l) for l in open(metrics_file)]
    train_recs = [r for r in records if "loss" in r]
    eval_recs = [r for r in records if "eval/pass@1(mean8)" in r]
    assert len(train_recs) >= 1
    assert len(eval_recs) >= 2  # initial + per-step (eval_every=1)
    for key in ("loss", "mean_format_reward", "mean_accuracy_reward",
                "min_accuracy_reward", "max_accuracy_reward",
                "mean_token_length", "episode", "total_batch_steps",
                "total_samples_processed", "timing/update_duration",
                "timing/reward_duration", "timing/generation_duration"):
        assert key in train_recs[0], key
    for key in ("eval/BoN(8)", "eval/mean_token_length",
                "timing/eval_duration"):
        assert key in eval_recs[0], key
