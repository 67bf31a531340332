"""Learner-count invariance: training the same flattened sample set on
1 learner (gradient accumulation) and on 2 learners (even split +
all-reduce mean) must produce the SAME adapter — the gradient estimator
is the mean over samples either way. This pins the distributed-update
math the 8-GPU run relies on (SURVEY §2.2 DP-over-learners)."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

PROBLEMS = ["What is 2+2?", "Compute 3*3.", "Find 10-4.", "Evaluate 5+5."]
ANSWERS = ["<think>a</think><answer>4</answer>",
           "<think>bb</think><answer>9</answer>",
           "<think>c</think><answer>6</answer>",
           "<think>dd</think><answer>10</answer>"]
REWARDS = [1.0, -0.5, 0.3, -0.8]


def _worker(rank, world_size, tmpdir, port, tag):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)

    from distrl_llm_amd.models.lora import lora_state_dict
    from distrl_llm_amd.parallel.worker import build_worker

    num_learners = world_size
    config = {
        "run_name": tag, "lora_save_path": os.path.join(tmpdir, tag),
        "lr": 1e-3, "max_prompt_tokens": 32, "max_new_tokens": 32,
        "episodes": 1, "num_candidates": 2, "batch_size": 4,
        # 1-learner world: 2 micro-batches of 2; 2-learner world: one
        # micro-batch of 2 per learner
        "train_batch_size": 2, "temperature": 1.0, "save_every": 0,
        "eval_every": 0, "model": "tiny-qwen2", "dataset": "synthetic",
        "number_of_actors": 0, "number_of_learners": num_learners,
        "learner": "grpo", "max_lora_rank": 4, "topk": 2,
        "learner_chunk_size": 2, "actor_gpu_usage": 0.9,
        "learner_gpu_usage": 0.35, "lora_alpha": 8, "lora_dropout": 0.0,
        "seed": 3,
    }
    trainer = build_worker(rank, world_size, config,
                           device=torch.device("cpu"),
                           engine_overrides={"num_kv_blocks": 128,
                                             "kv_block_size": 8,
                                             "max_seq_length": 256})
    # identical flattened samples, split evenly across learners (the
    # shape _update_handler receives from rank 0)
    per = len(PROBLEMS) // num_learners
    chunks = []
    for li in range(num_learners):
        s = slice(li * per, (li + 1) * per)
        chunks.append((PROBLEMS[s], ANSWERS[s],
                       np.asarray(REWARDS[s], dtype=np.float64)))
    for _ in range(3):  # multiple steps: optimizer state must track too
        trainer._handle("update", chunks)
    state = lora_state_dict(trainer.engine.model)
    torch.save({k: v.clone() for k, v in state.items()},
               os.path.join(tmpdir, f"{tag}_rank{rank}.pt"))
    trainer.fabric.close()


@pytest.mark.timeout(600)
def test_one_vs_two_learners_same_update(tmp_path):
    tmpdir = str(tmp_path)
    port = 24500 + os.getpid() % 500
    mp.spawn(_worker, nprocs=1, args=(1, tmpdir, port, "w1"), join=True)
    mp.spawn(_worker, nprocs=2, args=(2, tmpdir, port + 7, "w2"), join=True)

    s1 = torch.load(os.path.join(tmpdir, "w1_rank0.pt"))
    s2 = torch.load(os.path.join(tmpdir, "w2_rank0.pt"))
    moved = False
    for k in s1:
        torch.testing.assert_close(s1[k], s2[k], rtol=0, atol=0)
        if "lora_B" in k and s1[k].abs().max() > 0:
            moved = True
    assert moved, "updates must be non-trivial for the comparison to count"
