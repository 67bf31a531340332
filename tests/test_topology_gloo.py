"""Mixed-topology control-plane test: 2 actors + 2 learners over gloo on
CPU — the fullest per-node shape (actor chunking + multi-learner
all-reduce + broadcast in one world), the same structure the driver's
8-GPU run exercises. Verifies all four replicas converge to one adapter
and the round metrics account for every prompt exactly once."""

import json
import os
import zlib

import pytest
import torch
import torch.multiprocessing as mp


def _reward(completions, solutions):
    import numpy as np

    from distrl_llm_amd.rl.rewards import reward_function as base
    r = base(completions, solutions)
    r[:, 0] += np.array([(zlib.crc32(c[-64:].encode()) % 100) / 100.0
                         for c in completions])
    return r


def _worker(rank, world_size, tmpdir, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(tmpdir)

    from distrl_llm_amd.models.lora import lora_state_dict
    from distrl_llm_amd.parallel.worker import build_worker
    from distrl_llm_amd.rl.data import (ListDataset, process_dataset,
                                        r1_preprompt, synthetic_math_dataset)

    config = {
        "run_name": "topo_test", "project_name": "test",
        "lora_save_path": os.path.join(tmpdir, "topo_adapter"),
        "lr": 1e-3, "max_prompt_tokens": 48, "max_new_tokens": 10,
        "episodes": 1, "num_candidates": 2, "batch_size": 7,
        "train_batch_size": 2, "temperature": 1.0, "save_every": 100,
        "eval_every": 0, "model": "tiny-qwen2", "dataset": "synthetic",
        "number_of_actors": 2, "number_of_learners": 2, "learner": "pg",
        "max_lora_rank": 4, "topk": 2, "learner_chunk_size": 1,
        "actor_gpu_usage": 0.91, "learner_gpu_usage": 0.35,
        "lora_alpha": 8, "lora_dropout": 0.0, "seed": 13,
        "use_8bit_adam": True,
    }
    rows = process_dataset(None, synthetic_math_dataset(7, seed=4),
                           r1_preprompt)
    trainer = build_worker(
        rank, world_size, config, train_dataset=ListDataset(rows, seed=4),
        test_dataset=ListDataset(rows[:2], seed=4),
        reward_function=_reward, device=torch.device("cpu"),
        engine_overrides={"num_kv_blocks": 512, "kv_block_size": 8,
                          "max_seq_length": 512})
    trainer.train()
    state = lora_state_dict(trainer.engine.model)
    torch.save({k: v.clone() for k, v in state.items()},
               os.path.join(tmpdir, f"topo_state_rank{rank}.pt"))
    trainer.fabric.close()


@pytest.mark.timeout(600)
def test_two_actor_two_learner_topology(tmp_path):
    tmpdir = str(tmp_path)
    port = 27500 + os.getpid() % 500
    mp.spawn(_worker, nprocs=4, args=(4, tmpdir, port), join=True)

    states = [torch.load(os.path.join(tmpdir, f"topo_state_rank{r}.pt"))
              for r in range(4)]
    moved = False
    for key in states[0]:
        for r in range(1, 4):
            torch.testing.assert_close(states[0][key], states[r][key],
                                       rtol=0, atol=0)
        if "lora_B" in key and states[0][key].abs().max() > 0:
            moved = True
    assert moved

    # every prompt of the odd-sized batch (7, chunks 2a+2l with
    # learner_chunk_size 1) was processed exactly once per round
    records = [json.loads(l) for l in
               open(os.path.join(tmpdir, "metrics_topo_test.jsonl"))]
    train_recs = [r for r in records if "loss" in r]
    assert train_recs and train_recs[-1]["total_samples_processed"] == 7
