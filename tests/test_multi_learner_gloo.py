"""Multi-learner correctness: 1 actor + 2 learners over gloo. The gradient
all-reduce + replicated optimizer must keep every learner's adapter
bit-identical (replacing the reference's CPU gradient staging through
learner 0), and the round weight broadcast must land the
same adapter on the actor."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world_size, tmpdir, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(tmpdir)

    from distrl_llm_amd.models.lora import lora_state_dict
    from distrl_llm_amd.parallel.worker import build_worker
    from distrl_llm_amd.rl.data import (ListDataset, process_dataset,
                                        r1_preprompt, synthetic_math_dataset)
    from distrl_llm_amd.rl.rewards import reward_function as base_reward

    def reward_function(completions, solutions):
        # deterministic spread so advantages are nonzero and learners train
        import numpy as np
        import zlib
        r = base_reward(completions, solutions)
        r[:, 0] += np.array([(zlib.crc32(c[-64:].encode()) % 100) / 100.0
                             for c in completions])
        return r

    config = {
        "run_name": f"ml_test", "project_name": "test",
        "lora_save_path": os.path.join(tmpdir, "ml_adapter"),
        "lr": 1e-3, "max_prompt_tokens": 48, "max_new_tokens": 12,
        "episodes": 1, "num_candidates": 2, "batch_size": 6,
        "train_batch_size": 2, "temperature": 1.0, "save_every": 100,
        "eval_every": 0, "model": "tiny-qwen2", "dataset": "synthetic",
        "number_of_actors": 1, "number_of_learners": 2, "learner": "grpo",
        "use_vllm": True, "max_lora_rank": 4, "topk": 2,
        "learner_chunk_size": 2, "actor_gpu_usage": 0.91,
        "learner_gpu_usage": 0.35, "lora_alpha": 8, "lora_dropout": 0.0,
        "seed": 5, "use_8bit_adam": True,
    }
    rows = process_dataset(None, synthetic_math_dataset(6, seed=2), r1_preprompt)
    trainer = build_worker(
        rank, world_size, config, train_dataset=ListDataset(rows, seed=2),
        test_dataset=ListDataset(rows[:2], seed=2),
        reward_function=reward_function, device=torch.device("cpu"),
        engine_overrides={"num_kv_blocks": 512, "kv_block_size": 8,
                          "max_seq_length": 512})
    trainer.train()
    # every rank dumps its adapter state post-training
    state = lora_state_dict(trainer.engine.model)
    torch.save({k: v.clone() for k, v in state.items()},
               os.path.join(tmpdir, f"state_rank{rank}.pt"))
    trainer.fabric.close()


@pytest.mark.timeout(600)
def test_multi_learner_sync(tmp_path):
    tmpdir = str(tmp_path)
    port = 28500 + os.getpid() % 500
    mp.spawn(_worker, nprocs=3, args=(3, tmpdir, port), join=True)

    states = [torch.load(os.path.join(tmpdir, f"state_rank{r}.pt"))
              for r in range(3)]
    # learners (ranks 1, 2) must agree exactly; actor (rank 0) received the
    # broadcast of the same adapter
    moved = False
    for key in states[1]:
        torch.testing.assert_close(states[1][key], states[2][key],
                                   rtol=0, atol=0)
        torch.testing.assert_close(states[0][key], states[1][key],
                                   rtol=0, atol=0)
        if "lora_B" in key and states[1][key].abs().max() > 0:
            moved = True
    assert moved, "training should have updated at least one adapter tensor"
