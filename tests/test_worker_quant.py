"""Worker-level quantization selection: the reference picks 4-bit via
the '-bnb-4bit' model-name suffix (reference distributed_actor.py:17);
the config's explicit load_in_4bit overrides in either direction
(bench.py discloses it)."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, port, load_in_4bit, out_file):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from distrl_llm_amd.parallel.worker import build_worker
    config = {
        "run_name": "q", "lora_save_path": out_file + ".a", "lr": 1e-3,
        "max_prompt_tokens": 16, "max_new_tokens": 8, "episodes": 1,
        "num_candidates": 2, "batch_size": 2, "train_batch_size": 2,
        "temperature": 1.0, "save_every": 0, "eval_every": 0,
        "model": "tiny-qwen2", "dataset": "synthetic",
        "number_of_actors": 0, "number_of_learners": 1, "learner": "grpo",
        "max_lora_rank": 4, "topk": 2, "learner_chunk_size": 2,
        "actor_gpu_usage": 0.9, "learner_gpu_usage": 0.35,
        "lora_alpha": 8, "lora_dropout": 0.0, "seed": 1,
        "load_in_4bit": load_in_4bit,
    }
    trainer = build_worker(rank, 1, config, device=torch.device("cpu"),
                           engine_overrides={"num_kv_blocks": 64,
                                             "kv_block_size": 8,
                                             "max_seq_length": 64})
    mod = trainer.engine.model.model.layers[0].self_attn.q_proj
    quantized = getattr(mod, "weight_nf4", None) is not None
    trainer.fabric.close()
    with open(out_file, "w") as f:
        f.write(str(quantized))


@pytest.mark.timeout(300)
@pytest.mark.parametrize("load_in_4bit,expect", [(True, "True"),
                                                 (False, "False")])
def test_load_in_4bit_override(tmp_path, load_in_4bit, expect):
    out = str(tmp_path / "q.txt")
    port = 21500 + os.getpid() % 500 + (7 if load_in_4bit else 0)
    mp.spawn(_worker, nprocs=1, args=(port, load_in_4bit, out), join=True)
    assert open(out).read() == expect
