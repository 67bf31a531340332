#!/usr/bin/env python3
"""Flagship benchmark: whole-node RL samples/sec, Qwen2.5-7B GRPO.

Measures BASELINE.json's headline metric — RL samples/sec (whole node) for
Qwen2.5-7B 4-bit LoRA GRPO on synthetic MATH-shaped prompts with
random-init weights — on N MI355X GPUs of one node.

One timed "step" is one full RL round (the reference's unit of progress,
distributed_trainer.py:248-346): candidate generation for the round's
prompt batch (num_candidates completions per prompt, paged-KV engine),
reward + advantage + top-k math, learner gradient accumulation over all
kept samples, learner gradient all-reduce, optimizer step and LoRA weight
broadcast. Weak scaling: 10 prompts per GPU per round (the reference's
per-GPU prompt density: batch 30 on 3 GPUs).

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W]
N>1 is launched by the driver via torch.distributed.run with one rank per
GPU (RANK/WORLD_SIZE read from the env).
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

REFERENCE_SAMPLES_PER_SEC = 6.7  # ~100 steps/2h x 480 samples


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=2)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", type=str, default="unsloth/Qwen2.5-7B-Instruct-bnb-4bit")
    p.add_argument("--prompts-per-gpu", type=int, default=10)
    p.add_argument("--num-candidates", type=int, default=16)
    p.add_argument("--max-new-tokens", type=int, default=1200)
    p.add_argument("--max-prompt-tokens", type=int, default=350)
    p.add_argument("--learner", type=str, default="grpo")
    p.add_argument("--actors", type=int, default=-1,
                   help="-1: all ranks are dual-role learners (0 actors)")
    p.add_argument("--cpu", action="store_true", help="CPU/gloo debug mode")
    p.add_argument("--eos-mean", type=float, default=0.0,
                   help="EOS-realistic mode: per-candidate exponential "
                        "output-length caps with this mean (e.g. 450 — the "
                        "reference eval's observed mean; 0 = fixed-length "
                        "no-EOS rounds). Measures in-wave retirement.")
    p.add_argument("--tiny", action="store_true",
                   help="tiny model + short generations (debug only; NOT the "
                        "headline config)")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")

    use_cuda = torch.cuda.is_available() and not args.cpu
    device = None
    if use_cuda:
        local = int(os.environ.get("LOCAL_RANK", rank))
        device = torch.device(f"cuda:{local}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    model_name = args.model
    max_new, max_prompt = args.max_new_tokens, args.max_prompt_tokens
    if args.tiny:
        model_name = "tiny-qwen2" if not use_cuda else "small-qwen2"
        max_new, max_prompt = 16, 64

    num_learners = world_size if args.actors < 0 else world_size - args.actors
    num_actors = world_size - num_learners
    batch_size = args.prompts_per_gpu * world_size

    # hardware-tuned micro-batch (288 GB HBM; the reference's 8 was a 24 GB
    # OOM bound): same objective math (tests/test_learner.py micro-batch
    # invariance), sized so resident activations fit — ~2x layers/hidden of
    # 32B needs the smaller step
    big = "32b" in model_name.lower()
    micro = 2 if big else 16

    config = {
        "run_name": "bench",
        "project_name": "bench",
        "lora_save_path": "bench_lora",
        "lr": 2e-5,
        "max_prompt_tokens": max_prompt,
        "max_new_tokens": max_new,
        "episodes": 1,
        "num_candidates": args.num_candidates,
        "batch_size": batch_size,
        "train_batch_size": micro,
        "temperature": 1.2,
        "save_every": 10**9,
        "eval_every": 0,
        "model": model_name,
        "dataset": "synthetic",
        "number_of_actors": num_actors,
        "number_of_learners": num_learners,
        "learner": args.learner,
        "use_vllm": True,
        "max_lora_rank": 32,
        "topk": args.num_candidates,
        "learner_chunk_size": max(1, batch_size // world_size),
        "actor_gpu_usage": 0.5 if big else 0.91,
        "learner_gpu_usage": 0.2 if big else 0.35,
        "lora_alpha": 16,
        "lora_dropout": 0.0,
        "seed": 3407,
        "use_8bit_adam": True,
    }

    from distrl_llm_amd.parallel.worker import build_worker
    from distrl_llm_amd.rl.data import (ListDataset, process_dataset,
                                        r1_preprompt, synthetic_math_dataset)
    from distrl_llm_amd.rl.rewards import reward_function as real_reward_function

    def reward_function(completions, solutions):
        """Real reward stack + a deterministic per-candidate jitter.

        Random-init weights produce uniformly zero rewards, which would make
        every GRPO advantage zero and trip the degenerate-group skip — the
        learner would do no work and the bench would be a lie. The jitter
        (hash of the completion text, <=0.1) restores the reward spread real
        training has, so the timed region contains the full learner
        workload: forward, fused loss, backward, all-reduce, Adam step."""
        import zlib
        r = real_reward_function(completions, solutions)
        jit = np.array([(zlib.crc32(c[-256:].encode()) % 1000) / 1000.0 * 0.1
                        for c in completions])
        r[:, 0] += jit
        return r

    n_prompts = batch_size * (args.steps + args.warmup)
    rows = process_dataset(None, synthetic_math_dataset(n_prompts, seed=17),
                           r1_preprompt)
    train = ListDataset(rows, seed=17)

    overrides = {}
    if not use_cuda:
        overrides = {"num_kv_blocks": 2048, "kv_block_size": 8,
                     "max_seq_length": 512 if args.tiny else max_prompt + max_new}

    trainer = build_worker(rank, world_size, config, train_dataset=train,
                           test_dataset=ListDataset(rows[:2]),
                           reward_function=reward_function, device=device,
                           engine_overrides=overrides)

    if rank != 0:
        trainer.follower_loop()
        trainer.fabric.close()
        return

    # ---- rank 0 drives the bench ----
    batches = list(train.iter(batch_size=batch_size))
    sp_dict = dict(trainer.sampling_params.__dict__)
    if args.eos_mean > 0:
        sp_dict["geom_len_mean"] = args.eos_mean

    for i in range(args.warmup):
        trainer.rl_round(batches[i], sp_dict)

    trainer._cmd("barrier", None)
    t0 = time.time()
    per_step = []
    total_samples = 0
    for i in range(args.steps):
        ts = time.time()
        stats = trainer.rl_round(batches[args.warmup + i], sp_dict)
        total_samples += stats["num_samples"]
        per_step.append(time.time() - ts)
        print(f"[bench] step {i}: gen={stats['timing/generation_duration']:.2f}s "
              f"reward={stats['timing/reward_duration']:.2f}s "
              f"update={stats['timing/update_duration']:.2f}s "
              f"total={per_step[-1]:.2f}s", file=sys.stderr, flush=True)
    trainer._cmd("barrier", None)
    elapsed = time.time() - t0

    samples_per_sec = total_samples / elapsed
    result = {
        "metric": "rl_samples_per_sec",
        "value": round(samples_per_sec, 3),
        "unit": "samples/s",
        "n_gpus": world_size,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1000, 1),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": (round(samples_per_sec / REFERENCE_SAMPLES_PER_SEC, 3)
                        if not args.tiny and use_cuda else None),
        "dtype": "bf16" if use_cuda else "fp32",
        "data": "synthetic",
        "config": {
            "model": model_name,
            "algorithm": config["learner"],
            "load_in_4bit": "4bit" in model_name.lower(),
            "global_batch": batch_size,
            "num_candidates": args.num_candidates,
            "samples_per_step": batch_size * args.num_candidates,
            "train_micro_batch": micro,
            "seq_len": max_prompt + max_new,
            "max_new_tokens": max_new,
            "parallelism": f"dp{world_size} ({num_actors} actors + "
                           f"{num_learners} learners)",
        },
    }
    print(json.dumps(result), flush=True)

    trainer._cmd_seq = getattr(trainer, "_cmd_seq", 0) + 1
    trainer.fabric.broadcast_obj((trainer._cmd_seq, "stop", None), src=0)
    trainer.fabric.close()


if __name__ == "__main__":
    main()
