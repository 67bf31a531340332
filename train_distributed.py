#!/usr/bin/env python3
"""Distributed RL fine-tuning CLI — flag-for-flag with the reference
(reference train_distributed.py:11-35), running on the
MI355X-native engine: one spawned process per GPU over RCCL/xGMI (gloo on
CPU) instead of Ray actors.

Offline behavior: if the HF dataset cannot be loaded (no network), a
deterministic synthetic MATH-shaped dataset is used (BASELINE.json:
synthetic prompts / random-init weights) — pass --synthetic_dataset N to
force it.
"""

import argparse
import os

import torch


def parse_args():
    args = argparse.ArgumentParser()
    args.add_argument("--model", type=str, default="unsloth/Qwen2.5-7B-Instruct-bnb-4bit")
    args.add_argument("--dataset", type=str, default="HuggingFaceH4/MATH-500")
    args.add_argument("--run_name", type=str)
    args.add_argument("--project_name", type=str, default="math-reasoning")
    args.add_argument("--lora_save_path", type=str, default="lora_request_math")
    args.add_argument("--lr", type=float, default=2e-5)
    args.add_argument("--max_new_tokens", type=int, default=1200)
    args.add_argument("--max_prompt_tokens", type=int, default=350)
    args.add_argument("--temperature", type=float, default=1.2)
    args.add_argument("--episodes", type=int, default=15)
    args.add_argument("--num_candidates", type=int, default=16,
                      help="Number of sampled candidates per prompt")
    args.add_argument("--batch_size", type=int, default=30,
                      help="Total batch size split across all actors and learners")
    args.add_argument("--learner_chunk_size", type=int, default=8,
                      help="Per-learner generation sub-batch size")
    args.add_argument("--train_batch_size", type=int, default=8,
                      help="Learner micro-batch size for gradient accumulation")
    args.add_argument("--save_every", type=int, default=100)
    args.add_argument("--eval_every", type=int, default=10)
    args.add_argument("--number_of_actors", type=int, default=2)
    args.add_argument("--number_of_learners", type=int, default=1)
    args.add_argument("--learner", type=str, choices=["pg", "grpo"], default="pg")
    args.add_argument("--max_lora_rank", type=int, default=32)
    args.add_argument("--lora_alpha", type=int, default=16)
    args.add_argument("--lora_dropout", type=float, default=0)
    args.add_argument("--topk", type=int, default=16,
                      help="Top-k candidates per prompt kept for training")
    args.add_argument("--actor_gpu_usage", type=float, default=0.91)
    args.add_argument("--learner_gpu_usage", type=float, default=0.35)
    # native-framework extensions (not in the reference CLI)
    args.add_argument("--synthetic_dataset", type=int, default=0,
                      help="Use N synthetic MATH-shaped prompts instead of --dataset")
    args.add_argument("--load_adapter", type=str, default=None,
                      help="Warm-start: PEFT adapter directory to load into "
                           "every worker before training")
    args.add_argument("--resume", type=str, default=None,
                      help="Resume from a checkpoint directory written by the "
                           "save cadence (run_<name>/model_<step>): restores "
                           "adapter, optimizer, trainer counters, dataset "
                           "order and per-rank RNG streams")
    args.add_argument("--eval_only", action="store_true",
                      help="Evaluate (pass@1 / BoN@8) on the test split and "
                           "exit — combine with --load_adapter to score a "
                           "trained adapter")
    args.add_argument("--seed", type=int, default=3407)
    args.add_argument("--backend_device", type=str, default="auto",
                      choices=["auto", "cuda", "cpu"])
    return args.parse_args()


def build_config(args) -> dict:
    return {
        "run_name": args.run_name,
        "project_name": args.project_name,
        "lora_save_path": args.lora_save_path,
        "lr": args.lr,
        "max_prompt_tokens": args.max_prompt_tokens,
        "max_new_tokens": args.max_new_tokens,
        "episodes": args.episodes,
        "num_candidates": args.num_candidates,
        "batch_size": args.batch_size,
        "train_batch_size": args.train_batch_size,
        "temperature": args.temperature,
        "save_every": args.save_every,
        "eval_every": args.eval_every,
        "model": args.model,
        "dataset": args.dataset,
        "number_of_actors": args.number_of_actors,
        "number_of_learners": args.number_of_learners,
        "learner": args.learner,
        "use_vllm": True,  # kept for config-dict parity (always native engine)
        "max_lora_rank": args.max_lora_rank,
        "topk": args.topk,
        "learner_chunk_size": args.learner_chunk_size,
        "actor_gpu_usage": args.actor_gpu_usage,
        "learner_gpu_usage": args.learner_gpu_usage,
        "lora_alpha": args.lora_alpha,
        "lora_dropout": args.lora_dropout,
        "seed": args.seed,
        "load_adapter": args.load_adapter,
        "resume": args.resume,
        "eval_only": args.eval_only,
    }


def load_datasets(args, tokenizer):
    from distrl_llm_amd.rl.data import (ListDataset, load_local_rows,
                                        process_dataset, r1_preprompt,
                                        synthetic_math_dataset)
    rows = None
    if args.synthetic_dataset <= 0:
        if os.path.exists(args.dataset):
            # local file/dir (air-gapped real data): json/jsonl/parquet
            # with problem+solution (or MATH-style problem+answer) columns
            rows = load_local_rows(args.dataset)
        else:
            try:
                from datasets import load_dataset
                raw = load_dataset(args.dataset)["test"]
                raw = raw.map(lambda x: {"solution": x["answer"], "answer": x["answer"]})
                raw = raw.remove_columns(["answer"])
                rows = [dict(r) for r in raw]
            except Exception as e:
                print(f"Dataset load failed ({e}); falling back to synthetic data")
    if rows is None:
        n = args.synthetic_dataset if args.synthetic_dataset > 0 else 500
        rows = synthetic_math_dataset(n, seed=args.seed)
    rows = process_dataset(tokenizer, rows, r1_preprompt, postprompt="")
    # Seeded shuffle before the 90/10 split (reference train_distributed.py:44
    # uses train_test_split which shuffles; an unshuffled tail split would make
    # eval non-random for datasets ordered by subject/difficulty).
    import random as _random
    _random.Random(args.seed).shuffle(rows)
    n_test = max(1, len(rows) // 10)
    return ListDataset(rows[:-n_test], seed=args.seed), ListDataset(rows[-n_test:], seed=args.seed)


def _worker_main(rank, world_size, config, train_rows, test_rows, master_port):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(master_port)
    from distrl_llm_amd.parallel.worker import build_worker
    from distrl_llm_amd.rl.data import ListDataset
    from distrl_llm_amd.rl.rewards import reward_function
    trainer = build_worker(rank, world_size, config,
                           train_dataset=ListDataset(train_rows, seed=config["seed"]),
                           test_dataset=ListDataset(test_rows, seed=config["seed"]),
                           reward_function=reward_function)
    trainer.train()


def main():
    args = parse_args()
    config = build_config(args)
    world_size = args.number_of_actors + args.number_of_learners

    from distrl_llm_amd.models.hf_io import resolve_spec
    from distrl_llm_amd.utils.tokenizer import load_tokenizer
    tokenizer = load_tokenizer(args.model, resolve_spec(args.model).vocab_size)
    train_ds, test_ds = load_datasets(args, tokenizer)
    print(f"\nNumber of train samples: {len(train_ds)}\n")
    print(f"Number of test samples: {len(test_ds)}\n")

    if args.backend_device == "cuda" or (
            args.backend_device == "auto" and torch.cuda.is_available()):
        if torch.cuda.device_count() < world_size:
            raise RuntimeError(
                f"Not enough GPUs available. Available: "
                f"{torch.cuda.device_count()}, Required: {world_size}")

    import torch.multiprocessing as mp
    port = 29500 + (os.getpid() % 1000)
    mp.spawn(_worker_main, nprocs=world_size,
             args=(world_size, config, train_ds.rows, test_ds.rows, port),
             join=True)


if __name__ == "__main__":
    main()
