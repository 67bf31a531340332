#!/usr/bin/env python3
"""Learner-step microbench: time + kernel profile of the teacher-forced
forward/backward on Qwen2.5-7B (bench shapes: micro-batch of (B,1550))."""

import argparse
import sys
import time

import torch

sys.path.insert(0, ".")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--samples", type=int, default=32)
    p.add_argument("--micro", type=int, default=8)
    p.add_argument("--model", type=str, default="qwen2.5-7b")
    args = p.parse_args()

    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.train.learner import Learner
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer

    dev = torch.device("cuda:0")
    spec = get_spec(args.model)
    model = CausalLM(spec, lora_r=32, lora_alpha=16, dtype=torch.bfloat16,
                     device=dev).random_init(3407)
    model.quantize_nf4_()
    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    learner = Learner(model, tok, lr=2e-5, max_prompt_tokens=350,
                      max_new_tokens=1200, train_batch_size=args.micro)

    prompt = "x" * 1400  # ~350 tokens after truncation
    answer = "y" * 4800  # 1200 tokens after truncation
    problems = [prompt] * args.samples
    answers = [answer] * args.samples
    rewards = [0.1 * (i % 7 - 3) + 0.01 for i in range(args.samples)]

    # warmup
    learner.accumulate_gradients(problems[:args.micro], answers[:args.micro],
                                 rewards[:args.micro])
    learner.step()
    torch.cuda.synchronize()

    t0 = time.time()
    loss = learner.accumulate_gradients(problems, answers, rewards)
    learner.step()
    torch.cuda.synchronize()
    dt = time.time() - t0
    print(f"learner: {args.samples} samples in {dt:.2f}s "
          f"({args.samples/dt:.1f} samples/s, micro={args.micro}), "
          f"loss={loss:.4f}", flush=True)


if __name__ == "__main__":
    main()
