#!/usr/bin/env python3
"""BASELINE config 5 readiness: Qwen2.5-32B 4-bit on one MI355X — init,
quantize, fused-nf4 decode session, learner micro-step, memory report."""

import argparse
import sys
import time

import torch

sys.path.insert(0, ".")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="unsloth/Qwen2.5-32B-Instruct-bnb-4bit")
    ap.add_argument("--pool", type=float, default=0.3)
    args = ap.parse_args()
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.train.learner import Learner
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer

    dev = torch.device("cuda:0")
    spec = get_spec(args.model)
    t0 = time.time()
    model = CausalLM(spec, lora_r=32, lora_alpha=16, dtype=torch.bfloat16,
                     device=dev).random_init(3407)
    print(f"init {time.time()-t0:.1f}s, "
          f"alloc {torch.cuda.memory_allocated()/2**30:.1f} GiB", flush=True)
    t0 = time.time()
    model.quantize_nf4_()
    print(f"quantize {time.time()-t0:.1f}s, "
          f"alloc {torch.cuda.memory_allocated()/2**30:.1f} GiB", flush=True)

    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    t0 = time.time()
    engine = Engine(model, EngineConfig(max_seq_length=1550,
                                        gpu_memory_utilization=args.pool),
                    device=dev, seed=0)
    print(f"engine init {time.time()-t0:.1f}s, nf4_path={engine.fused.nf4}, "
          f"kv blocks {engine.pool.num_blocks}", flush=True)

    prompts = [tok.encode("Solve 12*11. " * 20)] * 8
    sp = SamplingParams(max_tokens=64, temperature=1.2, n=4, top_p=0.95)
    t0 = time.time()
    outs = engine.generate(prompts, sp, eos_token_id=tok.eos_token_id)
    torch.cuda.synchronize()
    dt = time.time() - t0
    n_seq = sum(len(o) for o in outs)
    n_tok = sum(len(ids) for o in outs for ids in o)
    print(f"generate: {n_seq} seqs, {n_tok} tokens in {dt:.1f}s "
          f"({n_tok/dt:.0f} tok/s), alloc "
          f"{torch.cuda.memory_allocated()/2**30:.1f} GiB", flush=True)

    learner = Learner(model, tok, lr=2e-5, max_prompt_tokens=350,
                      max_new_tokens=1200, train_batch_size=2)
    probs = ["p" * 1000] * 4
    answers = ["a" * 4000] * 4
    t0 = time.time()
    loss = learner.accumulate_gradients(probs, answers, [0.5, -0.5, 0.3, -0.3])
    learner.step()
    torch.cuda.synchronize()
    print(f"learner 4 samples (2 micro of (2,1550)): {time.time()-t0:.1f}s, "
          f"loss={loss:.4f}, peak "
          f"{torch.cuda.max_memory_allocated()/2**30:.1f} GiB", flush=True)


if __name__ == "__main__":
    main()
