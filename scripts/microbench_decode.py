#!/usr/bin/env python3
"""Decode-step microbench: per-step latency of the hipGraph decode session
on Qwen2.5-7B at a given batch size / context. Run under rocprofv3 for the
kernel breakdown."""

import argparse
import sys
import time

import torch

sys.path.insert(0, ".")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=160)
    p.add_argument("--prompt-len", type=int, default=410)
    p.add_argument("--steps", type=int, default=64)
    p.add_argument("--model", type=str, default="qwen2.5-7b")
    p.add_argument("--no-graph", action="store_true")
    p.add_argument("--no-4bit", action="store_true")
    args = p.parse_args()

    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.engine.decode_session import DecodeSession
    from distrl_llm_amd.models import CausalLM, get_spec

    dev = torch.device("cuda:0")
    spec = get_spec(args.model)
    t0 = time.time()
    model = CausalLM(spec, lora_r=32, lora_alpha=16, dtype=torch.bfloat16,
                     device=dev).random_init(3407)
    if not args.no_4bit:
        model.quantize_nf4_()
    print(f"model init {time.time()-t0:.1f}s", flush=True)

    engine = Engine(model, EngineConfig(max_seq_length=1550,
                                        gpu_memory_utilization=0.35),
                    device=dev, seed=0)
    if engine.fused is not None:
        engine.fused.refresh()
    sp = SamplingParams(max_tokens=1200, temperature=1.2, n=1, top_p=0.95)

    # fabricate prompts and prefill
    torch.manual_seed(0)
    prompts = [torch.randint(0, spec.vocab_size, (args.prompt_len,)).tolist()
               for _ in range(args.batch)]
    from distrl_llm_amd.engine.kvcache import Sequence
    seqs = []
    t0 = time.time()
    for i in range(0, args.batch, 16):
        batch = []
        for j, pr in enumerate(prompts[i:i + 16]):
            engine._seq_counter += 1
            batch.append(Sequence(engine._seq_counter, pr, i + j))
        logits = engine._prefill_batch(batch)
        for q, lg in zip(batch, logits):
            q.output_ids = [int(lg.argmax())]
            seqs.append(q)
    torch.cuda.synchronize()
    print(f"prefill {time.time()-t0:.1f}s for {args.batch} seqs", flush=True)

    session = DecodeSession(engine, seqs, sp, eos_token_id=None,
                            use_graph=not args.no_graph)
    if session.use_graph:
        t0 = time.time()
        session._capture()
        torch.cuda.synchronize()
        print(f"graph capture {time.time()-t0:.2f}s", flush=True)

    def run_steps(n):
        for _ in range(n):
            if session.graph is not None:
                session.graph.replay()
            else:
                session._step()

    run_steps(8)  # warmup
    torch.cuda.synchronize()
    t0 = time.time()
    run_steps(args.steps)
    torch.cuda.synchronize()
    dt = time.time() - t0
    print(f"decode: {dt/args.steps*1000:.2f} ms/step at batch={args.batch} "
          f"({args.batch*args.steps/dt:.0f} tok/s)", flush=True)


if __name__ == "__main__":
    main()
