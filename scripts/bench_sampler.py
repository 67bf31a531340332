#!/usr/bin/env python3
"""Single-kernel vs two-stage sampler at decode shapes."""
import sys, time, torch
sys.path.insert(0, ".")
from distrl_llm_amd.ops.build import build
ext = build()
dev = torch.device("cuda:0")
for B in (160, 64, 512):
    V = 152064
    logits = torch.randn(B, V, device=dev, dtype=torch.bfloat16)
    seeds = torch.randint(0, 2**31 - 1, (B,), device=dev, dtype=torch.int64)
    step = torch.zeros(1, dtype=torch.int64, device=dev)
    for name in ("sample_tokens", "sample_tokens2"):
        fn = getattr(ext, name)
        run = lambda: fn(logits, 1.2, 0.95, 0, seeds, step)
        for _ in range(10): run()
        torch.cuda.synchronize(); t0 = time.time()
        for _ in range(100): run()
        torch.cuda.synchronize()
        print(f"B={B} {name}: {(time.time()-t0)/100*1e6:.1f}us", flush=True)
