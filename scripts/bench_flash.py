#!/usr/bin/env python3
"""First-party flash attention vs torch SDPA (aotriton) at learner shapes."""
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distrl_llm_amd.ops.build import build  # noqa: E402
from distrl_llm_amd.ops import functional as OF  # noqa: E402

build()
dev = torch.device("cuda:0")


def bench(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e3  # ms


for B, Hq, Hkv, T, D in [(16, 28, 4, 1550, 128), (8, 28, 4, 1550, 128),
                         (16, 32, 8, 1024, 128)]:
    scale = D ** -0.5
    q = (torch.randn(B, Hq, T, D, device=dev) * 0.3).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, T, D, device=dev) * 0.3).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, T, D, device=dev) * 0.3).to(torch.bfloat16)
    dout = torch.randn_like(q)

    def ours_fwd():
        return OF.flash_attention(q, k, v, scale)

    def ours_fwdbwd():
        qg, kg, vg = (t.clone().requires_grad_(True) for t in (q, k, v))
        OF.flash_attention(qg, kg, vg, scale).backward(dout)

    group = Hq // Hkv
    ke = k.repeat_interleave(group, 1)
    ve = v.repeat_interleave(group, 1)

    def sdpa_fwd():
        return F.scaled_dot_product_attention(q, ke, ve, is_causal=True,
                                              scale=scale)

    def sdpa_fwdbwd():
        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)
        kge = kg.repeat_interleave(group, 1)
        vge = vg.repeat_interleave(group, 1)
        F.scaled_dot_product_attention(qg, kge, vge, is_causal=True,
                                       scale=scale).backward(dout)

    t1 = bench(ours_fwd)
    t2 = bench(sdpa_fwd)
    t3 = bench(ours_fwdbwd)
    t4 = bench(sdpa_fwdbwd)
    # attention FLOPs: 2*2*B*Hq*T^2/2*D (QK^T + PV, causal half)
    fl = 2 * B * Hq * T * T * D  # fwd FLOPs (causal: ~half of 2x this)
    print(f"B{B} Hq{Hq} Hkv{Hkv} T{T}: fwd ours {t1:6.2f}ms "
          f"({fl/t1/1e9:5.0f} GF/s eff) vs sdpa {t2:6.2f}ms | "
          f"fwd+bwd ours {t3:6.2f}ms vs sdpa(+rep) {t4:6.2f}ms", flush=True)
