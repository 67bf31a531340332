#!/usr/bin/env python3
"""Minimal flash-attention debug cases (V-path, softmax-path isolation)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distrl_llm_amd.ops.build import build  # noqa: E402

ext = build()
dev = torch.device("cuda:0")


def run(q, k, v, scale=1.0):
    o, lse = ext.flash_attn_fwd(q, k, v, scale)
    return o, lse


def ref(q, k, v, scale=1.0):
    s = torch.einsum("bhqd,bhkd->bhqk", q.float(), k.float()) * scale
    T = q.shape[2]
    mask = torch.ones(T, T, dtype=torch.bool, device=dev).tril()
    s = s.masked_fill(~mask, float("-inf"))
    return torch.einsum("bhqk,bhkd->bhqd", s.softmax(-1), v.float())


B, H, T, D = 1, 1, 32, 64

# case 1: K = 0 -> uniform P -> O[q] = mean(V[:q+1]); V = ramp over kv
q = torch.zeros(B, H, T, D, device=dev, dtype=torch.bfloat16)
k = torch.zeros(B, H, T, D, device=dev, dtype=torch.bfloat16)
v = torch.arange(T, device=dev, dtype=torch.float32).view(1, 1, T, 1)
v = (v.expand(B, H, T, D) / 8.0).to(torch.bfloat16).contiguous()
o, lse = run(q, k, v)
r = ref(q, k, v)
err = (o.float() - r).abs()
print("case1 (V ramp over kv): maxerr", err.max().item())
if err.max() > 1e-2:
    bad = (err > 1e-2).nonzero()[:8]
    print(" bad idx:", bad.tolist())
    print(" mine:", o.float()[0, 0, :4, :6])
    print(" ref :", r[0, 0, :4, :6])

# case 2: V = ramp over d (distinguishes d-columns)
v2 = torch.arange(D, device=dev, dtype=torch.float32).view(1, 1, 1, D)
v2 = (v2.expand(B, H, T, D) / 16.0).to(torch.bfloat16).contiguous()
o2, _ = run(q, k, v2)
r2 = ref(q, k, v2)
err2 = (o2.float() - r2).abs()
print("case2 (V ramp over d): maxerr", err2.max().item())
if err2.max() > 1e-2:
    print(" mine row0:", o2.float()[0, 0, 0, :16])
    print(" ref  row0:", r2[0, 0, 0, :16])

# case 3: random Q/K, V one-hot over kv (tests P values directly)
torch.manual_seed(1)
q3 = (torch.randn(B, H, T, D, device=dev) * 0.3).to(torch.bfloat16)
k3 = (torch.randn(B, H, T, D, device=dev) * 0.3).to(torch.bfloat16)
v3 = torch.zeros(T, D, device=dev)
v3[:, :T] = torch.eye(T, device=dev)
v3 = v3.view(1, 1, T, D).to(torch.bfloat16).expand(B, H, T, D).contiguous()
o3, lse3 = run(q3, k3, v3, scale=D ** -0.5)
r3 = ref(q3, k3, v3, scale=D ** -0.5)
err3 = (o3.float() - r3).abs()
print("case3 (P probe): maxerr", err3.max().item())
if err3.max() > 2e-2:
    bad = (err3 > 2e-2).nonzero()
    print(" n bad:", len(bad), "first:", bad[:6].tolist())
    qq = int(bad[0][2])
    print(" q row", qq, "mine:", o3.float()[0, 0, qq, :12])
    print("          ref :", r3[0, 0, qq, :12])

# case 4: lse check
sref = torch.einsum("bhqd,bhkd->bhqk", q3.float(), k3.float()) * (D ** -0.5)
mask = torch.ones(T, T, dtype=torch.bool, device=dev).tril()
sref = sref.masked_fill(~mask, float("-inf"))
lse_ref = sref.logsumexp(-1)
print("case4 lse maxerr:", (lse3 - lse_ref).abs().max().item())
