import os, sys, torch
sys.path.insert(0, ".")
from distrl_llm_amd.ops.build import build
from distrl_llm_amd.models.quant import prepack_nf4_fragments
from distrl_llm_amd.ops import reference as R
ext = build()
dev = torch.device("cuda:0")
torch.manual_seed(11)
N, K = 4608, 3584
w = torch.randn(N, K, device=dev) * 0.05
packed, absmax = R.quantize_nf4(w, 64)
w4f, amaxf = prepack_nf4_fragments(packed, absmax, N, K)
wdq = R.dequantize_nf4(packed, absmax, (N, K), 64).to(torch.bfloat16).float()
for M in (16, 160):
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    ref = x.float() @ wdq.t()
    ya = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
    yb = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
    err = (ya.float() - ref).abs()
    print(f"M={M}: maxerr {err.max().item():.4f} bad {(err>0.3).sum().item()} "
          f"nondet {(ya.float()-yb.float()).abs().max().item():.4f}")
