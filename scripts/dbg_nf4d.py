import os, sys, torch
sys.path.insert(0, ".")
from distrl_llm_amd.ops.build import build
from distrl_llm_amd.models.quant import prepack_nf4_fragments
from distrl_llm_amd.ops import reference as R
ext = build()
dev = torch.device("cuda:0")
torch.manual_seed(11)
M, N, K = 160, 4608, 3584
w = torch.randn(N, K, device=dev) * 0.05
packed, absmax = R.quantize_nf4(w, 64)
w4f, amaxf = prepack_nf4_fragments(packed, absmax, N, K)
x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
ya = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
yb = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
print("DBG", os.environ.get("DISTRL_NF4_DBG", "0"),
      "nondet:", (ya.float()-yb.float()).abs().max().item())
