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
wdq = R.dequantize_nf4(packed, absmax, (N, K), 64).to(torch.bfloat16).float()
x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)

ws = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
ks = ws.shape[0]
print("ksplit:", ks)
# slice refs: z0: panels 0-1 (k 0:1024), z1: 1024:2048, z2: 2048:3072, z3: 3072:3584
bounds = [0, 1024, 2048, 3072, 3584]
for z in range(ks):
    a, b = bounds[z], bounds[z+1]
    ref_z = x.float()[:, a:b] @ wdq[:, a:b].t()
    err = (ws[z] - ref_z).abs()
    print(f"z{z}: max err {err.max().item():.4f} bad>0.1 {(err>0.1).sum().item()}")
    if (err > 0.1).sum() > 0:
        bad = (err > 0.1).nonzero()
        ms = bad[:, 0].unique()
        print("   bad m rows:", ms[:20].tolist(), "count m:", len(ms))
        nb = bad[:, 1]
        print("   bad n range:", nb.min().item(), nb.max().item(), "count:", len(nb))
