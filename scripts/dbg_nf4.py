import os, sys, torch
sys.path.insert(0, ".")
from distrl_llm_amd.ops.build import build
from distrl_llm_amd.models.quant import prepack_bf16_fragments, prepack_nf4_fragments
from distrl_llm_amd.ops import reference as R
ext = build()
dev = torch.device("cuda:0")
torch.manual_seed(11)
M, N, K, r = 160, 4608, 3584, 96
w = torch.randn(N, K, device=dev) * 0.05
packed, absmax = R.quantize_nf4(w, 64)
w4f, amaxf = prepack_nf4_fragments(packed, absmax, N, K)
wdq = R.dequantize_nf4(packed, absmax, (N, K), 64).to(torch.bfloat16)
x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
base_ref = x.float() @ wdq.float().t()

import os as _os
if _os.environ.get("DISTRL_NF4_RETWS"):
    ws = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
    print("ws shape:", ws.shape)
    ysum = ws.sum(0)
    err_ws = (ysum - base_ref).abs()
    print("torch-reduced ws: max err", err_ws.max().item(),
          "bad:", (err_ws > 0.3).sum().item())
    ws2 = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
    d = (ws - ws2).abs()
    print("slab determinism:", d.max().item())
    if d.max() > 0.01:
        bad = (d > 0.01).nonzero()
        print("nondet slabs count:", len(bad))
        print("by z:", [int((bad[:,0]==z).sum()) for z in range(ws.shape[0])])
        print("sample idx (z,m,n):", bad[:6].tolist())
    raise SystemExit
# no-lora, no-bias
y0 = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
err = (y0.float() - base_ref).abs()
print("base: max err", err.max().item(), "bad>0.3:", (err > 0.3).sum().item())
if (err > 0.3).sum() > 0:
    bad = (err > 0.3).nonzero()[:10]
    print("bad idx:", bad.tolist())
    m0, n0 = bad[0].tolist()
    print("m,n:", m0, n0, "y:", y0[m0, n0].item(), "ref:", base_ref[m0, n0].item())
    # which mblock/ntile
    print("mblock:", m0 // 16, "ntile:", n0 // 16, "nblock:", n0 // 256)

# lora only difference
u32 = torch.randn(M, r, device=dev) * 0.3
B = (torch.randn(N, r, device=dev) * 0.05).to(torch.bfloat16)
bfr = prepack_bf16_fragments(B)
y1 = ext.nf4_gemm(x, w4f, amaxf, None, u32.contiguous(), bfr, N, K, r)
ref1 = base_ref + u32.to(torch.bfloat16).float() @ B.float().t()
err1 = (y1.float() - ref1).abs()
print("lora: max err", err1.max().item(), "bad:", (err1 > 0.3).sum().item())

# determinism check
ya = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
yb = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
print("determinism: max diff between two runs:", (ya.float()-yb.float()).abs().max().item())
yc = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
print("run3 vs run1:", (ya.float()-yc.float()).abs().max().item())
err_a = (ya.float() - base_ref).abs()
err_b = (yb.float() - base_ref).abs()
print("bad counts:", (err_a > 0.3).sum().item(), (err_b > 0.3).sum().item())

import os
if os.environ.get("DISTRL_NF4_RETWS"):
    ws = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
    print("ws shape:", ws.shape)
    ysum = ws.sum(0)
    err_ws = (ysum - base_ref).abs()
    print("torch-reduced ws: max err", err_ws.max().item(),
          "bad:", (err_ws > 0.3).sum().item())
    ws2 = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
    d = (ws - ws2).abs()
    print("slab determinism:", d.max().item())
    if d.max() > 0.01:
        bad = (d > 0.01).nonzero()[:8]
        print("nondet slab idx (z,m,n):", bad.tolist())
