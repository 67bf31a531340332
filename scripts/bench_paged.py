import sys, time, torch
sys.path.insert(0, ".")
from distrl_llm_amd.ops.build import build
ext = build()
dev = torch.device("cuda:0")
torch.manual_seed(0)
N, H, KV, D, bs = 160, 28, 4, 128, 16
ctx = 800
max_nb = (1550 + bs - 1) // bs
nb = N * max_nb + 8
kc = torch.randn(nb, bs, KV, D, device=dev, dtype=torch.bfloat16)
vc = torch.randn_like(kc)
q = torch.randn(N, H, D, device=dev, dtype=torch.bfloat16)
bt = torch.randperm(nb)[:N * max_nb].view(N, max_nb).int().to(dev)
cl = torch.full((N,), ctx, dtype=torch.int32, device=dev)
def run():
    return ext.paged_attention_decode(q, kc, vc, bt, cl, D ** -0.5)
for _ in range(10): run()
torch.cuda.synchronize()
t0 = time.time()
for _ in range(100): run()
torch.cuda.synchronize()
us = (time.time() - t0) / 100 * 1e6
kvb = N * ctx * KV * D * 2 * 2
print(f"paged decode N={N} ctx={ctx}: {us:.1f}us ({kvb/us*1e6/1e12:.2f} TB/s KV)")
