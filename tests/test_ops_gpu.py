"""Kernel numerics: each gfx950 HIP kernel vs the plain PyTorch fp32
reference (SURVEY.md §4 test strategy)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from distrl_llm_amd.ops.build import build
    return build()


def _dev():
    return torch.device("cuda:0")


# --------------------------------------------------------------- rmsnorm

@pytest.mark.parametrize("rows,hidden", [(7, 3584), (256, 896), (1, 128)])
def test_rmsnorm_fwd(ext, rows, hidden):
    torch.manual_seed(0)
    from distrl_llm_amd.ops import reference as R
    x = torch.randn(rows, hidden, device=_dev(), dtype=torch.bfloat16)
    w = torch.randn(hidden, device=_dev(), dtype=torch.bfloat16)
    y = ext.rmsnorm_fwd(x, w, 1e-6)
    ref = R.rmsnorm(x.float(), w.float(), 1e-6)
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)


def test_rmsnorm_bwd(ext):
    torch.manual_seed(1)
    rows, hidden = 5, 1024
    x32 = torch.randn(rows, hidden, device=_dev(), requires_grad=True)
    w32 = torch.randn(hidden, device=_dev())
    from distrl_llm_amd.ops import reference as R
    y = R.rmsnorm(x32, w32, 1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)
    dx_ref = x32.grad

    dx = ext.rmsnorm_bwd(dy.bfloat16(), x32.detach().bfloat16(),
                         w32.bfloat16(), 1e-6)
    torch.testing.assert_close(dx.float(), dx_ref, rtol=5e-2, atol=5e-2)


# -------------------------------------------------------------- silu_mul

def test_silu_mul(ext):
    torch.manual_seed(2)
    g = torch.randn(33, 1024, device=_dev(), dtype=torch.bfloat16)
    u = torch.randn(33, 1024, device=_dev(), dtype=torch.bfloat16)
    y = ext.silu_mul_fwd(g, u)
    ref = torch.nn.functional.silu(g.float()) * u.float()
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)

    dy = torch.randn_like(g)
    g32 = g.float().requires_grad_(True)
    u32 = u.float().requires_grad_(True)
    (torch.nn.functional.silu(g32) * u32).backward(dy.float())
    dg, du = ext.silu_mul_bwd(dy, g, u)
    torch.testing.assert_close(dg.float(), g32.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(du.float(), u32.grad, rtol=5e-2, atol=5e-2)


# ------------------------------------------------------------------ rope

def test_rope_inplace(ext):
    torch.manual_seed(3)
    from distrl_llm_amd.ops import reference as R
    T, H, KV, D, theta = 9, 8, 2, 128, 1e6
    q = torch.randn(T, H, D, device=_dev(), dtype=torch.bfloat16)
    k = torch.randn(T, KV, D, device=_dev(), dtype=torch.bfloat16)
    pos = torch.randint(0, 500, (T,), device=_dev())
    cos, sin = R.rope_cos_sin(pos, D, theta, device=_dev())
    q_ref, k_ref = R.apply_rope(q.float(), k.float(), cos, sin)

    inv_freq = 1.0 / (theta ** (torch.arange(0, D, 2, device=_dev(),
                                             dtype=torch.float32) / D))
    ext.rope_inplace(q, k, pos.int(), inv_freq)
    torch.testing.assert_close(q.float(), q_ref, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(k.float(), k_ref, rtol=2e-2, atol=2e-2)


# -------------------------------------------------------------- kv cache

def test_kv_scatter(ext):
    torch.manual_seed(4)
    from distrl_llm_amd.ops import reference as R
    T, KV, D, nb, bs = 21, 2, 64, 16, 16
    k = torch.randn(T, KV, D, device=_dev(), dtype=torch.bfloat16)
    v = torch.randn(T, KV, D, device=_dev(), dtype=torch.bfloat16)
    kc = torch.zeros(nb, bs, KV, D, device=_dev(), dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    kc_ref = kc.clone()
    vc_ref = vc.clone()
    slots = torch.randperm(nb * bs, device=_dev())[:T].int()
    slots[3] = -1  # skip entry
    ext.kv_cache_scatter(k, v, kc, vc, slots)
    R.kv_cache_scatter(k, v, kc_ref, vc_ref, slots.long())
    torch.testing.assert_close(kc, kc_ref)
    torch.testing.assert_close(vc, vc_ref)


# ----------------------------------------------------------- paged attn

@pytest.mark.parametrize("D,H,KV,ctxs", [
    (128, 28, 4, [1, 17, 333, 1550]),
    (64, 8, 2, [5, 100]),
])
def test_paged_attention_decode(ext, D, H, KV, ctxs):
    torch.manual_seed(5)
    from distrl_llm_amd.ops import reference as R
    bs = 16
    N = len(ctxs)
    max_nb = (max(ctxs) + bs - 1) // bs
    nb = max_nb * N + 2
    q = torch.randn(N, H, D, device=_dev(), dtype=torch.bfloat16)
    kc = torch.randn(nb, bs, KV, D, device=_dev(), dtype=torch.bfloat16)
    vc = torch.randn(nb, bs, KV, D, device=_dev(), dtype=torch.bfloat16)
    perm = torch.randperm(nb)[:N * max_nb]
    bt = perm.view(N, max_nb).int().to(_dev())
    ctx = torch.tensor(ctxs, dtype=torch.int32, device=_dev())
    scale = 1.0 / math.sqrt(D)

    out = ext.paged_attention_decode(q, kc, vc, bt, ctx, scale)
    ref = R.paged_attention_decode(q.float(), kc.float(), vc.float(),
                                   bt.long(), ctx, scale)
    torch.testing.assert_close(out.float(), ref, rtol=2e-2, atol=2e-2)


# -------------------------------------------------------------- sampling

@pytest.mark.parametrize("impl", ["sample_tokens", "sample_tokens2"])
def test_sample_tokens_distribution(ext, impl):
    """Statistical check: a peaked 3-token distribution sampled many times
    matches the renormalized top-p categorical within tolerance."""
    torch.manual_seed(6)
    V = 1000
    logits = torch.full((1, V), -20.0, device=_dev())
    logits[0, 10] = math.log(0.6)
    logits[0, 20] = math.log(0.3)
    logits[0, 30] = math.log(0.1)
    n = 4000
    rows = logits.expand(n, V).contiguous()
    seeds = torch.arange(n, device=_dev(), dtype=torch.int64) * 7919 + 13
    step = torch.zeros(1, dtype=torch.int64, device=_dev())
    fn = getattr(ext, impl)
    out = fn(rows, 1.0, 1.0, 0, seeds, step)
    counts = torch.bincount(out.cpu(), minlength=V).float() / n
    assert abs(counts[10] - 0.6) < 0.04
    assert abs(counts[20] - 0.3) < 0.04
    assert abs(counts[30] - 0.1) < 0.03
    assert counts.sum() == pytest.approx(1.0)

    # top_p = 0.65 keeps only token 10 (0.6 < 0.65 needs token 20 too: the
    # reference keeps the crossing token) -> {10, 20} renormalized
    out2 = fn(rows, 1.0, 0.65, 0, seeds, step)
    c2 = torch.bincount(out2.cpu(), minlength=V).float() / n
    assert c2[30] == 0.0
    assert abs(c2[10] - 0.6 / 0.9) < 0.05

    # top_k = 1 is greedy
    out3 = fn(rows, 1.0, 1.0, 1, seeds, step)
    assert (out3 == 10).all()


@pytest.mark.parametrize("impl", ["sample_tokens", "sample_tokens2"])
def test_sample_tokens_temperature(ext, impl):
    V = 128
    logits = torch.zeros(2000, V, device=_dev())
    logits[:, 5] = 2.0
    seeds = torch.arange(2000, device=_dev(), dtype=torch.int64)
    step = torch.zeros(1, dtype=torch.int64, device=_dev())
    # low temperature sharpens: nearly all mass on token 5
    out = getattr(ext, impl)(logits, 0.05, 1.0, 0, seeds, step)
    assert (out == 5).float().mean() > 0.99


# ------------------------------------------------------------ fused loss

def test_logprob_loss_kernel(ext):
    torch.manual_seed(7)
    from distrl_llm_amd.ops import reference as R
    B, T, V = 4, 33, 152064
    logits = torch.randn(B, T, V, device=_dev(), dtype=torch.bfloat16)
    targets = torch.randint(0, V, (B, T), device=_dev())
    tok, lse = ext.logprob_lse_fwd(logits, targets)
    ref_logp = R.logprob_gather(logits.float(), targets)
    ref_lse = torch.logsumexp(logits.float(), -1)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(tok, ref_logp, rtol=1e-4, atol=1e-4)

    w = torch.randn(B, T, device=_dev())
    dl = ext.logprob_loss_bwd(logits, targets, w, lse)
    probs = (logits.float() - lse.unsqueeze(-1)).exp()
    ref_dl = -w.unsqueeze(-1) * probs
    ref_dl.scatter_add_(-1, targets.unsqueeze(-1), w.unsqueeze(-1))
    torch.testing.assert_close(dl.float(), ref_dl, rtol=1e-2, atol=1e-3)


def test_autograd_loss_on_gpu():
    from distrl_llm_amd.ops import functional as OF
    torch.manual_seed(8)
    B, T, V = 2, 9, 5000
    logits = torch.randn(B, T, V, device=_dev(), dtype=torch.bfloat16,
                         requires_grad=True)
    targets = torch.randint(0, V, (B, T), device=_dev())
    mask = (torch.rand(B, T, device=_dev()) > 0.3).long()
    mask[:, 0] = 1
    rewards = torch.randn(B, device=_dev())
    loss = OF.logprob_loss(logits, targets, mask, rewards, 0.5)
    loss.backward()

    l2 = logits.detach().float().clone().requires_grad_(True)
    logp = l2.log_softmax(-1).gather(-1, targets.unsqueeze(-1)).squeeze(-1)
    m = mask.float()
    ref = -(((logp * m).sum(-1) / m.sum(-1)) * rewards).mean() * 0.5
    ref.backward()
    torch.testing.assert_close(loss.float(), ref, rtol=1e-2, atol=1e-3)
    torch.testing.assert_close(logits.grad.float(), l2.grad, rtol=5e-2,
                               atol=1e-4)


# ----------------------------------------------------------------- adam

def test_adam8bit_kernel_matches_cpu_path():
    from distrl_llm_amd.train.optim import Adam8bit
    torch.manual_seed(9)
    p_gpu = torch.nn.Parameter(torch.randn(3000, device=_dev()))
    p_cpu = torch.nn.Parameter(p_gpu.detach().cpu().clone())
    o_gpu = Adam8bit([p_gpu], lr=1e-2)
    o_cpu = Adam8bit([p_cpu], lr=1e-2)
    for i in range(10):
        g = torch.randn(3000)
        p_gpu.grad = g.to(_dev())
        p_cpu.grad = g.clone()
        o_gpu.step()
        o_cpu.step()
    torch.testing.assert_close(p_gpu.cpu(), p_cpu.detach(), rtol=1e-3,
                               atol=1e-3)


# ----------------------------------------------------- nf4 MFMA GEMM

def test_mfma_mapping_probe(ext):
    """Verify the assumed 16x16x32 bf16 fragment mapping: probe computes
    A @ B^T under the documented lane layout. Asymmetric operands so a
    transposed mapping cannot pass (guide G9)."""
    torch.manual_seed(10)
    a = torch.randn(16, 32, device=_dev(), dtype=torch.bfloat16)
    b = torch.randn(16, 32, device=_dev(), dtype=torch.bfloat16) * 0.5
    b[3] += 2.0  # extra asymmetry
    c = ext.mfma_probe(a, b)
    ref = a.float() @ b.float().t()
    torch.testing.assert_close(c, ref, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("M,N,K,r,bias", [
    (160, 512, 512, 0, False),
    (7, 256, 512, 0, True),
    (160, 4608, 3584, 96, True),
    (33, 2048, 1024, 64, False),
])
def test_nf4_gemm(ext, M, N, K, r, bias):
    from distrl_llm_amd.models.quant import (prepack_bf16_fragments,
                                             prepack_nf4_fragments)
    from distrl_llm_amd.ops import reference as R
    torch.manual_seed(11)
    dev = _dev()
    w = torch.randn(N, K, device=dev) * 0.05
    packed, absmax = R.quantize_nf4(w, 64)
    w4f, amaxf = prepack_nf4_fragments(packed, absmax, N, K)
    wdq = R.dequantize_nf4(packed, absmax, (N, K), 64).to(torch.bfloat16)
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    b = (torch.randn(N, device=dev, dtype=torch.bfloat16)
         if bias else None)

    ref = x.float() @ wdq.float().t()
    if bias:
        ref = ref + b.float()
    u = bfr = None
    if r > 0:
        u32 = torch.randn(M, r, device=dev) * 0.3
        B = (torch.randn(N, r, device=dev) * 0.05).to(torch.bfloat16)
        bfr = prepack_bf16_fragments(B)
        u = u32.contiguous()
        ref = ref + u32.to(torch.bfloat16).float() @ B.float().t()

    y = ext.nf4_gemm(x, w4f, amaxf, b, u, bfr, N, K, r)
    torch.testing.assert_close(y.float(), ref, rtol=3e-2, atol=3e-1)


def test_lora_u(ext):
    from distrl_llm_amd.models.quant import prepack_bf16_fragments
    torch.manual_seed(12)
    dev = _dev()
    M, K, r = 160, 3584, 96
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    A = (torch.randn(r, K, device=dev) * 0.05).to(torch.bfloat16)
    afrag = prepack_bf16_fragments(A)
    u = torch.zeros(M, r, device=dev, dtype=torch.float32)
    ext.lora_u(x, afrag, u, r, 4)
    ref = x.float() @ A.float().t()
    torch.testing.assert_close(u, ref, rtol=2e-2, atol=5e-1)


def test_engine_nf4_matches_merged():
    """The nf4 fused-GEMM decode path must agree with the merged-bf16 path
    up to nf4 quantization of the base weights (same quantized image both
    sides, adapters exact): greedy outputs identical."""
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    spec = get_spec("small-qwen2")
    dev = torch.device("cuda:0")
    model = CausalLM(spec, lora_r=8, lora_alpha=16, dtype=torch.bfloat16,
                     device=dev).random_init(21)
    model.quantize_nf4_()
    # give LoRA a nonzero contribution
    with torch.no_grad():
        for p in model.parameters():
            if p.requires_grad:
                p.add_(torch.randn_like(p) * 0.02)
    cfg = EngineConfig(max_seq_length=256, kv_block_size=16,
                       num_kv_blocks=512, max_num_seqs=64)
    prompts = [list(range(1, 40)), [5, 7, 11, 13]]
    sp = SamplingParams(max_tokens=8, temperature=0.0, n=1)

    import os
    os.environ["DISTRL_DECODE_NF4"] = "1"
    try:
        eng_nf4 = Engine(model, cfg, device=dev, seed=0)
        assert eng_nf4.fused.nf4, "nf4 path must be active for this spec"
        out_nf4 = eng_nf4.generate(prompts, sp, eos_token_id=None)
    finally:
        del os.environ["DISTRL_DECODE_NF4"]

    eng_bf16 = Engine(model, cfg, device=dev, seed=0)
    eng_bf16.fused.nf4 = False
    out_bf16 = eng_bf16.generate(prompts, sp, eos_token_id=None)

    for a, b in zip(out_nf4, out_bf16):
        agree = sum(x == y for x, y in zip(a[0], b[0]))
        assert agree >= 7, (a[0], b[0])


@pytest.mark.parametrize("D,H,KV,lens", [
    (128, 28, 4, [210, 17, 333]),
    (64, 8, 2, [100, 5, 64]),
])
def test_prefill_attention(ext, D, H, KV, lens):
    torch.manual_seed(13)
    from distrl_llm_amd.ops import reference as R
    total = sum(lens)
    q = torch.randn(total, H, D, device=_dev(), dtype=torch.bfloat16)
    k = torch.randn(total, KV, D, device=_dev(), dtype=torch.bfloat16)
    v = torch.randn(total, KV, D, device=_dev(), dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    tq0, trows, tkv0 = [], [], []
    start = 0
    for L in lens:
        for r0 in range(0, L, 16):
            tq0.append(start + r0)
            trows.append(min(16, L - r0))
            tkv0.append(start)
        start += L
    dev = _dev()
    out = ext.prefill_attention(
        q, k, v, torch.tensor(tq0, dtype=torch.int32, device=dev),
        torch.tensor(trows, dtype=torch.int32, device=dev),
        torch.tensor(tkv0, dtype=torch.int32, device=dev), max(lens), scale)
    ref = R.varlen_prefill_attention(q.float(), k.float(), v.float(), lens,
                                     scale)
    torch.testing.assert_close(out.float(), ref, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("B,Hq,Hkv,T,D", [
    (2, 8, 2, 100, 64),
    (1, 4, 4, 333, 128),
    (2, 28, 4, 160, 128),   # Qwen2.5-7B head geometry, short T
])
def test_flash_attention_fwd_bwd(ext, B, Hq, Hkv, T, D):
    """First-party flash attention vs fp32 torch reference: forward and all
    three input grads (SURVEY.md §2.4-B training attention row)."""
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    scale = D ** -0.5
    q = (torch.randn(B, Hq, T, D, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, T, D, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, T, D, device=dev) * 0.5).to(torch.bfloat16)
    dout = (torch.randn(B, Hq, T, D, device=dev) * 0.5).to(torch.bfloat16)

    from distrl_llm_amd.ops import functional as OF
    qg, kg, vg = (t.clone().requires_grad_(True) for t in (q, k, v))
    o = OF.flash_attention(qg, kg, vg, scale)
    o.backward(dout)

    # fp32 reference with GQA expansion
    group = Hq // Hkv
    qr = q.float().requires_grad_(True)
    kr = k.float().requires_grad_(True)
    vr = v.float().requires_grad_(True)
    ke = kr.repeat_interleave(group, 1)
    ve = vr.repeat_interleave(group, 1)
    s = torch.einsum("bhqd,bhkd->bhqk", qr, ke) * scale
    mask = torch.ones(T, T, dtype=torch.bool, device=dev).tril()
    s = s.masked_fill(~mask, float("-inf"))
    p = s.softmax(-1)
    orf = torch.einsum("bhqk,bhkd->bhqd", p, ve)
    orf.backward(dout.float())

    assert torch.allclose(o.float(), orf, atol=3e-2, rtol=3e-2), \
        (o.float() - orf).abs().max()
    assert torch.allclose(qg.grad.float(), qr.grad, atol=8e-2, rtol=8e-2), \
        (qg.grad.float() - qr.grad).abs().max()
    assert torch.allclose(kg.grad.float(), kr.grad, atol=8e-2, rtol=8e-2), \
        (kg.grad.float() - kr.grad).abs().max()
    assert torch.allclose(vg.grad.float(), vr.grad, atol=8e-2, rtol=8e-2), \
        (vg.grad.float() - vr.grad).abs().max()


def test_rope_training_fwd_bwd(ext):
    """Training RoPE fwd/bwd (ops/functional.rope_training): the fused
    HIP kernel forward + negated-frequency backward vs fp32 torch
    autograd (SURVEY.md §2.4-B RoPE fwd/bwd row; backward formula proven
    in test_autograd_formulas)."""
    from distrl_llm_amd.ops import functional as OF
    from distrl_llm_amd.ops import reference as R
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    B, T, H, KV, D = 2, 33, 4, 2, 64
    theta = 1e4
    pos = torch.arange(T, device=dev).repeat(B)
    inv_freq = (1.0 / (theta ** (torch.arange(0, D, 2, device=dev,
                                              dtype=torch.float32) / D)))
    q = (torch.randn(B, T, H, D, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, T, KV, D, device=dev) * 0.5).to(torch.bfloat16)
    dq = torch.randn_like(q)
    dk = torch.randn_like(k)

    qg = q.clone().requires_grad_(True)
    kg = k.clone().requires_grad_(True)
    qo, ko = OF.rope_training(qg, kg, pos, inv_freq)
    # bf16 upstream grads so backward exercises the KERNEL path (fp32
    # grads take the exact-torch fallback)
    torch.autograd.backward([qo, ko], [dq, dk])

    qr = q.float().requires_grad_(True)
    kr = k.float().requires_grad_(True)
    cos, sin = R.rope_cos_sin(pos.view(B, T), D, theta, dtype=torch.float32)
    qor, kor = R.apply_rope(qr, kr, cos, sin)
    (qor * dq.float()).sum().backward(retain_graph=True)
    (kor * dk.float()).sum().backward()

    assert torch.allclose(qo.float(), qor, atol=2e-2, rtol=2e-2)
    assert torch.allclose(ko.float(), kor, atol=2e-2, rtol=2e-2)
    assert torch.allclose(qg.grad.float(), qr.grad, atol=2e-2, rtol=2e-2)
    assert torch.allclose(kg.grad.float(), kr.grad, atol=2e-2, rtol=2e-2)
