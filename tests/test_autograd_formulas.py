"""Checks of the custom autograd backward formulas in ops/functional.py —
the same closed-form gradients the HIP backward kernels implement
(rmsnorm.hip, silu_mul.hip, loss.hip). A failure here means the GPU
kernels were derived from a wrong formula, not just a kernel bug.

The public CPU entry points route through the pure-torch composites (so
the engine/learner CPU paths get plain autograd); these tests invoke the
autograd.Functions directly with the extension dispatch disabled to
exercise the closed-form backward branch, and compare against autograd
through the composite."""

import pytest
import torch

from distrl_llm_amd.ops import functional as OF


@pytest.fixture
def no_ext(monkeypatch):
    """Force the Functions down their formula (non-ext) branch."""
    monkeypatch.setattr(OF, "_require_ext", lambda op: None)
    monkeypatch.setattr(OF, "extension_available", lambda: False)


def test_rmsnorm_backward_formula(no_ext):
    torch.manual_seed(0)
    x = torch.randn(5, 33, requires_grad=True)
    w = torch.randn(33, requires_grad=True)
    dy = torch.randn(5, 33)

    y = OF._RMSNormFn.apply(x, w, 1e-6)
    y.backward(dy)
    dx_formula = x.grad.clone()

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    r = torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-6)
    y2 = x2 * r * w2
    torch.testing.assert_close(y, y2, rtol=1e-5, atol=1e-6)
    y2.backward(dy)
    torch.testing.assert_close(dx_formula, x2.grad, rtol=1e-4, atol=1e-5)


def test_silu_mul_backward_formula(no_ext):
    torch.manual_seed(1)
    g = torch.randn(6, 21, requires_grad=True)
    u = torch.randn(6, 21, requires_grad=True)
    dy = torch.randn(6, 21)

    y = OF._SiluMulFn.apply(g, u)
    y.backward(dy)
    dg_f, du_f = g.grad.clone(), u.grad.clone()

    g2 = g.detach().clone().requires_grad_(True)
    u2 = u.detach().clone().requires_grad_(True)
    y2 = torch.nn.functional.silu(g2) * u2
    torch.testing.assert_close(y, y2, rtol=1e-5, atol=1e-6)
    y2.backward(dy)
    torch.testing.assert_close(dg_f, g2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(du_f, u2.grad, rtol=1e-4, atol=1e-5)


def test_logprob_loss_matches_composite_autograd():
    """The direct-dlogits backward equals autograd through the plain
    log_softmax/gather composite (the reference learner's math,
    reference distributed_actor.py:253-260,375). The fused Function is
    the live CPU path, so no dispatch patching is needed."""
    B, T, V = 3, 6, 19
    torch.manual_seed(2)
    logits = torch.randn(B, T, V, requires_grad=True)
    targets = torch.randint(0, V, (B, T))
    mask = (torch.rand(B, T) > 0.3).float()
    mask[:, 0] = 1.0
    rewards = torch.randn(B)

    loss = OF.logprob_loss(logits, targets, mask, rewards, loss_scale=0.7)
    loss.backward()
    g_fused = logits.grad.clone()

    logits2 = logits.detach().clone().requires_grad_(True)
    logp = torch.log_softmax(logits2, dim=-1)
    tok = logp.gather(-1, targets.unsqueeze(-1)).squeeze(-1)
    per_seq = (tok * mask).sum(-1) / mask.sum(-1).clamp_min(1.0)
    loss2 = -(per_seq * rewards).mean() * 0.7
    loss2.backward()
    torch.testing.assert_close(loss, loss2)
    torch.testing.assert_close(g_fused, logits2.grad, rtol=1e-5, atol=1e-6)


def test_logprob_loss_zero_mask_row_safe():
    """A fully-masked row (degenerate candidate) must contribute zero
    loss and zero gradient, not NaN (clamp_min(1) denominator)."""
    B, T, V = 2, 4, 7
    logits = torch.randn(B, T, V, requires_grad=True)
    targets = torch.randint(0, V, (B, T))
    mask = torch.ones(B, T)
    mask[1] = 0.0
    rewards = torch.tensor([1.0, 5.0])
    loss = OF.logprob_loss(logits, targets, mask, rewards)
    loss.backward()
    assert torch.isfinite(loss)
    assert torch.isfinite(logits.grad).all()
    assert logits.grad[1].abs().sum() == 0


def test_rope_backward_is_negated_frequency_rotation():
    """The training-RoPE backward formula (ops/functional._RopeTrainFn):
    grad of y = R(theta) x is R(-theta) dy. Verified against torch
    autograd through the reference rope on CPU fp64."""
    import torch
    from distrl_llm_amd.ops import reference as R
    torch.manual_seed(0)
    B, T, H, KV, D = 2, 5, 3, 1, 8
    theta = 1e4
    pos = torch.arange(T).repeat(B)
    inv_freq = 1.0 / (theta ** (torch.arange(0, D, 2).double() / D))
    q = torch.randn(B * T, H, D, dtype=torch.float64, requires_grad=True)
    k = torch.randn(B * T, KV, D, dtype=torch.float64, requires_grad=True)
    cos, sin = R.rope_cos_sin(pos, D, theta, dtype=torch.float64)
    qr, kr = R.apply_rope(q.view(B, T, H, D), k.view(B, T, KV, D),
                          cos.view(B, T, -1), sin.view(B, T, -1))
    dq = torch.randn_like(qr)
    dk = torch.randn_like(kr)
    (qr * dq).sum().backward(retain_graph=True)
    gq_auto = q.grad.clone()
    q.grad = None
    (kr * dk).sum().backward()
    gk_auto = k.grad.clone()

    # formula: rotate the output grads with NEGATED frequencies
    ncos, nsin = R.rope_cos_sin(pos, D, theta, dtype=torch.float64)
    gq_f, gk_f = R.apply_rope(dq, dk, ncos.view(B, T, -1),
                              -nsin.view(B, T, -1))
    assert torch.allclose(gq_f.reshape(B * T, H, D), gq_auto, atol=1e-10)
    assert torch.allclose(gk_f.reshape(B * T, KV, D), gk_auto, atol=1e-10)
