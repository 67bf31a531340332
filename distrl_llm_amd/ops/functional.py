"""Dispatch layer: HIP/CDNA4 kernels on ROCm devices, torch reference on CPU.

Policy (see package docstring): a CUDA(ROCm) tensor REQUIRES the compiled
gfx950 extension — if it is missing the op raises instead of silently
falling back to eager (so GPU tests can never pass on a non-native path).
Set ``DISTRL_ALLOW_EAGER_GPU=1`` only for debugging.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from . import reference as R
from .build import get_extension, extension_available


def _require_ext(op: str):
    ext = get_extension()
    if ext is None:
        if os.environ.get("DISTRL_ALLOW_EAGER_GPU", "0") == "1":
            return None
        raise RuntimeError(
            f"distrl_llm_amd op '{op}' called on a GPU tensor but the gfx950 "
            f"extension is not built/loadable. Run `python -m "
            f"distrl_llm_amd.ops.build` (or __graft_entry__.build()) first.")
    return ext


# --------------------------------------------------------------- rmsnorm

class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = _require_ext("rmsnorm")
        if ext is None:
            y = R.rmsnorm(x, weight, eps)
            ctx.save_for_backward(x, weight)
            ctx.eps = eps
            ctx.used_ext = False
            return y
        y = ext.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight)
        ctx.eps = eps
        ctx.used_ext = True
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        if ctx.used_ext:
            ext = get_extension()
            dx = ext.rmsnorm_bwd(dy.contiguous(), x, weight, ctx.eps)
        else:
            x32 = x.float()
            w32 = weight.float()
            dyw = dy.float() * w32
            var = x32.pow(2).mean(-1, keepdim=True)
            r = torch.rsqrt(var + ctx.eps)
            dot = (dyw * x32).mean(-1, keepdim=True)
            dx = (dyw * r - x32 * dot * r.pow(3)).to(x.dtype)
        return dx, None, None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if not x.is_cuda:
        return R.rmsnorm(x, weight, eps)
    if not torch.is_grad_enabled() or not x.requires_grad:
        ext = _require_ext("rmsnorm")
        if ext is None:
            return R.rmsnorm(x, weight, eps)
        return ext.rmsnorm_fwd(x.contiguous(), weight, eps)
    return _RMSNormFn.apply(x.contiguous(), weight, eps)


# -------------------------------------------------------------- silu_mul

class _SiluMulFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        ctx.save_for_backward(gate, up)
        if gate.is_cuda:
            ext = _require_ext("silu_mul")
            if ext is not None:
                return ext.silu_mul_fwd(gate.contiguous(), up.contiguous())
        return R.silu_mul(gate, up)

    @staticmethod
    def backward(ctx, dy):
        gate, up = ctx.saved_tensors
        if gate.is_cuda and extension_available():
            ext = get_extension()
            dg, du = ext.silu_mul_bwd(dy.contiguous(), gate.contiguous(), up.contiguous())
            return dg, du
        g32 = gate.float()
        sig = torch.sigmoid(g32)
        silu = g32 * sig
        dsilu = sig * (1 + g32 * (1 - sig))
        dg = (dy.float() * up.float() * dsilu).to(gate.dtype)
        du = (dy.float() * silu).to(up.dtype)
        return dg, du


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if not gate.is_cuda:
        return R.silu_mul(gate, up)
    if not torch.is_grad_enabled() or not (gate.requires_grad or up.requires_grad):
        ext = _require_ext("silu_mul")
        if ext is None:
            return R.silu_mul(gate, up)
        return ext.silu_mul_fwd(gate.contiguous(), up.contiguous())
    return _SiluMulFn.apply(gate, up)


# ------------------------------------------------------------------ rope

def apply_rope_inplace(q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
                       inv_freq: torch.Tensor, theta: float) -> Tuple[torch.Tensor, torch.Tensor]:
    """Inference-path RoPE: q (T, H, D), k (T, KV, D), positions (T,),
    inv_freq (D/2,) fp32 precomputed. On GPU uses the fused in-place
    kernel; on CPU returns rotated copies."""
    if q.is_cuda:
        ext = _require_ext("rope")
        if ext is not None:
            ext.rope_inplace(q, k, positions.to(torch.int32), inv_freq)
            return q, k
    cos, sin = R.rope_cos_sin(positions, q.shape[-1], theta, device=q.device)
    return R.apply_rope(q, k, cos, sin)


# ------------------------------------------------------------- kv cache

def kv_cache_scatter(k, v, key_cache, value_cache, slot_mapping):
    if k.is_cuda:
        ext = _require_ext("kv_cache_scatter")
        if ext is not None:
            ext.kv_cache_scatter(k.contiguous(), v.contiguous(), key_cache,
                                 value_cache, slot_mapping.to(torch.int32))
            return
    R.kv_cache_scatter(k, v, key_cache, value_cache, slot_mapping)


# ---------------------------------------------------- paged attn decode

def paged_attention_decode(q, key_cache, value_cache, block_tables,
                           context_lens, scale: float):
    if q.is_cuda:
        ext = _require_ext("paged_attention_decode")
        if ext is not None:
            return ext.paged_attention_decode(
                q.contiguous(), key_cache, value_cache,
                block_tables.to(torch.int32), context_lens.to(torch.int32),
                float(scale))
    return R.paged_attention_decode(q, key_cache, value_cache, block_tables,
                                    context_lens, scale)


# -------------------------------------------------------------- sampling

def sample_tokens(logits: torch.Tensor, temperature: float, top_p: float,
                  top_k: int, seeds: Optional[torch.Tensor] = None,
                  step: Optional[torch.Tensor] = None,
                  generator: Optional[torch.Generator] = None) -> torch.Tensor:
    if logits.is_cuda:
        ext = _require_ext("sample_tokens")
        if ext is not None and temperature > 0.0:
            if seeds is None:
                seeds = torch.randint(0, 2**31 - 1, (logits.shape[0],),
                                      device=logits.device, dtype=torch.int64,
                                      generator=generator)
            if step is None:
                step = torch.zeros(1, dtype=torch.int64, device=logits.device)
            fn = (ext.sample_tokens2 if logits.shape[0] <= 256
                  else ext.sample_tokens)
            return fn(logits.contiguous(), float(temperature),
                      float(top_p), int(top_k), seeds, step)
    return R.sample_tokens(logits, temperature, top_p, top_k, generator=generator)


# ------------------------------------------------- training RoPE fwd/bwd

class _RopeTrainFn(torch.autograd.Function):
    """First-party RoPE for the TRAINING path (SURVEY.md §2.4-B "RoPE
    fwd/bwd" row): forward rotates q/k with the fused HIP kernel
    (ops/csrc/rope.hip, already numerics-tested on the inference path);
    backward is the SAME kernel with NEGATED frequencies — the gradient
    of a rotation is the transposed rotation, and R(-theta) = R(theta)^T
    exactly (sin is odd, cos even). Formula verified on CPU against
    torch autograd (test_autograd_formulas), kernel fwd/bwd verified on
    GPU (test_ops_gpu::test_rope_training)."""

    @staticmethod
    def forward(ctx, q, k, positions, inv_freq):
        ext = _require_ext("rope")
        B, T, H, D = q.shape
        KV = k.shape[2]
        qf = q.reshape(B * T, H, D).contiguous().clone()
        kf = k.reshape(B * T, KV, D).contiguous().clone()
        ext.rope_inplace(qf, kf, positions.to(torch.int32), inv_freq)
        ctx.save_for_backward(positions, inv_freq)
        ctx.shapes = (B, T, H, KV, D)
        return qf.view(B, T, H, D), kf.view(B, T, KV, D)

    @staticmethod
    def backward(ctx, dq, dk):
        positions, inv_freq = ctx.saved_tensors
        B, T, H, KV, D = ctx.shapes
        if dq.dtype == torch.bfloat16 and dk.dtype == torch.bfloat16:
            ext = _require_ext("rope")
            dqf = dq.reshape(B * T, H, D).contiguous().clone()
            dkf = dk.reshape(B * T, KV, D).contiguous().clone()
            ext.rope_inplace(dqf, dkf, positions.to(torch.int32), -inv_freq)
            return (dqf.view(B, T, H, D), dkf.view(B, T, KV, D), None, None)
        # non-bf16 upstream grads (e.g. fp32 test harnesses): exact torch
        # rotation with negated sin — the kernel is bf16-only
        freqs = positions.to(torch.float32).unsqueeze(-1) * inv_freq
        cos = freqs.cos().view(B, T, -1)
        sin = freqs.sin().view(B, T, -1)
        gq, gk = R.apply_rope(dq, dk, cos, -sin)
        return gq, gk, None, None


def rope_training(q, k, positions, inv_freq):
    """q (B,T,H,D), k (B,T,KV,D) bf16 on GPU; positions flat (B*T,)."""
    return _RopeTrainFn.apply(q, k, positions, inv_freq)


# ----------------------------------------------------- nf4 base linear

class _NF4LinearFn(torch.autograd.Function):
    """Base projection through the fused nf4-dequant MFMA GEMM when the
    bf16 base image has been freed (LoRALinear.free_base_to_nf4_; big-
    model learner path — SURVEY.md §2.4-B "learner nf4 GEMM" row). The
    base is frozen, so backward only needs dX = dY @ W with W dequanted
    on the fly (transient, never resident)."""

    @staticmethod
    def forward(ctx, x, w4f, amaxf, packed, absmax, bias, N, K):
        ext = _require_ext("nf4_linear")
        x2 = x.reshape(-1, K)
        if x2.dtype != torch.bfloat16:
            x2 = x2.to(torch.bfloat16)
        y = ext.nf4_gemm(x2.contiguous(), w4f, amaxf, bias, None, None,
                         N, K, 0)
        ctx.save_for_backward(packed, absmax)
        ctx.NK = (N, K)
        ctx.in_dtype = x.dtype
        return y.view(*x.shape[:-1], N)

    @staticmethod
    def backward(ctx, dy):
        packed, absmax = ctx.saved_tensors
        N, K = ctx.NK
        W = R.dequantize_nf4(packed, absmax, (N, K), 64, torch.bfloat16)
        dx = (dy.reshape(-1, N).to(torch.bfloat16) @ W).to(ctx.in_dtype)
        return (dx.view(*dy.shape[:-1], K), None, None, None, None, None,
                None, None)


def nf4_linear(x, w4f, amaxf, packed, absmax, bias, N: int, K: int):
    return _NF4LinearFn.apply(x, w4f, amaxf, packed, absmax, bias, N, K)


# --------------------------------------------------- flash attention

class _FlashAttnFn(torch.autograd.Function):
    """First-party CDNA4 causal GQA flash attention (ops/csrc/attention.hip)
    — the learner's training attention without aotriton/Triton (SURVEY.md
    §2.4-B row "causal attention fwd/bwd"; reference hot path
    distributed_actor.py:241-243 + loss.backward())."""

    @staticmethod
    def forward(ctx, q, k, v, scale):
        ext = _require_ext("flash_attention")
        o, lse = ext.flash_attn_fwd(q, k, v, float(scale))
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = float(scale)
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        ext = _require_ext("flash_attention")
        dq, dk, dv = ext.flash_attn_bwd(dout.to(q.dtype).contiguous(),
                                        q, k, v, o, lse, ctx.scale)
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    scale: float) -> torch.Tensor:
    """Causal GQA attention, (B, H, T, D) layout, bf16. k/v carry the KV
    head count natively (no repeat_interleave materialization)."""
    return _FlashAttnFn.apply(q, k, v, scale)


# ------------------------------------------------------------ fused loss

class _LogprobLossFn(torch.autograd.Function):
    """Fused per-token log-softmax + gather + masked PG/GRPO loss.

    forward returns the scalar loss; backward emits dlogits directly
    without materializing a (B, T, V) log-prob tensor (SURVEY.md §2.4-B
    north star). loss = -mean_b( (sum_t logp*mask / sum_t mask) * R_b ).
    """

    @staticmethod
    def forward(ctx, logits, targets, mask, rewards, loss_scale):
        ext = _require_ext("logprob_loss") if logits.is_cuda else None
        m = mask.to(torch.float32)
        denom = m.sum(-1).clamp_min(1.0)
        coef = -rewards.to(torch.float32) / denom / logits.shape[0] * loss_scale
        logits = logits.contiguous()
        if ext is not None:
            tok, lse = ext.logprob_lse_fwd(logits, targets.contiguous())
        else:
            logp = logits.float().log_softmax(-1)
            tok = logp.gather(-1, targets.unsqueeze(-1)).squeeze(-1)
            lse = torch.logsumexp(logits.float(), -1)
        loss = ((tok * m).sum(-1) * coef).sum()
        ctx.save_for_backward(logits, targets, m, coef, lse)
        ctx.used_ext = ext is not None
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, m, coef, lse = ctx.saved_tensors
        w = (m * coef.unsqueeze(-1)) * dloss.float()
        if ctx.used_ext:
            ext = get_extension()
            dlogits = ext.logprob_loss_bwd(logits, targets, w, lse)
        else:
            probs = (logits.float() - lse.unsqueeze(-1)).exp()
            dlogits = probs * (-w).unsqueeze(-1)
            dlogits.scatter_add_(-1, targets.unsqueeze(-1), w.unsqueeze(-1))
            dlogits = dlogits.to(logits.dtype)
        return dlogits, None, None, None, None


def logprob_loss(logits: torch.Tensor, targets: torch.Tensor, mask: torch.Tensor,
                 rewards: torch.Tensor, loss_scale: float = 1.0) -> torch.Tensor:
    """PG/GRPO loss on answer-region logits. logits (B, T, V); targets/mask
    (B, T); rewards (B,). ``loss_scale`` folds the reference's
    loss/num_batches scaling (reference distributed_actor.py:382)."""
    return _LogprobLossFn.apply(logits, targets, mask, rewards, loss_scale)
