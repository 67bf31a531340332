"""Pure-PyTorch reference implementations of every custom op.

These are the numerics ground truth the HIP kernels are tested against
("Numerics tests for a HIP kernel compare it against a plain
PyTorch fp32 reference of the same op") and the CPU execution path for the
gloo plumbing config (BASELINE.json config 1).

Op inventory mirrors the implicit kernel surface of the reference's
dependency stack: RMSNorm, RoPE, SiLU-and-mul, KV-cache
scatter, paged attention (prefill + decode), fused sampling, nf4
quant/dequant, and the fused log-prob + advantage loss.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------- RMSNorm

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """y = x / rms(x) * w, computed in fp32 like the fused kernels."""
    dtype = x.dtype
    x32 = x.float()
    var = x32.pow(2).mean(-1, keepdim=True)
    y = x32 * torch.rsqrt(var + eps)
    return (y * weight.float()).to(dtype)


# ------------------------------------------------------------------- RoPE

def rope_cos_sin(positions: torch.Tensor, head_dim: int, theta: float,
                 dtype: torch.dtype = torch.float32,
                 device: Optional[torch.device] = None) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables for given integer positions; shape (..., head_dim//2)."""
    device = device if device is not None else positions.device
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, device=device,
                                             dtype=torch.float32) / head_dim))
    freqs = positions.to(torch.float32).unsqueeze(-1) * inv_freq
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def apply_rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
               sin: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rotate q, k. q: (..., n_heads, head_dim), k: (..., n_kv, head_dim);
    cos/sin: broadcastable (..., head_dim//2). Uses the HF 'rotate_half'
    convention (first half paired with second half)."""
    def rot(x):
        d = x.shape[-1] // 2
        x1, x2 = x[..., :d], x[..., d:]
        c = cos.unsqueeze(-2).to(x.dtype)
        s = sin.unsqueeze(-2).to(x.dtype)
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    return rot(q), rot(k)


# ------------------------------------------------------------- SiLU * mul

def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return F.silu(gate) * up


# --------------------------------------------------------------- KV cache

def kv_cache_scatter(k: torch.Tensor, v: torch.Tensor, key_cache: torch.Tensor,
                     value_cache: torch.Tensor, slot_mapping: torch.Tensor) -> None:
    """Scatter new K/V rows into the paged pool.

    k, v: (num_tokens, n_kv, head_dim); caches: (num_blocks, block_size,
    n_kv, head_dim); slot_mapping: (num_tokens,) flat slot = block *
    block_size + offset; slot < 0 means skip.
    """
    block_size = key_cache.shape[1]
    valid = slot_mapping >= 0
    slots = slot_mapping[valid]
    blk = torch.div(slots, block_size, rounding_mode="floor")
    off = slots % block_size
    key_cache[blk, off] = k[valid].to(key_cache.dtype)
    value_cache[blk, off] = v[valid].to(value_cache.dtype)


# ------------------------------------------------------- paged attention

def paged_attention_decode(q: torch.Tensor, key_cache: torch.Tensor,
                           value_cache: torch.Tensor, block_tables: torch.Tensor,
                           context_lens: torch.Tensor, scale: float) -> torch.Tensor:
    """Single-token decode attention over the paged KV pool.

    q: (num_seqs, n_heads, head_dim); caches: (num_blocks, block_size, n_kv,
    head_dim); block_tables: (num_seqs, max_blocks) int32; context_lens:
    (num_seqs,) — number of valid KV tokens per sequence (including the
    token written this step). GQA by head grouping.
    """
    num_seqs, n_heads, head_dim = q.shape
    n_kv = key_cache.shape[2]
    block_size = key_cache.shape[1]
    group = n_heads // n_kv
    out = torch.empty_like(q, dtype=torch.float32)
    for s in range(num_seqs):
        L = int(context_lens[s])
        nb = (L + block_size - 1) // block_size
        blocks = block_tables[s, :nb].long()
        keys = key_cache[blocks].reshape(nb * block_size, n_kv, head_dim)[:L]
        vals = value_cache[blocks].reshape(nb * block_size, n_kv, head_dim)[:L]
        qs = q[s].float()  # (H, D)
        keys = keys.float().repeat_interleave(group, dim=1)  # (L, H, D)
        vals = vals.float().repeat_interleave(group, dim=1)
        scores = torch.einsum("hd,lhd->hl", qs, keys) * scale
        probs = scores.softmax(-1)
        out[s] = torch.einsum("hl,lhd->hd", probs, vals)
    return out.to(q.dtype)


def varlen_prefill_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                             seq_lens, scale: float) -> torch.Tensor:
    """Causal attention over concatenated variable-length prompts.

    q: (total_tokens, n_heads, head_dim); k/v: (total_tokens, n_kv, head_dim);
    seq_lens: list[int] summing to total_tokens.
    """
    n_heads = q.shape[1]
    n_kv = k.shape[1]
    group = n_heads // n_kv
    outs = []
    start = 0
    for L in seq_lens:
        qs = q[start:start + L].float().transpose(0, 1)           # (H, L, D)
        ks = k[start:start + L].float().repeat_interleave(group, 1).transpose(0, 1)
        vs = v[start:start + L].float().repeat_interleave(group, 1).transpose(0, 1)
        scores = qs @ ks.transpose(-1, -2) * scale                # (H, L, L)
        mask = torch.ones(L, L, dtype=torch.bool, device=q.device).tril()
        scores = scores.masked_fill(~mask, float("-inf"))
        outs.append((scores.softmax(-1) @ vs).transpose(0, 1))    # (L, H, D)
        start += L
    return torch.cat(outs, 0).to(q.dtype)


# ---------------------------------------------------------------- sampling

def sample_tokens(logits: torch.Tensor, temperature: float, top_p: float,
                  top_k: int, generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Temperature / top-k / top-p sampling. logits: (B, V) -> (B,) int64.

    temperature == 0 means greedy (argmax).
    """
    if temperature == 0.0:
        return logits.argmax(-1)
    logits = logits.float() / temperature
    if top_k and top_k > 0 and top_k < logits.shape[-1]:
        kth = logits.topk(top_k, dim=-1).values[..., -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if top_p < 1.0:
        sorted_logits, sorted_idx = logits.sort(-1, descending=True)
        probs = sorted_logits.softmax(-1)
        cum = probs.cumsum(-1)
        # keep smallest set with cumulative prob >= top_p (token that crosses
        # the threshold is kept)
        cut = cum - probs >= top_p
        sorted_logits = sorted_logits.masked_fill(cut, float("-inf"))
        logits = torch.full_like(logits, float("-inf")).scatter(-1, sorted_idx, sorted_logits)
    probs = logits.softmax(-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


# -------------------------------------------------------------------- nf4

# The 16 nf4 code values (normalized normal-float quantiles), identical to
# the bitsandbytes nf4 codebook the reference's 4-bit models use.
NF4_CODE = torch.tensor([
    -1.0, -0.6961928009986877, -0.5250730514526367, -0.39491748809814453,
    -0.28444138169288635, -0.18477343022823334, -0.09105003625154495, 0.0,
    0.07958029955625534, 0.16093020141124725, 0.24611230194568634,
    0.33791524171829224, 0.44070982933044434, 0.5626170039176941,
    0.7229568362236023, 1.0,
])


def quantize_nf4(w: torch.Tensor, block_size: int = 64) -> Tuple[torch.Tensor, torch.Tensor]:
    """Blockwise nf4 quantization. w: any shape, numel % block_size == 0.

    Returns (packed uint8 of shape (numel//2,), absmax fp32 of shape
    (numel//block_size,)). Two 4-bit codes per byte, first value in the low
    nibble.
    """
    flat = w.detach().float().reshape(-1)
    n = flat.numel()
    if n % block_size != 0:
        raise ValueError(f"numel {n} not divisible by block_size {block_size}")
    blocks = flat.view(-1, block_size)
    absmax = blocks.abs().amax(dim=1).clamp_min(1e-12)
    normed = blocks / absmax.unsqueeze(1)
    code = NF4_CODE.to(w.device)
    idx = (normed.unsqueeze(-1) - code).abs().argmin(-1).to(torch.uint8)  # (nb, bs)
    idx = idx.view(-1)
    packed = (idx[0::2] | (idx[1::2] << 4)).contiguous()
    return packed, absmax.float()


def dequantize_nf4(packed: torch.Tensor, absmax: torch.Tensor, shape,
                   block_size: int = 64, dtype: torch.dtype = torch.float32) -> torch.Tensor:
    code = NF4_CODE.to(packed.device)
    lo = (packed & 0xF).long()
    hi = (packed >> 4).long()
    idx = torch.stack([lo, hi], dim=1).view(-1)
    vals = code[idx].view(-1, block_size) * absmax.unsqueeze(1)
    return vals.view(shape).to(dtype)


# ----------------------------------------------- fused log-prob + loss

def logprob_gather(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Per-token log p(target). logits: (B, T, V) (any float dtype, done in
    fp32), targets: (B, T) int64 -> (B, T) fp32."""
    logp = logits.float().log_softmax(-1)
    return logp.gather(-1, targets.unsqueeze(-1)).squeeze(-1)


def pg_loss(log_probs: torch.Tensor, mask: torch.Tensor,
            rewards: torch.Tensor) -> torch.Tensor:
    """-mean_b( (sum_t logp*mask / sum_t mask) * R_b )  — the reference PG
    loss (reference distributed_actor.py:375). GRPO's surrogate
    exp(logp - logp.detach()) has value 1 and the identical gradient
    (SURVEY.md §2.6-6), so one implementation serves both."""
    m = mask.to(log_probs.dtype)
    seq_mean = (log_probs * m).sum(-1) / m.sum(-1).clamp_min(1.0)
    return -(seq_mean * rewards.to(log_probs.dtype)).mean()


def adam8bit_reference(param, grad, m, v, step, lr, beta1=0.9, beta2=0.999,
                       eps=1e-8):
    """Plain fp32 Adam step used as ground truth for the 8-bit kernel."""
    m.mul_(beta1).add_(grad, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
    mhat = m / (1 - beta1 ** step)
    vhat = v / (1 - beta2 ** step)
    param.add_(-lr * mhat / (vhat.sqrt() + eps))
