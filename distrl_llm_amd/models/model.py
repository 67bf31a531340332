"""Decoder-only transformer (Qwen2 / Llama family) with built-in LoRA.

Native replacement for the reference's Unsloth ``FastLanguageModel``
(reference distributed_actor.py:58-69) and PEFT wrapper (helper.py:25-46):
one module owns the frozen base weights (bf16, optionally nf4-quantized
with the packed sidecar kept for the generation engine's fused 4-bit
kernels) plus the trainable LoRA adapters. The autograd (teacher-forced
training) forward lives here; the paged-KV generation path reads the same
weight tensors from ``engine/``.
"""

from __future__ import annotations

import math
import os
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import reference as R
from .lora import LoRALinear
from .spec import ModelSpec


class RMSNorm(nn.Module):
    def __init__(self, hidden: int, eps: float, dtype, device=None):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden, dtype=dtype, device=device),
                                   requires_grad=False)

    def forward(self, x):
        from ..ops import functional as OF
        return OF.rmsnorm(x, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, spec: ModelSpec, r: int, alpha: float, dropout: float,
                 dtype, device=None):
        super().__init__()
        self.spec = spec
        h, qs, kvs = spec.hidden_size, spec.q_size, spec.kv_size
        self.q_proj = LoRALinear(h, qs, spec.qkv_bias, r, alpha, dropout, dtype, device)
        self.k_proj = LoRALinear(h, kvs, spec.qkv_bias, r, alpha, dropout, dtype, device)
        self.v_proj = LoRALinear(h, kvs, spec.qkv_bias, r, alpha, dropout, dtype, device)
        self.o_proj = LoRALinear(qs, h, False, r, alpha, dropout, dtype, device)
        if spec.qk_norm:
            # Qwen3: per-head RMSNorm on q/k over head_dim, before RoPE
            # (HF Qwen3Attention.q_norm/k_norm)
            self.q_norm = RMSNorm(spec.head_dim, spec.rms_norm_eps, dtype,
                                  device)
            self.k_norm = RMSNorm(spec.head_dim, spec.rms_norm_eps, dtype,
                                  device)
        self.scale = spec.head_dim ** -0.5

    def forward(self, x, cos, sin, attn_bias):
        """attn_bias: additive (B,1,T,T) mask, or None for pure causal
        (right-padded layouts — lets SDPA pick its flash backend)."""
        B, T, _ = x.shape
        s = self.spec
        q = self.q_proj(x).view(B, T, s.num_heads, s.head_dim)
        k = self.k_proj(x).view(B, T, s.num_kv_heads, s.head_dim)
        v = self.v_proj(x).view(B, T, s.num_kv_heads, s.head_dim)
        if s.qk_norm:
            q = self.q_norm(q.reshape(-1, s.head_dim)).view_as(q)
            k = self.k_norm(k.reshape(-1, s.head_dim)).view_as(k)
        if (attn_bias is None and q.is_cuda and q.dtype == torch.bfloat16
                and os.environ.get("DISTRL_ROPE_KERNEL") == "1"):
            # first-party RoPE fwd/bwd (fused HIP kernel; backward =
            # negated-frequency rotation — ops/functional._RopeTrainFn).
            # Opt-in: formula CPU-proven + GPU-tested; the default stays
            # the torch path pending an end-to-end perf pass.
            from ..ops import functional as OF
            pos = torch.arange(T, device=x.device).repeat(B)
            inv_freq = 1.0 / (s.rope_theta ** (
                torch.arange(0, s.head_dim, 2, device=x.device,
                             dtype=torch.float32) / s.head_dim))
            q, k = OF.rope_training(q, k, pos, inv_freq)
        else:
            q, k = R.apply_rope(q, k, cos, sin)
        if (attn_bias is None and q.is_cuda and q.dtype == torch.bfloat16
                and s.head_dim in (64, 128)):
            # first-party CDNA4 flash attention (ops/csrc/attention.hip):
            # native GQA (no repeat_interleave materialization), causal,
            # fused online softmax — replaces torch SDPA's aotriton
            # (Triton-derived) backend on the learner hot path
            from ..ops import functional as OF
            q, k, v = (t.transpose(1, 2).contiguous() for t in (q, k, v))
            o = OF.flash_attention(q, k, v, self.scale)
        else:
            group = s.num_heads // s.num_kv_heads
            k = k.repeat_interleave(group, dim=2)
            v = v.repeat_interleave(group, dim=2)
            q, k, v = (t.transpose(1, 2) for t in (q, k, v))  # (B, H, T, D)
            if attn_bias is None:
                o = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                                   scale=self.scale)
            else:
                o = F.scaled_dot_product_attention(
                    q, k, v, attn_mask=attn_bias, scale=self.scale)
        o = o.transpose(1, 2).reshape(B, T, s.q_size)
        return self.o_proj(o)


class MLP(nn.Module):
    def __init__(self, spec: ModelSpec, r: int, alpha: float, dropout: float,
                 dtype, device=None):
        super().__init__()
        h, f = spec.hidden_size, spec.intermediate_size
        self.gate_proj = LoRALinear(h, f, False, r, alpha, dropout, dtype, device)
        self.up_proj = LoRALinear(h, f, False, r, alpha, dropout, dtype, device)
        self.down_proj = LoRALinear(f, h, False, r, alpha, dropout, dtype, device)

    def forward(self, x):
        from ..ops import functional as OF
        return self.down_proj(OF.silu_mul(self.gate_proj(x), self.up_proj(x)))


class DecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, r: int, alpha: float, dropout: float,
                 dtype, device=None):
        super().__init__()
        self.input_layernorm = RMSNorm(spec.hidden_size, spec.rms_norm_eps, dtype, device)
        self.self_attn = Attention(spec, r, alpha, dropout, dtype, device)
        self.post_attention_layernorm = RMSNorm(spec.hidden_size, spec.rms_norm_eps,
                                                dtype, device)
        self.mlp = MLP(spec, r, alpha, dropout, dtype, device)

    def forward(self, x, cos, sin, attn_bias):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin, attn_bias)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class _Inner(nn.Module):
    """Named 'model' so LoRA state-dict keys match PEFT's
    base_model.model.model.layers.N... layout."""

    def __init__(self, spec, r, alpha, dropout, dtype, device):
        super().__init__()
        self.embed_tokens = nn.Embedding(spec.vocab_size, spec.hidden_size,
                                         dtype=dtype, device=device)
        self.embed_tokens.weight.requires_grad_(False)
        self.layers = nn.ModuleList([
            DecoderLayer(spec, r, alpha, dropout, dtype, device)
            for _ in range(spec.num_layers)])
        self.norm = RMSNorm(spec.hidden_size, spec.rms_norm_eps, dtype, device)


class CausalLM(nn.Module):
    def __init__(self, spec: ModelSpec, lora_r: int = 0, lora_alpha: float = 16.0,
                 lora_dropout: float = 0.0, dtype: torch.dtype = torch.float32,
                 device=None):
        super().__init__()
        self.spec = spec
        self.dtype_ = dtype
        self.model = _Inner(spec, lora_r, lora_alpha, lora_dropout, dtype, device)
        if spec.tie_word_embeddings:
            self.lm_head = None
        else:
            self.lm_head = nn.Linear(spec.hidden_size, spec.vocab_size, bias=False,
                                     dtype=dtype, device=device)
            self.lm_head.weight.requires_grad_(False)

    # -------------------------------------------------------------- init

    @torch.no_grad()
    def random_init(self, seed: int = 3407) -> "CausalLM":
        """Deterministic random init (std 0.02) for synthetic benches
        (BASELINE.json: random-init weights). The reference seeds LoRA with
        random_state=3407 (reference helper.py:43)."""
        dev = next(self.parameters()).device
        # generate on-device when possible (7B on CPU would take minutes);
        # deterministic for a given (seed, architecture, device type)
        g = torch.Generator(device=dev).manual_seed(seed)

        def fill(t, std=0.02):
            t.copy_(torch.randn(t.shape, generator=g, dtype=torch.float32,
                                device=dev).mul_(std).to(t.dtype))

        fill(self.model.embed_tokens.weight)
        for layer in self.model.layers:
            for mod in (layer.self_attn.q_proj, layer.self_attn.k_proj,
                        layer.self_attn.v_proj, layer.self_attn.o_proj,
                        layer.mlp.gate_proj, layer.mlp.up_proj, layer.mlp.down_proj):
                fill(mod.weight)
                if mod.bias is not None:
                    mod.bias.zero_()
                if mod.r > 0:
                    a = torch.empty(mod.lora_A.shape, dtype=torch.float32,
                                    device=dev)
                    nn.init.kaiming_uniform_(a, a=math.sqrt(5), generator=g)
                    mod.lora_A.copy_(a.to(mod.lora_A.dtype))
                    mod.lora_B.zero_()
        if self.lm_head is not None:
            fill(self.lm_head.weight)
        return self

    @torch.no_grad()
    def quantize_nf4_(self, block_size: int = 64,
                      keep_bf16: Optional[bool] = None) -> "CausalLM":
        """nf4-quantize every base projection weight in place: the bf16
        weight is replaced with its quantize->dequantize image (so training
        numerics match the 4-bit model) and the packed nf4 + absmax sidecar
        is attached for the engine's fused 4-bit GEMM kernels.

        keep_bf16=False (default for >16B models on GPU) FREES the bf16
        base image instead — every base projection (learner forward,
        prefill) then runs through the fused nf4 GEMM with on-the-fly
        dequant for dX (SURVEY.md §2.4-B "learner nf4 GEMM" row; removes
        the 65 GiB redundant image at 32B / the 145 GiB one at 72B that
        OOM'd the 72B dual-role trial)."""
        if keep_bf16 is None:
            # auto: free only when the bf16 image threatens residency
            # (72B: image 145 GiB; 32B keeps its image — measured round 1
            # at 199 GiB peak with the faster hipBLASLt learner path)
            dev = next(self.parameters()).device
            n_params = sum(p.numel() for p in self.parameters())
            keep_bf16 = not (dev.type == "cuda" and n_params > 40e9)
        for mod in self.modules():
            if isinstance(mod, LoRALinear):
                packed, absmax = R.quantize_nf4(mod.weight, block_size)
                mod.weight_nf4 = packed
                mod.weight_absmax = absmax
                if keep_bf16:
                    mod.weight.copy_(R.dequantize_nf4(
                        packed, absmax, mod.weight.shape, block_size,
                        dtype=mod.weight.dtype))
                else:
                    mod.free_base_to_nf4_()
        return self

    # ----------------------------------------------------------- forward

    def _rope_tables(self, position_ids: torch.Tensor):
        return R.rope_cos_sin(position_ids, self.spec.head_dim,
                              self.spec.rope_theta, dtype=torch.float32)

    def forward(self, input_ids: torch.Tensor,
                attention_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Teacher-forced forward -> logits (B, T, V)."""
        return self.logits(self.forward_hidden(input_ids, attention_mask))

    def forward_hidden(self, input_ids: torch.Tensor,
                       attention_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Teacher-forced forward -> final hidden states (B, T, H), so the
        caller can evaluate the LM head only where needed (the learner
        skips the prompt region's vocab projection entirely).

        attention_mask: (B, T) 1 = real token, 0 = pad (left-padded prompts
        + right-padded answers, the learner layout of reference
        distributed_actor.py:217-239).
        """
        B, T = input_ids.shape
        if attention_mask is None:
            attention_mask = torch.ones(B, T, dtype=torch.long, device=input_ids.device)
        position_ids = (attention_mask.long().cumsum(-1) - 1).clamp_min(0)
        cos, sin = self._rope_tables(position_ids)

        # Right-pad-only layouts need no mask at all: trailing pads never
        # influence real tokens under causal attention -> SDPA can use its
        # flash backend. Left-padded rows need the full additive bias.
        right_pad_only = bool(
            (attention_mask[:, 0] == 1).all()
            and (attention_mask.long().diff(dim=-1) <= 0).all())
        if right_pad_only:
            bias = None
        else:
            causal = torch.ones(T, T, dtype=torch.bool, device=input_ids.device).tril()
            pad = attention_mask.bool().view(B, 1, 1, T)
            allow = causal.view(1, 1, T, T) & pad
            bias = torch.zeros(B, 1, T, T, dtype=self.dtype_, device=input_ids.device)
            bias.masked_fill_(~allow, torch.finfo(self.dtype_).min)

        x = self.model.embed_tokens(input_ids)
        for layer in self.model.layers:
            x = layer(x, cos, sin, bias)
        return self.model.norm(x)

    def logits(self, hidden: torch.Tensor) -> torch.Tensor:
        if self.lm_head is None:
            return hidden @ self.model.embed_tokens.weight.t()
        return self.lm_head(hidden)

    @property
    def device(self):
        return next(self.parameters()).device
