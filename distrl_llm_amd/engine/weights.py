"""Fused decode weights: per-layer merged (base + LoRA) projection tensors.

The decode hot loop must not pay 14 adapter GEMMs + 7 base GEMMs per layer
(measured ~10 ms/step at batch 160 on Qwen2.5-7B). LoRA only changes
between rounds, so the engine folds it once per weight-sync:

    W_eff = W_base + scale * (B @ A)

and concatenates the per-site projections — one qkv GEMM (q|k|v) and one
gate|up GEMM per layer — halving GEMM launches and giving hipBLASLt fatter
N at decode's skinny M. Refresh cost is ~100 ms for 7B (196 rank-32
GEMMs + adds), amortized over a whole generation round (SURVEY.md §2.4-A:
LoRA folded into the projection epilogue).
"""

from __future__ import annotations

from typing import List

import torch


class LayerWeights:
    __slots__ = ("qkv_w", "qkv_b", "o_w", "gateup_w", "down_w",
                 "in_norm", "post_norm")


def _effective(mod) -> torch.Tensor:
    w = mod.weight
    if mod.r > 0:
        w = w + (mod.lora_B.to(torch.float32) @ mod.lora_A.to(torch.float32)
                 ).to(w.dtype) * mod.scale
    return w


class FusedWeights:
    def __init__(self, model):
        self.model = model
        self.layers: List[LayerWeights] = []
        self._built = False

    @torch.no_grad()
    def refresh(self):
        """(Re)build the merged decode weights from the current base + LoRA
        tensors. Called once per weight sync, never inside the decode loop."""
        model = self.model
        first = not self._built
        for li, layer in enumerate(model.model.layers):
            at = layer.self_attn
            mlp = layer.mlp
            qkv = torch.cat([_effective(at.q_proj), _effective(at.k_proj),
                             _effective(at.v_proj)], dim=0).contiguous()
            gateup = torch.cat([_effective(mlp.gate_proj),
                                _effective(mlp.up_proj)], dim=0).contiguous()
            o_w = _effective(at.o_proj).contiguous()
            down_w = _effective(mlp.down_proj).contiguous()
            if first:
                lw = LayerWeights()
                lw.qkv_w = qkv
                lw.gateup_w = gateup
                lw.o_w = o_w
                lw.down_w = down_w
                if at.q_proj.bias is not None:
                    lw.qkv_b = torch.cat([at.q_proj.bias, at.k_proj.bias,
                                          at.v_proj.bias]).contiguous()
                else:
                    lw.qkv_b = None
                lw.in_norm = layer.input_layernorm.weight
                lw.post_norm = layer.post_attention_layernorm.weight
                self.layers.append(lw)
            else:
                lw = self.layers[li]
                lw.qkv_w.copy_(qkv)
                lw.gateup_w.copy_(gateup)
                lw.o_w.copy_(o_w)
                lw.down_w.copy_(down_w)
        self._built = True
