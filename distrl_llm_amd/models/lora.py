"""LoRA adapter layer + PEFT-format checkpoint I/O.

Replaces the reference's Unsloth/PEFT stack (reference helper.py:25-46:
rank-r adapters on the 7 projection modules q/k/v/o/gate/up/down,
lora_alpha scaling, dropout, bias="none") and the unsloth-zoo
save_lora/load_lora adapter directory (reference distributed_actor.py:12,
84-86,150). The on-disk format is the PEFT adapter directory —
``adapter_config.json`` + ``adapter_model.safetensors`` with
``base_model.model.model.layers.N.<module>.lora_{A,B}.weight`` keys — which
is the north-star checkpoint format.
"""

from __future__ import annotations

import json
import math
import os
from typing import Dict, List

import torch
import torch.nn as nn
import torch.nn.functional as F

TARGET_MODULES = ["q_proj", "k_proj", "v_proj", "o_proj",
                  "gate_proj", "up_proj", "down_proj"]


class LoRALinear(nn.Module):
    """Frozen base linear + trainable low-rank adapter.

    y = x W^T + b + scale * (x A^T) B^T, scale = alpha / r.
    A is kaiming-uniform initialized, B zeros (PEFT defaults), so a fresh
    adapter is an exact no-op — matching the reference's first-round
    behavior of generating from base weights before any save
    (SURVEY.md §2.6-3).
    """

    def __init__(self, in_features: int, out_features: int, bias: bool,
                 r: int = 0, alpha: float = 16.0, dropout: float = 0.0,
                 dtype: torch.dtype = torch.float32, device=None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.r = r
        self.alpha = alpha
        self.scale = alpha / r if r > 0 else 0.0
        self.weight = nn.Parameter(
            torch.empty(out_features, in_features, dtype=dtype, device=device),
            requires_grad=False)
        if bias:
            self.bias = nn.Parameter(
                torch.empty(out_features, dtype=dtype, device=device),
                requires_grad=False)
        else:
            self.register_parameter("bias", None)
        if r > 0:
            self.lora_A = nn.Parameter(
                torch.empty(r, in_features, dtype=dtype, device=device))
            self.lora_B = nn.Parameter(
                torch.empty(out_features, r, dtype=dtype, device=device))
            self.lora_dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()
        else:
            self.register_parameter("lora_A", None)
            self.register_parameter("lora_B", None)
        # nf4 sidecar (packed weights for the generation engine's 4-bit path)
        self.register_buffer("weight_nf4", None, persistent=False)
        self.register_buffer("weight_absmax", None, persistent=False)
        # fragment packs for the fused nf4 GEMM when the bf16 base image
        # is freed (big-model learner path) — see free_base_to_nf4_
        self.register_buffer("weight_w4f", None, persistent=False)
        self.register_buffer("weight_amaxf", None, persistent=False)

    def reset_lora(self, generator=None):
        if self.r > 0:
            nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5), generator=generator)
            nn.init.zeros_(self.lora_B)

    def free_base_to_nf4_(self):
        """Drop the bf16 base image; forward thereafter runs the fused
        nf4 GEMM against the prepacked fragments (dX dequants on the fly).
        Cuts resident base-weight memory 4.3x for the learner/prefill of
        very large models (72B bf16 image = 145 GiB does not fit beside
        the KV pool and activations)."""
        from .quant import prepack_nf4_fragments
        assert self.weight_nf4 is not None
        self.weight_w4f, self.weight_amaxf = prepack_nf4_fragments(
            self.weight_nf4, self.weight_absmax,
            self.out_features, self.in_features)
        self.weight.data = self.weight.data.new_empty(0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.weight.numel() == 0:
            from ..ops import functional as OF
            y = OF.nf4_linear(x, self.weight_w4f, self.weight_amaxf,
                              self.weight_nf4, self.weight_absmax,
                              self.bias, self.out_features,
                              self.in_features)
        else:
            y = F.linear(x, self.weight, self.bias)
        if self.r > 0:
            xd = self.lora_dropout(x)
            u = F.linear(xd, self.lora_A)
            # addmm folds the scale and the accumulation into the adapter
            # GEMM itself (no separate mul/add kernels in the hot path)
            y2 = y.reshape(-1, self.out_features)
            y = torch.addmm(y2, u.reshape(-1, self.r), self.lora_B.t(),
                            beta=1.0, alpha=self.scale).view_as(y)
        return y

    def extra_repr(self) -> str:
        return (f"in={self.in_features}, out={self.out_features}, "
                f"bias={self.bias is not None}, r={self.r}, alpha={self.alpha}")


def lora_state_dict(model: nn.Module) -> Dict[str, torch.Tensor]:
    """PEFT-keyed adapter state dict (see module docstring for key format)."""
    out = {}
    for name, mod in model.named_modules():
        if isinstance(mod, LoRALinear) and mod.r > 0:
            key = f"base_model.model.{name}"
            out[f"{key}.lora_A.weight"] = mod.lora_A.detach()
            out[f"{key}.lora_B.weight"] = mod.lora_B.detach()
    return out


def load_lora_state_dict(model: nn.Module, state: Dict[str, torch.Tensor]) -> int:
    loaded = 0
    mods = {f"base_model.model.{n}": m for n, m in model.named_modules()
            if isinstance(m, LoRALinear) and m.r > 0}
    for key, mod in mods.items():
        a = state.get(f"{key}.lora_A.weight")
        b = state.get(f"{key}.lora_B.weight")
        if a is None or b is None:
            raise KeyError(f"adapter state missing tensors for {key}")
        with torch.no_grad():
            mod.lora_A.copy_(a.to(mod.lora_A.dtype))
            mod.lora_B.copy_(b.to(mod.lora_B.dtype))
        loaded += 1
    return loaded


def save_adapter(model: nn.Module, path: str, base_model_name: str,
                 r: int, alpha: float, dropout: float = 0.0) -> None:
    """Write a PEFT adapter directory (adapter_config.json +
    adapter_model.safetensors)."""
    os.makedirs(path, exist_ok=True)
    cfg = {
        "peft_type": "LORA",
        "base_model_name_or_path": base_model_name,
        "r": r,
        "lora_alpha": alpha,
        "lora_dropout": dropout,
        "target_modules": TARGET_MODULES,
        "bias": "none",
        "task_type": "CAUSAL_LM",
        "use_rslora": False,
        "fan_in_fan_out": False,
        "inference_mode": False,
    }
    with open(os.path.join(path, "adapter_config.json"), "w") as f:
        json.dump(cfg, f, indent=2)
    from safetensors.torch import save_file
    state = {k: v.contiguous().cpu() for k, v in lora_state_dict(model).items()}
    save_file(state, os.path.join(path, "adapter_model.safetensors"))


def adapter_hyperparams(path: str):
    """(r, alpha, dropout) from a PEFT adapter directory's
    adapter_config.json — lets loaders build a matching model instead of
    guessing the rank."""
    import json
    with open(os.path.join(path, "adapter_config.json")) as f:
        cfg = json.load(f)
    return (int(cfg.get("r", 32)), float(cfg.get("lora_alpha", 16)),
            float(cfg.get("lora_dropout", 0.0)))


def load_adapter(model: nn.Module, path: str) -> int:
    """Load a PEFT adapter directory saved by save_adapter (or by PEFT)."""
    from safetensors.torch import load_file
    state = load_file(os.path.join(path, "adapter_model.safetensors"))
    return load_lora_state_dict(model, state)


def trainable_parameters(model: nn.Module) -> List[torch.nn.Parameter]:
    return [p for p in model.parameters() if p.requires_grad]
