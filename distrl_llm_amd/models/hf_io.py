"""HF-checkpoint interop: load/save base weights in the Hugging Face
safetensors layout, and derive a ModelSpec from a checkpoint's
``config.json``.

The reference loads real pretrained repos through Unsloth/HF
(reference distributed_actor.py:58-66); there is no network here, but a
user with a locally downloaded Qwen2/Llama checkpoint directory gets the
same capability: pass the directory as ``--model`` and the worker loads
its weights instead of random-initializing. Our parameter names already
mirror HF's exactly (``model.layers.N.self_attn.q_proj.weight`` ... —
chosen for PEFT key parity), so loading is a direct name-for-name copy.
"""

from __future__ import annotations

import json
import os
from typing import List, Optional

import torch

from .spec import ModelSpec

# checkpoint keys that are legitimately absent from our module tree
_IGNORABLE_SUBSTRINGS = ("rotary_emb.inv_freq",)


def is_hf_checkpoint_dir(path: str) -> bool:
    if not os.path.isdir(path):
        return False
    if os.path.exists(os.path.join(path, "model.safetensors")):
        return True
    if os.path.exists(os.path.join(path, "model.safetensors.index.json")):
        return True
    return any(f.endswith(".safetensors") and f != "adapter_model.safetensors"
               for f in os.listdir(path))


def spec_from_hf_config(path: str, name: Optional[str] = None) -> ModelSpec:
    """Build a ModelSpec from a checkpoint's config.json (Qwen2/Llama
    families — the architectures the reference targets)."""
    with open(os.path.join(path, "config.json")) as f:
        cfg = json.load(f)
    archs = cfg.get("architectures") or [""]
    arch = archs[0]
    if not any(a in arch for a in ("Qwen2", "Qwen3", "Llama", "Mistral")):
        raise ValueError(f"unsupported architecture {arch!r} in {path}")
    sw = cfg.get("sliding_window")
    if arch.startswith("Mistral") and sw is not None \
            and sw < cfg.get("max_position_embeddings", 32768):
        # pre-v0.3 Mistral uses sliding-window attention; this stack
        # implements plain causal attention (all BASELINE.json families),
        # so accepting the checkpoint would silently change semantics
        # beyond `sliding_window` tokens of context
        raise ValueError(
            f"windowed Mistral checkpoint (sliding_window={sw}) is not "
            f"supported; use a v0.3+ checkpoint (sliding_window null)")
    heads = cfg["num_attention_heads"]
    head_dim = cfg.get("head_dim") or cfg["hidden_size"] // heads
    return ModelSpec(
        name=name or os.path.basename(os.path.normpath(path)) or "local",
        vocab_size=cfg["vocab_size"],
        hidden_size=cfg["hidden_size"],
        intermediate_size=cfg["intermediate_size"],
        num_layers=cfg["num_hidden_layers"],
        num_heads=heads,
        num_kv_heads=cfg.get("num_key_value_heads", heads),
        head_dim=head_dim,
        rope_theta=float(cfg.get("rope_theta", 1e4)),
        rms_norm_eps=float(cfg.get("rms_norm_eps", 1e-6)),
        tie_word_embeddings=bool(cfg.get("tie_word_embeddings", False)),
        # Llama exposes attention_bias (default False); Qwen2 always
        # biases q/k/v; Qwen3 drops the bias and adds per-head q/k norms
        qkv_bias=bool(cfg.get("attention_bias", arch.startswith("Qwen2"))),
        max_position=int(cfg.get("max_position_embeddings", 32768)),
        qk_norm=arch.startswith("Qwen3"),
    )


def _shard_files(path: str) -> List[str]:
    idx = os.path.join(path, "model.safetensors.index.json")
    if os.path.exists(idx):
        with open(idx) as f:
            weight_map = json.load(f)["weight_map"]
        return [os.path.join(path, f) for f in sorted(set(weight_map.values()))]
    single = os.path.join(path, "model.safetensors")
    if os.path.exists(single):
        return [single]
    return [os.path.join(path, f) for f in sorted(os.listdir(path))
            if f.endswith(".safetensors") and f != "adapter_model.safetensors"]


def load_hf_checkpoint(model, path: str, strict: bool = True) -> int:
    """Copy a HF safetensors checkpoint's tensors into ``model`` (a
    CausalLM). Returns the number of tensors loaded. With ``strict``,
    every base (non-LoRA) parameter must be covered — except
    ``lm_head.weight`` on tie_word_embeddings models, where either the
    checkpoint omits it or ties it to the embedding."""
    from safetensors.torch import load_file

    params = dict(model.named_parameters())
    base_keys = {k for k in params if "lora_" not in k}
    loaded = set()
    with torch.no_grad():
        for file in _shard_files(path):
            for k, v in load_file(file).items():
                if k in params:
                    if params[k].shape != v.shape:
                        raise ValueError(
                            f"shape mismatch for {k}: checkpoint "
                            f"{tuple(v.shape)} vs model "
                            f"{tuple(params[k].shape)}")
                    params[k].copy_(v.to(params[k].dtype))
                    loaded.add(k)
                elif k == "lm_head.weight" and model.lm_head is None:
                    pass  # tied-embedding checkpoint shipping the tie anyway
                elif any(s in k for s in _IGNORABLE_SUBSTRINGS):
                    pass
                elif strict:
                    raise ValueError(f"unexpected checkpoint tensor {k!r}")
    missing = base_keys - loaded
    if strict and missing:
        raise ValueError(f"checkpoint {path} missing tensors: "
                         f"{sorted(missing)[:8]}{'...' if len(missing) > 8 else ''}")
    return len(loaded)


def save_hf_checkpoint(model, path: str) -> None:
    """Write the model's base weights as a single-file HF checkpoint
    (model.safetensors + config.json), loadable by load_hf_checkpoint
    and by transformers."""
    from safetensors.torch import save_file

    os.makedirs(path, exist_ok=True)
    state = {k: v.detach().contiguous().cpu()
             for k, v in model.named_parameters() if "lora_" not in k}
    save_file(state, os.path.join(path, "model.safetensors"))
    _write_hf_config(model.spec, path)


def _write_hf_config(s: ModelSpec, path: str) -> None:
    if s.qk_norm:
        arch, mtype = "Qwen3ForCausalLM", "qwen3"
    elif s.qkv_bias:
        arch, mtype = "Qwen2ForCausalLM", "qwen2"
    elif "mistral" in s.name:
        arch, mtype = "MistralForCausalLM", "mistral"
    else:
        arch, mtype = "LlamaForCausalLM", "llama"
    cfg = {
        "architectures": [arch],
        "model_type": mtype,
        "hidden_size": s.hidden_size,
        "intermediate_size": s.intermediate_size,
        "num_hidden_layers": s.num_layers,
        "num_attention_heads": s.num_heads,
        "num_key_value_heads": s.num_kv_heads,
        "head_dim": s.head_dim,
        "vocab_size": s.vocab_size,
        "rope_theta": s.rope_theta,
        "rms_norm_eps": s.rms_norm_eps,
        "tie_word_embeddings": s.tie_word_embeddings,
        "attention_bias": s.qkv_bias,
        "max_position_embeddings": s.max_position,
        "torch_dtype": "bfloat16",
    }
    if mtype == "mistral":
        # v0.3 semantics: no window (transformers defaults to 4096)
        cfg["sliding_window"] = None
    with open(os.path.join(path, "config.json"), "w") as f:
        json.dump(cfg, f, indent=2)


def save_merged_checkpoint(model, path: str) -> int:
    """Merge the trained LoRA adapters into the base weights and write a
    plain HF checkpoint (the deploy artifact the reference ecosystem
    calls ``save_pretrained_merged``): W_merged = W + scale * B@A per
    adapted projection. The model is not mutated. Returns the number of
    merged projection sites."""
    from safetensors.torch import save_file

    merged = {k: v.detach().clone().float()
              for k, v in model.named_parameters() if "lora_" not in k}
    n = 0
    for name, mod in model.named_modules():
        a = getattr(mod, "lora_A", None)
        b = getattr(mod, "lora_B", None)
        if a is None or b is None:
            continue
        key = f"{name}.weight"
        merged[key] = (merged[key]
                       + mod.scale * (b.detach().float()
                                      @ a.detach().float()))
        n += 1
    dtype = next(model.parameters()).dtype
    os.makedirs(path, exist_ok=True)
    save_file({k: v.to(dtype).contiguous().cpu() for k, v in merged.items()},
              os.path.join(path, "model.safetensors"))
    _write_hf_config(model.spec, path)
    return n


def resolve_spec(model_name_or_path: str) -> ModelSpec:
    """get_spec, extended: a local HF checkpoint directory derives its
    spec from config.json; known names hit the registry."""
    if (os.path.isdir(model_name_or_path)
            and os.path.exists(os.path.join(model_name_or_path,
                                            "config.json"))):
        return spec_from_hf_config(model_name_or_path)
    from .spec import get_spec
    return get_spec(model_name_or_path)
