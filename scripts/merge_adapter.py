#!/usr/bin/env python3
"""Merge a trained PEFT LoRA adapter into base weights and export a plain
HF checkpoint (the deploy artifact — the reference ecosystem's
`save_pretrained_merged`):

    python scripts/merge_adapter.py --model qwen2.5-7b \
        --adapter run_x/model_100 --out merged_ckpt

--model is a registry name (random-init base, for testing) or a local HF
checkpoint directory (real pretrained base).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from distrl_llm_amd.models.hf_io import (is_hf_checkpoint_dir,
                                         load_hf_checkpoint, resolve_spec,
                                         save_merged_checkpoint)
from distrl_llm_amd.models.lora import load_adapter
from distrl_llm_amd.models.model import CausalLM


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True)
    ap.add_argument("--adapter", required=True)
    ap.add_argument("--out", required=True)
    ap.add_argument("--lora_r", type=int, default=0,
                    help="0 = read from the adapter's adapter_config.json")
    ap.add_argument("--lora_alpha", type=float, default=0.0,
                    help="0 = read from the adapter's adapter_config.json")
    ap.add_argument("--seed", type=int, default=3407,
                    help="random-init seed when --model is a name")
    args = ap.parse_args()

    spec = resolve_spec(args.model)
    r, alpha = args.lora_r, args.lora_alpha
    if r <= 0 or alpha <= 0:
        from distrl_llm_amd.models.lora import adapter_hyperparams
        ar, aa, _ = adapter_hyperparams(args.adapter)
        r = r if r > 0 else ar
        alpha = alpha if alpha > 0 else aa
    model = CausalLM(spec, lora_r=r, lora_alpha=alpha,
                     dtype=torch.float32)
    model.random_init(args.seed)
    if is_hf_checkpoint_dir(args.model):
        load_hf_checkpoint(model, args.model)
    n_loaded = load_adapter(model, args.adapter)
    n_merged = save_merged_checkpoint(model, args.out)
    print(f"merged {n_merged} projection sites "
          f"({n_loaded} adapter tensors) -> {args.out}")


if __name__ == "__main__":
    main()
