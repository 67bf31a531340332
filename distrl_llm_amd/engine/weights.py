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

Replaces the reference's per-generate adapter hot-swap into vLLM
(``load_lora`` inside every fast_generate, reference
distributed_actor.py:148-150) — there the engine applies LoRA at runtime
per token; here the fold happens once per round, which 288 GB of HBM
makes free to cache.
"""

from __future__ import annotations

from typing import List

import torch


class LayerWeights:
    __slots__ = ("qkv_w", "qkv_b", "o_w", "gateup_w", "down_w",
                 "in_norm", "post_norm",
                 # nf4 fragment packs (base, built once) + adapter fragment
                 # packs (rebuilt per refresh) for the fused nf4 GEMM path
                 "qkv_w4", "qkv_amax", "o_w4", "o_amax",
                 "gateup_w4", "gateup_amax", "down_w4", "down_amax",
                 "qkv_afrag", "qkv_bfrag", "o_afrag", "o_bfrag",
                 "gateup_afrag", "gateup_bfrag", "down_afrag", "down_bfrag",
                 "qkv_r", "o_r", "gateup_r", "down_r")


def _effective(mod) -> torch.Tensor:
    w = mod.weight
    if mod.r > 0:
        w = w + (mod.lora_B.to(torch.float32) @ mod.lora_A.to(torch.float32)
                 ).to(w.dtype) * mod.scale
    return w


class FusedWeights:
    def __init__(self, model, use_nf4: bool = True):
        self.model = model
        self.layers: List[LayerWeights] = []
        self._built = False
        # nf4 fused-GEMM decode path: available when the model carries the
        # packed nf4 sidecars (load_in_4bit) and the kernels are built
        q0 = model.model.layers[0].self_attn.q_proj
        s = model.spec
        shapes_ok = (
            (s.q_size + 2 * s.kv_size) % 256 == 0
            and s.hidden_size % 256 == 0
            and (2 * s.intermediate_size) % 256 == 0
            and s.hidden_size % 64 == 0 and s.q_size % 64 == 0
            and s.intermediate_size % 64 == 0)
        # Both decode paths compute exactly nf4(W_base) + B@A: the merged
        # path pre-dequantizes that image into resident bf16 (288 GB HBM
        # makes the cache free) and currently measures faster at decode
        # batch sizes (10.8 vs 14.5 ms/step at batch 160); the fused nf4
        # GEMM path streams 4.3x fewer weight bytes and is selected with
        # DISTRL_DECODE_NF4=1 (kept fully tested).
        import os
        freed = (q0.weight_nf4 is not None and q0.weight.numel() == 0)
        env = os.environ.get("DISTRL_DECODE_NF4")
        if freed:
            # the bf16 base image was dropped (free_base_to_nf4_, big
            # models) — there is nothing to merge, the fused nf4 path is
            # the only decode path
            want_nf4 = True
        elif env is not None:
            want_nf4 = use_nf4 and env == "1"
        else:
            # default: merged cache for models whose bf16 copy is cheap;
            # fused nf4 path for very large models (e.g. 32B: a second
            # 65 GB merged copy is not worth the HBM)
            n_params = sum(p.numel() for p in model.parameters())
            want_nf4 = use_nf4 and n_params > 16e9
        self.nf4 = bool(want_nf4 and q0.weight_nf4 is not None
                        and q0.weight.is_cuda and shapes_ok)
        if freed and not self.nf4:
            raise RuntimeError(
                "base image freed but the fused nf4 decode path is "
                "unavailable for this shape — cannot fall back to merged")
        # hybrid: even when the merged-bf16 cache wins overall, the fused
        # nf4 kernel BEATS hipBLASLt on the deep-K down-projection
        # (61+13 vs 80 us/layer at decode batch 160 — profiles/r02) —
        # run just that site through it
        self.hybrid_down = bool(
            (not self.nf4) and q0.weight_nf4 is not None
            and q0.weight.is_cuda and shapes_ok
            and os.environ.get("DISTRL_HYBRID_DOWN", "1") == "1")
        self.lora_r = q0.r

    @staticmethod
    def _nf4_cat(mods):
        """Concatenate the flat nf4 packs / absmax of modules along N
        (rows are K-contiguous, so flat concatenation is row concat)."""
        packed = torch.cat([m.weight_nf4 for m in mods])
        absmax = torch.cat([m.weight_absmax for m in mods])
        N = sum(m.out_features for m in mods)
        K = mods[0].in_features
        return packed, absmax, N, K

    def _build_nf4_base(self, lw: "LayerWeights", layer):
        from ..models.quant import prepack_nf4_fragments
        at, mlp = layer.self_attn, layer.mlp
        for name, mods in (("qkv", [at.q_proj, at.k_proj, at.v_proj]),
                           ("o", [at.o_proj]),
                           ("gateup", [mlp.gate_proj, mlp.up_proj]),
                           ("down", [mlp.down_proj])):
            packed, absmax, N, K = self._nf4_cat(mods)
            w4f, amaxf = prepack_nf4_fragments(packed, absmax, N, K)
            setattr(lw, f"{name}_w4", w4f)
            setattr(lw, f"{name}_amax", amaxf)

    def _refresh_nf4_adapters(self, lw: "LayerWeights", layer, sites=None):
        """Per-round: stack the A matrices and build the block-diagonal,
        scale-folded B matrix per fused site, prepacked into fragments."""
        from ..models.quant import prepack_bf16_fragments
        at, mlp = layer.self_attn, layer.mlp
        for name, mods in (("qkv", [at.q_proj, at.k_proj, at.v_proj]),
                           ("o", [at.o_proj]),
                           ("gateup", [mlp.gate_proj, mlp.up_proj]),
                           ("down", [mlp.down_proj])):
            if sites is not None and name not in sites:
                continue
            A = torch.cat([m.lora_A.detach() for m in mods], dim=0)  # (r_tot, K)
            r_tot = A.shape[0]
            r_pad = (r_tot + 31) // 32 * 32  # kernel needs rank % 32 == 0
            N = sum(m.out_features for m in mods)
            if r_pad != r_tot:
                A = torch.cat([A, A.new_zeros(r_pad - r_tot, A.shape[1])], 0)
            B = A.new_zeros(N, r_pad)
            n0, r0 = 0, 0
            for m in mods:
                B[n0:n0 + m.out_features, r0:r0 + m.r] = (
                    m.lora_B.detach() * m.scale)
                n0 += m.out_features
                r0 += m.r
            # IN-PLACE refresh: decode sessions capture hipGraphs that
            # reference these tensors, and the graph cache reuses those
            # graphs across generation rounds — a fresh allocation here
            # would leave every cached graph reading freed memory (at
            # single-size waves the allocator happens to hand the same
            # block back, which HID this; multi-wave rounds surfaced it
            # as NaN logits -> the sampler degenerating to token 0)
            af = prepack_bf16_fragments(A.to(torch.bfloat16))
            bf = prepack_bf16_fragments(B.to(torch.bfloat16))
            try:
                old_af = getattr(lw, f"{name}_afrag")
                old_bf = getattr(lw, f"{name}_bfrag")
            except AttributeError:
                old_af = old_bf = None
            if (old_af is not None and old_af.shape == af.shape
                    and old_bf.shape == bf.shape):
                old_af.copy_(af)
                old_bf.copy_(bf)
            else:
                setattr(lw, f"{name}_afrag", af)
                setattr(lw, f"{name}_bfrag", bf)
            setattr(lw, f"{name}_r", r_pad)

    @torch.no_grad()
    def refresh(self):
        """(Re)build the merged decode weights from the current base + LoRA
        tensors. Called once per weight sync, never inside the decode loop.

        nf4 mode: the quantized base fragments are built ONCE (frozen);
        only the adapter fragment packs (A stacks, block-diagonal scaled B)
        are rebuilt per refresh — the adapter stays exact bf16.
        """
        model = self.model
        first = not self._built
        for li, layer in enumerate(model.model.layers):
            at = layer.self_attn
            if first:
                lw = LayerWeights()
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
            if self.nf4:
                if first:
                    self._build_nf4_base(lw, layer)
                self._refresh_nf4_adapters(lw, layer)
            else:
                self._refresh_merged(lw, layer, first)
                if self.hybrid_down:
                    if first:
                        from ..models.quant import prepack_nf4_fragments
                        packed, absmax, N, K = self._nf4_cat(
                            [layer.mlp.down_proj])
                        lw.down_w4, lw.down_amax = prepack_nf4_fragments(
                            packed, absmax, N, K)
                    self._refresh_nf4_adapters(lw, layer, sites=("down",))
        self._built = True

    def _refresh_merged(self, lw: "LayerWeights", layer, first: bool):
        at, mlp = layer.self_attn, layer.mlp
        qkv = torch.cat([_effective(at.q_proj), _effective(at.k_proj),
                         _effective(at.v_proj)], dim=0).contiguous()
        gateup = torch.cat([_effective(mlp.gate_proj),
                            _effective(mlp.up_proj)], dim=0).contiguous()
        o_w = _effective(at.o_proj).contiguous()
        down_w = _effective(mlp.down_proj).contiguous()
        if first:
            lw.qkv_w = qkv
            lw.gateup_w = gateup
            lw.o_w = o_w
            lw.down_w = down_w
        else:
            lw.qkv_w.copy_(qkv)
            lw.gateup_w.copy_(gateup)
            lw.o_w.copy_(o_w)
            lw.down_w.copy_(down_w)
