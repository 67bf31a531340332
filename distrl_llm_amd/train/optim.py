"""Optimizers: blockwise 8-bit Adam (HIP kernel) with fp32 fallback.

Replaces ``bnb.optim.Adam8bit`` (reference distributed_actor.py:209-211,
432-434): Adam whose m/v states are stored blockwise-quantized to 8 bits
with a per-block fp32 absmax (block 256). On GPU the update is one fused
HIP kernel per parameter; on CPU (plumbing config) the same quantized-state
math runs in torch so numerics are testable everywhere.
"""

from __future__ import annotations

from typing import Iterable, List

import torch


class Adam8bit(torch.optim.Optimizer):
    BLOCK = 256

    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8, weight_decay: float = 0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    def _init_state(self, p: torch.Tensor):
        n = p.numel()
        nb = (n + self.BLOCK - 1) // self.BLOCK
        dev = p.device
        return {
            "step": 0,
            "m_q": torch.zeros(n, dtype=torch.int8, device=dev),
            "v_q": torch.zeros(n, dtype=torch.uint8, device=dev),
            "m_absmax": torch.zeros(nb, dtype=torch.float32, device=dev),
            "v_absmax": torch.zeros(nb, dtype=torch.float32, device=dev),
        }

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr, (b1, b2), eps, wd = (group["lr"], group["betas"], group["eps"],
                                     group["weight_decay"])
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if not state:
                    state.update(self._init_state(p))
                state["step"] += 1
                if p.is_cuda:
                    from ..ops.build import get_extension
                    ext = get_extension()
                    if ext is not None:
                        ext.adam8bit_step(
                            p.view(-1), p.grad.contiguous().view(-1),
                            state["m_q"], state["v_q"],
                            state["m_absmax"], state["v_absmax"],
                            float(lr), float(b1), float(b2), float(eps),
                            float(wd), int(state["step"]))
                        continue
                self._step_torch(p, state, lr, b1, b2, eps, wd)
        return loss

    def load_state_dict(self, state_dict):
        """Dtype-preserving load. ``torch.optim.Optimizer.load_state_dict``
        casts floating state tensors to the param dtype, which would
        corrupt the fp32 block absmaxes when params are bf16 — map the
        saved state by position instead and only move devices."""
        saved_groups = state_dict["param_groups"]
        groups = self.param_groups
        if len(saved_groups) != len(groups) or any(
                len(sg["params"]) != len(g["params"])
                for sg, g in zip(saved_groups, groups)):
            raise ValueError("loaded state dict has a different parameter layout")
        id_map = {}
        for sg, g in zip(saved_groups, groups):
            for old_id, p in zip(sg["params"], g["params"]):
                id_map[old_id] = p
            for k, v in sg.items():
                if k != "params":
                    g[k] = v
        self.state.clear()
        for old_id, s in state_dict["state"].items():
            p = id_map[old_id]
            self.state[p] = {
                k: (v.to(p.device) if torch.is_tensor(v) else v)
                for k, v in s.items()}

    def _step_torch(self, p, state, lr, b1, b2, eps, wd):
        """Reference implementation of the quantized-state update (same
        math as the HIP kernel: dequant states -> Adam -> requant)."""
        n = p.numel()
        g = p.grad.reshape(-1).float()
        if wd:
            g = g + wd * p.reshape(-1).float()
        pad = (-n) % self.BLOCK
        if pad:
            g = torch.cat([g, torch.zeros(pad, device=g.device)])

        def deq(q, absmax, signed):
            scale = absmax.repeat_interleave(self.BLOCK)
            denom = 127.0 if signed else 255.0
            return q.float().reshape(-1) / denom * scale

        def q(x, signed):
            blocks = x.view(-1, self.BLOCK)
            absmax = blocks.abs().amax(1).clamp_min(1e-12)
            denom = 127.0 if signed else 255.0
            qv = torch.round(blocks / absmax.unsqueeze(1) * denom)
            if signed:
                qv = qv.clamp(-127, 127).to(torch.int8)
            else:
                qv = qv.clamp(0, 255).to(torch.uint8)
            return qv.view(-1), absmax

        m_q, v_q = state["m_q"], state["v_q"]
        if pad:
            m_q = torch.cat([m_q, torch.zeros(pad, dtype=m_q.dtype, device=m_q.device)])
            v_q = torch.cat([v_q, torch.zeros(pad, dtype=v_q.dtype, device=v_q.device)])
        m = deq(m_q, state["m_absmax"], True)
        v = deq(v_q, state["v_absmax"], False)
        m = b1 * m + (1 - b1) * g
        v = b2 * v + (1 - b2) * g * g
        t = state["step"]
        mhat = m / (1 - b1 ** t)
        vhat = v / (1 - b2 ** t)
        upd = (lr * mhat / (vhat.sqrt() + eps))[:n]
        p.view(-1).sub_(upd.to(p.dtype))
        mq, mam = q(m, True)
        vq, vam = q(v, False)
        state["m_q"].copy_(mq[:n])
        state["v_q"].copy_(vq[:n])
        state["m_absmax"].copy_(mam)
        state["v_absmax"].copy_(vam)


def make_optimizer(params: List[torch.nn.Parameter], lr: float,
                   use_8bit: bool = True) -> torch.optim.Optimizer:
    if use_8bit:
        return Adam8bit(params, lr=lr)
    return torch.optim.Adam(params, lr=lr)
