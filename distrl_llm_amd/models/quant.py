"""nf4 packing utilities + MFMA fragment prepacking for the gfx950 kernels.

The fused nf4 GEMM consumes weights in B-fragment order (one dword per
(n-tile, k-step, lane) = the lane's 8 nibbles — see ops/csrc/nf4_gemm.hip).
Because the base quantizer packs two nibbles per byte K-contiguously, the
fragment dword is exactly 4 consecutive packed bytes, so prepacking is a
pure int32 gather (no bit twiddling).
"""

from __future__ import annotations

from typing import Tuple

import torch


def _frag_maps(ntiles: int, ksteps: int, device):
    lane = torch.arange(64, device=device)
    n_map = (torch.arange(ntiles, device=device).view(-1, 1, 1) * 16
             + (lane % 16).view(1, 1, -1))
    k_map = (torch.arange(ksteps, device=device).view(1, -1, 1) * 4
             + (lane // 16).view(1, 1, -1))
    return n_map.expand(ntiles, ksteps, 64), k_map.expand(ntiles, ksteps, 64)


def prepack_nf4_fragments(packed: torch.Tensor, absmax: torch.Tensor,
                          N: int, K: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """packed: uint8 (N*K/2,), absmax: fp32 (N*K/64,) — the row-major (N, K)
    output of ops.reference.quantize_nf4. Returns (w4f int32 flat
    (ntiles*ksteps*64,), amaxf fp32 (ntiles*(K/64)*16,))."""
    assert N % 16 == 0 and K % 64 == 0
    device = packed.device
    ntiles, ksteps = N // 16, K // 32
    pd = packed.contiguous().view(N, K // 2).view(torch.int32)  # (N, K/8)
    n_map, k_map = _frag_maps(ntiles, ksteps, device)
    w4f = pd[n_map, k_map].contiguous()
    am = absmax.view(N, K // 64).view(ntiles, 16, K // 64)
    amaxf = am.permute(0, 2, 1).contiguous().float()
    return w4f.view(-1), amaxf.view(-1)


def prepack_bf16_fragments(w: torch.Tensor) -> torch.Tensor:
    """w: bf16 (N, K) with N % 16 == 0, K % 32 == 0. Returns int32
    (ntiles*ksteps*64*4,) in B-fragment order (16 B per (tile, kstep,
    lane) = the lane's 8 bf16 values)."""
    N, K = w.shape
    assert N % 16 == 0 and K % 32 == 0
    device = w.device
    ntiles, ksteps = N // 16, K // 32
    wv = w.contiguous().view(torch.int32).view(N, K // 2)  # dword = 2 bf16
    lane = torch.arange(64, device=device)
    n_map = (torch.arange(ntiles, device=device).view(-1, 1, 1, 1) * 16
             + (lane % 16).view(1, 1, -1, 1)).expand(ntiles, ksteps, 64, 4)
    d_map = (torch.arange(ksteps, device=device).view(1, -1, 1, 1) * 16
             + (lane // 16).view(1, 1, -1, 1) * 4
             + torch.arange(4, device=device).view(1, 1, 1, -1)
             ).expand(ntiles, ksteps, 64, 4)
    return wv[n_map, d_map].contiguous().view(-1)


def dequant_reference(w4f_src_packed: torch.Tensor, absmax: torch.Tensor,
                      N: int, K: int, dtype=torch.float32) -> torch.Tensor:
    from ..ops import reference as R
    return R.dequantize_nf4(w4f_src_packed, absmax, (N, K), 64, dtype)
