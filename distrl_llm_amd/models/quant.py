"""nf4 packing utilities + MFMA fragment prepacking for the gfx950 kernels.

The fused nf4 GEMM consumes weights in B-fragment order, layout v3: the
lane's 4 nibble-dwords for one 64-deep K chunk of a wave's 2 n-tiles are
contiguous ([N/32][K/64][64][4] — see ops/csrc/nf4_gemm.hip), so one chunk
streams as one dwordx4 load per lane (plus one dwordx2 of absmax).
Because the base quantizer packs two nibbles per byte K-contiguously,
each fragment dword is exactly 4 consecutive packed bytes, so prepacking
is a pure int32 gather (no bit twiddling).
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
    output of ops.reference.quantize_nf4.

    Layout v3 (vectorized chunk loads): the kernel's unit of streaming is
    one 64-deep K chunk of one wave's 2 n-tiles. All of a lane's data for
    a chunk is contiguous so it loads as 1x dwordx4 (weights) + 1x dwordx2
    (absmax):

      w4f  int32 [N/32][K/64][64 lanes][4]   (4 = ks*2 + nt)
      amaxf fp32 [N/32][K/64][16 lrow][2 nt]
    """
    assert N % 32 == 0 and K % 64 == 0
    device = packed.device
    # pad K to the kernel's widest super-panel (512) — padded columns get
    # absmax 0, so any nibble dequantizes to 0 and contributes nothing
    # (needed for shapes like Qwen2.5-72B's intermediate 29568 = 128*231)
    Kp = (K + 511) // 512 * 512
    if Kp != K:
        pb = packed.contiguous().view(N, K // 2)
        packed = torch.cat([pb, pb.new_zeros(N, (Kp - K) // 2)], 1).reshape(-1)
        am = absmax.contiguous().view(N, K // 64)
        absmax = torch.cat([am, am.new_zeros(N, (Kp - K) // 64)], 1).reshape(-1)
        K = Kp
    ntiles, ksteps = N // 16, K // 32
    ngr, nkb = N // 32, K // 64
    pd = packed.contiguous().view(N, K // 2).view(torch.int32)  # (N, K/8)
    n_map, k_map = _frag_maps(ntiles, ksteps, device)
    w4f1 = pd[n_map, k_map]                       # (ntiles, ksteps, 64)
    v1 = w4f1.view(ngr, 2, nkb, 2, 64)            # (g2, nt, kb, ks, lane)
    w4f = v1.permute(0, 2, 4, 3, 1).contiguous()  # (g2, kb, lane, ks, nt)
    am = absmax.view(N, nkb).view(ngr, 2, 16, nkb)  # (g2, nt, lrow, kb)
    amaxf = am.permute(0, 3, 2, 1).contiguous().float()  # (g2, kb, lrow, nt)
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
