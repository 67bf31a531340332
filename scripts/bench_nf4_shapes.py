#!/usr/bin/env python3
"""Per-shape nf4_gemm vs hipBLASLt-bf16 comparison at decode shapes."""

import os
import sys
import time

import torch

sys.path.insert(0, ".")


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6  # us


def main():
    from distrl_llm_amd.models.quant import (prepack_bf16_fragments,
                                             prepack_nf4_fragments)
    from distrl_llm_amd.ops import reference as R
    from distrl_llm_amd.ops.build import build
    ext = build()
    dev = torch.device("cuda:0")
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 160
    shapes = [("qkv", 4608, 3584, 96), ("o", 3584, 3584, 32),
              ("gateup", 37888, 3584, 64), ("down", 3584, 18944, 32)]
    for name, N, K, r in shapes:
        w = torch.randn(N, K, device=dev) * 0.05
        packed, absmax = R.quantize_nf4(w, 64)
        w4f, amaxf = prepack_nf4_fragments(packed, absmax, N, K)
        wb = w.to(torch.bfloat16)
        dequant_w = R.dequantize_nf4(packed, absmax, (N, K), 64, torch.float32)
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        A = (torch.randn(r, K, device=dev) * 0.05).to(torch.bfloat16)
        B = (torch.randn(N, r, device=dev) * 0.05).to(torch.bfloat16)
        afrag = prepack_bf16_fragments(A)
        bfrag = prepack_bf16_fragments(B)
        u = torch.zeros(M, r, device=dev, dtype=torch.float32)

        t_blas = bench(lambda: torch.nn.functional.linear(x, wb))
        t_nf4 = bench(lambda: ext.nf4_gemm(x, w4f, amaxf, None, None, None,
                                           N, K, 0))
        t_u = bench(lambda: ext.lora_u(x, afrag, u, r, 8))
        t_full = bench(lambda: ext.nf4_gemm(x, w4f, amaxf, None, u, bfrag,
                                            N, K, r))
        wbytes_nf4 = N * K // 2
        eff = wbytes_nf4 / (t_nf4 * 1e-6) / 1e12
        print(f"{name:7s} M={M} N={N} K={K}: blas-bf16 {t_blas:7.1f}us | "
              f"nf4 {t_nf4:7.1f}us ({eff:.2f} TB/s wstream) | "
              f"+lora {t_full:7.1f}us | lora_u {t_u:6.1f}us", flush=True)
        if os.environ.get("DISTRL_NF4_SWEEP") == "1":
            # correctness spot-check at current env, then tuning sweep
            y = ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
            ref = (x.float() @ dequant_w.t()).to(torch.bfloat16)
            err = (y.float() - ref.float()).abs().max().item()
            print(f"  maxerr vs fp32-dequant ref: {err:.4f}")
            for ks in (1, 2, 4, 8, 16):
                for nt_flag in ("1", "0"):
                    os.environ["DISTRL_NF4_KSPLIT"] = str(ks)
                    os.environ["DISTRL_NF4_NT"] = nt_flag
                    t = bench(lambda: ext.nf4_gemm(x, w4f, amaxf, None, None,
                                                   None, N, K, 0))
                    e = wbytes_nf4 / (t * 1e-6) / 1e12
                    print(f"  ksplit={ks:2d} nt={nt_flag}: {t:7.1f}us "
                          f"({e:.2f} TB/s)", flush=True)
            os.environ.pop("DISTRL_NF4_KSPLIT", None)
            os.environ.pop("DISTRL_NF4_NT", None)


if __name__ == "__main__":
    main()
