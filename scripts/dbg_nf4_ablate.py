#!/usr/bin/env python3
"""Ablate the nf4_gemm main-loop phases on the gate|up decode shape.

DBG=1 skips the shuffle-LUT dequant (B fragment = amv + (wbits&1)),
DBG=2 skips the x-LDS A-fragment reads (A = 1.0) — each isolates one
phase's cost (guide §5.4: ablate before optimizing). Also runs a single
kernel under rocprofv3-friendly conditions when PROF=1 (one launch per
process step, so --kernel-trace attributes cleanly).
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6


def main():
    from distrl_llm_amd.models.quant import prepack_nf4_fragments
    from distrl_llm_amd.ops import reference as R
    from distrl_llm_amd.ops.build import build
    ext = build()
    dev = torch.device("cuda:0")
    M = int(os.environ.get("ABL_M", "160"))
    N = int(os.environ.get("ABL_N", "37888"))
    K = int(os.environ.get("ABL_K", "3584"))
    ks = os.environ.get("ABL_KSPLIT", "2")
    w = torch.randn(N, K, device=dev) * 0.05
    packed, absmax = R.quantize_nf4(w, 64)
    w4f, amaxf = prepack_nf4_fragments(packed, absmax, N, K)
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    os.environ["DISTRL_NF4_KSPLIT"] = ks

    if os.environ.get("SWEEP2") == "1":
        # geometry sweep: waves-per-block x m-tile x ksplit
        for nw in (4, 8):
            for mtc in (3, 4, 5):
                for ksv in (1, 2, 4, 8):
                    os.environ["DISTRL_NF4_NW"] = str(nw)
                    os.environ["DISTRL_NF4_MTCAP"] = str(mtc)
                    os.environ["DISTRL_NF4_KSPLIT"] = str(ksv)
                    t = bench(lambda: ext.nf4_gemm(x, w4f, amaxf, None, None,
                                                   None, N, K, 0), iters=60)
                    eff = (N * K / 2) / (t * 1e-6) / 1e12
                    print(f"nw={nw} mt={mtc} ks={ksv}: {t:7.1f}us "
                          f"({eff:.2f} TB/s)", flush=True)
        for v in ("DISTRL_NF4_NW", "DISTRL_NF4_MTCAP", "DISTRL_NF4_KSPLIT"):
            os.environ.pop(v, None)
        return

    if os.environ.get("PROF") == "1":
        for _ in range(100):
            ext.nf4_gemm(x, w4f, amaxf, None, None, None, N, K, 0)
        torch.cuda.synchronize()
        return

    for dbg, label in ((0, "full"), (1, "no-dequant"), (2, "no-xlds"),
                       (3, "no-both")):
        if dbg == 3:
            continue  # not instantiated
        os.environ["DISTRL_NF4_DBG"] = str(dbg)
        t = bench(lambda: ext.nf4_gemm(x, w4f, amaxf, None, None, None,
                                       N, K, 0))
        eff = (N * K / 2) / (t * 1e-6) / 1e12
        print(f"M={M} N={N} K={K} ksplit={ks} {label:10s}: {t:7.1f}us "
              f"({eff:.2f} TB/s)", flush=True)
    os.environ.pop("DISTRL_NF4_DBG", None)


if __name__ == "__main__":
    main()
