#!/usr/bin/env python3
"""hipBLASLt layout A/B at decode shapes: W row-major (N,K) via F.linear
vs pre-transposed (K,N) via matmul — Tensile kernel selection is
layout-sensitive at skinny M."""

import sys
import time

import torch

sys.path.insert(0, ".")


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
    dev = torch.device("cuda:0")
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 160
    # simulate cold L3: many weight copies cycled (15 GB working set like
    # a real decode step)
    shapes = [("qkv", 4608, 3584), ("o", 3584, 3584),
              ("gateup", 37888, 3584), ("down", 3584, 18944)]
    for name, N, K in shapes:
        n_copies = max(1, int(2e9 // (N * K * 2)))  # ~2 GB of copies
        Ws = [torch.randn(N, K, device=dev, dtype=torch.bfloat16)
              for _ in range(n_copies)]
        Wts = [w.t().contiguous() for w in Ws]
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        i = [0]

        def lin():
            i[0] = (i[0] + 1) % n_copies
            return torch.nn.functional.linear(x, Ws[i[0]])

        def mm():
            i[0] = (i[0] + 1) % n_copies
            return x @ Wts[i[0]]

        t1 = bench(lin)
        t2 = bench(mm)
        wb = N * K * 2 / 1e6
        print(f"{name:7s} M={M} N={N} K={K} ({wb:.0f} MB): "
              f"linear(N,K) {t1:7.1f}us ({wb/t1*1e6/1e12:.2f} TB/s) | "
              f"matmul(K,N) {t2:7.1f}us ({wb/t2*1e6/1e12:.2f} TB/s)",
              flush=True)
        del Ws, Wts


if __name__ == "__main__":
    main()
