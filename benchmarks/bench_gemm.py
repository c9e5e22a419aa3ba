#!/usr/bin/env python3
"""Microbenchmark: the tall-M MFMA GEMM vs torch (rocBLAS) on the model
layer shapes (forward + data-grad of big-frontier linears)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import quiver  # noqa: F401,E402
from quiver import _ext  # noqa: E402


def time_fn(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    import argparse
    p = argparse.ArgumentParser()
    p.add_argument("--shape", default=None, help="M,K,N single shape")
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()
    shapes = [
        (1_060_000, 100, 256),  # GAT layer-1 projection (products frontier)
        (1_060_000, 256, 100),  # its data-grad
        (90_000, 100, 256),     # SAGE layer-1 post-aggregation
        (90_000, 256, 256),
        (13_000, 256, 47),      # last layer
    ]
    if args.shape:
        shapes = [tuple(int(x) for x in args.shape.split(","))]
    g = torch.Generator(device="cuda").manual_seed(0)
    print(f"{'M':>9} {'K':>4} {'N':>4} | {'mfma us':>8} {'torch us':>9} "
          f"{'speedup':>7} {'GB/s':>7} {'rel err':>9}")
    for m, k, n in shapes:
        a = torch.randn(m, k, device="cuda", generator=g)
        w = torch.randn(n, k, device="cuda", generator=g)  # torch layout
        bias = torch.randn(n, device="cuda", generator=g)
        wt = w.t().contiguous()
        t_q = time_fn(lambda: _ext.tall_gemm(a, wt, bias), iters=args.iters)
        t_t = time_fn(lambda: torch.nn.functional.linear(a, w, bias), iters=args.iters)
        c = _ext.tall_gemm(a, wt, bias)
        want = torch.nn.functional.linear(a, w, bias)
        rel = float((c - want).norm() / want.norm())
        gbs = (m * k + m * n + k * n) * 4 / t_q / 1e9
        print(f"{m:>9} {k:>4} {n:>4} | {t_q*1e6:>8.1f} {t_t*1e6:>9.1f} "
              f"{t_t/t_q:>7.2f} {gbs:>7.0f} {rel:>9.2e}")


if __name__ == "__main__":
    main()
