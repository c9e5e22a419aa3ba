#!/usr/bin/env python3
"""Microbenchmark: the split-K MFMA wgrad kernel vs torch (rocBLAS) on the
tall-skinny shapes the bench's SAGE/GAT layers hit.

C[M x N] = A^T @ B, A [K x M] = grad_out, B [K x N] = layer input.
"""
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
    p.add_argument("--shape", default=None,
                   help="K,M,N: profile a single shape")
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()
    shapes = [
        (90_000, 256, 100),   # products L1 wgrad (agg path)
        (90_000, 256, 256),
        (400_000, 256, 100),  # papers-scale frontier
        (13_000, 256, 256),   # L2
        (33_000, 47, 256),    # last layer
    ]
    if args.shape:
        shapes = [tuple(int(x) for x in args.shape.split(","))]
    g = torch.Generator(device="cuda").manual_seed(0)
    print(f"{'K':>8} {'M':>4} {'N':>4} | {'mfma us':>9} {'torch us':>9} "
          f"{'speedup':>7} {'mfma TF':>8} {'rel err':>9}")
    for k, m, n in shapes:
        a = torch.randn(k, m, device="cuda", generator=g)
        b = torch.randn(k, n, device="cuda", generator=g)
        t_q = time_fn(lambda: _ext.wgrad(a, b, True), iters=args.iters)
        t_t = time_fn(lambda: (a.t() @ b, a.sum(0)), iters=args.iters)
        c, bias = _ext.wgrad(a, b, True)
        want = a.double().t() @ b.double()
        rel = float((c.double() - want).norm() / want.norm())
        tf = 2 * k * m * n / t_q / 1e12
        print(f"{k:>8} {m:>4} {n:>4} | {t_q*1e6:>9.1f} {t_t*1e6:>9.1f} "
              f"{t_t/t_q:>7.2f} {tf:>8.1f} {rel:>9.2e}")


if __name__ == "__main__":
    main()
