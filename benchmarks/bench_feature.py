#!/usr/bin/env python3
"""Feature-collection throughput (GB/s of gathered feature bytes).

Mirrors the reference's feature benchmark (torch-quiver
benchmarks/feature/bench_feature.py: gathered bytes / gather time) on
synthetic features of the named shapes with a degree-skewed access stream.
Baselines to beat (other hardware): products 1-GPU 20% cache 14.82 GB/s;
2-GPU NVLink p2p clique 108.6 GB/s.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import quiver

SHAPES = {
    "products": dict(nodes=2_449_029, dim=100),
    "reddit": dict(nodes=232_965, dim=602),
    "papers100M": dict(nodes=10_000_000, dim=128),  # sliced for one box
}


def skewed_batches(nodes, batch, iters, seed=3, alpha=2.0):
    """Access stream where hot (low) ids dominate, like post-reorder reality."""
    rng = np.random.default_rng(seed)
    u = rng.random((iters, batch))
    ids = np.minimum((u ** alpha * nodes).astype(np.int64), nodes - 1)
    return [torch.from_numpy(row) for row in ids]


def bench(shape, cache, policy, device_list, batch=80_000, iters=50,
          warmup=5, rank=0, sorted_ids=False):
    cfg = SHAPES[shape]
    g = torch.Generator().manual_seed(0)
    feat = torch.randn(cfg["nodes"], cfg["dim"], generator=g)
    feature = quiver.Feature(rank, device_list=device_list,
                             device_cache_size=cache, cache_policy=policy)
    feature.from_cpu_tensor(feat)
    row_bytes = cfg["dim"] * 4
    batches = [b.cuda() for b in skewed_batches(cfg["nodes"], batch,
                                                warmup + iters)]
    if sorted_ids:
        batches = [torch.sort(b)[0] for b in batches]
    for i in range(warmup):
        feature[batches[i]]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    total = 0
    for i in range(iters):
        out = feature[batches[warmup + i]]
        total += out.numel() * out.element_size()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return dict(shape=shape, cache=cache, policy=policy,
                gbps=total / dt / 1e9, batch=batch, iters=iters,
                row_bytes=row_bytes, sorted_ids=sorted_ids)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--shape", default="products")
    p.add_argument("--cache", default="20%")
    p.add_argument("--policy", default="device_replicate")
    p.add_argument("--devices", default="0")
    p.add_argument("--batch", type=int, default=80_000)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--sorted", action="store_true")
    args = p.parse_args()
    cfg = SHAPES[args.shape]
    cache = args.cache
    if cache.endswith("%"):
        frac = float(cache[:-1]) / 100
        cache = int(cfg["nodes"] * cfg["dim"] * 4 * frac)
    devices = [int(d) for d in args.devices.split(",")]
    if args.policy == "p2p_clique_replicate":
        quiver.init_p2p(devices)
    res = bench(args.shape, cache, args.policy, devices, batch=args.batch,
                iters=args.iters, sorted_ids=args.sorted)
    print(json.dumps(res), flush=True)
