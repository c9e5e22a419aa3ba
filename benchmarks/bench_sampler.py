#!/usr/bin/env python3
"""Sampling throughput (SEPS = sampled edges / second).

Mirrors the reference's headline sampling benchmark
(torch-quiver benchmarks/sample/bench_sampler.py: SEPS counted as
sum(edge counts) / sample time) on synthetic graphs of the named shapes:
  - ogbn-products: 2.45M nodes / 123.7M edges, fanout [15,10,5]
  - reddit:        233k nodes / 114.6M edges, fanout [25,10]
Baselines to beat (other hardware): CPU 1.84M/2.0M, UVA 34.29M/33.15M SEPS.
"""
import argparse
import json
import sys
import time
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import quiver

SHAPES = {
    "products": dict(nodes=2_449_029, edges=123_718_280, fanout=[15, 10, 5],
                     train=196_615),
    "reddit": dict(nodes=232_965, edges=114_615_892, fanout=[25, 10],
                   train=153_431),
}


def make_graph(nodes, edges, seed=0, max_deg=20_000):
    rng = np.random.default_rng(seed)
    raw = rng.pareto(1.3, nodes) + 0.1
    deg = np.maximum((raw * (edges / raw.sum())).astype(np.int64), 1)
    np.clip(deg, 1, max_deg, out=deg)
    deg = -np.sort(-deg)
    indptr = np.zeros(nodes + 1, dtype=np.int64)
    np.cumsum(deg, out=indptr[1:])
    u = rng.random(int(indptr[-1]), dtype=np.float32)
    indices = np.clip((u * u * nodes).astype(np.int64), 0, nodes - 1)
    return torch.from_numpy(indptr), torch.from_numpy(indices)


def bench(mode, shape, batch=1024, iters=50, warmup=5, device=0):
    cfg = SHAPES[shape]
    indptr, indices = make_graph(cfg["nodes"], cfg["edges"])
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    sampler = quiver.GraphSageSampler(topo, cfg["fanout"], device=device,
                                      mode=mode)
    g = torch.Generator().manual_seed(7)
    batches = [torch.randint(0, cfg["train"], (batch,), generator=g)
               for _ in range(warmup + iters)]
    for i in range(warmup):
        sampler.sample(batches[i])
    if mode != "CPU":
        torch.cuda.synchronize()
    edges = 0
    t0 = time.perf_counter()
    for i in range(iters):
        _, _, adjs = sampler.sample(batches[warmup + i])
        edges += sum(adj.edge_index.shape[1] for adj in adjs)
    if mode != "CPU":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return dict(mode=mode, shape=shape, seps=edges / dt,
                ms_per_batch=dt / iters * 1000, batch=batch, iters=iters)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--modes", default="UVA,GPU,CPU")
    p.add_argument("--shapes", default="products,reddit")
    p.add_argument("--batch", type=int, default=1024)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--cpu-iters", type=int, default=5)
    args = p.parse_args()
    for shape in args.shapes.split(","):
        for mode in args.modes.split(","):
            if mode != "CPU" and not torch.cuda.is_available():
                continue
            iters = args.cpu_iters if mode == "CPU" else args.iters
            res = bench(mode, shape, batch=args.batch, iters=iters)
            print(json.dumps(res), flush=True)
