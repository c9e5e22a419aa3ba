#!/usr/bin/env python3
"""Flagship benchmark: GraphSAGE mini-batch training on a synthetic
ogbn-products-shaped graph (BASELINE.json metric: sampled-edges/sec +
GraphSAGE epoch time on MI355X).

One step = sample fanout [15,10,5] batch 1024 -> gather features -> SAGE
forward/backward -> optimizer step (+ DDP/RCCL gradient allreduce for N>1).
Setup mirrors the reference's headline E2E config (torch-quiver
docs/Introduction_en.md:140-149: products, 3-layer SAGE, batch 1024, 20%
device_replicate cache, UVA-sampled graph): graph topology pinned in host
DRAM and zero-copy sampled, hot 20% of features in HBM by degree order,
cold 80% in pinned host DRAM.

Synthetic data (no network for datasets): ogbn-products shape — 2,449,029
nodes, ~123.7M directed edges, 100 fp32 features, 47 classes, 196,615
train seeds.  Out-degrees are Pareto-distributed and node ids are
degree-sorted with skewed destination sampling (u^2), giving a power-law
cache-hit profile comparable to (slightly harder than) the real graph.

Weak scaling: each rank trains batches of 1024 from its own seed stream.
value = whole-job sampled-edges/sec through the FULL training step.
"""
import argparse
import json
import os
import time

import numpy as np
import torch
import torch.distributed as dist
import torch.nn.functional as F

import quiver
from quiver.nn import GAT, GraphSAGE

# graph presets: ogbn-products (default; the reference's headline E2E
# config) and ogbn-papers100M (reference benchmarks/ogbn-papers100M/
# dist_sampling_ogb_paper100M_quiver.py:128,192-197 — same fanout/batch,
# 8G cache, 128 feats; host-DRAM feature spill is the point of the config)
PRESETS = {
    # Both presets default to an HBM-resident graph: the CSR columns
    # (products 1 GB, papers100M 12.5 GB) trivially fit in 288 GB HBM3E —
    # the MI355X-idiomatic placement.  The FEATURE store keeps the named
    # configs' layout (hot cache in HBM, cold tier zero-copy/UVA in host
    # DRAM).  `--mode UVA` additionally keeps the graph in pinned host
    # memory (the reference's small-GPU setup); both are measured in
    # profiles/SUMMARY.md.
    "products": dict(nodes=2_449_029, edges=123_718_280, feat_dim=100,
                     classes=47, train=196_615, cache="196M", mode="GPU"),
    "papers100M": dict(nodes=111_059_956, edges=1_615_685_872, feat_dim=128,
                       classes=172, train=1_207_179, cache="8G", mode="GPU"),
}
BATCH = 1024
FANOUT = [15, 10, 5]
HIDDEN = 256

REF_EPOCH_SECONDS = {1: 11.1, 2: 5.8, 3: 4.7, 4: 3.25}  # products only


def make_graph(seed=0, nodes=None, edges=None, max_deg=20_000):
    """Directly build a power-law CSR: no COO sort needed.

    Degrees are Pareto with the tail clipped at max_deg (ogbn-products'
    real max degree is ~17.5k; an unclipped Pareto(1.3) at 2.45M nodes
    produces million-degree hubs no real dataset has)."""
    rng = np.random.default_rng(seed)
    raw = rng.pareto(1.3, nodes) + 0.1
    deg = np.maximum((raw * (edges / raw.sum() / 1.1)).astype(np.int64), 1)
    np.clip(deg, 1, max_deg, out=deg)
    # degree-sort descending so node id == hotness rank
    deg = -np.sort(-deg)
    scale = edges / deg.sum()
    deg = np.clip((deg * scale).astype(np.int64), 1, max_deg)
    indptr = np.zeros(nodes + 1, dtype=np.int64)
    np.cumsum(deg, out=indptr[1:])
    m = int(indptr[-1])
    # skewed destinations: hot (low-id) nodes are referenced more
    u = rng.random(m, dtype=np.float32)
    indices = (u * u * nodes).astype(np.int64)
    np.clip(indices, 0, nodes - 1, out=indices)
    return torch.from_numpy(indptr), torch.from_numpy(indices), m


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch", type=int, default=BATCH)
    p.add_argument("--mode", default=None, choices=["UVA", "GPU"])
    p.add_argument("--preset", default="products", choices=sorted(PRESETS))
    p.add_argument("--cache", default=None,
                   help="per-GPU HBM feature cache (default: preset's)")
    p.add_argument("--cache-policy", default=None,
                   choices=["device_replicate", "p2p_clique_replicate"],
                   help="default: device_replicate for 1 GPU (the reference's"
                        " headline 1-GPU config); p2p_clique_replicate for"
                        " N>1 (BASELINE config 3: feature store sharded +"
                        " replicated over xGMI — 8 x 20%% cache shards hold"
                        " the whole products feature tensor in HBM)")
    p.add_argument("--nodes", type=int, default=None)
    p.add_argument("--edges", type=int, default=None)
    p.add_argument("--no-overlap", action="store_true",
                   help="disable the sample+gather / compute prefetch overlap")
    p.add_argument("--prefetch-streams", type=int, default=2,
                   help="side streams for the sample+gather pipeline; 2 "
                        "round-robins batches so one batch's sample hides "
                        "under the other's host gather (measured: products "
                        "3.23->3.17, papers100M 2.86->2.20, GAT 6.23->5.89, "
                        "bf16 2.45->2.33 ms/step)")
    p.add_argument("--model", default="sage", choices=["sage", "gat"])
    p.add_argument("--placement", default="degree",
                   choices=["degree", "prob"],
                   help="hot-cache ordering: out-degree (default) or "
                        "multi-hop access probability from sample_prob "
                        "(reference cal_neighbor_prob).  NOTE: the "
                        "cal_next formula assumes a symmetrized graph "
                        "(true for the real OGB datasets); this synthetic "
                        "is directed, so prob placement mispredicts here "
                        "(measured hit rate 0.17 vs 0.39 for degree) — "
                        "degree stays the default")
    p.add_argument("--report-hit-rate", action="store_true",
                   help="measure the hot-cache hit rate over the warmup "
                        "batches (fraction of gathered rows served from "
                        "HBM) and print it before the timed region")
    p.add_argument("--dtype", default="fp32", choices=["fp32", "bf16"],
                   help="feature + model compute dtype.  The headline "
                        "config is fp32 (the reference's); bf16 halves "
                        "feature-store bytes (incl. the PCIe cold tier) "
                        "and is reported as its own honestly-labeled "
                        "variant")
    args = p.parse_args()

    ps = PRESETS[args.preset]
    if args.nodes is None:
        args.nodes = ps["nodes"]
    if args.edges is None:
        args.edges = ps["edges"]
    if args.cache is None:
        args.cache = ps["cache"]
    if args.mode is None:
        args.mode = ps["mode"]
    feat_dim, n_classes, n_train = ps["feat_dim"], ps["classes"], ps["train"]

    # refuse to start if the box cannot hold the host-side working set
    # (graph indices + features + pinned cold tier) — failing cleanly beats
    # driving the box out of memory
    need = args.edges * 8 + 2.2 * args.nodes * feat_dim * 4 + (1 << 32)
    try:
        avail = next(int(l.split()[1]) * 1024
                     for l in open("/proc/meminfo")
                     if l.startswith("MemAvailable"))
    except (OSError, StopIteration):
        avail = None
    if avail is not None and avail < need:
        raise SystemExit(f"host RAM too small for preset {args.preset}: "
                         f"need ~{need/2**30:.0f} GiB, "
                         f"available {avail/2**30:.0f} GiB")

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if args.cache_policy is None:
        args.cache_policy = ("p2p_clique_replicate" if world > 1
                             else "device_replicate")
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    assert world == args.gpus or world == 1, \
        f"WORLD_SIZE={world} but --gpus={args.gpus}"
    distributed = world > 1
    if distributed:
        dist.init_process_group("nccl")
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    t0 = time.perf_counter()
    indptr, indices, m = make_graph(seed=0, nodes=args.nodes,
                                    edges=args.edges)
    csr_topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    g = torch.Generator().manual_seed(0)
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    # chunked in-place fill: torch.randn of a 57 GB papers100M feature
    # tensor would take minutes; uniform_ is one pass
    feat_cpu = torch.empty(args.nodes, feat_dim, dtype=dtype)
    step = max(1, (1 << 28) // max(feat_dim, 1))
    for beg in range(0, args.nodes, step):
        feat_cpu[beg:beg + step].uniform_(-1.0, 1.0, generator=g)

    def build_feature(policy):
        if policy == "p2p_clique_replicate" and distributed:
            # collaborative build: each rank allocates ONLY its own
            # device's hot shard, peers' shards are reopened via hipIpc
            # (one hipMalloc per shard in the whole job, not per rank)
            f = quiver.Feature(local_rank, device_list=list(range(world)),
                               device_cache_size=args.cache,
                               cache_policy=policy, csr_topo=csr_topo)

            def all_gather(obj):
                objs = [None] * world
                dist.all_gather_object(objs, obj)
                return objs

            f.from_cpu_tensor_dist(feat_cpu, world, rank, all_gather,
                                   score=placement_score)
            return f
        if policy == "p2p_clique_replicate":
            quiver.init_p2p(list(range(world)))
            devices = list(range(world))
        else:
            devices = [local_rank]
        f = quiver.Feature(local_rank, device_list=devices,
                           device_cache_size=args.cache,
                           cache_policy=policy, csr_topo=csr_topo)
        f.from_cpu_tensor(feat_cpu, score=placement_score)
        return f

    sampler = quiver.GraphSageSampler(csr_topo, FANOUT, device=local_rank,
                                      mode=args.mode)
    placement_score = None
    if args.placement == "prob":
        # multi-hop access probability from the training seed distribution
        # (uniform over train_idx) drives the hot/cold ordering
        tg0 = torch.Generator().manual_seed(42)
        ti = torch.randint(0, args.nodes, (n_train,), generator=tg0)
        placement_score = sampler.sample_prob(ti.to(device), args.nodes).cpu()
    try:
        feature = build_feature(args.cache_policy)
    except Exception as e:  # noqa: BLE001 — the sharded layout must not
        # zero the run; a rank falling back alone stays functionally
        # correct (its device_replicate cache serves its own gathers, and
        # Feature keeps its exported shard alive for peers)
        if args.cache_policy != "p2p_clique_replicate":
            raise
        print(f"# rank {rank}: p2p_clique_replicate failed "
              f"({type(e).__name__}: {e}); falling back to "
              "device_replicate", flush=True)
        args.cache_policy = "device_replicate"
        feature = build_feature(args.cache_policy)

    if args.model == "gat":
        model = GAT(feat_dim, HIDDEN // 4, n_classes,
                    num_layers=len(FANOUT), heads=4, dropout=0.0).to(device)
    else:
        model = GraphSAGE(feat_dim, HIDDEN, n_classes,
                          num_layers=len(FANOUT), dropout=0.0).to(device)
    if args.dtype == "bf16":
        model = model.to(torch.bfloat16)
    if distributed:
        model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank])
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    # labels resident on GPU: indexing them with n_id stays device-side
    y = torch.randint(0, n_classes, (args.nodes,), generator=g).to(device)

    # per-rank seed stream: train nodes are a fixed uniform draw over ALL
    # nodes (real train_idx is not degree-biased; hot-head-only seeds would
    # overstate cache hit rate)
    tg = torch.Generator().manual_seed(42)
    train_idx = torch.randint(0, args.nodes, (n_train,), generator=tg)
    sg = torch.Generator().manual_seed(1234 + rank)
    n_batches = args.warmup + args.steps
    batches = [train_idx[torch.randint(0, n_train, (args.batch,),
                                       generator=sg)]
               for _ in range(n_batches)]
    if rank == 0:
        print(f"# setup done in {time.perf_counter()-t0:.1f}s "
              f"(edges={m/1e6:.1f}M)", flush=True)

    def train_on(n_id, bs, adjs, x):
        out = model(x, adjs)
        loss = F.nll_loss(out, y[n_id[:bs]])
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        return sum(adj.edge_index.shape[1] for adj in adjs)

    step_times = [] if os.environ.get("QUIVER_BENCH_STEP_TIMES") else None

    # ONE prefetcher spanning warmup + timed steps: a fresh prefetcher at
    # the timing boundary would create a new side stream (fresh caching-
    # allocator pool -> hundreds of MB re-allocated) and refill the
    # pipeline inside the timed window — measured as a ~10 ms first timed
    # step.  A continuous pipeline is also how a real epoch runs; the
    # depth batches produced ahead of t0 cancel against the depth
    # produced-but-unconsumed at the end.
    pf_iter = None
    if not args.no_overlap:
        pf = quiver.TrainingPrefetcher(sampler, feature, batches, depth=2,
                                       device=local_rank,
                                       num_streams=args.prefetch_streams)
        pf_iter = iter(pf)

    def run_range(lo, hi):
        total = 0
        t_prev = time.perf_counter()
        for i in range(lo, hi):
            if pf_iter is None:
                n_id, bs, adjs = sampler.sample(batches[i])
                x = feature[n_id]
            else:
                n_id, bs, adjs, x = next(pf_iter)
            total += train_on(n_id, bs, adjs, x)
            if step_times is not None:
                torch.cuda.synchronize()
                now = time.perf_counter()
                step_times.append((now - t_prev) * 1000)
                t_prev = now
        return total

    run_range(0, args.warmup)
    if distributed:
        dist.barrier()
    torch.cuda.synchronize()

    t_start = time.perf_counter()
    edges_done = run_range(args.warmup, args.warmup + args.steps)
    if distributed:
        dist.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t_start

    # MAX elapsed over ranks; SUM of edges over ranks
    if distributed:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t)
        e = torch.tensor([edges_done], device=device, dtype=torch.float64)
        dist.all_reduce(e, op=dist.ReduceOp.SUM)
        edges_done = int(e)

    if step_times is not None and rank == 0:
        print("# step ms:", " ".join(f"{t:.2f}" for t in step_times),
              flush=True)
    if rank == 0:
        ms_per_step = elapsed / args.steps * 1000
        steps_per_epoch = (n_train + args.batch * world - 1) // (args.batch *
                                                                 world)
        epoch_seconds = ms_per_step / 1000 * steps_per_epoch
        ref = (REF_EPOCH_SECONDS.get(world)
               if args.preset == "products" and args.dtype == "fp32"
               else None)  # the reference's published numbers are fp32
        vs_baseline = (ref / epoch_seconds) if ref else None
        print(json.dumps({
            "metric": "train-sampled-edges/sec (GraphSAGE e2e step, "
                      f"ogbn-{args.preset}-shaped synthetic)",
            "value": edges_done / elapsed,
            "unit": "edges/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": vs_baseline,
            "dtype": args.dtype,
            "data": f"synthetic (ogbn-{args.preset} shape: "
                    f"{args.nodes/1e6:.2f}M nodes, {m/1e6:.1f}M edges, "
                    f"{feat_dim} feats, power-law degrees), random-init "
                    "3-layer SAGE",
            "config": {
                "model": (f"graphsage-3L-h{HIDDEN}" if args.model == "sage"
                          else f"gat-3L-h{HIDDEN // 4}x4h")
                         + f" ogbn-{args.preset}",
                "global_batch": args.batch * world,
                "fanout": FANOUT,
                "parallelism": f"dp{world}",
                "sample_mode": args.mode,
                "cache": args.cache,
                "cache_policy": args.cache_policy,
                "placement": args.placement,
                "overlap": not args.no_overlap,
                "epoch_seconds_derived": epoch_seconds,
                "ref_epoch_seconds": ref,
            },
        }), flush=True)

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
