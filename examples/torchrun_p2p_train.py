#!/usr/bin/env python3
"""Multi-GPU DDP training with the collaboratively-built p2p feature store.

One process per GPU (torchrun), feature store sharded across the xGMI
clique: each rank hipMallocs ONLY its own device's hot shard and reopens
every peer's shard via hipIpc (Feature.from_cpu_tensor_dist) — remote
rows are read one-sided over xGMI inside the gather kernel.

Run:
  torchrun --standalone --nproc-per-node=8 --local-addr 127.0.0.1 \
      examples/torchrun_p2p_train.py

Reference analog: torch-quiver's NVLink-sharded p2p store built by one
owner + cudaIpc reopen (examples/multi_gpu/pyg/ogb-products/
dist_sampling_ogb_products_quiver.py).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn.functional as F

import quiver
from quiver.nn import GraphSAGE


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    dist.init_process_group("nccl")
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    # synthetic graph + features (shape only; swap in your dataset)
    n, dim, classes = 200_000, 128, 32
    g = torch.Generator().manual_seed(0)
    deg = torch.randint(1, 40, (n,), generator=g)
    indptr = torch.zeros(n + 1, dtype=torch.long)
    torch.cumsum(deg, 0, out=indptr[1:])
    indices = torch.randint(0, n, (int(indptr[-1]),), generator=g)
    csr_topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    feat_cpu = torch.randn(n, dim, generator=g)

    sampler = quiver.GraphSageSampler(csr_topo, [15, 10], device=local_rank,
                                      mode="GPU")
    feature = quiver.Feature(local_rank, device_list=list(range(world)),
                             device_cache_size=f"{n // world * dim * 4}",
                             cache_policy="p2p_clique_replicate",
                             csr_topo=csr_topo)

    # world/rank/all_gather default to the torch.distributed group
    feature.from_cpu_tensor_dist(feat_cpu)

    model = GraphSAGE(dim, 128, classes, num_layers=2, dropout=0.0).to(device)
    model = torch.nn.parallel.DistributedDataParallel(
        model, device_ids=[local_rank])
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    y = torch.randint(0, classes, (n,), generator=g).to(device)

    seeds = torch.randperm(n, generator=g)
    per_rank = seeds.split(seeds.numel() // world)[rank]
    for step, batch in enumerate(per_rank.split(1024)[:20]):
        n_id, bs, adjs = sampler.sample(batch)
        x = feature[n_id]
        out = model(x, adjs)
        loss = F.nll_loss(out, y[n_id[:bs]])
        opt.zero_grad(set_to_none=True)
        loss.backward()          # DDP allreduce over RCCL/xGMI
        opt.step()
        if rank == 0 and step % 5 == 0:
            print(f"step {step}: loss {loss.detach().item():.4f}", flush=True)
    dist.barrier()
    if rank == 0:
        print("done")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
