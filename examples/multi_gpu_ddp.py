#!/usr/bin/env python3
"""Multi-GPU DDP GraphSAGE training (BASELINE config 3).

Parent builds CSRTopo + sampler + Feature once; mp.spawn children receive
them through the registered ForkingPickler reducers (sampler rebuilds from
the shared CSRTopo, Feature reopens hipIpc handles), then train DDP with
RCCL gradient allreduce over xGMI.

Run: python examples/multi_gpu_ddp.py [world_size]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn.functional as F

import quiver
from quiver.nn import GraphSAGE


def run(rank, world, sampler, feature, y, train_idx, dim, classes):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29501")
    torch.cuda.set_device(rank)
    dist.init_process_group("nccl", rank=rank, world_size=world)

    model = GraphSAGE(dim, 256, classes, num_layers=3).cuda()
    model = torch.nn.parallel.DistributedDataParallel(model,
                                                      device_ids=[rank])
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    local_train = train_idx.split(train_idx.size(0) // world)[rank]
    loader = torch.utils.data.DataLoader(local_train, batch_size=1024,
                                         shuffle=True, drop_last=True)
    for epoch in range(2):
        t0 = time.perf_counter()
        for seeds in loader:
            n_id, batch_size, adjs = sampler.sample(seeds)
            xb = feature[n_id]
            adjs = [adj.to(rank) for adj in adjs]
            out = model(xb, adjs)
            loss = F.nll_loss(out, y[n_id[:batch_size].cpu()].cuda())
            opt.zero_grad()
            loss.backward()
            opt.step()
        dist.barrier()
        torch.cuda.synchronize()
        if rank == 0:
            print(f"epoch {epoch}: {time.perf_counter()-t0:.2f}s")
    dist.destroy_process_group()


def main(world=2, nodes=500_000, edges=10_000_000, dim=100, classes=47):
    import numpy as np
    avail = torch.cuda.device_count()
    if world > avail:
        print(f"# only {avail} GPU(s) visible: clamping world {world} -> "
              f"{avail}")
        world = max(1, avail)
    rng = np.random.default_rng(0)
    deg = np.maximum((rng.pareto(1.3, nodes) * 4).astype(np.int64), 1)
    indptr = np.zeros(nodes + 1, dtype=np.int64)
    np.cumsum(deg, out=indptr[1:])
    indices = rng.integers(0, nodes, int(indptr[-1]), dtype=np.int64)
    csr_topo = quiver.CSRTopo(indptr=torch.from_numpy(indptr),
                              indices=torch.from_numpy(indices))
    csr_topo.share_memory_()

    x_cpu = torch.randn(nodes, dim)
    y = torch.randint(0, classes, (nodes,))
    train_idx = torch.arange(nodes // 10)

    quiver.init_p2p(list(range(world)))
    sampler = quiver.GraphSageSampler(csr_topo, [15, 10, 5], device=0,
                                      mode="UVA")
    feature = quiver.Feature(0, device_list=list(range(world)),
                             device_cache_size="100M",
                             cache_policy="p2p_clique_replicate",
                             csr_topo=csr_topo)
    feature.from_cpu_tensor(x_cpu)

    mp.spawn(run, args=(world, sampler, feature, y, train_idx, dim, classes),
             nprocs=world, join=True)


if __name__ == "__main__":
    main(world=int(sys.argv[1]) if len(sys.argv) > 1 else 2)
