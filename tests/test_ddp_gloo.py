"""World-size-2 DDP training over gloo on CPU: same structure as bench.py's
distributed path (per-rank sampler + feature + DDP grad allreduce)."""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn.functional as F

import quiver
from quiver.nn import GraphSAGE


def _worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        g = torch.Generator().manual_seed(0)
        n = 300
        src = torch.randint(0, n, (3000,), generator=g)
        dst = torch.randint(0, n, (3000,), generator=g)
        topo = quiver.CSRTopo(torch.stack([src, dst]), node_count=n)
        x = torch.randn(n, 16, generator=g)
        y = torch.randint(0, 4, (n,), generator=g)

        sampler = quiver.GraphSageSampler(topo, [5, 5], mode="CPU")
        torch.manual_seed(0)  # same init on both ranks
        model = GraphSAGE(16, 32, 4, num_layers=2, dropout=0.0)
        model = torch.nn.parallel.DistributedDataParallel(model)
        opt = torch.optim.SGD(model.parameters(), lr=0.05)

        for step in range(3):
            seeds = torch.arange(rank * 50 + step, rank * 50 + step + 32)
            n_id, bs, adjs = sampler.sample(seeds)
            out = model(x[n_id], adjs)
            loss = F.nll_loss(out, y[seeds])
            opt.zero_grad()
            loss.backward()
            opt.step()

        # after DDP allreduce both ranks hold identical parameters
        flat = torch.cat([p.detach().flatten()
                          for p in model.module.parameters()])
        gathered = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        assert torch.allclose(gathered[0], gathered[1], atol=1e-6)
    finally:
        dist.destroy_process_group()


def test_ddp_training_gloo():
    mp.spawn(_worker, args=(2, 29517), nprocs=2, join=True)
