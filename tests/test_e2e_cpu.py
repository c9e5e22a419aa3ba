"""CPU end-to-end: CSRTopo -> CPU sampler -> SAGE training loop converges
on a synthetic community graph (config 1 of BASELINE.json)."""
import torch
import torch.nn.functional as F

import quiver
from quiver.nn import GraphSAGE


def _community_graph(n=200, dim=8, seed=0):
    g = torch.Generator().manual_seed(seed)
    # two communities, dense intra-community edges
    src, dst = [], []
    for c in range(2):
        base = c * (n // 2)
        for _ in range(n * 4):
            a = int(torch.randint(0, n // 2, (1,), generator=g)) + base
            b = int(torch.randint(0, n // 2, (1,), generator=g)) + base
            src.append(a)
            dst.append(b)
    edge_index = torch.tensor([src, dst])
    y = (torch.arange(n) >= n // 2).long()
    x = torch.randn(n, dim, generator=g) + y.float().unsqueeze(1) * 0.5
    return edge_index, x, y


def test_cpu_training_learns():
    torch.manual_seed(0)
    edge_index, x, y = _community_graph()
    topo = quiver.CSRTopo(edge_index, node_count=200)
    sampler = quiver.GraphSageSampler(topo, [5, 5], mode="CPU")
    model = GraphSAGE(8, 16, 2, num_layers=2, dropout=0.0)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    train_idx = torch.arange(200)
    first_loss = None
    for epoch in range(6):
        perm = torch.randperm(200)
        total = 0.0
        for beg in range(0, 200, 64):
            seeds = train_idx[perm[beg:beg + 64]]
            n_id, bs, adjs = sampler.sample(seeds)
            out = model(x[n_id], adjs)
            loss = F.nll_loss(out, y[seeds])
            opt.zero_grad()
            loss.backward()
            opt.step()
            total += float(loss.detach())
        if first_loss is None:
            first_loss = total
    assert total < first_loss * 0.8, (first_loss, total)


def test_mixed_sampler_iterates():
    edge_index, _, _ = _community_graph()
    topo = quiver.CSRTopo(edge_index, node_count=200)

    class Job(quiver.SampleJob):
        def __init__(self):
            self.data = [torch.arange(i * 10, i * 10 + 10) for i in range(8)]

        def __getitem__(self, i):
            return self.data[i]

        def __len__(self):
            return len(self.data)

        def shuffle(self):
            pass

    # GPU-free host: exercise the job/iterator logic with the CPU path via
    # UVA_ONLY is impossible, so use the mixed sampler's CPU workers only
    # when a GPU exists; here just check the job protocol wiring.
    job = Job()
    assert len(job) == 8
    assert job[0].numel() == 10
