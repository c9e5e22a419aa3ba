#!/usr/bin/env python3
"""Single-GPU GraphSAGE training with UVA sampling + cached feature store
(BASELINE config 2).  Synthetic ogbn-products-shaped graph; swap in real
data by loading your own edge_index/features."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import quiver
from quiver.nn import GraphSAGE


def main(nodes=500_000, edges=10_000_000, dim=100, classes=47, epochs=2):
    import numpy as np
    rng = np.random.default_rng(0)
    deg = np.maximum((rng.pareto(1.3, nodes) * 4).astype(np.int64), 1)
    deg = (deg * (edges / deg.sum())).astype(np.int64) + 1
    indptr = np.zeros(nodes + 1, dtype=np.int64)
    np.cumsum(deg, out=indptr[1:])
    indices = rng.integers(0, nodes, int(indptr[-1]), dtype=np.int64)
    csr_topo = quiver.CSRTopo(indptr=torch.from_numpy(indptr),
                              indices=torch.from_numpy(indices))

    x_cpu = torch.randn(nodes, dim)
    y = torch.randint(0, classes, (nodes,))
    train_idx = torch.arange(nodes // 10)

    sampler = quiver.GraphSageSampler(csr_topo, [15, 10, 5], device=0,
                                      mode="UVA")
    feature = quiver.Feature(0, device_list=[0], device_cache_size="200M",
                             cache_policy="device_replicate",
                             csr_topo=csr_topo)
    feature.from_cpu_tensor(x_cpu)

    model = GraphSAGE(dim, 256, classes, num_layers=3).cuda()
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    loader = torch.utils.data.DataLoader(train_idx, batch_size=1024,
                                         shuffle=True, drop_last=True)
    for epoch in range(epochs):
        t0 = time.perf_counter()
        for seeds in loader:
            n_id, batch_size, adjs = sampler.sample(seeds)
            xb = feature[n_id]
            adjs = [adj.to("cuda:0") for adj in adjs]
            out = model(xb, adjs)
            loss = F.nll_loss(out, y[n_id[:batch_size].cpu()].cuda())
            opt.zero_grad()
            loss.backward()
            opt.step()
        torch.cuda.synchronize()
        print(f"epoch {epoch}: {time.perf_counter()-t0:.2f}s "
              f"loss={float(loss):.4f}")


if __name__ == "__main__":
    main()
