"""One-GPU end-to-end training step + RCCL comm sanity."""
import pytest
import torch
import torch.nn.functional as F

import quiver
from quiver.nn import GraphSAGE, GAT

pytestmark = pytest.mark.gpu


@pytest.fixture
def training_setup():
    g = torch.Generator().manual_seed(0)
    n = 20000
    src = torch.randint(0, n, (400000,), generator=g)
    dst = torch.randint(0, n, (400000,), generator=g)
    topo = quiver.CSRTopo(torch.stack([src, dst]), node_count=n)
    x = torch.randn(n, 64, generator=g)
    y = torch.randint(0, 8, (n,), generator=g)
    return topo, x, y


@pytest.mark.parametrize("mode", ["UVA", "GPU"])
def test_training_step(training_setup, mode):
    topo, x, y = training_setup
    sampler = quiver.GraphSageSampler(topo, [10, 5], device=0, mode=mode)
    feature = quiver.Feature(0, device_list=[0], device_cache_size="2M",
                             cache_policy="device_replicate", csr_topo=topo)
    feature.from_cpu_tensor(x)
    model = GraphSAGE(64, 64, 8, num_layers=2).cuda()
    opt = torch.optim.Adam(model.parameters())
    y = y.cuda()
    losses = []
    for step in range(5):
        seeds = torch.randint(0, topo.node_count, (512,))
        n_id, bs, adjs = sampler.sample(seeds)
        feats = feature[n_id]
        adjs = [adj.to("cuda:0") for adj in adjs]
        out = model(feats, adjs)
        loss = F.nll_loss(out, y[n_id[:bs]])
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert all(torch.isfinite(torch.tensor(losses)))


def test_gat_training_step(training_setup):
    topo, x, y = training_setup
    sampler = quiver.GraphSageSampler(topo, [10, 5], device=0, mode="GPU")
    model = GAT(64, 32, 8, num_layers=2, heads=2).cuda()
    seeds = torch.randint(0, topo.node_count, (256,))
    n_id, bs, adjs = sampler.sample(seeds)
    feats = x[n_id.cpu()].cuda()
    adjs = [adj.to("cuda:0") for adj in adjs]
    out = model(feats, adjs)
    loss = F.nll_loss(out, y[n_id.cpu()][:bs].cuda())
    loss.backward()
    assert torch.isfinite(loss)


def test_rccl_comm_single_rank():
    quiver_id = quiver.getNcclId()
    assert isinstance(quiver_id, bytes) and len(quiver_id) == 128
    comm = quiver.NcclComm(0, 1, quiver_id, hosts=1, rank_per_host=1)
    t = torch.ones(1024, device="cuda")
    comm.allreduce(t)
    torch.cuda.synchronize()
    assert torch.equal(t, torch.ones(1024, device="cuda"))


def test_topo_single_clique():
    topo = quiver.p2pCliqueTopo([0])
    assert topo.p2p_clique_count == 1
    assert topo.get_clique_id(0) == 0


def test_prefetcher_async_chain_correct(training_setup):
    """TrainingPrefetcher's zero-sync chain yields the same feature rows a
    direct (synchronous) gather of the yielded frontier produces."""
    topo, x, y = training_setup
    sampler = quiver.GraphSageSampler(topo, [10, 5], device=0, mode="GPU")
    feature = quiver.Feature(0, device_list=[0], device_cache_size="1M",
                             cache_policy="device_replicate", csr_topo=topo)
    feature.from_cpu_tensor(x)
    batches = [torch.randint(0, topo.node_count, (256,))
               for _ in range(6)]
    seen = 0
    for n_id, bs, adjs, feats in quiver.TrainingPrefetcher(
            sampler, feature, batches, depth=2, device=0):
        assert bs == 256
        assert feats.shape[0] == n_id.shape[0]
        ref = feature[n_id]
        assert torch.equal(feats, ref)
        for adj in adjs:
            assert adj.edge_index[0].max() < adj.size[0]
            assert adj.edge_index[1].max() < adj.size[1]
        seen += 1
    assert seen == len(batches)
