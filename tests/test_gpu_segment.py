"""Fused segment-mean aggregation vs plain fp32 torch reference."""
import pytest
import torch

import quiver
from quiver.nn import _mean_aggregate, GraphSAGE

pytestmark = pytest.mark.gpu


def _torch_ref(x, src, dst, n_dst):
    agg = torch.zeros((n_dst, x.size(1)), dtype=x.dtype, device=x.device)
    agg.index_add_(0, dst, x[src])
    deg = torch.zeros(n_dst, dtype=x.dtype, device=x.device)
    deg.index_add_(0, dst, torch.ones_like(dst, dtype=x.dtype))
    return agg / deg.clamp(min=1).unsqueeze(-1)


@pytest.mark.parametrize("dim", [100, 256, 7, 1])
def test_segment_mean_forward(dim):
    g = torch.Generator().manual_seed(0)
    n_src, n_dst, e = 5000, 700, 9000
    x = torch.randn(n_src, dim, generator=g).cuda()
    dst = torch.sort(torch.randint(0, n_dst, (e,), generator=g)).values.cuda()
    src = torch.randint(0, n_src, (e,), generator=g).cuda()
    out = _mean_aggregate(x, src, dst, n_dst, sorted_dst=True)
    ref = _torch_ref(x, src, dst, n_dst)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def test_segment_mean_backward():
    g = torch.Generator().manual_seed(1)
    n_src, n_dst, e, dim = 800, 200, 3000, 64
    x = torch.randn(n_src, dim, generator=g).cuda().requires_grad_(True)
    x2 = x.detach().clone().requires_grad_(True)
    dst = torch.sort(torch.randint(0, n_dst, (e,), generator=g)).values.cuda()
    src = torch.randint(0, n_src, (e,), generator=g).cuda()
    go = torch.randn(n_dst, dim, generator=g).cuda()

    out = _mean_aggregate(x, src, dst, n_dst, sorted_dst=True)
    out.backward(go)
    ref = _torch_ref(x2, src, dst, n_dst)
    ref.backward(go)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4), \
        (x.grad - x2.grad).abs().max()


def test_sage_model_fused_matches_torch_path(small_graph):
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    sampler = quiver.GraphSageSampler(topo, [8, 4], device=0, mode="GPU")
    seeds = torch.arange(64)
    n_id, bs, adjs = sampler.sample(seeds)
    x = torch.randn(n_id.numel(), 32, device="cuda")
    torch.manual_seed(0)
    m_fused = GraphSAGE(32, 64, 8, num_layers=2, dropout=0.0,
                        sorted_dst=True).cuda()
    torch.manual_seed(0)
    m_plain = GraphSAGE(32, 64, 8, num_layers=2, dropout=0.0,
                        sorted_dst=False).cuda()
    adjs = [a.to("cuda:0") for a in adjs]
    out_f = m_fused(x, adjs)
    out_p = m_plain(x, adjs)
    assert torch.allclose(out_f, out_p, atol=1e-4), \
        (out_f - out_p).abs().max()
