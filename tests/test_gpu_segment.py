"""Fused segment-mean aggregation vs plain fp32 torch reference."""
import pytest
import torch

import quiver
from quiver import _ext
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


def test_segment_wsum_matches_torch():
    """Weighted segment sum (GAT aggregation) fwd + both backwards vs the
    plain torch reference."""
    from quiver import _ext
    g = torch.Generator().manual_seed(0)
    n_src, n_dst, H, C = 400, 90, 4, 16
    deg = torch.randint(0, 8, (n_dst,), generator=g)
    dst_ptr = torch.zeros(n_dst + 1, dtype=torch.long)
    dst_ptr[1:] = deg.cumsum(0)
    E = int(dst_ptr[-1])
    src = torch.randint(0, n_src, (E,), generator=g)
    dst = torch.repeat_interleave(torch.arange(n_dst), deg)

    x = torch.randn(n_src, H * C, generator=g, device="cpu")
    w = torch.randn(E, H, generator=g, device="cpu")
    xg = x.cuda().requires_grad_(True)
    wg = w.cuda().requires_grad_(True)
    out = _ext.segment_wsum(xg, wg, src.cuda(), dst_ptr.cuda(), H)

    x2 = x.cuda().requires_grad_(True)
    w2 = w.cuda().requires_grad_(True)
    msg = x2.view(n_src, H, C)[src.cuda()] * w2.unsqueeze(-1)
    want = torch.zeros(n_dst, H, C, device="cuda")
    want.index_add_(0, dst.cuda(), msg)
    want = want.reshape(n_dst, H * C)
    assert torch.allclose(out, want, atol=1e-4), \
        (out - want).abs().max().item()

    go = torch.randn(n_dst, H * C, generator=g).cuda()
    gx, gw = _ext.segment_wsum_backward(go, xg.detach(), wg.detach(),
                                        src.cuda(), dst_ptr.cuda(), H,
                                        True, True)
    want.backward(go)  # populates x2.grad / w2.grad through the torch chain
    assert torch.allclose(gx, x2.grad, atol=1e-3), \
        (gx - x2.grad).abs().max().item()
    assert torch.allclose(gw, w2.grad, atol=1e-3), \
        (gw - w2.grad).abs().max().item()


def test_gat_fused_autograd_matches_fallback():
    """GATConv with sorted_dst fused kernel == the index_add fallback."""
    from quiver.nn import GATConv
    g = torch.Generator().manual_seed(1)
    n_src, n_dst, E = 200, 60, 800
    src = torch.randint(0, n_src, (E,), generator=g)
    dst = torch.sort(torch.randint(0, n_dst, (E,), generator=g))[0]
    edge_index = torch.stack([src, dst]).cuda()
    x_src = torch.randn(n_src, 32, generator=g).cuda()
    x_dst = x_src[:n_dst]

    torch.manual_seed(7)
    conv_f = GATConv(32, 16, heads=2, sorted_dst=True).cuda()
    torch.manual_seed(7)
    conv_r = GATConv(32, 16, heads=2, sorted_dst=False).cuda()

    xf = x_src.clone().requires_grad_(True)
    xr = x_src.clone().requires_grad_(True)
    of = conv_f((xf, xf[:n_dst]), edge_index, (n_src, n_dst))
    orr = conv_r((xr, xr[:n_dst]), edge_index, (n_src, n_dst))
    assert torch.allclose(of, orr, atol=1e-4)
    of.pow(2).mean().backward()
    orr.pow(2).mean().backward()
    assert torch.allclose(xf.grad, xr.grad, atol=1e-4)
    for pf, pr in zip(conv_f.parameters(), conv_r.parameters()):
        assert torch.allclose(pf.grad, pr.grad, atol=1e-3)


def test_segment_softmax_matches_torch():
    from quiver import _ext
    g = torch.Generator().manual_seed(2)
    n_dst, H = 70, 4
    deg = torch.randint(0, 9, (n_dst,), generator=g)
    dst_ptr = torch.zeros(n_dst + 1, dtype=torch.long)
    dst_ptr[1:] = deg.cumsum(0)
    E = int(dst_ptr[-1])
    dst = torch.repeat_interleave(torch.arange(n_dst), deg)
    a = torch.randn(E, H, generator=g)

    ac = a.cuda().requires_grad_(True)
    out = _ext.segment_softmax(ac, dst_ptr.cuda(), H)

    # torch reference
    a2 = a.cuda().requires_grad_(True)
    amax = torch.full((n_dst, H), float("-inf"), device="cuda")
    amax = amax.scatter_reduce(0, dst.cuda().unsqueeze(-1).expand(E, H), a2,
                               reduce="amax", include_self=True)
    ex = (a2 - amax[dst.cuda()]).exp()
    den = torch.zeros(n_dst, H, device="cuda")
    den.index_add_(0, dst.cuda(), ex)
    want = ex / den[dst.cuda()].clamp(min=1e-16)
    assert torch.allclose(out, want, atol=1e-5), \
        (out - want).abs().max().item()

    go = torch.randn(E, H, generator=g).cuda()
    ga = _ext.segment_softmax_backward(go, out, dst_ptr.cuda(), H)
    want.backward(go)
    assert torch.allclose(ga, a2.grad, atol=1e-4), \
        (ga - a2.grad).abs().max().item()


def test_segment_mean_bf16_matches_fp32():
    """bf16 segment mean: fp32 accumulate inside, only stores round —
    compare against the plain fp32 torch reference with bf16 tolerance."""
    torch.manual_seed(0)
    n_src, n_dst, dim = 5000, 700, 100
    x32 = torch.randn(n_src, dim, device="cuda")
    x16 = x32.to(torch.bfloat16)
    deg = torch.randint(1, 30, (n_dst,), device="cuda")
    dst_ptr = torch.zeros(n_dst + 1, dtype=torch.long, device="cuda")
    torch.cumsum(deg, 0, out=dst_ptr[1:])
    E = int(dst_ptr[-1])
    src = torch.randint(0, n_src, (E,), device="cuda")
    out16 = _ext.segment_mean_gather(x16, src, dst_ptr)
    assert out16.dtype == torch.bfloat16
    # fp32 reference
    ref = torch.zeros(n_dst, dim, device="cuda")
    dst = torch.repeat_interleave(torch.arange(n_dst, device="cuda"), deg)
    ref.index_add_(0, dst, x16.float()[src])
    ref = ref / deg.clamp(min=1).unsqueeze(-1).float()
    assert torch.allclose(out16.float(), ref, atol=3e-2, rtol=3e-2)

    # backward: packed-bf16 atomics vs fp32 scatter reference
    g16 = torch.randn(n_dst, dim, device="cuda").to(torch.bfloat16)
    gx = _ext.segment_mean_gather_backward(g16, src, dst_ptr, n_src)
    assert gx.dtype == torch.bfloat16
    ref_gx = torch.zeros(n_src, dim, device="cuda")
    contrib = (g16.float() / deg.clamp(min=1).unsqueeze(-1).float())[dst]
    ref_gx.index_add_(0, src, contrib)
    assert torch.allclose(gx.float(), ref_gx, atol=6e-2, rtol=6e-2)


def test_sage_bf16_e2e_step():
    """One fwd+bwd of bf16 GraphSAGE through the fused kernels."""
    from quiver.nn import GraphSAGE
    import quiver
    g = torch.Generator().manual_seed(0)
    n = 3000
    deg = torch.randint(1, 20, (n,), generator=g)
    indptr = torch.zeros(n + 1, dtype=torch.long)
    torch.cumsum(deg, 0, out=indptr[1:])
    indices = torch.randint(0, n, (int(indptr[-1]),), generator=g)
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s = quiver.GraphSageSampler(topo, [8, 4], device=0, mode="GPU")
    n_id, bs, adjs = s.sample(torch.arange(64))
    x = torch.randn(n_id.numel(), 100, generator=g) \
        .to(torch.bfloat16).cuda().requires_grad_(True)
    model = GraphSAGE(100, 64, 10, num_layers=2, dropout=0.0) \
        .to(torch.bfloat16).cuda()
    out = model(x, adjs)
    assert out.dtype == torch.bfloat16
    loss = torch.nn.functional.nll_loss(
        out.float(), torch.randint(0, 10, (bs,), device="cuda"))
    loss.backward()
    assert x.grad is not None and torch.isfinite(x.grad.float()).all()
    for p in model.parameters():
        assert p.grad is None or torch.isfinite(p.grad.float()).all()


def test_gat_dots_matches_torch():
    """Fused attention dots (fwd+bwd) vs the plain torch chain."""
    from quiver.nn import _GatDots
    torch.manual_seed(0)
    N, H, C, n_dst = 5000, 4, 64, 700
    h = torch.randn(N, H * C, device="cuda", requires_grad=True)
    a_s = torch.randn(H * C, device="cuda", requires_grad=True)
    a_d = torch.randn(H * C, device="cuda", requires_grad=True)
    h2 = h.detach().clone().requires_grad_(True)
    a_s2 = a_s.detach().clone().requires_grad_(True)
    a_d2 = a_d.detach().clone().requires_grad_(True)

    asrc, adst = _GatDots.apply(h, a_s, a_d, n_dst, H)
    hv = h2.view(N, H, C)
    ref_src = (hv * a_s2.view(1, H, C)).sum(-1)
    ref_dst = (hv[:n_dst] * a_d2.view(1, H, C)).sum(-1)
    assert torch.allclose(asrc, ref_src, atol=1e-4, rtol=1e-4)
    assert torch.allclose(adst, ref_dst, atol=1e-4, rtol=1e-4)

    gs = torch.randn_like(asrc)
    gd = torch.randn_like(adst)
    (asrc * gs).sum().backward(retain_graph=True)
    (adst * gd).sum().backward()
    (ref_src * gs).sum().backward(retain_graph=True)
    (ref_dst * gd).sum().backward()
    assert torch.allclose(h.grad, h2.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(a_s.grad, a_s2.grad, atol=2e-3, rtol=1e-3)
    assert torch.allclose(a_d.grad, a_d2.grad, atol=2e-3, rtol=1e-3)


def test_gat_bf16_e2e_step():
    """bf16 GAT: projection/gather bf16, attention via the fp32 fused
    kernels behind a boundary cast — one fwd+bwd step must be finite and
    outputs bf16."""
    from quiver.nn import GAT
    import quiver
    g = torch.Generator().manual_seed(0)
    n = 3000
    deg = torch.randint(1, 20, (n,), generator=g)
    indptr = torch.zeros(n + 1, dtype=torch.long)
    torch.cumsum(deg, 0, out=indptr[1:])
    indices = torch.randint(0, n, (int(indptr[-1]),), generator=g)
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s = quiver.GraphSageSampler(topo, [8, 4], device=0, mode="GPU")
    n_id, bs, adjs = s.sample(torch.arange(64))
    x = torch.randn(n_id.numel(), 100, generator=g) \
        .to(torch.bfloat16).cuda().requires_grad_(True)
    model = GAT(100, 64, 10, num_layers=2, heads=4, dropout=0.0) \
        .to(torch.bfloat16).cuda()
    out = model(x, adjs)
    assert out.dtype == torch.bfloat16
    loss = torch.nn.functional.nll_loss(
        out.float(), torch.randint(0, 10, (bs,), device="cuda"))
    loss.backward()
    assert torch.isfinite(x.grad.float()).all()
    for name, p in model.named_parameters():
        if p.grad is not None:
            assert p.grad.dtype == p.dtype, name
            assert torch.isfinite(p.grad.float()).all(), name


def test_segment_wsum_bf16_matches_fp32():
    """bf16 weighted segment sum (x bf16, weights fp32): fwd + both
    backwards vs the fp32 reference."""
    torch.manual_seed(0)
    n_src, n_dst, H, C = 4000, 500, 4, 64
    x32 = torch.randn(n_src, H * C, device="cuda")
    x16 = x32.to(torch.bfloat16)
    deg = torch.randint(1, 20, (n_dst,), device="cuda")
    dst_ptr = torch.zeros(n_dst + 1, dtype=torch.long, device="cuda")
    torch.cumsum(deg, 0, out=dst_ptr[1:])
    E = int(dst_ptr[-1])
    src = torch.randint(0, n_src, (E,), device="cuda")
    w = torch.rand(E, H, device="cuda")
    out16 = _ext.segment_wsum(x16, w, src, dst_ptr, H)
    assert out16.dtype == torch.bfloat16
    out32 = _ext.segment_wsum(x16.float(), w, src, dst_ptr, H)
    assert torch.allclose(out16.float(), out32, atol=5e-2, rtol=5e-2)

    g16 = torch.randn(n_dst, H * C, device="cuda").to(torch.bfloat16)
    gx16, gw16 = _ext.segment_wsum_backward(g16, x16, w, src, dst_ptr, H,
                                            True, True)
    gx32, gw32 = _ext.segment_wsum_backward(g16.float(), x16.float(), w,
                                            src, dst_ptr, H, True, True)
    assert gx16.dtype == torch.bfloat16 and gw16.dtype == torch.float32
    assert torch.allclose(gx16.float(), gx32, atol=8e-2, rtol=8e-2)
    assert torch.allclose(gw16, gw32, atol=3e-2, rtol=3e-2)


def test_gat_dots_bf16_matches_fp32():
    from quiver.nn import _GatDots
    torch.manual_seed(1)
    N, H, C, n_dst = 4000, 4, 64, 500
    h32 = torch.randn(N, H * C, device="cuda", requires_grad=True)
    h16 = h32.detach().to(torch.bfloat16).requires_grad_(True)
    a_s = torch.randn(H * C, device="cuda", requires_grad=True)
    a_d = torch.randn(H * C, device="cuda", requires_grad=True)
    s16, d16 = _GatDots.apply(h16, a_s, a_d, n_dst, H)
    assert s16.dtype == torch.float32  # logits stay fp32
    s32, d32 = _GatDots.apply(h16.detach().float().requires_grad_(False),
                              a_s.detach(), a_d.detach(), n_dst, H)
    assert torch.allclose(s16, s32, atol=1e-3, rtol=1e-3)
    assert torch.allclose(d16, d32, atol=1e-3, rtol=1e-3)
    (s16.square().sum() + d16.square().sum()).backward()
    assert h16.grad.dtype == torch.bfloat16
    assert torch.isfinite(h16.grad.float()).all()
    assert a_s.grad is not None and torch.isfinite(a_s.grad).all()
