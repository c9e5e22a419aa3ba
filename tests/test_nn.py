import torch

from quiver.nn import SAGEConv, GATConv, GraphSAGE, GAT
from quiver.pyg.sage_sampler import Adj


def test_sage_conv_matches_dense_mean():
    torch.manual_seed(0)
    conv = SAGEConv(4, 3)
    x_src = torch.randn(5, 4)
    x_dst = x_src[:2]
    # dst 0 <- src {2,3}, dst 1 <- src {4}
    edge_index = torch.tensor([[2, 3, 4], [0, 0, 1]])
    out = conv((x_src, x_dst), edge_index, (5, 2))
    mean0 = (x_src[2] + x_src[3]) / 2
    mean1 = x_src[4]
    expect = torch.stack([
        conv.lin_l(mean0) + conv.lin_r(x_dst[0]),
        conv.lin_l(mean1) + conv.lin_r(x_dst[1]),
    ])
    assert torch.allclose(out, expect, atol=1e-6)


def test_sage_conv_isolated_node():
    conv = SAGEConv(4, 3)
    x = torch.randn(3, 4)
    edge_index = torch.tensor([[1], [0]])
    out = conv((x, x[:2]), edge_index, (3, 2))
    # dst 1 has no incoming edges: only the self path contributes + bias
    expect = conv.lin_l(torch.zeros(4)) + conv.lin_r(x[1])
    assert torch.allclose(out[1], expect, atol=1e-6)


def test_gat_conv_attention_normalized():
    torch.manual_seed(0)
    conv = GATConv(4, 3, heads=2)
    x = torch.randn(6, 4)
    edge_index = torch.tensor([[1, 2, 3, 4, 5], [0, 0, 0, 1, 1]])
    out = conv((x, x[:2]), edge_index, (6, 2))
    assert out.shape == (2, 6)
    assert torch.isfinite(out).all()
    # single-source dst: attention weight is 1 -> output = h_src + bias
    edge_index = torch.tensor([[3], [0]])
    out = conv((x, x[:1]), edge_index, (6, 1))
    h3 = conv.lin(x[3]).view(2, 3)
    assert torch.allclose(out[0], h3.reshape(-1) + conv.bias, atol=1e-5)


def _fake_adjs(batch, frontier1, frontier2):
    e1 = torch.stack([torch.arange(frontier1) % frontier1,
                      torch.arange(frontier1) % batch])
    e2 = torch.stack([torch.arange(frontier2) % frontier2,
                      torch.arange(frontier2) % frontier1])
    return [Adj(e2, torch.tensor([]), (frontier2, frontier1)),
            Adj(e1, torch.tensor([]), (frontier1, batch))]


def test_models_forward_backward():
    torch.manual_seed(0)
    for model in (GraphSAGE(8, 16, 4, num_layers=2),
                  GAT(8, 8, 4, num_layers=2, heads=2)):
        adjs = _fake_adjs(4, 10, 30)
        x = torch.randn(30, 8)
        out = model(x, adjs)
        assert out.shape == (4, 4)
        loss = out.sum()
        loss.backward()
        grads = [p.grad for p in model.parameters() if p.grad is not None]
        assert len(grads) > 0
