"""Numerics of the split-K weight-grad kernel (csrc/wgrad_kernels.hip)
against plain torch fp32 matmul, and of QLinear end-to-end autograd."""
import pytest
import torch

import quiver  # noqa: F401
from quiver import _ext
from quiver.nn import QLinear, _WGRAD_MIN_K

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("k,m,n", [(1000, 100, 256), (200_000, 100, 256),
                                   (50_000, 256, 256), (33_000, 47, 256),
                                   (70_000, 64, 64), (16, 3, 5)])
def test_wgrad_matches_torch(k, m, n):
    g = torch.Generator(device="cuda").manual_seed(0)
    a = torch.randn(k, m, device="cuda", generator=g)
    b = torch.randn(k, n, device="cuda", generator=g)
    c, bias = _ext.wgrad(a, b, True)
    # fp64 ground truth; fp32 split-K accumulation error grows ~sqrt(K),
    # so judge by relative error against the fp64 result
    want_c = (a.double().t() @ b.double())
    want_bias = a.double().sum(0)
    rel_c = (c.double() - want_c).norm() / want_c.norm()
    rel_b = (bias.double() - want_bias).norm() / max(want_bias.norm(), 1e-30)
    assert rel_c < 1e-5, float(rel_c)
    assert rel_b < 1e-4, float(rel_b)  # bias sums can cancel toward 0


def test_wgrad_no_bias():
    a = torch.randn(5000, 32, device="cuda")
    b = torch.randn(5000, 16, device="cuda")
    c, bias = _ext.wgrad(a, b, False)
    assert bias is None
    assert torch.allclose(c, a.t() @ b, atol=1e-2, rtol=1e-4)


def test_qlinear_autograd_matches_nn_linear():
    k = _WGRAD_MIN_K + 100
    torch.manual_seed(0)
    x = torch.randn(k, 100, device="cuda", requires_grad=True)
    ql = QLinear(100, 256).cuda()
    ref = torch.nn.Linear(100, 256).cuda()
    with torch.no_grad():
        ref.weight.copy_(ql.weight)
        ref.bias.copy_(ql.bias)
    x2 = x.detach().clone().requires_grad_(True)

    out = ql(x)
    out.pow(2).mean().backward()
    out2 = ref(x2)
    out2.pow(2).mean().backward()

    assert torch.allclose(out, out2, atol=1e-5)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5, rtol=1e-4)
    assert torch.allclose(ql.weight.grad, ref.weight.grad,
                          atol=1e-3, rtol=1e-3)
    assert torch.allclose(ql.bias.grad, ref.bias.grad, atol=1e-3, rtol=1e-3)


def test_wgrad_deterministic():
    """Split-K partials are reduced in fixed order (workspace + reduce
    kernel), so repeated calls on identical inputs are bitwise equal —
    unlike rocBLAS GSU."""
    g = torch.Generator(device="cuda").manual_seed(3)
    a = torch.randn(120_000, 256, device="cuda", generator=g)
    b = torch.randn(120_000, 100, device="cuda", generator=g)
    c0, bias0 = _ext.wgrad(a, b, True)
    for _ in range(3):
        c, bias = _ext.wgrad(a, b, True)
        assert torch.equal(c, c0)
        assert torch.equal(bias, bias0)
