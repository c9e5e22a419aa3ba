"""Minimal PyG-compatible GNN layers/models in plain torch.

The reference relies on torch_geometric for its models; this environment
has no PyG, so the SAGE/GAT model families used by examples, serving and
bench.py live here.  The layers follow PyG's bipartite convention: forward
((x_src, x_dst), edge_index, size) with edge_index[0] indexing x_src and
edge_index[1] indexing x_dst — exactly what GraphSageSampler's adjs feed.
Aggregations use index_add / scatter_reduce, which lower to native ROCm
kernels.
"""
from typing import Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import _ext

__all__ = ["SAGEConv", "GATConv", "GraphSAGE", "GAT", "QLinear"]


def _as_pair(x):
    if isinstance(x, (tuple, list)):
        return x[0], x[1]
    return x, x


class _SegmentMeanAgg(torch.autograd.Function):
    """Fused mean aggregation over dst-sorted edges (csrc/segment_kernels.hip).

    Replaces x[src] -> zeros -> index_add -> deg -> div with one kernel in
    each direction; the layer-1 input features carry no grad, so their
    backward is skipped entirely.
    """

    @staticmethod
    def forward(ctx, x, src, dst_ptr):
        ctx.save_for_backward(src, dst_ptr)
        ctx.n_src = x.size(0)
        return _ext.segment_mean_gather(x, src, dst_ptr)

    @staticmethod
    def backward(ctx, grad_out):
        src, dst_ptr = ctx.saved_tensors
        grad_x = _ext.segment_mean_gather_backward(
            grad_out.contiguous(), src, dst_ptr, ctx.n_src)
        return grad_x, None, None


class _SegmentWSum(torch.autograd.Function):
    """Weighted segment sum over dst-sorted edges (GAT attention
    aggregation, csrc/segment_kernels.hip).  Replaces the torch chain
    h_src[src] (materialize [E,H,C]) * alpha -> zeros -> index_add, which
    moves ~E*H*C*4 bytes twice per layer."""

    @staticmethod
    def forward(ctx, x, w, src, dst_ptr, heads):
        ctx.save_for_backward(x, w, src, dst_ptr)
        ctx.heads = heads
        return _ext.segment_wsum(x, w, src, dst_ptr, heads)

    @staticmethod
    def backward(ctx, grad_out):
        x, w, src, dst_ptr = ctx.saved_tensors
        gx, gw = _ext.segment_wsum_backward(
            grad_out.contiguous(), x, w, src, dst_ptr, ctx.heads,
            ctx.needs_input_grad[0], ctx.needs_input_grad[1])
        return (gx if ctx.needs_input_grad[0] else None,
                gw if ctx.needs_input_grad[1] else None, None, None, None)


class _GatDots(torch.autograd.Function):
    """Fused per-head attention dots: (h*att_src).sum(-1) and the same
    for att_dst over the prefix rows, in ONE read of h (the torch chain
    materializes a broadcast product and reduces it, ~3x the traffic on
    a ~1 GB projected frontier)."""

    @staticmethod
    def forward(ctx, h, att_src, att_dst, n_dst, heads):
        asrc, adst = _ext.gat_dots(h, att_src, att_dst, n_dst, heads)
        ctx.save_for_backward(h, att_src, att_dst)
        ctx.heads = heads
        return asrc, adst

    @staticmethod
    def backward(ctx, g_asrc, g_adst):
        h, att_src, att_dst = ctx.saved_tensors
        g_h, g_as, g_ad = _ext.gat_dots_backward(
            h, att_src, att_dst, g_asrc.contiguous(), g_adst.contiguous(),
            ctx.heads)
        return g_h, g_as, g_ad, None, None


class _GatAlpha(torch.autograd.Function):
    """Fused GAT attention coefficients over dst-sorted edges: per-node
    logit gathers + add + leaky_relu + segment softmax in one kernel pair
    (no [E,H] torch intermediates)."""

    @staticmethod
    def forward(ctx, asrc, adst, src, dst_ptr, heads, slope, n_edges):
        alpha = _ext.gat_alpha(asrc, adst, src, dst_ptr, heads, slope,
                               n_edges)
        ctx.save_for_backward(alpha, asrc, adst, src, dst_ptr)
        ctx.heads, ctx.slope = heads, slope
        return alpha

    @staticmethod
    def backward(ctx, grad_alpha):
        alpha, asrc, adst, src, dst_ptr = ctx.saved_tensors
        g_asrc, g_adst = _ext.gat_alpha_backward(
            grad_alpha, alpha, asrc, adst, src, dst_ptr, ctx.heads,
            ctx.slope)
        return g_asrc, g_adst, None, None, None, None, None


class _SegmentSoftmax(torch.autograd.Function):
    """Softmax over dst-sorted edge segments (csrc/segment_kernels.hip).
    torch's scatter_reduce(amax) path lowers to ~120 rocprim sort kernels
    per step on ROCm; this is two scalar kernels."""

    @staticmethod
    def forward(ctx, a, dst_ptr, heads):
        out = _ext.segment_softmax(a, dst_ptr, heads)
        ctx.save_for_backward(out, dst_ptr)
        ctx.heads = heads
        return out

    @staticmethod
    def backward(ctx, grad_out):
        out, dst_ptr = ctx.saved_tensors
        return (_ext.segment_softmax_backward(grad_out.contiguous(), out,
                                              dst_ptr, ctx.heads),
                None, None)


class _QLinearFn(torch.autograd.Function):
    """Linear layer whose weight gradient uses the split-K HIP kernel
    (csrc/wgrad_kernels.hip).

    Big-frontier linear GEMM routing (all measured, benchmarks/):
    - weight-grad: custom MFMA split-K (csrc/wgrad_kernels.hip,
      deterministic workspace reduce, 2.8-5.7x rocBLAS on these shapes)
      with the bias grad folded in;
    - narrow-output data-grads (in_features <= 128): custom tall-M MFMA
      kernel (csrc/gemm_kernels.hip, 1.1-1.25x);
    - forward and wide data-grads: rocBLAS (it wins there).
    Custom-kernel numerics are exact f32 fmaf chains
    (v_mfma_f32_32x32x2_f32).
    """

    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return F.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, grad_out):
        x, weight = ctx.saved_tensors
        grad_out = grad_out.contiguous()
        if ctx.needs_input_grad[0]:
            # narrow-output data-grads (in_features <= 128) beat rocBLAS
            # on the tall-M MFMA kernel (measured 1.1-1.25x; wide shapes
            # lose ~15% — rocBLAS keeps those; benchmarks/bench_gemm.py)
            if weight.size(1) <= 128 and grad_out.is_cuda \
                    and grad_out.dtype == torch.float32:
                grad_x = _ext.tall_gemm(grad_out, weight, None)
            else:
                grad_x = grad_out @ weight
        else:
            grad_x = None
        grad_w, grad_b = _ext.wgrad(grad_out, x, ctx.has_bias)
        return grad_x, grad_w, (grad_b if ctx.has_bias else None)


# frontier sizes below this use plain F.linear (rocBLAS wgrad is fine there)
_WGRAD_MIN_K = 16384


class QLinear(nn.Linear):
    """nn.Linear with the custom tall-skinny weight-grad path on GPU fp32
    inputs with a large leading (frontier) dimension."""

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.float32 and x.dim() == 2
                and x.size(0) >= _WGRAD_MIN_K and torch.is_grad_enabled()):
            return _QLinearFn.apply(x, self.weight, self.bias)
        return F.linear(x, self.weight, self.bias)


_ARANGE_CACHE = {}


def _arange(n, device):
    """Cached 0..n view (avoids one alloc+launch per layer per step)."""
    key = device.index if device.type == "cuda" else -1
    cached = _ARANGE_CACHE.get(key)
    if cached is None or cached.numel() < n:
        cached = torch.arange(max(n, 1 << 20), device=device)
        _ARANGE_CACHE[key] = cached
    return cached[:n]


_QUIVER_DEBUG = bool(int(__import__("os").environ.get("QUIVER_DEBUG", "0")))


def _dst_ptr_from_sorted(dst, n_dst):
    """Segment pointer from a dst-sorted edge list.  searchsorted silently
    returns garbage on unsorted input, so sorted_dst=True layers fed edges
    from a non-quiver sampler would be silently wrong — QUIVER_DEBUG=1
    turns that into a loud error."""
    if _QUIVER_DEBUG and dst.numel() > 1:
        assert bool((dst[1:] >= dst[:-1]).all()), \
            "sorted_dst=True but edge_index[1] is not sorted ascending"
    return torch.searchsorted(dst, _arange(n_dst + 1, dst.device))


def _mean_aggregate(x_src, src, dst, n_dst, sorted_dst=False):
    """Mean of x_src[src] grouped by dst.  Uses the fused HIP kernel when
    the caller guarantees dst is sorted ascending (our sampler's layout)
    on fp32 GPU tensors."""
    fused_ok = (sorted_dst and x_src.is_cuda and dst.numel() > 0
                and (x_src.dtype == torch.float32
                     or (x_src.dtype == torch.bfloat16
                         and x_src.size(1) % 2 == 0)))
    if fused_ok:
        dst_ptr = _dst_ptr_from_sorted(dst, n_dst)
        return _SegmentMeanAgg.apply(x_src, src, dst_ptr)
    agg = torch.zeros((n_dst, x_src.size(1)), dtype=x_src.dtype,
                      device=x_src.device)
    agg.index_add_(0, dst, x_src[src])
    deg = torch.zeros(n_dst, dtype=x_src.dtype, device=x_src.device)
    deg.index_add_(0, dst, torch.ones_like(dst, dtype=x_src.dtype))
    return agg / deg.clamp_(min=1).unsqueeze(-1)


class SAGEConv(nn.Module):
    """GraphSAGE-mean convolution.

    Set sorted_dst=True when edge_index[1] is sorted ascending (always the
    case for GraphSageSampler's adjs) to enable the fused segment-mean HIP
    kernel."""

    def __init__(self, in_channels, out_channels, bias=True,
                 sorted_dst=False):
        super().__init__()
        self.sorted_dst = sorted_dst
        self.lin_l = QLinear(in_channels, out_channels, bias=bias)  # neigh
        self.lin_r = QLinear(in_channels, out_channels, bias=False)  # self
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.xavier_uniform_(self.lin_l.weight)
        nn.init.xavier_uniform_(self.lin_r.weight)
        if self.lin_l.bias is not None:
            nn.init.zeros_(self.lin_l.bias)

    def forward(self, x, edge_index, size: Tuple[int, int] = None):
        x_src, x_dst = _as_pair(x)
        src, dst = edge_index[0], edge_index[1]
        n_dst = x_dst.size(0) if size is None else int(size[1])
        # aggregate-then-project: segment mean on in_channels, one GEMM after
        agg = _mean_aggregate(x_src, src, dst, n_dst,
                              sorted_dst=self.sorted_dst)
        out = self.lin_l(agg)
        if (x_dst.is_cuda and x_dst.dtype == torch.float32
                and x_dst.size(0) >= _WGRAD_MIN_K
                and torch.is_grad_enabled()):
            # large frontier: route the self path through QLinear so ITS
            # weight grad also takes the split-K kernel
            return out + self.lin_r(x_dst)
        # small/no-grad: self path fused into the neighbor GEMM via addmm
        return out.addmm_(x_dst, self.lin_r.weight.t()) if not out.requires_grad \
            else torch.addmm(out, x_dst, self.lin_r.weight.t())


class GATConv(nn.Module):
    """Graph attention convolution (multi-head, concat)."""

    def __init__(self, in_channels, out_channels, heads=1, concat=True,
                 negative_slope=0.2, dropout=0.0, bias=True,
                 sorted_dst=False):
        super().__init__()
        self.sorted_dst = sorted_dst
        self.heads = heads
        self.out_channels = out_channels
        self.concat = concat
        self.negative_slope = negative_slope
        self.dropout = dropout
        self.lin = QLinear(in_channels, heads * out_channels, bias=False)
        self.att_src = nn.Parameter(torch.empty(1, heads, out_channels))
        self.att_dst = nn.Parameter(torch.empty(1, heads, out_channels))
        out_dim = heads * out_channels if concat else out_channels
        self.bias = nn.Parameter(torch.zeros(out_dim)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.xavier_uniform_(self.lin.weight)
        nn.init.xavier_uniform_(self.att_src)
        nn.init.xavier_uniform_(self.att_dst)
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    def forward(self, x, edge_index, size: Tuple[int, int] = None):
        x_src, x_dst = _as_pair(x)
        src, dst = edge_index[0], edge_index[1]
        H, C = self.heads, self.out_channels
        n_dst = x_dst.size(0) if size is None else int(size[1])

        h2 = self.lin(x_src)
        # bf16 models: the projected features stay bf16 end to end — the
        # dots/wsum kernels take bf16 h with fp32 accumulation, while the
        # LOGITS (alpha) and softmax stay fp32 (standard practice, and
        # they are [*, H]-small).  Odd head widths fall back to one
        # boundary cast; torch's bf16 softmax fallback chain
        # (scatter_reduce amax -> ~120 rocprim sort kernels) is far
        # slower than either.
        out_dtype = h2.dtype
        native = (h2.is_cuda and self.sorted_dst and C % 4 == 0
                  and h2.dtype in (torch.float32, torch.bfloat16))
        if (h2.is_cuda and h2.dtype == torch.bfloat16 and self.sorted_dst
                and not native):
            h2 = h2.float()
        h_src = h2.view(-1, H, C)
        prefix = (x_dst.data_ptr() == x_src.data_ptr()
                  and x_dst.size(0) <= x_src.size(0)
                  and x_dst.stride() == x_src.stride()
                  and x_dst.size(1) == x_src.size(1))
        if prefix and native:
            # bipartite prefix convention (x_dst is x_src[:n_dst]) with
            # the fused dots kernel: one read of h for both logit sets
            alpha_src, alpha_dst = _GatDots.apply(
                h2, self.att_src.reshape(-1).float(),
                self.att_dst.reshape(-1).float(), x_dst.size(0), H)
        else:
            if prefix:
                h_dst = h_src[:x_dst.size(0)]
            else:
                h_dst = self.lin(x_dst).view(-1, H, C)
            alpha_src = (h_src * self.att_src).sum(-1)  # [N_src, H]
            alpha_dst = (h_dst * self.att_dst).sum(-1)  # [N_dst, H]
        fused = (self.sorted_dst and alpha_src.is_cuda
                 and alpha_src.dtype == torch.float32 and dst.numel() > 0)
        if fused:
            dst_ptr = _dst_ptr_from_sorted(dst, n_dst)
            alpha = _GatAlpha.apply(alpha_src, alpha_dst, src, dst_ptr, H,
                                    self.negative_slope, src.numel())
        else:
            alpha = alpha_src[src] + alpha_dst[dst]     # [E, H]
            alpha = F.leaky_relu(alpha, self.negative_slope)
            # segment softmax over incoming edges of each dst (torch path)
            alpha_max = torch.full((n_dst, H), float("-inf"),
                                   dtype=alpha.dtype, device=alpha.device)
            alpha_max = alpha_max.scatter_reduce(
                0, dst.unsqueeze(-1).expand_as(alpha), alpha, reduce="amax",
                include_self=True)
            alpha = (alpha - alpha_max[dst]).exp()
            denom = torch.zeros((n_dst, H), dtype=alpha.dtype,
                                device=alpha.device)
            denom.index_add_(0, dst, alpha)
            alpha = alpha / denom[dst].clamp(min=1e-16)
        if self.training and self.dropout > 0:
            alpha = F.dropout(alpha, p=self.dropout)

        if (self.sorted_dst and h_src.is_cuda and C % 4 == 0
                and h_src.dtype in (torch.float32, torch.bfloat16)
                and alpha.dtype == torch.float32
                and dst.numel() > 0):
            # fused weighted segment sum over the dst-sorted edges
            if not fused:
                dst_ptr = _dst_ptr_from_sorted(dst, n_dst)
            out = _SegmentWSum.apply(h_src.reshape(-1, H * C), alpha, src,
                                     dst_ptr, H)
            out = out if self.concat else out.view(n_dst, H, C).mean(1)
        else:
            msg = h_src[src] * alpha.unsqueeze(-1)      # [E, H, C]
            out = torch.zeros((n_dst, H, C), dtype=msg.dtype,
                              device=msg.device)
            out.index_add_(0, dst, msg)
            out = (out.reshape(n_dst, H * C) if self.concat
                   else out.mean(1))
        if self.bias is not None:
            out = out + self.bias
        return out.to(out_dtype) if out.dtype != out_dtype else out


class GraphSAGE(nn.Module):
    """Multi-layer GraphSAGE matching the reference examples' training/
    inference loop (forward(x, adjs) over per-hop bipartite graphs)."""

    def __init__(self, in_channels, hidden_channels, out_channels,
                 num_layers=3, dropout=0.5, sorted_dst=True):
        super().__init__()
        self.num_layers = num_layers
        self.dropout = dropout
        self.convs = nn.ModuleList()
        # sorted_dst: GraphSageSampler emits dst-sorted adjs -> fused kernel
        if num_layers == 1:
            self.convs.append(SAGEConv(in_channels, out_channels,
                                       sorted_dst=sorted_dst))
        else:
            self.convs.append(SAGEConv(in_channels, hidden_channels,
                                       sorted_dst=sorted_dst))
            for _ in range(num_layers - 2):
                self.convs.append(SAGEConv(hidden_channels, hidden_channels,
                                           sorted_dst=sorted_dst))
            self.convs.append(SAGEConv(hidden_channels, out_channels,
                                       sorted_dst=sorted_dst))

    def forward(self, x, adjs):
        for i, (edge_index, _, size) in enumerate(adjs):
            x_target = x[:size[1]]
            x = self.convs[i]((x, x_target), edge_index, size)
            if i != self.num_layers - 1:
                x = F.relu(x, inplace=True)
                if self.dropout > 0:
                    x = F.dropout(x, p=self.dropout, training=self.training)
        return torch.log_softmax(x, dim=-1)

    def full_forward(self, x, edge_index):
        n = x.size(0)
        # fused segment path needs dst-sorted edges; sort once up front
        order = torch.argsort(edge_index[1], stable=True)
        edge_index = edge_index[:, order]
        for i, conv in enumerate(self.convs):
            x = conv((x, x), edge_index, (n, n))
            if i != self.num_layers - 1:
                x = F.relu(x)
        return torch.log_softmax(x, dim=-1)


class GAT(nn.Module):
    def __init__(self, in_channels, hidden_channels, out_channels,
                 num_layers=2, heads=4, dropout=0.5, sorted_dst=True):
        super().__init__()
        self.num_layers = num_layers
        self.dropout = dropout
        self.convs = nn.ModuleList()
        # sorted_dst: GraphSageSampler emits dst-sorted adjs -> fused kernel
        if num_layers == 1:
            self.convs.append(GATConv(in_channels, out_channels, heads=1,
                                      sorted_dst=sorted_dst))
        else:
            self.convs.append(GATConv(in_channels, hidden_channels,
                                      heads=heads, sorted_dst=sorted_dst))
            for _ in range(num_layers - 2):
                self.convs.append(GATConv(hidden_channels * heads,
                                          hidden_channels, heads=heads,
                                          sorted_dst=sorted_dst))
            self.convs.append(GATConv(hidden_channels * heads, out_channels,
                                      heads=1, sorted_dst=sorted_dst))

    def forward(self, x, adjs):
        for i, (edge_index, _, size) in enumerate(adjs):
            x_target = x[:size[1]]
            x = self.convs[i]((x, x_target), edge_index, size)
            if i != self.num_layers - 1:
                x = F.elu(x, inplace=True)
                if self.dropout > 0:
                    x = F.dropout(x, p=self.dropout, training=self.training)
        return torch.log_softmax(x, dim=-1)
