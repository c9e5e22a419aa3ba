"""GPU sampler correctness: wave64 HIP kernels vs CPU/plain-torch ground
truth (numerics reference computed in fp32/torch on CPU)."""
import numpy as np
import pytest
import torch

import quiver

pytestmark = pytest.mark.gpu


@pytest.fixture(params=["UVA", "GPU"])
def gpu_sampler(request, small_graph):
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    return quiver.GraphSageSampler(topo, [8, 4], device=0,
                                   mode=request.param), topo


def test_sample_layer_matches_csr(gpu_sampler):
    sampler, topo = gpu_sampler
    indptr, indices = topo.indptr, topo.indices
    seeds = torch.arange(topo.node_count)
    out, cnt = sampler.sample_layer(seeds, 8)
    out, cnt = out.cpu(), cnt.cpu()
    off = 0
    for i in range(topo.node_count):
        deg = int(indptr[i + 1] - indptr[i])
        expect = min(deg, 8)
        assert cnt[i] == expect, i
        got = out[off:off + expect].tolist()
        actual = indices[indptr[i]:indptr[i + 1]].tolist()
        if deg <= 8:
            assert got == actual, i
        else:
            for g in got:
                assert g in actual
        off += expect
    assert off == out.numel()


def test_sample_no_duplicate_positions():
    # distinct neighbors => sampled set must be duplicate-free
    deg, k = 200, 16
    indptr = torch.tensor([0, deg] + [deg] * deg, dtype=torch.long)
    indices = torch.arange(1, deg + 1, dtype=torch.long)
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s = quiver.GraphSageSampler(topo, [k], device=0, mode="GPU")
    for _ in range(10):
        out, cnt = s.sample_layer(torch.tensor([0]), k)
        vals = out.cpu().tolist()
        assert len(vals) == k
        assert len(set(vals)) == k, vals


def test_sampling_uniformity_gpu():
    deg, k, trials = 64, 8, 3000
    indptr = torch.tensor([0, deg] + [deg] * deg, dtype=torch.long)
    indices = torch.arange(1, deg + 1, dtype=torch.long)
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s = quiver.GraphSageSampler(topo, [k], device=0, mode="GPU")
    seeds = torch.zeros(trials, dtype=torch.long)
    out, _ = s.sample_layer(seeds, k)
    freq = np.bincount(out.cpu().numpy(), minlength=deg + 1)[1:]
    p = freq / trials
    expected = k / deg
    sigma = np.sqrt(expected * (1 - expected) / trials)
    assert abs(p.mean() - expected) < 1e-6
    assert np.all(np.abs(p - expected) < 6 * sigma), p


def test_reindex_invariants_gpu(gpu_sampler):
    sampler, topo = gpu_sampler
    seeds = torch.tensor([3, 7, 11, 19])
    out, cnt = sampler.sample_layer(seeds, 8)
    frontier, row, col = sampler.reindex(seeds.cuda(), out, cnt)
    frontier, row, col = frontier.cpu(), row.cpu(), col.cpu()
    out, cnt = out.cpu(), cnt.cpu()
    assert frontier[:4].tolist() == seeds.tolist()
    assert len(set(frontier.tolist())) == frontier.numel()
    for j in range(out.numel()):
        assert frontier[col[j]] == out[j]
    assert torch.equal(row, torch.repeat_interleave(torch.arange(4), cnt))


def test_full_sample_two_hops_gpu(gpu_sampler):
    sampler, topo = gpu_sampler
    seeds = torch.arange(32)
    n_id, bs, adjs = sampler.sample(seeds)
    assert bs == 32
    assert adjs[-1].size[1] == 32
    assert adjs[0].size[1] == adjs[1].size[0]
    assert n_id.numel() == adjs[0].size[0]
    for adj in adjs:
        assert adj.edge_index[0].max() < adj.size[0]
        assert adj.edge_index[1].max() < adj.size[1]


def test_sample_prob_matches_reference():
    # chain graph: 0->1->2, k=1; verify against the formula computed in torch
    indptr = torch.tensor([0, 2, 3, 3], dtype=torch.long)
    indices = torch.tensor([1, 2, 2], dtype=torch.long)
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s = quiver.GraphSageSampler(topo, [1], device=0, mode="GPU")
    prob = s.sample_prob(torch.tensor([0]), 3).cpu()
    # last = [1,0,0]; deg = [2,1,0]
    # cur[0] = 1 (seed); cur[1] = 1-(1-0)*(1-last[2]*...) with N(1)={2}: 0
    # formula: cur[v] = 1-(1-last[v]) * prod_u (1 - last[u]*min(1,k/deg_u))
    # N(0) = {1,2}: both last=0 -> prod=1 -> cur[0] = 1-(1-1)*1 = 1
    # N(1) = {2}: last=0 -> cur[1] = 0;  N(2) empty -> cur[2] = 0
    assert torch.allclose(prob, torch.tensor([1.0, 0.0, 0.0]))

    prob2 = s.sample_prob(torch.tensor([1]), 3).cpu()
    # last=[0,1,0]: cur[0] = 1-(1-0)*[(1-1*min(1,1/1)) * (1-0)] = 1
    assert torch.allclose(prob2, torch.tensor([1.0, 1.0, 0.0]))


def test_deterministic_with_seed(small_graph):
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s = quiver.GraphSageSampler(topo, [8], device=0, mode="GPU")
    s.lazy_init_quiver()
    seeds = torch.arange(topo.node_count)
    s.quiver.set_seed(1234)
    a, _ = s.sample_layer(seeds, 8)
    s.quiver.set_seed(1234)
    b, _ = s.sample_layer(seeds, 8)
    assert torch.equal(a.cpu(), b.cpu())


def test_fused_sample_hops_edges_valid(gpu_sampler):
    """Fused multi-hop path: every emitted edge must exist in the CSR, the
    seed prefix must be preserved, and local ids must be in range."""
    sampler, topo = gpu_sampler
    indptr, indices = topo.indptr, topo.indices
    adj_sets = [set(indices[indptr[v]:indptr[v + 1]].tolist())
                for v in range(topo.node_count)]
    seeds = torch.arange(40)
    n_id, bs, adjs = sampler.sample(seeds)
    n_id = n_id.cpu()
    assert torch.equal(n_id[:bs], seeds)
    assert n_id.numel() == len(set(n_id.tolist()))  # frontier is unique
    # walk hops from the root layer (adjs reversed): target ids of hop h
    # index the previous frontier
    frontiers = [n_id]
    for adj in adjs:  # adjs[0] = outermost hop
        src, dst = adj.edge_index[0].cpu(), adj.edge_index[1].cpu()
        assert src.numel() == dst.numel()
        assert src.max() < adj.size[0] and dst.max() < adj.size[1]
    # hop-level neighbor check against CSR using the per-hop frontier
    # (reconstruct: last adj's size[1]==bs, frontier chain shares prefix)
    hop_frontier = n_id
    for adj in adjs:
        src, dst = adj.edge_index[0].cpu(), adj.edge_index[1].cpu()
        sub_frontier = hop_frontier[:adj.size[0]]
        targets = hop_frontier[:adj.size[1]]
        for e in range(0, src.numel(), 7):  # sampled subset for speed
            v = int(targets[dst[e]])
            u = int(sub_frontier[src[e]])
            assert u in adj_sets[v], (v, u)
        hop_frontier = targets


def test_fused_matches_perhop_deterministic_counts(small_graph):
    """Fused path agrees with the degree-determined quantities: first-layer
    sampled edge count = sum(min(deg(seed), k)), batch size preserved."""
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s_fused = quiver.GraphSageSampler(topo, [6, 3], device=0, mode="GPU")
    seeds = torch.arange(25)
    n1, b1, a1 = s_fused.sample(seeds)
    deg = (topo.indptr[1:] - topo.indptr[:-1])[seeds]
    want_e1 = int(torch.minimum(deg, torch.tensor(6)).sum())
    assert a1[-1].edge_index.shape[1] == want_e1
    assert b1 == 25


def test_cal_next_randomized_vs_cpu(small_graph):
    """cal_next on a random graph vs a CPU implementation of the formula
    (reference cuda_random.cu.hpp:71-104):
    cur[v] = 1 - (1 - last[v]) * prod_{u in N(v)} (1 - last[u]*min(1,k/deg_u))
    """
    indptr, indices = small_graph
    n = indptr.numel() - 1
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    deg = (indptr[1:] - indptr[:-1]).float()
    for k, seed in [(3, 0), (8, 1), (25, 2)]:
        s = quiver.GraphSageSampler(topo, [k], device=0, mode="GPU")
        s.lazy_init_quiver()
        g = torch.Generator().manual_seed(seed)
        last = torch.rand(n, generator=g)
        last[torch.rand(n, generator=g) < 0.5] = 0.0  # sparse support
        cur_gpu = torch.zeros(n, device="cuda:0")
        s.quiver.cal_neighbor_prob(0, last.to("cuda:0"), cur_gpu, k)
        cur_gpu = cur_gpu.cpu()
        # CPU reference
        sel = last * torch.minimum(torch.ones(n), k / deg.clamp(min=1))
        sel[deg == 0] = 0.0
        cur_ref = torch.empty(n)
        for v in range(n):
            nbrs = indices[indptr[v]:indptr[v + 1]]
            prod = float(torch.prod(1.0 - sel[nbrs])) if nbrs.numel() else 1.0
            cur_ref[v] = 1.0 - (1.0 - last[v]) * prod
        assert torch.allclose(cur_gpu, cur_ref, atol=1e-5), \
            f"k={k}: max err {(cur_gpu - cur_ref).abs().max()}"


def test_multihop_sample_prob_monotone(small_graph):
    """Multi-hop access probability: seeds stay at 1, probabilities lie in
    [0,1], and adding a hop can only grow each node's probability."""
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    n = topo.node_count
    train_idx = torch.arange(0, n, 7)
    s1 = quiver.GraphSageSampler(topo, [10], device=0, mode="GPU")
    s2 = quiver.GraphSageSampler(topo, [10, 10], device=0, mode="GPU")
    p1 = s1.sample_prob(train_idx, n).cpu()
    p2 = s2.sample_prob(train_idx, n).cpu()
    assert torch.all(p1[train_idx] == 1.0)
    assert torch.all((p1 >= 0) & (p1 <= 1.0 + 1e-6))
    assert torch.all(p2 >= p1 - 1e-5), "extra hop lowered access prob"
