import numpy as np
import torch

import quiver


def make_topo(indptr, indices):
    return quiver.CSRTopo(indptr=indptr, indices=indices)


def test_low_degree_rows_copied_exactly(small_graph):
    indptr, indices = small_graph
    topo = make_topo(indptr, indices)
    s = quiver.GraphSageSampler(topo, [8], mode="CPU")
    seeds = torch.arange(topo.node_count)
    out, cnt = s.sample_layer(seeds, 8)
    off = 0
    for i in range(topo.node_count):
        deg = int(indptr[i + 1] - indptr[i])
        expect = min(deg, 8)
        assert cnt[i] == expect
        got = out[off:off + expect].tolist()
        actual = indices[indptr[i]:indptr[i + 1]].tolist()
        if deg <= 8:
            assert got == actual
        else:
            # sampled without replacement from the neighbor positions
            assert len(got) == 8
            for g in got:
                assert g in actual
        off += expect


def test_sampling_is_subset_and_no_position_duplicates():
    # node 0 with 100 distinct neighbors; k=10 must give 10 distinct
    indptr = torch.tensor([0, 100], dtype=torch.long)
    indices = torch.arange(100, 200, dtype=torch.long)
    topo = quiver.CSRTopo(indptr=torch.tensor([0, 100] + [100] * 100),
                          indices=indices)
    s = quiver.GraphSageSampler(topo, [10], mode="CPU")
    for _ in range(20):
        out, cnt = s.sample_layer(torch.tensor([0]), 10)
        assert cnt[0] == 10
        assert len(set(out.tolist())) == 10
        assert all(100 <= v < 200 for v in out.tolist())


def test_sampling_uniformity():
    # reservoir sampling should select each neighbor with p = k/deg
    deg, k, trials = 50, 10, 4000
    indptr = torch.tensor([0, deg] + [deg] * deg, dtype=torch.long)
    indices = torch.arange(1, deg + 1, dtype=torch.long)
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    s = quiver.GraphSageSampler(topo, [k], mode="CPU")
    freq = np.zeros(deg + 1)
    for _ in range(trials):
        out, _ = s.sample_layer(torch.tensor([0]), k)
        freq[out.numpy()] += 1
    p = freq[1:] / trials
    expected = k / deg
    assert abs(p.mean() - expected) < 0.01
    # each neighbor within 5 sigma of binomial expectation
    sigma = np.sqrt(expected * (1 - expected) / trials)
    assert np.all(np.abs(p - expected) < 5 * sigma), p


def test_reindex_invariants(small_graph):
    indptr, indices = small_graph
    topo = make_topo(indptr, indices)
    s = quiver.GraphSageSampler(topo, [5], mode="CPU")
    seeds = torch.tensor([3, 7, 11, 19])
    out, cnt = s.sample_layer(seeds, 5)
    frontier, row, col = s.reindex(seeds, out, cnt)
    # seeds prefix
    assert frontier[:4].tolist() == seeds.tolist()
    # frontier unique
    assert len(set(frontier.tolist())) == frontier.numel()
    # row/col shapes and mapping back to global ids
    assert row.numel() == out.numel() == col.numel()
    assert int(cnt.sum()) == out.numel()
    for j in range(out.numel()):
        assert frontier[col[j]] == out[j]
    # row expansion matches counts
    expect_rows = torch.repeat_interleave(torch.arange(4), cnt)
    assert torch.equal(row, expect_rows)


def test_full_sample_two_hops(small_graph):
    indptr, indices = small_graph
    topo = make_topo(indptr, indices)
    s = quiver.GraphSageSampler(topo, [4, 3], mode="CPU")
    seeds = torch.arange(16)
    n_id, bs, adjs = s.sample(seeds)
    assert bs == 16
    assert len(adjs) == 2
    # layer order is reversed: adjs[0] is the outermost hop
    assert adjs[-1].size[1] == 16
    # target count of hop i == source count of hop i+1
    assert adjs[0].size[1] == adjs[1].size[0]
    assert n_id.numel() == adjs[0].size[0]
    # all edge endpoints are valid local ids
    for adj in adjs:
        assert adj.edge_index[0].max() < adj.size[0]
        assert adj.edge_index[1].max() < adj.size[1]
