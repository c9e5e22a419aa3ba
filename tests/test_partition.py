import torch

from quiver.partition import (partition_without_replication,
                              quiver_partition_feature,
                              load_quiver_feature_partition, select_nodes)


def test_partition_disjoint_and_covering():
    torch.manual_seed(0)
    n = 1000
    probs = [torch.rand(n) for _ in range(4)]
    parts = partition_without_replication("cpu", probs, None)
    assert len(parts) == 4
    allids = torch.cat(parts)
    assert allids.numel() == n
    assert len(set(allids.tolist())) == n  # disjoint + covering


def test_partition_prefers_own_hot_nodes():
    # interleaved preferences: every chunk contains both ranks' hot nodes,
    # so the greedy pick can honor them while staying balanced
    n = 400
    p0 = torch.zeros(n)
    p0[0::2] = 1.0
    p1 = torch.zeros(n)
    p1[1::2] = 1.0
    parts = partition_without_replication("cpu", [p0, p1], None)
    own0 = sum(1 for v in parts[0].tolist() if v % 2 == 0)
    own1 = sum(1 for v in parts[1].tolist() if v % 2 == 1)
    assert own0 > 180 and own1 > 180, (own0, own1)


def test_select_nodes():
    probs = [torch.tensor([0.0, 1.0, 0.0]), torch.tensor([0.0, 0.5, 2.0])]
    prob_sum, ids = select_nodes("cpu", probs, None)
    assert set(ids.flatten().tolist()) == {1, 2}


def test_partition_save_load(tmp_path):
    torch.manual_seed(1)
    probs = [torch.rand(256) for _ in range(2)]
    path = str(tmp_path / "parts")
    book, parts, caches = quiver_partition_feature(
        probs, path, cache_memory_budget="1K", per_feature_size=4)
    book2, part0, cache0 = load_quiver_feature_partition(0, path)
    assert torch.equal(book.cpu(), book2.cpu())
    assert torch.equal(parts[0].cpu(), part0.cpu())
    assert cache0 is not None and cache0.numel() > 0
    # book consistency
    for idx, part in enumerate(parts):
        assert (book[part] == idx).all()
