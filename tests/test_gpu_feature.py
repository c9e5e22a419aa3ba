"""ShardTensor / Feature GPU correctness vs plain-torch CPU ground truth:
HBM shard, pinned-host zero-copy shard, mixed tiers, dtypes, IPC."""
import pytest
import torch
import torch.multiprocessing as mp

import quiver
from quiver.shard_tensor import ShardTensor, ShardTensorConfig

pytestmark = pytest.mark.gpu


def _check_gather(cpu_tensor, st, n=2048):
    g = torch.Generator().manual_seed(7)
    idx = torch.randint(0, cpu_tensor.size(0), (n,), generator=g)
    got = st[idx.cuda()].cpu()
    assert torch.equal(got, cpu_tensor[idx]), (got[:3], cpu_tensor[idx][:3])


@pytest.mark.parametrize("dtype,dim", [
    (torch.float32, 100),   # 400 B rows (16B multiple)
    (torch.float16, 602),   # 1204 B rows (4B aligned)
    (torch.bfloat16, 7),    # 14 B rows (2B -> byte path)
    (torch.int64, 16),      # 128 B rows
    (torch.float32, 1),     # 4 B rows
])
def test_gather_all_gpu(dtype, dim):
    if dtype.is_floating_point:
        t = torch.randn(5000, dim).to(dtype)
    else:
        t = torch.randint(0, 1 << 40, (5000, dim), dtype=dtype)
    st = ShardTensor(0, ShardTensorConfig({}))
    st.append(t, 0)
    _check_gather(t, st)


def test_gather_gpu_plus_cpu_tiers():
    t = torch.randn(8000, 128)
    st = ShardTensor(0, ShardTensorConfig({}))
    st.append(t[:3000], 0)   # HBM
    st.append(t[3000:], -1)  # pinned host, zero-copy
    assert st.shape[0] == 8000
    _check_gather(t, st)


def test_gather_cpu_only_tier():
    t = torch.randn(4000, 32)
    st = ShardTensor(0, ShardTensorConfig({}))
    st.append(t, -1)
    _check_gather(t, st)


def test_scatter_update_roundtrip():
    t = torch.zeros(1000, 64)
    st = ShardTensor(0, ShardTensorConfig({}))
    st.append(t, 0)
    idx = torch.arange(0, 1000, 3)
    src = torch.randn(idx.numel(), 64).cuda()
    st.shard_tensor.scatter_update(idx.cuda(), src)
    got = st[idx.cuda()]
    assert torch.equal(got, src)


def test_from_cpu_tensor_budget_split():
    t = torch.randn(1000, 64)  # 256 B/row
    cfg = ShardTensorConfig({0: 256 * 300})  # 300 rows on GPU
    st = ShardTensor(0, cfg)
    st.from_cpu_tensor(t)
    assert st.shard_tensor.shard_rows() == [300, 700]
    assert st.shard_tensor.shard_devices() == [0, -1]
    _check_gather(t, st)


def test_feature_device_replicate_end_to_end(small_graph):
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    n = topo.node_count
    feat_cpu = torch.randn(n, 32)
    feature = quiver.Feature(0, device_list=[0], device_cache_size="16K",
                             cache_policy="device_replicate", csr_topo=topo)
    feature.from_cpu_tensor(feat_cpu.clone())
    idx = torch.randperm(n)[:200]
    got = feature[idx.cuda()].cpu()
    assert torch.equal(got, feat_cpu[idx])
    assert feature.size(0) == n
    assert feature.size(1) == 32
    assert feature.dim() == 2


def test_feature_no_cache_all_cpu():
    feat_cpu = torch.randn(500, 16)
    feature = quiver.Feature(0, device_list=[0], device_cache_size=0)
    feature.from_cpu_tensor(feat_cpu.clone())
    idx = torch.randperm(500)[:100]
    assert torch.equal(feature[idx.cuda()].cpu(), feat_cpu[idx])


def _ipc_child(rank, handle, feat_cpu, q):
    torch.cuda.set_device(0)
    feature = quiver.Feature.lazy_from_ipc_handle(handle)
    idx = torch.arange(100)
    got = feature[idx.cuda()].cpu()
    q.put(bool(torch.equal(got, feat_cpu[idx])))


def test_feature_ipc_across_processes():
    feat_cpu = torch.randn(1000, 32)
    feature = quiver.Feature(0, device_list=[0], device_cache_size="64K")
    feature.from_cpu_tensor(feat_cpu.clone())
    handle = feature.share_ipc()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    p = ctx.Process(target=_ipc_child, args=(0, handle, feat_cpu, q))
    p.start()
    ok = q.get()
    p.join(timeout=60)
    assert ok


def test_mixed_sampler_gpu(small_graph):
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)

    class Job(quiver.SampleJob):
        def __init__(self):
            self.data = [torch.arange(i * 20, i * 20 + 20) for i in range(6)]

        def __getitem__(self, i):
            return self.data[i]

        def __len__(self):
            return len(self.data)

        def shuffle(self):
            pass

    sampler = quiver.MixedGraphSageSampler(Job(), 1, topo, [4, 4],
                                           device=0, mode="GPU_ONLY")
    results = list(iter(sampler))
    assert len(results) == 6
    for n_id, bs, adjs in results:
        assert bs == 20
        assert len(adjs) == 2


@pytest.mark.parametrize("mode", ["GPU_CPU_MIXED", "UVA_CPU_MIXED"])
def test_mixed_sampler_gpu_cpu_work_stealing(small_graph, mode):
    """MIXED modes on real hardware: device sampler + CPU worker pool
    split one epoch adaptively; every job batch must come back exactly
    once with a valid computational graph."""
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)

    class Job(quiver.SampleJob):
        def __init__(self):
            self.data = [torch.arange(i * 25, i * 25 + 25)
                         for i in range(12)]

        def __getitem__(self, i):
            return self.data[i]

        def __len__(self):
            return len(self.data)

        def shuffle(self):
            pass

    sampler = quiver.MixedGraphSageSampler(Job(), 2, topo, [6, 4],
                                           device=0, mode=mode)
    results = list(iter(sampler))
    assert len(results) == 12
    seen_first = set()
    for n_id, bs, adjs in results:
        assert bs == 25
        assert len(adjs) == 2
        n_id = n_id.cpu()
        seen_first.add(int(n_id[0]))
        for adj in adjs:
            ei = adj.edge_index.cpu()
            assert ei[0].max() < adj.size[0]
            assert ei[1].max() < adj.size[1]
    assert len(seen_first) == 12  # every batch exactly once
    sampler.shutdown()
