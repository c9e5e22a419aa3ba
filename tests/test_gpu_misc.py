"""GPU tests for the disk (mmap) tier, p2p feature policy on one GPU, and
the serving pipeline with a GPU inference worker."""
import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

import quiver
from quiver.nn import GraphSAGE

pytestmark = pytest.mark.gpu


def test_feature_disk_tier(tmp_path):
    n, dim = 1000, 16
    full = torch.randn(n, dim)
    # rows 0..599 in memory, 600..999 on disk (mmap)
    mem_rows, disk_rows = 600, 400
    np.save(tmp_path / "disk.npy", full.numpy())
    disk_map = torch.full((n,), -1, dtype=torch.long)
    disk_map[:mem_rows] = torch.arange(mem_rows)

    feature = quiver.Feature(0, device_list=[0], device_cache_size="8K")
    feature.from_cpu_tensor(full[:mem_rows].clone())
    feature.set_mmap_file(str(tmp_path / "disk.npy"), disk_map)

    idx = torch.cat([torch.arange(500, 700), torch.arange(900, 1000)])
    got = feature[idx.cuda()].cpu()
    assert torch.allclose(got, full[idx])


def test_p2p_clique_policy_single_gpu(small_graph):
    indptr, indices = small_graph
    topo = quiver.CSRTopo(indptr=indptr, indices=indices)
    n = topo.node_count
    feat = torch.randn(n, 32)
    quiver.init_p2p([0])
    feature = quiver.Feature(0, device_list=[0], device_cache_size="16K",
                             cache_policy="p2p_clique_replicate",
                             csr_topo=topo)
    feature.from_cpu_tensor(feat.clone())
    idx = torch.randperm(n)[:128]
    assert torch.equal(feature[idx.cuda()].cpu(), feat[idx])


@pytest.mark.timeout(300)
def test_serving_gpu_pipeline(tmp_path):
    g = torch.Generator().manual_seed(0)
    n = 5000
    src = torch.randint(0, n, (50000,), generator=g)
    dst = torch.randint(0, n, (50000,), generator=g)
    topo = quiver.CSRTopo(torch.stack([src, dst]), node_count=n)
    x = torch.randn(n, 16, generator=g)
    model_path = str(tmp_path / "m.pt")
    torch.save(GraphSAGE(16, 32, 4, num_layers=2, dropout=0.0), model_path)

    stream_queues = [mp.get_context("spawn").Queue()]
    batcher = quiver.RequestBatcher(device_num=1,
                                    stream_queue_list=stream_queues,
                                    input_proc_per_device=1,
                                    sample_mode="GPU")
    hybrid = quiver.HybridSampler(topo, [5, 5], device_num=1,
                                  worker_num_per_device=1,
                                  batched_queue_list=
                                  batcher.batched_request_queue_list())
    hybrid.start()
    server = quiver.InferenceServer(model_path, [0], x,
                                    hybrid.sampled_request_queue_list(),
                                    sample_mode="GPU", csr_topo=topo,
                                    sizes=[5, 5], proc_num_per_device=1,
                                    uva_gpu="GPU")
    server.start(join=False)
    for i in range(4):
        stream_queues[0].put(torch.arange(i * 16, i * 16 + 16))
    out_q = server.result_queue_list()[0]
    results = [out_q.get(timeout=120) for _ in range(4)]
    batcher.stop()
    for r in results:
        assert r.shape == (16, 4)
        assert torch.isfinite(r).all()


def test_feature_from_mmap_device_config(tmp_path):
    """Pre-partitioned load path (reference Feature.from_mmap +
    DeviceConfig): rows selected from an mmap-ed numpy array by explicit
    per-device index tensors; local order set afterwards."""
    n, dim = 2000, 32
    full = np.random.default_rng(0).standard_normal((n, dim)).astype(
        np.float32)
    np.save(tmp_path / "feat.npy", full)
    mm = np.load(tmp_path / "feat.npy", mmap_mode="r")

    # non-identity placement: middle rows hot on GPU, the rest pinned host
    hot = torch.arange(1000, 1500)
    cold = torch.cat([torch.arange(0, 1000), torch.arange(1500, 2000)])
    feature = quiver.Feature(0, device_list=[0], device_cache_size=0)
    feature.from_mmap(mm, quiver.DeviceConfig({0: hot}, cold))
    # local order: feature row i of the store = global id order[i]
    feature.set_local_order(torch.cat([hot, cold]))

    idx = torch.randint(0, n, (777,))
    got = feature[idx.cuda()].cpu()
    assert torch.allclose(got, torch.from_numpy(full[idx.numpy()]))
