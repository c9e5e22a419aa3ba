"""Serving pipeline on a GPU-less host: RequestBatcher -> HybridSampler
(CPU pool) -> InferenceServer with device_list=['cpu']."""
import numpy as np
import torch
import torch.multiprocessing as mp
import pytest

import quiver
from quiver.nn import GraphSAGE
from quiver.serving import _Stop


@pytest.fixture
def serving_graph():
    g = torch.Generator().manual_seed(0)
    n = 100
    src = torch.randint(0, n, (800,), generator=g)
    dst = torch.randint(0, n, (800,), generator=g)
    topo = quiver.CSRTopo(torch.stack([src, dst]), node_count=n)
    x = torch.randn(n, 8, generator=g)
    return topo, x


@pytest.mark.timeout(120)
def test_cpu_serving_pipeline(serving_graph, tmp_path):
    topo, x = serving_graph
    model = GraphSAGE(8, 16, 4, num_layers=2, dropout=0.0)
    model_path = str(tmp_path / "model.pt")
    torch.save(model, model_path)

    stream_queues = [mp.get_context("spawn").Queue()]
    batcher = quiver.RequestBatcher(device_num=1,
                                    stream_queue_list=stream_queues,
                                    input_proc_per_device=1,
                                    sample_mode="CPU")
    hybrid = quiver.HybridSampler(topo, [4, 4], device_num=1,
                                  worker_num_per_device=1,
                                  batched_queue_list=
                                  batcher.batched_request_queue_list())
    hybrid.start()

    server = quiver.InferenceServer(model_path, ["cpu"], x,
                                    hybrid.sampled_request_queue_list(),
                                    sample_mode="CPU", csr_topo=topo,
                                    sizes=[4, 4], proc_num_per_device=1)
    server.start(join=False)

    n_req = 5
    for i in range(n_req):
        stream_queues[0].put(torch.arange(i * 10, i * 10 + 10))
    results = []
    out_q = server.result_queue_list()[0]
    for _ in range(n_req):
        results.append(out_q.get(timeout=60))
    batcher.stop()
    assert len(results) == n_req
    for r in results:
        assert r.shape == (10, 4)
        assert torch.isfinite(r).all()


def test_auto_routing_threshold(tmp_path):
    # neighbour_num predicts work; batches above threshold go to GPU queue
    neighbour = np.array([1000] * 10 + [1] * 90)
    path = str(tmp_path / "nbr.npy")
    np.save(path, neighbour)

    stream_queues = [mp.get_context("spawn").Queue()]
    batcher = quiver.RequestBatcher(device_num=1,
                                    stream_queue_list=stream_queues,
                                    input_proc_per_device=1,
                                    sample_mode="Auto", threshold=800,
                                    neighbour_path=path)
    cpu_qs, gpu_qs = batcher.batched_request_queue_list()
    stream_queues[0].put(np.array([0, 1]))       # work 2000 -> GPU
    stream_queues[0].put(np.array([50, 51]))     # work 2 -> CPU
    heavy = gpu_qs[0].get(timeout=30)
    light = cpu_qs[0].get(timeout=30)
    assert list(heavy) == [0, 1]
    assert list(light) == [50, 51]
    batcher.stop()
