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
    assert server.wait_ready(timeout=90) == 1  # warm-up barrier

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
    heavy, _t = gpu_qs[0].get(timeout=30)
    light, _t2 = cpu_qs[0].get(timeout=30)
    assert list(heavy) == [0, 1]
    assert list(light) == [50, 51]
    batcher.stop()


def test_auto_despatch_routing(tmp_path):
    """Threshold routing: heavy requests -> GPU queue, light -> CPU queue.
    Runs the despatcher body synchronously (no worker procs)."""
    import numpy as np
    nn = np.zeros(100, dtype=np.int64)
    nn[50:] = 1000  # nodes >= 50 are predicted-heavy
    path = tmp_path / "nn.npy"
    np.save(path, nn)

    batcher = quiver.RequestBatcher(
        device_num=1, stream_queue_list=[mp.get_context("spawn").Queue()],
        input_proc_per_device=0, sample_mode="Auto", request_mode="Serve",
        threshold=1500, neighbour_path=str(path))
    sq = batcher.stream_queue_list[0]
    light = np.array([1, 2, 3])            # work 0
    heavy = np.array([60, 61])             # work 2000 > 1500
    sq.put(light)
    sq.put(heavy)
    from quiver.serving import _Stop
    sq.put(_Stop())
    batcher.auto_despatch(0)               # runs to the sentinel

    cpu_q = batcher.cpu_batched_queue_list[0]
    gpu_q = batcher.gpu_batched_queue_list[0]
    got_cpu, t_cpu = cpu_q.get(timeout=10)
    got_gpu, t_gpu = gpu_q.get(timeout=10)
    assert list(got_cpu) == [1, 2, 3]
    assert list(got_gpu) == [60, 61]
    assert t_cpu > 0 and t_gpu > 0  # arrival stamps ride with requests
    assert isinstance(gpu_q.get(timeout=10), _Stop)
    assert isinstance(cpu_q.get(timeout=10), _Stop)


def test_drain_after_stop_single_sentinel(tmp_path):
    """_drain_after_stop consumes sentinels while draining and re-queues
    exactly ONE on exit (round-1 advisor finding: sentinels accumulated)."""
    import queue as pyqueue
    from quiver.serving import _drain_after_stop
    q = pyqueue.Queue()
    q.put("a")
    q.put(_Stop())
    q.put("b")
    q.put(_Stop())
    stop = _Stop()
    got = list(_drain_after_stop(q, stop))
    assert got == ["a", "b"]
    leftovers = []
    while not q.empty():
        leftovers.append(q.get_nowait())
    assert len(leftovers) == 1 and isinstance(leftovers[0], _Stop)


def test_auto_despatch_preparation_mode(tmp_path):
    """Preparation mode fans every request to BOTH queues (warm-up path),
    with arrival timestamps attached."""
    import numpy as np
    nn = np.ones(100, dtype=np.int64)
    path = tmp_path / "nn.npy"
    np.save(path, nn)
    batcher = quiver.RequestBatcher(
        device_num=1, stream_queue_list=[mp.get_context("spawn").Queue()],
        input_proc_per_device=0, sample_mode="Auto",
        request_mode="Preparation", threshold=10, neighbour_path=str(path))
    sq = batcher.stream_queue_list[0]
    sq.put(np.array([1, 2]))
    sq.put(_Stop())
    batcher.auto_despatch(0)
    for q in (batcher.cpu_batched_queue_list[0],
              batcher.gpu_batched_queue_list[0]):
        ids, t = q.get(timeout=10)
        assert list(ids) == [1, 2] and t > 0
        assert isinstance(q.get(timeout=10), _Stop)
