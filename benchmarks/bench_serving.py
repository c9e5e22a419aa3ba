#!/usr/bin/env python3
"""Serving benchmark: p99 latency + throughput at fixed request rate
(BASELINE.json config 5 — "Reddit GNN serving path: HybridSampler +
InferenceServer, p99 latency at fixed QPS").

Synthetic Reddit-shaped graph (232,965 nodes / ~114.6M directed edges /
602 features / 41 classes — no network for the real dataset), random-init
2-layer GraphSAGE, fanout [25, 10].

A client process emits single-request node batches into the stream queue
at a fixed rate; the pipeline is
  RequestBatcher -> (HybridSampler CPU pool | GPU queue) -> InferenceServer_Debug
and the debug server reports avg/p99 latency and served throughput
(reference analog: examples/serving/reddit/reddit_serving.py +
serving.py:236-360 stats).

Usage (GPU box):
  python benchmarks/bench_serving.py --mode GPU  --qps 300 --seconds 20
  python benchmarks/bench_serving.py --mode Auto --qps 300 --seconds 20
CPU-only smoke:
  python benchmarks/bench_serving.py --mode CPU --qps 20 --seconds 5 --cpu
"""
import argparse
import json
import os
import sys
import tempfile
import time

import numpy as np
import torch
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# Reddit shape
N_NODES = 232_965
N_EDGES = 114_615_892
FEAT_DIM = 602
N_CLASSES = 41
FANOUT = [25, 10]
HIDDEN = 128


def make_graph(seed, nodes, edges, max_deg=21_000):
    rng = np.random.default_rng(seed)
    raw = rng.pareto(1.4, nodes) + 0.5
    deg = np.maximum((raw * (edges / raw.sum())).astype(np.int64), 1)
    np.clip(deg, 1, max_deg, out=deg)
    deg = -np.sort(-deg)
    indptr = np.zeros(nodes + 1, dtype=np.int64)
    np.cumsum(deg, out=indptr[1:])
    m = int(indptr[-1])
    u = rng.random(m, dtype=np.float32)
    indices = (u * u * nodes).astype(np.int64)
    np.clip(indices, 0, nodes - 1, out=indices)
    return torch.from_numpy(indptr), torch.from_numpy(indices)


def client_loop(stream_queue, qps, seconds, request_size, nodes, seed):
    """Emit `request_size`-node requests at a fixed rate for `seconds`."""
    rng = np.random.default_rng(seed)
    period = 1.0 / qps
    n = int(qps * seconds)
    t0 = time.perf_counter()
    for i in range(n):
        target = t0 + i * period
        now = time.perf_counter()
        if target > now:
            time.sleep(target - now)
        stream_queue.put(rng.integers(0, nodes, request_size,
                                      dtype=np.int64))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--mode", default="GPU", choices=["GPU", "CPU", "Auto"])
    p.add_argument("--qps", type=float, default=300.0,
                   help="requests/second offered load")
    p.add_argument("--seconds", type=float, default=20.0)
    p.add_argument("--request-size", type=int, default=64,
                   help="seed nodes per request")
    p.add_argument("--cpu", action="store_true",
                   help="run everything on CPU (smoke test)")
    p.add_argument("--devices", type=int, nargs="*", default=None,
                   help="GPU ids for inference workers (default: all)")
    p.add_argument("--procs-per-device", type=int, default=None,
               help="inference workers per device (default: scaled from --qps)")
    p.add_argument("--cpu-workers-per-device", type=int,
               default=None,
               help="CPU sampler processes per device (default: scaled from --qps)")
    p.add_argument("--nodes", type=int, default=N_NODES)
    p.add_argument("--edges", type=int, default=N_EDGES)
    p.add_argument("--threshold", type=int, default=None,
                   help="Auto-mode routing threshold on predicted frontier")
    args = p.parse_args()

    import quiver
    from quiver.nn import GraphSAGE

    # size worker pools from offered load: a 64-seed request costs ~3 ms
    # of 1-thread CPU sampling and ~1 ms of GPU inference, so capacity
    # scales linearly in workers; keep utilization under ~50% so queueing
    # tails stay flat (measured: Auto@1000 QPS p99 149 ms with the 250-QPS
    # pool, 3.8 ms with this scaling)
    if args.cpu_workers_per_device is None:
        args.cpu_workers_per_device = max(8, int(args.qps / 80))

    t0 = time.perf_counter()
    indptr, indices = make_graph(0, args.nodes, args.edges)
    csr_topo = quiver.CSRTopo(indptr=indptr, indices=indices)

    if args.cpu:
        device_list = ["cpu"]
        feat_device = None
    else:
        n_dev = torch.cuda.device_count()
        device_list = (args.devices if args.devices is not None
                       else list(range(n_dev)))
        feat_device = device_list[0]
    device_num = len(device_list)

    g = torch.Generator().manual_seed(0)
    feat_cpu = torch.randn(args.nodes, FEAT_DIM, generator=g)
    if args.cpu:
        feature = feat_cpu.share_memory_()
    else:
        feature = quiver.Feature(feat_device, device_list=device_list,
                                 device_cache_size="400M",
                                 cache_policy="device_replicate",
                                 csr_topo=csr_topo)
        feature.from_cpu_tensor(feat_cpu)

    model = GraphSAGE(FEAT_DIM, HIDDEN, N_CLASSES, num_layers=len(FANOUT),
                      dropout=0.0)
    model_path = os.path.join(tempfile.gettempdir(), "bench_serving_model.pt")
    torch.save(model, model_path)

    neighbour_path = None
    if args.mode == "Auto":
        # estimated per-node sampled-frontier size drives Auto routing
        from quiver.generate_neighbour_num import generate_neighbour_num
        neighbour_path = os.path.join(tempfile.gettempdir(),
                                      "bench_serving_nn.npy")
        generate_neighbour_num(args.nodes, csr_topo, FANOUT, neighbour_path,
                               mode="CPU")

    threshold = args.threshold
    if threshold is None:
        # route the heaviest ~25% of requests to the GPU
        sample = np.random.default_rng(1).integers(
            0, args.nodes, (256, args.request_size))
        if neighbour_path is not None:
            nn_est = np.load(neighbour_path)
            work = np.take(nn_est, sample).sum(axis=1)
            threshold = int(np.percentile(work, 75))
        else:
            threshold = 0

    stream_queues = [mp.get_context("spawn").Queue() for _ in range(device_num)]
    batcher = quiver.RequestBatcher(device_num, stream_queues,
                                    input_proc_per_device=1,
                                    sample_mode=args.mode,
                                    request_mode="Serve",
                                    threshold=threshold,
                                    neighbour_path=neighbour_path)
    sampler_pool = None
    queues = batcher.batched_request_queue_list()
    if args.mode in ("CPU", "Auto"):
        sampler_pool = quiver.HybridSampler(
            csr_topo, FANOUT, device_num, args.cpu_workers_per_device,
            queues)
        sampler_pool.start()
        task_queues = sampler_pool.sampled_request_queue_list()
    else:
        task_queues = queues  # [cpu_batched, gpu_batched]

    ppd = args.procs_per_device
    if ppd is None:
        ppd = max(2, int(args.qps / 300) + 1) if args.mode == "Auto" else \
            max(1, int(args.qps / 400))
    if args.mode == "Auto" and ppd < 2:
        ppd = 2  # one GPU-fed + one CPU-fed worker per device
    server = quiver.InferenceServer_Debug(
        model_path, device_list, feature, task_queues, args.mode,
        csr_topo, FANOUT, ignord_length=int(args.qps),  # 1s warm-up ignored
        proc_num_per_device=ppd, uva_gpu="UVA")
    print(f"# setup {time.perf_counter()-t0:.1f}s; offered load "
          f"{args.qps} req/s x {args.seconds}s, request={args.request_size} "
          f"seeds, mode={args.mode}, devices={device_list}", flush=True)

    server.start(join=False)
    ready = server.wait_ready(timeout=180)
    print(f"# {ready}/{server.num_proc} inference workers warm", flush=True)
    clients = []
    for i, sq in enumerate(stream_queues):
        c = mp.Process(target=client_loop,
                       args=(sq, args.qps / device_num, args.seconds,
                             args.request_size, args.nodes, 7 + i),
                       daemon=True)
        c.start()
        clients.append(c)
    deadline = time.time() + args.seconds + 90
    stalled = False
    for c in clients:
        c.join(timeout=max(1.0, deadline - time.time()))
        if c.is_alive():
            stalled = True
    if stalled:
        print("# WARNING: client(s) still blocked flushing the stream "
              "queue after the deadline — pipeline cannot absorb the "
              "offered load; dumping stacks and continuing", flush=True)
    batcher.stop()

    def dump_worker_stacks():
        """Hang diagnosis: SIGUSR1 -> faulthandler stack dump to stderr."""
        import signal
        pids = [p.pid for p in batcher.procs]
        if sampler_pool is not None:
            pids += [p.pid for p in sampler_pool.procs]
        ctx = getattr(server, "spawn_ctx", None)
        if ctx is not None:
            pids += [p.pid for p in ctx.processes]
        for pid in pids:
            try:
                os.kill(pid, signal.SIGUSR1)
            except (OSError, TypeError):
                pass
        time.sleep(3)

    if stalled:
        dump_worker_stacks()

    stats = []
    deadline = time.time() + 120
    for q in server.result_queue_list():
        while time.time() < deadline:
            try:
                item = q.get(timeout=120)
            except Exception:
                print("# TIMEOUT waiting for worker stats — dumping worker "
                      "stacks to stderr", flush=True)
                dump_worker_stacks()
                raise
            if isinstance(item, dict):
                stats.append(item)
                break
            if not isinstance(item, torch.Tensor) and not \
                    isinstance(item, np.ndarray):
                break  # _Stop with no stats (worker served nothing)

    agg_tp = sum(s["throughput"] for s in stats)
    worst_p99 = max((s["tp99_latency_ms"] for s in stats), default=None)
    avg_lat = (sum(s["avg_latency"] * s["total"] for s in stats)
               / max(sum(s["total"] for s in stats), 1) if stats else None)
    print(json.dumps({
        "metric": "serving-p99-latency",
        "p99_ms": worst_p99,
        "avg_latency_s": avg_lat,
        "throughput_seeds_s": agg_tp,
        "offered_qps": args.qps,
        "mode": args.mode,
        "request_size": args.request_size,
        "devices": [str(d) for d in device_list],
        "model": f"graphsage-2L-h{HIDDEN} reddit-shaped synthetic",
        "fanout": FANOUT,
        "per_worker": stats,
    }), flush=True)


if __name__ == "__main__":
    # batcher/client/CPU-sampler procs fork (no CUDA touched in them);
    # InferenceServer always uses the spawn context internally.
    main()
    # skip interpreter-exit finalizers: daemon workers may hold undrained
    # queue items (results enqueued after a _Stop raced past them), and
    # their feeder threads would block a clean exit forever
    sys.stdout.flush()
    sys.stderr.flush()
    os._exit(0)
