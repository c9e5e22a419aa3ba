"""GNN serving pipeline: request batching -> hybrid CPU/GPU sampling ->
inference workers, connected by multiprocessing queues.

Capability parity with reference quiver/serving.py (RequestBatcher,
HybridSampler, InferenceServer, InferenceServer_Debug).  Differences:
  - workers stop cleanly on a _Stop sentinel (reference relies on
    daemon-kill / queue timeouts);
  - device_list entries may be "cpu" so the whole pipeline is testable on a
    GPU-less host (model then runs on CPU with the CPU sampler).
"""
import os
import time

import numpy as np
import torch
import torch.multiprocessing as mp

from .pyg import GraphSageSampler

__all__ = ["RequestBatcher", "HybridSampler", "InferenceServer",
           "InferenceServer_Debug"]

# All pipeline queues come from the spawn context: a fork-context mp.Queue
# passed into an mp.spawn child NEVER delivers (the child blocks in get()
# silently).  Spawn-context queues work for both forked (batcher/sampler)
# and spawned (inference) processes.
_ctx = mp.get_context("spawn")


class _Stop(object):
    pass


def _enable_stack_dump():
    """Worker-process init.

    1. SIGUSR1 -> dump this worker's python stacks to stderr (hang
       diagnosis: `kill -USR1 <pid>` or bench_serving's timeout handler).
    2. CPU-tensor sharing strategy -> file_system: queue payloads here go
       through mp.Manager proxies, and the default fd-passing strategy
       accumulates one fd in the manager process per shared tensor — at
       serving request rates that exhausts the fd limit and deadlocks the
       manager (observed as workers stuck in storage._share_fd_cpu_).
    """
    try:
        import faulthandler
        import signal
        faulthandler.register(signal.SIGUSR1)
    except (ImportError, AttributeError, ValueError):
        pass
    try:
        mp.set_sharing_strategy("file_system")
    except (RuntimeError, ValueError):
        pass


def _pin(cpu_range, idx):
    if cpu_range:
        try:
            os.sched_setaffinity(0, [cpu_range[idx % len(cpu_range)]])
        except OSError:
            pass


class RequestBatcher(object):
    """Routes incoming node-id request batches to per-device CPU or GPU
    queues; in Auto mode by predicted sampling work
    sum(neighbour_num[batch]) vs threshold."""

    def __init__(self, device_num, stream_queue_list, input_proc_per_device,
                 sample_mode="GPU", request_mode="CPU", threshold=800,
                 batch_time_limit=10, fixed_batch_size=512,
                 neighbour_path=None, cpu_range=[]):
        # direct (spawn-context) queues, NOT mp.Manager().Queue(): tensor
        # payloads through a manager proxy funnel every put/get through the
        # manager server process, which wedges under serving request rates.
        # Direct queues pass the shm handle endpoint-to-endpoint like torch
        # DataLoader workers.
        # bounded: overload applies backpressure to the stream queues
        # instead of building an unbounded in-flight backlog that makes
        # shutdown drain time unbounded
        self.cpu_batched_queue_list = [_ctx.Queue(maxsize=256)
                                       for _ in range(device_num)]
        self.gpu_batched_queue_list = [_ctx.Queue(maxsize=256)
                                       for _ in range(device_num)]
        self.stream_queue_list = stream_queue_list
        self.sample_mode = sample_mode
        self.device_num = device_num
        self.batch_time_limit = batch_time_limit / 1000
        self.threshold = threshold
        self.fixed_batch_size = fixed_batch_size
        self.neighbour_path = neighbour_path
        self.request_mode = request_mode
        self.cpu_range = cpu_range
        self.procs = []
        target = (self.auto_despatch if sample_mode == "Auto"
                  else self.fixed_despatch)
        for i in range(input_proc_per_device * device_num):
            p = mp.Process(target=target, args=(i,), daemon=True)
            p.start()
            self.procs.append(p)

    def fixed_despatch(self, idx):
        _enable_stack_dump()
        _pin(self.cpu_range, idx)
        torch.set_num_threads(1)  # routing is numpy-light; don't spawn pools
        stream_queue = self.stream_queue_list[idx]
        if self.sample_mode == "CPU":
            batched_queue = self.cpu_batched_queue_list[idx % self.device_num]
        else:
            batched_queue = self.gpu_batched_queue_list[idx % self.device_num]
        while True:
            item = stream_queue.get()
            if isinstance(item, _Stop):
                batched_queue.put(item)
                break
            # arrival timestamp rides with the request so downstream p99
            # includes queue wait (honest end-to-end latency)
            batched_queue.put((item, time.perf_counter()))

    def auto_despatch(self, idx):
        _enable_stack_dump()
        _pin(self.cpu_range, idx)
        torch.set_num_threads(1)
        stream_queue = self.stream_queue_list[idx]
        gpu_q = self.gpu_batched_queue_list[idx % self.device_num]
        cpu_q = self.cpu_batched_queue_list[idx % self.device_num]
        neighbour_num = np.load(self.neighbour_path)
        while True:
            item = stream_queue.get()
            if isinstance(item, _Stop):
                gpu_q.put(item)
                cpu_q.put(item)
                break
            if self.request_mode == "Preparation":
                gpu_q.put((item, time.perf_counter()))
                cpu_q.put((item, time.perf_counter()))
                continue
            predicted_work = np.take(neighbour_num, np.asarray(item)).sum()
            (gpu_q if predicted_work > self.threshold else cpu_q).put(
                (item, time.perf_counter()))

    def batched_request_queue_list(self):
        return [self.cpu_batched_queue_list, self.gpu_batched_queue_list]

    def stop(self):
        for q in self.stream_queue_list:
            q.put(_Stop())


class HybridSampler(object):
    """Pool of CPU sampler processes feeding per-device sampled queues."""

    def __init__(self, csr_topo, sizes, device_num, worker_num_per_device,
                 batched_queue_list, cpu_range=[]):
        self.csr_topo = csr_topo
        self.csr_topo.share_memory_()
        self.cpu_range = cpu_range
        self.device_num = device_num
        self.cpu_num_workers = device_num * worker_num_per_device
        self.sizes = sizes
        self.cpu_batched_queue_list = batched_queue_list[0]
        self.gpu_batched_queue_list = batched_queue_list[1]
        self.cpu_sampled_queue_list = [_ctx.Queue(maxsize=256)
                                       for _ in range(device_num)]
        self.procs = []

    def start(self):
        for i in range(self.cpu_num_workers):
            p = mp.Process(target=self.cpu_sampler_worker_loop,
                           args=(i, self.cpu_batched_queue_list,
                                 self.cpu_sampled_queue_list,
                                 self.device_num, self.sizes, self.csr_topo),
                           daemon=True)
            p.start()
            self.procs.append(p)

    def cpu_sampler_worker_loop(self, rank, sample_task_queue_list,
                                result_queue_list, device_num, sizes,
                                csr_topo):
        _enable_stack_dump()
        _pin(self.cpu_range, rank)
        # single-threaded: a 64-seed request samples in ~3 ms on one core,
        # while worker_count x default-OMP-pool oversubscription was the
        # round-1 Auto-mode ~100 ms p99 (8 workers x 8+ threads thrashing)
        torch.set_num_threads(1)
        cpu_sampler = GraphSageSampler(csr_topo, sizes, device="cpu",
                                       mode="CPU")
        task_queue = sample_task_queue_list[rank % device_num]
        result_queue = result_queue_list[rank % device_num]
        while True:
            start = time.perf_counter()
            item = task_queue.get()
            if isinstance(item, _Stop):
                task_queue.put(item)  # let sibling workers see it too
                result_queue.put(item)
                break
            ids, t_arrival = item if isinstance(item, tuple) else (item,
                                                                   start)
            res = cpu_sampler.sample(ids)
            result_queue.put((res, time.perf_counter() - start, t_arrival))

    def sampled_request_queue_list(self):
        return [self.cpu_sampled_queue_list, self.gpu_batched_queue_list]


def _drain_after_stop(q, stop_item):
    """The cpu_sampled queue has multiple producers; a _Stop from one
    sampler worker can overtake results another worker already queued.
    After seeing _Stop, keep consuming briefly so queued results are not
    dropped; keep the sentinel circulating for sibling consumers."""
    import queue as _queue
    stops = 0
    while stops < 5:
        try:
            item = q.get(timeout=0.5)
        except _queue.Empty:
            break
        if isinstance(item, _Stop):
            # consume (don't re-queue) sentinels while draining: re-putting
            # on every sighting made extra sentinels accumulate in the
            # shared queue.  Exactly one goes back at the end.
            stops += 1
            time.sleep(0.05)
            continue
        yield item
    q.put(stop_item)


def _feature_rows(feature, n_id, device):
    """Index the feature store with a frontier that may live on another
    device (GPU sampler output vs. plain CPU tensor store and vice versa)."""
    if isinstance(feature, torch.Tensor):
        if feature.device != n_id.device:
            n_id = n_id.to(feature.device)
        return feature[n_id].to(device)
    return feature[n_id].to(device)


def _load_model(model_path, device):
    if isinstance(model_path, torch.nn.Module):
        return model_path.to(device)
    return torch.load(model_path, weights_only=False).to(device)


def _resolve_device(device_list, rank):
    d = device_list[rank % len(device_list)]
    if d == "cpu":
        return "cpu", None
    return f"cuda:{d}", d


class InferenceServer(object):
    """Per-GPU inference workers: GPU workers sample+infer, CPU-fed workers
    infer batches pre-sampled by HybridSampler."""

    def __init__(self, model_path, device_list, x_feature, task_queue_list,
                 sample_mode, csr_topo, sizes, ignord_length=100,
                 proc_num_per_device=0, uva_gpu="GPU"):
        self.cpu_sampled_queue_list = task_queue_list[0]
        self.model_path = model_path
        self.device_list = device_list
        self.x_feature = x_feature
        self.gpu_task_queue_list = task_queue_list[1]
        self.sample_mode = sample_mode
        self.csr_topo = csr_topo
        self.sizes = sizes
        self.ignord_length = ignord_length
        self.proc_num_per_device = proc_num_per_device
        self.uva_gpu = uva_gpu
        self.num_proc = len(self.device_list) * self.proc_num_per_device
        self.output_queue_list = [_ctx.Queue()
                                  for _ in range(self.num_proc)]
        # workers announce themselves here once model+sampler are warm;
        # callers can gate offered load on wait_ready() so cold-start
        # (model load, first CUDA context) doesn't become a request backlog
        self.ready_queue = _ctx.Queue()

    def wait_ready(self, timeout=180.0):
        """Block until every inference worker finished warm-up."""
        import queue as _queue
        deadline = time.perf_counter() + timeout
        seen = 0
        while seen < self.num_proc:
            try:
                self.ready_queue.get(
                    timeout=max(0.1, deadline - time.perf_counter()))
                seen += 1
            except _queue.Empty:
                break
        return seen

    def start(self, join=True):
        self.spawn_ctx = mp.spawn(self.run,
                 args=(self.device_list, self.cpu_sampled_queue_list,
                       self.model_path, self.x_feature,
                       self.gpu_task_queue_list, self.sample_mode,
                       self.csr_topo, self.sizes, self.num_proc, self.uva_gpu,
                       self.output_queue_list),
                 nprocs=self.num_proc, join=join)

    def run(self, rank, device_list, cpu_sampled_queue_list, model_path,
            feature, gpu_sample_task_queue_list, sample_mode, csr_topo, sizes,
            num_proc, uva_gpu, output_queue_list):
        _enable_stack_dump()
        try:
            self._run(rank, device_list, cpu_sampled_queue_list, model_path,
                      feature, gpu_sample_task_queue_list, sample_mode,
                      csr_topo, sizes, num_proc, uva_gpu, output_queue_list)
        except Exception:
            # started with join=False the spawn context is never joined, so
            # surface worker failures on stderr instead of dying silently
            import traceback
            traceback.print_exc()
            raise

    def _run(self, rank, device_list, cpu_sampled_queue_list, model_path,
             feature, gpu_sample_task_queue_list, sample_mode, csr_topo,
             sizes, num_proc, uva_gpu, output_queue_list):
        output_queue = output_queue_list[rank]
        if sample_mode == "Auto":
            if rank < num_proc // 2:
                self.gpu_sampler_inference_loop(
                    rank, device_list, feature, gpu_sample_task_queue_list,
                    model_path, csr_topo, sizes, uva_gpu, output_queue)
            else:
                self.cpu_sampler_inference_loop(
                    rank - num_proc // 2, device_list, feature,
                    cpu_sampled_queue_list, model_path, output_queue)
        elif sample_mode == "GPU":
            self.gpu_sampler_inference_loop(
                rank, device_list, feature, gpu_sample_task_queue_list,
                model_path, csr_topo, sizes, uva_gpu, output_queue)
        else:
            self.cpu_sampler_inference_loop(rank, device_list, feature,
                                            cpu_sampled_queue_list,
                                            model_path, output_queue)

    def gpu_sampler_inference_loop(self, rank, device_list, feature,
                                   gpu_sample_task_queue_list, model_path,
                                   csr_topo, sizes, sample_mode,
                                   output_queue):
        device, dev_id = _resolve_device(device_list, rank)
        q = gpu_sample_task_queue_list[rank % len(device_list)]
        mode = "CPU" if device == "cpu" else sample_mode
        sampler = GraphSageSampler(csr_topo, sizes,
                                   device=(dev_id if dev_id is not None
                                           else "cpu"), mode=mode)
        model = _load_model(model_path, device)
        model.eval()
        self._warmup_and_signal(sampler, feature, model, device)
        with torch.no_grad():
            while True:
                item = q.get()
                if isinstance(item, _Stop):
                    q.put(item)
                    output_queue.put(item)
                    break
                ids = item[0] if isinstance(item, tuple) else item
                sample_task = torch.as_tensor(ids)
                n_id, batch_size, adjs = sampler.sample(sample_task)
                adjs = [adj.to(device) for adj in adjs]
                x_input = _feature_rows(feature, n_id, device)
                out = model(x_input, adjs)
                output_queue.put(out.cpu())

    def _warmup_and_signal(self, sampler, feature, model, device):
        """First-touch init (CUDA context, sampler engine, feature IPC)
        before announcing readiness."""
        try:
            with torch.no_grad():
                if sampler is not None:
                    n_id, _, adjs = sampler.sample(torch.tensor([0, 1]))
                    adjs = [adj.to(device) for adj in adjs]
                    model(_feature_rows(feature, n_id, device), adjs)
                else:
                    _feature_rows(feature,
                                  torch.tensor([0], dtype=torch.long),
                                  device)
                if device != "cpu":
                    torch.cuda.synchronize()
        except Exception:  # noqa: BLE001 - warm-up is best-effort
            pass
        self.ready_queue.put(os.getpid())

    def cpu_sampler_inference_loop(self, rank, device_list, feature,
                                   cpu_sampled_queue_list, model_path,
                                   output_queue):
        device, _ = _resolve_device(device_list, rank)
        q = cpu_sampled_queue_list[rank % len(device_list)]
        model = _load_model(model_path, device)
        model.eval()
        self._warmup_and_signal(None, feature, model, device)
        def infer(item):
            n_id, batch_size, adjs = item[0]
            adjs = [adj.to(device) for adj in adjs]
            x_input = _feature_rows(feature, n_id, device)
            out = model(x_input, adjs)
            output_queue.put(out.cpu())  # item[2] (arrival) used by Debug

        with torch.no_grad():
            while True:
                item = q.get()
                if isinstance(item, _Stop):
                    for extra in _drain_after_stop(q, item):
                        infer(extra)
                    output_queue.put(item)
                    break
                infer(item)

    def result_queue_list(self):
        return self.output_queue_list


class InferenceServer_Debug(InferenceServer):
    """InferenceServer variant with avg/p99 latency + throughput accounting
    (reference serving.py:236-360)."""

    def __init__(self, model_path, device_list, x_feature, task_queue_list,
                 sample_mode, csr_topo, sizes, ignord_length=100,
                 result_path=None, exp_id=0, proc_num_per_device=0,
                 uva_gpu="GPU"):
        super().__init__(model_path, device_list, x_feature, task_queue_list,
                         sample_mode, csr_topo, sizes, ignord_length,
                         proc_num_per_device, uva_gpu)
        self.result_path = result_path
        self.exp_id = exp_id

    @staticmethod
    def _report(result, ignord_length, tag, rank, res_path, exp_id):
        if not result:
            return None
        arr = np.array(result[ignord_length:] if
                       len(result) > ignord_length else result)
        lat = arr[:, 2] - arr[:, 0]
        avg_latency = np.average(lat, axis=0, weights=arr[:, 3])
        tp99_latency = np.percentile(lat, 99, axis=0) * 1000
        # split: queue wait (arrival -> worker pickup) vs service
        # (pickup -> done) — tells congestion apart from slow work
        wait = arr[:, 1] - arr[:, 0]
        serve = arr[:, 2] - arr[:, 1]
        span = max(np.max(arr[:, 2]) - np.min(arr[:, 2]), 1e-9)
        throughput = np.sum(arr[:, 3]) / span
        total = np.sum(arr[:, 3])
        print(f"{tag} Rank {rank}: Avg Latency: {avg_latency}, "
              f"TP99 Latency: {tp99_latency} ms, Throughput: {throughput}, "
              f"Total: {total} "
              f"(avg wait {np.mean(wait)*1e3:.2f} ms / "
              f"serve {np.mean(serve)*1e3:.2f} ms)", flush=True)
        if res_path is not None:
            np.save(os.path.join(res_path, f"{exp_id}_{tag}_{rank}"), arr)
        return dict(avg_latency=float(avg_latency),
                    tp99_latency_ms=float(tp99_latency),
                    throughput=float(throughput), total=float(total),
                    avg_wait_ms=float(np.mean(wait) * 1e3),
                    avg_serve_ms=float(np.mean(serve) * 1e3))

    def gpu_sampler_inference_loop(self, rank, device_list, feature,
                                   gpu_sample_task_queue_list, model_path,
                                   csr_topo, sizes, sample_mode,
                                   output_queue):
        device, dev_id = _resolve_device(device_list, rank)
        q = gpu_sample_task_queue_list[rank % len(device_list)]
        mode = "CPU" if device == "cpu" else sample_mode
        sampler = GraphSageSampler(csr_topo, sizes,
                                   device=(dev_id if dev_id is not None
                                           else "cpu"), mode=mode)
        model = _load_model(model_path, device)
        model.eval()
        self._warmup_and_signal(sampler, feature, model, device)
        result = []
        with torch.no_grad():
            while True:
                item = q.get()
                if isinstance(item, _Stop):
                    q.put(item)
                    stats = self._report(result, self.ignord_length, "GPU",
                                         rank, self.result_path, self.exp_id)
                    output_queue.put(stats if stats is not None else item)
                    break
                pickup = time.perf_counter()
                ids, start_time = (item if isinstance(item, tuple)
                                   else (item, pickup))
                sample_task = torch.as_tensor(ids)
                n_id, batch_size, adjs = sampler.sample(sample_task)
                sample_time = time.perf_counter()
                adjs = [adj.to(device) for adj in adjs]
                x_input = _feature_rows(feature, n_id, device)
                out = model(x_input, adjs)
                if device != "cpu":
                    torch.cuda.synchronize()
                end_time = time.perf_counter()
                result.append([start_time, sample_time, end_time, batch_size,
                               n_id.shape[0]])

    def cpu_sampler_inference_loop(self, rank, device_list, feature,
                                   cpu_sampled_queue_list, model_path,
                                   output_queue):
        device, _ = _resolve_device(device_list, rank)
        q = cpu_sampled_queue_list[rank % len(device_list)]
        model = _load_model(model_path, device)
        model.eval()
        self._warmup_and_signal(None, feature, model, device)
        result = []

        def infer(item):
            pickup = time.perf_counter()
            n_id, batch_size, adjs = item[0]
            start_time = item[2] if len(item) > 2 else pickup
            adjs = [adj.to(device) for adj in adjs]
            x_input = _feature_rows(feature, n_id, device)
            out = model(x_input, adjs)
            if device != "cpu":
                torch.cuda.synchronize()
            end_time = time.perf_counter()
            result.append([start_time, pickup, end_time, batch_size,
                           n_id.shape[0]])

        with torch.no_grad():
            while True:
                item = q.get()
                if isinstance(item, _Stop):
                    for extra in _drain_after_stop(q, item):
                        infer(extra)
                    stats = self._report(result, self.ignord_length, "CPU",
                                         rank, self.result_path, self.exp_id)
                    output_queue.put(stats if stats is not None else item)
                    break
                infer(item)
