"""Prefetching sample+gather pipeline.

Overlaps hop sampling and feature gathering of batch i+1 with model
compute of batch i: a worker thread drives the sampler and feature store
on its own HIP stream (the native calls release the GIL, so the main
thread keeps launching model kernels).  The reference runs these stages
sequentially; on MI355X the sampler's small kernels and the gather's
PCIe-bound cold reads coexist well with the model's GEMMs.

Usage:
    loader = quiver.TrainingPrefetcher(sampler, feature, batches, depth=2)
    for n_id, batch_size, adjs, x in loader:
        out = model(x, adjs)
        ...
"""
import queue
import threading

import torch

__all__ = ["TrainingPrefetcher"]


class _End:
    pass


class TrainingPrefetcher:
    def __init__(self, sampler, feature, seed_batches, depth=2, device=None):
        self.sampler = sampler
        self.feature = feature
        self.seed_batches = seed_batches
        self.depth = depth
        self.device = device if device is not None \
            else torch.cuda.current_device()

    def __iter__(self):
        q = queue.Queue(maxsize=self.depth)
        stop = threading.Event()
        stream = torch.cuda.Stream(self.device)

        def worker():
            try:
                with torch.cuda.stream(stream):
                    for seeds in self.seed_batches:
                        if stop.is_set():
                            break
                        n_id, bs, adjs = self.sampler.sample(seeds)
                        x = self.feature[n_id] if self.feature is not None \
                            else None
                        ev = torch.cuda.Event()
                        ev.record(stream)
                        q.put((n_id, bs, adjs, x, ev))
            except BaseException as e:  # propagate to consumer
                q.put(e)
                return
            q.put(_End())

        t = threading.Thread(target=worker, daemon=True)
        t.start()
        try:
            cur = torch.cuda.current_stream(self.device)
            while True:
                item = q.get()
                if isinstance(item, _End):
                    break
                if isinstance(item, BaseException):
                    raise item
                n_id, bs, adjs, x, ev = item
                # main stream waits for the side stream's work...
                cur.wait_event(ev)
                # ...and the producer-stream allocations must not be reused
                # until main-stream work on them completes
                n_id.record_stream(cur)
                for adj in adjs:
                    adj.edge_index.record_stream(cur)
                if x is not None:
                    x.record_stream(cur)
                yield n_id, bs, adjs, x
        finally:
            stop.set()
            # drain so the worker can exit
            try:
                while True:
                    q.get_nowait()
            except queue.Empty:
                pass
            t.join(timeout=10)
