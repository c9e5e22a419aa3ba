"""Prefetching sample+gather pipeline.

Overlaps hop sampling and feature gathering of batch i+1..i+depth with
model compute of batch i — WITHOUT a worker thread.  A two-thread design
(v1) serialized in practice: the producer thread was GIL-starved by the
main thread's kernel-launch loop, so sampling for batch i+1 only started
after compute for batch i had finished (kernel-trace evidence in
profiles/).  Here the main thread drives everything on one side HIP
stream: right after the training step for batch i is LAUNCHED (async),
it produces batch i+depth — the one host sync inside the fused sampler
(`sample_hops` sizes readback) blocks the CPU while the GPU concurrently
executes compute i on the main stream and sampling on the side stream.

Usage:
    loader = quiver.TrainingPrefetcher(sampler, feature, batches, depth=2)
    for n_id, batch_size, adjs, x in loader:
        out = model(x, adjs)
        ...
"""
from collections import deque

import torch

__all__ = ["TrainingPrefetcher"]


class TrainingPrefetcher:
    def __init__(self, sampler, feature, seed_batches, depth=2, device=None):
        self.sampler = sampler
        self.feature = feature
        self.seed_batches = seed_batches
        self.depth = max(1, depth)
        self.device = device if device is not None \
            else torch.cuda.current_device()

    def __iter__(self):
        cur = torch.cuda.current_stream(self.device)
        side = torch.cuda.Stream(self.device)
        it = iter(self.seed_batches)
        pending = deque()

        def produce():
            try:
                seeds = next(it)
            except StopIteration:
                return False
            with torch.cuda.stream(side):
                n_id, bs, adjs = self.sampler.sample(seeds)
                x = self.feature[n_id] if self.feature is not None else None
                ev = torch.cuda.Event()
                ev.record(side)
            pending.append((n_id, bs, adjs, x, ev))
            return True

        for _ in range(self.depth):
            if not produce():
                break
        while pending:
            n_id, bs, adjs, x, ev = pending.popleft()
            # main stream waits for the side stream's work for THIS batch...
            cur.wait_event(ev)
            # ...and side-stream allocations must not be reused until
            # main-stream work on them completes
            n_id.record_stream(cur)
            for adj in adjs:
                adj.edge_index.record_stream(cur)
            if x is not None:
                x.record_stream(cur)
            # consumer launches the training step for this batch inside the
            # yield; on re-entry we produce the next batch so its sample
            # sync overlaps that compute
            yield n_id, bs, adjs, x
            produce()
