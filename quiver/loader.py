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
import os
from collections import deque

import torch

__all__ = ["TrainingPrefetcher"]


class _GraphedChain:
    """depth+1 alternating hipGraph captures of the sample->gather chain.

    A captured graph replays the whole per-batch chain (~75 kernel
    launches) as ONE hipGraphLaunch, eliminating per-launch host cost.
    Requirements already guaranteed by the zero-sync chain design: no
    host syncs anywhere (sizes land in pinned memory via an async D2H
    node), upper-bound-sized static buffers, and a DEVICE-resident RNG
    word advanced by a captured bump kernel (a host-read seed would be
    frozen into the graph).

    Each slot owns its own capture + static buffers: batch i and batch
    i+1 are in flight simultaneously, so a single capture would overwrite
    live outputs.  The caller replays slots round-robin and must order
    each replay after the consumer of that slot's previous batch (an
    event on the main stream; see produce()).
    """

    def __init__(self, sampler, feature, seeds_proto, stream, depth):
        self.batch_size = seeds_proto.numel()
        self.slots = []
        self.turn = 0
        device = sampler.device
        H = len(sampler.sizes)
        # depth+1 slots: a replay then only needs the training step of
        # the batch (depth+1) ago to have finished, which is one produce
        # OLDER than the most recent main-stream event — so the replayed
        # chain still overlaps the current batch's compute
        for _ in range(depth + 1):
            static_seeds = seeds_proto.to(device).clone()
            # pinned buffer + its D2H copy stay OUTSIDE the capture:
            # hipHostMalloc (and torch pin_memory allocation) is not a
            # capturable operation
            sizes_pin = torch.empty(2 * H, dtype=torch.int64,
                                    pin_memory=True)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.stream(stream):
                with torch.cuda.graph(g, stream=stream):
                    raw, sizes_dev = sampler.quiver.sample_hops_raw(
                        static_seeds, sampler.sizes)
                    x_ub = None
                    if feature is not None:
                        n_dev = sizes_dev[2 * H - 1:2 * H]
                        x_ub = feature.gather_raw(raw[H - 1][0], n_dev)
            tok = (static_seeds, raw, sizes_dev, sizes_pin)
            self.slots.append((static_seeds, g, tok, x_ub))

    def run(self, seeds):
        """Replay the next slot on the current stream."""
        static_seeds, g, tok, x_ub = self.slots[self.turn]
        self.turn = (self.turn + 1) % len(self.slots)
        static_seeds.copy_(seeds, non_blocking=True)
        g.replay()
        tok[3].copy_(tok[2], non_blocking=True)  # sizes D2H, post-replay
        return tok, x_ub


class TrainingPrefetcher:
    def __init__(self, sampler, feature, seed_batches, depth=2, device=None,
                 num_streams=1):
        self.sampler = sampler
        self.feature = feature
        self.seed_batches = seed_batches
        self.depth = max(1, depth, num_streams)
        self.num_streams = max(1, num_streams)
        self.device = device if device is not None \
            else torch.cuda.current_device()

    def _can_chain_async(self):
        """Zero-sync produce path: fused GPU sampler + directly-accessible
        feature shards (no disk tier, no cross-clique pass)."""
        if not hasattr(self.sampler, "sample_async"):
            return False
        if getattr(self.sampler, "mode", None) not in ("GPU", "UVA"):
            return False
        if getattr(self.sampler, "sort_frontier", False):
            return False
        if self.feature is not None:
            if not hasattr(self.feature, "gather_raw"):
                return False
            if getattr(self.feature, "mmap_handle_", None) is not None:
                return False
            # pre-validate the known unsupported layout (shards needing the
            # cross-clique pass) instead of learning it from a RuntimeError
            # on the first produce
            try:
                st = self.feature._shard_tensor()
                if st._inaccessible_ranges():
                    return False
            except Exception:
                pass  # lazy IPC store: decided on first produce instead
        return True

    def __iter__(self):
        cur = torch.cuda.current_stream(self.device)
        # high-priority streams: the PCIe-latency-bound host gather needs
        # its blocks resident promptly even while model kernels churn CU
        # slots.  num_streams=2 round-robins batches over two streams so
        # one batch's sample phase hides under the other's gather.
        sides = [torch.cuda.Stream(self.device, priority=-1)
                 for _ in range(self.num_streams)]
        it = iter(self.seed_batches)
        pending = deque()
        main_evs = deque()
        chain_async = [self._can_chain_async()]
        # hipGraph-replay the chain (QUIVER_HIPGRAPH=1 opt-in).  Measured
        # on the products bench: capture works but replays ~7% SLOWER
        # than stream launches (4.55 vs 4.24 ms/step, 40 steps) — the
        # step is GPU-bound with CPU slack, so removing launch overhead
        # buys nothing and graph scheduling costs a little.  Kept for
        # launch-bound deployments (small batches / many hops).
        want_graph = (chain_async[0] and self.num_streams == 1
                      and os.environ.get("QUIVER_HIPGRAPH", "0") == "1")
        graphed = [None]
        produced = [0]
        rr = [0]

        def produce():
            try:
                seeds = next(it)
            except StopIteration:
                return False
            idx = rr[0] % len(sides)
            rr[0] += 1
            if chain_async[0] and want_graph and graphed[0] is None \
                    and produced[0] >= 1:
                # one eager batch has warmed the chain: capture (outside
                # the stream context — a failed capture can wedge the
                # stream in capture state, in which case it is replaced)
                try:
                    graphed[0] = _GraphedChain(self.sampler, self.feature,
                                               seeds, sides[idx],
                                               self.depth)
                except Exception as e:  # noqa: BLE001
                    import warnings
                    warnings.warn("TrainingPrefetcher: hipGraph capture "
                                  f"failed ({e}); running uncaptured",
                                  RuntimeWarning)
                    graphed[0] = False
                    sides[idx] = torch.cuda.Stream(self.device,
                                                   priority=-1)
            side = sides[idx]
            # record a main-stream event EVERY produce (graphed or not):
            # replays wait on the previous produce's event, and a stale
            # deque after a mix of eager and graphed batches would weaken
            # that ordering
            if want_graph:
                ev_main = torch.cuda.Event()
                ev_main.record(cur)
                main_evs.append(ev_main)
                if len(main_evs) > 2:
                    main_evs.popleft()
            with torch.cuda.stream(side):
                if chain_async[0] and isinstance(graphed[0], _GraphedChain) \
                        and seeds.numel() == graphed[0].batch_size:
                    # order this slot's replay after the training step of
                    # the batch that previously used this slot's static
                    # buffers.  With depth+1 slots that step was enqueued
                    # by the PREVIOUS produce call, so waiting on the
                    # previous produce's main-stream event (not this
                    # one's) keeps the replay overlapping the current
                    # batch's compute.
                    side.wait_event(main_evs[0])
                    try:
                        tok, x_ub = graphed[0].run(seeds)
                        ev = torch.cuda.Event()
                        ev.record(side)
                        pending.append(("gtok", tok, x_ub, ev))
                        produced[0] += 1
                        return True
                    except RuntimeError as e:
                        import warnings
                        warnings.warn("TrainingPrefetcher: hipGraph replay "
                                      f"failed ({e}); running uncaptured",
                                      RuntimeWarning)
                        graphed[0] = False
                if chain_async[0]:
                    try:
                        tok = self.sampler.sample_async(seeds)
                        x_ub = None
                        if self.feature is not None:
                            raw, sizes_dev = tok[1], tok[2]
                            n_dev = sizes_dev[2 * len(raw) - 1:2 * len(raw)]
                            x_ub = self.feature.gather_raw(raw[-1][0], n_dev)
                        ev = torch.cuda.Event()
                        ev.record(side)
                        pending.append(("tok", tok, x_ub, ev))
                        produced[0] += 1
                        return True
                    except RuntimeError as e:
                        # fall through to the sync path for the rest of the
                        # epoch — but say why, so a transient failure (e.g.
                        # OOM during gather_raw) is not silently converted
                        # into a slower epoch
                        import warnings
                        warnings.warn(
                            "TrainingPrefetcher: zero-sync chain disabled "
                            f"after: {e}", RuntimeWarning)
                        chain_async[0] = False
                n_id, bs, adjs = self.sampler.sample(seeds)
                x = self.feature[n_id] if self.feature is not None else None
                ev = torch.cuda.Event()
                ev.record(side)
            pending.append(("done", (n_id, bs, adjs), x, ev))
            return True

        for _ in range(self.depth):
            if not produce():
                break
        while pending:
            kind, payload, x, ev = pending.popleft()
            # main stream waits for the side stream's work for THIS batch
            cur.wait_event(ev)
            if kind in ("tok", "gtok"):
                # CPU-side wait for THIS batch only (event, not stream):
                # makes the pinned sizes valid; later batches keep running
                ev.synchronize()
                n_id, bs, adjs = self.sampler.sample_finalize(payload)
                if x is not None:
                    x = x[:n_id.size(0)]
            else:
                n_id, bs, adjs = payload
            if kind == "gtok":
                # graph-pool buffers have static lifetime: no allocator
                # bookkeeping, and the next replay of this slot is ordered
                # after this batch's training step by the producer's event
                pass
            else:
                # side-stream allocations must not be reused until
                # main-stream work on them completes
                if kind == "tok":
                    # the edge_index views alias the side-allocated
                    # upper-bound buffers read on the main stream
                    for f_ub, e_ub in payload[1]:
                        e_ub.record_stream(cur)
                n_id.record_stream(cur)
                for adj in adjs:
                    adj.edge_index.record_stream(cur)
                if x is not None:
                    x.record_stream(cur)
            # consumer launches the training step for this batch inside the
            # yield; on re-entry we produce batch i+depth so the whole
            # sample+gather chain overlaps that compute
            yield n_id, bs, adjs, x
            produce()
