"""Lightweight tracing / per-component stats.

Parity with the reference's opt-in stdtracer scopes (torch-quiver
srcs/cpp/include/quiver/trace.hpp + serving.py latency accounting),
re-done the ROCm way: python scope timers that aggregate into a stats
registry, enabled with QUIVER_TRACE=1, and per-scope GPU sync option so
spans mean what they say.  For kernel-level profiles use rocprofv3; these
scopes are for pipeline-stage accounting (sample / gather / forward /
allreduce).
"""
import os
import time
from collections import defaultdict
from contextlib import contextmanager

import torch

_ENABLED = os.environ.get("QUIVER_TRACE", "0") not in ("0", "", "false")
_stats = defaultdict(lambda: [0, 0.0, 0.0])  # name -> [count, total_s, max_s]


def enabled():
    return _ENABLED


def enable(flag=True):
    global _ENABLED
    _ENABLED = flag


@contextmanager
def trace_scope(name, sync_gpu=False):
    if not _ENABLED:
        yield
        return
    if sync_gpu and torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    try:
        yield
    finally:
        if sync_gpu and torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        s = _stats[name]
        s[0] += 1
        s[1] += dt
        s[2] = max(s[2], dt)


def stats():
    return {k: dict(count=v[0], total_s=v[1], avg_ms=v[1] / max(v[0], 1) * 1e3,
                    max_ms=v[2] * 1e3) for k, v in _stats.items()}


def reset():
    _stats.clear()


def report():
    out = []
    for name, s in sorted(stats().items(), key=lambda kv: -kv[1]["total_s"]):
        out.append(f"{name:30s} n={s['count']:-6d} avg={s['avg_ms']:8.3f}ms "
                   f"max={s['max_ms']:8.3f}ms total={s['total_s']:8.3f}s")
    text = "\n".join(out)
    if text:
        print(text, flush=True)
    return text
