import torch

import quiver
from quiver import trace


def test_trace_scopes_collect_stats():
    trace.enable(True)
    trace.reset()
    topo = quiver.CSRTopo(torch.tensor([[0, 1, 2], [1, 2, 0]]))
    s = quiver.GraphSageSampler(topo, [2], mode="CPU")
    s.sample(torch.tensor([0, 1]))
    st = trace.stats()
    assert "sampler.sample_layer" in st
    assert "sampler.reindex" in st
    assert st["sampler.sample_layer"]["count"] == 1
    assert trace.report()
    trace.enable(False)


def test_trace_disabled_is_noop():
    trace.enable(False)
    trace.reset()
    with trace.trace_scope("x"):
        pass
    assert trace.stats() == {}
