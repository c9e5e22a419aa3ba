"""CPU coverage for generate_neighbour_num (workload predictor) and the
ForkingPickler reductions (samplers crossing process boundaries)."""
import torch
import torch.multiprocessing as mp

import quiver
from quiver.generate_neighbour_num import generate_neighbour_num


def _ring_topo(n, k_each=2):
    # ring with k_each forward edges per node -> deterministic degrees
    src, dst = [], []
    for v in range(n):
        for j in range(1, k_each + 1):
            src.append(v)
            dst.append((v + j) % n)
    return quiver.CSRTopo(torch.stack([torch.tensor(src),
                                       torch.tensor(dst)]), node_count=n)


def test_generate_neighbour_num_exact_on_deterministic_graph(tmp_path):
    """Degrees (2) are below every fanout, so sampling is exhaustive and
    the exact per-node frontier size is computable by hand."""
    n = 24
    topo = _ring_topo(n, k_each=2)
    out = tmp_path / "nn.npy"
    res = generate_neighbour_num(n, topo, [4, 4], str(out), mode="CPU",
                                 sample=True)
    # frontier of v: {v} U {v+1,v+2} U {v+2..v+4} -> 5 unique nodes;
    # exact path counts |n_id| of sampler.sample([v])
    assert (res == 5).all(), res[:6]
    assert out.exists()

    est = generate_neighbour_num(n, topo, [4, 4], None, mode="CPU",
                                 sample=False)
    # estimator counts sampled neighbors per hop with duplicates:
    # 2 (hop1) + 4 (hop2) = 6 for every node
    assert (est == 6).all(), est[:6]


def _child_sample(sampler, out_q):
    n_id, bs, adjs = sampler.sample(torch.tensor([0, 1, 2]))
    out_q.put((int(n_id.numel()), int(bs), len(adjs)))


def test_sampler_crosses_process_via_reductions():
    """A CPU GraphSageSampler passed through mp is rebuilt from its IPC
    handle in the child (ForkingPickler reduction) and samples there."""
    topo = _ring_topo(32, k_each=3)
    topo.share_memory_()
    sampler = quiver.GraphSageSampler(topo, [3, 3], mode="CPU")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_child_sample, args=(sampler, q))
    p.start()
    n_id, bs, hops = q.get(timeout=60)
    p.join(timeout=30)
    assert bs == 3 and hops == 2 and n_id >= 3
