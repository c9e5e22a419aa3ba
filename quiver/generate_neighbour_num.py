"""Offline per-node expected sampled-subgraph size — the workload predictor
feeding RequestBatcher's Auto routing (reference
quiver/generate_neighbour_num.py).

Two paths:
  - sample=True: exact per-node frontier size by running the sampler on each
    node (reference behavior; slow, offline).
  - sample=False (default): fast batched estimate — each node's predicted
    work is the sum over hops of its sampled neighbor counts (duplicates
    kept), computed for all nodes in large batches.  For routing-threshold
    purposes this tracks the exact numbers closely at a fraction of the
    cost.
"""
import numpy as np
import torch
import torch.multiprocessing as mp

from .pyg import GraphSageSampler
from .utils import CSRTopo

__all__ = ["generate_neighbour_num"]


def _estimate_batched(sampler: GraphSageSampler, node_num, sizes,
                      batch=16384):
    out = np.zeros(node_num, dtype=np.int64)
    for beg in range(0, node_num, batch):
        end = min(node_num, beg + batch)
        seeds = torch.arange(beg, end, dtype=torch.long)
        totals = torch.zeros(end - beg, dtype=torch.long)
        owner = torch.arange(end - beg, dtype=torch.long)
        nodes = seeds
        for size in sizes:
            nbrs, cnt = sampler.sample_layer(nodes, size)
            cnt = cnt.cpu()
            totals.index_add_(0, owner, cnt)
            # next hop: each sampled neighbor belongs to its seed's owner
            owner = torch.repeat_interleave(owner, cnt)
            nodes = nbrs
        out[beg:end] = totals.numpy()
    return out


def _exact_per_node(sampler: GraphSageSampler, rank, node_num, num_proc):
    res = []
    for node in range(rank, node_num, num_proc):
        n_id, _, _ = sampler.sample(torch.tensor([node], dtype=torch.long))
        res.append((node, n_id.shape[0]))
    return res


def single_generate_neighbour_num(rank, node_num, csr_topo, num_proc, sizes,
                                  mode, result_path, device_list, sample,
                                  out_queue=None):
    device = device_list[rank % len(device_list)] if mode != "CPU" else "cpu"
    sampler = GraphSageSampler(csr_topo, sizes, device=device, mode=mode)
    if sample:
        res = _exact_per_node(sampler, rank, node_num, num_proc)
        if out_queue is not None:
            out_queue.put(res)
            return None
        arr = np.zeros(node_num, dtype=np.int64)
        for node, n in res:
            arr[node] = n
        return arr
    return _estimate_batched(sampler, node_num, sizes)


def generate_neighbour_num(node_num, edge_index, sizes, result_path,
                           device_list=("cpu",), parallel=False, mode="CPU",
                           num_proc=1, reverse=False, sample=False):
    """Compute neighbour_num[v] for all v and save as .npy at result_path."""
    if isinstance(edge_index, CSRTopo):
        csr_topo = edge_index
    else:
        if reverse:
            edge_index = edge_index.flip(0) if isinstance(
                edge_index, torch.Tensor) else edge_index[::-1]
        csr_topo = CSRTopo(edge_index, node_count=node_num)
    if not parallel or not sample:
        arr = single_generate_neighbour_num(0, node_num, csr_topo, 1, sizes,
                                            mode, result_path, device_list,
                                            sample)
    else:
        csr_topo.share_memory_()
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = []
        for r in range(num_proc):
            p = ctx.Process(target=single_generate_neighbour_num,
                            args=(r, node_num, csr_topo, num_proc, sizes,
                                  mode, result_path, device_list, True, q))
            p.start()
            procs.append(p)
        arr = np.zeros(node_num, dtype=np.int64)
        for _ in procs:
            for node, n in q.get():
                arr[node] = n
        for p in procs:
            p.join()
    if result_path is not None:
        np.save(result_path, arr)
    return arr
