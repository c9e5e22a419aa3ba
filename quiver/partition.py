"""Access-probability-driven feature partitioning.

Capability parity with reference quiver/partition.py (on-disk format:
result_path/feature_partition_N/{partition_res,cache_res}.pth +
feature_partition_book.pth).  The reference's
partition_feature_without_replication is an unfinished stub upstream
(partition.py:95-115 returns empty lists); here it is implemented with the
same probability-greedy chunked strategy as partition_without_replication.
Device-agnostic (runs on CPU when no GPU is present) and non-interactive.
"""
import os
import shutil
from typing import List

import torch

from . import utils as quiver_util

__all__ = [
    "quiver_partition_feature", "load_quiver_feature_partition",
    "partition_without_replication", "partition_feature_without_replication",
    "select_nodes",
]

QUIVER_MAGIC_NUMBER = 256
CHUNK_NUM = 32


def _default_device():
    if torch.cuda.is_available():
        return torch.cuda.current_device()
    return "cpu"


def partition_without_replication(device, probs: List[torch.Tensor], ids):
    """Greedy chunked partition of `ids` (or all nodes) into len(probs)
    parts: each chunk is scored per rank as own_prob*ranks - sum(other
    probs) and the top scores are picked round-robin, so every partition
    gets the nodes it accesses most while sizes stay balanced."""
    ranks = len(probs)
    if ids is not None:
        ids = ids.to(device)
    probs = [
        prob[ids].to(device) if ids is not None else prob.to(device)
        for prob in probs
    ]
    total_size = ids.size(0) if ids is not None else probs[0].size(0)
    res = [[] for _ in range(ranks)]
    chunk_size = (total_size + CHUNK_NUM - 1) // CHUNK_NUM
    chunk_beg = 0
    beg_rank = 0
    for _ in range(CHUNK_NUM):
        chunk_end = min(total_size, chunk_beg + chunk_size)
        this_chunk = chunk_end - chunk_beg
        if this_chunk <= 0:
            break
        chunk = torch.arange(chunk_beg, chunk_end, dtype=torch.int64,
                             device=device)
        score = [
            torch.full((this_chunk,), 1e-6, device=device)
            for _ in range(ranks)
        ]
        for rank in range(ranks):
            for dst_rank in range(ranks):
                if dst_rank == rank:
                    score[rank] += probs[dst_rank][chunk] * ranks
                else:
                    score[rank] -= probs[dst_rank][chunk]
        acc_size = 0
        rank_size = (this_chunk + ranks - 1) // ranks
        picked = torch.empty(0, dtype=torch.int64, device=device)
        for rank_ in range(beg_rank, beg_rank + ranks):
            rank = rank_ % ranks
            score[rank][picked] -= 1e6
            take = min(rank_size, this_chunk - acc_size)
            _, order = torch.sort(score[rank], descending=True)
            pick = order[:take]
            picked = torch.cat((picked, pick))
            res[rank].append(chunk[pick])
            acc_size += take
        beg_rank += 1
        chunk_beg += this_chunk
    out = []
    for rank in range(ranks):
        part = torch.cat(res[rank]) if res[rank] else torch.empty(
            0, dtype=torch.int64, device=device)
        out.append(ids[part] if ids is not None else part)
    return out


def select_nodes(device, probs: List[torch.Tensor], ids):
    nodes = probs[0].size(0)
    prob_sum = torch.zeros(nodes, device=device)
    for prob in probs:
        if ids is None:
            prob_sum += prob.to(device)
        else:
            prob_sum[ids] += prob[ids].to(device)
    node_ids = torch.nonzero(prob_sum)
    return prob_sum, node_ids


def partition_feature_without_replication(probs: List[torch.Tensor],
                                          chunk_size: int = QUIVER_MAGIC_NUMBER):
    """Partition every node by access probability; returns (parts, probs)."""
    device = _default_device()
    probs = [p.to(device) for p in probs]
    parts = partition_without_replication(device, probs, None)
    return parts, probs


def quiver_partition_feature(probs: List[torch.Tensor], result_path: str,
                             cache_memory_budget=0, per_feature_size=0,
                             chunk_size=QUIVER_MAGIC_NUMBER,
                             overwrite: bool = True):
    """Partition features by access probability and persist the result.

    Layout (same as reference):
        result_path/feature_partition_{i}/partition_res.pth
        result_path/feature_partition_{i}/cache_res.pth
        result_path/feature_partition_book.pth
    """
    if os.path.exists(result_path):
        if not overwrite:
            raise FileExistsError(result_path)
        shutil.rmtree(result_path)

    partition_num = len(probs)
    for idx in range(partition_num):
        os.makedirs(os.path.join(result_path, f"feature_partition_{idx}"))

    cache_memory_budget_bytes = quiver_util.parse_size(cache_memory_budget)
    per_feature_size_bytes = quiver_util.parse_size(per_feature_size)
    cache_count = int(cache_memory_budget_bytes /
                      (per_feature_size_bytes + 1e-6))
    per_partition_cache_count = cache_count // partition_num

    device = _default_device()
    partition_book = torch.zeros(probs[0].shape, dtype=torch.int64,
                                 device=device)
    partition_res, changed_probs = partition_feature_without_replication(
        probs, chunk_size)

    cache_res = [None] * partition_num
    if cache_count > 0:
        for idx in range(partition_num):
            _, prev_order = torch.sort(changed_probs[idx], descending=True)
            cache_res[idx] = prev_order[:per_partition_cache_count]

    for idx in range(partition_num):
        partition_book[partition_res[idx]] = idx
        torch.save(partition_res[idx],
                   os.path.join(result_path, f"feature_partition_{idx}",
                                "partition_res.pth"))
        torch.save(cache_res[idx],
                   os.path.join(result_path, f"feature_partition_{idx}",
                                "cache_res.pth"))
    torch.save(partition_book,
               os.path.join(result_path, "feature_partition_book.pth"))
    return partition_book, partition_res, cache_res


def load_quiver_feature_partition(partition_idx: int, result_path: str):
    if not os.path.exists(result_path):
        raise FileNotFoundError(result_path)
    partition_book = torch.load(
        os.path.join(result_path, "feature_partition_book.pth"))
    partition_res = torch.load(
        os.path.join(result_path, f"feature_partition_{partition_idx}",
                     "partition_res.pth"))
    cache_res = torch.load(
        os.path.join(result_path, f"feature_partition_{partition_idx}",
                     "cache_res.pth"))
    return partition_book, partition_res, cache_res
