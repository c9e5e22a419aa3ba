"""Multi-process CPU coverage of the exchange protocol (gloo, world=2).

The same _ExchangeMixin drives the RCCL path on GPU, so this validates the
id/feature wire protocol and the pairwise scheduler end to end.
"""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from quiver.comm import HostRankTable, schedule, TorchDistComm
from quiver.feature import PartitionInfo, DistFeature


def test_host_rank_table():
    t = HostRankTable(2, 2)
    assert t.ranks(0) == [0, 1]
    assert t.ranks(1) == [2, 3]
    assert t.host(3) == 1
    assert t.remote_peer(1, 1) == 3
    assert t.remote_peer(2, 0) == 0


def test_schedule_pairs_cover_requests():
    t = HostRankTable(2, 1)
    comm_mat = [[0, 5], [3, 0]]
    steps = schedule(comm_mat, t)
    pairs = [p for step in steps for p in step]
    assert (0, 1) in pairs and (1, 0) in pairs


class _CpuFeature:
    """Minimal feature backend for protocol tests."""

    def __init__(self, rows):
        self.rows = rows

    def __getitem__(self, ids):
        return self.rows[ids.cpu()]

    def size(self, dim):
        return self.rows.size(dim)


def _worker(rank, world, port, n_nodes, dim):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        full = torch.arange(n_nodes * dim, dtype=torch.float32).reshape(
            n_nodes, dim)
        # host h owns rows [h*half, (h+1)*half)
        half = n_nodes // world
        global2host = torch.arange(n_nodes) // half
        local = full[rank * half:(rank + 1) * half]
        comm = TorchDistComm(hosts=world, rank_per_host=1)
        info = PartitionInfo("cpu", rank, world, global2host)
        dist_feature = DistFeature(_CpuFeature(local), info, comm)
        ids = torch.randperm(n_nodes)[:10]
        got = dist_feature[ids]
        assert torch.equal(got, full[ids]), (rank, got, full[ids])
    finally:
        dist.destroy_process_group()


def test_dist_feature_exchange_gloo():
    port = 29511
    mp.spawn(_worker, args=(2, port, 40, 5), nprocs=2, join=True)


def test_schedule_properties_random():
    """Property test: every positive comm_mat entry between distinct hosts
    is scheduled exactly once, and within one round each host appears in
    at most one host-pair."""
    import random
    rng = random.Random(0)
    for trial in range(25):
        hosts = rng.randint(1, 5)
        rph = rng.randint(1, 3)
        t = HostRankTable(hosts, rph)
        ws = hosts * rph
        comm_mat = [[rng.choice([0, 0, 1, 7]) if t.host(i) != t.host(j)
                     else 0 for j in range(ws)] for i in range(ws)]
        steps = schedule(comm_mat, t)
        seen = set()
        for step in steps:
            hosts_in_step = set()
            host_pairs = set()
            for (src, dst) in step:
                pair = (t.host(src), t.host(dst))
                host_pairs.add(pair)
                assert (src, dst) not in seen, "pair scheduled twice"
                seen.add((src, dst))
            for hs, hd in host_pairs:
                # a host participates in at most one pair per round
                assert sum(1 for p in host_pairs if hs in p) <= 2
                hosts_in_step.update([hs, hd])
        # completeness: every requested (src_rank -> peer) with traffic is
        # scheduled (for the peer the table routes that src to)
        for src in range(ws):
            for host in range(hosts):
                if host == t.host(src):
                    continue
                dst = t.remote_peer(src, host)
                if comm_mat[src][dst] > 0:
                    assert (src, dst) in seen, (src, dst, comm_mat)
