"""Pure-logic unit tests (no GPU, no processes): exchange scheduler
properties, rank/host arithmetic, the mixed-sampler split policy."""
import torch

import quiver
from quiver.comm import HostRankTable, schedule
from quiver.pyg.sage_sampler import MixedGraphSageSampler
from quiver.utils import parse_size


def test_host_rank_table_arithmetic():
    t = HostRankTable(hosts=3, rank_per_host=4)
    assert t.world == 12
    assert t.ranks(1) == [4, 5, 6, 7]
    assert [t.host(r) for r in range(12)] == [0] * 4 + [1] * 4 + [2] * 4
    # a rank's remote peer shares its local index on the remote host
    assert t.remote_peer(5, 2) == 9
    assert t.remote_peers(5, [0, 2]) == [(5, 1), (5, 9)]
    mat = t.get_comm_mat(torch.arange(144))
    assert mat[1][2] == 1 * 12 + 2


def test_schedule_covers_all_nonzero_pairs_once():
    t = HostRankTable(hosts=4, rank_per_host=2)
    w = t.world
    comm = [[0] * w for _ in range(w)]
    # every host pair requests in both directions
    for src_h in range(4):
        for dst_h in range(4):
            if src_h == dst_h:
                continue
            for r in t.ranks(src_h):
                comm[r][t.remote_peer(r, dst_h)] = 5
    steps = schedule(comm, t)
    seen = set()
    for step in steps:
        # within one round each host participates in at most ONE host
        # pair (that is the point of the pairwise scheduler)
        host_of = {}
        for src, dst in step:
            pair = (src, dst)
            assert pair not in seen, "pair scheduled twice"
            seen.add(pair)
            hs, hd = t.host(src), t.host(dst)
            assert host_of.setdefault(hs, hd) == hd, \
                f"host {hs} talks to two hosts in one round"
    # every requested (src, dst) rank pair appears exactly once overall
    want = {(r, t.remote_peer(r, dh))
            for sh in range(4) for dh in range(4) if sh != dh
            for r in t.ranks(sh)}
    assert seen == want


def test_mixed_split_policy():
    s = MixedGraphSageSampler.__new__(MixedGraphSageSampler)
    s.num_workers = 4
    # no measurements yet: seed every worker
    assert s.decide_task_num(0.0, 0, 0.0, 0) == 4
    # device 10 ms/task, one CPU worker-second per task -> pool finishes
    # 4*10/1000... t_dev=0.01, t_cpu=1.0 -> 4*0.01/1.0/2 = 0.02 -> 0
    assert s.decide_task_num(0.1, 10, 10.0, 10) == 0
    # CPU as fast as device: hand out num_workers/2 per inline task
    assert s.decide_task_num(1.0, 10, 1.0, 10) == 2
    # CPU 4x faster: cap comes from the caller, policy says 8
    assert s.decide_task_num(1.0, 10, 0.25, 10) == 8


def test_parse_size_units():
    assert parse_size("1K") == 1024
    assert parse_size("1KB") == 1024
    assert parse_size("1.5M") == int(1.5 * 2**20)
    assert parse_size("2G") == 2 * 2**30
    assert parse_size(12345) == 12345
    assert parse_size("8") == 8


def test_partition_info_dispatch_cpu():
    """PartitionInfo: host bucketing + replicated hot nodes folded into
    the local range (reference feature.py:461-527), pure tensor logic."""
    from quiver.feature import PartitionInfo
    g2h = torch.tensor([0, 0, 1, 1, 2, 2, 0, 1])   # 8 nodes on 3 hosts
    replicate = torch.tensor([2, 4])                # hot nodes mirrored here
    info = PartitionInfo(device="cpu", host=0, hosts=3,
                         global2host=g2h.clone(), replicate=replicate)
    # replicated remote nodes now claim host 0
    assert int(info.global2host[2]) == 0 and int(info.global2host[4]) == 0
    ids = torch.tensor([0, 2, 3, 4, 5, 7])
    host_ids, host_orders = info.dispatch(ids)
    assert len(host_ids) == 3
    # host 0 serves: 0 (native), 2 and 4 (replicas) -> positions 0,1,3
    assert host_orders[0].tolist() == [0, 1, 3]
    # host 1 serves 3 and 7; host 2 serves 5
    assert host_orders[1].tolist() == [2, 5]
    assert host_orders[2].tolist() == [4]
    # every queried id lands on exactly one host
    total = sum(o.numel() for o in host_orders)
    assert total == ids.numel()


def test_topo_single_clique_without_gpu():
    """GPU-less Topo: all-true access matrix -> one clique, stable ids."""
    from quiver.utils import Topo
    t = Topo([0, 1, 2, 3])
    assert t.p2p_clique_count == 1
    assert t.cliques == [[0, 1, 2, 3]]
    for d in range(4):
        assert t.get_clique_id(d) == 0
    assert t.p2p_clique(2) == [0, 1, 2, 3]
