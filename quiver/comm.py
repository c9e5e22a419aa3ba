"""Inter-host feature exchange over RCCL.

Capability parity with reference quiver/comm.py: HostRankTable, the greedy
pairwise-round scheduler, and the two-phase (ids -> feats) exchange driven
by an allreduced ws*ws request-size matrix.

The exchange protocol is transport-agnostic so the same code runs over the
native RCCL communicator (GPU, xGMI/IB) and over torch.distributed gloo
(CPU tests): any object with send/recv/allreduce/rank/size/device works.
"""
import torch

from . import _ext


class HostRankTable:
    """rank <-> host arithmetic for a homogeneous layout: world ranks are
    numbered host-major, so rank r lives on host r // rank_per_host and a
    rank's peer on a remote host is the one sharing its local index."""

    def __init__(self, hosts, rank_per_host):
        self.hosts = hosts
        self.rank_per_host = rank_per_host

    @property
    def world(self):
        return self.hosts * self.rank_per_host

    def ranks(self, host):
        base = host * self.rank_per_host
        return list(range(base, base + self.rank_per_host))

    def host(self, rank):
        return rank // self.rank_per_host

    def remote_peer(self, rank, host):
        return host * self.rank_per_host + rank % self.rank_per_host

    def remote_peers(self, rank, hosts):
        return [(rank, self.remote_peer(rank, host)) for host in hosts]

    def get_comm_mat(self, flat_allreduce):
        w = self.world
        return flat_allreduce.cpu().reshape(w, w).tolist()


def schedule(comm_mat, table):
    """Greedy pairing of hosts into communication rounds."""
    steps = []
    traversed_pair = set()
    while True:
        step = []
        traversed_host = set()
        for src in range(table.hosts):
            if src in traversed_host:
                continue
            src_ranks = table.ranks(src)
            for dst in range(table.hosts):
                if dst in traversed_host:
                    continue
                if (src, dst) in traversed_pair:
                    continue
                traversed_pair.add((src, dst))
                finished = False
                for src_rank in src_ranks:
                    dst_rank = table.remote_peer(src_rank, dst)
                    if comm_mat[src_rank][dst_rank] <= 0:
                        continue
                    step.append((src_rank, dst_rank))
                    finished = True
                if finished:
                    traversed_host.add(src)
                    traversed_host.add(dst)
                    break
        if not step:
            break
        steps.append(step)
    return steps


class _ExchangeMixin:
    """Two-phase id/feature exchange; needs send/recv/allreduce/rank/size/
    device and a HostRankTable at self.table."""

    def _sync(self):
        if torch.cuda.is_available() and str(self.device) != "cpu":
            torch.cuda.current_stream().synchronize()

    def exchange(self, host2ids, feature):
        remote_sizes = torch.zeros(self.size * self.size, dtype=torch.int64)
        for host in range(self.table.hosts):
            ids = host2ids[host]
            remote_peer = self.table.remote_peer(self.rank, host)
            if ids is not None and remote_peer != self.rank:
                remote_sizes[self.rank * self.size + remote_peer] = ids.size(0)
        remote_sizes = remote_sizes.to(self.device)
        self.allreduce(remote_sizes)
        comm_mat = self.table.get_comm_mat(remote_sizes)
        steps = schedule(comm_mat, self.table)
        self._sync()
        req_ids = [None] * self.size
        res_feats = [None] * self.size
        for step in steps:
            for src, dst in step:
                if src == self.rank:
                    self.send(host2ids[self.table.host(dst)].to(self.device),
                              dst)
                if dst == self.rank:
                    ids = torch.zeros(comm_mat[src][dst], dtype=torch.int64,
                                      device=self.device)
                    self.recv(ids, src)
                    req_ids[src] = ids
        self._sync()
        for i, ids in enumerate(req_ids):
            if ids is not None:
                res_feats[i] = feature[ids].to(self.device)
        host2feats = [None] * self.table.hosts
        feat_dim = feature.size(1)
        for step in steps:
            for src, dst in step:
                if dst == self.rank:
                    self.send(res_feats[src], src)
                if src == self.rank:
                    feats = torch.zeros(comm_mat[src][dst], feat_dim,
                                        device=self.device)
                    self.recv(feats, dst)
                    host2feats[self.table.host(dst)] = feats
        self._sync()
        return host2feats


class NcclComm(_ExchangeMixin):
    """RCCL communicator on the current torch stream (GPU path)."""

    def __init__(self, rank, ws, id, hosts=None, rank_per_host=None):
        self.comm = _ext.NcclComm(rank, ws, id)
        self.device_ = torch.device("cuda", torch.cuda.current_device())
        if hosts is not None:
            self.table = HostRankTable(hosts, rank_per_host)
            self.host = self.table.host(rank)

    @property
    def rank(self):
        return self.comm.rank()

    @property
    def size(self):
        return self.comm.size()

    @property
    def device(self):
        return self.device_

    def send(self, tensor, dst):
        self.comm.send(tensor, dst)

    def recv(self, tensor, src):
        self.comm.recv(tensor, src)

    def allreduce(self, tensor):
        self.comm.allreduce(tensor)

    def allgather(self, src, dst):
        self.comm.allgather(src, dst)

    def alltoall(self, src, dst):
        self.comm.alltoall(src, dst)


class TorchDistComm(_ExchangeMixin):
    """Same exchange protocol over an initialized torch.distributed process
    group (gloo on CPU for tests, nccl/RCCL otherwise)."""

    def __init__(self, hosts, rank_per_host, device="cpu"):
        import torch.distributed as dist
        assert dist.is_initialized()
        self.dist = dist
        self.table = HostRankTable(hosts, rank_per_host)
        self.host = self.table.host(dist.get_rank())
        self.device_ = torch.device(device)

    @property
    def rank(self):
        return self.dist.get_rank()

    @property
    def size(self):
        return self.dist.get_world_size()

    @property
    def device(self):
        return self.device_

    def send(self, tensor, dst):
        self.dist.send(tensor.cpu() if self.device_.type == "cpu" else tensor,
                       dst)

    def recv(self, tensor, src):
        self.dist.recv(tensor, src)

    def allreduce(self, tensor):
        self.dist.all_reduce(tensor)


def getNcclId():
    return _ext.create_nccl_id()
