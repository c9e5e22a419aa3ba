"""Feature store: degree/probability-ordered multi-tier feature placement.

Tiers on an MI355X node: hot rows in local HBM3E (288 GB/GPU), warm rows in
peer-GPU HBM read one-sided over xGMI, cold rows in pinned host DRAM read
zero-copy, optional disk tier via numpy mmap.

Capability parity with reference quiver/feature.py (Feature, DeviceConfig,
PartitionInfo, DistFeature).
"""
from typing import List

import numpy as np
import torch

from . import _ext
from .shard_tensor import ShardTensor, ShardTensorConfig
from .trace import trace_scope
from .utils import Topo, CSRTopo, hot_order, parse_size

__all__ = ["Feature", "DistFeature", "PartitionInfo", "DeviceConfig"]


class DeviceConfig:
    def __init__(self, gpu_parts, cpu_part):
        self.gpu_parts = gpu_parts
        self.cpu_part = cpu_part


class Feature(object):
    """Partitioned/replicated feature store with GPU gather kernels.

    Args:
        rank: device running the gather kernels.
        device_list: devices participating in data placement.
        device_cache_size: per-device hot-cache budget ("200M", "4G", bytes).
        cache_policy: "device_replicate" (same hot cache on every GPU) or
            "p2p_clique_replicate" (hot cache sharded across the xGMI clique).
        csr_topo: graph topology for degree-ordered cache placement.
    """

    def __init__(self, rank: int, device_list: List[int],
                 device_cache_size=0, cache_policy="device_replicate",
                 csr_topo: CSRTopo = None):
        assert cache_policy in ["device_replicate", "p2p_clique_replicate"]
        self.device_cache_size = device_cache_size
        self.cache_policy = cache_policy
        self.device_list = device_list
        self.device_tensor_list = {}
        self.clique_tensor_list = {}
        self.rank = rank
        self.topo = Topo(self.device_list)
        self.csr_topo = csr_topo
        self.feature_order = None
        self.ipc_handle_ = None
        self.mmap_handle_ = None
        self.disk_map = None
        self.cpu_part = torch.zeros(0)
        assert self.clique_device_symmetry_check(), (
            f"\n{self.topo.info()}\nDifferent p2p clique sizes")

    def clique_device_symmetry_check(self):
        if self.cache_policy == "device_replicate":
            return True
        sizes = {len(c) for c in self.topo.cliques}
        return len(sizes) == 1

    def cal_size(self, cpu_tensor: torch.Tensor, cache_memory_budget: int):
        element_size = cpu_tensor.shape[1] * cpu_tensor.element_size()
        return cache_memory_budget // element_size

    def partition(self, cpu_tensor: torch.Tensor, cache_memory_budget: int):
        cache_size = self.cal_size(cpu_tensor, cache_memory_budget)
        return [cpu_tensor[:cache_size], cpu_tensor[cache_size:]]

    def from_cpu_tensor(self, cpu_tensor: torch.Tensor,
                        score: torch.Tensor = None):
        """Split a CPU tensor into hot cache(s) + pinned cold tier.

        score: optional per-node hotness used for cache placement instead
        of out-degree — pass the access probability from
        GraphSageSampler.sample_prob for probability-driven placement
        (reference cal_neighbor_prob, quiver_sample.cu:100-111).
        """
        if self.cache_policy == "device_replicate":
            cache_memory_budget = parse_size(self.device_cache_size)
            shuffle_ratio = 0.0
        else:
            clique0 = self.topo.cliques[0]
            cache_memory_budget = parse_size(self.device_cache_size) * len(clique0)
            shuffle_ratio = min(
                1.0, self.cal_size(cpu_tensor, cache_memory_budget)
                / cpu_tensor.size(0))

        if self.csr_topo is not None:
            if self.csr_topo.feature_order is None:
                prev_order, self.csr_topo.feature_order = hot_order(
                    self.csr_topo, shuffle_ratio, score=score)
                cpu_tensor = cpu_tensor[prev_order]
            self.feature_order = self.csr_topo.feature_order.to(self.rank) \
                if torch.cuda.is_available() else self.csr_topo.feature_order

        cache_part, self.cpu_part = self.partition(cpu_tensor,
                                                   cache_memory_budget)
        self.cpu_part = self.cpu_part.clone()

        if cache_part.shape[0] > 0 and self.cache_policy == "device_replicate":
            for device in self.device_list:
                shard_tensor = ShardTensor(self.rank, ShardTensorConfig({}))
                shard_tensor.append(cache_part, device)
                self.device_tensor_list[device] = shard_tensor
        elif cache_part.shape[0] > 0:
            for clique_id, clique_devices in enumerate(self.topo.cliques):
                block_size = self.cal_size(
                    cpu_tensor, cache_memory_budget // len(clique_devices))
                # when the aggregate clique budget exceeds the tensor
                # (e.g. 8 x 20%), balance shards across all devices
                # instead of filling the first few and leaving the rest
                # empty — xGMI read traffic then spreads over every link
                block_size = min(block_size,
                                 (cache_part.shape[0] +
                                  len(clique_devices) - 1)
                                 // len(clique_devices))
                shard_tensor = ShardTensor(self.rank, ShardTensorConfig({}))
                cur_pos = 0
                for idx, device in enumerate(clique_devices):
                    if idx == len(clique_devices) - 1:
                        shard_tensor.append(cache_part[cur_pos:], device)
                    else:
                        shard_tensor.append(
                            cache_part[cur_pos:cur_pos + block_size], device)
                        cur_pos += block_size
                self.clique_tensor_list[clique_id] = shard_tensor

        if self.cpu_part.numel() > 0:
            if self.cache_policy == "device_replicate":
                shard_tensor = self.device_tensor_list.get(
                    self.rank, None) or ShardTensor(self.rank,
                                                    ShardTensorConfig({}))
                shard_tensor.append(self.cpu_part, -1)
                self.device_tensor_list[self.rank] = shard_tensor
            else:
                clique_id = self.topo.get_clique_id(self.rank)
                shard_tensor = self.clique_tensor_list.get(
                    clique_id, None) or ShardTensor(self.rank,
                                                    ShardTensorConfig({}))
                shard_tensor.append(self.cpu_part, -1)
                self.clique_tensor_list[clique_id] = shard_tensor

    def from_cpu_tensor_dist(self, cpu_tensor: torch.Tensor, world=None,
                             rank=None, all_gather_object=None,
                             score: torch.Tensor = None):
        """Collaborative p2p_clique_replicate build, one process per GPU.

        Each torchrun rank hipMallocs ONLY its own device's hot shard and
        reopens every peer's shard via hipIpc — instead of each of N ranks
        allocating the full N-shard set (N x hot-cache HBM, N x H2D setup
        traffic).  The hot/cold ordering is computed deterministically
        (seeded) so all ranks agree without a broadcast.

        Layout parity with the single-process from_cpu_tensor on the same
        budget: shard r holds hot rows [r*block, (r+1)*block).

        all_gather_object(obj) -> list of every rank's obj (e.g. a wrapper
        over torch.distributed.all_gather_object).

        Reference design point: NVLink-sharded p2p store built by one
        owner + cudaIpc reopen (quiver_feature.cu:378-421, examples).

        world/rank/all_gather_object default to the initialized
        torch.distributed process group.
        """
        assert self.cache_policy == "p2p_clique_replicate", \
            "distributed build targets the sharded (p2p) layout"
        if all_gather_object is None:
            import torch.distributed as dist
            assert dist.is_initialized(), \
                "pass all_gather_object or init torch.distributed first"
            world = world if world is not None else dist.get_world_size()
            rank = rank if rank is not None else dist.get_rank()

            def all_gather_object(obj, _w=world):
                objs = [None] * _w
                dist.all_gather_object(objs, obj)
                return objs
        assert world is not None and rank is not None
        budget = parse_size(self.device_cache_size)
        total_budget = budget * world
        cache_rows = min(self.cal_size(cpu_tensor, total_budget),
                         cpu_tensor.size(0))
        prev_order = None
        if self.csr_topo is not None:
            if self.csr_topo.feature_order is None:
                shuffle_ratio = cache_rows / max(1, cpu_tensor.size(0))
                prev_order, self.csr_topo.feature_order = hot_order(
                    self.csr_topo, shuffle_ratio, score=score)
            else:
                prev_order = torch.argsort(self.csr_topo.feature_order)
            self.feature_order = self.csr_topo.feature_order.to(self.rank)

        def rows(beg, end):
            """Rows [beg, end) of the (virtually) reordered tensor —
            gathered per-slice so the full reordered tensor is never
            materialized."""
            if prev_order is None:
                return cpu_tensor[beg:end]
            return cpu_tensor[prev_order[beg:end]]

        # this rank's shard of the hot rows
        block = (cache_rows + world - 1) // world
        beg = min(rank * block, cache_rows)
        end = min(beg + block, cache_rows)
        own = ShardTensor(self.rank, ShardTensorConfig({}))
        handles = []
        if end > beg:
            own.shard_tensor.append(rows(beg, end).contiguous(), self.rank)
            handles = [it.share_ipc()
                       for it in own.shard_tensor.share_ipc()]
        # keep this rank's allocation alive on SELF before any fallible
        # step past the handle exchange: peers map it via hipIpc, and a
        # failure-then-GC here would free memory they hold open
        self._dist_own_keepalive = own
        gathered = all_gather_object((rank, self.rank, handles))

        # xGMI peer access across every device that owns a shard
        devices = sorted({dev for _, dev, hs in gathered if hs})
        if len(devices) > 1:
            _ext.init_p2p(devices)

        st = ShardTensor(self.rank, ShardTensorConfig({}))
        for r, dev, hs in sorted(gathered):
            if r == rank:
                for i in range(own.shard_tensor.shard_count()):
                    st.shard_tensor.append_from(own.shard_tensor, i)
            else:
                for h in hs:
                    st.shard_tensor.append_item(
                        _ext.ShardTensorItem.from_ipc(h))
        st._own_alloc = own  # keeps this rank's hipMalloc alive

        if cache_rows < cpu_tensor.size(0):
            self.cpu_part = rows(cache_rows, cpu_tensor.size(0)).contiguous()
            st.append(self.cpu_part, -1)
        clique_id = self.topo.get_clique_id(self.rank)
        self.clique_tensor_list[clique_id] = st

    def from_mmap(self, np_array, device_config: DeviceConfig):
        """Build from a (mmap) numpy array + explicit per-device partition.

        gpu_parts[device] is either a torch.Tensor of rows (np_array is None)
        or an index tensor selecting rows of np_array; cpu_part likewise.
        """
        assert len(device_config.gpu_parts) == len(self.device_list)

        def load_part(part):
            if isinstance(part, str):
                return torch.from_numpy(np.load(part))
            if isinstance(part, torch.Tensor):
                if np_array is None:
                    return part.to(dtype=torch.float32)
                return torch.from_numpy(np_array[part.numpy()]).to(
                    dtype=torch.float32)
            raise ValueError("gpu part must be tensor or .npy path")

        if self.cache_policy == "device_replicate":
            for device in self.device_list:
                cache_part = load_part(device_config.gpu_parts[device])
                if cache_part.shape[0] > 0:
                    shard_tensor = ShardTensor(self.rank,
                                               ShardTensorConfig({}))
                    shard_tensor.append(cache_part, device)
                    self.device_tensor_list[device] = shard_tensor
        else:
            for clique_id, clique_devices in enumerate(self.topo.cliques):
                shard_tensor = ShardTensor(self.rank, ShardTensorConfig({}))
                appended = False
                for device in clique_devices:
                    cache_part = load_part(device_config.gpu_parts[device])
                    if cache_part.shape[0] > 0:
                        shard_tensor.append(cache_part, device)
                        appended = True
                if appended:
                    self.clique_tensor_list[clique_id] = shard_tensor

        if device_config.cpu_part is not None:
            cpu_part = device_config.cpu_part
            if isinstance(cpu_part, str):
                cpu_part = torch.from_numpy(np.load(cpu_part))
            elif np_array is not None and cpu_part.dim() == 1:
                cpu_part = torch.from_numpy(np_array[cpu_part.numpy()]).to(
                    dtype=torch.float32)
            self.cpu_part = cpu_part
            if self.cpu_part.numel() > 0:
                if self.cache_policy == "device_replicate":
                    shard_tensor = self.device_tensor_list.get(
                        self.rank, None) or ShardTensor(
                            self.rank, ShardTensorConfig({}))
                    shard_tensor.append(self.cpu_part, -1)
                    self.device_tensor_list[self.rank] = shard_tensor
                else:
                    clique_id = self.topo.get_clique_id(self.rank)
                    shard_tensor = self.clique_tensor_list.get(
                        clique_id, None) or ShardTensor(
                            self.rank, ShardTensorConfig({}))
                    shard_tensor.append(self.cpu_part, -1)
                    self.clique_tensor_list[clique_id] = shard_tensor

    def set_local_order(self, local_order):
        """local_order[new_row] = original id -> feature_order[id] = row."""
        local_range = torch.arange(local_order.size(0), dtype=torch.int64,
                                   device=self.rank)
        self.feature_order = torch.zeros_like(local_range)
        self.feature_order[local_order.to(self.rank)] = local_range

    def set_mmap_file(self, path, disk_map):
        self.lazy_init_from_ipc_handle()
        self.mmap_handle_ = np.load(path, mmap_mode="r")
        self.disk_map = disk_map.to(self.rank)

    def read_mmap(self, ids):
        ids = ids.cpu().numpy()
        res = torch.from_numpy(self.mmap_handle_[ids])
        return res.to(device=self.rank, dtype=torch.float32)

    def _shard_tensor(self):
        if self.cache_policy == "device_replicate":
            return self.device_tensor_list[self.rank]
        return self.clique_tensor_list[self.topo.get_clique_id(self.rank)]

    def gather_raw(self, node_idx, n_dev):
        """Async gather over an upper-bound-sized frontier: the exact row
        count is read by the kernel from the 1-element device tensor
        `n_dev` — no host synchronization anywhere on the path.  Slack
        output rows are untouched; the caller slices.  Not available for
        the disk (mmap) tier or partially-accessible shard layouts."""
        self.lazy_init_from_ipc_handle()
        if self.mmap_handle_ is not None:
            raise RuntimeError("gather_raw does not support the disk tier")
        node_idx = node_idx.to(self.rank)
        if self.feature_order is not None:
            node_idx = self.feature_order[node_idx]
        return self._shard_tensor().gather_n(node_idx, n_dev)

    def __getitem__(self, node_idx: torch.Tensor):
        self.lazy_init_from_ipc_handle()
        node_idx = node_idx.to(self.rank)
        if self.mmap_handle_ is None:
            with trace_scope("feature.gather"):
                if self.feature_order is not None:
                    node_idx = self.feature_order[node_idx]
                return self._shard_tensor()[node_idx]
        # disk tier: disk_map < 0 -> mmap row, >= 0 -> in-memory row
        num_nodes = node_idx.size(0)
        disk_index = self.disk_map[node_idx]
        node_range = torch.arange(num_nodes, device=node_idx.device,
                                  dtype=torch.int64)
        disk_mask = disk_index < 0
        mem_mask = disk_index >= 0
        disk_ids = torch.masked_select(node_idx, disk_mask)
        disk_pos = torch.masked_select(node_range, disk_mask)
        mem_ids = torch.masked_select(node_idx, mem_mask)
        mem_pos = torch.masked_select(node_range, mem_mask)
        local_mem_ids = self.disk_map[mem_ids]
        disk_res = self.read_mmap(disk_ids)
        mem_res = self._shard_tensor()[local_mem_ids]
        res = torch.zeros((num_nodes, self.size(1)), device=node_idx.device,
                          dtype=mem_res.dtype)
        res[disk_pos] = disk_res.to(res.dtype)
        res[mem_pos] = mem_res
        return res

    def size(self, dim: int):
        self.lazy_init_from_ipc_handle()
        return self._shard_tensor().size(dim)

    def dim(self):
        return len(self.shape)

    @property
    def shape(self):
        self.lazy_init_from_ipc_handle()
        return self._shard_tensor().shape

    @property
    def ipc_handle(self):
        return self.ipc_handle_

    @ipc_handle.setter
    def ipc_handle(self, ipc_handle):
        self.ipc_handle_ = ipc_handle

    def share_ipc(self):
        self.cpu_part.share_memory_()
        gpu_ipc_handle_dict = {}
        if self.cache_policy == "device_replicate":
            for device in self.device_tensor_list:
                gpu_ipc_handle_dict[device] = \
                    self.device_tensor_list[device].share_ipc()[0]
        else:
            for clique_id in self.clique_tensor_list:
                gpu_ipc_handle_dict[clique_id] = \
                    self.clique_tensor_list[clique_id].share_ipc()[0]
        return (gpu_ipc_handle_dict,
                self.cpu_part if self.cpu_part.numel() > 0 else None,
                self.device_list, self.device_cache_size, self.cache_policy,
                self.csr_topo)

    def from_gpu_ipc_handle_dict(self, gpu_ipc_handle_dict, cpu_tensor):
        if self.cache_policy == "device_replicate":
            handle = gpu_ipc_handle_dict.get(self.rank, []), cpu_tensor, \
                self.rank
            shard_tensor = ShardTensor.new_from_share_ipc(handle, self.rank)
            self.device_tensor_list[self.rank] = shard_tensor
        else:
            clique_id = self.topo.get_clique_id(self.rank)
            handle = gpu_ipc_handle_dict.get(clique_id, []), cpu_tensor, \
                self.rank
            shard_tensor = ShardTensor.new_from_share_ipc(handle, self.rank)
            self.clique_tensor_list[clique_id] = shard_tensor
        self.cpu_part = cpu_tensor if cpu_tensor is not None \
            else torch.zeros(0)

    @classmethod
    def new_from_ipc_handle(cls, rank, ipc_handle):
        (gpu_ipc_handle_dict, cpu_part, device_list, device_cache_size,
         cache_policy, csr_topo) = ipc_handle
        feature = cls(rank, device_list, device_cache_size, cache_policy)
        feature.from_gpu_ipc_handle_dict(gpu_ipc_handle_dict, cpu_part)
        if csr_topo is not None:
            feature.feature_order = csr_topo.feature_order.to(rank)
        feature.csr_topo = csr_topo
        return feature

    @classmethod
    def lazy_from_ipc_handle(cls, ipc_handle):
        (gpu_ipc_handle_dict, cpu_part, device_list, device_cache_size,
         cache_policy, _) = ipc_handle
        feature = cls(device_list[0], device_list, device_cache_size,
                      cache_policy)
        feature.ipc_handle = ipc_handle
        return feature

    def lazy_init_from_ipc_handle(self):
        if self.ipc_handle is None:
            return
        self.rank = torch.cuda.current_device()
        (gpu_ipc_handle_dict, cpu_part, device_list, device_cache_size,
         cache_policy, csr_topo) = self.ipc_handle
        self.from_gpu_ipc_handle_dict(gpu_ipc_handle_dict, cpu_part)
        self.csr_topo = csr_topo
        if csr_topo is not None:
            self.feature_order = csr_topo.feature_order.to(self.rank)
        self.ipc_handle = None


class PartitionInfo:
    """Node -> host partition bookkeeping for DistFeature.

    Mirrors reference feature.py:461-527.
    """

    def __init__(self, device, host, hosts, global2host, replicate=None):
        self.global2host = global2host.to(device)
        self.host = host
        self.hosts = hosts
        self.device = device
        self.size = self.global2host.size(0)
        self.replicate = None if replicate is None else replicate.to(device)
        self.init_global2local()

    def init_global2local(self):
        total_range = torch.arange(self.size, dtype=torch.int64,
                                   device=self.device)
        self.global2local = torch.arange(self.size, dtype=torch.int64,
                                         device=self.device)
        for host in range(self.hosts):
            mask = self.global2host == host
            host_nodes = torch.masked_select(total_range, mask)
            host_size = host_nodes.size(0)
            if host == self.host:
                local_size = host_size
                if self.replicate is not None:
                    local_size += self.replicate.size(0)
            self.global2local[host_nodes] = torch.arange(
                host_size, dtype=torch.int64, device=self.device)
        if self.replicate is not None:
            # replicated hot nodes are folded into the local range's tail
            mask = self.global2host[self.replicate] != self.host
            remote_replicate = torch.masked_select(self.replicate, mask)
            base = int((self.global2host == self.host).sum())
            self.global2local[remote_replicate] = torch.arange(
                base, base + remote_replicate.size(0), dtype=torch.int64,
                device=self.device)
            self.global2host[remote_replicate] = self.host

    def dispatch(self, ids):
        """Bucket a frontier by owning host -> (host_ids, host_orders)."""
        host_ids = []
        host_orders = []
        ids_range = torch.arange(ids.size(0), dtype=torch.int64,
                                 device=self.device)
        owners = self.global2host[ids]
        for host in range(self.hosts):
            mask = owners == host
            h_ids = torch.masked_select(ids, mask)
            h_orders = torch.masked_select(ids_range, mask)
            host_ids.append(self.global2local[h_ids])
            host_orders.append(h_orders)
        return host_ids, host_orders


class DistFeature:
    """Multi-host feature store: local Feature + RCCL id/feature exchange.

    x = dist_feature[ids] is a synchronous collective — every rank in the
    communicator must call it together (reference feature.py:529-567).
    """

    def __init__(self, feature: Feature, info: PartitionInfo, comm):
        self.feature = feature
        self.info = info
        self.comm = comm

    def __getitem__(self, ids):
        ids = ids.to(self.comm.device)
        host_ids, host_orders = self.info.dispatch(ids)
        host_feats = self.comm.exchange(host_ids, self.feature)
        res = torch.zeros((ids.size(0), self.feature.size(1)),
                          device=self.comm.device)
        for feats, orders in zip(host_feats, host_orders):
            if feats is not None and feats.numel() > 0:
                res[orders] = feats
        local_ids = host_ids[self.info.host]
        local_orders = host_orders[self.info.host]
        if local_ids.numel() > 0:
            res[local_orders] = self.feature[local_ids]
        return res
