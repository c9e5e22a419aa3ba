"""Python ShardTensor: a virtual tensor spanning local HBM, peer-GPU HBM
(xGMI one-sided loads) and pinned host DRAM (zero-copy).

Capability parity with reference quiver/shard_tensor.py; the device kernel
is csrc/gather_kernels.hip.
"""
from typing import Dict, NamedTuple

import torch

from . import _ext
from .utils import parse_size


class Offset(NamedTuple):
    """Row span [start, end) a device's shard occupies in the virtual
    tensor (named for reference-API familiarity; it is just a row range)."""
    start: int
    end: int


class ShardTensorConfig:
    """device id -> memory budget ("200M", "4G" or bytes)."""

    def __init__(self, device_memory_budget: Dict[int, object]):
        self.tensor_offset_device: Dict[int, Offset] = {}
        self.device_memory_budget = {
            d: parse_size(b) for d, b in (device_memory_budget or {}).items()
        }

    @property
    def device_list(self):
        return list(self.device_memory_budget.keys())


class ShardTensor:
    def __init__(self, current_device: int, shard_tensor_config=None):
        self.shard_tensor = _ext.ShardTensor(current_device)
        self.current_device = current_device
        self.shard_tensor_config = shard_tensor_config or ShardTensorConfig({})
        self.cpu_tensor = None

    def partition(self, tensor, memory_budget):
        """How many rows fit in memory_budget bytes."""
        row_bytes = tensor[0].numel() * tensor.element_size()
        return parse_size(memory_budget) // row_bytes

    def append(self, tensor, device):
        start = self.shard_tensor.size(0) if self.shard_tensor.shard_count() else 0
        self.shard_tensor.append(tensor, device)
        end = start + tensor.size(0)
        if device >= 0:
            self.shard_tensor_config.tensor_offset_device[device] = Offset(
                start, end)
        if device == -1:
            self.cpu_tensor = tensor

    def from_cpu_tensor(self, tensor):
        cur_pos = 0
        for device_id, budget in self.shard_tensor_config.device_memory_budget.items():
            if cur_pos >= tensor.shape[0]:
                break
            size = min(self.partition(tensor, budget),
                       tensor.shape[0] - cur_pos)
            if size <= 0:
                continue
            self.append(tensor[cur_pos:cur_pos + size], device_id)
            cur_pos += size
        if cur_pos < tensor.shape[0]:
            self.append(tensor[cur_pos:], -1)

    # -- gather ------------------------------------------------------------
    def _inaccessible_ranges(self):
        """[(start, end, device)] of shards a local gather cannot read."""
        if not torch.cuda.is_available():
            return []
        mask = self.shard_tensor.access_mask_on(self.current_device)
        ends = self.shard_tensor.shard_ends()
        devices = self.shard_tensor.shard_devices()
        out = []
        start = 0
        for i, end in enumerate(ends):
            if not (mask >> i) & 1:
                out.append((start, end, devices[i]))
            start = end
        return out

    def collect_device(self, input_orders, nodes, inter_device, start, end,
                       wait_results):
        mask = (nodes >= start) & (nodes < end)
        request_nodes = torch.masked_select(nodes, mask)
        if request_nodes.numel() == 0:
            return
        part_orders = torch.masked_select(input_orders, mask)
        request_nodes = request_nodes.to(inter_device)
        result = self.shard_tensor.gather_on(inter_device, request_nodes)
        wait_results.append((part_orders, result.to(self.current_device)))

    def gather_n(self, nodes, n_dev):
        """Upper-bound gather with device-side exact count (async path);
        requires every shard to be directly accessible from this device."""
        if self._inaccessible_ranges():
            raise RuntimeError("gather_n: some shards need the cross-clique "
                               "pass; use __getitem__")
        nodes = nodes.to(self.current_device)
        return self.shard_tensor.gather_n(nodes, n_dev)

    def __getitem__(self, nodes):
        nodes = nodes.to(self.current_device)
        feature = self.shard_tensor[nodes]
        missing = self._inaccessible_ranges()
        if missing:
            input_orders = torch.arange(nodes.size(0), dtype=torch.long,
                                        device=nodes.device)
            wait_results = []
            for start, end, dev in missing:
                self.collect_device(input_orders, nodes, dev, start, end,
                                    wait_results)
            for orders, result in wait_results:
                feature[orders] = result
        return feature

    # -- metadata ----------------------------------------------------------
    @property
    def shape(self):
        return torch.Size(self.shard_tensor.shape())

    @property
    def device(self):
        return self.current_device

    def size(self, dim):
        return self.shard_tensor.size(dim)

    # -- IPC ---------------------------------------------------------------
    def share_ipc(self):
        # serialize native items to plain tuples so the handle can cross
        # process boundaries via pickle
        items = [it.share_ipc() for it in self.shard_tensor.share_ipc()]
        if self.cpu_tensor is not None:
            self.cpu_tensor.share_memory_()
        return items, self.cpu_tensor, self.current_device

    def from_ipc_handle(self, gpu_ipc_list, cpu_tensor):
        for item in gpu_ipc_list:
            self.shard_tensor.append_item(_ext.ShardTensorItem.from_ipc(item))
        if cpu_tensor is not None:
            self.shard_tensor.append(cpu_tensor, -1)
            self.cpu_tensor = cpu_tensor

    @classmethod
    def new_from_share_ipc(cls, ipc_handles, current_device):
        gpu_ipc_list, cpu_tensor, _ = ipc_handles
        st = cls(current_device, ShardTensorConfig({}))
        st.from_ipc_handle(gpu_ipc_list, cpu_tensor)
        return st
