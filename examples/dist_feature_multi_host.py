#!/usr/bin/env python3
"""Multi-host feature store (BASELINE config 4 building block).

Each host owns a probability-partitioned slice of the features
(quiver_partition_feature); a frontier gather dispatches ids by owning
host (PartitionInfo), exchanges id requests / feature responses over RCCL
(NcclComm.exchange) and serves local rows from the tiered Feature store.

Launch one process per GPU per host, e.g. on each host:
  torchrun --nnodes=2 --nproc-per-node=8 --master-addr=$MASTER \
      examples/dist_feature_multi_host.py
(also runs single-host for demonstration: torchrun --standalone
 --local-addr 127.0.0.1 --nproc-per-node=1 examples/...)

GPU-less demonstration (the exchange protocol is transport-agnostic —
gloo stands in for RCCL, a plain CPU tensor for the tiered store;
--rank-per-host 1 makes each process act as its own host):
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node=2 \
      examples/dist_feature_multi_host.py --rank-per-host 1
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import quiver


def main(nodes=200_000, dim=128, rank_per_host=None):
    use_gpu = torch.cuda.is_available()
    dist.init_process_group("nccl" if use_gpu else "gloo")
    rank = dist.get_rank()
    ws = dist.get_world_size()
    nproc_per_host = rank_per_host or int(
        os.environ.get("LOCAL_WORLD_SIZE",
                       os.environ.get("NPROC_PER_NODE", 1)))
    hosts = ws // nproc_per_host
    host = rank // nproc_per_host
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if use_gpu:
        torch.cuda.set_device(local_rank)

    # every host derives the same partition (seeded probs stand in for
    # sample_prob output on real data)
    g = torch.Generator().manual_seed(0)
    probs = [torch.rand(nodes, generator=g) for _ in range(hosts)]
    parts = quiver.partition.partition_without_replication("cpu", probs, None)
    global2host = torch.zeros(nodes, dtype=torch.int64)
    for h, part in enumerate(parts):
        global2host[part] = h

    feat_full = torch.arange(nodes, dtype=torch.float32).unsqueeze(1) \
        .repeat(1, dim)
    local_ids = parts[host]
    # order local rows by local id (PartitionInfo numbering)
    local_rows = feat_full[local_ids.sort().values]

    if use_gpu:
        feature = quiver.Feature(local_rank, device_list=[local_rank],
                                 device_cache_size="64M")
        feature.from_cpu_tensor(local_rows)
        # RCCL communicator: rank 0 creates the id, TCPStore-style bcast
        id_list = [quiver.getNcclId() if rank == 0 else None]
        dist.broadcast_object_list(id_list, src=0)
        comm = quiver.NcclComm(rank, ws, id_list[0], hosts=hosts,
                               rank_per_host=nproc_per_host)
        device = torch.device("cuda", torch.cuda.current_device())
    else:
        # GPU-less: the exchange protocol is transport-agnostic — gloo
        # stands in for RCCL and the local rows serve straight from a
        # CPU tensor (DistFeature only needs __getitem__/size)
        from quiver.comm import TorchDistComm
        feature = local_rows
        comm = TorchDistComm(hosts, nproc_per_host, device="cpu")
        device = torch.device("cpu")

    info = quiver.PartitionInfo(device, host, hosts, global2host)
    dist_feature = quiver.DistFeature(feature, info, comm)

    ids = torch.randint(0, nodes, (4096,),
                        generator=torch.Generator().manual_seed(rank))
    got = dist_feature[ids.to(device)]
    expect = feat_full[ids].to(device)
    assert torch.allclose(got, expect), "multi-host gather mismatch"
    print(f"rank {rank}: dist feature gather OK ({ids.numel()} ids, "
          f"{hosts} hosts)", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--rank-per-host", type=int, default=None)
    args = ap.parse_args()
    main(rank_per_host=args.rank_per_host)
