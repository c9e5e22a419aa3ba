"""quiver — MI355X-native distributed GNN sampling and feature-collection.

Same public API surface as quiver-team/torch-quiver (reference
srcs/python/quiver/__init__.py), re-built from scratch for MI355X:
wave64 HIP sampling/gather kernels, hipIpc + xGMI peer loads, RCCL
collectives, zero-copy pinned-host tiers.
"""
from .feature import Feature, DistFeature, PartitionInfo, DeviceConfig
from .pyg import GraphSageSampler, MixedGraphSageSampler, SampleJob
from .utils import CSRTopo
from .utils import Topo as p2pCliqueTopo
from .utils import init_p2p
from .generate_neighbour_num import generate_neighbour_num
from .comm import NcclComm, getNcclId
from .partition import quiver_partition_feature, load_quiver_feature_partition
from .serving import (RequestBatcher, HybridSampler, InferenceServer,
                      InferenceServer_Debug)
from . import multiprocessing  # noqa: F401  registers ForkingPickler reducers
from . import nn  # model zoo (SAGE/GAT) — PyG-compatible layers
from . import trace  # pipeline-stage tracing (QUIVER_TRACE=1)
from .loader import TrainingPrefetcher

__version__ = "0.1.0"

__all__ = [
    "Feature", "DistFeature", "PartitionInfo", "DeviceConfig", "CSRTopo",
    "GraphSageSampler", "MixedGraphSageSampler", "SampleJob",
    "quiver_partition_feature", "load_quiver_feature_partition",
    "p2pCliqueTopo", "init_p2p", "getNcclId", "NcclComm",
    "RequestBatcher", "HybridSampler", "InferenceServer",
    "InferenceServer_Debug", "generate_neighbour_num", "nn",
    "TrainingPrefetcher",
]
