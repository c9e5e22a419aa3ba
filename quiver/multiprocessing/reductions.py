"""ForkingPickler reducers: Feature and samplers cross process boundaries
as IPC handles (parity: reference quiver/multiprocessing/reductions.py)."""
from multiprocessing.reduction import ForkingPickler

import quiver


def rebuild_feature(ipc_handle):
    return quiver.Feature.lazy_from_ipc_handle(ipc_handle)


def reduce_feature(feature):
    return (rebuild_feature, (feature.share_ipc(),))


def rebuild_pyg_sampler(cls, ipc_handle):
    return cls.lazy_from_ipc_handle(ipc_handle)


def reduce_pyg_sampler(sampler):
    return (rebuild_pyg_sampler, (type(sampler), sampler.share_ipc()))


def init_reductions():
    ForkingPickler.register(quiver.Feature, reduce_feature)
    ForkingPickler.register(quiver.pyg.GraphSageSampler, reduce_pyg_sampler)
    ForkingPickler.register(quiver.pyg.MixedGraphSageSampler,
                            reduce_pyg_sampler)
