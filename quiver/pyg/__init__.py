from .sage_sampler import GraphSageSampler, MixedGraphSageSampler, SampleJob, Adj

__all__ = ["GraphSageSampler", "MixedGraphSageSampler", "SampleJob", "Adj"]
