from .base import (BaseSampler, EdgeIndex, EdgeSamplerInput,
                   HeteroSamplerOutput, NegativeSampling,
                   NegativeSamplingMode, NeighborOutput, NodeSamplerInput,
                   RemoteSamplerInput, SamplerOutput, SamplingConfig,
                   SamplingType)
from .neighbor_sampler import NeighborSampler
from .negative_sampler import RandomNegativeSampler

__all__ = [
    "BaseSampler", "EdgeIndex", "EdgeSamplerInput", "HeteroSamplerOutput",
    "NegativeSampling", "NegativeSamplingMode", "NeighborOutput",
    "NodeSamplerInput", "RemoteSamplerInput", "SamplerOutput",
    "SamplingConfig", "SamplingType",
    "NeighborSampler", "RandomNegativeSampler",
]
