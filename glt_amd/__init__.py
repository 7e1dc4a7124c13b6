"""glt_amd — MI355X-native graph learning engine.

A from-scratch GNN sampling-and-training framework for AMD Instinct MI355X
(CDNA4 / gfx950) with the capabilities of alibaba/graphlearn-for-pytorch:
GPU neighbor/negative/subgraph sampling as hand-written HIP kernels, a
tiered feature store over HBM3E + xGMI peers + pinned-host UVA, and a
distributed runtime (partitioned graphs, RPC sampling workers, shm sample
channels, RCCL collectives).  See SURVEY.md for the structural map.
"""
__version__ = "0.1.0"

import os

# RCCL over xGMI: the host driver supports dmabuf IPC only.
os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
# Quiet LibTorch warning spam (reference python/__init__.py:19 behavior).
os.environ.setdefault("TORCH_CPP_LOG_LEVEL", "ERROR")

import torch  # noqa: F401  (loads libc10/libtorch before our extension)

from . import _C  # native core (built in-tree; fail loudly if missing)
from . import channel, data, distributed, loader, models, parallel, \
    partition, sampler, utils
from .data import Dataset, DeviceGroup, Feature, Graph, Topology
from .loader import (LinkLoader, LinkNeighborLoader, NeighborLoader,
                     NodeLoader, SubGraphLoader)
from .sampler import (EdgeSamplerInput, NegativeSampling, NeighborSampler,
                      NodeSamplerInput, RandomNegativeSampler,
                      SamplingConfig, SamplingType)
from .typing import EdgeType, NodeType, as_str, reverse_edge_type
from .utils import seed_everything

__all__ = [
    "_C", "data", "loader", "models", "partition", "sampler", "utils",
    "Dataset", "DeviceGroup", "Feature", "Graph", "Topology",
    "LinkLoader", "LinkNeighborLoader", "NeighborLoader", "NodeLoader",
    "SubGraphLoader", "EdgeSamplerInput", "NegativeSampling",
    "NeighborSampler", "NodeSamplerInput", "RandomNegativeSampler",
    "SamplingConfig", "SamplingType", "seed_everything",
]
