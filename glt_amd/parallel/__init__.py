"""Parallelism strategies index.

GLT's scaling axes and where they live in glt_amd:

- Data parallelism (DDP over RCCL/xGMI): standard torch DDP; bench.py and
  examples/multi_gpu wire it (one process per GPU).
- Graph partition parallelism: glt_amd.distributed (DistNeighborSampler
  fan-out + stitch over partitioned DistDatasets).
- Sampling<->training pipeline parallelism: subprocess producers + pinned
  shm channels (distributed/dist_sampling_producer.py) and the in-process
  side-stream prefetcher (loader/node_loader.py).
- Feature-storage parallelism over the 8-GPU xGMI node: DeviceGroup
  sharding (data/feature.py) and hip-IPC peer shards
  (data/xgmi_feature.py), both served by the UnifiedFeatureStore gather.
- Collective feature exchange: DistFeature.all2all_get (RCCL
  all_to_all_single), enabled with worker_options(use_all2all=True).

This module re-exports those entry points for discoverability.
"""
from ..data.feature import DeviceGroup
from ..data.xgmi_feature import XgmiShardedFeature
from ..distributed.dist_feature import DistFeature
from ..distributed.dist_neighbor_sampler import DistNeighborSampler
from ..distributed.dist_sampling_producer import (
    DistCollocatedSamplingProducer, DistMpSamplingProducer)

__all__ = [
    "DeviceGroup", "XgmiShardedFeature", "DistFeature",
    "DistNeighborSampler", "DistCollocatedSamplingProducer",
    "DistMpSamplingProducer",
]
