"""DistNeighborLoader / DistLinkNeighborLoader / DistSubGraphLoader
(parity: reference python/distributed/dist_neighbor_loader.py,
dist_link_neighbor_loader.py, dist_subgraph_loader.py)."""
from typing import List, Optional, Union

import torch

from ..sampler import (EdgeSamplerInput, NegativeSampling, NodeSamplerInput,
                       RemoteSamplerInput, SamplingConfig, SamplingType)
from .dist_dataset import DistDataset
from .dist_loader import DistLoader


class DistNeighborLoader(DistLoader):
    def __init__(self, data: Optional[DistDataset],
                 num_neighbors: List[int], input_nodes,
                 batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 with_weight: bool = False, edge_dir: str = "out",
                 collect_features: bool = True,
                 to_device: Optional[torch.device] = None,
                 worker_options=None):
        if isinstance(input_nodes, RemoteSamplerInput):
            inp = input_nodes  # server resolves the seed split/path
        elif isinstance(input_nodes, tuple) and not isinstance(
                input_nodes, NodeSamplerInput):
            inp = NodeSamplerInput(node=torch.as_tensor(input_nodes[1]),
                                   input_type=input_nodes[0])
        elif isinstance(input_nodes, NodeSamplerInput):
            inp = input_nodes
        else:
            inp = NodeSamplerInput(node=torch.as_tensor(input_nodes))
        config = SamplingConfig(
            sampling_type=SamplingType.NODE, num_neighbors=num_neighbors,
            batch_size=batch_size, shuffle=shuffle, drop_last=drop_last,
            with_edge=with_edge, collect_features=collect_features,
            with_weight=with_weight, edge_dir=edge_dir)
        super().__init__(data, inp, config, to_device, worker_options)


class DistLinkNeighborLoader(DistLoader):
    def __init__(self, data: Optional[DistDataset],
                 num_neighbors: List[int], edge_label_index,
                 edge_label=None, neg_sampling=None, batch_size: int = 1,
                 shuffle: bool = False, drop_last: bool = False,
                 with_edge: bool = False, with_weight: bool = False,
                 edge_dir: str = "out", collect_features: bool = True,
                 to_device: Optional[torch.device] = None,
                 worker_options=None):
        if isinstance(edge_label_index, tuple) and isinstance(
                edge_label_index[0], (tuple, str)):
            input_type, eli = edge_label_index
        else:
            input_type, eli = None, edge_label_index
        eli = torch.as_tensor(eli)
        if isinstance(neg_sampling, str):
            neg_sampling = NegativeSampling(neg_sampling)
        inp = EdgeSamplerInput(row=eli[0], col=eli[1], label=edge_label,
                               input_type=input_type,
                               neg_sampling=neg_sampling)
        config = SamplingConfig(
            sampling_type=SamplingType.LINK, num_neighbors=num_neighbors,
            batch_size=batch_size, shuffle=shuffle, drop_last=drop_last,
            with_edge=with_edge, collect_features=collect_features,
            with_neg=neg_sampling is not None, with_weight=with_weight,
            edge_dir=edge_dir)
        super().__init__(data, inp, config, to_device, worker_options)


class DistSubGraphLoader(DistLoader):
    def __init__(self, data: Optional[DistDataset], input_nodes,
                 num_neighbors: Optional[List[int]] = None,
                 batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 edge_dir: str = "out", collect_features: bool = True,
                 to_device: Optional[torch.device] = None,
                 worker_options=None):
        inp = NodeSamplerInput(node=torch.as_tensor(input_nodes))
        config = SamplingConfig(
            sampling_type=SamplingType.SUBGRAPH,
            num_neighbors=num_neighbors, batch_size=batch_size,
            shuffle=shuffle, drop_last=drop_last, with_edge=with_edge,
            collect_features=collect_features, edge_dir=edge_dir)
        super().__init__(data, inp, config, to_device, worker_options)
