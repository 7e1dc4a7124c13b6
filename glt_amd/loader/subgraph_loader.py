"""SubGraphLoader (parity: reference python/loader/subgraph_loader.py):
induces the full edge set among (optionally expanded) seed node batches."""
from typing import List, Optional

import torch

from ..data import Dataset
from ..sampler import NeighborSampler, NodeSamplerInput
from .node_loader import NodeLoader, _SeedIterator


class SubGraphLoader(NodeLoader):
    def __init__(self, data: Dataset, input_nodes,
                 num_neighbors: Optional[List[int]] = None,
                 batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 device: Optional[torch.device] = None,
                 to_device: Optional[torch.device] = None,
                 seed: Optional[int] = None, **kwargs):
        sampler = NeighborSampler(data.get_graph(),
                                  num_neighbors=num_neighbors, device=device,
                                  with_edge=with_edge, seed=seed)
        super().__init__(data, sampler, input_nodes, batch_size, shuffle,
                         drop_last, with_edge, to_device)

    def __next__(self):
        seeds = next(self._it)
        out = self.sampler.subgraph(NodeSamplerInput(seeds))
        return self._collate_fn(out)
