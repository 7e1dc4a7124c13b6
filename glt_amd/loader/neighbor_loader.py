"""NeighborLoader — the PyG drop-in local loader (parity: reference
python/loader/neighbor_loader.py:27-112)."""
from typing import List, Optional, Union

import torch

from ..data import Dataset
from ..sampler import NeighborSampler
from .node_loader import NodeLoader


class NeighborLoader(NodeLoader):
    def __init__(self, data: Dataset, num_neighbors: List[int], input_nodes,
                 batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 with_weight: bool = False,
                 device: Optional[torch.device] = None,
                 to_device: Optional[torch.device] = None,
                 edge_dir: Optional[str] = None, seed: Optional[int] = None,
                 as_pyg_v1: bool = False, prefetch: int = 0, **kwargs):
        edge_dir = edge_dir or data.edge_dir
        sampler = NeighborSampler(
            data.get_graph() if not isinstance(data.graph, dict)
            else data.graph,
            num_neighbors=num_neighbors, device=device, with_edge=with_edge,
            with_weight=with_weight, edge_dir=edge_dir, seed=seed)
        self.as_pyg_v1 = as_pyg_v1
        super().__init__(data, sampler, input_nodes, batch_size, shuffle,
                         drop_last, with_edge, to_device, prefetch)

    def __next__(self):
        data = super().__next__()
        if self.as_pyg_v1:
            from ..sampler.base import EdgeIndex

            n = data.node.numel()
            return (data.batch_size, data.node,
                    [EdgeIndex(data.edge_index, data.edge, (n, n))])
        return data
