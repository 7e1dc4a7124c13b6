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
            # PyG-v1 contract: one EdgeIndex per layer, deepest hop first,
            # sizes = (num rows of the layer input, num rows of its output)
            from ..sampler.base import EdgeIndex

            nsn = [int(v) for v in (data.num_sampled_nodes or [])]
            nse = [int(v) for v in (data.num_sampled_edges or [])]
            if nsn and nse:
                adjs = []
                n_pref = [sum(nsn[:i + 1]) for i in range(len(nsn))]
                e_off = 0
                for h, ne in enumerate(nse):
                    ei = data.edge_index[:, e_off:e_off + ne]
                    eid = (data.edge[e_off:e_off + ne]
                           if data.edge is not None else None)
                    size = (n_pref[min(h + 1, len(n_pref) - 1)], n_pref[h])
                    adjs.append(EdgeIndex(ei, eid, size))
                    e_off += ne
                adjs = adjs[::-1]
            else:
                n = data.node.numel()
                adjs = [EdgeIndex(data.edge_index, data.edge, (n, n))]
            return data.batch_size, data.node, adjs
        return data
