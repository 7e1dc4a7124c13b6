"""Random (hash-free, shuffled-chunk) node partitioner (parity: reference
python/partition/random_partitioner.py:63-86)."""
from typing import Dict, Union

import torch

from .base import PartitionerBase
from .partition_book import GLTPartitionBook


class RandomPartitioner(PartitionerBase):
    def _partition_node_ids(self, ntype=None):
        n = (self.num_nodes[ntype] if isinstance(self.num_nodes, dict)
             else self.num_nodes)
        perm = torch.randperm(n)
        per = (n + self.num_parts - 1) // self.num_parts
        node_pb = torch.zeros(n, dtype=torch.uint8)
        ids_list = []
        for p in range(self.num_parts):
            ids = perm[p * per:(p + 1) * per]
            node_pb[ids] = p
            ids_list.append(ids.sort().values)
        return ids_list, GLTPartitionBook(node_pb)
