"""DistGraph: local partition graph + partition books (parity: reference
python/distributed/dist_graph.py)."""
from typing import Dict, Optional, Union

import torch

from ..data import Graph
from ..partition import PartitionBook
from ..typing import EdgeType, NodeType


class DistGraph:
    def __init__(self, num_partitions: int, partition_idx: int,
                 local_graph: Union[Graph, Dict[EdgeType, Graph]],
                 node_pb: Union[PartitionBook, Dict[NodeType,
                                                    PartitionBook]],
                 edge_pb: Union[PartitionBook, Dict[EdgeType,
                                                    PartitionBook]] = None):
        self.num_partitions = num_partitions
        self.partition_idx = partition_idx
        self.local_graph = local_graph
        self.node_pb = node_pb
        self.edge_pb = edge_pb
        self.data_cls = "hetero" if isinstance(local_graph, dict) else "homo"

    def get_graph(self, etype: Optional[EdgeType] = None):
        if self.data_cls == "hetero":
            return self.local_graph.get(etype) if etype is not None \
                else self.local_graph
        return self.local_graph

    def get_node_partitions(self, ids: torch.Tensor,
                            ntype: Optional[NodeType] = None):
        pb = self.node_pb[ntype] if isinstance(self.node_pb, dict) else \
            self.node_pb
        return pb[ids]

    def get_edge_partitions(self, eids: torch.Tensor,
                            etype: Optional[EdgeType] = None):
        pb = self.edge_pb[etype] if isinstance(self.edge_pb, dict) else \
            self.edge_pb
        return pb[eids]
