"""PyG-style remote backend for server-client mode.

Client-side FeatureStore/GraphStore adapters over the DistServer getters
(capability parity: reference dist_server.py:87-127 +
test/python/test_pyg_remote_backend.py).  When torch_geometric is
installed these duck-type its FeatureStore/GraphStore protocols closely
enough for NeighborLoader-style access; standalone they are plain remote
tensor accessors.
"""
from typing import List, Optional, Tuple

import torch

from ..typing import EdgeType, NodeType
from . import dist_client


class RemoteFeatureStore:
    """Fetch node features/labels from the assigned server(s)."""

    def __init__(self, server_rank: Optional[int] = None):
        ranks = ([server_rank] if server_rank is not None
                 else dist_client.get_assigned_servers())
        self.server_ranks = ranks

    def _server(self) -> int:
        return self.server_ranks[0]

    def get_tensor(self, ids: torch.Tensor,
                   ntype: Optional[NodeType] = None,
                   attr: str = "x") -> torch.Tensor:
        if attr == "y":
            return dist_client.request_server(
                self._server(), "get_node_label", ids.cpu(), ntype)
        return dist_client.request_server(
            self._server(), "get_node_feature", ids.cpu(), ntype)

    def get_tensor_size(self, ntype: Optional[NodeType] = None):
        return dist_client.request_server(self._server(),
                                          "get_tensor_size", ntype)

    def get_partition_id(self, ids: torch.Tensor,
                         ntype: Optional[NodeType] = None):
        return dist_client.request_server(
            self._server(), "get_node_partition_id", ids.cpu(), ntype)


class RemoteGraphStore:
    """Fetch graph topology from the assigned server(s)."""

    def __init__(self, server_rank: Optional[int] = None):
        ranks = ([server_rank] if server_rank is not None
                 else dist_client.get_assigned_servers())
        self.server_ranks = ranks

    def get_edge_index(self, etype: Optional[EdgeType] = None,
                       layout: str = "COO") -> torch.Tensor:
        return dist_client.request_server(self.server_ranks[0],
                                          "get_edge_index", etype, layout)

    def get_meta(self):
        return dist_client.request_server(self.server_ranks[0],
                                          "get_dataset_meta")
