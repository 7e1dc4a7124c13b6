"""Hot-feature reordering (parity: reference python/data/reorder.py:19-36)."""
from typing import Tuple

import torch

from .graph import Topology


def sort_by_in_degree(
    feature: torch.Tensor, split_ratio: float, topo: Topology
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Reorder feature rows so high in-degree (hot) rows come first.

    Returns (reordered_features, id2index) where id2index maps the original
    node id to its new row position.
    """
    # in-degree of node v = number of times v appears as a column
    indeg = torch.bincount(topo.indices, minlength=topo.num_nodes)
    order = torch.argsort(indeg, descending=True, stable=True)
    id2index = torch.empty_like(order)
    id2index[order] = torch.arange(order.numel(), device=order.device)
    return feature[order].contiguous(), id2index
