"""Common type aliases (parity: reference python/typing.py)."""
from typing import Dict, Tuple, Union

import torch

NodeType = str
EdgeType = Tuple[str, str, str]

# Reverse direction of an edge type (used for 'in' edge_dir sampling).
# Matches the reference convention (reference python/typing.py:39-46):
# self-loops of a single node type keep their relation name.
def reverse_edge_type(etype: EdgeType) -> EdgeType:
    src, rel, dst = etype
    if src == dst:
        return (dst, rel, src)
    if rel.split("_", 1)[0] == "rev":
        return (dst, rel.split("_", 1)[1], src)
    return (dst, "rev_" + rel, src)


def as_str(etype: Union[NodeType, EdgeType]) -> str:
    if isinstance(etype, (tuple, list)):
        return "__".join(etype)
    return etype


def str2etype(s: str) -> Union[NodeType, EdgeType]:
    parts = s.split("__")
    if len(parts) == 3:
        return tuple(parts)
    return s


TensorDataType = Union[torch.Tensor, Dict[str, torch.Tensor]]
