"""Minimal torch_geometric-compatible Data / HeteroData containers.

Only the surface the glt_amd loaders and models touch: free-form tensor
attributes, ``edge_index``, ``num_nodes``, ``.to()/.cpu()/.pin_memory()``,
hetero node/edge stores indexable by type.
"""
from typing import Any, Dict, Iterator, Optional, Tuple

import torch

from ..typing import EdgeType, NodeType


def _apply(value, fn):
    if torch.is_tensor(value):
        return fn(value)
    if isinstance(value, dict):
        return {k: _apply(v, fn) for k, v in value.items()}
    if isinstance(value, (list, tuple)):
        return type(value)(_apply(v, fn) for v in value)
    return value


class Data:
    def __init__(self, x: Optional[torch.Tensor] = None,
                 edge_index: Optional[torch.Tensor] = None,
                 y: Optional[torch.Tensor] = None, **kwargs):
        self._store: Dict[str, Any] = {}
        if x is not None:
            self.x = x
        if edge_index is not None:
            self.edge_index = edge_index
        if y is not None:
            self.y = y
        for k, v in kwargs.items():
            setattr(self, k, v)

    # -- attribute bag ------------------------------------------------------
    def __setattr__(self, key, value):
        if key.startswith("_"):
            super().__setattr__(key, value)
        else:
            self._store[key] = value

    def __getattr__(self, key):
        store = self.__dict__.get("_store")
        if store is not None and key in store:
            return store[key]
        if key.startswith("_") or key in ("num_nodes", "num_edges"):
            raise AttributeError(key)
        return None

    def __delattr__(self, key):
        self._store.pop(key, None)

    def __contains__(self, key):
        return key in self._store

    def __getitem__(self, key):
        return self._store[key]

    def __setitem__(self, key, value):
        self._store[key] = value

    def keys(self):
        return list(self._store.keys())

    def items(self):
        return self._store.items()

    # -- sizes --------------------------------------------------------------
    @property
    def num_nodes(self) -> Optional[int]:
        if "num_nodes" in self._store:
            return self._store["num_nodes"]
        x = self._store.get("x")
        if x is not None:
            return x.size(0)
        n = self._store.get("node")
        if n is not None:
            return n.size(0)
        ei = self._store.get("edge_index")
        if ei is not None and ei.numel() > 0:
            return int(ei.max()) + 1
        return None

    @num_nodes.setter
    def num_nodes(self, v):
        self._store["num_nodes"] = v

    @property
    def num_edges(self) -> int:
        ei = self._store.get("edge_index")
        return 0 if ei is None else ei.size(1)

    # -- transforms ---------------------------------------------------------
    def to(self, device, non_blocking: bool = False):
        for k, v in list(self._store.items()):
            self._store[k] = _apply(
                v, lambda t: t.to(device, non_blocking=non_blocking))
        return self

    def cpu(self):
        return self.to(torch.device("cpu"))

    def cuda(self, device=None, non_blocking: bool = True):
        return self.to(device or torch.device("cuda"),
                       non_blocking=non_blocking)

    def pin_memory(self):
        for k, v in list(self._store.items()):
            self._store[k] = _apply(
                v, lambda t: t.pin_memory() if not t.is_cuda else t)
        return self

    def __repr__(self):
        parts = []
        for k, v in self._store.items():
            if torch.is_tensor(v):
                parts.append(f"{k}={list(v.shape)}")
            else:
                parts.append(f"{k}={v}")
        return f"Data({', '.join(parts)})"


def apply_to_tensors(data, fn):
    """Apply fn to every tensor held by a Data or HeteroData in place."""
    if isinstance(data, HeteroData):
        for s in list(data._node_stores.values()) + \
                list(data._edge_stores.values()):
            apply_to_tensors(s, fn)
        for k, v in list(data._global.items()):
            data._global[k] = _apply(v, fn)
        return data
    for k, v in list(data._store.items()):
        data._store[k] = _apply(v, fn)
    return data


class _TypeStore(Data):
    """Per-node-type / per-edge-type store inside HeteroData."""


class HeteroData:
    def __init__(self):
        object.__setattr__(self, "_node_stores", {})
        object.__setattr__(self, "_edge_stores", {})
        object.__setattr__(self, "_global", {})

    def __getitem__(self, key) -> _TypeStore:
        if isinstance(key, tuple):
            key = tuple(key)
            return self._edge_stores.setdefault(key, _TypeStore())
        return self._node_stores.setdefault(key, _TypeStore())

    def __setattr__(self, key, value):
        self._global[key] = value

    def __getattr__(self, key):
        g = object.__getattribute__(self, "_global")
        if key in g:
            return g[key]
        if key.startswith("_"):
            raise AttributeError(key)
        return None

    @property
    def node_types(self):
        return list(self._node_stores.keys())

    @property
    def edge_types(self):
        return list(self._edge_stores.keys())

    def node_items(self) -> Iterator[Tuple[NodeType, _TypeStore]]:
        return self._node_stores.items()

    def edge_items(self) -> Iterator[Tuple[EdgeType, _TypeStore]]:
        return self._edge_stores.items()

    @property
    def x_dict(self):
        return {k: s.x for k, s in self._node_stores.items() if s.x is not None}

    @property
    def edge_index_dict(self):
        return {k: s.edge_index for k, s in self._edge_stores.items()
                if s.edge_index is not None}

    def to(self, device, non_blocking: bool = False):
        for s in self._node_stores.values():
            s.to(device, non_blocking=non_blocking)
        for s in self._edge_stores.values():
            s.to(device, non_blocking=non_blocking)
        for k, v in list(self._global.items()):
            self._global[k] = _apply(
                v, lambda t: t.to(device, non_blocking=non_blocking))
        return self

    def cpu(self):
        return self.to(torch.device("cpu"))

    def cuda(self, device=None, non_blocking: bool = True):
        return self.to(device or torch.device("cuda"),
                       non_blocking=non_blocking)

    def __repr__(self):
        return (f"HeteroData(nodes={list(self._node_stores)}, "
                f"edges={list(self._edge_stores)})")
