"""Sampler input/output dataclasses + sampling config.

API parity: reference python/sampler/base.py:43-462 (PyG-compatible
NodeSamplerInput / EdgeSamplerInput / SamplerOutput / HeteroSamplerOutput /
NeighborOutput / NegativeSampling / SamplingType / SamplingConfig).
"""
import math
from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Dict, List, NamedTuple, Optional, Tuple, Union

import torch

from ..typing import EdgeType, NodeType


class EdgeIndex(NamedTuple):
    """PyG v1-style (edge_index, e_id, size) triple."""
    edge_index: torch.Tensor
    e_id: Optional[torch.Tensor]
    size: Tuple[int, int]

    def to(self, *args, **kwargs):
        return EdgeIndex(
            self.edge_index.to(*args, **kwargs),
            self.e_id.to(*args, **kwargs) if self.e_id is not None else None,
            self.size)


@dataclass
class NodeSamplerInput:
    node: torch.Tensor
    input_type: Optional[NodeType] = None

    def __getitem__(self, index) -> "NodeSamplerInput":
        return NodeSamplerInput(self.node[index], self.input_type)

    def __len__(self):
        return self.node.numel()

    def share_memory(self):
        self.node.share_memory_()
        return self

    def to(self, device):
        return NodeSamplerInput(self.node.to(device), self.input_type)

    @classmethod
    def cast(cls, x):
        if isinstance(x, cls):
            return x
        return cls(torch.as_tensor(x))


class NegativeSamplingMode(Enum):
    binary = "binary"
    triplet = "triplet"


class NegativeSampling:
    def __init__(self, mode: Union[NegativeSamplingMode, str],
                 amount: Union[int, float] = 1,
                 weight: Optional[torch.Tensor] = None):
        self.mode = NegativeSamplingMode(mode)
        self.amount = amount
        self.weight = weight
        if self.is_triplet() and isinstance(self.amount, float):
            self.amount = math.ceil(self.amount)

    def is_binary(self):
        return self.mode == NegativeSamplingMode.binary

    def is_triplet(self):
        return self.mode == NegativeSamplingMode.triplet

    def share_memory(self):
        if self.weight is not None:
            self.weight.share_memory_()
        return self

    def to(self, device):
        return NegativeSampling(
            self.mode, self.amount,
            self.weight.to(device) if self.weight is not None else None)


@dataclass
class EdgeSamplerInput:
    row: torch.Tensor
    col: torch.Tensor
    label: Optional[torch.Tensor] = None
    input_type: Optional[EdgeType] = None
    neg_sampling: Optional[NegativeSampling] = None

    def __getitem__(self, index) -> "EdgeSamplerInput":
        return EdgeSamplerInput(
            self.row[index], self.col[index],
            self.label[index] if self.label is not None else None,
            self.input_type, self.neg_sampling)

    def __len__(self):
        return self.row.numel()

    def share_memory(self):
        self.row.share_memory_()
        self.col.share_memory_()
        if self.label is not None:
            self.label.share_memory_()
        return self

    def to(self, device):
        return EdgeSamplerInput(
            self.row.to(device), self.col.to(device),
            self.label.to(device) if self.label is not None else None,
            self.input_type, self.neg_sampling)


@dataclass
class SamplerOutput:
    node: torch.Tensor
    row: torch.Tensor
    col: torch.Tensor
    edge: Optional[torch.Tensor] = None
    batch: Optional[torch.Tensor] = None
    num_sampled_nodes: Optional[Union[List[int], torch.Tensor]] = None
    num_sampled_edges: Optional[Union[List[int], torch.Tensor]] = None
    device: Optional[torch.device] = None
    metadata: Optional[Any] = None

    def to(self, device):
        def mv(t):
            return t.to(device) if torch.is_tensor(t) else t

        return SamplerOutput(mv(self.node), mv(self.row), mv(self.col),
                             mv(self.edge), mv(self.batch),
                             self.num_sampled_nodes, self.num_sampled_edges,
                             device, self.metadata)


@dataclass
class HeteroSamplerOutput:
    node: Dict[NodeType, torch.Tensor]
    row: Dict[EdgeType, torch.Tensor]
    col: Dict[EdgeType, torch.Tensor]
    edge: Optional[Dict[EdgeType, torch.Tensor]] = None
    batch: Optional[Dict[NodeType, torch.Tensor]] = None
    num_sampled_nodes: Optional[Dict[NodeType, Any]] = None
    num_sampled_edges: Optional[Dict[EdgeType, Any]] = None
    edge_types: Optional[List[EdgeType]] = None
    input_type: Optional[Union[NodeType, EdgeType]] = None
    device: Optional[torch.device] = None
    metadata: Optional[Any] = None

    def get_edge_index(self) -> Dict[EdgeType, torch.Tensor]:
        return {k: torch.stack([v, self.col[k]]) for k, v in self.row.items()}


@dataclass
class NeighborOutput:
    nbr: torch.Tensor
    nbr_num: torch.Tensor
    edge: Optional[torch.Tensor] = None

    def to(self, device):
        return NeighborOutput(
            self.nbr.to(device), self.nbr_num.to(device),
            self.edge.to(device) if self.edge is not None else None)


class SamplingType(Enum):
    NODE = 0
    LINK = 1
    SUBGRAPH = 2
    RANDOM_WALK = 3


@dataclass
class SamplingConfig:
    sampling_type: SamplingType = SamplingType.NODE
    num_neighbors: Optional[List[int]] = None
    batch_size: int = 1
    shuffle: bool = False
    drop_last: bool = False
    with_edge: bool = False
    collect_features: bool = False
    with_neg: bool = False
    with_weight: bool = False
    edge_dir: str = "out"
    seed_stride: int = 1  # random-walk length when RANDOM_WALK


@dataclass
class RemoteSamplerInput:
    """Seed specification resolved on the SERVER side (server-client mode):
    either a named dataset split ('train'/'val'/'test') or a .pt file path
    readable by the server (parity: reference sampler/base.py
    RemoteSamplerInput semantics).
    """
    split: Optional[str] = None
    path: Optional[str] = None
    input_type: Optional[NodeType] = None

    def resolve(self, dataset) -> "NodeSamplerInput":
        import torch as _torch

        if self.path:
            seeds = _torch.load(self.path, weights_only=False)
        else:
            attr = {"train": "train_idx", "val": "val_idx",
                    "test": "test_idx"}[self.split or "train"]
            seeds = getattr(dataset, attr)
            if seeds is None:
                raise ValueError(
                    f"dataset has no '{self.split}' split on the server")
        return NodeSamplerInput(node=_torch.as_tensor(seeds),
                                input_type=self.input_type)

    def __len__(self):  # unknown client-side; producers resolve remotely
        return 0


class BaseSampler:
    """Abstract sampler interface (parity: reference sampler/base.py:444+)."""

    def sample_from_nodes(self, inputs: NodeSamplerInput, **kwargs):
        raise NotImplementedError

    def sample_from_edges(self, inputs: EdgeSamplerInput, **kwargs):
        raise NotImplementedError

    @property
    def edge_permutation(self):
        return None
