"""NodeLoader: iterate seed batches -> sampled PyG batches.

Parity: reference python/loader/node_loader.py:54-115 (seed DataLoader +
collate features/labels into Data/HeteroData).
"""
from typing import Optional, Union

import torch

from ..data import Dataset
from ..sampler import (BaseSampler, NodeSamplerInput, SamplerOutput,
                       HeteroSamplerOutput)
from ..utils.tracing import trace_region
from .transform import to_data, to_hetero_data


class _SeedIterator:
    def __init__(self, seeds: torch.Tensor, batch_size: int, shuffle: bool,
                 drop_last: bool, generator=None):
        self.seeds = seeds
        self.batch_size = batch_size
        self.drop_last = drop_last
        n = seeds.numel()
        if shuffle:
            self.order = torch.randperm(n, generator=generator)
        else:
            self.order = None
        self.pos = 0
        self.n = n

    def __iter__(self):
        return self

    def __next__(self):
        if self.pos >= self.n:
            raise StopIteration
        end = min(self.pos + self.batch_size, self.n)
        if self.drop_last and end - self.pos < self.batch_size:
            raise StopIteration
        idx = slice(self.pos, end)
        batch = (self.seeds[self.order[idx]] if self.order is not None
                 else self.seeds[idx])
        self.pos = end
        return batch

    def __len__(self):
        if self.drop_last:
            return self.n // self.batch_size
        return (self.n + self.batch_size - 1) // self.batch_size


class NodeLoader:
    def __init__(self, data: Dataset, node_sampler: BaseSampler,
                 input_nodes, batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 to_device: Optional[torch.device] = None):
        self.data = data
        self.sampler = node_sampler
        self.input_nodes = NodeSamplerInput.cast(
            input_nodes if not isinstance(input_nodes, tuple)
            else NodeSamplerInput(node=input_nodes[1],
                                  input_type=input_nodes[0]))
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.drop_last = drop_last
        self.with_edge = with_edge
        self.to_device = to_device

    def __iter__(self):
        self._it = _SeedIterator(self.input_nodes.node, self.batch_size,
                                 self.shuffle, self.drop_last)
        return self

    def __len__(self):
        return len(_SeedIterator(self.input_nodes.node, self.batch_size,
                                 False, self.drop_last))

    def __next__(self):
        seeds = next(self._it)
        inp = NodeSamplerInput(seeds, self.input_nodes.input_type)
        out = self.sampler.sample_from_nodes(inp)
        with trace_region("collate"):
            result = self._collate_fn(out)
        return result

    # -- feature/label collection ------------------------------------------
    def _collate_fn(self, out: Union[SamplerOutput, HeteroSamplerOutput]):
        if isinstance(out, HeteroSamplerOutput):
            return self._hetero_collate(out)
        x = None
        nf = self.data.get_node_feature()
        if nf is not None:
            x = nf[out.node]
        ef = None
        if self.data.get_edge_feature() is not None and out.edge is not None:
            ef = self.data.get_edge_feature()[out.edge]
        y = None
        labels = self.data.get_node_label()
        if labels is not None:
            seed_nodes = (out.batch if out.batch is not None else out.node)
            y = labels.to(seed_nodes.device)[seed_nodes] \
                if not labels.is_cuda else labels[seed_nodes]
        data = to_data(out, batch_labels=y, node_feats=x, edge_feats=ef)
        if self.to_device is not None:
            data = data.to(self.to_device, non_blocking=True)
        return data

    def _hetero_collate(self, out: HeteroSamplerOutput):
        node_feats = {}
        for nt, nodes in out.node.items():
            f = self.data.get_node_feature(nt)
            if f is not None and nodes.numel() > 0:
                node_feats[nt] = f[nodes]
        edge_feats = {}
        if out.edge is not None:
            for et, eids in out.edge.items():
                f = self.data.get_edge_feature(et)
                if f is not None and eids.numel() > 0:
                    edge_feats[et] = f[eids]
        labels = {}
        for nt, b in (out.batch or {}).items():
            lab = self.data.get_node_label(nt)
            if lab is not None:
                labels[nt] = lab.to(b.device)[b] if not lab.is_cuda \
                    else lab[b]
        data = to_hetero_data(out, batch_label_dict=labels or None,
                              node_feat_dict=node_feats or None,
                              edge_feat_dict=edge_feats or None,
                              edge_dir=getattr(self.sampler, "edge_dir",
                                               "out"))
        if self.to_device is not None:
            data = data.to(self.to_device, non_blocking=True)
        return data
