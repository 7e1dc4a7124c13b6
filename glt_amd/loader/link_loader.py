"""LinkLoader / LinkNeighborLoader (parity: reference
python/loader/link_loader.py:100-198, link_neighbor_loader.py)."""
from typing import List, Optional, Union

import torch

from ..data import Dataset
from ..sampler import (BaseSampler, EdgeSamplerInput, NegativeSampling,
                       NeighborSampler)
from .node_loader import NodeLoader


class _EdgeSeedIterator:
    def __init__(self, row, col, label, batch_size, shuffle, drop_last):
        self.row, self.col, self.label = row, col, label
        self.batch_size = batch_size
        self.drop_last = drop_last
        n = row.numel()
        self.order = torch.randperm(n) if shuffle else None
        self.pos, self.n = 0, n

    def __iter__(self):
        return self

    def __len__(self):
        if self.drop_last:
            return self.n // self.batch_size
        return (self.n + self.batch_size - 1) // self.batch_size

    def __next__(self):
        if self.pos >= self.n:
            raise StopIteration
        end = min(self.pos + self.batch_size, self.n)
        if self.drop_last and end - self.pos < self.batch_size:
            raise StopIteration
        sl = (self.order[self.pos:end] if self.order is not None
              else slice(self.pos, end))
        self.pos = end
        return (self.row[sl], self.col[sl],
                self.label[sl] if self.label is not None else None)


class LinkLoader(NodeLoader):
    """Iterates seed edges; supports binary/triplet negative sampling."""

    def __init__(self, data: Dataset, link_sampler: BaseSampler,
                 edge_label_index, edge_label=None,
                 neg_sampling: Optional[Union[NegativeSampling, str]] = None,
                 batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 to_device: Optional[torch.device] = None,
                 prefetch: int = 0):
        self.data = data
        self.sampler = link_sampler
        if isinstance(edge_label_index, tuple):
            # hetero: (edge_type, [2, E] tensor)
            self.input_type, eli = edge_label_index
        else:
            self.input_type, eli = None, edge_label_index
        self.edge_row, self.edge_col = eli[0], eli[1]
        self.edge_label = edge_label
        if isinstance(neg_sampling, str):
            neg_sampling = NegativeSampling(neg_sampling)
        self.neg_sampling = neg_sampling
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.drop_last = drop_last
        self.with_edge = with_edge
        self.to_device = to_device
        self.prefetch = prefetch
        self._prefetcher = None

    def __iter__(self):
        self._it = _EdgeSeedIterator(self.edge_row, self.edge_col,
                                     self.edge_label, self.batch_size,
                                     self.shuffle, self.drop_last)
        if self._prefetcher is not None:
            self._prefetcher.stop()
            self._prefetcher = None
        if self.prefetch > 0 and torch.cuda.is_available():
            from .node_loader import _Prefetcher

            self._prefetcher = _Prefetcher(self, self._it, self.prefetch)
        return self

    def __len__(self):
        return len(_EdgeSeedIterator(self.edge_row, self.edge_col, None,
                                     self.batch_size, False, self.drop_last))

    def _produce(self, item):
        row, col, label = item
        inp = EdgeSamplerInput(row=row, col=col, label=label,
                               input_type=self.input_type,
                               neg_sampling=self.neg_sampling)
        out = self.sampler.sample_from_edges(inp)
        return self._collate_fn(out)

    def __next__(self):
        if self._prefetcher is not None:
            return self._prefetcher.next()
        return self._produce(next(self._it))

    def shutdown(self):
        if self._prefetcher is not None:
            self._prefetcher.stop()
            self._prefetcher = None

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass


class LinkNeighborLoader(LinkLoader):
    def __init__(self, data: Dataset, num_neighbors: List[int],
                 edge_label_index, edge_label=None, neg_sampling=None,
                 batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 with_weight: bool = False,
                 device: Optional[torch.device] = None,
                 to_device: Optional[torch.device] = None,
                 edge_dir: Optional[str] = None, seed: Optional[int] = None,
                 prefetch: int = 0, **kwargs):
        edge_dir = edge_dir or data.edge_dir
        sampler = NeighborSampler(
            data.get_graph() if not isinstance(data.graph, dict)
            else data.graph,
            num_neighbors=num_neighbors, device=device, with_edge=with_edge,
            with_weight=with_weight, edge_dir=edge_dir, seed=seed)
        super().__init__(data, sampler, edge_label_index, edge_label,
                         neg_sampling, batch_size, shuffle, drop_last,
                         with_edge, to_device, prefetch=prefetch)
