"""Partition books (parity: reference python/partition/partition_book.py)."""
from typing import Union

import torch


class PartitionBook:
    def __getitem__(self, indices: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    @property
    def offset(self) -> int:
        return 0

    @property
    def device(self):
        return torch.device("cpu")


class GLTPartitionBook(PartitionBook):
    """Dense id -> partition tensor."""

    def __init__(self, book: torch.Tensor):
        self.book = book.to(torch.uint8) if book.dtype != torch.uint8 \
            else book

    def __getitem__(self, indices):
        return self.book[indices.cpu().long()].to(torch.long)

    def __len__(self):
        return self.book.numel()

    def to(self, device):
        self.book = self.book.to(device)
        return self

    def share_memory_(self):
        if not self.book.is_cuda:
            self.book.share_memory_()
        return self


class RangePartitionBook(PartitionBook):
    """Contiguous ranges: partition p owns ids [bounds[p-1], bounds[p])."""

    def __init__(self, bounds: torch.Tensor, partition_idx: int):
        assert bounds.dim() == 1
        self.partition_bounds = bounds.long()
        self.partition_idx = partition_idx
        self._start = 0 if partition_idx == 0 else \
            int(bounds[partition_idx - 1])

    def __getitem__(self, indices):
        return torch.searchsorted(self.partition_bounds,
                                  indices.cpu().long(), right=True)

    def __len__(self):
        return int(self.partition_bounds[-1])

    @property
    def offset(self) -> int:
        return self._start

    def id2index(self, ids: torch.Tensor) -> torch.Tensor:
        return ids.long() - self._start

    def share_memory_(self):
        self.partition_bounds.share_memory_()
        return self
