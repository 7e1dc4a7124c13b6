"""Thin python wrapper over the native UnifiedFeatureStore.

Parity: reference python/data/unified_tensor.py (UnifiedTensor wrapper over
csrc UnifiedTensor).  Rows are appended as device segments or host tensors
(auto-mapped); gather dispatches to the wave-per-row HIP kernel.
"""
from typing import List, Optional

import torch


class UnifiedTensor:
    def __init__(self, device: int, dtype: torch.dtype = torch.float32):
        from .. import _C

        self._C = _C
        self.device = device
        self.dtype = dtype
        self._store = _C.UnifiedFeatureStore(device)
        self._keepalive: List[torch.Tensor] = []

    def append_device_tensor(self, t: torch.Tensor):
        assert t.is_cuda
        self._store.append(t.contiguous())
        self._keepalive.append(t)

    def append_cpu_tensor(self, t: torch.Tensor):
        t = t.contiguous()
        mapped = self._C.host_mapped_view(t, self.device)
        self._store.append(mapped)
        self._keepalive += [t, mapped]

    def __getitem__(self, ids: torch.Tensor) -> torch.Tensor:
        return self._store.gather(
            ids.to(torch.device("cuda", self.device), non_blocking=True))

    @property
    def shape(self):
        return (self._store.rows(), self._store.dim())

    def size(self, dim: int):
        return self.shape[dim]
