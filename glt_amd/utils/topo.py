"""Topology conversions (COO <-> CSR/CSC) in pure torch.

The reference leans on torch_sparse.SparseTensor for this
(reference python/utils/topo.py:29-91); we implement the conversions with
torch.sort / bincount so there is no external sparse dependency, and we
keep per-row column order sorted — the negative sampler's binary-search
membership test relies on it.
"""
from typing import Optional, Tuple

import torch


def coo_to_csr(
    row: torch.Tensor,
    col: torch.Tensor,
    edge_id: Optional[torch.Tensor] = None,
    edge_weight: Optional[torch.Tensor] = None,
    num_rows: Optional[int] = None,
) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor],
           Optional[torch.Tensor]]:
    """Returns (indptr, indices, edge_ids, edge_weights) with indices sorted
    within each row.  edge_id defaults to the input COO edge position."""
    assert row.dim() == 1 and col.dim() == 1 and row.numel() == col.numel()
    row = row.long()
    col = col.long()
    if num_rows is None:
        num_rows = int(row.max()) + 1 if row.numel() > 0 else 0
    if edge_id is None:
        edge_id = torch.arange(row.numel(), dtype=torch.long,
                               device=row.device)
    # Sort by (row, col): stable two-pass sort.
    perm = torch.argsort(col, stable=True)
    row_s = row[perm]
    perm2 = torch.argsort(row_s, stable=True)
    perm = perm[perm2]
    indices = col[perm]
    counts = torch.bincount(row, minlength=num_rows)
    indptr = torch.zeros(num_rows + 1, dtype=torch.long, device=row.device)
    torch.cumsum(counts, 0, out=indptr[1:])
    eids = edge_id[perm]
    ew = edge_weight[perm] if edge_weight is not None else None
    return indptr, indices, eids, ew


def coo_to_csc(row, col, edge_id=None, edge_weight=None,
               num_cols: Optional[int] = None):
    """CSC of (row, col) == CSR of the reversed graph."""
    return coo_to_csr(col, row, edge_id, edge_weight, num_rows=num_cols)


def sort_csr_indices(indptr: torch.Tensor, indices: torch.Tensor,
                     *aux: Optional[torch.Tensor]):
    """Sort `indices` within each CSR row (plus parallel aux arrays)."""
    num_rows = indptr.numel() - 1
    row = torch.repeat_interleave(
        torch.arange(num_rows, device=indptr.device),
        indptr[1:] - indptr[:-1])
    key = row * (int(indices.max()) + 2 if indices.numel() else 1) + indices
    perm = torch.argsort(key)
    out = [indices[perm]]
    for a in aux:
        out.append(a[perm] if a is not None else None)
    return tuple(out)
