"""Autograd wrapper over the fused HIP segment-mean kernel.

Used by SAGEConv when the batch lives on the GPU and edges are sorted by
target (the glt_amd sampler's native edge order); falls back to torch
index_add on CPU or unsorted input.
"""
import torch

_boundary_cache = {}


def _boundaries(n: int, device) -> torch.Tensor:
    """Cached 0..n arange per device (saves one kernel + one python op per
    conv call; the step is launch-bound)."""
    key = (device.type, device.index)
    buf = _boundary_cache.get(key)
    if buf is None or buf.numel() <= n:
        buf = torch.arange(max(n + 1, 1 << 20), device=device)
        _boundary_cache[key] = buf
    return buf[: n + 1]



class _SegmentMean(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, col, offsets, n_src):
        from .. import _C

        ctx.save_for_backward(col, offsets)
        ctx.n_src = n_src
        return _C.segment_mean_fwd(x, col, offsets, offsets.numel() - 1)

    @staticmethod
    def backward(ctx, dy):
        from .. import _C

        col, offsets = ctx.saved_tensors
        # kernel accumulates in an fp32 arena regardless of dy dtype
        dx = _C.segment_mean_bwd(dy.contiguous(), col, offsets, ctx.n_src)
        if dx.dtype != dy.dtype:
            dx = dx.to(dy.dtype)
        return dx, None, None, None


class _SegmentMeanCat(torch.autograd.Function):
    """Fused ``[segment_mean(x) | x[:n_tgt]]`` — one kernel instead of
    segment-mean + dim-1 torch.cat (which costs two full copyBuffer
    passes over [n, F] per conv layer; see profiles/).  Homo-path only:
    targets must be the row prefix of x."""

    @staticmethod
    def forward(ctx, x, col, offsets, n_tgt):
        from .. import _C

        ctx.save_for_backward(col, offsets)
        ctx.n_src = x.size(0)
        return _C.segment_mean_cat_fwd(x, col, offsets, n_tgt)

    @staticmethod
    def backward(ctx, dy):
        from .. import _C

        col, offsets = ctx.saved_tensors
        dx = _C.segment_mean_cat_bwd(dy.contiguous(), col, offsets,
                                     ctx.n_src)
        if dx.dtype != dy.dtype:
            dx = dx.to(dy.dtype)
        return dx, None, None, None


def segment_mean(x: torch.Tensor, tgt: torch.Tensor, src: torch.Tensor,
                 n_tgt: int) -> torch.Tensor:
    """Mean of x[src[e]] over contiguous tgt segments.

    Requires tgt ascending (glt_amd batches satisfy this); returns
    [n_tgt, F].
    """
    offsets = torch.searchsorted(tgt, _boundaries(n_tgt, tgt.device))
    return _SegmentMean.apply(x.contiguous(), src.contiguous(), offsets,
                              x.size(0))


def segment_mean_cat(x: torch.Tensor, tgt: torch.Tensor, src: torch.Tensor,
                     n_tgt: int) -> torch.Tensor:
    """[segment_mean | root] fused: returns [n_tgt, 2F] where the left F
    columns are the neighbor mean and the right F columns are x[:n_tgt]."""
    offsets = torch.searchsorted(tgt, _boundaries(n_tgt, tgt.device))
    return _SegmentMeanCat.apply(x.contiguous(), src.contiguous(), offsets,
                                 n_tgt)
