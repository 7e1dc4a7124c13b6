"""Autograd wrapper for the fused GAT edge-softmax + aggregation kernels."""
import torch


class _GatFused(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h_src, a_src, a_dst, src, offsets, slope):
        from .. import _C

        # contiguity once, shared by fwd and the saved tensors bwd reads
        h_src = h_src.contiguous()
        a_src = a_src.contiguous()
        a_dst = a_dst.contiguous()
        out, m, z = _C.gat_fused_fwd(h_src, a_src, a_dst, src, offsets,
                                     slope)
        ctx.save_for_backward(h_src, a_src, a_dst, src, offsets, out, m, z)
        ctx.slope = slope
        return out

    @staticmethod
    def backward(ctx, dout):
        from .. import _C

        h_src, a_src, a_dst, src, offsets, out, m, z = ctx.saved_tensors
        dh, das, dad = _C.gat_fused_bwd(h_src, a_src, a_dst, src, offsets,
                                        out, m, z, dout, ctx.slope)
        return dh, das, dad, None, None, None


def gat_softmax_aggregate(h_src, a_src, a_dst, tgt, src, n_tgt,
                          negative_slope=0.2):
    """out[t,h,:] = sum_e softmax_t(leaky(a_dst[t,h]+a_src[src_e,h]))
    * h_src[src_e,h,:], over edges sorted by target."""
    from .segment import _boundaries

    offsets = torch.searchsorted(tgt, _boundaries(n_tgt, tgt.device))
    return _GatFused.apply(h_src, a_src, a_dst, src.contiguous(), offsets,
                           negative_slope)
