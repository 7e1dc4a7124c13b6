"""MFMA f32 projection linear.

`mfma_linear(x, weight, bias)` = x @ weight.T + bias via the hand-written
v_mfma_f32_16x16x4_f32 GEMM (exact fp32).  Forward and input-gradient run
the custom kernel; the weight gradient (a K-huge reduction GEMM) stays on
rocBLAS/hipBLASLt, which is the right tool for that shape.
"""
import torch


class _MfmaLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        from .. import _C

        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        # weight is [out, in]; kernel wants B = weight.T [in, out]
        return _C.sage_gemm(x, weight.t().contiguous(), bias)

    @staticmethod
    def backward(ctx, dy):
        from .. import _C

        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if weight.size(1) % 64 == 0:
                dx = _C.sage_gemm(dy, weight.contiguous(), None)
            else:  # in-dim not tile-able: library GEMM
                dx = dy @ weight
        if ctx.needs_input_grad[1]:
            dw = dy.t() @ x  # K-huge reduction: library GEMM
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = dy.sum(0)
        return dx, dw, db


def mfma_linear(x: torch.Tensor, weight: torch.Tensor,
                bias: torch.Tensor = None) -> torch.Tensor:
    """Drop-in F.linear for fp32 CUDA inputs with out_features % 64 == 0."""
    return _MfmaLinear.apply(x, weight, bias)


def use_mfma_linear(x: torch.Tensor, weight: torch.Tensor) -> bool:
    """Dispatch policy from MI355X measurements: the custom
    kernel ~doubles rocBLAS on narrow outputs (n<=128: Tensile picks a
    32-wide tile at ~32 TF there) and ties it at n=256, where we keep the
    library."""
    return (x.is_cuda and x.dtype == torch.float32
            and weight.dtype == torch.float32
            and weight.size(0) % 64 == 0 and weight.size(0) <= 128)
