"""MFMA f32 projection linear (optionally with fused ReLU epilogue).

`mfma_linear(x, weight, bias, relu=...)` = relu?(x @ weight.T + bias) via
the hand-written v_mfma_f32 GEMM (exact fp32).  Forward and input-gradient
run the custom kernel; the weight gradient (a K-huge reduction GEMM) stays
on rocBLAS/hipBLASLt, which is the right tool for that shape.

Measured dispatch policy: the custom kernel ~doubles rocBLAS
at n<=128 outputs and ties it at n=256 — so the un-fused form is used only
for narrow outputs, while the ReLU-fused form is a net win wherever it
applies (it removes a full activation read+write round trip).
"""
import torch


class _MfmaLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, relu):
        from .. import _C

        out = _C.sage_gemm(x, weight.t().contiguous(), bias, relu)
        ctx.save_for_backward(x, weight, out)
        ctx.has_bias = bias is not None
        ctx.relu = relu
        return out

    @staticmethod
    def backward(ctx, dy):
        from .. import _C

        x, weight, out = ctx.saved_tensors
        dy = dy.contiguous()
        if ctx.relu:
            dy = dy * (out > 0).to(dy.dtype)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if weight.size(1) % 64 == 0:
                dx = _C.sage_gemm(dy, weight.contiguous(), None, False)
            else:  # in-dim not tile-able: library GEMM
                dx = dy @ weight
        if ctx.needs_input_grad[1]:
            dw = dy.t() @ x  # K-huge reduction: library GEMM
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = dy.sum(0)
        return dx, dw, db, None


def mfma_linear(x: torch.Tensor, weight: torch.Tensor,
                bias: torch.Tensor = None, relu: bool = False
                ) -> torch.Tensor:
    """Drop-in F.linear (+fused ReLU) for fp32 CUDA inputs with
    out_features % 64 == 0."""
    return _MfmaLinear.apply(x, weight, bias, relu)


def use_mfma_linear(x: torch.Tensor, weight: torch.Tensor,
                    relu: bool = False) -> bool:
    if not (x.is_cuda and x.dtype == torch.float32
            and weight.dtype == torch.float32
            and weight.size(0) % 64 == 0):
        return False
    # ~2x rocBLAS at narrow outputs; at n>=256 rocBLAS wins in-context
    # (measured end-to-end: 394 vs 372 b/s with the fused path), so the
    # fused-ReLU form is offered only where the kernel wins outright
    return weight.size(0) <= 128
