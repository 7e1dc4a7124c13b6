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


class _CastLinear(torch.autograd.Function):
    """Reduced-precision (bf16) linear over fp32 master parameters.

    Manual mixed precision — measured faster than torch.autocast at these
    shapes (autocast's cast caching + guard overhead regressed the
    launch-bound step).  Forward casts the weight/bias once
    to the input's dtype (hipBLASLt runs the GEMM at the bf16 MFMA rate,
    ~2x fp32); backward produces fp32 grads for the fp32 leaves so Adam
    keeps full-precision state.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, relu):
        w16 = weight.to(x.dtype)
        if _use_bf16_mfma(x, weight):
            from .. import _C

            out = _C.gemm_bt_bf16(x, w16, bias, relu, False)
        else:
            b16 = bias.to(x.dtype) if bias is not None else None
            out = torch.nn.functional.linear(x, w16, b16)
            if relu:
                out = torch.relu_(out)
        if relu:  # out is only needed for the ReLU mask in backward
            ctx.save_for_backward(x, w16, out)
        else:
            ctx.save_for_backward(x, w16)
        ctx.has_bias = bias is not None
        ctx.relu = relu
        return out

    @staticmethod
    def backward(ctx, dy):
        if ctx.relu:
            x, w16, out = ctx.saved_tensors
        else:
            x, w16 = ctx.saved_tensors
        dy = dy.contiguous()
        if ctx.relu:
            # one-kernel relu mask (vs gt + mul two-kernel form)
            dy = torch.ops.aten.threshold_backward(dy, out, 0)
        dx = dw = db = None
        mfma = _use_bf16_mfma(x, w16)
        if mfma:
            from .. import _C

            if ctx.needs_input_grad[1]:
                # fused dW = dy^T @ x and db = colsum(dy), fp32 out
                dw, db = _C.gemm_kt_bf16(
                    dy, x, ctx.has_bias and ctx.needs_input_grad[2])
            if ctx.needs_input_grad[0]:
                if w16.size(1) % 16 == 0:
                    dx = _C.gemm_bt_bf16(dy, w16.t().contiguous(), None,
                                         False, False)
                else:
                    dx = dy @ w16
        else:
            if ctx.needs_input_grad[0]:
                dx = dy @ w16
            if ctx.needs_input_grad[1]:
                dw = (dy.t() @ x).to(torch.float32)
            if ctx.has_bias and ctx.needs_input_grad[2]:
                db = dy.sum(0).to(torch.float32)
        if ctx.has_bias and ctx.needs_input_grad[2] and db is None:
            db = dy.sum(0).to(torch.float32)
        return dx, dw, db, None


def _use_bf16_mfma(x: torch.Tensor, weight: torch.Tensor) -> bool:
    """Dispatch policy for the hand-written bf16 MFMA GEMMs.  Requires
    device bf16; small-K/N tails are handled by the kernels, but tiny
    classifier heads (N<64) stay on hipBLASLt where the custom tiling
    has no parallelism to win with."""
    import os

    if os.environ.get("GLT_DISABLE_BF16_MFMA"):
        return False
    return (x.is_cuda and x.dtype == torch.bfloat16
            and weight.size(0) >= 32)


def cast_linear(x: torch.Tensor, weight: torch.Tensor,
                bias: torch.Tensor = None, relu: bool = False
                ) -> torch.Tensor:
    """F.linear with the compute in x's (reduced) dtype and fp32 master
    weights; optional fused-at-the-boundary ReLU."""
    return _CastLinear.apply(x, weight, bias, relu)


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
