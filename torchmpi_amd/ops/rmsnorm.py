"""Fused RMSNorm backed by the gfx950 kernel (csrc/kernels/rmsnorm.hip):
1 read + 1 write per call vs ~7 activation passes for the eager form.
Used by models/llama.py on bf16 CUDA inputs; falls back to eager otherwise.
"""
import torch

from .. import _C


class _RmsFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        y, inv = _C.rms_forward(x, weight, eps)
        ctx.save_for_backward(x, weight, inv)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, inv = ctx.saved_tensors
        dx, dw = _C.rms_backward(x, dy.contiguous(), weight, inv)
        return dx, dw, None


class _RmsAddFunction(torch.autograd.Function):
    """(y, h) = (rmsnorm(res + branch), res + branch) in one pass — the
    Llama pre-norm residual add fused into the norm kernel. Backward: both
    inputs receive dh + rms_dx(dy), computed in one pass too."""

    @staticmethod
    def forward(ctx, res, branch, weight, eps):
        y, h, inv = _C.rms_forward_add(res, branch, weight, eps)
        ctx.save_for_backward(h, weight, inv)
        return y, h

    @staticmethod
    def backward(ctx, dy, dh):
        h, weight, inv = ctx.saved_tensors
        dres = dh.contiguous() if dh is not None else None
        dx, dw = _C.rms_backward(h, dy.contiguous(), weight, inv, dres)
        return dx, dx, dw, None


def fused_rmsnorm_supported(x: torch.Tensor) -> bool:
    return (x.is_cuda and x.dtype == torch.bfloat16 and x.is_contiguous()
            and bool(_C.rms_ok(x)))


def fused_rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float):
    """weight must be fp32 (llama keeps norm weights in fp32 master)."""
    return _RmsFunction.apply(x, weight.float(), eps)


def fused_add_rmsnorm(res: torch.Tensor, branch: torch.Tensor,
                      weight: torch.Tensor, eps: float):
    """Returns (normed, new_residual) for the pre-norm pattern
    h = res + branch; y = rmsnorm(h)."""
    return _RmsAddFunction.apply(res, branch, weight.float(), eps)
