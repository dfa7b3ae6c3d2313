"""Fused NHWC BatchNorm backed by the gfx950 kernels
(csrc/kernels/batchnorm.hip). Drop-in replacement for nn.BatchNorm2d on
channels_last bf16/f32 tensors — MIOpen's spatial BN is ~35% of the
ResNet-50 step on MI355X (profiles/).

The kernels optionally fuse the FOLLOWING relu and a residual add into the
BN epilogue (y = relu(bn(x) + res)): the separate elementwise add/relu
passes were ~20% of the ResNet-50 step (profiles/
resnet50_bf16_kernel_stats.md). Modules that know their successor call
`forward(x, relu=True, res=identity)` (see models/resnet.py Bottleneck).

State-dict compatible with nn.BatchNorm2d; convert_batchnorm(model) swaps
all BatchNorm2d modules in place.
"""
from typing import Optional

import torch
import torch.nn as nn

from .. import _C


class _BnFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, weight, bias, running_mean, running_var, training,
                momentum, eps, relu):
        x = x.contiguous(memory_format=torch.channels_last)
        if res is not None:
            res = res.contiguous(memory_format=torch.channels_last)
        y, mean, invstd, mask = _C.bn_forward(x, weight, bias, running_mean,
                                              running_var, training, momentum,
                                              eps, relu, res)
        # Backward needs the relu mask: the 1-bit mask tensor on the vector
        # path (1/16th of y), y itself on the scalar fallback / eval path.
        need_y = relu and (mask.numel() == 0 or not training)
        ctx.save_for_backward(x, weight, mean, invstd,
                              y if need_y else torch.empty(0), mask)
        ctx.training = training
        ctx.relu = relu
        ctx.has_res = res is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, invstd, y, mask = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        if not ctx.training:
            # eval: stats are constants — no batch-statistic terms.
            dyf = dy.float()
            if ctx.relu:
                dyf = dyf * (y.float() > 0)
            m = mean.view(1, -1, 1, 1)
            istd = invstd.view(1, -1, 1, 1)
            xhat = (x.float() - m) * istd
            dx = (dyf * (weight.view(1, -1, 1, 1) * istd)).to(x.dtype)
            dweight = (dyf * xhat).sum((0, 2, 3))
            dbias = dyf.sum((0, 2, 3))
            dres = dyf.to(x.dtype) if ctx.has_res else None
            return (dx.contiguous(memory_format=torch.channels_last), dres,
                    dweight, dbias, None, None, None, None, None, None)
        out = _C.bn_backward(x, dy, weight, mean, invstd,
                             y if (ctx.relu and y.numel()) else None,
                             ctx.relu, ctx.has_res,
                             mask if mask.numel() else None)
        dres = out[3] if ctx.has_res else None
        return (out[0], dres, out[1], out[2], None, None, None, None, None,
                None)


class FusedBatchNorm2d(nn.Module):
    """Per-channel batchnorm with fp32 stats/affine and bf16/f32 NHWC data
    path on our kernels; falls back to torch BN on CPU / non-4D input.
    forward(x, relu=..., res=...) fuses a following ReLU / residual add."""

    supports_fusion = True  # duck-typed by models that fuse relu/residual

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def forward(self, x, relu: bool = False,
                res: Optional[torch.Tensor] = None):
        if not x.is_cuda or x.dim() != 4:
            y = torch.nn.functional.batch_norm(
                x, self.running_mean, self.running_var, self.weight, self.bias,
                self.training, self.momentum, self.eps)
            if res is not None:
                y = y + res
            return torch.relu(y) if relu else y
        if self.training:
            self.num_batches_tracked += 1
        # autocast: keep our fp32-stat kernels; input stays in its dtype
        with torch.autocast("cuda", enabled=False):
            xw = x
            if xw.dtype not in (torch.bfloat16, torch.float32):
                xw = xw.to(torch.bfloat16)
            rw = res
            if rw is not None and rw.dtype != xw.dtype:
                rw = rw.to(xw.dtype)
            return _BnFunction.apply(xw, rw, self.weight, self.bias,
                                     self.running_mean, self.running_var,
                                     self.training, self.momentum, self.eps,
                                     relu)

    @classmethod
    def from_batchnorm(cls, bn: nn.BatchNorm2d) -> "FusedBatchNorm2d":
        m = cls(bn.num_features, bn.eps, bn.momentum if bn.momentum else 0.1)
        m = m.to(bn.weight.device)
        with torch.no_grad():
            m.weight.copy_(bn.weight.float())
            m.bias.copy_(bn.bias.float())
            m.running_mean.copy_(bn.running_mean.float())
            m.running_var.copy_(bn.running_var.float())
            m.num_batches_tracked.copy_(bn.num_batches_tracked)
        return m


def convert_batchnorm(module: nn.Module) -> nn.Module:
    """Recursively replace nn.BatchNorm2d with FusedBatchNorm2d."""
    for name, child in module.named_children():
        if isinstance(child, nn.BatchNorm2d):
            setattr(module, name, FusedBatchNorm2d.from_batchnorm(child))
        else:
            convert_batchnorm(child)
    return module
