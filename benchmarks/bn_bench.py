#!/usr/bin/env python3
"""A/B: FusedBatchNorm2d vs torch.nn.BatchNorm2d (MIOpen) on ResNet shapes,
bf16 channels_last, fwd+bwd wall time."""
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from torchmpi_amd.ops.batchnorm import FusedBatchNorm2d  # noqa: E402

SHAPES = [  # (N, C, H, W) — ResNet-50 @ 224
    (256, 64, 112, 112),
    (256, 256, 56, 56),
    (256, 512, 28, 28),
    (256, 1024, 14, 14),
    (256, 2048, 7, 7),
]


def bench(mod, x, iters=20):
    g = torch.randn_like(x)
    for _ in range(5):
        y = mod(x)
        y.backward(g)
        x.grad = None
        mod.weight.grad = None
        mod.bias.grad = None
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        y = mod(x)
        y.backward(g)
        x.grad = None
        mod.weight.grad = None
        mod.bias.grad = None
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    dev = torch.device("cuda", 0)
    print(f"{'shape':>22} {'torch ms':>9} {'fused ms':>9} {'speedup':>8}")
    for shape in SHAPES:
        x = (torch.randn(shape, device=dev, dtype=torch.bfloat16)
             .contiguous(memory_format=torch.channels_last).requires_grad_(True))
        ref = torch.nn.BatchNorm2d(shape[1]).to(dev)
        ours = FusedBatchNorm2d(shape[1]).to(dev)
        t_ref = bench(ref, x)
        t_our = bench(ours, x)
        print(f"{str(shape):>22} {t_ref:9.3f} {t_our:9.3f} {t_ref / t_our:8.2f}x")


if __name__ == "__main__":
    main()
