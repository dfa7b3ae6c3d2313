#!/usr/bin/env python3
"""Bandwidth microbench for the gfx950 kernels (reduce_into, pack/unpack).

reduce_into streams 2 reads + 1 write; at the ~6.3 TB/s achievable HBM rate
the ceiling is ~2.1 TB/s of 'algorithmic' (dst-size) throughput x 3.
"""
import json
import sys
import time

import torch


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    sys.path.insert(0, "/root/repo")
    from torchmpi_amd.ops import Bucket, reduce_into, scale_
    dev = torch.device("cuda", 0)
    results = {}

    for dtype, name in [(torch.bfloat16, "bf16"), (torch.float32, "f32")]:
        es = 2 if dtype == torch.bfloat16 else 4
        for mb in (1, 16, 256):
            n = (mb << 20) // es
            a = torch.randn(n, device=dev).to(dtype)
            b = torch.randn(n, device=dev).to(dtype)
            el = bench(lambda: reduce_into(a, b, "sum"))
            moved = 3 * n * es  # read a, read b, write a
            results[f"reduce_{name}_{mb}MB_GBps"] = round(moved / el / 1e9, 1)

    # torch baseline for comparison (a += b)
    n = (256 << 20) // 2
    a = torch.randn(n, device=dev).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    el = bench(lambda: a.add_(b))
    results["torch_add__bf16_256MB_GBps"] = round(3 * n * 2 / el / 1e9, 1)

    # scale
    el = bench(lambda: scale_(a, 1.0000001))
    results["scale_bf16_256MB_GBps"] = round(2 * n * 2 / el / 1e9, 1)

    # pack/unpack: 64 tensors of 1M fp32 -> bf16 bucket
    ts = [torch.randn(1 << 20, device=dev) for _ in range(64)]
    bucket = Bucket(ts, bucket_dtype=torch.bfloat16)
    el = bench(lambda: bucket.pack())
    moved = sum(t.numel() * 4 for t in ts) + bucket.total * 2
    results["pack_f32_to_bf16_256MB_GBps"] = round(moved / el / 1e9, 1)
    el = bench(lambda: bucket.unpack(0.125))
    results["unpack_bf16_to_f32_256MB_GBps"] = round(moved / el / 1e9, 1)

    print(json.dumps(results, indent=1))


if __name__ == "__main__":
    main()
