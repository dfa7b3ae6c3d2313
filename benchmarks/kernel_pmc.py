#!/usr/bin/env python3
"""Tiny PMC-counter probe for the gfx950 kernels (run under
rocprofv3 --pmc FETCH_SIZE WRITE_SIZE): a handful of launches of
reduce_into and pack/unpack at 256 MB so per-kernel HBM bytes can be
checked against the expected traffic."""
import sys

import torch

sys.path.insert(0, "/root/repo")
from torchmpi_amd.ops import Bucket, reduce_into  # noqa: E402

dev = torch.device("cuda", 0)
n = (256 << 20) // 2
a = torch.randn(n, device=dev).to(torch.bfloat16)
b = torch.randn(n, device=dev).to(torch.bfloat16)
for _ in range(3):
    reduce_into(a, b, "sum")
torch.cuda.synchronize()

ts = [torch.randn(1 << 20, device=dev) for _ in range(64)]
bucket = Bucket(ts, bucket_dtype=torch.bfloat16)
for _ in range(3):
    bucket.pack()
    bucket.unpack(0.125)
torch.cuda.synchronize()
print("pmc probe done")
