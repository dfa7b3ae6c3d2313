#!/usr/bin/env python3
"""Allreduce bandwidth sweep (reference: benchmarks/ allreduce bandwidth
scripts, SURVEY.md section 4 'benchmarks doubling as soak tests').

Sweeps size x dtype, reports algbw and NCCL-convention busbw
(2(W-1)/W x size / time). Run under torchrun, one rank per GPU:

  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      benchmarks/allreduce_bw.py
"""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import torchmpi_amd as tm
    tm.start()
    W, rank = tm.size(), tm.rank()
    use_gpu = torch.cuda.is_available()
    dev = (torch.device("cuda", torch.cuda.current_device()) if use_gpu
           else torch.device("cpu"))

    sizes = [1 << p for p in range(10, 29)]  # 1 KiB .. 256 MiB
    results = []
    for dtype, es in ((torch.bfloat16, 2), (torch.float32, 4)):
        for bytes_ in sizes:
            n = bytes_ // es
            t = torch.ones(n, dtype=dtype, device=dev)
            iters = max(3, min(50, (64 << 20) // bytes_))
            for _ in range(3):
                tm.allreduce(t)
            if use_gpu:
                torch.cuda.synchronize()
            tm.barrier()
            t0 = time.perf_counter()
            for _ in range(iters):
                tm.allreduce(t)
            if use_gpu:
                torch.cuda.synchronize()
            el = (time.perf_counter() - t0) / iters
            tm.barrier()
            algbw = bytes_ / el / 1e9
            busbw = (2 * (W - 1) / W) * algbw if W > 1 else algbw
            results.append({"dtype": str(dtype).split(".")[-1],
                            "bytes": bytes_, "us": round(el * 1e6, 1),
                            "algbw_GBps": round(algbw, 2),
                            "busbw_GBps": round(busbw, 2)})
    if rank == 0:
        print(json.dumps({"world": W, "device": str(dev),
                          "results": results}, indent=1))
    tm.stop()


if __name__ == "__main__":
    main()
