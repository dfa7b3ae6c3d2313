#!/usr/bin/env python3
"""External reference: the SAME size sweep as allreduce_bw.py but through
torch.distributed's library ncclAllReduce (RCCL's own tuned collective).

This is a measurement yardstick only — the framework's transport is the
composed p2p engine (csrc/rccl_engine.hip); BASELINE/docs/tuning.md call
for our composed allreduce to land within ~15% of the library number once
tuned. Run identically:

  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      benchmarks/allreduce_bw_ncclref.py
"""
import json
import os
import time

import torch
import torch.distributed as dist


def main():
    use_gpu = torch.cuda.is_available()
    backend = "nccl" if use_gpu else "gloo"
    dist.init_process_group(backend)
    W, rank = dist.get_world_size(), dist.get_rank()
    if use_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    dev = torch.device("cuda") if use_gpu else torch.device("cpu")

    sizes = [1 << p for p in range(10, 29)]
    results = []
    for dtype, es in ((torch.bfloat16, 2), (torch.float32, 4)):
        if not use_gpu and dtype == torch.bfloat16:
            continue
        for bytes_ in sizes:
            n = bytes_ // es
            t = torch.ones(n, dtype=dtype, device=dev)
            iters = max(3, min(50, (64 << 20) // bytes_))
            for _ in range(3):
                dist.all_reduce(t)
            if use_gpu:
                torch.cuda.synchronize()
            dist.barrier()
            t0 = time.perf_counter()
            for _ in range(iters):
                dist.all_reduce(t)
            if use_gpu:
                torch.cuda.synchronize()
            el = (time.perf_counter() - t0) / iters
            dist.barrier()
            algbw = bytes_ / el / 1e9
            busbw = (2 * (W - 1) / W) * algbw if W > 1 else algbw
            results.append({"dtype": str(dtype).split(".")[-1],
                            "bytes": bytes_, "us": round(el * 1e6, 1),
                            "algbw_GBps": round(algbw, 2),
                            "busbw_GBps": round(busbw, 2)})
    if rank == 0:
        print(json.dumps({"world": W, "device": str(dev),
                          "transport": f"torch.distributed/{backend}",
                          "results": results}, indent=1))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
