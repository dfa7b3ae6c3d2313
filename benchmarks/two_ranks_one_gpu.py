#!/usr/bin/env python3
"""Experiment: run the full RCCL engine with world_size=2, both ranks on the
SAME GPU (device 0).

FINDING (recorded 2026-09-13, ROCm 7.2 RCCL): ncclCommInitRank returns
'invalid usage' for two ranks on one device — RCCL does not support
multi-rank-per-GPU, so multi-rank engine paths cannot be exercised on a
1-GPU box. Multi-rank coverage therefore comes from (a) the exported-
schedule simulation tests (tests/test_ring_schedule.py), (b) the CPU
transport running the same algorithms under real multi-process tests, and
(c) bench.py's analytic self-check at startup on the real 8-GPU node.
Kept for re-testing on future RCCL versions.
"""
import os
import sys

import torch
import torch.multiprocessing as mp


def worker(rank, world, port):
    sys.path.insert(0, "/root/repo")
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      LOCAL_RANK="0")  # both on device 0
    import torchmpi_amd as tm
    tm.start(use_gpu=True)
    dev = torch.device("cuda", 0)
    try:
        for n in (1000, 1 << 20):
            t = torch.full((n,), float(rank + 1), device=dev)
            tm.allreduce(t)
            torch.cuda.synchronize()
            assert torch.equal(t, torch.full((n,), 3.0, device=dev)), t[:4]
        # bf16 + average
        t = torch.full((1 << 20,), float(rank + 1), device=dev,
                       dtype=torch.bfloat16)
        tm.allreduce(t, average=True)
        torch.cuda.synchronize()
        assert torch.allclose(t.float(), torch.full((1 << 20,), 1.5, device=dev))
        # broadcast + allgather + sendrecv
        b = (torch.arange(1000, device=dev, dtype=torch.float32) if rank == 0
             else torch.zeros(1000, device=dev))
        tm.broadcast(b, root=0)
        torch.cuda.synchronize()
        assert torch.equal(b, torch.arange(1000, device=dev, dtype=torch.float32))
        g = tm.allgather(torch.full((10,), float(rank), device=dev))
        torch.cuda.synchronize()
        assert torch.equal(g[0], torch.zeros(10, device=dev))
        assert torch.equal(g[1], torch.ones(10, device=dev))
        rs_in = torch.arange(100, device=dev, dtype=torch.float32) * (rank + 1)
        rs_out = torch.zeros(50, device=dev)
        tm.reduce_scatter(rs_in, rs_out)
        torch.cuda.synchronize()
        expect = (torch.arange(100, device=dev, dtype=torch.float32) * 3)[
            rank * 50:(rank + 1) * 50]
        assert torch.equal(rs_out, expect)
        if rank == 0:
            print("TWO_RANKS_OK")
    finally:
        tm.stop()


if __name__ == "__main__":
    mp.spawn(worker, (2, 29655), nprocs=2, join=True)
