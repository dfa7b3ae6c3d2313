#!/usr/bin/env python3
"""Allreduce bandwidth sweep (reference: benchmarks/ allreduce bandwidth
scripts, SURVEY.md section 4 'benchmarks doubling as soak tests').

Sweeps size x dtype, reports algbw and NCCL-convention busbw
(2(W-1)/W x size / time). Run under torchrun, one rank per GPU:

  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      benchmarks/allreduce_bw.py

First-contact safety (round-1 verdict item 2): before the sweep, a
verify/degrade ladder runs the same analytic checks as bench.py — a wrong
multi-ring result degrades to single-ring / no-graphs instead of producing
garbage numbers, and every stage is under a watchdog so a grouped-p2p
mismatch hangs one stage (process exits 124 with a diagnostic), not the
round. Tuning flags: --rings N --no-graph --sizes ... --dtypes ...
"""
import argparse
import json
import os
import sys
import threading
import time

# Capture RCCL warnings in the log on first contact with a new topology.
os.environ.setdefault("NCCL_DEBUG", "WARN")

import torch  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


class Watchdog:
    """Kills the process (exit 124) if a stage exceeds its deadline: a hung
    grouped-p2p mismatch must end THIS process loudly, not hang the job."""

    def __init__(self, seconds):
        self.seconds = seconds
        self.stage = "init"
        self.deadline = None
        self._lock = threading.Lock()
        t = threading.Thread(target=self._run, daemon=True)
        t.start()

    def arm(self, stage):
        with self._lock:
            self.stage = stage
            self.deadline = time.monotonic() + self.seconds

    def disarm(self):
        with self._lock:
            self.deadline = None

    def _run(self):
        while True:
            time.sleep(1.0)
            with self._lock:
                expired = (self.deadline is not None
                           and time.monotonic() > self.deadline)
                stage = self.stage
            if expired:
                sys.stderr.write(
                    f"[allreduce_bw watchdog] rank {os.environ.get('RANK')} "
                    f"stage '{stage}' exceeded {self.seconds}s — aborting\n")
                sys.stderr.flush()
                os._exit(124)


def verify_ladder(tm, device, wd):
    """bench.py's degrade ladder: eager multi-ring wrong -> single-ring +
    no-graphs; graph replay wrong -> graphs off. Verdict is GLOBAL (every
    rank takes the same branch). Returns a status string."""
    W = tm.size()
    if W < 2 or device.type != "cuda":
        return "n/a"
    n = 1 << 20
    t = torch.empty(n, device=device)
    expect = float(W * (W + 1) / 2)
    eng = tm.communicator().engine

    def check():
        t.fill_(float(tm.rank() + 1))
        tm.allreduce(t)
        torch.cuda.synchronize()
        ok = bool(torch.allclose(t, torch.full_like(t, expect)))
        flags = tm.allgather(torch.tensor([1.0 if ok else 0.0]))
        return bool(flags.min() > 0.5)

    for it in range(3):
        wd.arm(f"verify iter {it}")
        if check():
            continue
        if it == 0:
            eng.set_max_rings(1)
            eng.set_graphs_enabled(False)
            wd.arm("verify single-ring fallback")
            if all(check() for _ in range(2)):
                return "degraded(single-ring,no-graphs)"
            raise RuntimeError("allreduce wrong even on the single-ring "
                               "path — refusing to produce numbers")
        eng.set_graphs_enabled(False)
        wd.arm("verify graph-disable fallback")
        if not check():
            raise RuntimeError("eager allreduce wrong after graph disable")
        return "graphs-disabled(replay-mismatch)"
    wd.disarm()
    return "ok"


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rings", type=int, default=0,
                   help="cap rings per collective (0 = engine default)")
    p.add_argument("--no-graph", action="store_true")
    p.add_argument("--sizes", type=str, default="",
                   help="comma-separated byte sizes (default 1KiB..256MiB)")
    p.add_argument("--dtypes", type=str, default="bf16,f32")
    p.add_argument("--stage-timeout", type=float, default=float(
        os.environ.get("TORCHMPI_BENCH_STAGE_TIMEOUT", "120")))
    args = p.parse_args()

    wd = Watchdog(args.stage_timeout)
    wd.arm("start()")

    import torchmpi_amd as tm
    tm.start()
    W, rank = tm.size(), tm.rank()
    use_gpu = torch.cuda.is_available()
    dev = (torch.device("cuda", torch.cuda.current_device()) if use_gpu
           else torch.device("cpu"))

    if use_gpu and W > 1:
        eng = tm.communicator().engine
        if args.rings > 0:
            eng.set_max_rings(args.rings)
        if args.no_graph:
            eng.set_graphs_enabled(False)

    wd.arm("verify ladder")
    status = verify_ladder(tm, dev, wd)

    if args.sizes:
        sizes = [int(s) for s in args.sizes.split(",")]
    else:
        sizes = [1 << p for p in range(10, 29)]  # 1 KiB .. 256 MiB
    dmap = {"bf16": (torch.bfloat16, 2), "f32": (torch.float32, 4),
            "f16": (torch.float16, 2), "f64": (torch.float64, 8)}
    dtypes = [dmap[d] for d in args.dtypes.split(",")]

    results = []
    for dtype, es in dtypes:
        for bytes_ in sizes:
            n = bytes_ // es
            label = f"{str(dtype).split('.')[-1]}/{bytes_}B"
            wd.arm(f"warmup {label}")
            t = torch.ones(n, dtype=dtype, device=dev)
            iters = max(3, min(50, (64 << 20) // bytes_))
            for _ in range(3):
                tm.allreduce(t)
            if use_gpu:
                torch.cuda.synchronize()
            # spot-check the analytic value at this exact size/dtype before
            # timing it (fill rank+1 -> W(W+1)/2)
            t.fill_(float(tm.rank() + 1))
            tm.allreduce(t)
            if use_gpu:
                torch.cuda.synchronize()
            expect = float(W * (W + 1) / 2)
            ok = bool((t.float() - expect).abs().max() <= max(1e-3 * expect,
                                                              1e-6))
            flags = tm.allgather(torch.tensor([1.0 if ok else 0.0]))
            if flags.min() < 0.5:
                results.append({"dtype": str(dtype).split(".")[-1],
                                "bytes": bytes_, "error": "WRONG VALUE"})
                continue
            t.fill_(1.0)
            tm.barrier()
            wd.arm(f"timing {label}")
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
    wd.disarm()
    if rank == 0:
        print(json.dumps({"world": W, "device": str(dev),
                          "selfcheck": status,
                          "rings": (args.rings or "default"),
                          "graphs": (not args.no_graph),
                          "results": results}, indent=1))
    tm.stop()


if __name__ == "__main__":
    main()
