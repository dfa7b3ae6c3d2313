#!/usr/bin/env python3
"""Llama-3-8B data-parallel training step benchmark:
bucket-fused bf16 gradient allreduce (cast-fused pack kernels) overlapped
with backward. Reports tokens/s, step time, and EXPOSED communication time
(step time with sync minus step time with hooks detached, same everything).

Run:  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
        benchmarks/llama_dp.py --config llama3_8b --seq 4096 --micro-batch 1
Single GPU smoke: python benchmarks/llama_dp.py --config debug
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run_steps(model, sync, opt, tokens, targets, n, use_sync=True):
    for _ in range(n):
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", dtype=torch.bfloat16,
                            enabled=tokens.is_cuda):
            loss = model(tokens, targets)
        loss.backward()
        if use_sync:
            sync.finish()
        opt.step()
    return loss


def timed(fn):
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return time.perf_counter() - t0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="debug",
                    choices=["debug", "llama3_8b"])
    ap.add_argument("--seq", type=int, default=0)
    ap.add_argument("--micro-batch", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--bucket-mb", type=int, default=50)
    args = ap.parse_args()

    import torchmpi_amd as tm
    from torchmpi_amd.models import LlamaModel, llama3_8b, llama_debug
    from torchmpi_amd.nn import GradientSynchronizer, synchronize_parameters

    tm.start()
    W, rank = tm.size(), tm.rank()
    use_gpu = torch.cuda.is_available()
    dev = (torch.device("cuda", torch.cuda.current_device()) if use_gpu
           else torch.device("cpu"))

    cfg = llama3_8b() if args.config == "llama3_8b" else llama_debug()
    seq = args.seq or (4096 if args.config == "llama3_8b" else 128)
    B = args.micro_batch

    torch.manual_seed(0)
    model = LlamaModel(cfg).to(dev)
    synchronize_parameters(model)
    sync = GradientSynchronizer(model, bucket_mb=args.bucket_mb, average=True,
                                comm_dtype=torch.bfloat16 if use_gpu else None)
    opt = torch.optim.SGD(model.parameters(), lr=1e-4, momentum=0.9)

    tokens = torch.randint(0, cfg.vocab_size, (B, seq), device=dev)
    targets = torch.randint(0, cfg.vocab_size, (B, seq), device=dev)

    run_steps(model, sync, opt, tokens, targets, args.warmup)
    tm.barrier()
    el_sync = timed(lambda: run_steps(model, sync, opt, tokens, targets,
                                      args.steps, True))
    tm.barrier()
    # Exposure baseline: identical steps with gradient sync removed.
    sync.detach()
    run_steps(model, sync, opt, tokens, targets, 1, use_sync=False)
    tm.barrier()
    el_nosync = timed(lambda: run_steps(model, sync, opt, tokens, targets,
                                        args.steps, False))
    tm.barrier()

    el = torch.tensor([el_sync, el_nosync], dtype=torch.float64)
    allel = tm.allgather(el)
    el_sync = float(allel[:, 0].max())
    el_nosync = float(allel[:, 1].max())

    if rank == 0:
        step_ms = el_sync / args.steps * 1e3
        nosync_ms = el_nosync / args.steps * 1e3
        exposed = max(0.0, step_ms - nosync_ms)
        grad_bytes = sum(p.numel() for p in model.parameters()
                         if p.requires_grad) * 2  # bf16 on the wire
        print(json.dumps({
            "bench": "llama_dp",
            "model": args.config,
            "world": W,
            "seq_len": seq,
            "micro_batch": B,
            "global_tokens_per_step": W * B * seq,
            "tokens_per_sec": round(W * B * seq * args.steps / el_sync, 1),
            "step_ms": round(step_ms, 2),
            "step_ms_no_sync": round(nosync_ms, 2),
            "exposed_comm_ms": round(exposed, 2),
            "exposed_comm_pct": round(100 * exposed / step_ms, 2),
            "grad_bf16_MB": round(grad_bytes / 1e6, 1),
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
        }))
    tm.stop()


if __name__ == "__main__":
    main()
