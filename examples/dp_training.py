#!/usr/bin/env python3
"""Synchronous data-parallel SGD with overlapped gradient allreduce
(reference: examples/mnist/mnist_allreduce.lua, SURVEY.md stack 3.5).

Run:  python examples/dp_training.py --world 2          (spawn, CPU)
      torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
          examples/dp_training.py                        (one rank per GPU)

Synthetic data (no network access for datasets); each rank trains on its
own shard (shard = sample_index % world, like the reference examples).
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_model():
    torch.manual_seed(1234)
    return torch.nn.Sequential(
        torch.nn.Flatten(),
        torch.nn.Linear(28 * 28, 256), torch.nn.ReLU(),
        torch.nn.Linear(256, 128), torch.nn.ReLU(),
        torch.nn.Linear(128, 10),
    )


def main_worker(checkpoint=None, resume=False):
    import torchmpi_amd as tm
    from torchmpi_amd.nn import GradientSynchronizer, synchronize_parameters

    tm.start()
    rank, W = tm.rank(), tm.size()
    dev = (torch.device("cuda", torch.cuda.current_device())
           if torch.cuda.is_available() else torch.device("cpu"))

    model = make_model().to(dev)
    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    start_epoch = 0
    if resume and checkpoint and os.path.exists(checkpoint):
        # Checkpoint/resume at the reference's capability level (SURVEY.md
        # section 5): torch.save on rank 0 behind a barrier; every rank loads.
        state = torch.load(checkpoint, map_location=dev, weights_only=False)
        model.load_state_dict(state["model"])
        opt.load_state_dict(state["opt"])
        start_epoch = state["epoch"] + 1
        if rank == 0:
            print(f"resumed from {checkpoint} at epoch {start_epoch}")
    synchronize_parameters(model)          # replicas agree at t=0
    sync = GradientSynchronizer(model, average=True)

    # Synthetic "dataset": 4096 samples, rank-sharded sample_index % W.
    # TORCHMPI_EXAMPLE_FAST=1 shrinks the run (CI under load).
    fast = os.environ.get("TORCHMPI_EXAMPLE_FAST") == "1"
    n_samples = 1024 if fast else 4096
    n_epochs = 1 if fast else 3
    torch.manual_seed(999)
    data = torch.randn(n_samples, 1, 28, 28)
    labels = (data.view(n_samples, -1).sum(-1) * 10).long().abs() % 10
    my_idx = torch.arange(rank, n_samples, W)

    for epoch in range(start_epoch, n_epochs):
        perm = torch.randperm(len(my_idx))
        total_loss, n = 0.0, 0
        for i in range(0, len(my_idx), 64):
            batch = my_idx[perm[i:i + 64]]
            x, y = data[batch].to(dev), labels[batch].to(dev)
            opt.zero_grad(set_to_none=False)
            loss = torch.nn.functional.cross_entropy(model(x), y)
            loss.backward()           # bucket allreduces fire from hooks
            sync.finish()             # wait + unpack (grads now averaged)
            opt.step()
            total_loss += float(loss) * len(batch)
            n += len(batch)
        if rank == 0:
            print(f"epoch {epoch}: loss {total_loss / n:.4f}")
        tm.barrier()               # grads/steps complete everywhere
        if checkpoint and rank == 0:
            torch.save({"model": model.state_dict(),
                        "opt": opt.state_dict(), "epoch": epoch}, checkpoint)
        tm.barrier()
    tm.stop()


def _spawn(rank, world, port, checkpoint=None, resume=False):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      LOCAL_RANK=str(rank))
    main_worker(checkpoint, resume)


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=0,
                    help="spawn N local processes (else use torchrun env)")
    ap.add_argument("--checkpoint", type=str, default=None)
    ap.add_argument("--resume", action="store_true")
    args = ap.parse_args()
    if args.world > 0:
        import torch.multiprocessing as mp
        mp.spawn(_spawn, (args.world, _free_port(), args.checkpoint, args.resume),
                 nprocs=args.world, join=True)
    else:
        main_worker(args.checkpoint, args.resume)
