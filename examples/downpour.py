#!/usr/bin/env python3
"""Downpour-SGD on the sharded parameter server (reference:
examples/mnist parameterserver example, SURVEY.md stack 3.4).

Each worker trains asynchronously on its own shard, pushing gradient deltas
to the PS with the 'add' rule every step and pulling fresh parameters every
`--pull-every` steps. No global barrier inside the training loop.

Run: python examples/downpour.py --world 4
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main_worker(pull_every=4):
    import torchmpi_amd as tm

    tm.start()
    rank, W = tm.rank(), tm.size()
    torch.manual_seed(1234)
    model = torch.nn.Sequential(
        torch.nn.Flatten(), torch.nn.Linear(64, 64), torch.nn.ReLU(),
        torch.nn.Linear(64, 10))
    tm.nn.synchronize_parameters(model)

    # Flatten params into one vector for the PS (the reference registered
    # the getParameters() flat tensor).
    flat = torch.nn.utils.parameters_to_vector(model.parameters()).detach()
    ps = tm.ParameterServer()
    ps.register(flat)
    lr = 0.05

    torch.manual_seed(4321 + rank)
    pending = None
    n_steps = 10 if os.environ.get("TORCHMPI_EXAMPLE_FAST") == "1" else 40
    for step in range(n_steps):
        x = torch.randn(32, 1, 8, 8).view(32, -1)
        y = torch.randint(0, 10, (32,))
        model.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        grad_flat = torch.nn.utils.parameters_to_vector(
            [p.grad for p in model.parameters()])
        # async push of -lr * grad with the 'add' rule
        if pending is not None:
            pending.wait()
        pending = ps.send(flat, "add", data=(-lr) * grad_flat)
        # local (stale) update so the worker keeps moving
        with torch.no_grad():
            torch.nn.utils.vector_to_parameters(
                torch.nn.utils.parameters_to_vector(model.parameters())
                - lr * grad_flat, model.parameters())
        if (step + 1) % pull_every == 0:
            pending.wait()
            pending = None
            ps.receive(flat).wait()
            with torch.no_grad():
                torch.nn.utils.vector_to_parameters(flat, model.parameters())
    if pending is not None:
        pending.wait()
    tm.barrier()
    ps.receive(flat).wait()
    if rank == 0:
        print(f"downpour done; final loss {float(loss):.4f}; "
              f"param norm {float(flat.norm()):.4f}")
    ps.shutdown()
    tm.barrier()
    tm.stop()


def _spawn(rank, world, port):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      LOCAL_RANK=str(rank))
    main_worker()


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=2)
    args = ap.parse_args()
    import torch.multiprocessing as mp
    mp.spawn(_spawn, (args.world, _free_port()), nprocs=args.world, join=True)
