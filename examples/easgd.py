#!/usr/bin/env python3
"""EASGD (elastic averaging SGD) on the parameter server (reference:
examples/mnist *easgd* example).

The PS holds the center variable x~. Every tau steps each worker pulls x~,
applies the elastic force x <- x - alpha*(x - x~) locally, and pushes
+alpha*(x - x~) to the center with the 'add' rule (the center moves toward
the workers; staleness-tolerant, no barrier in the loop).

Run: python examples/easgd.py --world 4
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main_worker(alpha=0.5, tau=4):
    import torchmpi_amd as tm

    tm.start()
    rank = tm.rank()
    torch.manual_seed(7)
    model = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.ReLU(),
                                torch.nn.Linear(64, 4))
    tm.nn.synchronize_parameters(model)

    center = torch.nn.utils.parameters_to_vector(model.parameters()).detach().clone()
    ps = tm.ParameterServer()
    ps.register(center)  # PS now holds x~ = initial params

    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    torch.manual_seed(100 + rank)
    n_steps = 12 if os.environ.get("TORCHMPI_EXAMPLE_FAST") == "1" else 40
    for step in range(n_steps):
        x = torch.randn(64, 32)
        y = torch.randint(0, 4, (64,))
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        if (step + 1) % tau == 0:
            xt = torch.empty_like(center)
            ps.receive(center, out=xt).wait()        # pull x~
            xv = torch.nn.utils.parameters_to_vector(model.parameters()).detach()
            force = alpha * (xv - xt)
            ps.send(center, "add", data=force).wait()  # center += alpha (x - x~)
            with torch.no_grad():
                torch.nn.utils.vector_to_parameters(xv - force,
                                                    model.parameters())
    tm.barrier()
    final = torch.empty_like(center)
    ps.receive(center, out=final).wait()
    if rank == 0:
        print(f"easgd done; final loss {float(loss):.4f}; "
              f"center norm {float(final.norm()):.4f}")
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
