"""Distributed test harness: spawn real processes (the reference tested
under real mpirun — SURVEY.md section 4; here: multiprocessing.spawn with
the TCP rendezvous on 127.0.0.1)."""
import os
import socket
import sys
import traceback

import torch.multiprocessing as mp

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def find_free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, fn, world, port, use_gpu, args):
    sys.path.insert(0, REPO_ROOT)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torchmpi_amd as tm
    try:
        tm.start(use_gpu=use_gpu)
        fn(tm, *args)
        tm.stop()
    except Exception:
        traceback.print_exc()
        raise


def run_distributed(fn, world: int = 2, use_gpu: bool = False, args=()):
    """Run fn(tm, *args) in `world` fresh processes."""
    port = find_free_port()
    mp.spawn(_worker, (fn, world, port, use_gpu, args), nprocs=world, join=True)
