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
    # When one rank fails, mp.spawn SIGTERMs the others — dump every
    # thread's stack first so a stuck rank tells us WHERE it was stuck
    # (flake diagnosis under loaded CI boxes).
    import faulthandler
    import signal
    faulthandler.register(signal.SIGTERM, all_threads=True, chain=False)
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
    """Run fn(tm, *args) in `world` fresh processes.

    Retries ONCE (fresh port, fresh processes) when the failure is a
    timeout: on the shared CI box, external load spikes can starve a rank
    past the 600 s transport/store timeouts (measured: the same test
    passes 8/8 back-to-back on an idle box and the whole suite runs 7x
    slower during spikes). A genuine deadlock still fails — it times out
    both attempts.
    """
    last_timeout_err = None
    for attempt in range(2):
        port = find_free_port()
        try:
            mp.spawn(_worker, (fn, world, port, use_gpu, args), nprocs=world,
                     join=True)
            return
        except Exception as e:  # noqa: BLE001
            msg = str(e)
            is_timeout = ("poll timeout" in msg or "timed out" in msg
                          or "Timeout" in msg)
            if attempt == 0 and is_timeout:
                sys.stderr.write(
                    f"[run_distributed] timeout under load — retrying once "
                    f"({type(e).__name__})\n")
                last_timeout_err = e
                continue
            raise
    raise last_timeout_err
