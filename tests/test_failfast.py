"""Failure semantics: the reference is MPI-era
fail-fast — a dead rank kills the job. Here: a collective against a dead
peer must raise promptly ("peer closed connection"), never hang."""
import os
import subprocess
import sys
import time

from .util import REPO_ROOT, find_free_port

_SCRIPT = r"""
import os, sys, time
sys.path.insert(0, os.environ["TMA_REPO"])
import torch
import torchmpi_amd as tm

tm.start(use_gpu=False)
rank = tm.rank()
tm.barrier()
if rank == 1:
    os._exit(7)  # die without teardown, mid-job
time.sleep(0.5)  # let rank 1's sockets close
t = torch.ones(1 << 16)
try:
    tm.allreduce(t)
except RuntimeError as e:
    print("FAILFAST OK:", str(e)[:60])
    sys.exit(0)
print("NO ERROR RAISED")
sys.exit(1)
"""


def test_peer_death_fails_fast():
    port = find_free_port()
    env = dict(os.environ, TMA_REPO=REPO_ROOT, MASTER_ADDR="127.0.0.1",
               MASTER_PORT=str(port), WORLD_SIZE="2",
               TORCHMPI_TCP_TIMEOUT_MS="30000")
    procs = []
    for r in (0, 1):
        e = dict(env, RANK=str(r), LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, "-c", _SCRIPT],
                                      env=e, cwd=REPO_ROOT,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    t0 = time.time()
    out0, err0 = procs[0].communicate(timeout=120)
    procs[1].wait(timeout=30)
    elapsed = time.time() - t0
    assert procs[1].returncode == 7
    assert procs[0].returncode == 0, (out0, err0[-1500:])
    assert "FAILFAST OK" in out0, (out0, err0[-1000:])
    # promptly: well under the 30 s poll timeout (socket close, not timeout)
    assert elapsed < 25, f"fail-fast took {elapsed:.1f}s"
