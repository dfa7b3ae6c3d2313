"""GPU parameter server over xGMI/hipIpc peer mappings.

Runs with TWO PROCESSES SHARING DEVICE 0 (hipIpc works same-device across
processes, unlike multi-rank RCCL), so the one-sided copy/atomic-add/read
path is fully exercised on a 1-GPU box; on the 8-GPU node the same code
runs over the xGMI fabric."""
import os

import pytest
import torch

from .util import REPO_ROOT, find_free_port

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs an AMD GPU")


def _ps_worker(rank, world, port, scenario):
    import sys
    sys.path.insert(0, REPO_ROOT)
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      LOCAL_RANK="0")  # all ranks share device 0
    import torchmpi_amd as tm
    tm.start(use_gpu=True)
    dev = torch.device("cuda", 0)
    ps = tm.ParameterServer()
    W = tm.size()

    if scenario == "rules":
        n = 1037
        p = torch.zeros(n, device=dev)
        ps.register(p)
        tm.barrier()
        # concurrent atomic 'add' from every rank
        ps.send(p, "add", data=torch.full((n,), float(rank + 1), device=dev)).wait()
        tm.barrier()
        got = torch.empty(n, device=dev)
        ps.receive(p, out=got).wait()
        expect = float(W * (W + 1) / 2)
        assert torch.allclose(got, torch.full((n,), expect, device=dev)), got[:5]
        tm.barrier()
        # 'copy' rule from rank 0
        if rank == 0:
            ps.send(p, "copy",
                    data=torch.arange(n, dtype=torch.float32, device=dev)).wait()
        tm.barrier()
        ps.receive(p, out=got).wait()
        assert torch.equal(got, torch.arange(n, dtype=torch.float32, device=dev))
    elif scenario == "rules16":
        # bf16 'add' rule (32-bit system-scope CAS path) with an ODD count:
        # exercises the tail-word CAS and non-16B shard offsets.
        n = 1031
        p = torch.zeros(n, device=dev, dtype=torch.bfloat16)
        ps.register(p)
        tm.barrier()
        ps.send(p, "add",
                data=torch.full((n,), float(rank + 1), device=dev,
                                dtype=torch.bfloat16)).wait()
        tm.barrier()
        got = torch.empty(n, device=dev, dtype=torch.bfloat16)
        ps.receive(p, out=got).wait()
        expect = float(W * (W + 1) / 2)
        assert torch.allclose(got.float(), torch.full((n,), expect, device=dev),
                              rtol=1e-2), got[:5]
        tm.barrier()
        if rank == 0:
            ps.send(p, "copy",
                    data=torch.arange(n, device=dev).to(torch.bfloat16)).wait()
        tm.barrier()
        ps.receive(p, out=got).wait()
        assert torch.equal(got, torch.arange(n, device=dev).to(torch.bfloat16))
    elif scenario == "shards":
        p = torch.full((64,), float(rank), device=dev)
        ps.register(p)  # shard k comes from rank k's initial value
        tm.barrier()
        ps.receive(p).wait()
        half = 64 // W
        for k in range(W):
            assert torch.equal(p[k * half:(k + 1) * half],
                               torch.full((half,), float(k), device=dev))

    tm.barrier()
    ps.shutdown()
    tm.stop()


@requires_gpu
@pytest.mark.parametrize("scenario", ["rules", "rules16", "shards"])
def test_gpu_ps_two_procs_one_device(scenario):
    import torch.multiprocessing as mp
    port = find_free_port()
    mp.spawn(_ps_worker, (2, port, scenario), nprocs=2, join=True)
