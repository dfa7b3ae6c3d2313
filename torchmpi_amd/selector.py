"""Collective implementation selector (reference: mpi.collectiveSelector in
torchmpi/init.lua — SURVEY.md section 2.2 — which picked cpu/gpu x
mpi/nccl/p2p x sync/async per call).

In torchmpi_amd the choice is algorithmic, not backend-al (there is exactly
one transport per device class); this module exposes WHICH algorithm a call
will use, mirroring the engine's dispatch rules, for introspection, tests
and docs.
"""
import math
from dataclasses import dataclass

import torch

from .utils import env_int


@dataclass
class Selection:
    device: str       # "cpu" | "gpu"
    transport: str    # "tcp" | "rccl-p2p"
    algorithm: str    # e.g. "ring", "multi-ring(7)", "recursive-halving"
    graphable: bool   # will be hipGraph-cached after first use


def select_allreduce(tensor: torch.Tensor, world_size: int) -> Selection:
    """Mirror of RcclEngine::allreduce / CpuComm::allreduce dispatch."""
    nbytes = tensor.numel() * tensor.element_size()
    rhd_max = env_int("TORCHMPI_RHD_MAX_BYTES", 256 << 10)
    pow2 = world_size > 0 and (world_size & (world_size - 1)) == 0

    if not tensor.is_cuda:
        if pow2 and nbytes <= rhd_max and tensor.numel() >= world_size:
            return Selection("cpu", "tcp", "recursive-halving", False)
        return Selection("cpu", "tcp", "ring", False)

    if world_size == 1:
        return Selection("gpu", "rccl-p2p", "local", False)
    if pow2 and nbytes <= rhd_max and tensor.numel() >= world_size:
        return Selection("gpu", "rccl-p2p", "recursive-halving", False)
    multiring_min = env_int("TORCHMPI_MULTIRING_MIN_BYTES", 1 << 20)
    graph_min = env_int("TORCHMPI_GRAPH_MIN_BYTES", 1 << 20)
    nrings = _nrings(world_size)
    if nbytes >= multiring_min and nrings > 1:
        return Selection("gpu", "rccl-p2p", f"multi-ring({nrings})",
                         nbytes >= graph_min)
    return Selection("gpu", "rccl-p2p", "ring", nbytes >= graph_min)


def _nrings(world: int) -> int:
    from .parallel.communicator import ring_permutations
    return len(ring_permutations(world))


def describe(tensor: torch.Tensor, world_size: int = None) -> str:
    from . import communicator, is_started
    if world_size is None:
        world_size = communicator().size() if is_started() else 1
    s = select_allreduce(tensor, world_size)
    return (f"allreduce[{tensor.numel()}x{tensor.dtype}] @W={world_size}: "
            f"{s.device}/{s.transport}/{s.algorithm}"
            f"{' +hipGraph' if s.graphable else ''}")
