"""Communicator: hierarchical process groups over TCP (CPU tensors) and
RCCL-over-xGMI (GPU tensors).

Re-designs TorchMPI's communicator stack + resource cache
(lib/resources.cpp): a Communicator owns
  - a TCP socket mesh + worker thread (CPU collectives, control plane),
  - lazily, an RcclEngine: R link-disjoint ring communicators over RCCL
    with dedicated HIP streams (GPU collectives).
`split(color, key)` creates sub-communicators (reference MPI_Comm_split
keyed from Lua).
"""
import math
import os
from typing import List, Optional

import torch

from .. import _C
from ..bootstrap import Rendezvous
from ..utils import env_int, env_str, get_logger

# 7 directed-edge-disjoint Hamiltonian cycles on 8 ranks: every GPU sends on
# all 7 of its outgoing xGMI links and receives on all 7 incoming ones
# (verified: the 7 cycles cover all 56 directed links exactly once).
K8_RINGS = [
    [0, 5, 7, 2, 1, 6, 4, 3],
    [0, 4, 5, 6, 2, 7, 3, 1],
    [0, 1, 4, 6, 5, 2, 3, 7],
    [0, 7, 1, 5, 3, 4, 2, 6],
    [0, 6, 3, 2, 4, 1, 7, 5],
    [0, 3, 5, 4, 7, 6, 1, 2],
    [0, 2, 5, 1, 3, 6, 7, 4],
]


def ring_permutations(world: int) -> List[List[int]]:
    """Link-disjoint ring orders for `world` ranks on a fully-connected
    xGMI node. A step-k ring is Hamiltonian iff gcd(k, world) == 1."""
    if world == 1:
        return [[0]]
    if world == 8:
        rings = [list(r) for r in K8_RINGS]
    else:
        rings = []
        for k in range(1, world):
            if math.gcd(k, world) != 1:
                continue
            rings.append([(k * i) % world for i in range(world)])
            if len(rings) >= 7:
                break
    want = env_int("TORCHMPI_NUM_RINGS", 0)
    if want > 0:
        rings = rings[:want]
    return rings


class Communicator:
    """A process group. `global_ranks` lists member global ranks (sorted by
    group rank); `rdv` is a Rendezvous scoped uniquely to this communicator."""

    def __init__(self, rdv: Rendezvous, global_ranks: List[int],
                 device: Optional[torch.device] = None):
        self.rdv = rdv
        self.global_ranks = list(global_ranks)
        self.global_rank = rdv.rank
        assert self.global_rank in self.global_ranks, \
            "rank not a member of this communicator"
        self._rank = self.global_ranks.index(self.global_rank)
        self._size = len(self.global_ranks)
        self.device = device
        self._engine = None
        self._split_counter = 0
        self.log = get_logger()

        bind = env_str("TORCHMPI_BIND_ADDR", "127.0.0.1")
        self._cc = _C.CpuComm(self._rank, bind)
        my_addr = f"{bind}:{self._cc.listen_port()}".encode()
        addrs = rdv.exchange("cpuaddr", my_addr, ranks=self.global_ranks)
        self._cc.connect(self._rank, [a.decode() for a in addrs])

    # -- identity ----------------------------------------------------------

    def rank(self) -> int:
        return self._rank

    def size(self) -> int:
        return self._size

    # -- engines -----------------------------------------------------------

    @property
    def cpu(self):
        return self._cc

    @property
    def engine(self):
        """The RCCL engine (created on first GPU collective)."""
        if self._engine is None:
            if not torch.cuda.is_available():
                raise RuntimeError(
                    "torchmpi_amd: GPU collective requested but no GPU is "
                    "available (the native RCCL engine refuses to fall back)")
            dev = self.device if self.device is not None else torch.device(
                "cuda", torch.cuda.current_device())
            rings = ring_permutations(self._size)
            nuids = len(rings)
            if self._rank == 0:
                uids = [_C.nccl_unique_id() for _ in range(nuids)]
                for i, u in enumerate(uids):
                    self.rdv.set(f"ncclid/{i}", u)
            else:
                uids = [self.rdv.get(f"ncclid/{i}") for i in range(nuids)]
            self.log.info(f"init RCCL engine: world={self._size} rings={len(rings)}")
            self._engine = _C.RcclEngine(self._rank, self._size, dev.index,
                                         uids, rings)
        return self._engine

    # -- split -------------------------------------------------------------

    def split(self, color: Optional[int], key: int = 0) -> Optional["Communicator"]:
        """Collective. Ranks with the same color form a sub-communicator,
        ordered by (key, rank). color=None -> no membership (returns None)."""
        self._split_counter += 1
        my = torch.tensor(
            [(-1 if color is None else int(color)), int(key), self.global_rank],
            dtype=torch.int64)
        out = torch.empty(self._size * 3, dtype=torch.int64)
        self._cc.allgather(my, out).wait()
        rows = out.view(self._size, 3).tolist()
        if color is None:
            return None
        members = [(k, gr) for c, k, gr in rows if c == int(color)]
        members.sort()
        granks = [gr for _, gr in members]
        sub_rdv = self.rdv.scoped(f"split{self._split_counter}/{int(color)}")
        return Communicator(sub_rdv, granks, self.device)

    def split_by_node(self) -> "Communicator":
        """Intra-node communicator (reference: hostname-keyed MPI_Comm_split).
        Single-node on MI355X boxes, so this is usually a full dup."""
        import hashlib
        import socket
        host = socket.gethostname().encode()
        h = int.from_bytes(hashlib.sha1(host).digest()[:4], "little") % (2 ** 31)
        return self.split(color=h, key=self.global_rank)

    # -- barrier -----------------------------------------------------------

    def barrier(self):
        """Host-level barrier over the TCP mesh (reference MPI_Barrier)."""
        self._cc.barrier().wait()

    def shutdown(self):
        if self._engine is not None:
            self._engine = None
        self._cc = None
