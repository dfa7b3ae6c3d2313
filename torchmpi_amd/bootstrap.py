"""Rendezvous: TCP store replacing MPI_Init/mpirun (SURVEY.md stack 3.1).

Reads the torchrun-compatible env (RANK, WORLD_SIZE, MASTER_ADDR,
MASTER_PORT, LOCAL_RANK) or explicit arguments, and provides a tiny
key-value barrier/exchange facility used to wire up the TCP socket mesh and
to distribute RCCL unique ids.
"""
import datetime
import os
from typing import Optional

import torch.distributed as dist

from .utils import env_str


class Rendezvous:
    """A namespaced view over one shared TCPStore."""

    def __init__(self, store, rank: int, world_size: int, prefix: str = "tma"):
        self.store = store
        self.rank = rank
        self.world_size = world_size
        self.prefix = prefix

    def scoped(self, suffix: str) -> "Rendezvous":
        return Rendezvous(self.store, self.rank, self.world_size,
                          f"{self.prefix}/{suffix}")

    def set(self, key: str, value: bytes) -> None:
        self.store.set(f"{self.prefix}/{key}", value)

    def get(self, key: str) -> bytes:
        return self.store.get(f"{self.prefix}/{key}")

    def exchange(self, key: str, value: bytes, ranks=None) -> list:
        """Each participating rank publishes `value`; returns all values in
        rank order. `ranks` is the global-rank list (default: all)."""
        if ranks is None:
            ranks = list(range(self.world_size))
        self.set(f"{key}/{self.rank}", value)
        return [self.get(f"{key}/{r}") for r in ranks]


_rendezvous: Optional[Rendezvous] = None


def create_rendezvous(rank=None, world_size=None, master_addr=None,
                      master_port=None) -> Rendezvous:
    global _rendezvous
    if _rendezvous is not None:
        return _rendezvous
    rank = int(os.environ.get("RANK", 0)) if rank is None else rank
    world_size = (int(os.environ.get("WORLD_SIZE", 1))
                  if world_size is None else world_size)
    master_addr = master_addr or env_str("MASTER_ADDR", "127.0.0.1")
    master_port = (int(os.environ.get("MASTER_PORT", 29561))
                   if master_port is None else master_port)
    store = dist.TCPStore(
        master_addr,
        master_port,
        world_size,
        is_master=(rank == 0),
        timeout=datetime.timedelta(seconds=600),
        wait_for_workers=False,
        use_libuv=True,
    )
    _rendezvous = Rendezvous(store, rank, world_size)
    return _rendezvous


def destroy_rendezvous() -> None:
    global _rendezvous
    _rendezvous = None
