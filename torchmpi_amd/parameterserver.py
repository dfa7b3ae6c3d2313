"""Sharded parameter server (reference: torchmpi/parameterserver.lua +
lib/parameterserver.cpp, SURVEY.md sections 2.1/3.4).

Each registered tensor is sharded round-robin-contiguously across ranks;
clients push updates with 'copy' or 'add' rules and pull values
asynchronously. Used to implement Downpour-SGD and EASGD
(examples/downpour_mnist.py, examples/easgd.py).

GPU tensors are staged through host memory (the PS is the async-SGD control
path; the bandwidth-critical path is the collective engine).
"""
from typing import Dict, Optional

import torch

from . import _C
from .bootstrap import Rendezvous
from .utils import env_str

_RULES = {"copy": 0, "add": 1}


class PsHandle:
    def __init__(self, work, post=None):
        self._work = work
        self._post = post

    def wait(self):
        self._work.wait()
        if self._post is not None:
            self._post()
            self._post = None
        return self

    def ready(self):
        return self._work.ready()


class ParameterServer:
    """One PS context per communicator. Collective constructor."""

    _counter = 0

    def __init__(self, comm=None):
        from . import communicator
        self.comm = comm or communicator()
        ParameterServer._counter += 1
        rdv: Rendezvous = self.comm.rdv.scoped(f"ps{ParameterServer._counter}")
        bind = env_str("TORCHMPI_BIND_ADDR", "127.0.0.1")
        self._ctx = _C.PsContext(self.comm.rank(), bind)
        my_addr = f"{bind}:{self._ctx.listen_port()}".encode()
        addrs = rdv.exchange("psaddr", my_addr, ranks=self.comm.global_ranks)
        self._ctx.start(self.comm.rank(), [a.decode() for a in addrs])
        self._tids: Dict[int, int] = {}
        self._staging: Dict[int, torch.Tensor] = {}
        self._next_tid = 0

    def register(self, tensor: torch.Tensor) -> int:
        """Collective: shard `tensor`'s current value across ranks.
        Returns the tensor id."""
        tid = self._next_tid
        self._next_tid += 1
        host = tensor.detach().reshape(-1).cpu().contiguous()
        self._ctx.register_tensor(tid, host)
        self._tids[id(tensor)] = tid
        if tensor.is_cuda:
            self._staging[tid] = torch.empty_like(host, pin_memory=True)
        return tid

    def _tid_of(self, tensor: torch.Tensor) -> int:
        tid = self._tids.get(id(tensor))
        if tid is None:
            raise KeyError("tensor is not registered with this parameter server")
        return tid

    def send(self, tensor: torch.Tensor, rule: str = "add",
             data: Optional[torch.Tensor] = None) -> PsHandle:
        """Push `data` (default: the registered tensor's value) to the shard
        owners with `rule` in {'copy','add'}."""
        tid = self._tid_of(tensor)
        src = tensor if data is None else data
        flat = src.detach().reshape(-1)
        if flat.is_cuda:
            stage = self._staging[tid]
            stage.copy_(flat, non_blocking=False)
            flat = stage
        else:
            flat = flat.contiguous()
        return PsHandle(self._ctx.send(tid, flat, _RULES[rule]))

    def receive(self, tensor: torch.Tensor,
                out: Optional[torch.Tensor] = None) -> PsHandle:
        """Pull the current sharded value of registered `tensor` into `out`
        (default: the tensor itself), async."""
        tid = self._tid_of(tensor)
        flat = (tensor if out is None else out).detach().reshape(-1)
        if flat.is_cuda:
            stage = self._staging[tid]
            w = self._ctx.receive(tid, stage)

            def post():
                flat.copy_(stage)
            return PsHandle(w, post)
        return PsHandle(self._ctx.receive(tid, flat))

    prefetch = receive  # reference: prefetch == async receive

    def free(self, tensor: torch.Tensor) -> None:
        tid = self._tids.pop(id(tensor), None)
        if tid is not None:
            self._ctx.free_tensor(tid)
            self._staging.pop(tid, None)

    def shutdown(self):
        self._ctx.stop()
