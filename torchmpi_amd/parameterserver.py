"""Sharded parameter server (reference: torchmpi/parameterserver.lua +
lib/parameterserver.cpp, SURVEY.md sections 2.1/3.4).

Each registered tensor is sharded contiguously across ranks; clients push
updates with 'copy' or 'add' rules and pull values asynchronously. Used to
implement Downpour-SGD and EASGD (examples/downpour.py, examples/easgd.py).

Two transports, picked per tensor:
  - GPU tensors: ONE-SIDED over xGMI — shards live in fine-grained device
    memory mapped into every client via hipIpc; 'copy'/receive are remote
    copy kernels, 'add' is a system-scope atomicAdd kernel (csrc/ipc_ps.hip).
    No server thread exists; the fabric is the server. The reference needed
    an MPI server loop for this — on a fully-connected MI355X
    node it reduces to peer mappings.
  - CPU tensors: the TCP-mesh PS with a per-rank event-loop thread
    (csrc/ps.cpp), matching the reference's design.
"""
from typing import Dict, Optional

import torch

from . import _C
from .bootstrap import Rendezvous
from .utils import env_str

_RULES = {"copy": 0, "add": 1}


class PsHandle:
    def __init__(self, work=None, event=None, post=None, keep=None):
        self._work = work
        self._event = event
        self._post = post
        self._keep = keep  # tensors the in-flight op reads/writes

    def wait(self):
        if self._work is not None:
            self._work.wait()
        if self._event is not None:
            self._event.synchronize()
        if self._post is not None:
            self._post()
            self._post = None
        return self

    def ready(self):
        if self._work is not None:
            return self._work.ready()
        if self._event is not None:
            return self._event.query()
        return True


class ParameterServer:
    """One PS context per communicator. Collective constructor."""

    _counter = 0

    def __init__(self, comm=None):
        from . import communicator
        self.comm = comm or communicator()
        ParameterServer._counter += 1
        self.rdv: Rendezvous = self.comm.rdv.scoped(f"ps{ParameterServer._counter}")
        bind = env_str("TORCHMPI_BIND_ADDR", "127.0.0.1")
        self._ctx = _C.PsContext(self.comm.rank(), bind)
        my_addr = f"{bind}:{self._ctx.listen_port()}".encode()
        addrs = self.rdv.exchange("psaddr", my_addr, ranks=self.comm.global_ranks)
        self._ctx.start(self.comm.rank(), [a.decode() for a in addrs])
        self._gpu = None  # lazy: created on first GPU registration
        self._tids: Dict[int, int] = {}
        self._kind: Dict[int, str] = {}
        self._next_tid = 0

    def _gpu_ctx(self):
        if self._gpu is None:
            dev = (self.comm.device
                   if self.comm.device is not None
                   else torch.device("cuda", torch.cuda.current_device()))
            self._gpu = _C.IpcPs(self.comm.rank(), self.comm.size(), dev.index)
        return self._gpu

    def register(self, tensor: torch.Tensor) -> int:
        """Collective: shard `tensor`'s current value across ranks."""
        tid = self._next_tid
        self._next_tid += 1
        flat = tensor.detach().reshape(-1).contiguous()
        if tensor.is_cuda:
            ctx = self._gpu_ctx()
            handle = ctx.register_local(tid, flat)
            handles = self.rdv.exchange(f"ipc{tid}", handle,
                                        ranks=self.comm.global_ranks)
            ctx.open_peers(tid, [bytes(h) for h in handles])
            self._kind[tid] = "gpu"
        else:
            self._ctx.register_tensor(tid, flat)
            self._kind[tid] = "cpu"
        self._tids[id(tensor)] = tid
        # Registration is collective: no client may push before every owner
        # has its shard store (reference: PS init posts persistent receives
        # before any traffic).
        self.comm.barrier()
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
        src = (tensor if data is None else data).detach().reshape(-1).contiguous()
        if self._kind[tid] == "gpu":
            self._gpu.send(tid, src, _RULES[rule])
            ev = torch.cuda.Event()
            ev.record()
            return PsHandle(event=ev, keep=src)
        return PsHandle(work=self._ctx.send(tid, src, _RULES[rule]), keep=src)

    def receive(self, tensor: torch.Tensor,
                out: Optional[torch.Tensor] = None) -> PsHandle:
        """Pull the current sharded value of registered `tensor` into `out`
        (default: the tensor itself), async."""
        tid = self._tid_of(tensor)
        dst = (tensor if out is None else out).detach().reshape(-1)
        if self._kind[tid] == "gpu":
            assert dst.is_contiguous(), "ps receive: need contiguous destination"
            self._gpu.receive(tid, dst)
            ev = torch.cuda.Event()
            ev.record()
            return PsHandle(event=ev, keep=dst)
        flat = dst.contiguous()
        if flat.data_ptr() != dst.data_ptr():
            w = self._ctx.receive(tid, flat)

            def post():
                dst.copy_(flat)
            return PsHandle(work=w, post=post, keep=(flat, dst))
        return PsHandle(work=self._ctx.receive(tid, flat), keep=flat)

    prefetch = receive  # reference: prefetch == async receive

    def free(self, tensor: torch.Tensor) -> None:
        """Collective: all ranks free together (a live client fetching a
        freed shard would kill the owner's event loop)."""
        self.comm.barrier()
        tid = self._tids.pop(id(tensor), None)
        if tid is not None:
            if self._kind.get(tid) == "gpu":
                self._gpu.free_tensor(tid)
            else:
                self._ctx.free_tensor(tid)
            self._kind.pop(tid, None)

    def shutdown(self):
        """Collective: barrier FIRST (with the event loop still serving), so
        a fast rank cannot stop serving while a slow rank's request is in
        flight (each client waits its own handles before calling this)."""
        self.comm.barrier()
        self._ctx.stop()
        self._gpu = None
