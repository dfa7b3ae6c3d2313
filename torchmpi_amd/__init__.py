"""torchmpi_amd — MI355X-native collective communication layer for PyTorch.

From-scratch re-design of the capabilities of facebookarchive/TorchMPI
(torchmpi/init.lua public API — SURVEY.md section 2.2): start/stop/rank/
size/barrier, tensor collectives (broadcast, reduce, allreduce, allgather,
sendreceive + reduce_scatter/gather/scatter/alltoall), sync and async
variants, a communicator stack with split, an nn synchronization layer and
a sharded parameter server — over TCP (CPU tensors) and RCCL-on-xGMI with
hand-written gfx950 kernels (GPU tensors). No MPI, no CUDA paths.

Typical use (one process per GPU, torchrun-compatible env)::

    import torchmpi_amd as tm
    tm.start(use_gpu=True)
    tm.nn.synchronize_parameters(model)
    ...
    tm.allreduce(loss)          # sync (stream-ordered on GPU)
    h = tm.allreduce(grad, async_op=True)
    ...
    h.wait()
    tm.stop()
"""
import os
from typing import List, Optional

import torch

try:
    from . import _C  # built by setup.py build_ext --inplace
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "torchmpi_amd native extension is not built. Run: "
        "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace "
        f"(import error: {e})") from e

from .bootstrap import create_rendezvous, destroy_rendezvous
from .parallel.communicator import Communicator
from .parallel.hierarchical import hierarchical_allreduce
from .utils import get_logger
from .utils.trace import trace_dump, trace_enabled, trace_op

RedOp = _C.RedOp

_OPS = {
    "sum": RedOp.SUM,
    "max": RedOp.MAX,
    "min": RedOp.MIN,
    "prod": RedOp.PROD,
}

_world: Optional[Communicator] = None
_comm_stack: List[Communicator] = []


def _op(op) -> "_C.RedOp":
    if isinstance(op, str):
        return _OPS[op.lower()]
    return op


# ---------------------------------------------------------------------------
# Lifecycle (reference: mpi.start/stop, SURVEY.md stack 3.1)
# ---------------------------------------------------------------------------

def start(use_gpu: Optional[bool] = None, rank: Optional[int] = None,
          world_size: Optional[int] = None, master_addr: Optional[str] = None,
          master_port: Optional[int] = None) -> Communicator:
    """Initialize the world communicator. Collective: every rank calls it.

    use_gpu=None auto-detects. Device binding follows the reference
    (lib/torch_mpi_cuda.cpp): local rank modulo visible device count.
    """
    global _world, _comm_stack
    if _world is not None:
        return _world
    rdv = create_rendezvous(rank, world_size, master_addr, master_port)
    if use_gpu is None:
        use_gpu = torch.cuda.is_available()
    device = None
    if use_gpu:
        if not torch.cuda.is_available():
            raise RuntimeError("torchmpi_amd.start(use_gpu=True): no GPU found; "
                               "refusing silent CPU fallback")
        local = int(os.environ.get("LOCAL_RANK", rdv.rank))
        device = torch.device("cuda", local % torch.cuda.device_count())
        torch.cuda.set_device(device)
    _world = Communicator(rdv.scoped("world"), list(range(rdv.world_size)), device)
    _comm_stack = [_world]
    get_logger().info(
        f"started: rank {_world.rank()}/{_world.size()} device={device}")
    return _world


def stop() -> None:
    """Tear down all communicators (collective)."""
    global _world, _comm_stack
    if _world is None:
        return
    trace_dump()
    barrier()
    for c in _comm_stack:
        c.shutdown()
    _comm_stack = []
    _world = None
    destroy_rendezvous()


def is_started() -> bool:
    return _world is not None


def world() -> Communicator:
    if _world is None:
        raise RuntimeError("torchmpi_amd.start() has not been called")
    return _world


def communicator() -> Communicator:
    """The current communicator (top of the stack)."""
    if not _comm_stack:
        raise RuntimeError("torchmpi_amd.start() has not been called")
    return _comm_stack[-1]


def rank() -> int:
    return communicator().rank()


def size() -> int:
    return communicator().size()


def barrier(comm: Optional[Communicator] = None) -> None:
    (comm or communicator()).barrier()


# ---------------------------------------------------------------------------
# Communicator stack (reference: torchmpi push/set communicator)
# ---------------------------------------------------------------------------

def split(color: Optional[int], key: int = 0,
          comm: Optional[Communicator] = None) -> Optional[Communicator]:
    return (comm or communicator()).split(color, key)


def push_communicator(comm_or_color, key: int = 0) -> Optional[Communicator]:
    """Push a sub-communicator onto the stack. Accepts an existing
    Communicator or a split color (reference: string-keyed push)."""
    if isinstance(comm_or_color, Communicator):
        sub = comm_or_color
    else:
        color = comm_or_color
        if isinstance(color, str):
            import hashlib
            color = int.from_bytes(
                hashlib.sha1(color.encode()).digest()[:4], "little") % (2 ** 31)
        sub = communicator().split(color, key)
    if sub is not None:
        _comm_stack.append(sub)
    return sub


def pop_communicator() -> None:
    if len(_comm_stack) <= 1:
        raise RuntimeError("cannot pop the world communicator")
    _comm_stack.pop()


# ---------------------------------------------------------------------------
# Collectives (reference: mpi.*Tensor / mpi.async.*Tensor)
# ---------------------------------------------------------------------------

class Handle:
    """Async completion handle (reference: integer handles + syncHandle).

    On GPU, wait() makes the *caller's current stream* wait (cheap, keeps
    overlap); wait_host() blocks the host. On CPU both block the host.
    """

    def __init__(self, inner, is_gpu: bool, post=None):
        self._inner = inner
        self._is_gpu = is_gpu
        self._post = post

    def wait(self):
        self._inner.wait()
        if self._post is not None:
            self._post()
            self._post = None
        return self

    def wait_host(self):
        if self._is_gpu:
            self._inner.wait_host()
        else:
            self._inner.wait()
        if self._post is not None:
            self._post()
            self._post = None
        return self

    def ready(self) -> bool:
        return self._inner.query() if self._is_gpu else self._inner.ready()


def sync_handle(h: Handle):
    """Reference mpi.syncHandle."""
    return h.wait()


def _finish(h: Handle, async_op: bool):
    return h if async_op else h.wait()


def _traced(fn):
    """Chrome-trace wrapper (TORCHMPI_TRACE=prefix)."""
    import functools

    @functools.wraps(fn)
    def wrapper(tensor, *a, **kw):
        if not trace_enabled():
            return fn(tensor, *a, **kw)
        nbytes = (tensor.numel() * tensor.element_size()
                  if hasattr(tensor, "numel") else 0)
        with trace_op(fn.__name__, nbytes):
            return fn(tensor, *a, **kw)
    return wrapper


@_traced
def allreduce(tensor: torch.Tensor, op="sum", async_op: bool = False,
              average: bool = False, comm: Optional[Communicator] = None):
    """In-place allreduce. average=True fuses a 1/world scale (GPU: in the
    reduction kernel epilogue)."""
    c = comm or communicator()
    if tensor.is_cuda:
        alpha = 1.0 / c.size() if average else 1.0
        w = c.engine.allreduce(tensor, _op(op), alpha)
        return _finish(Handle(w, True), async_op)
    w = c.cpu.allreduce(tensor, _op(op))
    post = None
    if average:
        ws = c.size()

        def post():  # noqa: E306
            tensor.div_(ws)
    return _finish(Handle(w, False, post), async_op)


@_traced
def reduce(tensor: torch.Tensor, op="sum", root: int = 0, async_op: bool = False,
           comm: Optional[Communicator] = None):
    """Reduce to `root` (in place at root; other ranks' tensors unchanged)."""
    c = comm or communicator()
    if tensor.is_cuda:
        return _finish(Handle(c.engine.reduce(tensor, _op(op), root), True), async_op)
    return _finish(Handle(c.cpu.reduce(tensor, _op(op), root), False), async_op)


@_traced
def broadcast(tensor: torch.Tensor, root: int = 0, async_op: bool = False,
              comm: Optional[Communicator] = None):
    c = comm or communicator()
    if tensor.is_cuda:
        return _finish(Handle(c.engine.broadcast(tensor, root), True), async_op)
    return _finish(Handle(c.cpu.broadcast(tensor, root), False), async_op)


@_traced
def allgather(tensor: torch.Tensor, out: Optional[torch.Tensor] = None,
              async_op: bool = False, comm: Optional[Communicator] = None):
    """Gather every rank's tensor; returns a tensor of shape
    [world_size, *tensor.shape] (or fills `out`)."""
    c = comm or communicator()
    if out is None:
        out = torch.empty((c.size(),) + tuple(tensor.shape), dtype=tensor.dtype,
                          device=tensor.device)
    if tensor.is_cuda:
        h = Handle(c.engine.allgather(tensor, out), True)
    else:
        h = Handle(c.cpu.allgather(tensor, out), False)
    if async_op:
        return h, out
    h.wait()
    return out


@_traced
def reduce_scatter_(tensor: torch.Tensor, out: torch.Tensor, op="sum",
                    async_op: bool = False, comm: Optional[Communicator] = None):
    """In-place-workspace variant: reduces `tensor` across ranks and scatters
    segment `rank` into `out`, CLOBBERING `tensor` (it is the reduction
    workspace). Zero-copy — use when the input is disposable (e.g. the
    hierarchical allreduce's flat buffer)."""
    c = comm or communicator()
    if tensor.is_cuda:
        return _finish(Handle(c.engine.reduce_scatter(tensor, out, _op(op)), True),
                       async_op)
    return _finish(Handle(c.cpu.reduce_scatter(tensor, out, _op(op)), False),
                   async_op)


@_traced
def reduce_scatter(tensor: torch.Tensor, out: torch.Tensor, op="sum",
                   async_op: bool = False, comm: Optional[Communicator] = None):
    """Reduce `tensor` across ranks and scatter segment `rank` into `out`.
    Input-preserving (MPI_Reduce_scatter semantics): the reduction runs on a
    private copy, one extra D2D/memcpy of the input."""
    work = tensor.clone()
    return reduce_scatter_(work, out, op, async_op=async_op, comm=comm)


@_traced
def sendreceive(send_tensor: torch.Tensor, dst: int, recv_tensor: torch.Tensor,
                src: int, async_op: bool = False,
                comm: Optional[Communicator] = None):
    """Pairwise exchange (reference mpi.sendreceiveTensor / MPI_Sendrecv)."""
    c = comm or communicator()
    if send_tensor.is_cuda:
        return _finish(
            Handle(c.engine.sendrecv(send_tensor, dst, recv_tensor, src), True),
            async_op)
    return _finish(
        Handle(c.cpu.sendrecv(send_tensor, dst, recv_tensor, src), False), async_op)


@_traced
def send(tensor: torch.Tensor, dst: int, async_op: bool = False,
         comm: Optional[Communicator] = None):
    c = comm or communicator()
    if tensor.is_cuda:
        return _finish(Handle(c.engine.send(tensor, dst), True), async_op)
    return _finish(Handle(c.cpu.send(tensor, dst), False), async_op)


@_traced
def recv(tensor: torch.Tensor, src: int, async_op: bool = False,
         comm: Optional[Communicator] = None):
    c = comm or communicator()
    if tensor.is_cuda:
        return _finish(Handle(c.engine.recv(tensor, src), True), async_op)
    return _finish(Handle(c.cpu.recv(tensor, src), False), async_op)


def gather(tensor: torch.Tensor, root: int = 0, out: Optional[torch.Tensor] = None,
           comm: Optional[Communicator] = None):
    c = comm or communicator()
    if c.rank() == root and out is None:
        out = torch.empty((c.size(),) + tuple(tensor.shape), dtype=tensor.dtype,
                          device=tensor.device)
    if tensor.is_cuda:
        if out is None:
            out = torch.empty(0, dtype=tensor.dtype, device=tensor.device)
        Handle(c.engine.gather(tensor, out, root), True).wait()
    else:
        if out is None:
            out = torch.empty(0, dtype=tensor.dtype)
        Handle(c.cpu.gather(tensor, out, root), False).wait()
    return out if c.rank() == root else None


def scatter(tensor_in: Optional[torch.Tensor], out: torch.Tensor, root: int = 0,
            comm: Optional[Communicator] = None):
    c = comm or communicator()
    if tensor_in is None:
        tensor_in = torch.empty(0, dtype=out.dtype, device=out.device)
    if out.is_cuda:
        Handle(c.engine.scatter(tensor_in, out, root), True).wait()
        return out
    Handle(c.cpu.scatter(tensor_in, out, root), False).wait()
    return out


def alltoall(tensor: torch.Tensor, out: Optional[torch.Tensor] = None,
             comm: Optional[Communicator] = None):
    c = comm or communicator()
    if out is None:
        out = torch.empty_like(tensor)
    if tensor.is_cuda:
        Handle(c.engine.alltoall(tensor, out), True).wait()
        return out
    Handle(c.cpu.alltoall(tensor, out), False).wait()
    return out


# Async namespace mirror (reference: mpi.async.*).
class _AsyncNS:
    @staticmethod
    def allreduce(tensor, op="sum", average=False, comm=None):
        return allreduce(tensor, op, async_op=True, average=average, comm=comm)

    @staticmethod
    def broadcast(tensor, root=0, comm=None):
        return broadcast(tensor, root, async_op=True, comm=comm)

    @staticmethod
    def reduce(tensor, op="sum", root=0, comm=None):
        return reduce(tensor, op, root, async_op=True, comm=comm)

    @staticmethod
    def allgather(tensor, out=None, comm=None):
        return allgather(tensor, out, async_op=True, comm=comm)

    @staticmethod
    def sendreceive(send_tensor, dst, recv_tensor, src, comm=None):
        return sendreceive(send_tensor, dst, recv_tensor, src, async_op=True,
                           comm=comm)


asynchronous = _AsyncNS()

from . import nn  # noqa: E402
from . import selector  # noqa: E402
from . import ops  # noqa: E402
from .parameterserver import ParameterServer  # noqa: E402

__all__ = [
    "start", "stop", "is_started", "rank", "size", "barrier", "world",
    "communicator", "split", "push_communicator", "pop_communicator",
    "allreduce", "reduce", "broadcast", "allgather", "reduce_scatter",
    "reduce_scatter_",
    "sendreceive", "send", "recv", "gather", "scatter", "alltoall",
    "asynchronous", "sync_handle", "Handle", "RedOp", "Communicator",
    "hierarchical_allreduce", "ParameterServer", "nn", "ops", "selector",
]
