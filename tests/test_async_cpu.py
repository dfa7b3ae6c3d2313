"""Async engine: many in-flight handles, per-communicator ordering
(reference async paths of test/collectives.lua — SURVEY.md section 4)."""
import torch

from .util import run_distributed


def _many_handles_body(tm):
    W, r = tm.size(), tm.rank()
    tensors = [torch.full((1000,), float(r + 1 + i)) for i in range(32)]
    handles = [tm.allreduce(t, async_op=True) for t in tensors]
    for i, h in enumerate(handles):
        h.wait()
        s = sum(k + 1 + i for k in range(W))
        assert torch.equal(tensors[i], torch.full((1000,), float(s)))


def test_many_inflight_handles():
    run_distributed(_many_handles_body, world=2)


def _ordering_body(tm):
    # Ops on one communicator execute in submission order: a broadcast
    # submitted after an allreduce must see the allreduced value.
    W, r = tm.size(), tm.rank()
    t = torch.full((100,), float(r + 1))
    h1 = tm.allreduce(t, async_op=True)
    h2 = tm.broadcast(t, root=0, async_op=True)
    h2.wait()
    h1.wait()
    assert torch.equal(t, torch.full((100,), float(W * (W + 1) / 2)))


def test_ordering():
    run_distributed(_ordering_body, world=4)


def _mixed_sync_async_body(tm):
    W, r = tm.size(), tm.rank()
    a = torch.full((10,), float(r))
    h = tm.allreduce(a, async_op=True)
    b = torch.full((10,), 2.0 * r)
    tm.allreduce(b)  # sync op after async on the same communicator
    h.wait()
    assert torch.equal(a, torch.full((10,), float(sum(range(W)))))
    assert torch.equal(b, torch.full((10,), 2.0 * sum(range(W))))


def test_mixed_sync_async():
    run_distributed(_mixed_sync_async_body, world=2)


def _ready_poll_body(tm):
    t = torch.ones(10000)
    h = tm.allreduce(t, async_op=True)
    # ready() must eventually become True and stay consistent with wait()
    for _ in range(10000):
        if h.ready():
            break
    h.wait()
    assert h.ready()


def test_ready_poll():
    run_distributed(_ready_poll_body, world=2)


def _async_namespace_body(tm):
    # reference mpi.async.* namespace parity + sync_handle
    W, r = tm.size(), tm.rank()
    t = torch.full((50,), float(r + 1))
    h = tm.asynchronous.allreduce(t)
    tm.sync_handle(h)
    assert torch.equal(t, torch.full((50,), float(W * (W + 1) / 2)))
    b = torch.arange(5.0) if r == 0 else torch.zeros(5)
    tm.sync_handle(tm.asynchronous.broadcast(b, root=0))
    assert torch.equal(b, torch.arange(5.0))
    h, out = tm.asynchronous.allgather(torch.full((3,), float(r)))
    h.wait()
    for k in range(W):
        assert torch.equal(out[k], torch.full((3,), float(k)))
    s = torch.full((4,), float(r))
    rv = torch.zeros(4)
    tm.sync_handle(tm.asynchronous.sendreceive(s, (r + 1) % W, rv, (r - 1) % W))
    assert torch.equal(rv, torch.full((4,), float((r - 1) % W)))


def test_async_namespace():
    run_distributed(_async_namespace_body, world=2)
