"""Parameter server: shard placement, copy/add rules, prefetch coherence,
handle sync (reference test/parameterserver.lua — SURVEY.md section 4)."""
import torch

from .util import run_distributed


def _rules_body(tm):
    W, r = tm.size(), tm.rank()
    ps = tm.ParameterServer()
    n = 103  # deliberately not divisible by W
    p = torch.zeros(n)
    ps.register(p)

    # 'add' from every rank: value becomes sum of rank deltas.
    ps.send(p, "add", data=torch.full((n,), float(r + 1))).wait()
    tm.barrier()
    got = torch.empty(n)
    ps.receive(p, out=got).wait()
    assert torch.allclose(got, torch.full((n,), float(W * (W + 1) / 2))), got[:5]

    # 'copy' from rank 0 overwrites.
    tm.barrier()
    if r == 0:
        ps.send(p, "copy", data=torch.arange(n, dtype=torch.float32)).wait()
    tm.barrier()
    ps.receive(p, out=got).wait()
    assert torch.equal(got, torch.arange(n, dtype=torch.float32))
    ps.shutdown()


def test_ps_rules():
    run_distributed(_rules_body, world=4)


def _mutual_large_fetch_body(tm):
    """Regression (advisor, round 1): two ranks simultaneously fetching each
    other's shards LARGER than the combined socket buffers used to mutually
    block both event loops (blocking kData sends). The loops now drain
    outbound queues non-blockingly, so this completes."""
    W, r = tm.size(), tm.rank()
    n = 8 << 20  # 32 MB fp32: shard snapshots far exceed socket buffers
    ps = tm.ParameterServer()
    p = torch.full((n,), float(r + 1))
    ps.register(p)
    tm.barrier()
    got = torch.empty(n)
    ps.receive(p, out=got).wait()  # both ranks fetch at the same time
    base = n // W
    off = 0
    for k in range(W):
        cnt = base + (1 if k < n % W else 0)
        assert torch.equal(got[off], torch.tensor(float(k + 1))), (k, got[off])
        off += cnt
    ps.shutdown()


def test_ps_mutual_large_fetch():
    run_distributed(_mutual_large_fetch_body, world=2)


def _receive_into_registered_body(tm):
    W, r = tm.size(), tm.rank()
    ps = tm.ParameterServer()
    p = torch.full((50,), float(r))  # different initial values per rank
    ps.register(p)  # shards: rank k's slice comes from rank k's copy
    tm.barrier()
    ps.receive(p).wait()
    # Each shard slice equals its owner's initial value.
    base, rem = divmod(50, W)
    off = 0
    for k in range(W):
        cnt = base + (1 if k < rem else 0)
        assert torch.equal(p[off:off + cnt], torch.full((cnt,), float(k)))
        off += cnt
    ps.shutdown()


def test_ps_shard_placement():
    run_distributed(_receive_into_registered_body, world=4)


def _downpour_body(tm):
    """Downpour-style loop: push grads with 'add', pull params periodically
    (reference examples/mnist parameterserver example)."""
    W, r = tm.size(), tm.rank()
    ps = tm.ParameterServer()
    params = torch.zeros(64)
    ps.register(params)
    for step in range(5):
        grad = torch.full((64,), 0.1 * (r + 1))
        ps.send(params, "add", data=-0.1 * grad).wait()
        tm.barrier()
    ps.receive(params).wait()
    expect = -0.01 * sum(k + 1 for k in range(W)) * 5
    assert torch.allclose(params, torch.full((64,), expect), atol=1e-5), params[:4]
    ps.shutdown()


def test_ps_downpour_loop():
    run_distributed(_downpour_body, world=2)


def _multi_tensor_body(tm):
    W, r = tm.size(), tm.rank()
    ps = tm.ParameterServer()
    a = torch.zeros(10)
    b = torch.zeros(20, dtype=torch.float64)
    ps.register(a)
    ps.register(b)
    ps.send(a, "add", data=torch.ones(10)).wait()
    ps.send(b, "add", data=torch.full((20,), 2.0, dtype=torch.float64)).wait()
    tm.barrier()
    ga, gb = torch.empty(10), torch.empty(20, dtype=torch.float64)
    ps.receive(a, out=ga).wait()
    ps.receive(b, out=gb).wait()
    assert torch.allclose(ga, torch.full((10,), float(W)))
    assert torch.allclose(gb, torch.full((20,), 2.0 * W, dtype=torch.float64))
    ps.shutdown()


def test_ps_multi_tensor():
    run_distributed(_multi_tensor_body, world=2)
