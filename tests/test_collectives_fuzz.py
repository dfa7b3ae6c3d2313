"""Seeded randomized sweep over the verb surface (reference: the
'all combinations' collectives driver). One spawn per
world size; inside, 40 random (verb, dtype, size, root) draws with analytic
expected values — catches edge sizes (n < W, primes, 16B-misaligned counts)
the fixed sweeps miss."""
import random

import torch

from .util import run_distributed

DTYPES = [torch.float32, torch.float64, torch.int32, torch.int64,
          torch.bfloat16]


def _tol(dtype):
    return {torch.bfloat16: 0.05, torch.float16: 1e-2}.get(dtype, 1e-6)


def _fuzz_body(tm, seed):
    W, r = tm.size(), tm.rank()
    rng = random.Random(seed)  # same stream on every rank
    for trial in range(40):
        verb = rng.choice(["allreduce", "broadcast", "reduce", "allgather",
                           "sendrecv", "reduce_scatter"])
        dtype = rng.choice(DTYPES)
        n = rng.choice([1, 2, 3, W - 1 if W > 1 else 1, W, W + 1, 17, 63,
                        101, 1000, 4097, 65537])
        root = rng.randrange(W)
        if verb == "allreduce":
            t = torch.full((n,), float(r + 1)).to(dtype)
            tm.allreduce(t)
            expect = torch.full((n,), float(W * (W + 1) / 2))
            assert torch.allclose(t.float(), expect, rtol=_tol(dtype)), \
                (trial, verb, dtype, n)
        elif verb == "broadcast":
            t = (torch.arange(n).to(dtype) if r == root
                 else torch.zeros(n).to(dtype))
            tm.broadcast(t, root=root)
            assert torch.allclose(t.float(), torch.arange(n).float(),
                                  rtol=_tol(dtype)), (trial, verb, dtype, n)
        elif verb == "reduce":
            t = torch.full((n,), float(r + 1)).to(dtype)
            tm.reduce(t, root=root)
            if r == root:
                expect = torch.full((n,), float(W * (W + 1) / 2))
                assert torch.allclose(t.float(), expect, rtol=_tol(dtype)), \
                    (trial, verb, dtype, n)
        elif verb == "allgather":
            out = tm.allgather(torch.full((n,), float(r)).to(dtype))
            for k in range(W):
                assert torch.allclose(out[k].float(), torch.full((n,), float(k)),
                                      rtol=_tol(dtype)), (trial, verb, dtype, n)
        elif verb == "sendrecv":
            s = torch.full((n,), float(r)).to(dtype)
            rv = torch.zeros(n).to(dtype)
            tm.sendreceive(s, (r + 1) % W, rv, (r - 1) % W)
            assert torch.allclose(rv.float(),
                                  torch.full((n,), float((r - 1) % W)),
                                  rtol=_tol(dtype)), (trial, verb, dtype, n)
        elif verb == "reduce_scatter":
            if dtype in (torch.bfloat16,):
                dtype = torch.float32
            t = torch.arange(n).to(dtype) * (r + 1)
            base, rem = divmod(n, W)
            cnt = base + (1 if r < rem else 0)
            off = r * base + min(r, rem)
            out = torch.zeros(max(cnt, 0)).to(dtype)
            tm.reduce_scatter(t, out)
            sfac = W * (W + 1) / 2
            expect = (torch.arange(off, off + cnt).float() * sfac)
            assert torch.allclose(out.float(), expect, rtol=_tol(dtype)), \
                (trial, verb, dtype, n, cnt)


def test_fuzz_w2():
    run_distributed(_fuzz_body, world=2, args=(1234,))


def test_fuzz_w3():
    run_distributed(_fuzz_body, world=3, args=(77,))


def test_fuzz_w5():
    run_distributed(_fuzz_body, world=5, args=(9,))
