"""Communicator split / stack / hierarchical composition (reference:
multiple-communicator tests + hierarchical 2-level paths — SURVEY.md
section 4 and section 2.3 row 4)."""
import torch

from .util import run_distributed


def _split_parity_body(tm):
    W, r = tm.size(), tm.rank()
    sub = tm.split(color=r % 2, key=r)
    assert sub.size() == W // 2 + (W % 2 if r % 2 == 0 else 0)
    t = torch.full((100,), float(r + 1))
    tm.allreduce(t, comm=sub)
    expect = sum(k + 1 for k in range(W) if k % 2 == r % 2)
    assert torch.equal(t, torch.full((100,), float(expect)))
    # collectives on the parent still work after a split
    t2 = torch.full((10,), 1.0)
    tm.allreduce(t2)
    assert torch.equal(t2, torch.full((10,), float(W)))


def test_split_parity():
    run_distributed(_split_parity_body, world=4)


def _split_none_body(tm):
    r = tm.rank()
    sub = tm.split(color=0 if r == 0 else None)
    if r == 0:
        assert sub is not None and sub.size() == 1
    else:
        assert sub is None


def test_split_none_color():
    run_distributed(_split_none_body, world=2)


def _push_pop_body(tm):
    W, r = tm.size(), tm.rank()
    sub = tm.push_communicator(r % 2, key=r)
    assert tm.size() == sub.size()
    t = torch.full((10,), float(r + 1))
    tm.allreduce(t)  # uses the pushed sub-communicator
    expect = sum(k + 1 for k in range(W) if k % 2 == r % 2)
    assert torch.equal(t, torch.full((10,), float(expect)))
    tm.pop_communicator()
    assert tm.size() == W


def test_push_pop_stack():
    run_distributed(_push_pop_body, world=4)


def _hierarchical_body(tm):
    W, r = tm.size(), tm.rank()
    assert W == 4
    # 2 groups of 2: intra = same half, inter = same index within half
    intra = tm.split(color=r // 2, key=r)
    inter = tm.split(color=r % 2, key=r)
    t = torch.arange(1000, dtype=torch.float32) + (r + 1)
    tm.hierarchical_allreduce(t, intra, inter)
    expect = torch.arange(1000, dtype=torch.float32) * W + sum(
        k + 1 for k in range(W))
    assert torch.allclose(t, expect), (t[:4], expect[:4])


def test_hierarchical_allreduce():
    run_distributed(_hierarchical_body, world=4)


def _hierarchical_avg_body(tm):
    W, r = tm.size(), tm.rank()
    intra = tm.split(color=r // 2, key=r)
    inter = tm.split(color=r % 2, key=r)
    t = torch.arange(1000, dtype=torch.float32) + (r + 1)
    tm.hierarchical_allreduce(t, intra, inter, average=True)
    # mean over ALL wi*we ranks (regression: was scaled only by 1/inter.size)
    expect = torch.arange(1000, dtype=torch.float32) + \
        sum(k + 1 for k in range(W)) / W
    assert torch.allclose(t, expect), (t[:4], expect[:4])


def test_hierarchical_allreduce_average():
    run_distributed(_hierarchical_avg_body, world=4)


def _nested_split_body(tm):
    W, r = tm.size(), tm.rank()
    half = tm.split(color=r // 2, key=r)
    quarter = half.split(color=half.rank(), key=0)
    assert quarter.size() == 1
    t = torch.full((5,), float(r))
    tm.allreduce(t, comm=quarter)
    assert torch.equal(t, torch.full((5,), float(r)))


def test_nested_split():
    run_distributed(_nested_split_body, world=4)


def _split_by_node_body(tm):
    sub = tm.split_by_node() if hasattr(tm, "split_by_node") else \
        tm.communicator().split_by_node()
    assert sub.size() == tm.size()  # single host in CI


def test_split_by_node():
    run_distributed(_split_by_node_body, world=2)


def _hier_2x4_body(tm):
    """BASELINE config #5 topology: two groups of four (intra), four pairs
    (inter); hierarchical allreduce must equal the flat allreduce."""
    W, r = tm.size(), tm.rank()
    assert W == 8
    intra = tm.split(color=r // 4, key=r)   # 2 x 4-rank groups
    inter = tm.split(color=r % 4, key=r)    # 4 x 2-rank groups
    assert intra.size() == 4 and inter.size() == 2
    n = 4099  # not divisible by 4 -> unequal shards path
    t = torch.arange(n, dtype=torch.float32) * (r + 1)
    flat = t.clone()
    tm.allreduce(flat)
    tm.hierarchical_allreduce(t, intra, inter)
    assert torch.allclose(t, flat), (t[:4], flat[:4])


def test_hierarchical_2x4():
    run_distributed(_hier_2x4_body, world=8)


def _push_string_key_body(tm):
    """Reference parity: TorchMPI pushed communicators by STRING key; all
    ranks pushing the same string land in one subgroup."""
    W, r = tm.size(), tm.rank()
    sub = tm.push_communicator("myKey")
    assert sub is not None and sub.size() == W  # same key => same color
    t = torch.full((10,), float(r + 1))
    tm.allreduce(t)  # acts on top of stack (the pushed sub-communicator)
    assert torch.equal(t, torch.full((10,), float(W * (W + 1) / 2)))
    tm.pop_communicator()
    t2 = torch.full((4,), 1.0)
    tm.allreduce(t2)  # back on world
    assert torch.equal(t2, torch.full((4,), float(W)))


def test_push_communicator_string_key():
    run_distributed(_push_string_key_body, world=3)
