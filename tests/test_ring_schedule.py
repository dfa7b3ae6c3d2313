"""Simulation tests of the GPU engine's ring schedules.

The 8-GPU xGMI path cannot run on a 1-GPU CI box, so correctness must hold
by construction: _C.ring_schedule exposes the exact step lists the engine
executes (same C++ function), and these tests simulate the message passing
for every world size and ring permutation, checking:
  - reduce-scatter: rank g ends owning the fully reduced segment g
  - allgather: every rank ends with every reduced segment
  - link-disjointness of the 8-GPU ring set (all 56 directed xGMI links)
"""
import math

import pytest

from torchmpi_amd import _C
from torchmpi_amd.parallel.communicator import K8_RINGS, ring_permutations


def simulate_allreduce(world, perm):
    """Each rank holds W segment-values; simulate the engine's RS + AG
    schedules over the ring defined by perm. Values: rank r contributes
    (r+1) to every segment; full sum = W(W+1)/2."""
    full = world * (world + 1) // 2
    data = [[r + 1 for _ in range(world)] for r in range(world)]
    pos_of = {g: i for i, g in enumerate(perm)}

    rs = _C.ring_schedule(world, perm, 0)
    assert len(rs) == world - 1
    for step in rs:
        # collect all sends first (simultaneous exchange), then apply
        incoming = {}
        for p in range(world):
            send_seg, recv_seg = step[p]
            succ = perm[(p + 1) % world]
            incoming[succ] = (send_seg, data[perm[p]][send_seg])
            # consistency: successor's recv_seg must equal our send_seg
            assert step[(p + 1) % world][1] == send_seg
        for g, (seg, val) in incoming.items():
            data[g][seg] += val

    # After RS: rank g owns segment g fully reduced.
    for g in range(world):
        assert data[g][g] == full, (world, perm, g, data[g])

    ag = _C.ring_schedule(world, perm, 1)
    for step in ag:
        incoming = {}
        for p in range(world):
            send_seg, recv_seg = step[p]
            succ = perm[(p + 1) % world]
            assert step[(p + 1) % world][1] == send_seg
            incoming[succ] = (send_seg, data[perm[p]][send_seg])
        for g, (seg, val) in incoming.items():
            data[g][seg] = val

    for g in range(world):
        for s in range(world):
            assert data[g][s] == full, (world, perm, g, s, data[g])


@pytest.mark.parametrize("world", [2, 3, 4, 5, 6, 7, 8])
def test_all_ring_permutations_allreduce(world):
    for perm in ring_permutations(world):
        assert sorted(perm) == list(range(world))
        simulate_allreduce(world, perm)


def test_k8_rings_cover_all_links():
    edges = set()
    for ring in K8_RINGS:
        for i in range(8):
            e = (ring[i], ring[(i + 1) % 8])
            assert e not in edges, f"directed link {e} reused"
            edges.add(e)
    assert len(edges) == 56  # all 8x7 directed xGMI links


def test_step_rings_hamiltonian():
    for world in range(2, 9):
        for perm in ring_permutations(world):
            # each perm must be a single cycle visiting everyone
            assert len(set(perm)) == world


def test_ring_count_matches_engine_env(monkeypatch):
    monkeypatch.setenv("TORCHMPI_NUM_RINGS", "2")
    rings = ring_permutations(8)
    assert len(rings) == 2
    monkeypatch.delenv("TORCHMPI_NUM_RINGS")
    assert len(ring_permutations(8)) == 7


def test_segment_parts_partition():
    """Multi-ring verbs split every global segment into per-ring parts; the
    parts must disjointly cover each segment and rank g's parts must be
    contiguous (= its global segment)."""
    for world in (2, 4, 8):
        for total in (7, 64, 1000, 12345):
            # global segments
            base, rem = divmod(total, world)
            off, cnt, o = [], [], 0
            for g in range(world):
                c = base + (1 if g < rem else 0)
                off.append(o); cnt.append(c); o += c
            for nrings in (1, 2, 7):
                covered = [0] * world
                prev_end = dict((g, off[g]) for g in range(world))
                for r in range(nrings):
                    off_r, cnt_r = _C.segment_parts(off, cnt, r, nrings, 4)
                    for g in range(world):
                        if cnt_r[g] == 0:
                            continue
                        assert off_r[g] == prev_end[g]  # contiguous parts
                        prev_end[g] = off_r[g] + cnt_r[g]
                        covered[g] += cnt_r[g]
                assert covered == cnt, (world, total, nrings, covered, cnt)


def test_selector_mirrors_dispatch():
    import torch
    from torchmpi_amd.selector import select_allreduce
    # small pow2 world -> recursive halving
    s = select_allreduce(torch.zeros(1000), 4)
    assert s.algorithm == "recursive-halving" and s.device == "cpu"
    # large -> ring on CPU
    s = select_allreduce(torch.zeros(1 << 22), 4)
    assert s.algorithm == "ring"
    # non-pow2 world: never RHD
    s = select_allreduce(torch.zeros(100), 3)
    assert s.algorithm == "ring"
