"""In-place (aliased) transposes via ManyPencilArray — §8(f) row 1
(multiarrays.jl:106-143; aliasing path Transpositions.jl:250-264; tests
modelled on test/pencils.jl:224-239)."""

import math

import numpy as np
import pytest

import oracle as orc
from pencilarrays_amd import (
    ManyPencilArray, Pencil, PencilArray, Topology, Transposition,
    build_plan, gather_sim, run_transpose_sim,
)
from util import seeded_parents


def _pencil_chain(dims=(16, 21, 41), pdims=(2, 2)):
    topo = Topology(pdims)
    p1 = Pencil(topo, dims, (1, 2))
    p2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    p3 = Pencil(topo, dims, (0, 1), permute=(2, 1, 0))
    return topo, (p1, p2, p3)


def test_many_pencil_array_layout():
    topo, pens = _pencil_chain()
    m = ManyPencilArray(pens, rank=0)
    assert len(m) == 3
    n_expected = max(p.length_local(0) for p in pens)
    assert m.data.shape[0] == n_expected
    for a, p in zip(m.arrays, pens):
        assert a.pencil is p
        assert np.shares_memory(a.data, m.data)


def test_aliased_plan_structure():
    topo, (p1, p2, _) = _pencil_chain()
    for rank in range(topo.nranks):
        pl = build_plan(p1, p2, rank, aliased=True)
        assert pl.local is None
        assert pl.self_pack is not None and pl.self_unpack is not None
        # recv buffer now includes the self tail (Transpositions.jl:317)
        assert pl.recv_nelem_total == p2.length_local(rank)


def test_inplace_roundtrip_sim():
    """u1 -> u2 -> u3 -> u2 -> u1 entirely in place (one buffer per rank),
    bit-exact (pencils.jl:224-239 recipe), vs the oracle at each hop."""
    dims, pdims = (16, 21, 41), (2, 2)
    topo, pens = _pencil_chain(dims, pdims)
    p1, p2, p3 = pens
    nr = topo.nranks

    g, parents = seeded_parents(dims, pdims, (1, 2), (0, 1, 2), (),
                                np.float64)
    ms = [ManyPencilArray(pens, r) for r in range(nr)]
    for r in range(nr):
        ms[r].first.data[:] = parents[r]
    orig = [m.first.data.copy() for m in ms]

    def hop(i, j, din, pin, dout, pout, check_parents):
        run_transpose_sim([m[j] for m in ms], [m[i] for m in ms])
        exp = orc.transpose_oracle(check_parents, dims, pdims, din, pin,
                                   dout, pout, ())
        for r in range(nr):
            ln = pens[j].length_local(r)
            assert np.array_equal(ms[r][j].data[:ln], exp[r]), f"rank {r}"
        return exp

    id3 = (0, 1, 2)
    e2 = hop(0, 1, (1, 2), id3, (0, 2), (1, 2, 0), parents)
    e3 = hop(1, 2, (0, 2), (1, 2, 0), (0, 1), (2, 1, 0), e2)
    hop(2, 1, (0, 1), (2, 1, 0), (0, 2), (1, 2, 0), e3)
    hop(1, 0, (0, 2), (1, 2, 0), (1, 2), id3, e2)
    for r in range(nr):
        ln = p1.length_local(r)
        assert np.array_equal(ms[r].first.data[:ln], orig[r][:ln])


def test_inplace_local_permute_single_rank():
    """Same decomposition, aliased, pure permutation (permute_local!
    in-place staging, Transpositions.jl:250-264)."""
    dims = (12, 10, 8)
    topo = Topology((1, 1))
    pa_ = Pencil(topo, dims, (1, 2))
    pb = Pencil(topo, dims, (1, 2), permute=(2, 0, 1))
    rng = np.random.default_rng(3)
    buf = rng.standard_normal(pa_.length_local(0))
    src = PencilArray(pa_, 0, buf)
    dst = PencilArray(pb, 0, buf)  # aliased!
    t = Transposition(dst, src)
    assert t.aliased
    src_copy = buf.copy()
    t.execute()
    exp = orc.transpose_oracle([src_copy], dims, (1, 1), (1, 2), (0, 1, 2),
                               (1, 2), (2, 0, 1), ())[0]
    assert np.array_equal(buf, exp)


def test_nonaliased_not_staged():
    topo, (p1, p2, _) = _pencil_chain()
    pl = build_plan(p1, p2, 0, aliased=False)
    assert pl.self_pack is None and pl.local is not None
