"""CPU transpose tests (host mirror, in-process simulation): the reference's
procedural parity recipe (test/transpose.jl) against the independent oracle.
"""

import math

import numpy as np
import pytest

import oracle as orc
from pencilarrays_amd import (
    Pencil, PencilArray, Topology, build_plan, gather_sim, run_transpose_sim,
)
from util import SWEEP, seeded_parents


def _make_arrays(dims, pdims, decomp, perm, extra, dtype, seed=0xC0FFEE):
    topo = Topology(pdims)
    pen = Pencil(topo, dims, decomp, permute=perm)
    g, parents = seeded_parents(dims, pdims, decomp, perm, extra, dtype, seed)
    arrs = [PencilArray(pen, r, parents[r].copy(), extra)
            for r in range(topo.nranks)]
    return topo, pen, g, arrs


@pytest.mark.parametrize("cfg", SWEEP, ids=lambda c: f"{c[0]}x{c[1]}_{c[2]}to{c[4]}")
def test_sim_transpose_matches_oracle(cfg):
    dims, pdims, di, pi, do, po, extra, dtype = cfg
    topo, pin, g, srcs = _make_arrays(dims, pdims, di, pi, extra, dtype)
    pout = Pencil(topo, dims, do, permute=po)
    dests = [PencilArray.empty(pout, r, dtype=dtype, extra_dims=extra)
             for r in range(topo.nranks)]
    run_transpose_sim(dests, srcs)
    exp = orc.transpose_oracle([s.data for s in srcs], dims, pdims, di, pi,
                               do, po, extra)
    for r in range(topo.nranks):
        assert np.array_equal(dests[r].data, exp[r]), f"rank {r}"
    # gather-compare (test/transpose.jl:6-22)
    assert np.array_equal(gather_sim(srcs), gather_sim(dests))


def test_roundtrip_bit_identity():
    """u1 -> u2 -> u3 -> u2 -> u1 restores u1 bit-exactly
    (test/transpose.jl:48-60), dims (16,21,41), the reference pencils."""
    dims, pdims = (16, 21, 41), (2, 2)
    topo = Topology(pdims)
    pen1 = Pencil(topo, dims, (1, 2))
    pen2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    pen3 = Pencil(topo, dims, (0, 1), permute=(2, 1, 0))
    nr = topo.nranks

    rngs = [np.random.default_rng(42 + r) for r in range(nr)]
    u1 = [PencilArray(pen1, r,
                      rngs[r].standard_normal(pen1.length_local(r)) + 10 * r)
          for r in range(nr)]
    u1_orig = [x.data.copy() for x in u1]
    u2 = [PencilArray.empty(pen2, r) for r in range(nr)]
    u3 = [PencilArray.empty(pen3, r) for r in range(nr)]

    run_transpose_sim(u2, u1)
    assert np.array_equal(gather_sim(u1), gather_sim(u2))
    run_transpose_sim(u3, u2)
    assert np.array_equal(gather_sim(u2), gather_sim(u3))
    run_transpose_sim(u2, u3)
    run_transpose_sim(u1, u2)
    for r in range(nr):
        assert np.array_equal(u1[r].data, u1_orig[r])


def test_two_hop_transpose_raises():
    # test/transpose.jl:45: direct u1 -> u3 is not possible
    topo = Topology((2, 2))
    pen1 = Pencil(topo, (16, 21, 41), (1, 2))
    pen3 = Pencil(topo, (16, 21, 41), (0, 1))
    with pytest.raises(ValueError):
        build_plan(pen1, pen3, 0)


def test_extra_dims_mismatch_raises():
    from pencilarrays_amd import Transposition
    topo = Topology((1, 1))
    pen = Pencil(topo, (4, 4, 4), (1, 2))
    a = PencilArray.empty(pen, 0, extra_dims=(3,))
    b = PencilArray.empty(pen, 0)
    with pytest.raises(ValueError):
        Transposition(b, a)


def test_plan_counts_and_offsets():
    """Send/recv totals and block symmetry across all ranks."""
    dims, pdims = (16, 21, 41), (2, 4)
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, (1, 2))
    Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    plans = [build_plan(Pi, Po, r) for r in range(topo.nranks)]
    for r, p in enumerate(plans):
        total_send = sum(b.send_nelem for b in p.peers if b.peer_k != p.my_k)
        assert total_send == p.send_nelem_total
        assert (p.recv_nelem_total
                + p.peers[p.my_k].recv_nelem) == Po.length_local(r)
        # symmetry: my send to peer == peer's recv from me
        for b in p.peers:
            q = plans[b.global_rank]
            assert q.peers[p.my_k].recv_nelem == b.send_nelem
            assert q.peers[p.my_k].send_nelem == b.recv_nelem


def test_single_rank_transposition_execute():
    """Transposition.execute() without torch.distributed (world=1)."""
    from pencilarrays_amd import Transposition
    topo = Topology((1, 1))
    pen1 = Pencil(topo, (42, 31, 29), (1, 2))
    pen2 = Pencil(topo, (42, 31, 29), (0, 2), permute=(1, 2, 0))
    rng = np.random.default_rng(7)
    src = PencilArray(pen1, 0, rng.standard_normal(pen1.length_local(0)))
    dst = PencilArray.empty(pen2, 0)
    Transposition(dst, src).execute()
    exp = orc.transpose_oracle([src.data], (42, 31, 29), (1, 1), (1, 2),
                               (0, 1, 2), (0, 2), (1, 2, 0), ())
    assert np.array_equal(dst.data, exp[0])
