"""The PencilFFTs consumer pattern end-to-end: a distributed 3-D FFT
computed as per-axis local FFTs + pencil transposes, compared against a
single whole-array FFT.  This exercises exactly what the reference's
memory-order permutations exist for (README.md:29-31: PencilArrays is the
layer under PencilFFTs): after each transpose the next FFT axis is the
CONTIGUOUS memory axis of the new pencil.

CPU version runs the multi-rank simulation on a 2x2 grid; the GPU version
(world=1, torch.fft + the native engine's transposes) is in test_gpu.py.
"""

import numpy as np
import pytest

from pencilarrays_amd import (
    ManyPencilArray, Pencil, PencilArray, Topology, run_transpose_sim,
    gather_sim,
)

import oracle as orc


def fft_pencils(topo, dims):
    """x-, y-, z-pencils with the PencilFFTs memory layouts
    (test/transpose.jl:28-30 pencil chain)."""
    p1 = Pencil(topo, dims, (1, 2))                      # x contiguous
    p2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))   # y contiguous
    p3 = Pencil(topo, dims, (0, 1), permute=(2, 1, 0))   # z contiguous
    return p1, p2, p3


def local_fft_along(x: PencilArray, logical_dim: int) -> None:
    """FFT along one GLOBAL logical dimension, which must be complete on
    every rank (that's what the pencil chain guarantees); operates on the
    memory view in place."""
    mv = x.parent_memview()  # axes in memory order, axis 0 fastest
    # the logical dim sits at memory position inv(perm)[dim]
    from pencilarrays_amd.permutations import perm_inv
    mem_axis = perm_inv(x.pencil.perm)[logical_dim]
    assert mem_axis == 0, "FFT axis should be the contiguous memory axis"
    out = np.fft.fft(mv, axis=mem_axis)
    mv[...] = out


def test_distributed_fft_matches_fftn():
    dims, pdims = (16, 12, 10), (2, 2)
    topo = Topology(pdims)
    p1, p2, p3 = fft_pencils(topo, dims)
    nr = topo.nranks

    rng = np.random.default_rng(99)
    g = (rng.standard_normal(dims) + 1j * rng.standard_normal(dims)
         ).astype(np.complex128)

    u1 = []
    for r in range(nr):
        parent = orc.parent_from_global(g, dims, pdims, (1, 2), (0, 1, 2), r)
        u1.append(PencilArray(p1, r, parent.astype(np.complex128)))
    u2 = [PencilArray.empty(p2, r, dtype=np.complex128) for r in range(nr)]
    u3 = [PencilArray.empty(p3, r, dtype=np.complex128) for r in range(nr)]

    for r in range(nr):
        local_fft_along(u1[r], 0)       # FFT along x on the x-pencil
    run_transpose_sim(u2, u1)
    for r in range(nr):
        local_fft_along(u2[r], 1)       # FFT along y on the y-pencil
    run_transpose_sim(u3, u2)
    for r in range(nr):
        local_fft_along(u3[r], 2)       # FFT along z on the z-pencil

    got = gather_sim(u3)
    want = np.fft.fftn(g)
    assert np.allclose(got, want, rtol=1e-10, atol=1e-8)


def test_distributed_fft_inplace_many_pencil():
    """Same, fully in place over one ManyPencilArray buffer per rank (the
    PencilFFTs in-place plan layout, multiarrays.jl)."""
    dims, pdims = (16, 12, 10), (2, 2)
    topo = Topology(pdims)
    pens = fft_pencils(topo, dims)
    nr = topo.nranks

    rng = np.random.default_rng(7)
    g = (rng.standard_normal(dims) + 1j * rng.standard_normal(dims)
         ).astype(np.complex128)

    ms = [ManyPencilArray(pens, r, dtype=np.complex128) for r in range(nr)]
    for r in range(nr):
        parent = orc.parent_from_global(g, dims, pdims, (1, 2), (0, 1, 2), r)
        ms[r].first.data[:parent.size] = parent

    for r in range(nr):
        local_fft_along(ms[r][0], 0)
    run_transpose_sim([m[1] for m in ms], [m[0] for m in ms])
    for r in range(nr):
        local_fft_along(ms[r][1], 1)
    run_transpose_sim([m[2] for m in ms], [m[1] for m in ms])
    for r in range(nr):
        local_fft_along(ms[r][2], 2)

    got = gather_sim([m[2] for m in ms])
    assert np.allclose(got, np.fft.fftn(g), rtol=1e-10, atol=1e-8)
