"""Pencil / topology metadata: split formula, axes coverage, to_local,
compatibility rules."""

import math

import pytest

from pencilarrays_amd import Pencil, Topology, dims_create
from pencilarrays_amd.pencil import local_data_range


def test_split_formula_matches_reference():
    # data_ranges.jl:4-9: 1-based (N(p-1))÷P+1 : (Np)÷P.  0-based half-open
    # [N*p//P, N*(p+1)//P).  Check equality and exact coverage.
    for N in [1, 2, 7, 16, 21, 41, 1000]:
        for P in [1, 2, 3, 5, 8, 41, 50]:
            prev_hi = 0
            for p in range(P):
                lo, hi = local_data_range(p, P, N)
                # julia: a = (N*(p1-1))÷P + 1 with p1 = p+1 -> a-1 == lo
                a = (N * p) // P + 1
                b = (N * (p + 1)) // P
                assert lo == a - 1 and hi == b
                assert lo == prev_hi  # contiguous, disjoint
                prev_hi = hi
            assert prev_hi == N  # full coverage


def test_axes_cover_global():
    topo = Topology((2, 3))
    pen = Pencil(topo, (16, 21, 41), (1, 2))
    seen = set()
    for r in range(topo.nranks):
        region = pen.axes_for_rank(r)
        for i0 in range(*region[0]):
            for i1 in range(*region[1]):
                for i2 in range(*region[2]):
                    key = (i0, i1, i2)
                    assert key not in seen
                    seen.add(key)
    assert len(seen) == 16 * 21 * 41


def test_default_decomposition():
    # Pencils.jl:387-390: the M rightmost dimensions
    topo = Topology((2, 2))
    pen = Pencil(topo, (4, 8, 12))
    assert pen.decomp_dims == (1, 2)


def test_to_local():
    topo = Topology((2, 2))
    pen = Pencil(topo, (16, 21, 41), (1, 2), permute=(1, 2, 0))
    r = 3
    region = pen.axes_for_rank(r)
    loc = pen.to_local(r, region)
    assert all(lo == 0 for lo, _ in loc)
    assert tuple(hi - lo for lo, hi in loc) == pen.size_local(r)
    locm = pen.to_local(r, region, memory_order=True)
    assert tuple(hi - lo for lo, hi in locm) == pen.size_local(r, True)


def test_compatibility_rules():
    topo = Topology((2, 2))
    p1 = Pencil(topo, (16, 21, 41), (1, 2))
    p3 = Pencil(topo, (16, 21, 41), (0, 1))
    # (1,2) vs (0,1): two differences -> the reference throws
    # (Transpositions.jl:182-199; test/transpose.jl:45)
    with pytest.raises(ValueError):
        p1.transpose_dim(p3)
    p2 = Pencil(topo, (16, 21, 41), (0, 2))
    assert p1.transpose_dim(p2) == 0
    assert p2.transpose_dim(p3) == 1
    assert p1.transpose_dim(p1.with_(permute=(2, 1, 0))) is None
    # different global size
    q = Pencil(topo, (16, 21, 40), (1, 2))
    with pytest.raises(ValueError):
        p1.transpose_dim(q)


def test_invalid_construction():
    topo = Topology((2, 2))
    with pytest.raises(ValueError):
        Pencil(topo, (16, 21, 41), (1, 1))  # repeated dims
    with pytest.raises(ValueError):
        Pencil(topo, (16, 21, 41), (1, 3))  # out of range
    with pytest.raises(ValueError):
        Pencil(topo, (16,), (0, 1))  # M > N via wrong length


def test_topology_rank_maps():
    topo = Topology((2, 4))
    assert topo.nranks == 8
    # MPI row-major rank order (reorder=false): last coordinate fastest
    assert topo.cart_rank((0, 0)) == 0
    assert topo.cart_rank((0, 3)) == 3
    assert topo.cart_rank((1, 0)) == 4
    for r in range(8):
        assert topo.cart_rank(topo.cart_coords(r)) == r
    assert topo.subgroup_ranks(5, 0) == [1, 5]      # vary dim 0
    assert topo.subgroup_ranks(5, 1) == [4, 5, 6, 7]  # vary dim 1


def test_dims_create():
    assert dims_create(8, 2) == (4, 2)
    assert dims_create(4, 2) == (2, 2)
    assert dims_create(2, 2) == (2, 1)
    assert dims_create(1, 2) == (1, 1)
    assert dims_create(6, 2) == (3, 2)
    assert dims_create(12, 3) == (3, 2, 2)
    assert dims_create(7, 2) == (7, 1)
    for n, m in [(8, 2), (12, 3), (5, 2)]:
        assert math.prod(dims_create(n, m)) == n


def test_from_nprocs_convenience():
    """Pencil(dims, comm) convenience form (Pencils.jl:106-116): implicit
    topology via Dims_create over the default decomp dims 2:N (0-based
    1..N-1), mirroring the doctest `Pencil((4, 8, 12), MPI.COMM_WORLD)` ->
    decomposed dimensions (2, 3)."""
    from pencilarrays_amd import Pencil
    p = Pencil.from_nprocs((4, 8, 12), 4)
    assert p.decomp_dims == (1, 2)          # Julia (2, 3)
    assert tuple(p.topology.dims) == (2, 2)  # Dims_create(4, 2)
    assert p.topology.nranks == 4
    # explicit single decomposed dim: Pencil((4, 8, 12), (1,), comm)
    p2 = Pencil.from_nprocs((4, 8, 12), 3, decomp_dims=(0,))
    assert p2.decomp_dims == (0,)
    assert tuple(p2.topology.dims) == (3,)
