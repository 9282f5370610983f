"""Pencil decomposition metadata.

Restates the metadata half of src/Pencils/ (Pencils.jl, data_ranges.jl):

- the block-distribution split formula ``local_data_range``
  (data_ranges.jl:4-9): 1-based ``(N*(p-1))÷P+1 : (N*p)÷P`` becomes the
  half-open 0-based ``[(N*p)//P, (N*(p+1))//P)``.
- ``complete_dims`` / ``generate_axes_matrix`` (data_ranges.jl:15-45): the
  per-process axis ranges ``axes_all``.
- ``axes_local`` / ``axes_local_perm`` (Pencils.jl:221-236).
- ``to_local`` (Pencils.jl:579-587).
- compatibility rules for transposes (Transpositions.jl:182-199).

Ranges are half-open 0-based ``(lo, hi)`` int pairs; dims and permutations are
0-based.  A "region" is an N-tuple of ranges in logical dimension order.
"""

from __future__ import annotations

import math
from typing import Optional, Sequence, Tuple

from .permutations import (
    Perm,
    check_perm,
    identity_perm,
    perm_apply,
)
from .topology import Topology

Range = Tuple[int, int]          # half-open [lo, hi)
Region = Tuple[Range, ...]       # one range per logical dimension


def local_data_range(p: int, P: int, N: int) -> Range:
    """data_ranges.jl:4-9, 0-based half-open."""
    assert 0 <= p < P
    return (N * p) // P, (N * (p + 1)) // P


def complete_dims(n: int, dims: Sequence[int], vals: Sequence[int],
                  fill: int = 0) -> Tuple[int, ...]:
    """data_ranges.jl:15-26 with 0-based dims; positions not in ``dims`` get
    ``fill`` (the reference fills 1 = a single "process"/coordinate; with
    0-based coordinates the filler coordinate is 0 and the filler process
    count is 1 — pass ``fill`` accordingly)."""
    out = [fill] * n
    for d, v in zip(dims, vals):
        out[d] = v
    return tuple(out)


def range_intersect(a: Range, b: Range) -> Range:
    lo, hi = max(a[0], b[0]), min(a[1], b[1])
    return (lo, max(lo, hi))  # empty ranges normalise to hi == lo


def region_intersect(a: Region, b: Region) -> Region:
    return tuple(range_intersect(x, y) for x, y in zip(a, b))


def region_lengths(r: Region) -> Tuple[int, ...]:
    return tuple(hi - lo for lo, hi in r)


def region_nelem(r: Region) -> int:
    return math.prod(region_lengths(r))


class Pencil:
    """Decomposition of an N-d array over an M-d process grid (M ≤ N).

    Mirrors Pencil{N,M} (Pencils.jl:151-251).  Unlike the reference, the
    staging buffers live in the native engine's plan, not here; this object is
    pure metadata and is cheap to share.
    """

    def __init__(
        self,
        topology: Topology,
        size_global: Sequence[int],
        decomp_dims: Optional[Sequence[int]] = None,
        permute: Optional[Sequence[int]] = None,
    ):
        n = len(size_global)
        m = topology.ndims
        if decomp_dims is None:
            # default: the M rightmost dimensions (Pencils.jl:387-390)
            decomp_dims = tuple(range(n - m, n))
        decomp_dims = tuple(int(d) for d in decomp_dims)
        # _check_selected_dimensions (Pencils.jl:393-406)
        if m > n:
            raise ValueError(
                f"number of decomposed dimensions M={m} cannot exceed N={n}")
        if len(decomp_dims) != m:
            raise ValueError("decomp_dims length must match topology ndims")
        if len(set(decomp_dims)) != m:
            raise ValueError(f"dimensions may not be repeated. Got {decomp_dims}.")
        if not all(0 <= d < n for d in decomp_dims):
            raise ValueError(f"dimensions must be in 0:{n-1}. Got {decomp_dims}.")

        self.topology = topology
        self.size_global: Tuple[int, ...] = tuple(int(s) for s in size_global)
        self.decomp_dims: Tuple[int, ...] = decomp_dims
        self.perm: Perm = (identity_perm(n) if permute is None
                           else check_perm(permute))
        self.ndims = n

    @classmethod
    def from_nprocs(cls, size_global: Sequence[int], nprocs: int,
                    decomp_dims: Optional[Sequence[int]] = None,
                    permute: Optional[Sequence[int]] = None) -> "Pencil":
        """``Pencil(size_global, [decomp_dims,] comm)`` (Pencils.jl:106-116):
        the convenience constructor that creates the process topology
        implicitly — `MPI.Dims_create` over the decomposed dimensions
        (default: dimensions 1..N-1 0-based, i.e. the reference's 2:N)."""
        n = len(size_global)
        if decomp_dims is None:
            decomp_dims = tuple(range(1, n))
        m = len(decomp_dims)
        from .topology import dims_create
        topo = Topology(dims_create(nprocs, m))
        return cls(topo, size_global, decomp_dims, permute=permute)

    # ---- axes ----------------------------------------------------------

    def axes_for_coords(self, coords: Sequence[int]) -> Region:
        """Region owned by the process at topology coords (data_ranges.jl:39-41)."""
        n = self.ndims
        procs = complete_dims(n, self.decomp_dims, self.topology.dims, fill=1)
        cs = complete_dims(n, self.decomp_dims, coords, fill=0)
        return tuple(
            local_data_range(cs[d], procs[d], self.size_global[d])
            for d in range(n)
        )

    def axes_for_rank(self, rank: int) -> Region:
        return self.axes_for_coords(self.topology.cart_coords(rank))

    def range_local(self, rank: int, memory_order: bool = False) -> Region:
        r = self.axes_for_rank(rank)
        return perm_apply(self.perm, r) if memory_order else r

    def range_remote(self, rank_or_coords, memory_order: bool = False) -> Region:
        """range_remote(p, coords|rank, order) (Pencils.jl:529-536)."""
        if isinstance(rank_or_coords, tuple):
            r = self.axes_for_coords(rank_or_coords)
        else:
            r = self.axes_for_rank(int(rank_or_coords))
        return perm_apply(self.perm, r) if memory_order else r

    def size_local(self, rank: int, memory_order: bool = False) -> Tuple[int, ...]:
        return region_lengths(self.range_local(rank, memory_order))

    def length_local(self, rank: int) -> int:
        return math.prod(self.size_local(rank))

    def to_local(self, rank: int, region: Region,
                 memory_order: bool = False) -> Region:
        """Global→local index ranges (Pencils.jl:579-587), 0-based: local =
        global − first(axes_local)."""
        axes_local = self.axes_for_rank(rank)
        out = tuple(
            (g[0] - al[0], g[1] - al[0])
            for g, al in zip(region, axes_local)
        )
        return perm_apply(self.perm, out) if memory_order else out

    # ---- compatibility (Transpositions.jl:182-199, :111) ---------------

    def check_compatible(self, other: "Pencil") -> None:
        if self.topology != other.topology:
            raise ValueError("pencil topologies must be the same.")
        if self.size_global != other.size_global:
            raise ValueError(
                f"global data sizes must be the same between different pencil "
                f"configurations. Got {self.size_global} != {other.size_global}.")
        ndiff = sum(a != b for a, b in
                    zip(self.decomp_dims, other.decomp_dims))
        if ndiff > 1:
            raise ValueError(
                f"pencil decompositions must differ in at most one dimension. "
                f"Got decomposed dimensions {self.decomp_dims} and "
                f"{other.decomp_dims}.")

    def transpose_dim(self, other: "Pencil") -> Optional[int]:
        """Index R of the single differing decomposed dimension
        (Transpositions.jl:111), or None if decompositions are identical."""
        self.check_compatible(other)
        for j, (a, b) in enumerate(zip(self.decomp_dims, other.decomp_dims)):
            if a != b:
                return j
        return None

    # ---- misc ----------------------------------------------------------

    def with_(self, decomp_dims=None, permute=None) -> "Pencil":
        """Derived pencil (Pencils.jl:257-271): same topology & global size."""
        return Pencil(
            self.topology, self.size_global,
            decomp_dims=self.decomp_dims if decomp_dims is None else decomp_dims,
            permute=self.perm if permute is None else permute,
        )

    def __repr__(self):
        return (f"Pencil(size_global={self.size_global}, "
                f"decomp_dims={self.decomp_dims}, perm={self.perm}, "
                f"grid={self.topology.dims})")
