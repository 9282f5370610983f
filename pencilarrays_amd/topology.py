"""Cartesian process-grid topology (the reference's MPITopology).

Restates src/Pencils/MPITopologies.jl for a single node of GPUs, with no MPI:

- ``MPITopology(comm, pdims)`` builds an MPI Cartesian communicator with
  ``reorder = false`` (MPITopologies.jl:125-131), so the rank↔coordinate map is
  MPI's fixed row-major ordering: rank = c[0]*prod(dims[1:]) + ... + c[M-1]
  (last coordinate fastest).  ``get_cart_ranks`` (:208-226) materialises that
  map; we compute it in closed form.
- Per-dimension 1-D sub-communicators (``create_subcomms``, :244-251) keep the
  coordinate order, so the rank of a process *within* subgroup R equals its
  Cartesian coordinate along R (``get_cart_ranks_subcomm``, :229-242).
- ``dims_create`` (:138-144) mirrors MPI_Dims_create: factor nproc into M
  dims, as balanced as possible, non-increasing.

All coordinates/dims here are 0-based Python tuples.
"""

from __future__ import annotations

import math
from typing import List, Tuple


def dims_create(nproc: int, m: int) -> Tuple[int, ...]:
    """MPI_Dims_create(nproc, m): balanced factorisation, non-increasing.

    (MPITopologies.jl:138-144 defers to MPI; this is the MPICH algorithm's
    result for the sizes we use: most-balanced factorisation, sorted
    descending.)
    """
    if m <= 0:
        raise ValueError("m must be positive")
    best: List[int] = [nproc] + [1] * (m - 1)

    def search(remaining: int, slots: int, cap: int, acc: List[int]):
        nonlocal best
        if slots == 1:
            if remaining <= cap:
                cand = acc + [remaining]  # non-increasing by construction
                if (max(cand), cand) < (max(best), best):
                    best = cand
            return
        f = min(remaining, cap)
        while f >= 1:
            if remaining % f == 0:
                search(remaining // f, slots - 1, f, acc + [f])
            f -= 1

    search(nproc, m, nproc, [])
    return tuple(best)


class Topology:
    """An M-dimensional Cartesian process grid (one process per GPU).

    Mirrors MPITopology{M} (MPITopologies.jl:72-119): ``dims``,
    ``coords_local`` per rank, rank maps, and per-dimension subgroups.
    """

    def __init__(self, pdims: Tuple[int, ...]):
        if any(d <= 0 for d in pdims):
            raise ValueError(f"invalid process grid {pdims}")
        self.dims: Tuple[int, ...] = tuple(int(d) for d in pdims)
        self.ndims = len(self.dims)
        self.nranks = math.prod(self.dims)
        # row-major strides (MPI_Cart_create reorder=false rank order:
        # last coordinate fastest)
        strides = [1] * self.ndims
        for i in range(self.ndims - 2, -1, -1):
            strides[i] = strides[i + 1] * self.dims[i + 1]
        self._strides = tuple(strides)

    def cart_rank(self, coords: Tuple[int, ...]) -> int:
        """MPI.Cart_rank (MPITopologies.jl:220-222), 0-based row-major."""
        assert len(coords) == self.ndims
        r = 0
        for c, d, s in zip(coords, self.dims, self._strides):
            if not (0 <= c < d):
                raise ValueError(f"coordinate {coords} outside grid {self.dims}")
            r += c * s
        return r

    def cart_coords(self, rank: int) -> Tuple[int, ...]:
        if not (0 <= rank < self.nranks):
            raise ValueError(f"rank {rank} outside topology of {self.nranks}")
        return tuple((rank // s) % d for s, d in zip(self._strides, self.dims))

    def subgroup_ranks(self, rank: int, r_dim: int) -> List[int]:
        """Global ranks of the 1-D subgroup through ``rank`` along topology
        dimension ``r_dim``, ordered by coordinate (= subgroup rank order,
        MPITopologies.jl:229-242)."""
        coords = list(self.cart_coords(rank))
        out = []
        for k in range(self.dims[r_dim]):
            coords[r_dim] = k
            out.append(self.cart_rank(tuple(coords)))
        return out

    def __eq__(self, other):
        return isinstance(other, Topology) and self.dims == other.dims

    def __repr__(self):
        s = "×".join(map(str, self.dims))
        return f"Topology({s} processes)"
