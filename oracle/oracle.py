"""CPU oracle (direct-spec form) for the pencil transpose — TEST INFRASTRUCTURE.

ORACLE ONLY: nothing under oracle/ may be imported by the product path.  Only
tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may use it.

This file restates WHAT transpose! computes, independently of the package's
plan/copy machinery (which restates HOW the reference computes it).  The spec,
from the reference:

  For pencils Pi, Po over the same topology and global size
  (Transpositions.jl:182-199), transpose!(dest, src) makes
  ``gather(dest) == gather(src)`` hold (test/transpose.jl:6-22): every global
  logical element keeps its value; only ownership (which rank holds it) and
  layout (the memory-order permutation of the parent array) change.

  - ownership: rank with topology coords c owns, along each decomposed
    logical dim d = decomp_dims[j], the 0-based half-open range
    [N*c[j]//P[j], N*(c[j]+1)//P[j])  (data_ranges.jl:4-9), full range on
    non-decomposed dims (data_ranges.jl:15-45).
  - layout: the parent holds the local logical block with axes gathered by
    the pencil's permutation, axis 0 fastest (column-major — Julia parent,
    arrays.jl:134-138: dims = size_local(pencil, MemoryOrder())).

So: dest_parent(rank) = global[owned region in logical order], axes
transposed to Po memory order, flattened Fortran-order.

Pinning: validated in tests against (a) an analytic input
u[i0,i1,...] = Σ i_d · Π_{d'<d} N_{d'} whose placement is closed-form
(BASELINE.md), (b) the package's independently-written simulation path,
(c) the line-faithful C restatement (oracle/ref_impl.c) which follows the
reference's pack→exchange→unpack flow.  The reference itself ships no golden
files — its pins are procedural (round-trip identity test/transpose.jl:60 and
gather-equality :6-22), all reproduced in tests/.
"""

from __future__ import annotations

import math
from typing import List, Sequence, Tuple

import numpy as np

Region = Tuple[Tuple[int, int], ...]


def _split(c: int, P: int, N: int) -> Tuple[int, int]:
    # data_ranges.jl:4-9 (0-based half-open)
    return (N * c) // P, (N * (c + 1)) // P


def _coords(rank: int, pdims: Sequence[int]) -> Tuple[int, ...]:
    # MPI_Cart_create(reorder=false) row-major rank order
    cs = []
    rem = rank
    for i in range(len(pdims)):
        stride = math.prod(pdims[i + 1:])
        cs.append(rem // stride)
        rem %= stride
    return tuple(cs)


def owned_region(size_global: Sequence[int], pdims: Sequence[int],
                 decomp_dims: Sequence[int], rank: int) -> Region:
    cs = _coords(rank, pdims)
    region = [(0, s) for s in size_global]
    for j, d in enumerate(decomp_dims):
        region[d] = _split(cs[j], pdims[j], size_global[d])
    return tuple(region)


def parent_from_global(global_arr: np.ndarray, size_global, pdims,
                       decomp_dims, perm, rank,
                       extra_dims: Tuple[int, ...] = ()) -> np.ndarray:
    """Expected parent flat buffer (1-D) of the PencilArray on ``rank``."""
    region = owned_region(size_global, pdims, decomp_dims, rank)
    n = len(size_global)
    sl = tuple(slice(lo, hi) for lo, hi in region) + \
        tuple(slice(None) for _ in extra_dims)
    block = global_arr[sl]                       # logical order
    order = tuple(perm) + tuple(range(n, n + len(extra_dims)))
    block_mem = np.transpose(block, order)       # memory order
    return np.asfortranarray(block_mem).ravel(order="F")


def global_from_parents(parents: Sequence[np.ndarray], size_global, pdims,
                        decomp_dims, perm,
                        extra_dims: Tuple[int, ...] = ()) -> np.ndarray:
    """Inverse: assemble the global logical array from per-rank parent flats
    (the gather recipe, gather.jl:59-95)."""
    n = len(size_global)
    shape = tuple(size_global) + tuple(extra_dims)
    out = np.empty(shape, dtype=parents[0].dtype)
    nranks = math.prod(pdims)
    inv = [0] * n
    for i, v in enumerate(perm):
        inv[v] = i
    order = tuple(inv) + tuple(range(n, n + len(extra_dims)))
    for rank in range(nranks):
        region = owned_region(size_global, pdims, decomp_dims, rank)
        mem_dims = tuple(region[d][1] - region[d][0] for d in perm) + \
            tuple(extra_dims)
        blk_mem = parents[rank].reshape(mem_dims, order="F")
        blk = np.transpose(blk_mem, order)       # back to logical order
        sl = tuple(slice(lo, hi) for lo, hi in region) + \
            tuple(slice(None) for _ in extra_dims)
        out[sl] = blk
    return out


def transpose_oracle(src_parents: Sequence[np.ndarray], size_global, pdims,
                     decomp_in, perm_in, decomp_out, perm_out,
                     extra_dims: Tuple[int, ...] = ()) -> List[np.ndarray]:
    """All ranks' expected dest parents for transpose!(Po ← Pi)."""
    g = global_from_parents(src_parents, size_global, pdims, decomp_in,
                            perm_in, extra_dims)
    nranks = math.prod(pdims)
    return [
        parent_from_global(g, size_global, pdims, decomp_out, perm_out, r,
                           extra_dims)
        for r in range(nranks)
    ]


def analytic_global(size_global, extra_dims: Tuple[int, ...] = (),
                    dtype=np.float64) -> np.ndarray:
    """u[i0,i1,...] = linear index (i0 fastest) — closed-form placement
    checkable input (BASELINE.md 'analytic linear-index pattern')."""
    shape = tuple(size_global) + tuple(extra_dims)
    n = math.prod(shape)
    return np.arange(n, dtype=dtype).reshape(shape, order="F")
