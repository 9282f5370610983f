"""Shared test helpers: ctypes wrapper for the C oracle, config sweep lists,
and input generators."""

from __future__ import annotations

import ctypes
import math
import os
import sys
from typing import List, Sequence, Tuple

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "oracle"))
import oracle as orc  # noqa: E402

I64 = ctypes.c_int64
I32 = ctypes.c_int32

# The reference's transpose-test sweep, 0-based (test/transpose.jl:24-74,
# test/pencils.jl:459-542): (dims, pdims, decomp_in, perm_in, decomp_out,
# perm_out, extra_dims, dtype).
ID3 = (0, 1, 2)
SWEEP = [
    # test/transpose.jl pencils 1->2->3 chain with the reference permutations
    ((16, 21, 41), (2, 2), (1, 2), ID3, (0, 2), (1, 2, 0), (), np.float64),
    ((16, 21, 41), (2, 2), (0, 2), (1, 2, 0), (0, 1), (2, 1, 0), (), np.float64),
    # without permutations (transpose.jl:63-67)
    ((16, 21, 41), (2, 2), (1, 2), ID3, (0, 2), ID3, (), np.float64),
    # unsorted decomp dims, #57 (transpose.jl:70-74)
    ((16, 21, 41), (2, 2), (1, 2), ID3, (1, 0), ID3, (), np.float64),
    # uneven grid
    ((16, 21, 41), (2, 3), (1, 2), ID3, (0, 2), (1, 2, 0), (), np.float64),
    # world = 1 (config 1 class)
    ((42, 31, 29), (1, 1), (1, 2), ID3, (0, 2), ID3, (), np.float64),
    # same decomposition: plain copy / pure local permutation
    # (pencils.jl:483-520 class)
    ((16, 21, 41), (2, 2), (1, 2), ID3, (1, 2), (2, 0, 1), (), np.float64),
    ((16, 21, 41), (2, 2), (1, 2), ID3, (1, 2), ID3, (), np.float64),
    # extra dims (pencils.jl:459-480)
    ((8, 9, 10), (2, 2), (1, 2), ID3, (0, 2), (1, 2, 0), (3,), np.float64),
    ((8, 9, 10), (2, 2), (1, 2), ID3, (0, 2), (1, 2, 0), (4, 3), np.float64),
    # ComplexF32 (pencils.jl:523-542 uses ComplexF32)
    ((16, 21, 41), (2, 2), (1, 2), ID3, (0, 2), (1, 2, 0), (), np.complex64),
    # slab / 1-D decomposition
    ((16, 21, 41), (3,), (1,), ID3, (0,), (1, 2, 0), (), np.float64),
    ((16, 21, 41), (3,), (1,), ID3, (2,), ID3, (), np.float64),
    # empty ranks: P > N along a decomposed dim
    ((3, 21, 41), (4, 2), (0, 2), ID3, (1, 2), ID3, (), np.float64),
    # decomposition including dim 0, ComplexF64
    ((5, 4, 41), (4, 2), (0, 1), ID3, (2, 1), (2, 1, 0), (), np.complex128),
    # 2-D data
    ((17, 23), (2, 2), (0, 1), (0, 1), (0, 1), (1, 0), (), np.float64),
    # 3-D (all-dims) decomposition, ComplexF32: only the permutation can
    # change (pencils.jl:522-542)
    ((16, 21, 41), (2, 2, 1), (0, 1, 2), ID3, (0, 1, 2), (1, 2, 0), (),
     np.complex64),
    ((12, 10, 8), (2, 1, 2), (0, 1, 2), (1, 2, 0), (0, 1, 2), (2, 1, 0), (),
     np.complex64),
    # 4-D data (the reference is N-dimensional; Pencil{N,M} any N)
    ((6, 7, 8, 9), (2, 2), (1, 3), (0, 1, 2, 3), (0, 3), (3, 0, 1, 2), (),
     np.float64),
    ((10, 9, 8, 7), (2, 3), (1, 2), (1, 0, 3, 2), (1, 3), (2, 3, 0, 1), (),
     np.float32),
    # 5-D data (deep descriptors: N=5 exercises MAXND-adjacent paths)
    ((5, 6, 4, 7, 3), (2, 2), (1, 3), (0, 1, 2, 3, 4), (0, 3),
     (4, 0, 2, 1, 3), (), np.float64),
    ((4, 5, 6, 3, 7), (2, 1), (2, 3), (2, 0, 4, 1, 3), (4, 3),
     (1, 3, 0, 4, 2), (), np.complex64),
]


def seeded_parents(dims, pdims, decomp, perm, extra, dtype, seed=0xC0FFEE):
    """Per-rank parent flats of a seeded random global array (values are
    random bits; only movement matters — BASELINE.md)."""
    rng = np.random.default_rng(seed)
    shape = tuple(dims) + tuple(extra)
    if np.issubdtype(np.dtype(dtype), np.complexfloating):
        g = (rng.standard_normal(shape) + 1j * rng.standard_normal(shape)
             ).astype(dtype)
    else:
        g = rng.standard_normal(shape).astype(dtype)
    nranks = math.prod(pdims)
    return g, [orc.parent_from_global(g, dims, pdims, decomp, perm, r, extra)
               for r in range(nranks)]


class COracle:
    def __init__(self, path: str):
        self.lib = ctypes.CDLL(path)
        self.lib.oracle_local_len.restype = ctypes.c_int64

    def local_len(self, dims, pdims, decomp, extra, rank) -> int:
        N, M, E = len(dims), len(pdims), len(extra)
        return self.lib.oracle_local_len(
            N, (I64 * N)(*dims), M, (I64 * M)(*pdims), (I32 * M)(*decomp),
            E, (I64 * E)(*extra) if E else None, rank)

    def transpose_all(self, src_parents: Sequence[np.ndarray], dims, pdims,
                      di, pi, do, po, extra, esz) -> List[np.ndarray]:
        N, M, E = len(dims), len(pdims), len(extra)
        nr = math.prod(pdims)
        dsts = [np.zeros(self.local_len(dims, pdims, do, extra, r) * esz,
                         dtype=np.uint8) for r in range(nr)]
        srcs_u8 = [np.ascontiguousarray(s).view(np.uint8) for s in src_parents]
        sp = (ctypes.c_void_p * nr)(*[s.ctypes.data for s in srcs_u8])
        dp = (ctypes.c_void_p * nr)(*[d.ctypes.data for d in dsts])
        self.lib.oracle_transpose_all(
            N, (I64 * N)(*dims), M, (I64 * M)(*pdims),
            (I32 * M)(*di), (I32 * N)(*pi), (I32 * M)(*do), (I32 * N)(*po),
            E, (I64 * E)(*extra) if E else None, I64(esz), sp, dp)
        return dsts
