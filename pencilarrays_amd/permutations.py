"""Permutation algebra for pencil decompositions.

Reimplements (0-based, tuples of ints) the four StaticPermutations.jl operators
the reference hot path uses.  The reference does not vendor
StaticPermutations.jl (PencilArrays Project.toml:20,46 pins v0.3); the operator
semantics below are derived from the reference's own usage and tests:

- ``perm * t``  gathers:      r[i] = t[perm[i]]          (arrays.jl:327-337,
  docstring example arrays.jl:19-31: parent dims = perm * logical dims)
- ``perm \\ t`` inverse-apply: r[perm[i]] = t[i]          (arrays.jl:116,205)
- ``p / q``    relative perm: (p/q) * (q*t) == p*t       (Transpositions.jl:506)
- ``append(p, E)`` extends with identity on E extra dims (Transpositions.jl:243,
  :602, :643-645)
- ``inv(p)``   inverse: inv(p)[p[i]] = i                 (gather.jl:37)
- ``isidentity``                                          (Transpositions.jl:638,:652)

Convention used throughout this package: a permutation ``q`` of length N maps
memory axes to logical dimensions — **q[i] is the logical dimension stored at
memory position i, with memory position 0 the fastest-varying axis** (the
reference's Julia parent array is column-major, so its first dimension is the
fastest; we keep that axis order and only flip to 0-based indices).
``identity_perm(N)`` plays the role of ``NoPermutation()``.
"""

from __future__ import annotations

from typing import Sequence, Tuple

Perm = Tuple[int, ...]


def identity_perm(n: int) -> Perm:
    return tuple(range(n))


def is_perm(p: Sequence[int]) -> bool:
    return sorted(p) == list(range(len(p)))


def check_perm(p: Sequence[int]) -> Perm:
    # Pencils.jl:382-385 (check_permutation)
    if not is_perm(p):
        raise ValueError(f"invalid permutation of dimensions: {p}")
    return tuple(p)


def is_identity(p: Sequence[int]) -> bool:
    return all(v == i for i, v in enumerate(p))


def perm_apply(p: Sequence[int], t: Sequence) -> tuple:
    """``p * t`` — gather: r[i] = t[p[i]]."""
    return tuple(t[i] for i in p)


def perm_inv(p: Sequence[int]) -> Perm:
    """``inv(p)``: inv(p)[p[i]] = i."""
    out = [0] * len(p)
    for i, v in enumerate(p):
        out[v] = i
    return tuple(out)


def perm_unapply(p: Sequence[int], t: Sequence) -> tuple:
    """``p \\ t`` — inverse apply: r[p[i]] = t[i]  ⇔  r = inv(p) * t."""
    return perm_apply(perm_inv(p), t)


def perm_relative(p: Sequence[int], q: Sequence[int]) -> Perm:
    """``p / q`` — the permutation r with r * (q * t) == p * t for all t.

    r[i] = inv(q)[p[i]].  Used at Transpositions.jl:506 as
    ``perm = permutation(Po) / permutation(Pi)``.
    """
    qi = perm_inv(q)
    return tuple(qi[v] for v in p)


def perm_append(p: Sequence[int], e: int) -> Perm:
    """``append(p, Val(E))`` — extend with identity on E extra (slowest) dims."""
    n = len(p)
    return tuple(p) + tuple(range(n, n + e))
