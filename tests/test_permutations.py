"""Permutation-operator semantics (pins StaticPermutations.jl v0.3 behaviour
as observed through the reference: arrays.jl:19-31, test/permutations.jl,
Transpositions.jl:506)."""

import itertools

import pytest

from pencilarrays_amd import (
    identity_perm, perm_append, perm_apply, perm_inv, perm_relative,
    perm_unapply,
)
from pencilarrays_amd.permutations import check_perm, is_identity


def test_gather_semantics():
    # Permutation(2,3,1) (Julia) == (1,2,0): parent dims = perm * logical dims
    # (arrays.jl:19-31 example: logical (10,20,30), perm (2,3,1) -> (20,30,10))
    assert perm_apply((1, 2, 0), (10, 20, 30)) == (20, 30, 10)


def test_inverse_and_unapply():
    for p in itertools.permutations(range(4)):
        t = (11, 22, 33, 44)
        assert perm_apply(perm_inv(p), perm_apply(p, t)) == t
        assert perm_unapply(p, perm_apply(p, t)) == t
        assert perm_apply(p, perm_unapply(p, t)) == t


def test_relative_perm_identity():
    # (p/q) * (q*t) == p*t for all p, q (Transpositions.jl:506 contract)
    t = tuple(range(100, 105))
    for p in itertools.permutations(range(5)):
        for q in [(1, 0, 2, 4, 3), (4, 3, 2, 1, 0), tuple(range(5))]:
            r = perm_relative(p, q)
            assert perm_apply(r, perm_apply(q, t)) == perm_apply(p, t)


def test_append():
    assert perm_append((1, 2, 0), 2) == (1, 2, 0, 3, 4)
    assert perm_append((), 3) == (0, 1, 2)


def test_identity_checks():
    assert is_identity(identity_perm(5))
    assert not is_identity((1, 0, 2))
    with pytest.raises(ValueError):
        check_perm((0, 0, 1))
    with pytest.raises(ValueError):
        check_perm((0, 2))
