"""Shared staging pool (CPU-side logic): grow-only, versioned, one pair per
device — the engine analogue of the reference's send_buf/recv_buf shared
across derived pencils (Pencils.jl:187-189, 257-271)."""

import pytest

torch = pytest.importorskip("torch")

from pencilarrays_amd.native import StagingPool  # noqa: E402


def test_pool_grow_only_and_versioning():
    pool = StagingPool(torch.device("cpu"))
    s1, r1 = pool.reserve(100, 200)
    v1 = pool.version
    assert s1.numel() >= 100 and r1.numel() >= 200

    # smaller request: no growth, same tensors, same version
    s2, r2 = pool.reserve(50, 80)
    assert s2 is s1 and r2 is r1 and pool.version == v1

    # equal request: still no growth
    s3, r3 = pool.reserve(100, 200)
    assert s3 is s1 and r3 is r1 and pool.version == v1

    # larger send only: version bumps, recv untouched
    s4, r4 = pool.reserve(300, 100)
    assert s4.numel() >= 300 and r4 is r1 and pool.version == v1 + 1

    # larger recv only
    s5, r5 = pool.reserve(10, 500)
    assert s5 is s4 and r5.numel() >= 500 and pool.version == v1 + 2
