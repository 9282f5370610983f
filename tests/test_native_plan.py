"""Native (C++) plan building vs the Python host mirror: the block tables and
normalized copy descriptors must be identical.  Host-only — runs without a
GPU (libpencilhip.so loads on CPU; only execution needs a device)."""

import os

import numpy as np
import pytest

from pencilarrays_amd import Pencil, Topology, build_plan
from util import SWEEP

native = pytest.importorskip("pencilarrays_amd.native")

if not os.path.exists(native.lib_path()):
    pytest.skip("libpencilhip.so not built", allow_module_level=True)


def _desc_tuple(d):
    if d is None:
        return None
    return (tuple(d.dims), tuple(d.sstrides), d.soffset,
            tuple(d.dstrides), d.doffset)


@pytest.mark.parametrize("cfg", SWEEP, ids=lambda c: f"{c[0]}x{c[1]}_{c[2]}to{c[4]}")
def test_native_plan_matches_python(cfg):
    dims, pdims, di, pi, do, po, extra, dtype = cfg
    esz = np.dtype(dtype).itemsize
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, di, permute=pi)
    Po = Pencil(topo, dims, do, permute=po)
    for rank in range(topo.nranks):
        py = build_plan(Pi, Po, rank, extra)
        nat = native.NativePlan(Pi, Po, rank, esz, extra)

        assert nat.r_dim == (-1 if py.r_dim is None else py.r_dim)
        assert nat.nproc_sub == py.nproc_sub
        assert nat.my_k == py.my_k

        sb, rb = nat.buffer_sizes()
        assert sb == py.send_nelem_total * esz
        assert rb == py.recv_nelem_total * esz

        assert _desc_tuple(py.local) == nat.copydesc(0)

        if py.r_dim is not None:
            for blk in py.peers:
                info = nat.block_info(blk.peer_k)
                assert info[0] == blk.peer_k
                assert info[1] == blk.global_rank
                assert info[4] == blk.send_nelem
                assert info[5] == blk.recv_nelem
                if blk.peer_k != py.my_k:
                    assert info[2] == blk.send_offset
                    assert info[3] == blk.recv_offset
                assert bool(info[6]) == (blk.pack is not None)
                assert bool(info[7]) == (blk.unpack is not None)
                assert _desc_tuple(blk.pack) == nat.copydesc(1, blk.peer_k)
                assert _desc_tuple(blk.unpack) == nat.copydesc(2, blk.peer_k)


@pytest.mark.parametrize("cfg", SWEEP[:8], ids=lambda c: f"{c[0]}x{c[1]}")
def test_native_aliased_plan_matches_python(cfg):
    dims, pdims, di, pi, do, po, extra, dtype = cfg
    esz = np.dtype(dtype).itemsize
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, di, permute=pi)
    Po = Pencil(topo, dims, do, permute=po)
    for rank in range(topo.nranks):
        py = build_plan(Pi, Po, rank, extra, aliased=True)
        nat = native.NativePlan(Pi, Po, rank, esz, extra, aliased=True)
        sb, rb = nat.buffer_sizes()
        assert sb == py.send_nelem_total * esz
        assert rb == py.recv_nelem_total * esz
        assert nat.copydesc(0) is None  # no fused local in aliased mode
        assert _desc_tuple(py.self_pack) == nat.copydesc(3)
        assert _desc_tuple(py.self_unpack) == nat.copydesc(4)
        if py.r_dim is not None:
            for blk in py.peers:
                assert _desc_tuple(blk.pack) == nat.copydesc(1, blk.peer_k)
                assert _desc_tuple(blk.unpack) == nat.copydesc(2, blk.peer_k)


def test_plan_create_rejects_two_hop():
    topo = Topology((2, 2))
    Pi = Pencil(topo, (16, 21, 41), (1, 2))
    Po = Pencil(topo, (16, 21, 41), (0, 1))
    with pytest.raises(RuntimeError):
        native.NativePlan(Pi, Po, 0, 8)


def test_abi_exports_complete():
    """Every symbol declared in include/pencilhip.h is exported."""
    import ctypes
    import re
    lib = native.load()
    header = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "include", "pencilhip.h")
    text = open(header).read()
    syms = re.findall(r"^(?:pa_status|void|int|int64_t|const char \*)\s*"
                      r"(pa_\w+)\s*\(", text, re.M)
    assert len(syms) >= 20
    for s in set(syms):
        assert hasattr(lib, s), f"missing export {s}"


def test_plan_create_error_strings_mirror_reference():
    """assert_compatible's error messages (Transpositions.jl:182-199) are
    kept verbatim so a host porting error-handling code sees the same
    text."""
    topo = Topology((2, 2))
    Pi = Pencil(topo, (16, 21, 41), (1, 2))
    # two-hop decomposition change
    Po2 = Pencil(topo, (16, 21, 41), (0, 1))
    with pytest.raises(RuntimeError,
                       match="differ in at most one dimension"):
        native.NativePlan(Pi, Po2, 0, 8)
    # different global size
    Po3 = Pencil(topo, (16, 21, 40), (0, 2))
    with pytest.raises(RuntimeError,
                       match="global data sizes must be the same"):
        native.NativePlan(Pi, Po3, 0, 8)
    # different topology
    topo2 = Topology((4, 1))
    Po4 = Pencil(topo2, (16, 21, 41), (0, 2))
    with pytest.raises(RuntimeError,
                       match="topologies must be the same"):
        native.NativePlan(Pi, Po4, 0, 8)


def test_pencil_create_rejects_bad_args():
    import ctypes
    lib = native.load()
    I64, I32, VP = ctypes.c_int64, ctypes.c_int32, ctypes.c_void_p
    topo = VP()
    assert lib.pa_topology_create(2, (I64 * 2)(2, 2),
                                  ctypes.byref(topo)) == 0
    out = VP()
    # repeated decomp dim (_check_selected_dimensions, Pencils.jl:393-406)
    st = lib.pa_pencil_create(topo, 3, (I64 * 3)(8, 8, 8),
                              (I32 * 2)(1, 1), None, ctypes.byref(out))
    assert st != 0 and b"repeated" in lib.pa_last_error()
    # decomp dim out of range
    st = lib.pa_pencil_create(topo, 3, (I64 * 3)(8, 8, 8),
                              (I32 * 2)(1, 5), None, ctypes.byref(out))
    assert st != 0 and b"out of range" in lib.pa_last_error()
    # invalid permutation (not a bijection)
    st = lib.pa_pencil_create(topo, 3, (I64 * 3)(8, 8, 8),
                              (I32 * 2)(1, 2), (I32 * 3)(0, 0, 2),
                              ctypes.byref(out))
    assert st != 0 and b"invalid permutation" in lib.pa_last_error()
    # M > N
    st = lib.pa_pencil_create(topo, 1, (I64 * 1)(8),
                              (I32 * 2)(0, 0), None, ctypes.byref(out))
    assert st != 0
    lib.pa_topology_destroy(topo)


def test_plan_create_rejects_bad_rank_and_esz():
    topo = Topology((2, 2))
    Pi = Pencil(topo, (16, 21, 41), (1, 2))
    Po = Pencil(topo, (16, 21, 41), (0, 2))
    with pytest.raises(RuntimeError, match="rank out of range"):
        native.NativePlan(Pi, Po, 7, 8)
    with pytest.raises(RuntimeError, match="elem_size"):
        native.NativePlan(Pi, Po, 0, 0)
