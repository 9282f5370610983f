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
