"""MPIIODriver-format I/O (§8(f) row 4): write/read round trips in both
layouts, cross-decomposition reads of discontiguous datasets, metadata
fields (mirrors test/io.jl's write+read recipe)."""

import json
import math
import os

import numpy as np
import pytest

import oracle as orc
from pencilarrays_amd import Pencil, PencilArray, Topology
from pencilarrays_amd.pencilio import MPIIOFile
from util import seeded_parents


def _arrays(dims, pdims, decomp, perm, extra, dtype):
    topo = Topology(pdims)
    pen = Pencil(topo, dims, decomp, permute=perm)
    g, parents = seeded_parents(dims, pdims, decomp, perm, extra, dtype)
    xs = [PencilArray(pen, r, parents[r].copy(), extra)
          for r in range(topo.nranks)]
    return g, pen, xs


def _write_all(path, name, xs, chunks):
    """Single-process simulation of the per-rank collective write: one
    MPIIOFile per rank (only rank 0 truncates / writes metadata), rank 0's
    write first so the file is sized before the other windows land."""
    fs = [MPIIOFile(path, "w", rank=x.rank) for x in xs]
    for f, x in zip(fs, xs):
        f.write(name, x, chunks=chunks)
    for f in fs:
        f.close()
    return fs[0]


@pytest.mark.parametrize("perm", [(0, 1, 2), (1, 2, 0)],
                         ids=["noperm", "perm120"])
@pytest.mark.parametrize("chunks", [False, True], ids=["discontig", "chunks"])
def test_roundtrip_same_decomposition(tmp_path, perm, chunks):
    dims, pdims = (16, 21, 41), (2, 2)
    g, pen, xs = _arrays(dims, pdims, (1, 2), perm, (), np.float64)
    path = str(tmp_path / "data.bin")
    _write_all(path, "u", xs, chunks)

    f = MPIIOFile(path, "r", rank=0)
    assert "u" in f.meta["datasets"]
    for r in range(len(xs)):
        y = PencilArray.empty(pen, r)
        fr = MPIIOFile(path, "r", rank=r)
        fr.read("u", y)
        assert np.array_equal(y.data, xs[r].data), f"rank {r}"


def test_discontiguous_cross_decomposition_read(tmp_path):
    """The discontiguous layout is decomposition-independent
    (mpi_io.jl:183-186): write on a 2x2 grid, read the same file on 1x1 and
    on 4x1 — every reader sees the same global array."""
    dims = (16, 21, 41)
    g, pen, xs = _arrays(dims, (2, 2), (1, 2), (1, 2, 0), (), np.float64)
    import tempfile
    path = str(tmp_path / "x.bin")
    _write_all(path, "u", xs, chunks=False)

    # read on a single rank, identity permutation
    topo1 = Topology((1, 1))
    pen1 = Pencil(topo1, dims, (1, 2), permute=(1, 2, 0))
    y = PencilArray.empty(pen1, 0)
    MPIIOFile(path, "r", rank=0).read("u", y)
    got = orc.global_from_parents([y.data], dims, (1, 1), (1, 2), (1, 2, 0))
    assert np.array_equal(got, g)

    # read on a 4x1 grid
    topo4 = Topology((4, 1))
    pen4 = Pencil(topo4, dims, (1, 2), permute=(1, 2, 0))
    outs = []
    for r in range(4):
        z = PencilArray.empty(pen4, r)
        MPIIOFile(path, "r", rank=r).read("u", z)
        outs.append(z.data)
    got4 = orc.global_from_parents(outs, dims, (4, 1), (1, 2), (1, 2, 0))
    assert np.array_equal(got4, g)


def test_file_bytes_discontiguous_layout(tmp_path):
    """The raw file IS the global array in memory order, column-major —
    byte-level format pin."""
    dims = (8, 6, 5)
    perm = (1, 2, 0)
    g, pen, xs = _arrays(dims, (2, 2), (1, 2), perm, (), np.float64)
    path = str(tmp_path / "y.bin")
    _write_all(path, "u", xs, chunks=False)
    raw = np.fromfile(path, dtype=np.float64)
    gmem = np.transpose(g, perm)  # memory order
    assert np.array_equal(raw, np.asfortranarray(gmem).ravel(order="F"))


def test_metadata_fields(tmp_path):
    dims = (8, 6, 5)
    g, pen, xs = _arrays(dims, (2, 2), (1, 2), (1, 2, 0), (), np.complex64)
    path = str(tmp_path / "m.bin")
    _write_all(path, "vec", xs, chunks=False)
    meta = json.load(open(path + ".json"))
    assert meta["driver"]["type"] == "MPIIODriver"
    d = meta["datasets"]["vec"]
    assert d["element_type"] == "ComplexF32"
    assert d["permutation"] == [2, 3, 1]         # 1-based, Julia convention
    assert d["decomposed_dims"] == [2, 3]
    assert d["dims_logical"] == [8, 6, 5]
    assert d["dims_memory"] == [6, 5, 8]
    assert d["offset_bytes"] == 0
    assert d["size_bytes"] == 8 * 6 * 5 * 8
    assert d["chunks"] is False
    assert d["julia_endian_bom"] == "0x04030201"


def test_two_datasets_appended(tmp_path):
    dims = (8, 6, 5)
    g, pen, xs = _arrays(dims, (1, 1), (1, 2), (0, 1, 2), (), np.float64)
    path = str(tmp_path / "two.bin")
    f = MPIIOFile(path, "w", rank=0)
    f.write("a", xs[0], chunks=False)
    f.write("b", xs[0], chunks=True)
    f.close()
    meta = json.load(open(path + ".json"))
    assert meta["datasets"]["b"]["offset_bytes"] == \
        meta["datasets"]["a"]["size_bytes"]
    y = PencilArray.empty(pen, 0)
    MPIIOFile(path, "r", rank=0).read("b", y)
    assert np.array_equal(y.data, xs[0].data)
    # metadata-less read of dataset a
    z = PencilArray.empty(pen, 0)
    MPIIOFile(path, "r", rank=0).read_raw(z, offset=0)
    assert np.array_equal(z.data, xs[0].data)


def test_read_type_mismatch_raises(tmp_path):
    dims = (8, 6, 5)
    g, pen, xs = _arrays(dims, (1, 1), (1, 2), (0, 1, 2), (), np.float64)
    path = str(tmp_path / "t.bin")
    f = MPIIOFile(path, "w", rank=0)
    f.write("u", xs[0])
    f.close()
    bad = PencilArray.empty(pen, 0, dtype="float32")
    with pytest.raises(TypeError):
        MPIIOFile(path, "r", rank=0).read("u", bad)


def test_read_size_bytes_mismatch_raises(tmp_path):
    """check_metadata's size_bytes assert (mpi_io.jl:306-307): corrupted
    metadata must be rejected, not read past."""
    dims = (8, 6, 5)
    g, pen, xs = _arrays(dims, (1, 1), (1, 2), (0, 1, 2), (), np.float64)
    path = str(tmp_path / "t.bin")
    f = MPIIOFile(path, "w", rank=0)
    f.write("u", xs[0])
    f.close()
    meta = json.load(open(path + ".json"))
    meta["datasets"]["u"]["size_bytes"] += 8
    json.dump(meta, open(path + ".json", "w"))
    y = PencilArray.empty(pen, 0)
    with pytest.raises(ValueError, match="size_bytes"):
        MPIIOFile(path, "r", rank=0).read("u", y)


def test_write_unsupported_dtype_clear_error(tmp_path):
    """float16 now maps to Float16; a truly unmapped dtype gets a clear
    TypeError, not a KeyError."""
    from pencilarrays_amd.pencilio import _julia_type
    assert _julia_type(np.dtype(np.float16)) == "Float16"
    assert _julia_type(np.dtype(np.int8)) == "Int8"
    with pytest.raises(TypeError, match="unsupported element type"):
        _julia_type(np.dtype([("a", np.int32)]))  # structured dtype
