"""GPU parity tests (single MI355X): every kernel path of the native engine
against the CPU oracle, bit-exact.

- end-to-end pa_transpose_execute on world=1 configs (the fused local path,
  with and without permutations, complex dtypes, extra dims, odd sizes);
- every pack/unpack/local descriptor of multi-rank plans executed on device
  via pa_device_copy vs the CPU executor (covers the exact N>1 kernels
  without needing N GPUs);
- a full multi-rank transpose simulated on ONE GPU (device-to-device copies
  standing in for RCCL) vs the oracle;
- RCCL single-rank communicator bootstrap.

All tests prefill destinations with a sentinel and compare ENTIRE buffers, so
out-of-window writes are caught, not just wrong values inside the window.
"""

import math

import numpy as np
import pytest

import oracle as orc
from pencilarrays_amd import (
    Pencil, PencilArray, Topology, Transposition, build_plan,
)
from pencilarrays_amd.copyexec import apply_copy
from util import SWEEP, seeded_parents

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs a ROCm GPU", allow_module_level=True)

from pencilarrays_amd import native  # noqa: E402

WORLD1 = [
    # (dims, din, pin, dout, pout, extra, dtype)
    ((42, 31, 29), (1, 2), (0, 1, 2), (0, 2), (0, 1, 2), (), np.float64),
    ((42, 31, 29), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.float64),
    ((64, 64, 64), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.float64),
    ((64, 64, 64), (1, 2), (0, 1, 2), (0, 2), (2, 0, 1), (), np.float64),
    ((257, 129, 65), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.float64),
    ((42, 31, 29), (1, 2), (2, 1, 0), (0, 2), (1, 2, 0), (), np.float64),
    ((42, 31, 29), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.complex64),
    ((42, 31, 29), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.complex128),
    ((42, 31, 29), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.float32),
    ((42, 31, 29), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.float16),
    ((24, 18, 12), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (3,), np.float64),
    ((24, 18, 12), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (4, 3), np.float64),
    ((1, 7, 5), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0), (), np.float64),
    # 5-D (deep descriptors on device)
    ((5, 6, 4, 7, 3), (1, 3), (0, 1, 2, 3, 4), (0, 3), (4, 0, 2, 1, 3), (),
     np.float64),
]

_T_DTYPE = {
    np.dtype(np.float16): torch.float16,
    np.dtype(np.float64): torch.float64,
    np.dtype(np.float32): torch.float32,
    np.dtype(np.complex64): torch.complex64,
    np.dtype(np.complex128): torch.complex128,
}


def _to_gpu(a: np.ndarray):
    return torch.from_numpy(np.ascontiguousarray(a)).to("cuda:0")


def _sentinel_like(n, dtype):
    t = torch.empty(n, dtype=_T_DTYPE[np.dtype(dtype)], device="cuda:0")
    t.view(torch.uint8).fill_(0xAB)
    return t


@pytest.mark.parametrize("cfg", WORLD1,
                         ids=lambda c: f"{c[0]}_{c[3]}p{c[4]}_{np.dtype(c[6]).name}")
def test_world1_execute_vs_oracle(cfg):
    dims, di, pi, do, po, extra, dtype = cfg
    topo = Topology((1, 1))
    Pi = Pencil(topo, dims, di, permute=pi)
    Po = Pencil(topo, dims, do, permute=po)
    g, parents = seeded_parents(dims, (1, 1), di, pi, extra, dtype)
    src = PencilArray(Pi, 0, _to_gpu(parents[0]), extra)
    n_out = Po.length_local(0) * math.prod(extra or (1,))
    dst = PencilArray(Po, 0, _sentinel_like(n_out, dtype), extra)
    Transposition(dst, src).execute()
    torch.cuda.synchronize()
    exp = orc.transpose_oracle(parents, dims, (1, 1), di, pi, do, po, extra)[0]
    got = dst.data.cpu().numpy()
    assert np.array_equal(got, exp)


MULTI = [c for c in SWEEP if math.prod(c[1]) > 1][:10]


@pytest.mark.parametrize("cfg", MULTI,
                         ids=lambda c: f"{c[0]}x{c[1]}_{c[2]}to{c[4]}")
def test_device_descs_match_cpu_executor(cfg):
    """Runs every plan descriptor on the GPU via pa_device_copy and compares
    buffers (with sentinels) against the CPU executor."""
    dims, pdims, di, pi, do, po, extra, dtype = cfg
    esz = np.dtype(dtype).itemsize
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, di, permute=pi)
    Po = Pencil(topo, dims, do, permute=po)
    g, parents = seeded_parents(dims, pdims, di, pi, extra, dtype)
    lib = native.load()
    import ctypes
    I64 = ctypes.c_int64

    def dev_copy(desc, src_t, dst_t):
        nd = len(desc.dims)
        st = lib.pa_device_copy(
            nd, (I64 * nd)(*desc.dims), (I64 * nd)(*desc.sstrides),
            I64(desc.soffset), (I64 * nd)(*desc.dstrides), I64(desc.doffset),
            I64(esz), ctypes.c_void_p(src_t.data_ptr()),
            ctypes.c_void_p(dst_t.data_ptr()), None)
        assert st == 0, lib.pa_last_error().decode()

    for rank in range(topo.nranks):
        plan = build_plan(Pi, Po, rank, extra)
        src_np = parents[rank]
        src_t = _to_gpu(src_np)
        descs = []
        if plan.local is not None:
            descs.append(("local", plan.local,
                          Po.length_local(rank) * math.prod(extra or (1,))))
        for blk in plan.peers:
            if blk.pack is not None:
                descs.append((f"pack{blk.peer_k}", blk.pack,
                              plan.send_nelem_total))
            if blk.unpack is not None:
                # unpack reads from a contiguous recv buffer; synthesise it
                # with the CPU pack of the matching peer block
                descs.append((f"unpack{blk.peer_k}", blk.unpack,
                              Po.length_local(rank) * math.prod(extra or (1,))))

        for name, desc, out_n in descs:
            if name.startswith("unpack"):
                src_flat = np.zeros(plan.recv_nelem_total, dtype=dtype)
                rng = np.random.default_rng(1234)
                src_flat[:] = rng.standard_normal(src_flat.shape).astype(dtype)
                src_dev = _to_gpu(src_flat)
            else:
                src_flat = src_np
                src_dev = src_t
            exp = np.empty(out_n, dtype=dtype)
            exp.view(np.uint8)[:] = 0xAB
            got_t = _sentinel_like(out_n, dtype)
            apply_copy(desc, src_flat, exp)
            dev_copy(desc, src_dev, got_t)
            torch.cuda.synchronize()
            got = got_t.cpu().numpy()
            assert np.array_equal(got, exp), f"rank {rank} {name}"


@pytest.mark.parametrize("cfg", MULTI[:6],
                         ids=lambda c: f"{c[0]}x{c[1]}_{c[2]}to{c[4]}")
def test_multirank_sim_on_one_gpu(cfg):
    """Full multi-rank transpose on one GPU: pack/local/unpack with native
    kernels, device-to-device tensor copies standing in for RCCL."""
    dims, pdims, di, pi, do, po, extra, dtype = cfg
    esz = np.dtype(dtype).itemsize
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, di, permute=pi)
    Po = Pencil(topo, dims, do, permute=po)
    g, parents = seeded_parents(dims, pdims, di, pi, extra, dtype)
    nr = topo.nranks
    import ctypes
    I64 = ctypes.c_int64
    lib = native.load()

    def dev_copy(desc, src_t, dst_t):
        nd = len(desc.dims)
        st = lib.pa_device_copy(
            nd, (I64 * nd)(*desc.dims), (I64 * nd)(*desc.sstrides),
            I64(desc.soffset), (I64 * nd)(*desc.dstrides), I64(desc.doffset),
            I64(esz), ctypes.c_void_p(src_t.data_ptr()),
            ctypes.c_void_p(dst_t.data_ptr()), None)
        assert st == 0, lib.pa_last_error().decode()

    plans = [build_plan(Pi, Po, r, extra) for r in range(nr)]
    srcs = [_to_gpu(parents[r]) for r in range(nr)]
    pex = math.prod(extra or (1,))
    dsts = [_sentinel_like(Po.length_local(r) * pex, dtype) for r in range(nr)]
    sends = [_sentinel_like(max(p.send_nelem_total, 1), dtype) for p in plans]
    recvs = [_sentinel_like(max(p.recv_nelem_total, 1), dtype) for p in plans]

    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.pack is not None:
                dev_copy(blk.pack, srcs[r], sends[r])
        if p.local is not None:
            dev_copy(p.local, srcs[r], dsts[r])
    torch.cuda.synchronize()
    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.peer_k == p.my_k or blk.send_nelem == 0:
                continue
            q = plans[blk.global_rank]
            rblk = q.peers[p.my_k]
            recvs[blk.global_rank][rblk.recv_offset:
                                   rblk.recv_offset + rblk.recv_nelem] = \
                sends[r][blk.send_offset:blk.send_offset + blk.send_nelem]
    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.unpack is not None:
                dev_copy(blk.unpack, recvs[r], dsts[r])
    torch.cuda.synchronize()

    exp = orc.transpose_oracle(parents, dims, pdims, di, pi, do, po, extra)
    for r in range(nr):
        assert np.array_equal(dsts[r].cpu().numpy(), exp[r]), f"rank {r}"


def test_inplace_world1_on_gpu():
    """Aliased (in-place) transpose through pa_transpose_execute: staged
    self path on device, bit-exact vs oracle (Transpositions.jl:250-264)."""
    dims = (64, 48, 40)
    topo = Topology((1, 1))
    p1 = Pencil(topo, dims, (1, 2))
    p2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    n = max(p1.length_local(0), p2.length_local(0))
    buf = torch.empty(n, dtype=torch.float64, device="cuda:0")
    rng = np.random.default_rng(11)
    src_np = rng.standard_normal(p1.length_local(0))
    buf[:p1.length_local(0)] = torch.from_numpy(src_np).cuda()
    src = PencilArray(p1, 0, buf[:p1.length_local(0)])
    dst = PencilArray(p2, 0, buf[:p2.length_local(0)])
    t = Transposition(dst, src)
    assert t.aliased
    t.execute()
    torch.cuda.synchronize()
    exp = orc.transpose_oracle([src_np], dims, (1, 1), (1, 2), (0, 1, 2),
                               (0, 2), (1, 2, 0), ())[0]
    assert np.array_equal(dst.data.cpu().numpy(), exp)
    # and back, in place again
    back_exp = src_np
    t2 = Transposition(src, dst)
    assert t2.aliased
    t2.execute()
    torch.cuda.synchronize()
    assert np.array_equal(src.data.cpu().numpy(), back_exp)


def test_multirank_aliased_sim_on_one_gpu():
    """Distributed IN-PLACE transpose simulated on one GPU: one shared
    buffer per rank (src and dst alias), staged self blocks + packs +
    exchange (device copies) + unpacks, all with native kernels, vs the
    oracle."""
    import ctypes
    dims, pdims = (16, 21, 41), (2, 2)
    dtype = np.float64
    esz = 8
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, (1, 2))
    Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    g, parents = seeded_parents(dims, pdims, (1, 2), (0, 1, 2), (), dtype)
    nr = topo.nranks
    lib = native.load()
    I64 = ctypes.c_int64

    def dev_copy(desc, src_t, dst_t):
        nd = len(desc.dims)
        st = lib.pa_device_copy(
            nd, (I64 * nd)(*desc.dims), (I64 * nd)(*desc.sstrides),
            I64(desc.soffset), (I64 * nd)(*desc.dstrides), I64(desc.doffset),
            I64(esz), ctypes.c_void_p(src_t.data_ptr()),
            ctypes.c_void_p(dst_t.data_ptr()), None)
        assert st == 0, lib.pa_last_error().decode()

    plans = [build_plan(Pi, Po, r, aliased=True) for r in range(nr)]
    bufs = []
    for r in range(nr):
        n = max(Pi.length_local(r), Po.length_local(r))
        b = torch.empty(n, dtype=torch.float64, device="cuda:0")
        b[:Pi.length_local(r)] = torch.from_numpy(parents[r]).cuda()
        bufs.append(b)
    sends = [_sentinel_like(max(p.send_nelem_total, 1), dtype) for p in plans]
    recvs = [_sentinel_like(max(p.recv_nelem_total, 1), dtype) for p in plans]

    # 1. packs (remote + staged self), all reads of src before any dst write
    for r, p in enumerate(plans):
        assert p.local is None and p.self_pack is not None
        for blk in p.peers:
            if blk.pack is not None:
                dev_copy(blk.pack, bufs[r], sends[r])
        dev_copy(p.self_pack, bufs[r], recvs[r])
    torch.cuda.synchronize()
    # 2. exchange
    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.peer_k == p.my_k or blk.send_nelem == 0:
                continue
            q = plans[blk.global_rank]
            rblk = q.peers[p.my_k]
            recvs[blk.global_rank][rblk.recv_offset:
                                   rblk.recv_offset + rblk.recv_nelem] = \
                sends[r][blk.send_offset:blk.send_offset + blk.send_nelem]
    # 3. unpacks (incl. staged self) into the SAME buffers
    for r, p in enumerate(plans):
        dev_copy(p.self_unpack, recvs[r], bufs[r])
        for blk in p.peers:
            if blk.unpack is not None:
                dev_copy(blk.unpack, recvs[r], bufs[r])
    torch.cuda.synchronize()

    exp = orc.transpose_oracle(parents, dims, pdims, (1, 2), (0, 1, 2),
                               (0, 2), (1, 2, 0), ())
    for r in range(nr):
        got = bufs[r][:Po.length_local(r)].cpu().numpy()
        assert np.array_equal(got, exp[r]), f"rank {r}"


def test_randomized_descriptor_sweep():
    """Property sweep: 15 seeded-random configurations (dims, grid, decomps,
    perms, dtype) — every plan descriptor executed on device vs the CPU
    executor, whole-buffer sentinel compare."""
    import ctypes
    import itertools
    rng = np.random.default_rng(20260915)
    lib = native.load()
    I64 = ctypes.c_int64
    from pencilarrays_amd.plan import build_plan as bp

    def dev_copy(desc, esz, src_t, dst_t):
        nd = len(desc.dims)
        st = lib.pa_device_copy(
            nd, (I64 * nd)(*desc.dims), (I64 * nd)(*desc.sstrides),
            I64(desc.soffset), (I64 * nd)(*desc.dstrides), I64(desc.doffset),
            I64(esz), ctypes.c_void_p(src_t.data_ptr()),
            ctypes.c_void_p(dst_t.data_ptr()), None)
        assert st == 0, lib.pa_last_error().decode()

    for trial in range(15):
        nd = int(rng.integers(2, 5))
        dims = tuple(int(rng.integers(1, 24)) for _ in range(nd))
        m = int(rng.integers(1, min(nd, 2) + 1))
        pdims = tuple(int(rng.integers(1, 4)) for _ in range(m))
        all_dims = list(range(nd))
        di = tuple(rng.permutation(all_dims)[:m].tolist())
        do = list(di)
        avail = [d for d in all_dims if d not in di]
        if avail:  # change one decomposed dim; else same-decomposition case
            do[int(rng.integers(0, m))] = int(rng.permutation(avail)[0])
        do = tuple(do)
        pi = tuple(rng.permutation(nd).tolist())
        po = tuple(rng.permutation(nd).tolist())
        dtype = [np.float64, np.float32, np.complex64][trial % 3]
        esz = np.dtype(dtype).itemsize

        topo = Topology(pdims)
        Pi = Pencil(topo, dims, di, permute=pi)
        Po = Pencil(topo, dims, do, permute=po)
        g, parents = seeded_parents(dims, pdims, di, pi, (), dtype,
                                    seed=1000 + trial)
        from pencilarrays_amd.copyexec import apply_copy
        for rank in range(topo.nranks):
            plan = bp(Pi, Po, rank)
            descs = []
            pex = 1
            if plan.local is not None:
                descs.append((plan.local, parents[rank],
                              Po.length_local(rank)))
            for blk in plan.peers:
                if blk.pack is not None:
                    descs.append((blk.pack, parents[rank],
                                  plan.send_nelem_total))
                if blk.unpack is not None:
                    rbuf = rng.standard_normal(
                        plan.recv_nelem_total).astype(dtype)
                    descs.append((blk.unpack, rbuf, Po.length_local(rank)))
            for desc, src_np, out_n in descs:
                if out_n == 0 or desc.nelem == 0:
                    continue
                exp = np.empty(out_n, dtype=dtype)
                exp.view(np.uint8)[:] = 0xAB
                got_t = _sentinel_like(out_n, dtype)
                src_t = _to_gpu(np.ascontiguousarray(src_np))
                apply_copy(desc, np.ascontiguousarray(src_np), exp)
                dev_copy(desc, esz, src_t, got_t)
                torch.cuda.synchronize()
                assert np.array_equal(got_t.cpu().numpy(), exp), \
                    f"trial {trial} rank {rank} {desc}"


def test_fullsize_1024_roundtrip_properties():
    """Size-independent properties at the BASELINE full size (1024^3 f64,
    world=1, PencilFFTs-permuted): x->y->x round trip restores the input
    bit-exactly, and the order-independent modular checksum of bit patterns
    is conserved at every stage (pure data movement)."""
    dims = (1024, 1024, 1024)
    topo = Topology((1, 1))
    p1 = Pencil(topo, dims, (1, 2))
    p2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    n = p1.length_local(0)
    g = torch.Generator(device="cuda:0").manual_seed(0xC0FFEE)
    u1 = PencilArray(p1, 0, torch.randn(n, generator=g, dtype=torch.float64,
                                        device="cuda:0"))
    u2 = PencilArray(p2, 0, torch.empty(n, dtype=torch.float64,
                                        device="cuda:0"))
    back = PencilArray(p1, 0, torch.empty(n, dtype=torch.float64,
                                          device="cuda:0"))

    def cks(t):
        return int(t.view(torch.int64).sum().item())

    c0 = cks(u1.data)
    Transposition(u2, u1).execute()
    torch.cuda.synchronize()
    assert cks(u2.data) == c0
    Transposition(back, u2).execute()
    torch.cuda.synchronize()
    assert torch.equal(back.data, u1.data)  # bit-exact round trip
    del u1, u2, back
    torch.cuda.empty_cache()


def test_odd_element_sizes_via_byteify():
    """Element sizes outside {4,8,16} (the reference allows arbitrary isbits
    types) route through the byte-ified fallback: a 2-byte and a 6-byte
    'element' permuted copy vs the CPU executor."""
    import ctypes
    from pencilarrays_amd.plan import CopyDesc, normalize_desc
    from pencilarrays_amd.copyexec import apply_copy
    lib = native.load()
    I64 = ctypes.c_int64
    rng = np.random.default_rng(77)
    for esz, di, dj in [(2, 37, 53), (6, 23, 31)]:
        # transpose-shaped desc: src contiguous axis0, dst contiguous axis1
        desc = normalize_desc(CopyDesc(
            dims=(di, dj), sstrides=(1, di), soffset=0,
            dstrides=(dj, 1), doffset=0))
        n = di * dj
        src = rng.integers(0, 255, n * esz, dtype=np.uint8)
        exp = np.full(n * esz, 0xAB, dtype=np.uint8)
        # CPU: byte-ified application
        bdesc = normalize_desc(CopyDesc(
            dims=(esz,) + desc.dims, sstrides=(1,) + tuple(
                s * esz for s in desc.sstrides), soffset=desc.soffset * esz,
            dstrides=(1,) + tuple(s * esz for s in desc.dstrides),
            doffset=desc.doffset * esz))
        apply_copy(bdesc, src, exp)
        src_t = torch.from_numpy(src).cuda()
        got_t = torch.full((n * esz,), 0xAB, dtype=torch.uint8,
                           device="cuda:0")
        nd = len(desc.dims)
        st = lib.pa_device_copy(
            nd, (I64 * nd)(*desc.dims), (I64 * nd)(*desc.sstrides),
            I64(desc.soffset), (I64 * nd)(*desc.dstrides), I64(desc.doffset),
            I64(esz), ctypes.c_void_p(src_t.data_ptr()),
            ctypes.c_void_p(got_t.data_ptr()), None)
        assert st == 0, lib.pa_last_error().decode()
        torch.cuda.synchronize()
        assert np.array_equal(got_t.cpu().numpy(), exp), f"esz={esz}"


def test_launch_beyond_2e32_workitems():
    """Regression: the HSA dispatch limit is 2^32-1 work-items per grid
    dimension; launches above it must split across gridDim.y (a 2048^3 f64
    copy is exactly 2^32 16-B words and used to fail).  Cheap variant: a
    byte-element copy of 2^32+13 bytes (odd, so no word-scaling) needs
    ~4.3e9 threads."""
    import ctypes
    lib = native.load()
    I64 = ctypes.c_int64
    n = (1 << 32) + 13
    src = torch.empty(n, dtype=torch.uint8, device="cuda:0")
    src[:256].fill_(7)
    src[-256:].fill_(9)
    dst = torch.zeros(n, dtype=torch.uint8, device="cuda:0")
    st = lib.pa_device_copy(
        1, (I64 * 1)(n), (I64 * 1)(1), I64(0), (I64 * 1)(1), I64(0),
        I64(1), ctypes.c_void_p(src.data_ptr()),
        ctypes.c_void_p(dst.data_ptr()), None)
    assert st == 0, lib.pa_last_error().decode()
    torch.cuda.synchronize()
    assert torch.equal(dst[:256].cpu(), src[:256].cpu())
    assert torch.equal(dst[-256:].cpu(), src[-256:].cpu())
    assert int(dst[n // 2].item()) == int(src[n // 2].item())
    del src, dst
    torch.cuda.empty_cache()


def test_repeated_execute_stress():
    """100 back-to-back async executes on one plan (alternating x->y / y->x
    on the same buffers, no intermediate syncs): stream/event ordering must
    hold; final state checked bit-exactly."""
    dims = (128, 96, 80)
    topo = Topology((1, 1))
    p1 = Pencil(topo, dims, (1, 2))
    p2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    n = p1.length_local(0)
    rng = np.random.default_rng(123)
    u1_np = rng.standard_normal(n)
    u1 = PencilArray(p1, 0, _to_gpu(u1_np))
    u2 = PencilArray(p2, 0, _sentinel_like(n, np.float64))
    t12 = Transposition(u2, u1)
    t21 = Transposition(u1, u2)
    for _ in range(50):
        t12.execute(sync=False)
        t21.execute(sync=False)
    torch.cuda.synchronize()
    assert np.array_equal(u1.data.cpu().numpy(), u1_np)
    exp2 = orc.transpose_oracle([u1_np], dims, (1, 1), (1, 2), (0, 1, 2),
                                (0, 2), (1, 2, 0), ())[0]
    t12.execute()
    torch.cuda.synchronize()
    assert np.array_equal(u2.data.cpu().numpy(), exp2)


def test_rccl_single_rank_bootstrap():
    lib = native.load()
    import ctypes
    n = lib.pa_unique_id_size()
    assert n >= 64
    buf = ctypes.create_string_buffer(n)
    assert lib.pa_get_unique_id(buf) == 0, lib.pa_last_error().decode()
    comm = native.NativeComm.create(bytes(buf.raw), 1, 0)
    assert comm.handle
    # pa_allreduce through RCCL (world=1: identity), f64 sum in place
    t = torch.arange(1024, dtype=torch.float64, device="cuda:0")
    st = lib.pa_allreduce(comm.handle, ctypes.c_void_p(t.data_ptr()),
                          ctypes.c_void_p(t.data_ptr()),
                          ctypes.c_int64(1024), 0, 0, None)
    assert st == 0, lib.pa_last_error().decode()
    torch.cuda.synchronize()
    assert torch.equal(t.cpu(), torch.arange(1024, dtype=torch.float64))


def test_fft_pattern_on_gpu():
    """PencilFFTs consumer pattern on device: per-axis torch.fft on the
    contiguous memory axis of each pencil + native transposes, vs
    torch.fft.fftn of the whole array (world=1, permuted pencils)."""
    from pencilarrays_amd.permutations import perm_inv
    dims = (64, 48, 40)
    topo = Topology((1, 1))
    p1 = Pencil(topo, dims, (1, 2))
    p2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    p3 = Pencil(topo, dims, (0, 1), permute=(2, 1, 0))

    rng = np.random.default_rng(5)
    g = (rng.standard_normal(dims) + 1j * rng.standard_normal(dims)
         ).astype(np.complex128)
    parent = np.asfortranarray(g).ravel(order="F")

    u1 = PencilArray(p1, 0, _to_gpu(parent))
    u2 = PencilArray(p2, 0, _sentinel_like(p2.length_local(0), np.complex128))
    u3 = PencilArray(p3, 0, _sentinel_like(p3.length_local(0), np.complex128))

    def fft_axis0(x):
        mv = x.parent_memview()  # torch view, axis 0 fastest
        assert perm_inv(x.pencil.perm)[fft_axis0.dim] == 0
        out = torch.fft.fft(mv, dim=0)
        mv.copy_(out)

    fft_axis0.dim = 0
    fft_axis0(u1)
    Transposition(u2, u1).execute()
    fft_axis0.dim = 1
    fft_axis0(u2)
    Transposition(u3, u2).execute()
    fft_axis0.dim = 2
    fft_axis0(u3)
    torch.cuda.synchronize()

    # u3 parent (mem order (2,1,0)) -> logical order
    got = u3.logical_view().cpu().numpy()
    want = np.fft.fftn(g)
    assert np.allclose(got, want, rtol=1e-10, atol=1e-7)


def test_roundtrip_on_gpu():
    """u1 -> u2 -> u1 world=1 with permuted pencils restores u1 bit-exactly
    (test/transpose.jl:48-60 recipe)."""
    dims = (96, 70, 33)
    topo = Topology((1, 1))
    p1 = Pencil(topo, dims, (1, 2))
    p2 = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    rng = np.random.default_rng(42)
    u1_np = rng.standard_normal(p1.length_local(0))
    u1 = PencilArray(p1, 0, _to_gpu(u1_np))
    u2 = PencilArray(p2, 0, _sentinel_like(p2.length_local(0), np.float64))
    Transposition(u2, u1).execute()
    back = PencilArray(p1, 0, _sentinel_like(p1.length_local(0), np.float64))
    Transposition(back, u2).execute()
    torch.cuda.synchronize()
    assert np.array_equal(back.data.cpu().numpy(), u1_np)
