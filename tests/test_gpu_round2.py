"""Round-2 GPU tests:

- shared staging pool: an x->y->z chain reuses ONE send/recv pair with zero
  steady-state allocation (the reference's buffer-sharing contract,
  Pencils.jl:257-271; its JLArray test asserts no reallocation,
  test/array_types.jl:118-127);
- per-stage HIP-event timing through the ABI (TimerOutputs analogue);
- chunked-exchange unpack descriptors executed ON DEVICE
  (PENCILHIP_EXCHANGE_CHUNKS path, chunk_of_raw) vs the full unpack;
- full-size (1024^3) analytic linear-index input: closed-form placement
  checked per sampled slab + exact whole-array checksum, so layout bugs
  cannot hide behind oracle-memory limits;
- reductions routed through the native engine's pa_allreduce.
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


def _to_gpu(a: np.ndarray):
    return torch.from_numpy(np.ascontiguousarray(a)).to("cuda:0")


# ---------------------------------------------------------------------------
# shared staging pool
# ---------------------------------------------------------------------------

def test_chain_shares_staging_and_no_steady_state_allocs():
    """x->y->z->y->x chain at world 1 (aliased staging via in-place mode is
    separate): all plans bind the same pool tensors; after warmup, repeated
    executes allocate nothing and the pool never grows."""
    dims = (96, 80, 72)
    topo = Topology((1, 1))
    Pi = Pencil(topo, dims, (1, 2))
    Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    Pz = Pencil(topo, dims, (0, 1), permute=(2, 1, 0))

    g, parents = seeded_parents(dims, (1, 1), (1, 2), (0, 1, 2), (),
                                np.float64)
    u1 = PencilArray(Pi, 0, _to_gpu(parents[0]))
    u2 = PencilArray(Po, 0, torch.empty(Po.length_local(0),
                                        dtype=torch.float64, device="cuda:0"))
    u3 = PencilArray(Pz, 0, torch.empty(Pz.length_local(0),
                                        dtype=torch.float64, device="cuda:0"))

    t12 = Transposition(u2, u1)
    t23 = Transposition(u3, u2)
    t12.execute()
    t23.execute()
    torch.cuda.synchronize()

    pool = native.staging_pool(torch.device("cuda", 0))
    # both natives bound to the same pool tensors
    assert t12._native._send.data_ptr() == t23._native._send.data_ptr()
    assert t12._native._recv.data_ptr() == t23._native._recv.data_ptr()

    v0 = pool.version
    a0 = torch.cuda.memory_allocated()
    for _ in range(10):
        t12.execute()
        t23.execute()
    torch.cuda.synchronize()
    assert pool.version == v0, "staging pool reallocated in steady state"
    assert torch.cuda.memory_allocated() == a0, \
        "steady-state executes allocated device memory"

    # results still bit-exact vs oracle after all the reuse
    exp2 = orc.transpose_oracle(parents, dims, (1, 1), (1, 2), (0, 1, 2),
                                (0, 2), (1, 2, 0), ())[0]
    assert np.array_equal(u2.data.cpu().numpy(), exp2)


def test_pool_growth_rebinds_existing_plans():
    """A later, larger plan grows the pool; an older plan must re-bind to
    the new tensors on its next execute (version check) and stay correct."""
    topo = Topology((1, 1))
    dims_small, dims_big = (24, 18, 12), (96, 80, 72)

    def mk(dims):
        Pi = Pencil(topo, dims, (1, 2))
        Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
        g, parents = seeded_parents(dims, (1, 1), (1, 2), (0, 1, 2), (),
                                    np.float64)
        src = PencilArray(Pi, 0, _to_gpu(parents[0]))
        dst = PencilArray(Po, 0, torch.empty(
            Po.length_local(0), dtype=torch.float64, device="cuda:0"))
        exp = orc.transpose_oracle(parents, dims, (1, 1), (1, 2), (0, 1, 2),
                                   (0, 2), (1, 2, 0), ())[0]
        return Transposition(dst, src), dst, exp

    # in-place small plan first (in-place => uses recv staging => pool)
    PiS = Pencil(topo, dims_small, (1, 2))
    PoS = Pencil(topo, dims_small, (0, 2), permute=(1, 2, 0))
    nel = max(PiS.length_local(0), PoS.length_local(0))
    buf = torch.empty(nel, dtype=torch.float64, device="cuda:0")
    g, parents = seeded_parents(dims_small, (1, 1), (1, 2), (0, 1, 2), (),
                                np.float64)
    buf[:PiS.length_local(0)].copy_(_to_gpu(parents[0]))
    srcS = PencilArray(PiS, 0, buf[:PiS.length_local(0)])
    dstS = PencilArray(PoS, 0, buf[:PoS.length_local(0)])
    tS = Transposition(dstS, srcS)  # aliased -> staged through recv pool
    assert tS.aliased

    tB, dstB, expB = mk(dims_big)  # aliased=False, but big in-place next:
    tS.execute()
    torch.cuda.synchronize()
    expS = orc.transpose_oracle(parents, dims_small, (1, 1), (1, 2),
                                (0, 1, 2), (0, 2), (1, 2, 0), ())[0]
    assert np.array_equal(dstS.data.cpu().numpy(), expS)

    # big IN-PLACE plan grows the pool's recv buffer
    nelB = max(Pencil(topo, dims_big, (1, 2)).length_local(0), 1)
    bufB = torch.empty(nelB, dtype=torch.float64, device="cuda:0")
    gB, parentsB = seeded_parents(dims_big, (1, 1), (1, 2), (0, 1, 2), (),
                                  np.float64)
    bufB.copy_(_to_gpu(parentsB[0]))
    PiB = Pencil(topo, dims_big, (1, 2))
    PoB = Pencil(topo, dims_big, (0, 2), permute=(1, 2, 0))
    srcB2 = PencilArray(PiB, 0, bufB)
    dstB2 = PencilArray(PoB, 0, bufB)
    tB2 = Transposition(dstB2, srcB2)
    assert tB2.aliased
    tB2.execute()
    torch.cuda.synchronize()
    expB2 = orc.transpose_oracle(parentsB, dims_big, (1, 1), (1, 2),
                                 (0, 1, 2), (0, 2), (1, 2, 0), ())[0]
    assert np.array_equal(dstB2.data.cpu().numpy(), expB2)

    # the small plan re-binds (pool grew) and still gives exact results
    buf[:PiS.length_local(0)].copy_(_to_gpu(parents[0]))
    tS.execute()
    torch.cuda.synchronize()
    assert np.array_equal(dstS.data.cpu().numpy(), expS)


# ---------------------------------------------------------------------------
# per-stage timing
# ---------------------------------------------------------------------------

def test_stage_times_world1():
    dims = (128, 96, 64)
    topo = Topology((1, 1))
    Pi = Pencil(topo, dims, (1, 2))
    Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    g, parents = seeded_parents(dims, (1, 1), (1, 2), (0, 1, 2), (),
                                np.float64)
    src = PencilArray(Pi, 0, _to_gpu(parents[0]))
    dst = PencilArray(Po, 0, torch.empty(
        Po.length_local(0), dtype=torch.float64, device="cuda:0"))
    t = Transposition(dst, src)
    t.execute()
    nat = t._native.native
    # not enabled yet -> error surfaced
    with pytest.raises(RuntimeError, match="timing not enabled"):
        nat.stage_times()
    nat.enable_timing(True)
    t.execute()
    torch.cuda.synchronize()
    st = nat.stage_times()
    # world-1: no exchange; the fused local copy is the whole step
    assert st["exchange"] is None
    assert st["local"] is not None and st["local"] > 0
    assert st["pack"] is not None and st["pack"] < 0.5  # no packs: ~0
    # and results are still exact with timing on
    exp = orc.transpose_oracle(parents, dims, (1, 1), (1, 2), (0, 1, 2),
                               (0, 2), (1, 2, 0), ())[0]
    assert np.array_equal(dst.data.cpu().numpy(), exp)


# ---------------------------------------------------------------------------
# chunked-exchange unpack on device
# ---------------------------------------------------------------------------

def _nat_plan(Pi, Po, rank, esz, extra=()):
    return native.NativePlan(Pi, Po, rank, esz, extra)


@pytest.mark.parametrize("chunks", [2, 3, 7])
def test_chunked_unpack_descriptors_on_device(chunks):
    """Every peer's chunk-of-raw unpack descriptors, executed on device via
    pa_device_copy, must reproduce the full unpack bit-exactly (sentinel
    whole-buffer compare).  This is the exact code path the chunked RCCL
    exchange uses for its overlapped unpacks."""
    import ctypes
    dims, pdims = (16, 21, 41), (2, 2)
    di, pi, do, po = (1, 2), (0, 1, 2), (0, 2), (1, 2, 0)
    dtype = np.float64
    esz = np.dtype(dtype).itemsize
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, di, permute=pi)
    Po = Pencil(topo, dims, do, permute=po)
    lib = native.load()

    for rank in range(math.prod(pdims)):
        nat = _nat_plan(Pi, Po, rank, esz)
        pplan = build_plan(Pi, Po, rank, ())
        rng = np.random.default_rng(1234 + rank)
        recv = rng.standard_normal(max(pplan.recv_nelem_total, 1)
                                   ).astype(dtype)
        n_out = Po.length_local(rank)
        recv_d = _to_gpu(recv)

        full = torch.empty(max(n_out, 1), dtype=torch.float64,
                           device="cuda:0")
        chunked = torch.empty_like(full)
        full.view(torch.uint8).fill_(0xCD)
        chunked.view(torch.uint8).fill_(0xCD)

        def run_desc(desc, dst_t):
            nd = len(desc[0])
            I64A = ctypes.c_int64 * nd
            st = lib.pa_device_copy(
                nd, I64A(*desc[0]), I64A(*desc[1]),
                ctypes.c_int64(desc[2]), I64A(*desc[3]),
                ctypes.c_int64(desc[4]), ctypes.c_int64(esz),
                ctypes.c_void_p(recv_d.data_ptr()),
                ctypes.c_void_p(dst_t.data_ptr()), None)
            assert st == 0, native.load().pa_last_error().decode()

        for k in range(nat.nproc_sub):
            d_full = nat.copydesc(2, k)
            if d_full is None:
                continue
            run_desc(d_full, full)
            raw = nat.copydesc(5, k)
            assert raw is not None
            outer = raw[0][-1]
            for c in range(chunks):
                lo = outer * c // chunks
                hi = outer * (c + 1) // chunks
                if hi <= lo:
                    continue
                dims_c = raw[0][:-1] + (hi - lo,)
                sub = (dims_c, raw[1], raw[2] + lo * raw[1][-1],
                       raw[3], raw[4] + lo * raw[3][-1])
                run_desc(sub, chunked)
        torch.cuda.synchronize()
        assert torch.equal(full.view(torch.uint8), chunked.view(torch.uint8))


# ---------------------------------------------------------------------------
# full-size analytic placement
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("po", [(1, 2, 0), (0, 1, 2)],
                         ids=["permuted", "identity"])
def test_analytic_placement_1024_full_size(po):
    """1024^3 x->y at world 1 with u[l0,l1,l2] = l0 + N0*l1 + N0*N1*l2 (the
    global linear index, int64 bits in an f64 array).  Placement is checked
    closed-form on sampled output slabs — expected values computed from
    index arithmetic alone, independent of the plan — plus the exact
    wrap-around checksum of the whole array."""
    N = (1024, 1024, 1024)
    n = math.prod(N)
    topo = Topology((1, 1))
    Pi = Pencil(topo, N, (1, 2))
    Po = Pencil(topo, N, (0, 2), permute=po)

    src_t = torch.empty(n, dtype=torch.int64, device="cuda:0")
    CH = 1 << 28
    for off in range(0, n, CH):
        m = min(CH, n - off)
        src_t[off:off + m] = torch.arange(off, off + m, dtype=torch.int64,
                                          device="cuda:0")
    src = PencilArray(Pi, 0, src_t.view(torch.float64))
    dst_t = torch.empty(n, dtype=torch.float64, device="cuda:0")
    dst = PencilArray(Po, 0, dst_t)
    Transposition(dst, src).execute()
    torch.cuda.synchronize()

    di = dst_t.view(torch.int64)
    # exact wrap-around checksum: sum over arange(n) mod 2^64
    total = (n * (n - 1) // 2) % (1 << 64)
    got = int(di.sum().item()) % (1 << 64)
    assert got == total

    # closed-form placement on sampled slabs: memory axes of dst are
    # md[a] = N[po[a]], column-major parent; element at memory coords
    # (m0,m1,m2) is logical l with l[po[a]] = m_a and must equal
    # l0 + N0*l1 + N0*N1*l2.
    md = tuple(N[p] for p in po)
    view = di.view(md[2], md[1], md[0])  # row-major view of col-major flat
    l = [None, None, None]
    m0 = torch.arange(md[0], dtype=torch.int64, device="cuda:0")
    m1 = torch.arange(md[1], dtype=torch.int64, device="cuda:0")
    for m2 in [0, 1, md[2] // 2, md[2] - 1, 777]:
        l[po[0]] = m0.view(1, md[0])
        l[po[1]] = m1.view(md[1], 1)
        l[po[2]] = torch.tensor(m2, dtype=torch.int64, device="cuda:0")
        expected = l[0] + N[0] * l[1] + N[0] * N[1] * l[2]
        assert torch.equal(view[m2], expected.expand(md[1], md[0])), \
            f"placement mismatch in slab m2={m2}"


# ---------------------------------------------------------------------------
# reductions through the native engine
# ---------------------------------------------------------------------------

def test_native_allreduce_tensor_world1():
    """allreduce_tensor via a single-rank RCCL comm: identity, all ops and
    dtypes the ABI maps."""
    uid = native._nccl_uid()
    comm = native.NativeComm.create(uid, 1, 0)
    for dt, val in [(torch.float64, 3.5), (torch.float32, -2.0),
                    (torch.int64, 7), (torch.int32, -9)]:
        t = torch.tensor([val, val * 2], dtype=dt, device="cuda:0")
        for op in ("sum", "prod", "min", "max"):
            before = t.clone()
            native.allreduce_tensor(comm, t, op)
            torch.cuda.synchronize()
            assert torch.equal(t, before), (dt, op)


def test_pool_growth_with_async_work_in_flight():
    """Pool growth while deferred-wait executes are still enqueued: the
    reserve() device-sync must drain the in-flight chain before retiring
    the old buffers; afterwards the old plan re-binds and stays exact."""
    topo = Topology((1, 1))
    dims_s, dims_b = (48, 40, 32), (128, 112, 96)

    def inplace_pair(dims, seed):
        Pi = Pencil(topo, dims, (1, 2))
        Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
        n = Pi.length_local(0)
        buf = torch.empty(n, dtype=torch.float64, device="cuda:0")
        g, parents = seeded_parents(dims, (1, 1), (1, 2), (0, 1, 2), (),
                                    np.float64, seed=seed)
        buf.copy_(_to_gpu(parents[0]))
        src = PencilArray(Pi, 0, buf)
        dst = PencilArray(Po, 0, buf)
        t = Transposition(dst, src)
        assert t.aliased
        exp = orc.transpose_oracle(parents, dims, (1, 1), (1, 2),
                                   (0, 1, 2), (0, 2), (1, 2, 0), ())[0]
        return t, buf, parents[0], exp

    tS, bufS, origS, expS = inplace_pair(dims_s, 111)
    # enqueue a chain of deferred-wait executes (forward+back repeatedly,
    # ending forward) without any sync
    PiS = tS.src.pencil
    back = Transposition(
        PencilArray(PiS, 0, bufS),
        PencilArray(tS.dest.pencil, 0, bufS))
    for _ in range(4):
        tS.execute(sync=False)
        back.execute(sync=False)
    tS.execute(sync=False)  # in flight when the big plan grows the pool

    tB, bufB, origB, expB = inplace_pair(dims_b, 222)  # reserve() syncs here
    tB.execute()
    torch.cuda.synchronize()
    assert np.array_equal(bufB.cpu().numpy(), expB)
    assert np.array_equal(bufS.cpu().numpy(), expS)  # chain landed intact

    # old plan re-binds to the grown pool and still bit-exact
    bufS.copy_(_to_gpu(origS))
    tS.execute()
    torch.cuda.synchronize()
    assert np.array_equal(bufS.cpu().numpy(), expS)


def test_gather_of_cuda_array_world1():
    """gather on device arrays converts to host exactly (gather.jl:47: GPU
    data converted to CPU before placement)."""
    dims = (10, 8, 6)
    topo = Topology((1, 1))
    Pi = Pencil(topo, dims, (1, 2), permute=(2, 0, 1))
    g, parents = seeded_parents(dims, (1, 1), (1, 2), (2, 0, 1), (),
                                np.float64)
    x = PencilArray(Pi, 0, _to_gpu(parents[0]))
    from pencilarrays_amd import gather_sim
    got = gather_sim([x])
    assert got.shape == dims
    assert np.array_equal(got, g)


def test_set_comm_size_mismatch_rejected():
    """pa_plan_set_comm validates the communicator against the subgroup
    (nranks == P, rank == my coordinate along R)."""
    uid = native._nccl_uid()
    comm = native.NativeComm.create(uid, 1, 0)
    topo = Topology((2, 1))
    Pi = Pencil(topo, (16, 12, 8), (1, 2))
    Po = Pencil(topo, (16, 12, 8), (0, 2))
    nat = native.NativePlan(Pi, Po, 0, 8)
    assert nat.nproc_sub == 2
    with pytest.raises(RuntimeError, match="comm size 1 != subgroup size 2"):
        nat.set_comm(comm)
