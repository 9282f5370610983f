"""ctypes host for the native engine (libpencilhip.so, C ABI in
include/pencilhip.h).

The GPU product path runs EXCLUSIVELY through this module: plan building,
pack/unpack/fused-copy kernels and the RCCL exchange all happen inside the
native library.  If the library is missing on a GPU box this module raises —
there is no eager/PyTorch fallback.

RCCL bootstrap: the 128-byte unique id of each 1-D subgroup communicator is
exchanged through the torch.distributed store (any backend); the library then
calls ncclCommInitRank itself (replacing MPI.Cart_sub,
MPITopologies.jl:244-251).  A Julia/MPI host would broadcast the id over MPI
instead — see INTEGRATION.md.
"""

from __future__ import annotations

import ctypes
import os
from typing import Dict, Optional, Tuple

I64 = ctypes.c_int64
I32 = ctypes.c_int32
VP = ctypes.c_void_p

_LIB = None


def lib_path() -> str:
    return os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "libpencilhip.so")


def load() -> ctypes.CDLL:
    global _LIB
    if _LIB is not None:
        return _LIB
    path = lib_path()
    if not os.path.exists(path):
        raise RuntimeError(
            f"native engine not built: {path} missing. Run "
            f"`python -c 'import __graft_entry__; __graft_entry__.build()'` "
            f"(hipcc, seconds). The GPU path has no fallback.")
    lib = ctypes.CDLL(path)
    lib.pa_last_error.restype = ctypes.c_char_p
    lib.pa_pencil_length_local.restype = I64
    _LIB = lib
    return lib


def _check(lib, status: int, what: str):
    if status != 0:
        raise RuntimeError(f"{what}: {lib.pa_last_error().decode()}")


class NativePlan:
    """pa_plan plus the pa_topology/pa_pencil handles it needs.  Plan
    building is host-only (works without a GPU); execute needs one."""

    def __init__(self, Pi, Po, rank: int, elem_size: int,
                 extra_dims: Tuple[int, ...] = (), aliased: bool = False):
        lib = self.lib = load()
        topo = Pi.topology
        # both pa_pencils are built against ONE pa_topology handle, so the
        # engine-side topology check can't see a host-level mismatch —
        # validate here (assert_compatible, Transpositions.jl:184-186)
        if tuple(Po.topology.dims) != tuple(topo.dims):
            raise RuntimeError("pencil topologies must be the same.")
        m = topo.ndims
        self._topo = VP()
        _check(lib, lib.pa_topology_create(
            m, (I64 * m)(*topo.dims), ctypes.byref(self._topo)),
            "pa_topology_create")

        def mk_pencil(p):
            h = VP()
            n = p.ndims
            perm = (I32 * n)(*p.perm)
            _check(lib, lib.pa_pencil_create(
                self._topo, n, (I64 * n)(*p.size_global),
                (I32 * m)(*p.decomp_dims), perm, ctypes.byref(h)),
                "pa_pencil_create")
            return h

        self._pi = mk_pencil(Pi)
        self._po = mk_pencil(Po)
        self._plan = VP()
        e = len(extra_dims)
        _check(lib, lib.pa_plan_create(
            self._pi, self._po, I64(elem_size), e,
            (I64 * e)(*extra_dims) if e else None, rank,
            1 if aliased else 0,
            ctypes.byref(self._plan)), "pa_plan_create")
        self.nproc_sub = lib.pa_plan_nproc_sub(self._plan)
        self.r_dim = lib.pa_plan_r_dim(self._plan)
        self.my_k = lib.pa_plan_my_k(self._plan)

    def buffer_sizes(self) -> Tuple[int, int]:
        s, r = I64(), I64()
        _check(self.lib, self.lib.pa_plan_buffer_sizes(
            self._plan, ctypes.byref(s), ctypes.byref(r)),
            "pa_plan_buffer_sizes")
        return s.value, r.value

    def set_buffers(self, send_ptr: int, recv_ptr: int):
        _check(self.lib, self.lib.pa_plan_set_buffers(
            self._plan, VP(send_ptr), VP(recv_ptr)), "pa_plan_set_buffers")

    def set_comm(self, comm: "NativeComm"):
        _check(self.lib, self.lib.pa_plan_set_comm(self._plan, comm.handle),
               "pa_plan_set_comm")

    def execute(self, src_ptr: int, dst_ptr: int, stream: int):
        _check(self.lib, self.lib.pa_transpose_execute(
            self._plan, VP(src_ptr), VP(dst_ptr), VP(stream)),
            "pa_transpose_execute")

    def wait(self, stream: int):
        _check(self.lib, self.lib.pa_transpose_wait(self._plan, VP(stream)),
               "pa_transpose_wait")

    def enable_timing(self, enable: bool = True):
        """Per-stage HIP-event timing (the TimerOutputs analogue,
        Transpositions.jl:173-177)."""
        _check(self.lib, self.lib.pa_plan_enable_timing(
            self._plan, 1 if enable else 0), "pa_plan_enable_timing")

    def stage_times(self) -> dict:
        """{pack, local, exchange, unpack} in ms for the LAST completed
        execute (call after wait()); absent stages are None."""
        out = (ctypes.c_double * 4)()
        _check(self.lib, self.lib.pa_plan_stage_times(self._plan, out),
               "pa_plan_stage_times")
        keys = ("pack", "local", "exchange", "unpack")
        return {k: (None if out[i] < 0 else out[i])
                for i, k in enumerate(keys)}

    # ---- introspection (host-side; parity tests vs plan.py) ----------

    def block_info(self, k: int):
        out = (I64 * 8)()
        _check(self.lib, self.lib.pa_plan_block_info(self._plan, k, out),
               "pa_plan_block_info")
        return tuple(out)

    def copydesc(self, which: int, k: int = 0):
        nd = I64()
        dims = (I64 * 8)()
        sstr = (I64 * 8)()
        dstr = (I64 * 8)()
        soff = I64()
        doff = I64()
        st = self.lib.pa_plan_copydesc(
            self._plan, which, k, ctypes.byref(nd), dims, sstr,
            ctypes.byref(soff), dstr, ctypes.byref(doff))
        if st != 0:
            return None
        n = nd.value
        return (tuple(dims[:n]), tuple(sstr[:n]), soff.value,
                tuple(dstr[:n]), doff.value)

    def __del__(self):
        lib = getattr(self, "lib", None)
        if lib is None:
            return
        if getattr(self, "_plan", None):
            lib.pa_plan_destroy(self._plan)
        for h in (getattr(self, "_pi", None), getattr(self, "_po", None)):
            if h:
                lib.pa_pencil_destroy(h)
        if getattr(self, "_topo", None):
            lib.pa_topology_destroy(self._topo)


class NativeComm:
    def __init__(self, handle):
        self.handle = handle

    @classmethod
    def create(cls, uid: bytes, nranks: int, rank: int) -> "NativeComm":
        lib = load()
        h = VP()
        _check(lib, lib.pa_comm_create(
            ctypes.create_string_buffer(uid, len(uid)), nranks, rank,
            ctypes.byref(h)), "pa_comm_create")
        return cls(h)


_COMM_CACHE: Dict[tuple, NativeComm] = {}


def exchange_uid(topology, r_dim, rank: int, uid_fn) -> Tuple[bytes, int, int]:
    """Exchange the subgroup's RCCL unique id through the torch.distributed
    store: the subgroup leader (coordinate 0 along r_dim) generates it with
    ``uid_fn()`` and publishes it under a key unique to the subgroup; the
    others fetch it, acknowledge, and the leader deletes the key — so a
    reused store (restart, comm re-creation) can never serve a stale uid.
    ``r_dim=None`` means the FULL topology (the world communicator used by
    reductions).  Returns (uid, subgroup_size, subgroup_rank).
    Assumes dist rank == topology rank (one process per GPU)."""
    if r_dim is None:
        ranks = list(range(topology.nranks))
    else:
        ranks = topology.subgroup_ranks(rank, r_dim)
    import torch.distributed as dist
    if not (dist.is_available() and dist.is_initialized()):
        raise RuntimeError(
            "multi-GPU transpose needs torch.distributed initialised for the "
            "RCCL unique-id exchange")
    if dist.get_world_size() != topology.nranks:
        raise RuntimeError(
            f"torch.distributed world size {dist.get_world_size()} != "
            f"topology nranks {topology.nranks} (one process per GPU, dist "
            f"rank == topology rank)")
    if dist.get_rank() != rank:
        raise RuntimeError(
            f"torch.distributed rank {dist.get_rank()} != topology rank "
            f"{rank}")
    store = dist.distributed_c10d._get_default_store()
    sub_rank = ranks.index(rank)
    store_key = f"pencilhip_uid_{topology.dims}_{r_dim}_{min(ranks)}"
    ack_key = store_key + "_ack"
    if sub_rank == 0:
        uid = uid_fn()
        store.set(store_key, uid)
        # wait until every follower has read the uid, then retire the key
        import time as _time
        while store.add(ack_key, 0) < len(ranks) - 1:
            _time.sleep(0.001)
        try:
            store.delete_key(store_key)
            store.delete_key(ack_key)
        except Exception:
            pass  # stores without delete_key: keys stay (benign in-session)
    else:
        uid = bytes(store.get(store_key))
        store.add(ack_key, 1)
    return uid, len(ranks), sub_rank


def _nccl_uid() -> bytes:
    lib = load()
    n = lib.pa_unique_id_size()
    buf = ctypes.create_string_buffer(n)
    _check(lib, lib.pa_get_unique_id(buf), "pa_get_unique_id")
    return bytes(buf.raw)


def subgroup_comm(topology, r_dim: int, rank: int) -> Optional[NativeComm]:
    """RCCL communicator for the 1-D subgroup through ``rank`` along
    ``r_dim`` (replacing topology.subcomms[R], MPITopologies.jl:244-251)."""
    ranks = topology.subgroup_ranks(rank, r_dim)
    if len(ranks) == 1:
        return None
    key = (tuple(topology.dims), r_dim, tuple(ranks))
    if key in _COMM_CACHE:
        return _COMM_CACHE[key]
    uid, nranks, sub_rank = exchange_uid(topology, r_dim, rank, _nccl_uid)
    comm = NativeComm.create(uid, nranks, sub_rank)
    _COMM_CACHE[key] = comm
    return comm


def world_comm(topology, rank: int) -> Optional[NativeComm]:
    """RCCL communicator over the FULL topology (every rank): the engine
    backend for reductions' single Allreduce (reductions.jl:17-18)."""
    if topology.nranks == 1:
        return None
    key = (tuple(topology.dims), None, topology.nranks)
    if key in _COMM_CACHE:
        return _COMM_CACHE[key]
    uid, nranks, sub_rank = exchange_uid(topology, None, rank, _nccl_uid)
    comm = NativeComm.create(uid, nranks, sub_rank)
    _COMM_CACHE[key] = comm
    return comm


def allreduce_tensor(comm: NativeComm, t, op: str):
    """In-place ncclAllReduce of a contiguous cuda tensor through the native
    engine (pa_allreduce; reductions.jl:17-18)."""
    import torch
    lib = load()
    dtmap = {torch.float64: 0, torch.float32: 1, torch.int64: 2,
             torch.int32: 3, torch.uint8: 4}
    opmap = {"sum": 0, "prod": 1, "min": 2, "max": 3}
    if t.dtype not in dtmap:
        raise TypeError(f"pa_allreduce: unsupported dtype {t.dtype}")
    stream = torch.cuda.current_stream().cuda_stream
    _check(lib, lib.pa_allreduce(
        comm.handle, VP(t.data_ptr()), VP(t.data_ptr()), I64(t.numel()),
        dtmap[t.dtype], opmap[op], VP(stream)), "pa_allreduce")
    return t


class StagingPool:
    """Per-device shared staging buffers — the engine's analogue of the
    reference's send_buf/recv_buf shared across derived pencils
    (Pencils.jl:187-189, 257-271; the JLArray test even asserts no
    reallocation, test/array_types.jl:118-127).  Grow-only; every
    NativeTransposition on the device slices the same two tensors, so an
    x->y->z chain holds ONE send/recv pair.  Safe for same-stream chaining
    (stream order serialises reuse), like the reference's shared buffers
    under deferred waits."""

    def __init__(self, device):
        import torch
        self.torch = torch
        self.device = device
        self.send = torch.empty(0, dtype=torch.uint8, device=device)
        self.recv = torch.empty(0, dtype=torch.uint8, device=device)
        self.version = 0

    def reserve(self, send_bytes: int, recv_bytes: int):
        grow = (send_bytes > self.send.numel()
                or recv_bytes > self.recv.numel())
        if grow:
            # growth is a plan-creation-time event; drain in-flight work
            # (incl. the engine's comm stream, invisible to torch) before
            # retiring the old buffers
            if self.device.type == "cuda":
                self.torch.cuda.synchronize(self.device)
            if send_bytes > self.send.numel():
                self.send = self.torch.empty(
                    send_bytes, dtype=self.torch.uint8, device=self.device)
            if recv_bytes > self.recv.numel():
                self.recv = self.torch.empty(
                    recv_bytes, dtype=self.torch.uint8, device=self.device)
            self.version += 1
        return self.send, self.recv


_POOLS: Dict[int, StagingPool] = {}


def staging_pool(device) -> StagingPool:
    idx = device.index if device.index is not None else 0
    if idx not in _POOLS:
        _POOLS[idx] = StagingPool(device)
    return _POOLS[idx]


class NativeTransposition:
    """GPU execution of one Transposition: staging buffers from torch's
    allocator, kernels + RCCL inside the native library."""

    def __init__(self, t):
        import torch
        self.torch = torch
        plan = t.plan
        src = t.src
        esz = src.data.element_size()
        self.np_plan = plan
        self.native = NativePlan(plan.Pi, plan.Po, plan.rank, esz,
                                 plan.extra_dims, aliased=plan.aliased)
        sb, rb = self.native.buffer_sizes()
        dev = src.data.device
        # Staging comes from the per-device shared pool (the reference's
        # buffer sharing across derived pencils, Pencils.jl:257-271) unless
        # PENCILHIP_PRIVATE_STAGING=1 asks for plan-private buffers.
        # never hand the engine a null pointer (a null send_buf would
        # trigger its hipMalloc fallback and orphan the pool's recv buffer)
        self._need = (max(sb, 1), max(rb, 1))
        if os.environ.get("PENCILHIP_PRIVATE_STAGING") == "1":
            self._pool = None
            self._send = torch.empty(max(sb, 1), dtype=torch.uint8,
                                     device=dev)
            self._recv = torch.empty(max(rb, 1), dtype=torch.uint8,
                                     device=dev)
            self.native.set_buffers(self._send.data_ptr(),
                                    self._recv.data_ptr())
        else:
            self._pool = staging_pool(dev)
            self._bind_pool()
        if self.native.nproc_sub > 1:
            comm = subgroup_comm(plan.Pi.topology, self.native.r_dim,
                                 plan.rank)
            if comm is not None:
                self.native.set_comm(comm)

    def _bind_pool(self):
        send, recv = self._pool.reserve(*self._need)
        self._send, self._recv = send, recv
        self.native.set_buffers(send.data_ptr(), recv.data_ptr())
        self._pool_version = self._pool.version

    def execute(self, src_tensor, dst_tensor, sync: bool = True):
        torch = self.torch
        assert src_tensor.is_cuda and dst_tensor.is_cuda
        assert src_tensor.is_contiguous() and dst_tensor.is_contiguous()
        if self._pool is not None and \
                self._pool.version != self._pool_version:
            self._bind_pool()  # another plan grew the pool: re-point
        stream = torch.cuda.current_stream().cuda_stream
        self.native.execute(src_tensor.data_ptr(), dst_tensor.data_ptr(),
                            stream)
        if sync:
            self.native.wait(stream)
