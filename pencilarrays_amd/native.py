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


def exchange_uid(topology, r_dim: int, rank: int, uid_fn) -> Tuple[bytes, int, int]:
    """Exchange the subgroup's RCCL unique id through the torch.distributed
    store: the subgroup leader (coordinate 0 along r_dim) generates it with
    ``uid_fn()`` and publishes it under a key unique to the subgroup; the
    others fetch it.  Returns (uid, subgroup_size, subgroup_rank).
    Assumes dist rank == topology rank (one process per GPU)."""
    ranks = topology.subgroup_ranks(rank, r_dim)
    import torch.distributed as dist
    if not (dist.is_available() and dist.is_initialized()):
        raise RuntimeError(
            "multi-GPU transpose needs torch.distributed initialised for the "
            "RCCL unique-id exchange")
    store = dist.distributed_c10d._get_default_store()
    sub_rank = ranks.index(rank)
    store_key = f"pencilhip_uid_{topology.dims}_{r_dim}_{min(ranks)}"
    if sub_rank == 0:
        uid = uid_fn()
        store.set(store_key, uid)
    else:
        uid = bytes(store.get(store_key))
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
        self._send = torch.empty(max(sb, 1), dtype=torch.uint8, device=dev)
        self._recv = torch.empty(max(rb, 1), dtype=torch.uint8, device=dev)
        self.native.set_buffers(self._send.data_ptr(), self._recv.data_ptr())
        if self.native.nproc_sub > 1:
            comm = subgroup_comm(plan.Pi.topology, self.native.r_dim,
                                 plan.rank)
            if comm is not None:
                self.native.set_comm(comm)

    def execute(self, src_tensor, dst_tensor, sync: bool = True):
        torch = self.torch
        assert src_tensor.is_cuda and dst_tensor.is_cuda
        assert src_tensor.is_contiguous() and dst_tensor.is_contiguous()
        stream = torch.cuda.current_stream().cuda_stream
        self.native.execute(src_tensor.data_ptr(), dst_tensor.data_ptr(),
                            stream)
        if sync:
            self.native.wait(stream)
