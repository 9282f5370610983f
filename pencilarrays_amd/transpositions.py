"""transpose!: the global pencil redistribution (the hot path).

Host orchestration mirroring src/Transpositions/Transpositions.jl:

- ``Transposition(dest, src)`` — plan object (Transpositions.jl:94-119), here
  amortisable across calls (the reference re-plans per call, :165-167).
- ``transpose(t)`` / ``transpose_into(dest, src)`` — :142-180.

Execution backends:

- **numpy + torch.distributed (gloo or none)** — the host mirror used by CPU
  tests (including world_size>1 gloo runs, standing in for the reference's
  "N MPI ranks on one box" test harness, test/runtests.jl:29-54).  Pack /
  unpack / fused-local copies run the SAME CopyDescs as the GPU engine, via
  numpy.  Exchange uses isend/irecv pairs (tag 42, Transpositions.jl:469-477).
- **torch cuda tensors** — the product path: every copy and the RCCL exchange
  run inside the native HIP engine (`libpencilhip.so`).  If the native engine
  is unavailable this path raises — there is no silent eager fallback.
"""

from __future__ import annotations

from typing import Sequence

import numpy as np

from .array import PencilArray
from .copyexec import apply_copy
from .pencil import Pencil
from .plan import TransposePlan, build_plan

MPI_TAG = 42  # Transpositions.jl:469


def _arrays_alias(a, b) -> bool:
    """Base.mightalias equivalent (Transpositions.jl:250): do the two parent
    buffers share memory?  (ManyPencilArray in-place transposes.)"""
    if isinstance(a, np.ndarray) and isinstance(b, np.ndarray):
        return np.shares_memory(a, b)
    if not isinstance(a, np.ndarray) and not isinstance(b, np.ndarray):
        try:
            sa, sb = a.untyped_storage(), b.untyped_storage()
            if sa.data_ptr() != sb.data_ptr():
                # conservative overlap check on address ranges
                a0, a1 = a.data_ptr(), a.data_ptr() + a.nbytes
                b0, b1 = b.data_ptr(), b.data_ptr() + b.nbytes
                return a0 < b1 and b0 < a1
            return True
        except Exception:
            return False
    return False


class Transposition:
    def __init__(self, dest: PencilArray, src: PencilArray,
                 method: str = "grouped"):
        """``Transposition(Ao, Ai; method)`` (Transpositions.jl:94-119).

        ``method`` is accepted for signature parity with the reference's
        PointToPoint/Alltoallv choice (:56-68): on MI355X both map to the
        same grouped ncclSend/ncclRecv exchange over xGMI (RCCL has no
        alltoallv; grouped p2p IS the alltoall realisation), so the value
        ("grouped", "point_to_point", "alltoallv") does not change the
        execution.
        """
        if dest.extra_dims != src.extra_dims:
            raise ValueError(
                f"incompatible number of extra dimensions of PencilArrays: "
                f"{src.extra_dims} != {dest.extra_dims}")
        if dest.rank != src.rank:
            raise ValueError("dest and src must live on the same rank")
        if method not in ("grouped", "point_to_point", "alltoallv"):
            raise ValueError(f"unknown transpose method {method!r}")
        self.method = method
        self.src = src
        self.dest = dest
        self.aliased = _arrays_alias(src.data, dest.data)
        self.plan: TransposePlan = build_plan(
            src.pencil, dest.pencil, src.rank, src.extra_dims,
            aliased=self.aliased)
        self._native = None  # set lazily for the GPU path

    # ------------------------------------------------------------------
    # CPU (numpy) execution — host mirror / test infrastructure.
    # ------------------------------------------------------------------

    def _execute_numpy(self, use_dist: bool):
        plan = self.plan
        src_flat = self.src.data
        dst_flat = self.dest.data
        itemsize = src_flat.itemsize

        if plan.r_dim is None or plan.nproc_sub == 1:
            if plan.local is not None:
                apply_copy(plan.local, src_flat, dst_flat)
            elif plan.self_pack is not None:
                # in-place: stage through recv buffer (Transpositions.jl:250-264)
                recv_buf = np.empty(plan.recv_nelem_total, dtype=src_flat.dtype)
                apply_copy(plan.self_pack, src_flat, recv_buf)
                apply_copy(plan.self_unpack, recv_buf, dst_flat)
            return

        if not use_dist:
            raise RuntimeError(
                "distributed transpose requires torch.distributed to be "
                "initialised (or use run_transpose_sim for in-process tests)")

        import torch
        import torch.distributed as dist

        # one process per rank: dist rank/world must match the topology
        # (a mismatch would silently address the wrong peers)
        topo = plan.Pi.topology
        if dist.get_world_size() != topo.nranks:
            raise RuntimeError(
                f"torch.distributed world size {dist.get_world_size()} != "
                f"topology nranks {topo.nranks}")
        if dist.get_rank() != plan.rank:
            raise RuntimeError(
                f"torch.distributed rank {dist.get_rank()} != topology "
                f"rank {plan.rank}")

        send_buf = np.empty(plan.send_nelem_total, dtype=src_flat.dtype)
        recv_buf = np.empty(plan.recv_nelem_total, dtype=src_flat.dtype)

        # 1. pack all remote blocks (Transpositions.jl:346-431); in aliased
        # mode also stage the self block into the recv tail BEFORE any write
        # of dst (:394-404)
        for blk in plan.peers:
            if blk.pack is not None:
                apply_copy(blk.pack, src_flat, send_buf)
        if plan.self_pack is not None:
            apply_copy(plan.self_pack, src_flat, recv_buf)

        # 2. exchange: per-peer nonblocking send/recv (:463-479)
        reqs = []
        for blk in plan.peers:
            if blk.peer_k == plan.my_k:
                continue
            if blk.recv_nelem > 0:
                rt = torch.from_numpy(
                    recv_buf[blk.recv_offset:blk.recv_offset + blk.recv_nelem])
                reqs.append(dist.irecv(rt, src=blk.global_rank, tag=MPI_TAG))
            if blk.send_nelem > 0:
                st = torch.from_numpy(
                    send_buf[blk.send_offset:blk.send_offset + blk.send_nelem])
                reqs.append(dist.isend(st, dst=blk.global_rank, tag=MPI_TAG))

        # 3. fused local (self) block overlaps the exchange (:394-404 + :530)
        if plan.local is not None:
            apply_copy(plan.local, src_flat, dst_flat)
        elif plan.self_unpack is not None:
            apply_copy(plan.self_unpack, recv_buf, dst_flat)

        for r in reqs:
            r.wait()

        # 4. unpack received blocks (:489-536)
        for blk in plan.peers:
            if blk.unpack is not None:
                apply_copy(blk.unpack, recv_buf, dst_flat)

    # ------------------------------------------------------------------
    # GPU execution — native HIP engine only.
    # ------------------------------------------------------------------

    def _execute_torch_cuda(self, sync: bool):
        from . import native
        if self._native is None:
            self._native = native.NativeTransposition(self)
        self._native.execute(self.src.data, self.dest.data, sync=sync)

    # ------------------------------------------------------------------

    def execute(self, sync: bool = True):
        """Run the transpose.  ``sync=False`` is the reference's
        ``transpose!(t; waitall=false)`` contract (Transpositions.jl:142-158):
        the call returns with work enqueued on the stream; call :meth:`wait`
        (== ``MPI.Waitall(t)``, :128-131) before reusing the source or
        relying on the destination outside the stream."""
        if self.src.is_torch:
            if not self.src.data.is_cuda:
                raise RuntimeError(
                    "torch CPU tensors are not a supported backend: use "
                    "numpy arrays for the CPU host mirror, or cuda tensors "
                    "for the native engine")
            self._execute_torch_cuda(sync)
        else:
            import torch.distributed as dist
            self._execute_numpy(use_dist=dist.is_available() and dist.is_initialized())
        return self.dest

    def wait(self):
        """MPI.Waitall(t) (Transpositions.jl:128-131)."""
        if self._native is not None:
            import torch
            self._native.native.wait(torch.cuda.current_stream().cuda_stream)
        return self


def transpose_into(dest: PencilArray, src: PencilArray,
                   method: str = "grouped") -> PencilArray:
    """``transpose!(dest, src; method)`` (Transpositions.jl:161-169)."""
    if dest is src:
        return dest
    return Transposition(dest, src, method=method).execute()


# ----------------------------------------------------------------------
# In-process multi-rank simulation (test harness only): runs every rank's
# pack/exchange/unpack inside one process, memcpy standing in for the
# exchange.  This is how CPU tests cover P>1 without spawning processes.
# ----------------------------------------------------------------------

def run_transpose_sim(dests: Sequence[PencilArray],
                      srcs: Sequence[PencilArray]) -> None:
    nranks = len(srcs)
    plans = [build_plan(s.pencil, d.pencil, r, s.extra_dims,
                        aliased=_arrays_alias(s.data, d.data))
             for r, (d, s) in enumerate(zip(dests, srcs))]

    send_bufs = [np.empty(p.send_nelem_total, dtype=srcs[r].data.dtype)
                 for r, p in enumerate(plans)]
    recv_bufs = [np.empty(p.recv_nelem_total, dtype=srcs[r].data.dtype)
                 for r, p in enumerate(plans)]

    # pack on every rank (incl. staged self blocks, before any dst write)
    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.pack is not None:
                apply_copy(blk.pack, srcs[r].data, send_bufs[r])
        if p.self_pack is not None:
            apply_copy(p.self_pack, srcs[r].data, recv_bufs[r])
    for r, p in enumerate(plans):
        if p.local is not None:
            apply_copy(p.local, srcs[r].data, dests[r].data)
        elif p.self_unpack is not None:
            apply_copy(p.self_unpack, recv_bufs[r], dests[r].data)

    # exchange: copy each send block into the receiver's recv buffer
    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.peer_k == p.my_k or blk.send_nelem == 0:
                continue
            # The matching recv block on the peer: the peer's PeerBlock whose
            # peer coordinate equals MY coordinate along R.
            q = plans[blk.global_rank]
            rblk = q.peers[p.my_k]
            assert rblk.global_rank == r and rblk.recv_nelem == blk.send_nelem, \
                (r, blk, rblk)
            recv_bufs[blk.global_rank][
                rblk.recv_offset:rblk.recv_offset + rblk.recv_nelem] = \
                send_bufs[r][blk.send_offset:blk.send_offset + blk.send_nelem]

    # unpack on every rank
    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.unpack is not None:
                apply_copy(blk.unpack, recv_bufs[r], dests[r].data)
