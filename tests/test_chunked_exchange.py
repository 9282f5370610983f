"""Chunked-exchange split logic (PENCILHIP_EXCHANGE_CHUNKS > 1): applying
the per-chunk unpack descriptors sequentially must reproduce the full
unpack bit-exactly, and chunk byte ranges must tile each peer block exactly
with identical splits on sender and receiver.  (The RCCL leg itself needs
>1 GPU and runs at round end; this pins everything else.)"""

import math

import numpy as np

from pencilarrays_amd import Pencil, Topology, build_plan
from pencilarrays_amd.copyexec import apply_copy
from pencilarrays_amd.permutations import perm_apply
from pencilarrays_amd.plan import CopyDesc, normalize_desc


def raw_unpack_desc(plan, blk):
    """Rebuild the un-normalized unpack descriptor (axes = Pi memory order +
    extras, column-major buffer) the engine chunks along its last axis."""
    Pi, Po, rank = plan.Pi, plan.Po, plan.rank
    n = Pi.ndims
    E = len(plan.extra_dims)
    from pencilarrays_amd.plan import _parent_strides
    from pencilarrays_amd.permutations import perm_inv, perm_relative
    bdims = tuple(perm_apply(Pi.perm, tuple(hi - lo for lo, hi in blk.recv_region))) \
        + plan.extra_dims
    bstr = []
    acc = 1
    for d in bdims:
        bstr.append(acc)
        acc *= d
    mem_o, pst_o = _parent_strides(Po, rank, plan.extra_dims)
    q_rel = perm_relative(Po.perm, Pi.perm)
    q_full = tuple(q_rel) + tuple(n + i for i in range(E))
    dst_local = Po.to_local(rank, blk.recv_region, memory_order=True)
    starts = tuple(lo for lo, _ in dst_local) + (0,) * E
    dstr = [0] * (n + E)
    doff = 0
    inv_q = [0] * (n + E)
    for i, v in enumerate(q_full):
        inv_q[v] = i
    for j in range(n + E):
        dstr[j] = pst_o[inv_q[j]]
    for i in range(n + E):
        doff += starts[i] * pst_o[i]
    return CopyDesc(tuple(bdims), tuple(bstr), blk.recv_offset,
                    tuple(dstr), doff)


def chunk_of_raw(raw: CopyDesc, lo: int, hi: int) -> CopyDesc:
    last = len(raw.dims) - 1
    dims = list(raw.dims)
    dims[last] = hi - lo
    return normalize_desc(CopyDesc(
        tuple(dims), raw.sstrides,
        raw.soffset + lo * raw.sstrides[last],
        raw.dstrides, raw.doffset + lo * raw.dstrides[last]))


def test_chunked_unpack_equals_full():
    dims, pdims = (16, 21, 41), (2, 4)
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, (1, 2))
    Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    rng = np.random.default_rng(31337)
    for rank in range(topo.nranks):
        plan = build_plan(Pi, Po, rank)
        if plan.recv_nelem_total == 0:
            continue
        recv = rng.standard_normal(plan.recv_nelem_total)
        out_n = Po.length_local(rank)
        full = np.empty(out_n)
        full.view(np.uint8)[:] = 0xAB
        for blk in plan.peers:
            if blk.unpack is not None:
                apply_copy(blk.unpack, recv, full)
        for C in (2, 3, 5, 8, 64):
            got = np.empty(out_n)
            got.view(np.uint8)[:] = 0xAB
            for blk in plan.peers:
                if blk.unpack is None:
                    continue
                raw = raw_unpack_desc(plan, blk)
                # engine invariant: normalize(raw) == the stored unpack desc
                nr = normalize_desc(raw)
                assert (nr.dims, nr.sstrides, nr.soffset, nr.dstrides,
                        nr.doffset) == (blk.unpack.dims, blk.unpack.sstrides,
                                        blk.unpack.soffset,
                                        blk.unpack.dstrides,
                                        blk.unpack.doffset)
                outer = raw.dims[-1]
                rowelems = blk.recv_nelem // outer
                covered = 0
                for c in range(C):
                    lo = outer * c // C
                    hi = outer * (c + 1) // C
                    if hi <= lo:
                        continue
                    # chunk byte range tiles the block contiguously
                    assert lo * rowelems == covered
                    covered = hi * rowelems
                    apply_copy(chunk_of_raw(raw, lo, hi), recv, got)
                assert covered == blk.recv_nelem
            assert np.array_equal(got, full), f"rank {rank} C={C}"


def test_native_raw_unpack_matches_python():
    """The engine's stored raw unpack descriptor (pa_plan_copydesc which=5)
    equals the Python reconstruction the chunk tests use."""
    import os
    import pytest
    from pencilarrays_amd import native
    if not os.path.exists(native.lib_path()):
        pytest.skip("libpencilhip.so not built")
    dims, pdims = (16, 21, 41), (2, 4)
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, (1, 2))
    Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    for rank in range(topo.nranks):
        plan = build_plan(Pi, Po, rank)
        nat = native.NativePlan(Pi, Po, rank, 8)
        for blk in plan.peers:
            if blk.unpack is None:
                assert nat.copydesc(5, blk.peer_k) is None
                continue
            raw = raw_unpack_desc(plan, blk)
            got = nat.copydesc(5, blk.peer_k)
            assert got == (raw.dims, raw.sstrides, raw.soffset,
                           raw.dstrides, raw.doffset), \
                f"rank {rank} peer {blk.peer_k}"


def test_sender_receiver_chunk_symmetry():
    """For every (sender, receiver) pair: the sender's send-block outer
    extent equals the receiver's recv-block outer extent, so both sides cut
    identical chunk byte ranges."""
    dims, pdims = (16, 21, 41), (2, 4)
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, (1, 2))
    Po = Pencil(topo, dims, (0, 2), permute=(1, 2, 0))
    plans = [build_plan(Pi, Po, r) for r in range(topo.nranks)]
    for r, p in enumerate(plans):
        for blk in p.peers:
            if blk.peer_k == p.my_k:
                continue
            q = plans[blk.global_rank]
            rblk = q.peers[p.my_k]
            # my send block to peer == peer's recv block from me, so the raw
            # dims sequences (Pi mem order) are equal -> same outer extent
            send_dims = perm_apply(Pi.perm, tuple(
                hi - lo for lo, hi in blk.send_region))
            recv_dims = perm_apply(Pi.perm, tuple(
                hi - lo for lo, hi in rblk.recv_region))
            assert send_dims == recv_dims
