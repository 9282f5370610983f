"""Transposition plan: the block tables of the global pencil transpose.

Restates the control path of src/Transpositions/Transpositions.jl as pure
metadata.  For a given rank, input pencil Pi and output pencil Po (with the
decompositions differing along one topology dimension R, Transpositions.jl:111)
this computes exactly which sub-blocks are packed/sent/received/unpacked and
with which layout:

- peer enumeration = ``get_remote_indices`` (:542-552): coordinates equal to
  mine except along R, k = 0..P-1; subgroup rank of peer k is k
  (MPITopologies.jl:229-242).
- send block to peer k = intersect(Pi.axes_local, Po.axes_all[k])   (:383)
- recv block from peer k = intersect(Po.axes_local, Pi.axes_all[k]) (:388,:521)
- the self block (k == my coord along R) is placed at the END of the receive
  buffer (:394-404); other blocks at offsets accumulated in k order (:414-415).
- pack order inside a block: column-major (first memory axis fastest) over the
  block's extents in Pi *memory* order, extra dims outermost (copy_range!,
  :554-586).
- unpack: the received block, reshaped column-major to its extents gathered by
  permutation(Pi) (:527, :599-600), is scattered into the Po parent window at
  ``perm * o_range_iperm`` with ``perm = permutation(Po)/permutation(Pi)``
  (:506, :602).

Every data movement is expressed as a :class:`CopyDesc` — an N-d strided copy
``dst[Σ j_i·dstride_i] = src[Σ j_i·sstride_i]`` over ``dims`` — which is what
the HIP copy engine, the numpy executor and the C++ engine all execute.
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import List, Optional, Tuple

from .pencil import (
    Pencil,
    Region,
    region_intersect,
    region_lengths,
    region_nelem,
)
from .permutations import perm_apply, perm_inv, perm_relative

# --------------------------------------------------------------------------


@dataclass(frozen=True)
class CopyDesc:
    """dst[doffset + Σ j_i·dstrides_i] = src[soffset + Σ j_i·sstrides_i],
    j over ``dims`` (all in elements)."""
    dims: Tuple[int, ...]
    sstrides: Tuple[int, ...]
    soffset: int
    dstrides: Tuple[int, ...]
    doffset: int

    @property
    def nelem(self) -> int:
        return math.prod(self.dims) if self.dims else 1


def _colmajor_strides(dims: Tuple[int, ...]) -> Tuple[int, ...]:
    st, acc = [], 1
    for d in dims:
        st.append(acc)
        acc *= d
    return tuple(st)


def normalize_desc(d: CopyDesc) -> CopyDesc:
    """Drop unit axes, sort by ascending src stride, merge adjacent axes that
    are contiguous on BOTH sides.  Semantics-preserving relabeling."""
    axes = [(dim, ss, ds) for dim, ss, ds in
            zip(d.dims, d.sstrides, d.dstrides) if dim != 1]
    if not axes:
        return CopyDesc((1,), (1,), d.soffset, (1,), d.doffset)
    axes.sort(key=lambda a: a[1])
    merged = [axes[0]]
    for dim, ss, ds in axes[1:]:
        pdim, pss, pds = merged[-1]
        if ss == pss * pdim and ds == pds * pdim:
            merged[-1] = (pdim * dim, pss, pds)
        else:
            merged.append((dim, ss, ds))
    dims, ss, ds = zip(*merged)
    return CopyDesc(dims, ss, d.soffset, ds, d.doffset)


# --------------------------------------------------------------------------


@dataclass
class PeerBlock:
    peer_k: int                 # coordinate along R == subgroup rank
    global_rank: int            # rank in the full topology
    send_region: Region         # global logical ranges I send to this peer
    recv_region: Region         # global logical ranges I receive from it
    send_nelem: int
    recv_nelem: int
    send_offset: int            # element offset into send staging buffer
    recv_offset: int            # element offset into recv staging buffer
    pack: Optional[CopyDesc]    # src parent -> send buffer (None if empty/self)
    unpack: Optional[CopyDesc]  # recv buffer -> dst parent (None if empty/self)


@dataclass
class TransposePlan:
    rank: int
    Pi: Pencil
    Po: Pencil
    extra_dims: Tuple[int, ...]
    r_dim: Optional[int]            # R; None = same decomposition (local path)
    nproc_sub: int                  # subgroup size P (1 if local path)
    my_k: int                       # my coordinate along R
    peers: List[PeerBlock] = field(default_factory=list)
    local: Optional[CopyDesc] = None  # fused self/local permuted copy
    send_nelem_total: int = 0
    recv_nelem_total: int = 0       # staging incl. self tail when aliased
    aliased: bool = False
    # aliased (in-place) mode: the self block stages through the recv-buffer
    # tail instead of the fused direct copy (the reference's aliasing path:
    # Transpositions.jl:250-264 for the local case, :394-404 placement for
    # the distributed case), so every read of src completes before any write
    # of dst.
    self_pack: Optional[CopyDesc] = None    # src -> recv_buf tail
    self_unpack: Optional[CopyDesc] = None  # recv_buf tail -> dst

    @property
    def subgroup_global_ranks(self) -> List[int]:
        if self.r_dim is None:
            return [self.rank]
        return self.Pi.topology.subgroup_ranks(self.rank, self.r_dim)


def _parent_strides(p: Pencil, rank: int,
                    extra_dims: Tuple[int, ...]) -> Tuple[Tuple[int, ...], Tuple[int, ...]]:
    """(mem_dims, strides) of the local parent array: memory-order local dims
    plus extra dims appended, column-major (axis 0 fastest) — byte-identical
    to the reference's Julia parent (arrays.jl:134-138)."""
    mem = tuple(p.size_local(rank, memory_order=True)) + tuple(extra_dims)
    return mem, _colmajor_strides(mem)


def _window_desc_src(p: Pencil, rank: int, region: Region,
                     extra_dims: Tuple[int, ...]) -> Tuple[Tuple[int, ...], Tuple[int, ...], int]:
    """dims/strides/offset of a global ``region`` as a window of the local
    parent of pencil ``p``, axes in p's memory order (+ extra axes)."""
    local = p.to_local(rank, region, memory_order=True)  # mem-order local ranges
    _, pst = _parent_strides(p, rank, extra_dims)
    n = p.ndims
    dims = tuple(hi - lo for lo, hi in local) + tuple(extra_dims)
    strides = pst  # one per mem axis then extra axes
    offset = sum(lo * pst[i] for i, (lo, _) in enumerate(local))
    return dims, strides, offset


def build_plan(Pi: Pencil, Po: Pencil, rank: int,
               extra_dims: Tuple[int, ...] = (),
               aliased: bool = False) -> TransposePlan:
    extra_dims = tuple(int(e) for e in extra_dims)
    R = Pi.transpose_dim(Po)  # validates compatibility
    n = Pi.ndims
    E = len(extra_dims)

    # Relative permutation Po-mem-axis -> Pi-mem-axis (Transpositions.jl:506):
    # dest memory axis i' holds logical dim permo[i'], which sits at source
    # memory axis inv(permi)[permo[i']].
    permi = perm_apply(Pi.perm, tuple(range(n)))  # == Pi.perm
    permo = Po.perm
    q_rel = perm_relative(permo, permi)           # len n
    q_full = tuple(q_rel) + tuple(n + i for i in range(E))
    inv_q = perm_inv(q_full)

    plan = TransposePlan(
        rank=rank, Pi=Pi, Po=Po, extra_dims=extra_dims,
        r_dim=R, nproc_sub=1 if R is None else Pi.topology.dims[R],
        my_k=0, aliased=aliased,
    )

    mem_o, pst_o = _parent_strides(Po, rank, extra_dims)

    def unpack_like(buf_dims: Tuple[int, ...], dst_local_mem: Region,
                    src_strides: Tuple[int, ...], src_offset: int) -> CopyDesc:
        """CopyDesc moving a block laid out on ``src`` (axes = Pi-mem order +
        extras, strides given) into the Po parent window starting at
        ``dst_local_mem`` (Po-mem-order local ranges)."""
        # buffer axis j feeds dest mem axis inv_q-of... dst axis i' reads
        # buffer axis q_full[i']; so dstride[j] = pst_o[inv_q[j]] and the
        # dest offset uses the window start of axis inv_q[j].
        starts = tuple(lo for lo, _ in dst_local_mem) + (0,) * E
        dstrides = tuple(pst_o[inv_q[j]] for j in range(n + E))
        doffset = sum(starts[i] * pst_o[i] for i in range(n + E))
        return normalize_desc(CopyDesc(
            dims=buf_dims, sstrides=src_strides, soffset=src_offset,
            dstrides=dstrides, doffset=doffset,
        ))

    def staged_self(region_in: Region, region_out: Region, recv_off: int):
        """Self block via the recv-buffer tail (in-place/aliased mode)."""
        sdims, sst, soff = _window_desc_src(Pi, rank, region_in, extra_dims)
        plan.self_pack = normalize_desc(CopyDesc(
            dims=sdims, sstrides=sst, soffset=soff,
            dstrides=_colmajor_strides(sdims), doffset=recv_off,
        ))
        bdims = (perm_apply(Pi.perm, region_lengths(region_out))
                 + tuple(extra_dims))
        dst_local = Po.to_local(rank, region_out, memory_order=True)
        plan.self_unpack = unpack_like(
            bdims, dst_local, _colmajor_strides(bdims), recv_off)

    if R is None:
        # Same decomposition: plain copy or local permutation
        # (transpose_impl!(::Nothing), Transpositions.jl:214-271; in-place
        # variant stages through recv_buf, :250-264).
        region = Pi.axes_for_rank(rank)
        if region_nelem(region) > 0:
            if aliased:
                staged_self(region, region, 0)
                plan.recv_nelem_total = (region_nelem(region)
                                         * math.prod(extra_dims or (1,)))
            else:
                sdims, sst, soff = _window_desc_src(Pi, rank, region,
                                                    extra_dims)
                dst_local = Po.to_local(rank, region, memory_order=True)
                plan.local = unpack_like(sdims, dst_local, sst, soff)
        return plan

    topo = Pi.topology
    coords = topo.cart_coords(rank)
    plan.my_k = coords[R]
    P = topo.dims[R]

    axes_local_i = Pi.axes_for_rank(rank)
    axes_local_o = Po.axes_for_rank(rank)

    isend = 0
    irecv = 0
    # length of data exchanged with myself (Transpositions.jl:303-306)
    self_region = region_intersect(axes_local_i, axes_local_o)
    length_self = region_nelem(self_region) * math.prod(extra_dims or (1,))
    length_recv_remote = (Po.length_local(rank) * math.prod(extra_dims or (1,))
                          - length_self)

    for k in range(P):
        peer_coords = list(coords)
        peer_coords[R] = k
        peer_coords = tuple(peer_coords)
        grank = topo.cart_rank(peer_coords)

        srange = region_intersect(axes_local_i, Po.axes_for_coords(peer_coords))
        rrange = region_intersect(axes_local_o, Pi.axes_for_coords(peer_coords))
        ns = region_nelem(srange) * math.prod(extra_dims or (1,))
        nr = region_nelem(rrange) * math.prod(extra_dims or (1,))

        blk = PeerBlock(
            peer_k=k, global_rank=grank,
            send_region=srange, recv_region=rrange,
            send_nelem=ns, recv_nelem=nr,
            send_offset=0, recv_offset=0, pack=None, unpack=None,
        )

        if k == plan.my_k:
            # Fused self path: direct src-window -> dst-window permuted copy
            # (replaces the reference's copy to the recv_buf tail :394-404
            # followed by copy_permuted! — same values, half the HBM traffic).
            # In aliased (in-place) mode the direct copy would read windows
            # the unpacks overwrite, so the self block stages through the
            # recv tail exactly like the reference.
            assert nr == length_self and ns == length_self
            blk.recv_offset = length_recv_remote  # kept for reference parity
            if ns > 0:
                if aliased:
                    staged_self(srange, rrange, length_recv_remote)
                else:
                    sdims, sst, soff = _window_desc_src(Pi, rank, srange,
                                                        extra_dims)
                    dst_local = Po.to_local(rank, rrange, memory_order=True)
                    plan.local = unpack_like(sdims, dst_local, sst, soff)
        else:
            blk.send_offset = isend
            blk.recv_offset = irecv
            if ns > 0:
                sdims, sst, soff = _window_desc_src(Pi, rank, srange, extra_dims)
                blk.pack = normalize_desc(CopyDesc(
                    dims=sdims, sstrides=sst, soffset=soff,
                    dstrides=_colmajor_strides(sdims), doffset=isend,
                ))
            if nr > 0:
                # buffer holds the block in Pi memory order of ITS window
                bdims = (perm_apply(Pi.perm, region_lengths(rrange))
                         + tuple(extra_dims))
                dst_local = Po.to_local(rank, rrange, memory_order=True)
                blk.unpack = unpack_like(
                    bdims, dst_local, _colmajor_strides(bdims), irecv)
            isend += ns
            irecv += nr

        plan.peers.append(blk)

    plan.send_nelem_total = isend
    assert irecv == length_recv_remote
    plan.recv_nelem_total = irecv + (length_self if aliased else 0)
    return plan
