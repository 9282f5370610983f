"""gather: collect the full (global, logical-order) array — the parity oracle
recipe used by every reference transpose test (src/gather.jl, used by
test/transpose.jl:6-22).

Two forms:
- :func:`gather_sim` — in-process: takes all ranks' PencilArrays, returns the
  global array.  Mirrors gather.jl:59-95 placement: dest[axes_all[n]] = block.
- :func:`gather_dist` — torch.distributed form (gloo/nccl): non-root ranks
  inverse-permute their parent (gather.jl:32-40) and send (:49-56); root
  receives per-rank blocks and places them (:68-95).  Returns the array on
  root, None elsewhere.
"""

from __future__ import annotations

from typing import Optional, Sequence

import numpy as np

from .array import PencilArray

MPI_TAG = 42  # gather.jl:23


def _logical_block(x: PencilArray) -> np.ndarray:
    """Local data as a contiguous array in logical order (+ extra dims) —
    the inverse-permuted copy of gather.jl:32-40."""
    lv = x.logical_view()
    if x.is_torch:
        lv = lv.cpu().numpy()
    return np.ascontiguousarray(lv)


def gather_sim(arrays: Sequence[PencilArray]) -> np.ndarray:
    p0 = arrays[0].pencil
    extra = arrays[0].extra_dims
    shape = tuple(p0.size_global) + extra
    first = arrays[0].data
    dtype = first.dtype if isinstance(first, np.ndarray) else \
        np.dtype(str(first.dtype).replace("torch.", ""))
    dest = np.empty(shape, dtype=dtype)
    for r, x in enumerate(arrays):
        assert x.rank == r
        region = x.pencil.axes_for_rank(r)
        sl = tuple(slice(lo, hi) for lo, hi in region) + \
            tuple(slice(None) for _ in extra)
        dest[sl] = _logical_block(x)
    return dest


def gather_dist(x: PencilArray, root: int = 0) -> Optional[np.ndarray]:
    import torch
    import torch.distributed as dist

    comm_rank = dist.get_rank()
    block = _logical_block(x)

    if comm_rank != root:
        dist.send(torch.from_numpy(block), dst=root, tag=MPI_TAG)
        return None

    p = x.pencil
    extra = x.extra_dims
    shape = tuple(p.size_global) + extra
    dest = np.empty(shape, dtype=block.dtype)
    topo = p.topology
    for r in range(topo.nranks):
        region = p.axes_for_rank(r)
        sl = tuple(slice(lo, hi) for lo, hi in region) + \
            tuple(slice(None) for _ in extra)
        if r == root:
            dest[sl] = block
        else:
            dims = tuple(hi - lo for lo, hi in region) + extra
            buf = torch.empty(dims, dtype=torch.from_numpy(block).dtype)
            dist.recv(buf, src=r, tag=MPI_TAG)
            dest[sl] = buf.numpy()
    return dest
