"""CPU execution of :class:`CopyDesc` strided copies via numpy.

This is the host-side mirror of the HIP copy engine, used by the CPU
(gloo-tested) path and by unit tests.  The GPU product path never runs this —
it executes the same descriptors with HIP kernels through the C ABI.
"""

from __future__ import annotations

import numpy as np

from .plan import CopyDesc


def apply_copy(desc: CopyDesc, src_flat: np.ndarray, dst_flat: np.ndarray) -> None:
    """dst[doffset + Σ j·dstrides] = src[soffset + Σ j·sstrides]."""
    if desc.nelem == 0:
        return
    assert src_flat.ndim == 1 and dst_flat.ndim == 1
    isz = src_flat.itemsize
    assert dst_flat.itemsize == isz
    sv = np.lib.stride_tricks.as_strided(
        src_flat[desc.soffset:],
        shape=desc.dims,
        strides=tuple(s * isz for s in desc.sstrides),
    )
    dv = np.lib.stride_tricks.as_strided(
        dst_flat[desc.doffset:],
        shape=desc.dims,
        strides=tuple(s * isz for s in desc.dstrides),
    )
    dv[...] = sv
