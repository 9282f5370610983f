"""PencilArray: local array + decomposition metadata.

Mirrors the slice of src/arrays.jl the hot path and its parity checks need:

- the parent (local storage) has the dimensions of the pencil in **memory**
  order plus optional extra dims appended on the slowest side
  (arrays.jl:134-138); memory axis 0 is the fastest-varying, so the flat
  buffer is byte-identical to the reference's column-major Julia parent.
- logical-order accessors ``size_local`` / ``range_local`` (size.jl,
  arrays.jl:317-340 index-permutation semantics).

Storage backends:
- numpy ndarray (host mirror, used by CPU/gloo tests and the oracle recipes);
- torch tensor on ROCm (``cuda``) — the product path, whose data movement is
  done exclusively by the native HIP engine.

The parent is held as a FLAT 1-D buffer plus (mem_dims,) metadata; views in
memory or logical order are materialised on demand.  This avoids committing to
either C or Fortran axis conventions in the storage itself.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np

from .pencil import Pencil
from .permutations import perm_inv


class PencilArray:
    def __init__(self, pencil: Pencil, rank: int, data_flat,
                 extra_dims: Tuple[int, ...] = ()):
        self.pencil = pencil
        self.rank = rank
        self.extra_dims = tuple(int(e) for e in extra_dims)
        mem = tuple(pencil.size_local(rank, memory_order=True)) + self.extra_dims
        self.mem_dims = mem
        if data_flat.ndim != 1 or data_flat.shape[0] != math.prod(mem):
            raise ValueError(
                f"array has incorrect dimensions: {data_flat.shape}. "
                f"Expected flat length {math.prod(mem)} for memory dims {mem}.")
        self.data = data_flat  # flat, length prod(mem_dims)

    # ---- constructors --------------------------------------------------

    @classmethod
    def empty(cls, pencil: Pencil, rank: int, dtype="float64",
              extra_dims: Tuple[int, ...] = (), backend: str = "numpy",
              device=None):
        mem = tuple(pencil.size_local(rank, memory_order=True)) + tuple(extra_dims)
        n = math.prod(mem)
        if backend == "numpy":
            flat = np.empty(n, dtype=dtype)
        elif backend == "torch":
            import torch
            flat = torch.empty(n, dtype=getattr(torch, str(dtype)) if isinstance(dtype, str) else dtype,
                               device=device)
        else:
            raise ValueError(backend)
        return cls(pencil, rank, flat, extra_dims)

    # ---- views ---------------------------------------------------------

    @property
    def is_torch(self) -> bool:
        return not isinstance(self.data, np.ndarray)

    def parent_memview(self):
        """Parent array with axes in memory order, axis 0 fastest.

        numpy: a Fortran-ordered view of shape ``mem_dims``.
        torch: a view of shape ``reversed(mem_dims)`` permuted so that the
        returned tensor has shape ``mem_dims`` with axis 0 fastest.
        """
        if self.is_torch:
            t = self.data.view(tuple(reversed(self.mem_dims)))
            nd = len(self.mem_dims)
            return t.permute(tuple(range(nd - 1, -1, -1)))
        return self.data.reshape(self.mem_dims, order="F")

    def logical_view(self):
        """Local array with axes in LOGICAL order (+ extra axes): index
        [i0,...,iN-1] like the reference's ``u[i,j,k]`` (arrays.jl:327-337)."""
        n = self.pencil.ndims
        e = len(self.extra_dims)
        mv = self.parent_memview()
        # mem axis i holds logical dim perm[i]; logical dim d is mem axis
        # inv(perm)[d].
        invp = perm_inv(self.pencil.perm)
        order = tuple(invp) + tuple(range(n, n + e))
        if self.is_torch:
            return mv.permute(order)
        return mv.transpose(order)

    def size_local(self, memory_order: bool = False) -> Tuple[int, ...]:
        s = self.pencil.size_local(self.rank, memory_order)
        return tuple(s) + self.extra_dims

    def similar(self, pencil: Optional[Pencil] = None, dtype=None) -> "PencilArray":
        """similar(x, [p::Pencil]) (arrays.jl:287-303): a new uninitialised
        PencilArray with the same (or the given) decomposition, same backend
        and device."""
        p = self.pencil if pencil is None else pencil
        if self.is_torch:
            import torch
            dt = self.data.dtype if dtype is None else dtype
            return PencilArray.empty(p, self.rank, dtype=dt,
                                     extra_dims=self.extra_dims,
                                     backend="torch",
                                     device=self.data.device)
        dt = self.data.dtype if dtype is None else dtype
        return PencilArray.empty(p, self.rank, dtype=dt,
                                 extra_dims=self.extra_dims)

    def __len__(self):
        return int(math.prod(self.mem_dims))
