"""ManyPencilArray: M aliased PencilArray views over one buffer
(src/multiarrays.jl:106-143): the buffer is sized for the LARGEST of the
pencils (:123), and transposing between two member arrays is an in-place
transpose — Transposition detects the aliasing and stages the self block
through the recv buffer (Transpositions.jl:250-264), exactly like the
reference.  This is what PencilFFTs' in-place plans use."""

from __future__ import annotations

import math
from typing import Sequence, Tuple

import numpy as np

from .array import PencilArray
from .pencil import Pencil


class ManyPencilArray:
    def __init__(self, pencils: Sequence[Pencil], rank: int,
                 dtype="float64", extra_dims: Tuple[int, ...] = (),
                 backend: str = "numpy", device=None):
        extra_dims = tuple(int(e) for e in extra_dims)
        pex = math.prod(extra_dims) if extra_dims else 1
        lengths = [p.length_local(rank) * pex for p in pencils]
        n = max(lengths) if lengths else 0
        if backend == "numpy":
            flat = np.zeros(n, dtype=dtype)
        elif backend == "torch":
            import torch
            tdt = getattr(torch, str(dtype)) if isinstance(dtype, str) else dtype
            flat = torch.zeros(n, dtype=tdt, device=device)
        else:
            raise ValueError(backend)
        self.data = flat
        self.pencils = tuple(pencils)
        self.arrays = tuple(
            PencilArray(p, rank, flat[:ln], extra_dims)
            for p, ln in zip(pencils, lengths))

    @property
    def first(self) -> PencilArray:
        return self.arrays[0]

    @property
    def last(self) -> PencilArray:
        return self.arrays[-1]

    def __getitem__(self, i: int) -> PencilArray:
        return self.arrays[i]

    def __len__(self) -> int:
        return len(self.arrays)
