"""pencilarrays_amd: an MI355X-native global-transpose engine.

A from-scratch replacement for the hot path of PencilArrays.jl
(Transpositions.transpose!, the global pencil redistribution), built for
AMD Instinct MI355X (gfx950): hand-written HIP copy/transpose kernels behind a
C ABI (`libpencilhip.so`), RCCL over xGMI for the inter-GPU exchange, with the
reference's Pencil / PencilArray / Transposition host API mirrored so the
engine is a drop-in for that path.

Reference: jipolanco/PencilArrays.jl v0.19.11 (file:line cites throughout).
"""

from .permutations import (
    identity_perm,
    perm_append,
    perm_apply,
    perm_inv,
    perm_relative,
    perm_unapply,
)
from .topology import Topology, dims_create
from .pencil import Pencil
from .array import PencilArray
from .plan import CopyDesc, TransposePlan, build_plan, normalize_desc
from .transpositions import Transposition, run_transpose_sim, transpose_into
from .multiarrays import ManyPencilArray
from .pencilio import MPIIOFile
from . import reductions
from .gather import gather_dist, gather_sim

# name parity with the reference's `gather` (gather.jl:17): the distributed
# form, returning the global array on root and None elsewhere.
gather = gather_dist

__all__ = [
    "Topology", "dims_create", "Pencil", "PencilArray",
    "Transposition", "transpose_into", "run_transpose_sim", "ManyPencilArray",
    "MPIIOFile", "reductions",
    "gather", "gather_sim", "gather_dist",
    "CopyDesc", "TransposePlan", "build_plan", "normalize_desc",
    "identity_perm", "perm_apply", "perm_unapply", "perm_inv",
    "perm_relative", "perm_append",
]
