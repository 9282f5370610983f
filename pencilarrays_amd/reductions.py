"""Global reductions over PencilArrays (src/reductions.jl:9-38):
local mapreduce over the parent + one Allreduce across the full topology —
§8(f) row 3 of the scope table.

Backends for the single collective (exactly one, as the reference does one
MPI.Allreduce, reductions.jl:17-18):

- **device data**: the native engine's `pa_allreduce` (ncclAllReduce over a
  full-topology RCCL communicator, bootstrapped like the exchange comms) —
  the engine does the work, torch.distributed only carries the unique id;
- **host data / unsupported dtypes**: `torch.distributed.all_reduce` on a
  CPU tensor (gloo in CPU tests).

With no process group initialised (world = 1) the local value is global.
"""

from __future__ import annotations

from typing import Callable, Optional

import numpy as np

from .array import PencilArray

_NUMPY_OPS = {
    "sum": np.sum, "prod": np.prod, "min": np.min, "max": np.max,
}


def _dist():
    try:
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized():
            return dist
    except ImportError:
        pass
    return None


def _neutral(op: str, dtype: np.dtype):
    """Identity element of `op` IN THE ARRAY'S DTYPE (a rank with an empty
    local block must contribute the same dtype as every other rank)."""
    dtype = np.dtype(dtype)
    if op == "sum":
        return dtype.type(0)
    if op == "prod":
        return dtype.type(1)
    info = (np.iinfo(dtype) if np.issubdtype(dtype, np.integer)
            else np.finfo(dtype))
    if op == "min":
        return dtype.type(info.max)
    if op == "max":
        return dtype.type(info.min)
    raise ValueError(op)


_NATIVE_DTYPES = None  # torch dtype -> supported by pa_allreduce


def _native_allreduce_scalar(x: PencilArray, local, op: str):
    """One ncclAllReduce through the native engine on a 1-element device
    tensor (reductions.jl:17-18 with RCCL doing the collective).  Returns
    None if this path does not apply (CPU data, world 1, dtype unsupported,
    no native library)."""
    import torch
    data = x.data
    if isinstance(data, np.ndarray) or not data.is_cuda:
        return None
    dist = _dist()
    if dist is None or dist.get_world_size() == 1:
        return None
    global _NATIVE_DTYPES
    if _NATIVE_DTYPES is None:
        _NATIVE_DTYPES = {torch.float64, torch.float32, torch.int64,
                          torch.int32, torch.uint8}
    t = torch.as_tensor([local], device=data.device)
    if t.dtype not in _NATIVE_DTYPES:
        return None
    from . import native
    comm = native.world_comm(x.pencil.topology, x.rank)
    if comm is None:
        return None
    native.allreduce_tensor(comm, t, op)
    torch.cuda.current_stream().synchronize()
    return t.item()


def _allreduce_scalar(x: PencilArray, local, op: str):
    dist = _dist()
    if dist is None:
        return local
    v = _native_allreduce_scalar(x, local, op)
    if v is not None:
        return v
    import torch
    red = {"sum": dist.ReduceOp.SUM, "prod": dist.ReduceOp.PRODUCT,
           "min": dist.ReduceOp.MIN, "max": dist.ReduceOp.MAX}[op]
    t = torch.as_tensor([local])  # CPU tensor: works over gloo
    dist.all_reduce(t, op=red)
    return t.item()


def mapreduce(f: Optional[Callable], op: str, x: PencilArray):
    """mapreduce(f, op, u) (reductions.jl:9-18): f elementwise over the
    LOCAL parent, `op`-reduced locally, then one Allreduce.  op in
    {"sum","prod","min","max"}.  The local scalar is cast to the array's
    dtype so every rank (including ones with empty local blocks)
    contributes the same type."""
    data = x.data
    if isinstance(data, np.ndarray):
        np_dtype = data.dtype
        v = data if f is None else f(data)
        if v.size == 0:
            local = _neutral(op, np_dtype)
        else:
            local = np_dtype.type(_NUMPY_OPS[op](v))
    else:  # torch
        import torch
        np_dtype = np.dtype(str(data.dtype).replace("torch.", ""))
        v = data if f is None else f(data)
        if v.numel() == 0:
            local = _neutral(op, np_dtype)
        else:
            local = np_dtype.type(
                {"sum": torch.sum, "prod": torch.prod,
                 "min": torch.min, "max": torch.max}[op](v).item())
    return _allreduce_scalar(x, local, op)


def sum_(x: PencilArray):
    return mapreduce(None, "sum", x)


def minimum(x: PencilArray):
    return mapreduce(None, "min", x)


def maximum(x: PencilArray):
    return mapreduce(None, "max", x)


def _bool_allreduce(x: PencilArray, local: bool, op: str) -> bool:
    """any/all: logical OR/AND == max/min over {0,1} in int32."""
    return bool(_allreduce_scalar(x, np.int32(local), op))


def any_(f: Callable, x: PencilArray) -> bool:
    """any(f, u) (reductions.jl:26-32): local any + logical-OR Allreduce."""
    data = x.data
    if isinstance(data, np.ndarray):
        local = bool(np.any(f(data))) if data.size else False
    else:
        local = bool(f(data).any().item()) if data.numel() else False
    return _bool_allreduce(x, local, "max")


def all_(f: Callable, x: PencilArray) -> bool:
    """all(f, u) (reductions.jl:34-38)."""
    data = x.data
    if isinstance(data, np.ndarray):
        local = bool(np.all(f(data))) if data.size else True
    else:
        local = bool(f(data).all().item()) if data.numel() else True
    return _bool_allreduce(x, local, "min")
