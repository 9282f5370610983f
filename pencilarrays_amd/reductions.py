"""Global reductions over PencilArrays (src/reductions.jl:9-38):
local mapreduce over the parent + an Allreduce across the full topology —
§8(f) row 3 of the scope table.

The collective is `torch.distributed.all_reduce` (RCCL on GPUs, gloo in CPU
tests — one collective, exactly as the reference does one MPI.Allreduce);
a Julia/C host uses the C ABI's `pa_allreduce` (ncclAllReduce) instead.
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


def _allreduce_scalar(value, op: str, device=None):
    dist = _dist()
    if dist is None:
        return value
    import torch
    red = {"sum": dist.ReduceOp.SUM, "prod": dist.ReduceOp.PRODUCT,
           "min": dist.ReduceOp.MIN, "max": dist.ReduceOp.MAX,
           "band": dist.ReduceOp.BAND if hasattr(dist.ReduceOp, "BAND")
           else dist.ReduceOp.MIN,
           "bor": dist.ReduceOp.BOR if hasattr(dist.ReduceOp, "BOR")
           else dist.ReduceOp.MAX}[op]
    t = torch.as_tensor([value], device=device)
    dist.all_reduce(t, op=red)
    return t.item()


def mapreduce(f: Optional[Callable], op: str, x: PencilArray):
    """mapreduce(f, op, u) (reductions.jl:9-18): f elementwise over the
    LOCAL parent, `op`-reduced locally, then Allreduce'd.  op in
    {"sum","prod","min","max"}."""
    data = x.data
    if isinstance(data, np.ndarray):
        v = data if f is None else f(data)
        if v.size == 0:
            local = _neutral(op, data.dtype)
        else:
            local = _NUMPY_OPS[op](v)
        return _allreduce_scalar(local, op)
    else:  # torch
        import torch
        v = data if f is None else f(data)
        if v.numel() == 0:
            local = _neutral(op, np.dtype(str(v.dtype).replace("torch.", "")))
        else:
            local = {"sum": torch.sum, "prod": torch.prod,
                     "min": torch.min, "max": torch.max}[op](v).item()
        dev = data.device if data.is_cuda else None
        return _allreduce_scalar(local, op, device=dev)


def _neutral(op: str, dtype):
    if op == "sum":
        return dtype.type(0) if hasattr(dtype, "type") else 0
    if op == "prod":
        return 1
    if op == "min":
        return np.inf
    if op == "max":
        return -np.inf
    raise ValueError(op)


def sum_(x: PencilArray):
    return mapreduce(None, "sum", x)


def minimum(x: PencilArray):
    return mapreduce(None, "min", x)


def maximum(x: PencilArray):
    return mapreduce(None, "max", x)


def any_(f: Callable, x: PencilArray) -> bool:
    """any(f, u) (reductions.jl:26-32): local any + logical-OR Allreduce."""
    data = x.data
    if isinstance(data, np.ndarray):
        local = bool(np.any(f(data))) if data.size else False
    else:
        local = bool(f(data).any().item()) if data.numel() else False
    return bool(_allreduce_scalar(int(local), "max"))


def all_(f: Callable, x: PencilArray) -> bool:
    """all(f, u) (reductions.jl:34-38)."""
    data = x.data
    if isinstance(data, np.ndarray):
        local = bool(np.all(f(data))) if data.size else True
    else:
        local = bool(f(data).all().item()) if data.numel() else True
    return bool(_allreduce_scalar(int(local), "min"))
