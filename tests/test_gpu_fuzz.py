"""Property fuzz on the GPU: seeded-random world-1 transpose configurations
through the REAL engine (pa_transpose_execute on device) vs the independent
direct-spec oracle, bit-exact, with sentinel-filled destinations and
round-trip identity.  Covers random shape/permutation/dtype mixes across
every kernel path the dispatcher can pick (vector-store tile, scalar tile,
linear runs, 1-D copy incl. the NT gate, byte-ified odd sizes)."""

import math
import os

import numpy as np
import pytest

import oracle as orc
from pencilarrays_amd import Pencil, PencilArray, Topology, Transposition
from util import seeded_parents

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs a ROCm GPU", allow_module_level=True)

_T_DTYPE = {
    np.dtype(np.float16): torch.float16,
    np.dtype(np.float64): torch.float64,
    np.dtype(np.float32): torch.float32,
    np.dtype(np.complex64): torch.complex64,
    np.dtype(np.complex128): torch.complex128,
}


def _sentinel(n, dtype):
    t = torch.empty(max(n, 1), dtype=_T_DTYPE[np.dtype(dtype)],
                    device="cuda:0")
    t.view(torch.uint8).fill_(0xAB)
    return t


def test_gpu_fuzz_world1_vs_oracle():
    rng = np.random.default_rng(0xBEEFCAFE)
    DTYPES = [np.float64, np.float32, np.complex64, np.complex128,
              np.float16]
    trials = int(os.environ.get("PENCILHIP_FUZZ_TRIALS", "40"))
    for trial in range(trials):
        nd = int(rng.integers(2, 5))
        dims = tuple(int(rng.integers(1, 40)) for _ in range(nd))
        m = int(rng.integers(1, min(nd, 3)))
        all_dims = list(range(nd))
        di = tuple(rng.permutation(all_dims)[:m].tolist())
        do = list(di)
        avail = [d for d in all_dims if d not in di]
        if avail and rng.random() < 0.9:
            do[int(rng.integers(0, m))] = int(rng.permutation(avail)[0])
        do = tuple(do)
        pi = tuple(rng.permutation(nd).tolist())
        po = tuple(rng.permutation(nd).tolist())
        extra = (int(rng.integers(2, 4)),) if rng.random() < 0.3 else ()
        dtype = DTYPES[trial % len(DTYPES)]
        pdims = (1,) * m

        topo = Topology(pdims)
        Pi = Pencil(topo, dims, di, permute=pi)
        Po = Pencil(topo, dims, do, permute=po)
        g, parents = seeded_parents(dims, pdims, di, pi, extra, dtype,
                                    seed=9000 + trial)
        pex = math.prod(extra) if extra else 1
        src = PencilArray(
            Pi, 0,
            torch.from_numpy(np.ascontiguousarray(parents[0])).to("cuda:0"),
            extra)
        dst = PencilArray(Po, 0,
                          _sentinel(Po.length_local(0) * pex, dtype), extra)
        Transposition(dst, src).execute()
        torch.cuda.synchronize()
        exp = orc.transpose_oracle(parents, dims, pdims, di, pi, do, po,
                                   extra)[0]
        got = dst.data.cpu().numpy()
        assert np.array_equal(got, exp), \
            f"trial {trial}: {dims} {pdims} {di}{pi}->{do}{po} " \
            f"{np.dtype(dtype).name} extra={extra}"

        # round trip back through the engine
        back = PencilArray(Pi, 0,
                           _sentinel(Pi.length_local(0) * pex, dtype), extra)
        Transposition(back, dst).execute()
        torch.cuda.synchronize()
        assert np.array_equal(back.data.cpu().numpy(), parents[0]), \
            f"trial {trial} roundtrip"


def test_gpu_fuzz_inplace_world1_vs_oracle():
    """15 seeded-random IN-PLACE (aliased) world-1 configs: src and dst
    share one buffer, the engine stages through the shared recv pool
    (Transpositions.jl:250-264 semantics), bit-exact vs oracle."""
    rng = np.random.default_rng(0xA11A5ED)
    DTYPES = [np.float64, np.float32, np.complex64]
    trials = int(os.environ.get("PENCILHIP_FUZZ_TRIALS_INPLACE", "15"))
    for trial in range(trials):
        nd = int(rng.integers(2, 5))
        dims = tuple(int(rng.integers(2, 32)) for _ in range(nd))
        m = int(rng.integers(1, min(nd, 3)))
        all_dims = list(range(nd))
        di = tuple(rng.permutation(all_dims)[:m].tolist())
        do = list(di)
        avail = [d for d in all_dims if d not in di]
        if avail and rng.random() < 0.9:
            do[int(rng.integers(0, m))] = int(rng.permutation(avail)[0])
        do = tuple(do)
        pi = tuple(rng.permutation(nd).tolist())
        po = tuple(rng.permutation(nd).tolist())
        dtype = DTYPES[trial % len(DTYPES)]
        pdims = (1,) * m

        topo = Topology(pdims)
        Pi = Pencil(topo, dims, di, permute=pi)
        Po = Pencil(topo, dims, do, permute=po)
        g, parents = seeded_parents(dims, pdims, di, pi, (), dtype,
                                    seed=4000 + trial)
        n = max(Pi.length_local(0), Po.length_local(0))
        buf = torch.empty(n, dtype=_T_DTYPE[np.dtype(dtype)],
                          device="cuda:0")
        buf[:Pi.length_local(0)].copy_(
            torch.from_numpy(np.ascontiguousarray(parents[0])).to("cuda:0"))
        src = PencilArray(Pi, 0, buf[:Pi.length_local(0)])
        dst = PencilArray(Po, 0, buf[:Po.length_local(0)])
        t = Transposition(dst, src)
        assert t.aliased, f"trial {trial}: aliasing not detected"
        t.execute()
        torch.cuda.synchronize()
        exp = orc.transpose_oracle(parents, dims, pdims, di, pi, do, po,
                                   ())[0]
        assert np.array_equal(dst.data.cpu().numpy(), exp), \
            f"trial {trial}: {dims} {di}{pi}->{do}{po} " \
            f"{np.dtype(dtype).name} in-place"
