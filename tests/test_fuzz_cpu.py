"""Property fuzz (CPU): 40 seeded-random full-transpose configurations —
in-process multi-rank simulation vs the independent direct-spec oracle,
bit-exact, plus round-trip identity where the reverse transpose is legal."""

import math

import numpy as np

import oracle as orc
from pencilarrays_amd import Pencil, PencilArray, Topology, run_transpose_sim
from util import seeded_parents


def test_fuzz_sim_vs_oracle():
    rng = np.random.default_rng(0xFACADE)
    for trial in range(40):
        nd = int(rng.integers(2, 5))
        dims = tuple(int(rng.integers(1, 20)) for _ in range(nd))
        m = int(rng.integers(1, min(nd, 3)))
        pdims = tuple(int(rng.integers(1, 4)) for _ in range(m))
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
        dtype = [np.float64, np.float32, np.complex64,
                 np.complex128][trial % 4]

        topo = Topology(pdims)
        Pi = Pencil(topo, dims, di, permute=pi)
        Po = Pencil(topo, dims, do, permute=po)
        g, parents = seeded_parents(dims, pdims, di, pi, extra, dtype,
                                    seed=7000 + trial)
        nr = topo.nranks
        srcs = [PencilArray(Pi, r, parents[r].copy(), extra)
                for r in range(nr)]
        dests = [PencilArray.empty(Po, r, dtype=dtype, extra_dims=extra)
                 for r in range(nr)]
        run_transpose_sim(dests, srcs)
        exp = orc.transpose_oracle(parents, dims, pdims, di, pi, do, po,
                                   extra)
        for r in range(nr):
            assert np.array_equal(dests[r].data, exp[r]), \
                f"trial {trial} rank {r}: {dims} {pdims} {di}{pi}->{do}{po}"

        # round trip back
        back = [PencilArray.empty(Pi, r, dtype=dtype, extra_dims=extra)
                for r in range(nr)]
        run_transpose_sim(back, dests)
        for r in range(nr):
            assert np.array_equal(back[r].data, parents[r]), \
                f"trial {trial} rank {r} roundtrip"

        # in-place variant of the same trial: one shared buffer per rank
        pex = math.prod(extra) if extra else 1
        bufs = []
        for r in range(nr):
            n = max(Pi.length_local(r), Po.length_local(r)) * pex
            b = np.zeros(n, dtype=dtype)
            b[:parents[r].size] = parents[r]
            bufs.append(b)
        srcs_ip = [PencilArray(Pi, r, bufs[r][:parents[r].size], extra)
                   for r in range(nr)]
        dests_ip = [PencilArray(Po, r,
                                bufs[r][:Po.length_local(r) * pex], extra)
                    for r in range(nr)]
        run_transpose_sim(dests_ip, srcs_ip)
        for r in range(nr):
            assert np.array_equal(dests_ip[r].data, exp[r]), \
                f"trial {trial} rank {r} in-place"
