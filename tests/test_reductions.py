"""Reductions over PencilArrays (reductions.jl:9-38 semantics): local
mapreduce + one Allreduce.  Single-rank vs numpy on the gathered array;
world-2 gloo for the collective."""

import os

import numpy as np
import pytest
import torch.multiprocessing as mp

from pencilarrays_amd import Pencil, PencilArray, Topology
from pencilarrays_amd import reductions as red
from util import seeded_parents


def test_single_rank_matches_numpy():
    dims = (16, 21, 41)
    topo = Topology((1, 1))
    pen = Pencil(topo, dims, (1, 2), permute=(1, 2, 0))
    g, parents = seeded_parents(dims, (1, 1), (1, 2), (1, 2, 0), (),
                                np.float64)
    x = PencilArray(pen, 0, parents[0])
    assert np.isclose(red.sum_(x), g.sum())
    assert red.minimum(x) == g.min()
    assert red.maximum(x) == g.max()
    assert red.any_(lambda v: v > g.max() - 1e-12, x)
    assert not red.any_(lambda v: v > g.max() + 1, x)
    assert red.all_(lambda v: v >= g.min(), x)
    assert not red.all_(lambda v: v > g.min(), x)
    assert np.isclose(red.mapreduce(np.abs, "max", x), np.abs(g).max())


def _worker(rank, world, port):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import sys
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        sys.path.insert(0, repo)
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import numpy as np
        from pencilarrays_amd import Pencil, PencilArray, Topology
        from pencilarrays_amd import reductions as red
        from util import seeded_parents

        dims, pdims = (16, 21, 41), (2, 1)
        topo = Topology(pdims)
        pen = Pencil(topo, dims, (1, 2))
        g, parents = seeded_parents(dims, pdims, (1, 2), (0, 1, 2), (),
                                    np.float64)
        x = PencilArray(pen, rank, parents[rank])
        assert np.isclose(red.sum_(x), g.sum())
        assert red.minimum(x) == g.min()
        assert red.maximum(x) == g.max()
        assert red.any_(lambda v: v == g.max(), x)  # max lives on ONE rank
        assert red.all_(lambda v: v >= g.min(), x)
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_gloo_world2_reductions():
    mp.spawn(_worker, args=(2, 29751), nprocs=2, join=True)
