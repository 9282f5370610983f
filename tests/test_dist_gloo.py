"""Multi-process CPU transpose over torch.distributed (gloo) — the stand-in
for the reference's `mpiexec -n N` test harness (test/runtests.jl:29-54) and
the coverage of the N>1 exchange path that runs on RCCL on the GPU.

World sizes 2 and 4 on localhost; same parity recipe: oracle-compare on every
rank + gather-compare + round trip."""

import os

import numpy as np
import pytest
import torch.multiprocessing as mp

WORLD_CONFIGS = [
    # (world, pdims, dims, din, pin, dout, pout)
    (2, (2, 1), (16, 21, 41), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0)),
    (2, (1, 2), (16, 21, 41), (1, 2), (0, 1, 2), (1, 0), (0, 1, 2)),
    (4, (2, 2), (16, 21, 41), (1, 2), (0, 1, 2), (0, 2), (1, 2, 0)),
    (4, (2, 2), (16, 21, 41), (0, 2), (1, 2, 0), (0, 1), (2, 1, 0)),
    (4, (4,), (16, 21, 41), (1,), (0, 1, 2), (0,), (1, 2, 0)),
]


def _worker(rank, world, pdims, dims, din, pin, dout, pout, port):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import sys
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        sys.path.insert(0, repo)
        sys.path.insert(0, os.path.join(repo, "oracle"))
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import oracle as orc
        from pencilarrays_amd import (
            Pencil, PencilArray, Topology, Transposition, gather_dist,
            transpose_into,
        )
        from util import seeded_parents

        topo = Topology(pdims)
        Pi = Pencil(topo, dims, din, permute=pin)
        Po = Pencil(topo, dims, dout, permute=pout)
        g, parents = seeded_parents(dims, pdims, din, pin, (), np.float64)
        src = PencilArray(Pi, rank, parents[rank].copy())
        dst = PencilArray.empty(Po, rank)
        transpose_into(dst, src)

        # per-rank oracle compare
        exp = orc.transpose_oracle(parents, dims, pdims, din, pin, dout,
                                   pout, ())
        assert np.array_equal(dst.data, exp[rank]), f"rank {rank} mismatch"

        # gather-compare on root (test/transpose.jl:6-22)
        gs = gather_dist(src)
        gd = gather_dist(dst)
        if rank == 0:
            assert gs is not None and gd is not None
            assert np.array_equal(gs, gd)

        # round trip back
        back = PencilArray.empty(Pi, rank)
        transpose_into(back, dst)
        assert np.array_equal(back.data, parents[rank])
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("cfg", WORLD_CONFIGS,
                         ids=lambda c: f"w{c[0]}_{c[1]}_{c[3]}to{c[5]}")
def test_gloo_transpose(cfg):
    world, pdims, dims, din, pin, dout, pout = cfg
    port = 29511 + abs(hash(cfg)) % 2000
    mp.spawn(_worker, args=(world, pdims, dims, din, pin, dout, pout, port),
             nprocs=world, join=True)


def _worker_inplace_chain(rank, world, port):
    """Distributed IN-PLACE x->y->z->y->x chain over one ManyPencilArray
    buffer per rank (the PencilFFTs in-place pattern), gloo world=4."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import sys
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        sys.path.insert(0, repo)
        sys.path.insert(0, os.path.join(repo, "oracle"))
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import oracle as orc
        from pencilarrays_amd import (
            ManyPencilArray, Pencil, Topology, Transposition,
        )
        from util import seeded_parents

        dims, pdims = (16, 21, 41), (2, 2)
        topo = Topology(pdims)
        pens = (
            Pencil(topo, dims, (1, 2)),
            Pencil(topo, dims, (0, 2), permute=(1, 2, 0)),
            Pencil(topo, dims, (0, 1), permute=(2, 1, 0)),
        )
        g, parents = seeded_parents(dims, pdims, (1, 2), (0, 1, 2), (),
                                    np.float64)
        m = ManyPencilArray(pens, rank)
        n1 = pens[0].length_local(rank)
        m.first.data[:n1] = parents[rank]
        orig = parents[rank].copy()

        t12 = Transposition(m[1], m[0])
        assert t12.aliased
        t12.execute()
        t23 = Transposition(m[2], m[1])
        t23.execute()
        # verify z-pencil state against the oracle
        exp = orc.transpose_oracle(parents, dims, pdims, (1, 2), (0, 1, 2),
                                   (0, 1), (2, 1, 0), ())
        ln = pens[2].length_local(rank)
        assert np.array_equal(m[2].data[:ln], exp[rank]), f"rank {rank}"
        # back down the chain, still in place
        Transposition(m[1], m[2]).execute()
        Transposition(m[0], m[1]).execute()
        assert np.array_equal(m.first.data[:n1], orig)
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_gloo_inplace_chain_world4():
    mp.spawn(_worker_inplace_chain, args=(4, 29971), nprocs=4, join=True)


def _worker_uid_exchange(rank, world, port):
    """The RCCL unique-id bootstrap (native.exchange_uid) with an injected
    uid generator: every subgroup member must receive its leader's 128-byte
    id, with per-subgroup keys and coordinate-ordered subgroup ranks —
    exercised cross-process on gloo (the NCCL call itself needs a GPU)."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import sys
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        sys.path.insert(0, repo)
        from pencilarrays_amd import Topology
        from pencilarrays_amd.native import exchange_uid

        topo = Topology((2, 2))
        for r_dim in (0, 1):
            ranks = topo.subgroup_ranks(rank, r_dim)
            leader = ranks[0]
            fake = bytes([leader]) * 128  # distinct per subgroup leader

            uid, nranks, sub_rank = exchange_uid(
                topo, r_dim, rank, uid_fn=lambda: fake)
            assert nranks == 2
            assert sub_rank == ranks.index(rank)
            assert sub_rank == topo.cart_coords(rank)[r_dim]
            assert uid == bytes([leader]) * 128, \
                f"rank {rank} dim {r_dim}: wrong uid"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_gloo_uid_exchange_world4():
    mp.spawn(_worker_uid_exchange, args=(4, 29181), nprocs=4, join=True)
