"""Probe: can the FULL RCCL exchange path run with 2 ranks sharing ONE GPU?

NCCL historically rejects two ranks on one device; if RCCL permits it, a
1-GPU box can execute the real multi-rank path end to end (pack -> grouped
ncclSend/Recv on the comm stream -> event-ordered unpack) and check it
bit-exactly against the oracle — de-risking the 8-GPU round-end run.

Run under torchrun:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29571 \
    tools/probe_rccl_2ranks_1gpu.py [--chunks N]

Control plane is gloo (barriers + uid exchange only); the engine's RCCL
comms are the only NCCL-family communicators.
"""

import argparse
import json
import math
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))

import numpy as np  # noqa: E402
import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

import oracle as orc  # noqa: E402
from pencilarrays_amd import Pencil, PencilArray, Topology, Transposition  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--chunks", type=int, default=1)
    ap.add_argument("--dims", type=int, nargs=3, default=[64, 48, 40])
    args = ap.parse_args()
    if args.chunks > 1:
        os.environ["PENCILHIP_EXCHANGE_CHUNKS"] = str(args.chunks)

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)  # BOTH ranks on the one GPU

    dims = tuple(args.dims)
    pdims = (world, 1)
    di, pi, do, po = (1, 2), (0, 1, 2), (0, 2), (1, 2, 0)
    topo = Topology(pdims)
    Pi = Pencil(topo, dims, di, permute=pi)
    Po = Pencil(topo, dims, do, permute=po)

    # seeded global + per-rank parents (same generator as tests/util.py)
    rng = np.random.default_rng(0xC0FFEE)
    garr = rng.standard_normal(dims).astype(np.float64)
    my_parent = orc.parent_from_global(garr, dims, pdims, di, pi, rank, ())

    src = PencilArray(Pi, rank, torch.from_numpy(
        np.ascontiguousarray(my_parent)).to("cuda:0"))
    dst = PencilArray(Po, rank, torch.empty(
        Po.length_local(rank), dtype=torch.float64, device="cuda:0"))

    t = Transposition(dst, src)
    try:
        t.execute()
        torch.cuda.synchronize()
    except Exception as exc:
        print(json.dumps({"rank": rank, "ok": False,
                          "error": str(exc)[:400]}), flush=True)
        dist.barrier()
        return

    exp = orc.transpose_oracle(
        [orc.parent_from_global(garr, dims, pdims, di, pi, r, ())
         for r in range(world)],
        dims, pdims, di, pi, do, po, ())[rank]
    got = dst.data.cpu().numpy()
    exact = bool(np.array_equal(got, exp))

    # also run the deferred-wait chain a few times on the reused plan
    for _ in range(5):
        t.execute(sync=False)
    t.wait()
    torch.cuda.synchronize()
    exact2 = bool(np.array_equal(dst.data.cpu().numpy(), exp))

    print(json.dumps({"rank": rank, "ok": True, "bit_exact": exact,
                      "reuse_exact": exact2, "chunks": args.chunks,
                      "nproc_sub": t._native.native.nproc_sub}), flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
