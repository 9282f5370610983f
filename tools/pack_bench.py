# Time the N>1 pack/unpack descriptors (8-GPU 1024^3 x->y shapes) standalone
# via pa_device_copy + HIP events: the HBM legs of the multi-GPU roofline.
import ctypes, sys, torch
sys.path.insert(0, "/root/repo")
from pencilarrays_amd import Topology, Pencil, build_plan, native

lib = native.load()
I64 = ctypes.c_int64

def time_desc(desc, esz, src_t, dst_t, reps=10):
    nd = len(desc.dims)
    args = (nd, (I64*nd)(*desc.dims), (I64*nd)(*desc.sstrides), I64(desc.soffset),
            (I64*nd)(*desc.dstrides), I64(desc.doffset), I64(esz),
            ctypes.c_void_p(src_t.data_ptr()), ctypes.c_void_p(dst_t.data_ptr()), None)
    lib.pa_device_copy(*args); torch.cuda.synchronize()
    best = 1e30
    for _ in range(reps):
        a = torch.cuda.Event(enable_timing=True); b = torch.cuda.Event(enable_timing=True)
        a.record(); lib.pa_device_copy(*args); b.record(); b.synchronize()
        best = min(best, a.elapsed_time(b))
    gb = desc.nelem * esz * 2 / 1e9
    return best, gb / (best*1e-3)

for grid, name in [((2,4), "8gpu 2x4"), ((2,2), "4gpu 2x2"), ((4,2), "8gpu 4x2")]:
    topo = Topology(grid)
    Pi = Pencil(topo, (1024,)*3, (1,2))
    for perm, pname in [(None, "id"), ((1,2,0), "perm")]:
        Po = Pencil(topo, (1024,)*3, (0,2), permute=perm)
        p = build_plan(Pi, Po, 0)
        src = torch.randn(Pi.length_local(0), dtype=torch.float64, device="cuda:0")
        send = torch.empty(max(p.send_nelem_total,1), dtype=torch.float64, device="cuda:0")
        recv = torch.randn(max(p.recv_nelem_total,1), dtype=torch.float64, device="cuda:0")
        dst = torch.empty(Po.length_local(0), dtype=torch.float64, device="cuda:0")
        for blk in p.peers:
            if blk.pack is not None:
                ms, gbs = time_desc(blk.pack, 8, src, send)
                print(f"{name} {pname} pack  k={blk.peer_k} dims={blk.pack.dims} {ms:.3f} ms {gbs:.0f} GB/s")
            if blk.unpack is not None:
                ms, gbs = time_desc(blk.unpack, 8, recv, dst)
                print(f"{name} {pname} unpk  k={blk.peer_k} dims={blk.unpack.dims} {ms:.3f} ms {gbs:.0f} GB/s")
        if p.local is not None:
            ms, gbs = time_desc(p.local, 8, src, dst)
            print(f"{name} {pname} local dims={p.local.dims} {ms:.3f} ms {gbs:.0f} GB/s")
        del src, send, recv, dst; torch.cuda.empty_cache()
