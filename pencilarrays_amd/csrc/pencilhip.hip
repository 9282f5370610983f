/* pencilhip.hip — MI355X-native (gfx950) global-transpose engine.
 *
 * Implements include/pencilhip.h: the engine under PencilArrays.jl's
 * Transpositions.transpose! hot path, built from scratch for CDNA4:
 *
 *  - every data movement (pack, unpack, fused local/self copy) is a strided
 *    copy descriptor executed by one of three HIP kernels chosen at plan
 *    time: a vector 1-D copy, a batched linear-runs copy, or an LDS-tiled
 *    N-d transpose (coalesced reads AND writes, 64-wide wavefronts);
 *  - the subgroup exchange is grouped ncclSend/ncclRecv (RCCL over xGMI) on
 *    the caller's HIP stream — replacing MPI.Isend/Irecv / MPI.Alltoallv!
 *    (Transpositions.jl:419-428, 463-479);
 *  - the self block bypasses the staging buffers entirely (the reference
 *    stages it through recv_buf, :394-404 + :588-606 = 32 B/elem of HBM
 *    traffic; the fused kernel moves 16 B/elem), bit-identical results.
 *
 * Algorithm restated from (0-based, half-open ranges):
 *   split formula           data_ranges.jl:4-9
 *   axes/regions            data_ranges.jl:15-45
 *   to_local                Pencils.jl:579-587
 *   topology / subgroups    MPITopologies.jl:125-136, 208-251
 *   plan + peer blocks      Transpositions.jl:94-119, 282-344, 346-431,
 *                           489-536, 542-552
 *   pack/unpack semantics   Transpositions.jl:554-667 (copy_range! /
 *                           copy_permuted!, perm = permutation(Po)/
 *                           permutation(Pi) at :506)
 *
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 -shared -fPIC
 *        pencilhip.hip -I../../include -L/opt/rocm/lib -lrccl
 *        -o ../libpencilhip.so
 */

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <sched.h>

#include <algorithm>
#include <chrono>
#include <cstdarg>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "pencilhip.h"

/* ================= error handling ================================== */

static thread_local std::string g_err;

extern "C" const char *pa_last_error(void) { return g_err.c_str(); }

static pa_status fail(const char *fmt, ...)
{
    char buf[512];
    va_list ap;
    va_start(ap, fmt);
    vsnprintf(buf, sizeof buf, fmt, ap);
    va_end(ap);
    g_err = buf;
    return 1;
}

#define HIP_CHECK(x)                                                         \
    do {                                                                     \
        hipError_t err_ = (x);                                               \
        if (err_ != hipSuccess)                                              \
            return fail("HIP error %s at %s:%d: %s", hipGetErrorName(err_),  \
                        __FILE__, __LINE__, hipGetErrorString(err_));        \
    } while (0)

#define NCCL_CHECK(x)                                                        \
    do {                                                                     \
        ncclResult_t err_ = (x);                                             \
        if (err_ != ncclSuccess)                                             \
            return fail("RCCL error at %s:%d: %s", __FILE__, __LINE__,       \
                        ncclGetErrorString(err_));                           \
    } while (0)

/* ================= kernels ========================================= */

#define MAXND 8

struct DescDev {
    int nd;
    int64_t dims[MAXND], sstr[MAXND], dstr[MAXND];
    int64_t soff, doff, total;
};

/* 1-D contiguous copy, T = 4/8/16-byte word.  The N=1 identity-permutation
 * transpose collapses to this (a straight device copy).  Direct one-element-
 * per-thread mapping: measured 6268 GB/s vs 4729 for a 2048-block
 * grid-stride loop (tools/probe_copy.hip, profiles/r01_probe_copy.txt) —
 * 99.6% of the chip's float4-copy ceiling. */
template <typename T>
__global__ __launch_bounds__(256) void k_copy_1d(const T *__restrict__ src,
                                                 T *__restrict__ dst,
                                                 int64_t n)
{
    /* 2-D grid: the HSA dispatch packet's per-dimension work-item count is
     * 32-bit, so >2^32 threads (e.g. 2048^3 f64 = exactly 2^32 uint4) must
     * split across gridDim.y. */
    const int64_t b = (int64_t)blockIdx.y * gridDim.x + blockIdx.x;
    const int64_t i = b * blockDim.x + threadIdx.x;
    if (i < n) dst[i] = src[i];
}

/* Nontemporal variant for streams far past the 256 MiB Infinity Cache:
 * bypassing L2/LLC gains ~5% at >=8 GiB payloads but LOSES ~7% when the
 * working set is cache-resident (probe: P1D direct/nt-direct at 128 MiB /
 * 8 GiB / 32 GiB, profiles/r2_probe_kernels8/9.txt) — dispatch gates on
 * payload size. */
typedef unsigned int v4u_nt __attribute__((ext_vector_type(4)));
__global__ __launch_bounds__(256) void k_copy_1d_nt(
    const uint4 *__restrict__ src, uint4 *__restrict__ dst, int64_t n)
{
    const int64_t b = (int64_t)blockIdx.y * gridDim.x + blockIdx.x;
    const int64_t i = b * blockDim.x + threadIdx.x;
    if (i < n) {
        v4u_nt v = __builtin_nontemporal_load((const v4u_nt *)&src[i]);
        __builtin_nontemporal_store(v, (v4u_nt *)&dst[i]);
    }
}

/* Batched linear runs: axis 0 contiguous on both sides (coalesced); outer
 * axes strided.  Used for pack (strided window -> contiguous buffer when the
 * fastest axis survives) and its inverse.  Direct mapping (see k_copy_1d). */
template <typename T>
__global__ __launch_bounds__(256) void k_copy_linear(const T *__restrict__ src,
                                                     T *__restrict__ dst,
                                                     DescDev d)
{
    const int64_t run = d.dims[0];
    const int64_t b = (int64_t)blockIdx.y * gridDim.x + blockIdx.x;
    const int64_t idx = b * blockDim.x + threadIdx.x;
    if (idx >= d.total) return;
    int64_t o = idx / run;
    const int64_t i = idx - o * run;
    int64_t so = d.soff + i, doo = d.doff + i;
    for (int a = 1; a < d.nd; a++) {
        const int64_t j = o % d.dims[a];
        o /= d.dims[a];
        so += j * d.sstr[a];
        doo += j * d.dstr[a];
    }
    dst[doo] = src[so];
}

/* Generic gather/scatter (rare fallback: no contiguous axis on either side
 * after normalization, e.g. degenerate unit-extent windows). */
template <typename T>
__global__ __launch_bounds__(256) void k_copy_generic(const T *__restrict__ src,
                                                      T *__restrict__ dst,
                                                      DescDev d)
{
    int64_t idx = ((int64_t)blockIdx.y * gridDim.x + blockIdx.x) *
                      blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * gridDim.y * blockDim.x;
    for (; idx < d.total; idx += stride) {
        int64_t rem = idx, so = d.soff, doo = d.doff;
        for (int a = 0; a < d.nd; a++) {
            const int64_t j = rem % d.dims[a];
            rem /= d.dims[a];
            so += j * d.sstr[a];
            doo += j * d.dstr[a];
        }
        dst[doo] = src[so];
    }
}

/* LDS-tiled N-d transpose: src contiguous along axis 0, dst contiguous along
 * axis TA.  Reads coalesced along axis 0 into a padded LDS tile, barrier,
 * writes coalesced along axis TA.  TILE×TILE elements per workgroup,
 * 64×NROWS threads (wave64-shaped).  Remaining axes are batch.
 *
 * The unpack of a permuted pencil (copy_permuted!, Transpositions.jl:588-667)
 * is exactly this kernel; the reference's GPU path does a generic
 * permutedims! plus an extra temporary copy (:651-667 "TODO avoid
 * allocation") — here it is one kernel, no temporary. */
template <typename T, int TILE_I, int TILE_J, int NROWS, int JCHUNK>
__global__ __launch_bounds__(64 * NROWS) void k_transpose_tile(
    const T *__restrict__ src, T *__restrict__ dst, DescDev d, int ta,
    int64_t ntile_i, int64_t njchunk, int64_t nblocks)
{
    __shared__ T tile[TILE_J][TILE_I + 1];

    const int tx = threadIdx.x; /* 0..63  : fast axis */
    const int ty = threadIdx.y; /* 0..NROWS-1 */

    const int64_t bid = (int64_t)blockIdx.y * gridDim.x + blockIdx.x;
    if (bid >= nblocks) return;
    const int64_t t_i = bid % ntile_i;
    int64_t rest = bid / ntile_i;
    const int64_t chunk = rest % njchunk;
    int64_t batch = rest / njchunk;

    /* batch offsets over axes != 0, != ta */
    int64_t so_b = d.soff, do_b = d.doff;
    for (int a = 1; a < d.nd; a++) {
        if (a == ta) continue;
        const int64_t j = batch % d.dims[a];
        batch /= d.dims[a];
        so_b += j * d.sstr[a];
        do_b += j * d.dstr[a];
    }

    const int64_t i0 = t_i * TILE_I; /* along axis 0 */
    const int64_t ni = d.dims[0] - i0 < TILE_I ? d.dims[0] - i0 : TILE_I;

    /* sweep JCHUNK consecutive j-tiles with one workgroup: its reads walk
     * one contiguous src region and each of its TILE_I output rows is
     * written as a sequential stream (probe-measured +1.5% over one tile
     * per WG at the real 1024^3-permuted shape, profiles/probe_ab). */
    for (int jt = 0; jt < JCHUNK; jt++) {
        const int64_t j0 = (chunk * JCHUNK + jt) * TILE_J; /* along ta */
        if (j0 >= d.dims[ta]) break;
        const int64_t nj = d.dims[ta] - j0 < TILE_J ? d.dims[ta] - j0 : TILE_J;

        /* load: lanes sweep axis 0 (src-contiguous), rows sweep axis ta */
        {
            const int64_t base = so_b + i0 /* *1 */ + j0 * d.sstr[ta];
            for (int j = ty; j < nj; j += NROWS) {
                const int64_t row = base + (int64_t)j * d.sstr[ta];
                for (int i = tx; i < ni; i += 64)
                    tile[j][i] = src[row + i];
            }
        }
        __syncthreads();
        /* store: lanes sweep axis ta (dst-contiguous), rows sweep axis 0 */
        {
            const int64_t base = do_b + j0 /* *1 */ + i0 * d.dstr[0];
            for (int i = ty; i < ni; i += NROWS) {
                const int64_t row = base + (int64_t)i * d.dstr[0];
                for (int j = tx; j < nj; j += 64)
                    dst[row + j] = tile[j][i];
            }
        }
        __syncthreads(); /* before the next tile reuses the LDS */
    }
}

/* LDS-tiled transpose with VECTOR (16-B) STORES: 64(i) x 128(j) tile, each
 * lane storing a j-pair as one uint4 — 1024-B write bursts per instruction.
 * Interleaved-median probe (profiles/r2_probe_kernels*.txt): +1.5-1.8% over
 * the scalar 128x64 sweep at the 1024^3-permuted shape (write path is
 * L2-writeback-bound, wider bursts per instruction help where wider tiles
 * do not).  8-byte elements only; caller guarantees every dst stride
 * (except the unit ta axis) and the dst offset are EVEN and the dst base
 * pointer is 16-B aligned, so every vector store is aligned.  Tail j-tiles
 * (nj < TJ or odd) take a scalar epilogue. */
template <typename T, int TI, int TJ, int NROWS, int JCHUNK>
__global__ __launch_bounds__(64 * NROWS) void k_transpose_tile_vs(
    const T *__restrict__ src, T *__restrict__ dst, DescDev d, int ta,
    int64_t ntile_i, int64_t njchunk, int64_t nblocks)
{
    static_assert(sizeof(T) == 8, "vector-store tile is 8-byte-elem only");
    /* TRANSPOSED LDS layout (tile[i][j], +2 pad): the store phase reads
     * 16 B of consecutive j per lane — one wide conflict-free LDS read.
     * The original tile[j][i] layout measured 0.6 LDS-conflict cycles per
     * active cycle (lane-pair row-strided reads); this layout removes them
     * and wins the interleaved A/B by a small margin (probe vs2 p2,
     * profiles/r2_probe_vs2*.txt). */
    __shared__ T tile[TI][TJ + 2];

    const int tx = threadIdx.x;
    const int ty = threadIdx.y;

    const int64_t bid = (int64_t)blockIdx.y * gridDim.x + blockIdx.x;
    if (bid >= nblocks) return;
    const int64_t t_i = bid % ntile_i;
    int64_t rest = bid / ntile_i;
    const int64_t chunk = rest % njchunk;
    int64_t batch = rest / njchunk;

    int64_t so_b = d.soff, do_b = d.doff;
    for (int a = 1; a < d.nd; a++) {
        if (a == ta) continue;
        const int64_t j = batch % d.dims[a];
        batch /= d.dims[a];
        so_b += j * d.sstr[a];
        do_b += j * d.dstr[a];
    }

    const int64_t i0 = t_i * TI;
    const int64_t ni = d.dims[0] - i0 < TI ? d.dims[0] - i0 : TI;

    for (int jt = 0; jt < JCHUNK; jt++) {
        const int64_t j0 = (chunk * JCHUNK + jt) * TJ;
        if (j0 >= d.dims[ta]) break;
        const int64_t nj = d.dims[ta] - j0 < TJ ? d.dims[ta] - j0 : TJ;

        {
            const int64_t base = so_b + i0 + j0 * d.sstr[ta];
            for (int j = ty; j < nj; j += NROWS) {
                const int64_t row = base + (int64_t)j * d.sstr[ta];
                for (int i = tx; i < ni; i += 64)
                    tile[i][j] = src[row + i];
            }
        }
        __syncthreads();
        {
            const int64_t base = do_b + j0 + i0 * d.dstr[0];
            const int64_t njv = nj & ~(int64_t)1; /* even part */
            for (int i = ty; i < ni; i += NROWS) {
                T *row = dst + base + (int64_t)i * d.dstr[0];
                for (int j2 = 2 * tx; j2 < njv; j2 += 128) {
                    uint4 q;
                    ((T *)&q)[0] = tile[i][j2];
                    ((T *)&q)[1] = tile[i][j2 + 1];
                    *(uint4 *)&row[j2] = q;
                }
                if ((nj & 1) && tx == 0) /* odd tail element */
                    row[nj - 1] = tile[i][nj - 1];
            }
        }
        __syncthreads();
    }
}

/* ================= descriptor normalization & dispatch ============= */

struct CopyDescH {
    int nd = 0;
    int64_t dims[MAXND] = {0}, sstr[MAXND] = {0}, dstr[MAXND] = {0};
    int64_t soff = 0, doff = 0;
    int64_t total = 0;
};

static CopyDescH normalize_desc(int nd, const int64_t *dims,
                                const int64_t *sstr, int64_t soff,
                                const int64_t *dstr, int64_t doff)
{
    struct Axis { int64_t dim, ss, ds; };
    std::vector<Axis> ax;
    for (int i = 0; i < nd; i++)
        if (dims[i] != 1) ax.push_back({dims[i], sstr[i], dstr[i]});
    if (ax.empty()) ax.push_back({1, 1, 1});
    std::sort(ax.begin(), ax.end(),
              [](const Axis &a, const Axis &b) { return a.ss < b.ss; });
    std::vector<Axis> m{ax[0]};
    for (size_t i = 1; i < ax.size(); i++) {
        Axis &p = m.back();
        if (ax[i].ss == p.ss * p.dim && ax[i].ds == p.ds * p.dim)
            p.dim *= ax[i].dim;
        else
            m.push_back(ax[i]);
    }
    CopyDescH out;
    out.nd = (int)m.size();
    out.total = 1;
    for (int i = 0; i < out.nd; i++) {
        out.dims[i] = m[i].dim;
        out.sstr[i] = m[i].ss;
        out.dstr[i] = m[i].ds;
        out.total *= m[i].dim;
    }
    out.soff = soff;
    out.doff = doff;
    return out;
}

/* Try to reinterpret a linear-runs descriptor in W-byte words (W >= esz). */
static bool word_scale(const CopyDescH &d, int64_t esz, int64_t W,
                       CopyDescH *out)
{
    if ((d.dims[0] * esz) % W || (d.soff * esz) % W || (d.doff * esz) % W)
        return false;
    for (int a = 1; a < d.nd; a++)
        if ((d.sstr[a] * esz) % W || (d.dstr[a] * esz) % W) return false;
    *out = d;
    out->dims[0] = d.dims[0] * esz / W;
    out->soff = d.soff * esz / W;
    out->doff = d.doff * esz / W;
    for (int a = 1; a < d.nd; a++) {
        out->sstr[a] = d.sstr[a] * esz / W;
        out->dstr[a] = d.dstr[a] * esz / W;
    }
    out->total = out->dims[0];
    for (int a = 1; a < out->nd; a++) out->total *= out->dims[a];
    return true;
}

static DescDev to_dev(const CopyDescH &d)
{
    DescDev o;
    o.nd = d.nd;
    for (int i = 0; i < MAXND; i++) {
        o.dims[i] = i < d.nd ? d.dims[i] : 1;
        o.sstr[i] = i < d.nd ? d.sstr[i] : 0;
        o.dstr[i] = i < d.nd ? d.dstr[i] : 0;
    }
    o.soff = d.soff;
    o.doff = d.doff;
    o.total = d.total;
    return o;
}

/* Exact one-element-per-thread grid (probe-measured fastest for the
 * memory-bound copies: the dispatcher streams blocks, no software loop). */
static int64_t grid_exact(int64_t work_items, int per_block)
{
    int64_t blocks = (work_items + per_block - 1) / per_block;
    if (blocks < 1) blocks = 1;
    return blocks;
}

/* Split a block count into a 2-D grid: the HSA dispatch limit is 2^32-1
 * WORK-ITEMS per grid dimension, so gridDim.x is capped at that / threads
 * and the remainder goes to gridDim.y. */
static pa_status grid2d(int64_t blocks, int threads_per_block, dim3 *out)
{
    const int64_t max_x = 0xFFFFFFFFll / threads_per_block;
    int64_t gx = blocks < max_x ? blocks : max_x;
    int64_t gy = (blocks + gx - 1) / gx;
    if (gy > 0xFFFFFFFFll) return fail("grid too large");
    *out = dim3((uint32_t)gx, (uint32_t)gy);
    return 0;
}

static int grid_for(int64_t work_items, int per_block)
{
    int64_t blocks = grid_exact(work_items, per_block);
    const int64_t cap = 8192; /* generic fallback only */
    if (blocks > cap) blocks = cap;
    return (int)blocks;
}

static pa_status launch_desc(const CopyDescH &dn, int64_t esz,
                             const void *src, void *dst, hipStream_t stream)
{
    if (dn.total == 0) return 0;
    const char *s = (const char *)src;
    char *d = (char *)dst;

    const bool lin = (dn.sstr[0] == 1 && dn.dstr[0] == 1);
    if (lin) {
        CopyDescH w;
        int64_t W = 0;
        for (int64_t cand : {16, 8, 4})
            if (cand >= esz || (esz % cand) == 0)
                if (word_scale(dn, esz, cand, &w)) { W = cand; break; }
        if (!W) { /* odd element size: fall back to bytes */
            if (!word_scale(dn, esz, 1, &w)) return fail("word_scale(1)");
            W = 1;
        }
        if (w.nd == 1) {
            dim3 blocks;
            pa_status gst = grid2d(grid_exact(w.total, 256), 256, &blocks);
            if (gst) return gst;
            if (W == 16 && w.total * 16 >= (512ll << 20))
                hipLaunchKernelGGL(k_copy_1d_nt, blocks, dim3(256),
                                   0, stream, (const uint4 *)s + w.soff,
                                   (uint4 *)d + w.doff, w.total);
            else if (W == 16)
                hipLaunchKernelGGL(k_copy_1d<uint4>, blocks, dim3(256),
                                   0, stream, (const uint4 *)s + w.soff,
                                   (uint4 *)d + w.doff, w.total);
            else if (W == 8)
                hipLaunchKernelGGL(k_copy_1d<uint64_t>, blocks,
                                   dim3(256), 0, stream,
                                   (const uint64_t *)s + w.soff,
                                   (uint64_t *)d + w.doff, w.total);
            else if (W == 4)
                hipLaunchKernelGGL(k_copy_1d<uint32_t>, blocks,
                                   dim3(256), 0, stream,
                                   (const uint32_t *)s + w.soff,
                                   (uint32_t *)d + w.doff, w.total);
            else
                hipLaunchKernelGGL(k_copy_1d<uint8_t>, blocks, dim3(256),
                                   0, stream, (const uint8_t *)s + w.soff,
                                   (uint8_t *)d + w.doff, w.total);
        } else {
            DescDev dd = to_dev(w);
            dim3 blocks;
            pa_status gst = grid2d(grid_exact(w.total, 256), 256, &blocks);
            if (gst) return gst;
            if (W == 16)
                hipLaunchKernelGGL(k_copy_linear<uint4>, blocks,
                                   dim3(256), 0, stream, (const uint4 *)s,
                                   (uint4 *)d, dd);
            else if (W == 8)
                hipLaunchKernelGGL(k_copy_linear<uint64_t>, blocks,
                                   dim3(256), 0, stream, (const uint64_t *)s,
                                   (uint64_t *)d, dd);
            else if (W == 4)
                hipLaunchKernelGGL(k_copy_linear<uint32_t>, blocks,
                                   dim3(256), 0, stream, (const uint32_t *)s,
                                   (uint32_t *)d, dd);
            else
                hipLaunchKernelGGL(k_copy_linear<uint8_t>, blocks,
                                   dim3(256), 0, stream, (const uint8_t *)s,
                                   (uint8_t *)d, dd);
        }
        HIP_CHECK(hipGetLastError());
        return 0;
    }

    /* transpose case: src-contiguous axis 0, find dst-contiguous axis */
    int ta = -1;
    if (dn.sstr[0] == 1)
        for (int a = 1; a < dn.nd; a++)
            if (dn.dstr[a] == 1) { ta = a; break; }

    if (ta > 0 && (esz == 4 || esz == 8 || esz == 16)) {
        DescDev dd = to_dev(dn);
        int64_t nbatch = 1;
        for (int a = 1; a < dn.nd; a++)
            if (a != ta) nbatch *= dn.dims[a];
        /* tile shapes probe-measured on MI355X (profiles/r01_probe_copy*,
         * r2_probe_kernels*): 8-B elements prefer the 64x128 vector-store
         * tile (+1.5-1.8% median) when alignment permits; otherwise (and
         * for 4-B) the scalar 128(i)x64(j) r16 sweep; 16-B: 32x32. */
        bool vec_ok = false;
        if (esz == 8) {
            vec_ok = ((dn.doff & 1) == 0) && ((dn.dstr[0] & 1) == 0) &&
                     (((uintptr_t)d & 15) == 0);
            for (int a2 = 1; a2 < dn.nd && vec_ok; a2++)
                if (a2 != ta && (dn.dstr[a2] & 1)) vec_ok = false;
        }
        if (vec_ok) {
            constexpr int TI = 64, TJ = 128, NR = 16;
            const int64_t nti = (dn.dims[0] + TI - 1) / TI;
            const int64_t ntj = (dn.dims[ta] + TJ - 1) / TJ;
            int jc = 1; /* same occupancy/tail-balance thresholds as below */
            if (ntj * nti * nbatch >= 32 * 4096 && ntj >= 256)
                jc = 32;
            else if (ntj * nti * nbatch >= 8 * 4096 && ntj >= 64)
                jc = 8;
            const int64_t njc = (ntj + jc - 1) / jc;
            const int64_t nblocks = nti * njc * nbatch;
            dim3 grid;
            pa_status gst = grid2d(nblocks, 64 * NR, &grid);
            if (gst) return gst;
#define LAUNCH_VS(JCV)                                                       \
    hipLaunchKernelGGL((k_transpose_tile_vs<uint64_t, TI, TJ, NR, JCV>),     \
                       grid, dim3(64, NR), 0, stream, (const uint64_t *)s,   \
                       (uint64_t *)d, dd, ta, nti, njc, nblocks)
            if (jc == 32)
                LAUNCH_VS(32);
            else if (jc == 8)
                LAUNCH_VS(8);
            else
                LAUNCH_VS(1);
#undef LAUNCH_VS
            HIP_CHECK(hipGetLastError());
            return 0;
        }
        if (esz == 8 || esz == 4) {
            constexpr int TI = 128, TJ = 64, NR = 16;
            const int64_t nti = (dn.dims[0] + TI - 1) / TI;
            const int64_t ntj = (dn.dims[ta] + TJ - 1) / TJ;
            /* adaptive sweep depth: long sweeps win on tall-skinny shapes
             * (+1.5% A/B) but must keep >= ~4k workgroups for occupancy and
             * tail balance on small/batched shapes */
            int jc = 1;
            if (ntj * nti * nbatch >= 32 * 4096 && ntj >= 256)
                jc = 32;
            else if (ntj * nti * nbatch >= 8 * 4096 && ntj >= 64)
                jc = 8;
            const int64_t njc = (ntj + jc - 1) / jc;
            const int64_t nblocks = nti * njc * nbatch;
            dim3 grid;
            pa_status gst = grid2d(nblocks, 64 * NR, &grid);
            if (gst) return gst;
#define LAUNCH_TT(T, JCV)                                                        hipLaunchKernelGGL((k_transpose_tile<T, TI, TJ, NR, JCV>), grid,                                dim3(64, NR), 0, stream, (const T *)s, (T *)d, dd,                           ta, nti, njc, nblocks)
            if (esz == 8) {
                if (jc == 32)
                    LAUNCH_TT(uint64_t, 32);
                else if (jc == 8)
                    LAUNCH_TT(uint64_t, 8);
                else
                    LAUNCH_TT(uint64_t, 1);
            } else {
                if (jc == 32)
                    LAUNCH_TT(uint32_t, 32);
                else if (jc == 8)
                    LAUNCH_TT(uint32_t, 8);
                else
                    LAUNCH_TT(uint32_t, 1);
            }
#undef LAUNCH_TT
        } else {
            constexpr int TI = 32, TJ = 32, NR = 8, JC = 1;
            const int64_t nti = (dn.dims[0] + TI - 1) / TI;
            const int64_t ntj = (dn.dims[ta] + TJ - 1) / TJ;
            const int64_t nblocks = nti * ntj * nbatch;
            dim3 grid;
            pa_status gst = grid2d(nblocks, 64 * NR, &grid);
            if (gst) return gst;
            hipLaunchKernelGGL((k_transpose_tile<uint4, TI, TJ, NR, JC>),
                               grid, dim3(64, NR), 0, stream,
                               (const uint4 *)s, (uint4 *)d, dd, ta, nti, ntj,
                               nblocks);
        }
        HIP_CHECK(hipGetLastError());
        return 0;
    }

    /* generic fallback */
    if (esz == 16 || esz == 8 || esz == 4) {
        DescDev dd = to_dev(dn);
        const dim3 blocks(grid_for(dn.total, 256)); /* grid-stride kernel */
        if (esz == 16)
            hipLaunchKernelGGL(k_copy_generic<uint4>, blocks, dim3(256),
                               0, stream, (const uint4 *)s, (uint4 *)d, dd);
        else if (esz == 8)
            hipLaunchKernelGGL(k_copy_generic<uint64_t>, blocks,
                               dim3(256), 0, stream, (const uint64_t *)s,
                               (uint64_t *)d, dd);
        else
            hipLaunchKernelGGL(k_copy_generic<uint32_t>, blocks,
                               dim3(256), 0, stream, (const uint32_t *)s,
                               (uint32_t *)d, dd);
        HIP_CHECK(hipGetLastError());
        return 0;
    }

    /* any other element size (the reference allows arbitrary isbits
     * element types): byte-ify — prepend a stride-1 axis of esz bytes,
     * which makes the descriptor linear-copyable at byte granularity
     * (slow for tiny elements, always correct) */
    {
        if (dn.nd + 1 > MAXND) return fail("too many dims to byte-ify");
        int64_t dims[MAXND + 1], ss[MAXND + 1], ds[MAXND + 1];
        dims[0] = esz;
        ss[0] = 1;
        ds[0] = 1;
        for (int a = 0; a < dn.nd; a++) {
            dims[a + 1] = dn.dims[a];
            ss[a + 1] = dn.sstr[a] * esz;
            ds[a + 1] = dn.dstr[a] * esz;
        }
        CopyDescH b = normalize_desc(dn.nd + 1, dims, ss, dn.soff * esz, ds,
                                     dn.doff * esz);
        return launch_desc(b, 1, src, dst, stream);
    }
}

/* ================= metadata ======================================== */

struct Range { int64_t lo, hi; };

struct pa_topology {
    int m;
    std::vector<int64_t> dims;
    int64_t nranks;
};

struct pa_pencil {
    pa_topology topo; /* copied: a pencil owns its grid description */
    int n;
    std::vector<int64_t> size_global;
    std::vector<int32_t> decomp; /* length m */
    std::vector<int32_t> perm;   /* length n; perm[i] = logical dim at mem i */
};

static void split_range(int64_t c, int64_t P, int64_t N, Range *r)
{
    r->lo = (N * c) / P; /* data_ranges.jl:4-9 */
    r->hi = (N * (c + 1)) / P;
}

static void cart_coords(const pa_topology &t, int rank, int64_t *coords)
{
    int64_t rem = rank;
    for (int i = 0; i < t.m; i++) {
        int64_t stride = 1;
        for (int j = i + 1; j < t.m; j++) stride *= t.dims[j];
        coords[i] = rem / stride;
        rem %= stride;
    }
}

static int cart_rank(const pa_topology &t, const int64_t *coords)
{
    int64_t r = 0;
    for (int i = 0; i < t.m; i++) r = r * t.dims[i] + coords[i];
    return (int)r;
}

static void axes_for_coords(const pa_pencil &p, const int64_t *coords,
                            Range *out)
{
    for (int d = 0; d < p.n; d++) {
        out[d].lo = 0;
        out[d].hi = p.size_global[d];
    }
    for (int j = 0; j < p.topo.m; j++)
        split_range(coords[j], p.topo.dims[j], p.size_global[p.decomp[j]],
                    &out[p.decomp[j]]);
}

static void axes_for_rank(const pa_pencil &p, int rank, Range *out)
{
    int64_t coords[MAXND];
    cart_coords(p.topo, rank, coords);
    axes_for_coords(p, coords, out);
}

static void range_intersect(const Range &a, const Range &b, Range *out)
{
    out->lo = std::max(a.lo, b.lo);
    out->hi = std::max(out->lo, std::min(a.hi, b.hi));
}

static int64_t region_nelem(int n, const Range *r)
{
    int64_t v = 1;
    for (int d = 0; d < n; d++) v *= (r[d].hi - r[d].lo);
    return v;
}

/* ================= comm ============================================ */

struct pa_comm {
    ncclComm_t comm;
    int nranks, rank;
};

/* ================= plan ============================================ */

struct PeerBlockC {
    int k, grank;
    int64_t send_off, recv_off, send_n, recv_n;
    bool has_pack = false, has_unpack = false;
    CopyDescH pack, unpack;
    /* chunked-exchange support: the send/recv blocks viewed as
     * (rowelems x outer) with outer = the last raw (Pi-mem-order + extras)
     * axis — identical split on sender and receiver by construction. */
    int64_t send_outer = 1, send_rowelems = 0;
    int64_t recv_outer = 1, recv_rowelems = 0;
    CopyDescH unpack_raw; /* un-normalized: nd = n+E, axis nd-1 = outer */
    bool has_unpack_raw = false;
};

struct pa_plan {
    pa_pencil Pi, Po;
    int rank;
    int E;
    std::vector<int64_t> extra;
    int64_t esz;
    int R;    /* -1 = same decomposition */
    int P;    /* subgroup size */
    int myk;  /* my coordinate along R */
    std::vector<PeerBlockC> peers;
    bool has_local = false;
    CopyDescH local;
    /* aliased (in-place) mode: self block staged through the recv tail
     * (Transpositions.jl:250-264, :394-404) */
    bool aliased = false;
    bool has_self = false;
    CopyDescH self_pack, self_unpack;
    int64_t send_total = 0, recv_total = 0; /* elements (remote blocks) */
    void *send_buf = nullptr, *recv_buf = nullptr;
    bool own_bufs = false;
    pa_comm *comm = nullptr;
    /* two-stream overlap: the exchange runs on comm_stream while the fused
     * local copy runs on the caller's stream (the reference overlaps the
     * local block with the exchange the same way: it copies it first and
     * unpacks it while recvs are in flight, Transpositions.jl:510-517). */
    hipStream_t comm_stream = nullptr;
    hipEvent_t ev_pack = nullptr, ev_comm = nullptr;
    /* >1: split the exchange into this many chunks and unpack each chunk
     * while later chunks are still in flight (the reference's Waitany
     * overlap, Transpositions.jl:510-517).  From PENCILHIP_EXCHANGE_CHUNKS;
     * default 1 (single grouped exchange). */
    int chunks = 1;
    std::vector<hipEvent_t> ev_chunks;
    /* per-stage timing (the TimerOutputs analogue, Transpositions.jl:
     * 173-177,327,337): opt-in HIP events with timing around pack / local /
     * exchange / unpack of the LAST execute.  tm[0..1] pack start/end and
     * tm[2] local end on the caller's stream; tm[3..4] exchange start/end on
     * the comm stream; tm[5..6] unpack start/end on the caller's stream. */
    bool timing = false;
    hipEvent_t tm[7] = {nullptr, nullptr, nullptr, nullptr,
                        nullptr, nullptr, nullptr};
    bool tm_valid[7] = {false, false, false, false, false, false, false};
};

static int64_t prod_extra(const pa_plan &pl)
{
    int64_t v = 1;
    for (auto e : pl.extra) v *= e;
    return v;
}

/* parent memory dims (memory-order local sizes + extra) and column-major
 * strides (axis 0 fastest — the Julia parent, arrays.jl:134-138). */
static void parent_strides(const pa_pencil &p, int rank,
                           const std::vector<int64_t> &extra, int64_t *mem,
                           int64_t *st, int *knd)
{
    Range ax[MAXND];
    axes_for_rank(p, rank, ax);
    int k = 0;
    for (int i = 0; i < p.n; i++, k++)
        mem[k] = ax[p.perm[i]].hi - ax[p.perm[i]].lo;
    for (auto e : extra) mem[k++] = e;
    int64_t acc = 1;
    for (int i = 0; i < k; i++) {
        st[i] = acc;
        acc *= mem[i];
    }
    *knd = k;
}

/* window of a global region in the parent of pencil p (axes in p's memory
 * order + extras): dims/strides/offset (to_local, Pencils.jl:579-587). */
static void window_desc(const pa_pencil &p, int rank,
                        const std::vector<int64_t> &extra, const Range *region,
                        int64_t *dims, int64_t *str, int64_t *off, int *knd)
{
    Range ax[MAXND];
    axes_for_rank(p, rank, ax);
    int64_t mem[2 * MAXND], pst[2 * MAXND];
    int k;
    parent_strides(p, rank, extra, mem, pst, &k);
    int64_t o = 0;
    for (int i = 0; i < p.n; i++) {
        const int d = p.perm[i];
        dims[i] = region[d].hi - region[d].lo;
        o += (region[d].lo - ax[d].lo) * pst[i];
    }
    for (size_t e = 0; e < extra.size(); e++) dims[p.n + e] = extra[e];
    for (int i = 0; i < k; i++) str[i] = pst[i];
    *off = o;
    *knd = k;
}

/* buffer->dest unpack descriptor: buffer axes = Pi memory order (+extras),
 * column-major with given strides; dest axis i' reads buffer axis
 * inv(perm_i)[perm_o[i']] (the relative permutation, Transpositions.jl:506).
 */
static CopyDescH unpack_desc_raw_out(const pa_plan &pl, const Range *grange,
                                     const int64_t *bufdims,
                                     const int64_t *bufstr, int64_t bufoff,
                                     CopyDescH *raw_out);

static CopyDescH unpack_desc(const pa_plan &pl, const Range *grange,
                             const int64_t *bufdims, const int64_t *bufstr,
                             int64_t bufoff)
{
    return unpack_desc_raw_out(pl, grange, bufdims, bufstr, bufoff, nullptr);
}

static CopyDescH unpack_desc_raw_out(const pa_plan &pl, const Range *grange,
                                     const int64_t *bufdims,
                                     const int64_t *bufstr, int64_t bufoff,
                                     CopyDescH *raw_out)
{
    const pa_pencil &Po = pl.Po;
    const pa_pencil &Pi = pl.Pi;
    const int n = Pi.n;
    const int E = pl.E;

    Range axo[MAXND];
    axes_for_rank(Po, pl.rank, axo);
    int64_t mem_o[2 * MAXND], pst_o[2 * MAXND];
    int k;
    parent_strides(Po, pl.rank, pl.extra, mem_o, pst_o, &k);

    int32_t ipi[MAXND];
    for (int i = 0; i < n; i++) ipi[Pi.perm[i]] = i;

    int64_t dstr[2 * MAXND], doff = 0;
    for (int ip = 0; ip < n; ip++) {
        const int d = Po.perm[ip];
        dstr[ipi[d]] = pst_o[ip];
        doff += (grange[d].lo - axo[d].lo) * pst_o[ip];
    }
    for (int e = 0; e < E; e++) dstr[n + e] = pst_o[n + e];

    if (raw_out) {
        raw_out->nd = n + E;
        for (int i = 0; i < n + E; i++) {
            raw_out->dims[i] = bufdims[i];
            raw_out->sstr[i] = bufstr[i];
            raw_out->dstr[i] = dstr[i];
        }
        raw_out->soff = bufoff;
        raw_out->doff = doff;
        raw_out->total = 1;
        for (int i = 0; i < n + E; i++) raw_out->total *= bufdims[i];
    }
    return normalize_desc(n + E, bufdims, bufstr, bufoff, dstr, doff);
}


/* staged self block (aliased mode): pack src window -> recv tail, unpack
 * recv tail -> dst window */
static void staged_self_descs(pa_plan *pl, const Range *sr, const Range *rr,
                              int64_t recv_off)
{
    const pa_pencil &Pi = pl->Pi;
    const int n = Pi.n, e = pl->E;
    int64_t dims[2 * MAXND], str[2 * MAXND], off;
    int kk;
    window_desc(Pi, pl->rank, pl->extra, sr, dims, str, &off, &kk);
    int64_t cstr[2 * MAXND], acc = 1;
    for (int i = 0; i < kk; i++) {
        cstr[i] = acc;
        acc *= dims[i];
    }
    pl->self_pack = normalize_desc(kk, dims, str, off, cstr, recv_off);
    int64_t bdims[2 * MAXND];
    for (int i = 0; i < n; i++)
        bdims[i] = rr[Pi.perm[i]].hi - rr[Pi.perm[i]].lo;
    for (int e2 = 0; e2 < e; e2++) bdims[n + e2] = pl->extra[e2];
    int64_t bstr[2 * MAXND];
    acc = 1;
    for (int i = 0; i < n + e; i++) {
        bstr[i] = acc;
        acc *= bdims[i];
    }
    pl->self_unpack = unpack_desc(*pl, rr, bdims, bstr, recv_off);
    pl->has_self = true;
}


/* chunk of a raw (un-normalized) unpack descriptor: outer rows [lo, hi) of
 * its last axis */
static CopyDescH chunk_of_raw(const CopyDescH &raw, int64_t lo, int64_t hi)
{
    const int last = raw.nd - 1;
    int64_t dims[2 * MAXND], sstr[2 * MAXND], dstr[2 * MAXND];
    for (int i = 0; i < raw.nd; i++) {
        dims[i] = raw.dims[i];
        sstr[i] = raw.sstr[i];
        dstr[i] = raw.dstr[i];
    }
    dims[last] = hi - lo;
    return normalize_desc(raw.nd, dims, sstr, raw.soff + lo * sstr[last],
                          dstr, raw.doff + lo * dstr[last]);
}

static inline int64_t chunk_lo(int64_t outer, int c, int C)
{
    return outer * c / C;
}

extern "C" {

/* ---- topology ----------------------------------------------------- */

pa_status pa_topology_create(int m, const int64_t *pdims, pa_topology **out)
{
    if (m <= 0 || m > MAXND) return fail("invalid topology ndims %d", m);
    auto *t = new pa_topology;
    t->m = m;
    t->nranks = 1;
    for (int i = 0; i < m; i++) {
        if (pdims[i] <= 0) { delete t; return fail("invalid pdims"); }
        t->dims.push_back(pdims[i]);
        t->nranks *= pdims[i];
    }
    *out = t;
    return 0;
}

void pa_topology_destroy(pa_topology *t) { delete t; }
int pa_topology_nranks(const pa_topology *t) { return (int)t->nranks; }

pa_status pa_topology_coords(const pa_topology *t, int rank, int64_t *coords)
{
    if (rank < 0 || rank >= t->nranks) return fail("rank out of range");
    cart_coords(*t, rank, coords);
    return 0;
}

/* ---- pencil ------------------------------------------------------- */

pa_status pa_pencil_create(pa_topology *t, int n, const int64_t *size_global,
                           const int32_t *decomp_dims, const int32_t *perm,
                           pa_pencil **out)
{
    if (n <= 0 || n + 2 > MAXND) return fail("invalid ndims %d", n);
    if (t->m > n) return fail("M (%d) cannot exceed N (%d)", t->m, n);
    auto *p = new pa_pencil;
    p->topo = *t;
    p->n = n;
    p->size_global.assign(size_global, size_global + n);
    p->decomp.assign(decomp_dims, decomp_dims + t->m);
    /* _check_selected_dimensions (Pencils.jl:393-406) */
    for (int j = 0; j < t->m; j++) {
        if (p->decomp[j] < 0 || p->decomp[j] >= n) {
            delete p;
            return fail("decomp dim %d out of range", p->decomp[j]);
        }
        for (int j2 = 0; j2 < j; j2++)
            if (p->decomp[j2] == p->decomp[j]) {
                delete p;
                return fail("repeated decomp dim %d", p->decomp[j]);
            }
    }
    if (perm) {
        p->perm.assign(perm, perm + n);
        std::vector<int> seen(n, 0);
        for (int i = 0; i < n; i++) {
            if (p->perm[i] < 0 || p->perm[i] >= n || seen[p->perm[i]]++) {
                delete p;
                return fail("invalid permutation");
            }
        }
    } else {
        for (int i = 0; i < n; i++) p->perm.push_back(i);
    }
    *out = p;
    return 0;
}

void pa_pencil_destroy(pa_pencil *p) { delete p; }

pa_status pa_pencil_range_local(const pa_pencil *p, int rank, int memory_order,
                                int64_t *lo, int64_t *hi)
{
    if (rank < 0 || rank >= p->topo.nranks) return fail("rank out of range");
    Range ax[MAXND];
    axes_for_rank(*p, rank, ax);
    for (int i = 0; i < p->n; i++) {
        const int d = memory_order ? p->perm[i] : i;
        lo[i] = ax[d].lo;
        hi[i] = ax[d].hi;
    }
    return 0;
}

pa_status pa_pencil_size_local(const pa_pencil *p, int rank, int memory_order,
                               int64_t *out)
{
    int64_t lo[MAXND], hi[MAXND];
    pa_status st = pa_pencil_range_local(p, rank, memory_order, lo, hi);
    if (st) return st;
    for (int i = 0; i < p->n; i++) out[i] = hi[i] - lo[i];
    return 0;
}

int64_t pa_pencil_length_local(const pa_pencil *p, int rank)
{
    Range ax[MAXND];
    axes_for_rank(*p, rank, ax);
    return region_nelem(p->n, ax);
}

/* ---- comm --------------------------------------------------------- */

int pa_unique_id_size(void) { return (int)sizeof(ncclUniqueId); }

pa_status pa_get_unique_id(char *id)
{
    NCCL_CHECK(ncclGetUniqueId((ncclUniqueId *)id));
    return 0;
}

pa_status pa_comm_create(const char *id, int nranks, int rank, pa_comm **out)
{
    auto *c = new pa_comm;
    c->nranks = nranks;
    c->rank = rank;
    ncclUniqueId uid;
    memcpy(&uid, id, sizeof uid);
    /* Non-blocking init with a bounded wait: a blocking ncclCommInitRank
     * hangs forever if any subgroup rank is missing or disagrees — in an
     * unattended multi-process run that is an undiagnosable timeout.
     * Timeout via PENCILHIP_COMM_TIMEOUT_S (default 180 s). */
    ncclConfig_t cfg = NCCL_CONFIG_INITIALIZER;
    cfg.blocking = 0;
    ncclResult_t r = ncclCommInitRankConfig(&c->comm, nranks, uid, rank, &cfg);
    if (r != ncclSuccess && r != ncclInProgress) {
        delete c;
        return fail("ncclCommInitRank: %s", ncclGetErrorString(r));
    }
    double timeout_s = 180.0;
    if (const char *te = getenv("PENCILHIP_COMM_TIMEOUT_S")) {
        double v = atof(te);
        if (v > 0) timeout_s = v;
    }
    const auto t0 = std::chrono::steady_clock::now();
    ncclResult_t st = ncclInProgress;
    while (true) {
        ncclResult_t qr = ncclCommGetAsyncError(c->comm, &st);
        if (qr != ncclSuccess) {
            (void)ncclCommAbort(c->comm);
            delete c;
            return fail("ncclCommGetAsyncError: %s", ncclGetErrorString(qr));
        }
        if (st == ncclSuccess) break;
        if (st != ncclInProgress) {
            (void)ncclCommAbort(c->comm);
            delete c;
            return fail("ncclCommInitRank (async): %s",
                        ncclGetErrorString(st));
        }
        const double waited = std::chrono::duration<double>(
            std::chrono::steady_clock::now() - t0).count();
        if (waited > timeout_s) {
            (void)ncclCommAbort(c->comm);
            delete c;
            return fail("ncclCommInitRank timed out after %.0f s (subgroup "
                        "nranks=%d rank=%d): a peer is missing or "
                        "disagrees on the unique id", timeout_s, nranks,
                        rank);
        }
        sched_yield();
    }
    *out = c;
    return 0;
}

void pa_comm_destroy(pa_comm *c)
{
    if (!c) return;
    ncclCommDestroy(c->comm);
    delete c;
}

/* ---- plan --------------------------------------------------------- */

pa_status pa_plan_create(const pa_pencil *pin, const pa_pencil *pout,
                         int64_t elem_size, int e, const int64_t *extra_dims,
                         int rank, int flags, pa_plan **out)
{
    /* assert_compatible (Transpositions.jl:182-199) */
    if (pin->topo.m != pout->topo.m || pin->topo.dims != pout->topo.dims)
        return fail("pencil topologies must be the same.");
    if (pin->n != pout->n || pin->size_global != pout->size_global)
        return fail("global data sizes must be the same between different "
                    "pencil configurations.");
    int ndiff = 0, R = -1;
    for (int j = 0; j < pin->topo.m; j++)
        if (pin->decomp[j] != pout->decomp[j]) {
            ndiff++;
            if (R < 0) R = j;
        }
    if (ndiff > 1)
        return fail("pencil decompositions must differ in at most one "
                    "dimension.");
    if (rank < 0 || rank >= pin->topo.nranks) return fail("rank out of range");
    if (elem_size <= 0) return fail("invalid elem_size");
    if (pin->n + e > MAXND)
        return fail("too many dims: N+E must be <= %d", MAXND);

    auto *pl = new pa_plan;
    pl->Pi = *pin;
    pl->Po = *pout;
    pl->rank = rank;
    pl->E = e;
    if (e) pl->extra.assign(extra_dims, extra_dims + e);
    pl->esz = elem_size;
    pl->R = R;
    pl->aliased = (flags & 1) != 0;
    if (const char *ce = getenv("PENCILHIP_EXCHANGE_CHUNKS")) {
        int c = atoi(ce);
        if (c >= 1 && c <= 64) pl->chunks = c;
    }

    const int n = pin->n;
    const int64_t pex = prod_extra(*pl);

    if (R < 0) {
        pl->P = 1;
        pl->myk = 0;
        /* local path (transpose_impl!(::Nothing), Transpositions.jl:214-271;
         * aliased variant stages through recv_buf, :250-264) */
        Range region[MAXND];
        axes_for_rank(*pin, rank, region);
        if (region_nelem(n, region) > 0) {
            if (pl->aliased) {
                staged_self_descs(pl, region, region, 0);
                pl->recv_total = region_nelem(n, region) * pex;
            } else {
                int64_t dims[2 * MAXND], str[2 * MAXND], off;
                int k;
                window_desc(*pin, rank, pl->extra, region, dims, str, &off,
                            &k);
                pl->local = unpack_desc(*pl, region, dims, str, off);
                pl->has_local = true;
            }
        }
        *out = pl;
        return 0;
    }

    pl->P = (int)pin->topo.dims[R];
    int64_t coords[MAXND];
    cart_coords(pin->topo, rank, coords);
    pl->myk = (int)coords[R];

    Range axl_i[MAXND], axl_o[MAXND];
    axes_for_rank(*pin, rank, axl_i);
    axes_for_rank(*pout, rank, axl_o);

    int64_t isend = 0, irecv = 0;
    for (int k = 0; k < pl->P; k++) {
        int64_t pc[MAXND];
        memcpy(pc, coords, sizeof(int64_t) * pin->topo.m);
        pc[R] = k;
        PeerBlockC blk;
        blk.k = k;
        blk.grank = cart_rank(pin->topo, pc);

        Range axp_o[MAXND], axp_i[MAXND], sr[MAXND], rr[MAXND];
        axes_for_coords(*pout, pc, axp_o);
        axes_for_coords(*pin, pc, axp_i);
        for (int d = 0; d < n; d++) {
            range_intersect(axl_i[d], axp_o[d], &sr[d]); /* :383 */
            range_intersect(axl_o[d], axp_i[d], &rr[d]); /* :388,:521 */
        }
        blk.send_n = region_nelem(n, sr) * pex;
        blk.recv_n = region_nelem(n, rr) * pex;

        if (k == pl->myk) {
            blk.send_off = 0;
            blk.recv_off = 0;
            /* fused self path: direct window->window permuted copy
             * (replaces :394-404 + :588-606; same values, half the HBM
             * traffic).  Aliased (in-place) mode stages through the recv
             * tail like the reference so every src read precedes any dst
             * write. */
            if (blk.send_n > 0) {
                if (pl->aliased) {
                    /* recv_off = length of remote recvs; filled below once
                     * known: record block extents now, patch offset later */
                    blk.recv_off = -1; /* placeholder, fixed after loop */
                } else {
                    int64_t dims[2 * MAXND], str[2 * MAXND], off;
                    int kk;
                    window_desc(*pin, rank, pl->extra, sr, dims, str, &off,
                                &kk);
                    pl->local = unpack_desc(*pl, rr, dims, str, off);
                    pl->has_local = true;
                }
            }
        } else {
            blk.send_off = isend;
            blk.recv_off = irecv;
            if (blk.send_n > 0) {
                int64_t dims[2 * MAXND], str[2 * MAXND], off;
                int kk;
                window_desc(*pin, rank, pl->extra, sr, dims, str, &off, &kk);
                int64_t cstr[2 * MAXND], acc = 1;
                for (int i = 0; i < kk; i++) {
                    cstr[i] = acc;
                    acc *= dims[i];
                }
                blk.pack = normalize_desc(kk, dims, str, off, cstr, isend);
                blk.has_pack = true;
                blk.send_outer = dims[kk - 1];
                blk.send_rowelems = blk.send_n / (dims[kk - 1] ? dims[kk - 1] : 1);
            }
            if (blk.recv_n > 0) {
                /* buffer dims: recv block extents gathered by Pi's
                 * permutation (+ extras), column-major (:527,:599-600) */
                int64_t bdims[2 * MAXND];
                for (int i = 0; i < n; i++)
                    bdims[i] = rr[pin->perm[i]].hi - rr[pin->perm[i]].lo;
                for (int e2 = 0; e2 < e; e2++) bdims[n + e2] = pl->extra[e2];
                int64_t bstr[2 * MAXND], acc = 1;
                for (int i = 0; i < n + e; i++) {
                    bstr[i] = acc;
                    acc *= bdims[i];
                }
                blk.unpack = unpack_desc_raw_out(*pl, rr, bdims, bstr,
                                                 irecv, &blk.unpack_raw);
                blk.has_unpack = true;
                blk.has_unpack_raw = true;
                blk.recv_outer = bdims[(n + e) - 1];
                blk.recv_rowelems =
                    blk.recv_n / (blk.recv_outer ? blk.recv_outer : 1);
            }
            isend += blk.send_n;
            irecv += blk.recv_n;
        }
        pl->peers.push_back(blk);
    }
    pl->send_total = isend;
    pl->recv_total = irecv;
    if (pl->aliased) {
        /* self block at the END of the recv buffer (:394-404) */
        PeerBlockC &self = pl->peers[pl->myk];
        self.recv_off = irecv;
        if (self.send_n > 0) {
            Range pc_sr[MAXND], pc_rr[MAXND];
            Range axp_o[MAXND], axp_i[MAXND];
            axes_for_coords(*pout, coords, axp_o);
            axes_for_coords(*pin, coords, axp_i);
            for (int d = 0; d < n; d++) {
                range_intersect(axl_i[d], axp_o[d], &pc_sr[d]);
                range_intersect(axl_o[d], axp_i[d], &pc_rr[d]);
            }
            staged_self_descs(pl, pc_sr, pc_rr, irecv);
            pl->recv_total = irecv + self.send_n;
        }
    }
    *out = pl;
    return 0;
}

void pa_plan_destroy(pa_plan *p)
{
    if (!p) return;
    if (p->own_bufs) {
        if (p->send_buf) (void)hipFree(p->send_buf);
        if (p->recv_buf) (void)hipFree(p->recv_buf);
    }
    if (p->ev_pack) (void)hipEventDestroy(p->ev_pack);
    if (p->ev_comm) (void)hipEventDestroy(p->ev_comm);
    for (auto e : p->ev_chunks) (void)hipEventDestroy(e);
    for (auto e : p->tm)
        if (e) (void)hipEventDestroy(e);
    if (p->comm_stream) (void)hipStreamDestroy(p->comm_stream);
    delete p;
}

pa_status pa_plan_set_comm(pa_plan *p, pa_comm *c)
{
    if (p->R >= 0 && p->P > 1) {
        if (c->nranks != p->P)
            return fail("comm size %d != subgroup size %d", c->nranks, p->P);
        if (c->rank != p->myk)
            return fail("comm rank %d != my subgroup coordinate %d", c->rank,
                        p->myk);
    }
    p->comm = c;
    return 0;
}

pa_status pa_plan_buffer_sizes(const pa_plan *p, int64_t *send_bytes,
                               int64_t *recv_bytes)
{
    *send_bytes = p->send_total * p->esz;
    *recv_bytes = p->recv_total * p->esz;
    return 0;
}

pa_status pa_plan_set_buffers(pa_plan *p, void *send_buf, void *recv_buf)
{
    if (p->own_bufs) {
        if (p->send_buf) (void)hipFree(p->send_buf);
        if (p->recv_buf) (void)hipFree(p->recv_buf);
        p->own_bufs = false;
    }
    p->send_buf = send_buf;
    p->recv_buf = recv_buf;
    return 0;
}

pa_status pa_transpose_execute(pa_plan *p, const void *src_parent,
                               void *dst_parent, void *stream_)
{
    hipStream_t stream = (hipStream_t)stream_;

    const bool need_bufs = p->send_total > 0 || p->recv_total > 0;
    if (need_bufs && !p->send_buf && !p->own_bufs) {
        if (p->send_total)
            HIP_CHECK(hipMalloc(&p->send_buf, p->send_total * p->esz));
        if (p->recv_total)
            HIP_CHECK(hipMalloc(&p->recv_buf, p->recv_total * p->esz));
        p->own_bufs = true;
    }

    if (p->timing) {
        for (auto &e : p->tm)
            if (!e) HIP_CHECK(hipEventCreate(&e));
        for (auto &v : p->tm_valid) v = false;
    }
#define TM_REC(i, strm)                                                      \
    do {                                                                     \
        if (p->timing) {                                                     \
            HIP_CHECK(hipEventRecord(p->tm[i], strm));                       \
            p->tm_valid[i] = true;                                           \
        }                                                                    \
    } while (0)

    /* 1. pack every remote block (Transpositions.jl:346-431); in aliased
     * mode also stage the self block into the recv tail before any dst
     * write (:394-404) */
    TM_REC(0, stream);
    for (auto &blk : p->peers)
        if (blk.has_pack) {
            pa_status st =
                launch_desc(blk.pack, p->esz, src_parent, p->send_buf, stream);
            if (st) return st;
        }
    if (p->has_self) {
        pa_status st = launch_desc(p->self_pack, p->esz, src_parent,
                                   p->recv_buf, stream);
        if (st) return st;
    }
    TM_REC(1, stream);

    /* 2. exchange: grouped ncclSend/ncclRecv over xGMI on a dedicated comm
     * stream ordered after the pack kernels (replaces :419-428/:463-479;
     * the pre-send device sync of :472-473 is unnecessary — everything is
     * stream/event-ordered).  The fused local copy (step 3) runs on the
     * caller's stream CONCURRENTLY with the exchange — the engine's
     * equivalent of the reference's unpack-local-while-recvs-fly overlap
     * (Transpositions.jl:510-517). */
    bool exchanging = false;
    if (p->R >= 0 && p->P > 1) {
        for (auto &blk : p->peers)
            if (blk.k != p->myk && (blk.send_n || blk.recv_n))
                exchanging = true;
    }
    const int C = (exchanging && p->chunks > 1) ? p->chunks : 1;
    if (exchanging) {
        if (!p->comm)
            return fail("subgroup exchange requires pa_plan_set_comm");
        if (!p->comm_stream) {
            HIP_CHECK(hipStreamCreateWithFlags(&p->comm_stream,
                                               hipStreamNonBlocking));
            HIP_CHECK(hipEventCreateWithFlags(&p->ev_pack,
                                              hipEventDisableTiming));
            HIP_CHECK(hipEventCreateWithFlags(&p->ev_comm,
                                              hipEventDisableTiming));
        }
        while ((int)p->ev_chunks.size() < C) {
            hipEvent_t e;
            HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
            p->ev_chunks.push_back(e);
        }
        HIP_CHECK(hipEventRecord(p->ev_pack, stream));
        HIP_CHECK(hipStreamWaitEvent(p->comm_stream, p->ev_pack, 0));
        TM_REC(3, p->comm_stream);
        /* C == 1: one grouped exchange.  C > 1: the exchange is split into C
         * groups of matching per-peer sub-blocks (outer rows of the block,
         * identical split on sender and receiver), and each chunk's unpack
         * overlaps the later chunks' transfers — the reference's Waitany
         * overlap (Transpositions.jl:510-517) in stream/event form. */
        for (int c = 0; c < C; c++) {
            NCCL_CHECK(ncclGroupStart());
            for (auto &blk : p->peers) {
                if (blk.k == p->myk) continue;
                if (blk.recv_n) {
                    const int64_t lo = chunk_lo(blk.recv_outer, c, C);
                    const int64_t hi = chunk_lo(blk.recv_outer, c + 1, C);
                    if (hi > lo)
                        NCCL_CHECK(ncclRecv(
                            (char *)p->recv_buf +
                                (blk.recv_off + lo * blk.recv_rowelems) *
                                    p->esz,
                            (size_t)((hi - lo) * blk.recv_rowelems * p->esz),
                            ncclUint8, blk.k, p->comm->comm, p->comm_stream));
                }
                if (blk.send_n) {
                    const int64_t lo = chunk_lo(blk.send_outer, c, C);
                    const int64_t hi = chunk_lo(blk.send_outer, c + 1, C);
                    if (hi > lo)
                        NCCL_CHECK(ncclSend(
                            (const char *)p->send_buf +
                                (blk.send_off + lo * blk.send_rowelems) *
                                    p->esz,
                            (size_t)((hi - lo) * blk.send_rowelems * p->esz),
                            ncclUint8, blk.k, p->comm->comm, p->comm_stream));
                }
            }
            NCCL_CHECK(ncclGroupEnd());
            HIP_CHECK(hipEventRecord(p->ev_chunks[c], p->comm_stream));
        }
        TM_REC(4, p->comm_stream);
        HIP_CHECK(hipEventRecord(p->ev_comm, p->comm_stream));
    }

    /* 3. fused local/self copy — on the caller's stream, overlapping the
     * exchange (aliased mode: unpack the staged self block instead) */
    if (p->has_local) {
        pa_status st =
            launch_desc(p->local, p->esz, src_parent, dst_parent, stream);
        if (st) return st;
    }
    if (p->has_self) {
        pa_status st = launch_desc(p->self_unpack, p->esz, p->recv_buf,
                                   dst_parent, stream);
        if (st) return st;
    }
    TM_REC(2, stream);

    /* 4. unpack received blocks (:489-536).  C == 1: all after the single
     * exchange completes.  C > 1: chunk c unpacks as soon as its group
     * lands, overlapping chunks c+1.. in flight. */
    if (C <= 1) {
        if (exchanging)
            HIP_CHECK(hipStreamWaitEvent(stream, p->ev_comm, 0));
        TM_REC(5, stream);
        for (auto &blk : p->peers)
            if (blk.has_unpack) {
                pa_status st = launch_desc(blk.unpack, p->esz, p->recv_buf,
                                           dst_parent, stream);
                if (st) return st;
            }
        TM_REC(6, stream);
    } else {
        for (int c = 0; c < C; c++) {
            HIP_CHECK(hipStreamWaitEvent(stream, p->ev_chunks[c], 0));
            if (c == 0) TM_REC(5, stream);
            for (auto &blk : p->peers) {
                if (!blk.has_unpack_raw) continue;
                const int64_t lo = chunk_lo(blk.recv_outer, c, C);
                const int64_t hi = chunk_lo(blk.recv_outer, c + 1, C);
                if (hi <= lo) continue;
                CopyDescH d = chunk_of_raw(blk.unpack_raw, lo, hi);
                pa_status st =
                    launch_desc(d, p->esz, p->recv_buf, dst_parent, stream);
                if (st) return st;
            }
        }
        TM_REC(6, stream);
    }
#undef TM_REC

    return 0;
}

/* ---- per-stage timing (TimerOutputs analogue) ---------------------- */

pa_status pa_plan_enable_timing(pa_plan *p, int enable)
{
    p->timing = enable != 0;
    return 0;
}

pa_status pa_plan_stage_times(pa_plan *p, double out[4])
{
    /* Valid after the last execute has completed (pa_transpose_wait).
     * out = {pack_ms, local_ms, exchange_ms, unpack_ms}; -1 for stages the
     * plan does not have. */
    if (!p->timing) return fail("timing not enabled (pa_plan_enable_timing)");
    float ms;
    auto span = [&](int a, int b) -> double {
        if (!p->tm_valid[a] || !p->tm_valid[b]) return -1.0;
        if (hipEventElapsedTime(&ms, p->tm[a], p->tm[b]) != hipSuccess)
            return -1.0;
        return (double)ms;
    };
    out[0] = span(0, 1); /* pack (+ staged self pack)      */
    out[1] = span(1, 2); /* fused local / self unpack      */
    out[2] = span(3, 4); /* exchange (comm stream)         */
    out[3] = span(5, 6); /* unpack of received blocks      */
    return 0;
}

pa_status pa_transpose_wait(pa_plan *p, void *stream_)
{
    (void)p;
    HIP_CHECK(hipStreamSynchronize((hipStream_t)stream_));
    return 0;
}

/* ---- introspection ------------------------------------------------ */

int pa_plan_nproc_sub(const pa_plan *p) { return p->P; }
int pa_plan_r_dim(const pa_plan *p) { return p->R; }
int pa_plan_my_k(const pa_plan *p) { return p->myk; }

pa_status pa_plan_block_info(const pa_plan *p, int k, int64_t out[8])
{
    if (p->R < 0 || k < 0 || k >= (int)p->peers.size())
        return fail("no peer block %d", k);
    const PeerBlockC &b = p->peers[k];
    out[0] = b.k;
    out[1] = b.grank;
    out[2] = b.send_off;
    out[3] = b.recv_off;
    out[4] = b.send_n;
    out[5] = b.recv_n;
    out[6] = b.has_pack;
    out[7] = b.has_unpack;
    return 0;
}

pa_status pa_plan_copydesc(const pa_plan *p, int which, int k, int64_t *nd,
                           int64_t *dims, int64_t *sstr, int64_t *soff,
                           int64_t *dstr, int64_t *doff)
{
    const CopyDescH *d = nullptr;
    if (which == 0) {
        if (!p->has_local) return fail("no local copy in plan");
        d = &p->local;
    } else if (which == 3) {
        if (!p->has_self) return fail("no staged self pack in plan");
        d = &p->self_pack;
    } else if (which == 4) {
        if (!p->has_self) return fail("no staged self unpack in plan");
        d = &p->self_unpack;
    } else if (which == 5) {
        if (p->R < 0 || k < 0 || k >= (int)p->peers.size())
            return fail("no peer block %d", k);
        if (!p->peers[k].has_unpack_raw) return fail("no raw unpack");
        d = &p->peers[k].unpack_raw;
        /* raw: may exceed normalized nd cap assumptions of callers; nd <= 2*MAXND
         * but CopyDescH holds MAXND — enforced at plan build (n+E <= MAXND). */
    } else {
        if (p->R < 0 || k < 0 || k >= (int)p->peers.size())
            return fail("no peer block %d", k);
        const PeerBlockC &b = p->peers[k];
        if (which == 1) {
            if (!b.has_pack) return fail("no pack for peer %d", k);
            d = &b.pack;
        } else if (which == 2) {
            if (!b.has_unpack) return fail("no unpack for peer %d", k);
            d = &b.unpack;
        } else
            return fail("bad which %d", which);
    }
    *nd = d->nd;
    for (int i = 0; i < d->nd; i++) {
        dims[i] = d->dims[i];
        sstr[i] = d->sstr[i];
        dstr[i] = d->dstr[i];
    }
    *soff = d->soff;
    *doff = d->doff;
    return 0;
}

/* ---- reductions ---------------------------------------------------- */

pa_status pa_allreduce(pa_comm *c, const void *sendbuf, void *recvbuf,
                       int64_t count, int dtype, int op, void *stream)
{
    static const ncclDataType_t dts[] = {ncclFloat64, ncclFloat32, ncclInt64,
                                         ncclInt32, ncclUint8};
    static const ncclRedOp_t ops[] = {ncclSum, ncclProd, ncclMin, ncclMax};
    if (dtype < 0 || dtype > 4) return fail("bad dtype %d", dtype);
    if (op < 0 || op > 3) return fail("bad op %d", op);
    NCCL_CHECK(ncclAllReduce(sendbuf, recvbuf, (size_t)count, dts[dtype],
                             ops[op], c->comm, (hipStream_t)stream));
    return 0;
}

/* ---- standalone device copy --------------------------------------- */

pa_status pa_device_copy(int nd, const int64_t *dims, const int64_t *sstr,
                         int64_t soff, const int64_t *dstr, int64_t doff,
                         int64_t elem_size, const void *src, void *dst,
                         void *stream)
{
    if (nd <= 0 || nd > MAXND) return fail("bad nd");
    CopyDescH d = normalize_desc(nd, dims, sstr, soff, dstr, doff);
    return launch_desc(d, elem_size, src, dst, (hipStream_t)stream);
}

} /* extern "C" */
