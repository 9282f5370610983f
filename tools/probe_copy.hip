/* probe_copy.hip — kernel-variant microbenchmark for the copy engine.
 * Standalone; run on an MI355X to pick the fastest variants, which are then
 * baked into csrc/pencilhip.hip.  Measures GB/s (algorithmic: rd+wr bytes /
 * time) with HIP events, best-of-reps, plus a checksum correctness check.
 *
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 probe_copy.hip -o probe_copy
 */

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>

#define CHK(x)                                                               \
    do {                                                                     \
        hipError_t e = (x);                                                  \
        if (e != hipSuccess) {                                               \
            fprintf(stderr, "HIP error %s @%d\n", hipGetErrorString(e),      \
                    __LINE__);                                               \
            exit(1);                                                         \
        }                                                                    \
    } while (0)

/* ---------------- 1-D copy variants ---------------- */

__global__ __launch_bounds__(256) void c_gridstride(const uint4 *__restrict__ s,
                                                    uint4 *__restrict__ d,
                                                    int64_t n)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t st = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += st) d[i] = s[i];
}

__global__ __launch_bounds__(256) void c_direct(const uint4 *__restrict__ s,
                                                uint4 *__restrict__ d,
                                                int64_t n)
{
    const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) d[i] = s[i];
}

typedef unsigned int v4u __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(256) void c_nt(const v4u *__restrict__ s,
                                            v4u *__restrict__ d, int64_t n)
{
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t st = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += st)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&s[i]), &d[i]);
}

__global__ __launch_bounds__(256) void c_unroll4(const uint4 *__restrict__ s,
                                                 uint4 *__restrict__ d,
                                                 int64_t n)
{
    const int64_t st = (int64_t)gridDim.x * blockDim.x;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t n4 = n - 3 * st;
    for (; i < n4; i += 4 * st) {
        uint4 a = s[i], b = s[i + st], c = s[i + 2 * st], e = s[i + 3 * st];
        d[i] = a;
        d[i + st] = b;
        d[i + 2 * st] = c;
        d[i + 3 * st] = e;
    }
    for (; i < n; i += st) d[i] = s[i];
}

__global__ __launch_bounds__(512) void c_direct512(const uint4 *__restrict__ s,
                                                   uint4 *__restrict__ d,
                                                   int64_t n)
{
    const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) d[i] = s[i];
}

__global__ __launch_bounds__(256) void c_nt_unroll4(const v4u *__restrict__ s,
                                                    v4u *__restrict__ d,
                                                    int64_t n)
{
    const int64_t st = (int64_t)gridDim.x * blockDim.x;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t n4 = n - 3 * st;
    for (; i < n4; i += 4 * st) {
        v4u a = __builtin_nontemporal_load(&s[i]);
        v4u b = __builtin_nontemporal_load(&s[i + st]);
        v4u c = __builtin_nontemporal_load(&s[i + 2 * st]);
        v4u e = __builtin_nontemporal_load(&s[i + 3 * st]);
        __builtin_nontemporal_store(a, &d[i]);
        __builtin_nontemporal_store(b, &d[i + st]);
        __builtin_nontemporal_store(c, &d[i + 2 * st]);
        __builtin_nontemporal_store(e, &d[i + 3 * st]);
    }
    for (; i < n; i += st) d[i] = s[i];
}

/* ---------------- transpose variants ----------------
 * Model: 2-D transpose of a NI x NJ f64 matrix (src row-major along i,
 * dst row-major along j): src[i + NI*j] -> dst[j + NJ*i].
 * This is the unpack kernel's shape with batch=1. */

template <int TILE, int NROWS>
__global__ __launch_bounds__(64 * NROWS) void t_scalar(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t nti)
{
    __shared__ uint64_t tile[TILE][TILE + 1];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t t_j = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * TILE, j0 = t_j * TILE;
    const int64_t ni = min((int64_t)TILE, NI - i0);
    const int64_t nj = min((int64_t)TILE, NJ - j0);
    for (int j = ty; j < nj; j += NROWS)
        for (int i = tx; i < ni; i += 64)
            tile[j][i] = src[(i0 + i) + NI * (j0 + j)];
    __syncthreads();
    for (int i = ty; i < ni; i += NROWS)
        for (int j = tx; j < nj; j += 64)
            dst[(j0 + j) + NJ * (i0 + i)] = tile[j][i];
}

/* rectangular tile: TI along src-contiguous axis, TJ along dst-contiguous
 * axis (longer reads vs writes tradeoff). */
template <int TI, int TJ, int NROWS>
__global__ __launch_bounds__(64 * NROWS) void t_rect(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t nti)
{
    __shared__ uint64_t tile[TJ][TI + 1];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t t_j = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * TI, j0 = t_j * TJ;
    for (int j = ty; j < TJ; j += NROWS)
        for (int i = tx; i < TI; i += 64)
            tile[j][i] = src[(i0 + i) + NI * (j0 + j)];
    __syncthreads();
    for (int i = ty; i < TI; i += NROWS)
        for (int j = tx; j < TJ; j += 64)
            dst[(j0 + j) + NJ * (i0 + i)] = tile[j][i];
}

/* 64x64 with nontemporal global accesses */
typedef unsigned long v1u64 __attribute__((ext_vector_type(1)));
template <int TILE, int NROWS>
__global__ __launch_bounds__(64 * NROWS) void t_nt(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t nti)
{
    __shared__ uint64_t tile[TILE][TILE + 1];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t t_j = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * TILE, j0 = t_j * TILE;
    for (int j = ty; j < TILE; j += NROWS)
        for (int i = tx; i < TILE; i += 64)
            tile[j][i] =
                __builtin_nontemporal_load(&src[(i0 + i) + NI * (j0 + j)]);
    __syncthreads();
    for (int i = ty; i < TILE; i += NROWS)
        for (int j = tx; j < TILE; j += 64)
            __builtin_nontemporal_store(tile[j][i],
                                        &dst[(j0 + j) + NJ * (i0 + i)]);
}

/* t_rect with t_j-fastest block order: consecutive blocks advance along the
 * dst-contiguous axis, so concurrent blocks read AND write adjacent regions
 * (DRAM page locality) on tall-skinny shapes. */
template <int TI, int TJ, int NROWS>
__global__ __launch_bounds__(64 * NROWS) void t_rect_jf(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t ntj)
{
    __shared__ uint64_t tile[TJ][TI + 1];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int64_t t_j = (int64_t)blockIdx.x % ntj;
    const int64_t t_i = (int64_t)blockIdx.x / ntj;
    const int64_t i0 = t_i * TI, j0 = t_j * TJ;
    for (int j = ty; j < TJ; j += NROWS)
        for (int i = tx; i < TI; i += 64)
            tile[j][i] = src[(i0 + i) + NI * (j0 + j)];
    __syncthreads();
    for (int i = ty; i < TI; i += NROWS)
        for (int j = tx; j < TJ; j += 64)
            dst[(j0 + j) + NJ * (i0 + i)] = tile[j][i];
}

/* j-sweeping workgroup: each WG owns one i-block and a CONTIGUOUS span of
 * j-tiles, so its reads advance through one contiguous src region and each
 * of its TI output rows is written as a sequential stream (long DRAM
 * bursts on both sides). */
template <int TI, int TJ, int NROWS, int JCHUNK>
__global__ __launch_bounds__(64 * NROWS) void t_sweep(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t nti)
{
    __shared__ uint64_t tile[TJ][TI + 1];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t chunk = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * TI;
    const int64_t jbase = chunk * (int64_t)JCHUNK * TJ;
    for (int jt = 0; jt < JCHUNK; jt++) {
        const int64_t j0 = jbase + (int64_t)jt * TJ;
        if (j0 >= NJ) break;
        for (int j = ty; j < TJ; j += NROWS)
            for (int i = tx; i < TI; i += 64)
                tile[j][i] = src[(i0 + i) + NI * (j0 + j)];
        __syncthreads();
        for (int i = ty; i < TI; i += NROWS)
            for (int j = tx; j < TJ; j += 64)
                dst[(j0 + j) + NJ * (i0 + i)] = tile[j][i];
        __syncthreads();
    }
}

/* sweep + 16-B vector global accesses (interior tiles, TI=128: row = 64
 * lanes x uint4) */
template <int NROWS, int JCHUNK>
__global__ __launch_bounds__(64 * NROWS) void t_sweep_vec(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t nti)
{
    constexpr int TI = 128, TJ = 64;
    __shared__ uint64_t tile[TJ][TI + 2];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t chunk = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * TI;
    for (int jt = 0; jt < JCHUNK; jt++) {
        const int64_t j0 = (chunk * (int64_t)JCHUNK + jt) * TJ;
        if (j0 >= NJ) break;
        for (int j = ty; j < TJ; j += NROWS) {
            const uint4 v = *(const uint4 *)&src[(i0 + 2 * tx) +
                                                 NI * (j0 + j)];
            tile[j][2 * tx] = ((const uint64_t *)&v)[0];
            tile[j][2 * tx + 1] = ((const uint64_t *)&v)[1];
        }
        __syncthreads();
        /* write: lanes 0..31 cover the 32 j-pairs of row i, lanes 32..63
         * the j-pairs of row i+NROWS-offsetted partner (2 rows per y-step) */
        {
            const int jj = 2 * (tx & 31);
            const int half = tx >> 5;
            for (int i = 2 * ty + half; i < TI; i += 2 * NROWS) {
                uint4 v;
                ((uint64_t *)&v)[0] = tile[jj][i];
                ((uint64_t *)&v)[1] = tile[jj + 1][i];
                *(uint4 *)&dst[(j0 + jj) + NJ * (i0 + i)] = v;
            }
        }
        __syncthreads();
    }
}

/* sweep + double-buffered LDS (no barrier between store(jt) and load(jt+1));
 * LDS 2x tiles => 1 WG/CU */
template <int NROWS, int JCHUNK>
__global__ __launch_bounds__(64 * NROWS) void t_sweep_db(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t nti)
{
    constexpr int TI = 128, TJ = 64;
    __shared__ uint64_t tile[2][TJ][TI + 1];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t chunk = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * TI;
    int cur = 0;
    /* preload tile 0 */
    {
        const int64_t j0 = chunk * (int64_t)JCHUNK * TJ;
        if (j0 >= NJ) return;
        for (int j = ty; j < TJ; j += NROWS)
            for (int i = tx; i < TI; i += 64)
                tile[0][j][i] = src[(i0 + i) + NI * (j0 + j)];
    }
    for (int jt = 0; jt < JCHUNK; jt++) {
        const int64_t j0 = (chunk * (int64_t)JCHUNK + jt) * TJ;
        if (j0 >= NJ) break;
        __syncthreads(); /* tile[cur] complete */
        /* prefetch next tile into the other buffer while storing cur */
        const int64_t j0n = j0 + TJ;
        if (jt + 1 < JCHUNK && j0n < NJ)
            for (int j = ty; j < TJ; j += NROWS)
                for (int i = tx; i < TI; i += 64)
                    tile[cur ^ 1][j][i] = src[(i0 + i) + NI * (j0n + j)];
        for (int i = ty; i < TI; i += NROWS) {
            const int64_t row = (j0) + NJ * (i0 + i);
            for (int j = tx; j < TJ; j += 64)
                dst[row + j] = tile[cur][j][i];
        }
        cur ^= 1;
    }
}

/* vectorized 16-B loads/stores: interior tiles only (caller guarantees
 * NI,NJ multiples of TILE).  Lanes 0..31 load row 2*ty, lanes 32..63 row
 * 2*ty+1 (uint4 = 2 f64 along i).  Write phase symmetric along j. */
template <int TILE, int NROWS>
__global__ __launch_bounds__(64 * NROWS) void t_vec(
    const uint64_t *__restrict__ src, uint64_t *__restrict__ dst, int64_t NI,
    int64_t NJ, int64_t nti)
{
    __shared__ uint64_t tile[TILE][TILE + 2];
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int lane = tx & 31;      /* 0..31 */
    const int half = tx >> 5;      /* 0 or 1 */
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t t_j = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * TILE, j0 = t_j * TILE;

    /* load: rows 2*r+half for r = ty, ty+NROWS, ... ; each lane one uint4 */
    for (int r = ty; r < TILE / 2; r += NROWS) {
        const int j = 2 * r + half;
        const uint4 v = *(const uint4 *)&src[(i0 + 2 * lane) + NI * (j0 + j)];
        tile[j][2 * lane] = ((const uint64_t *)&v)[0];
        tile[j][2 * lane + 1] = ((const uint64_t *)&v)[1];
    }
    __syncthreads();
    /* store: output rows i = 2*r+half; lane covers j = 2*lane, 2*lane+1 */
    for (int r = ty; r < TILE / 2; r += NROWS) {
        const int i = 2 * r + half;
        uint4 v;
        ((uint64_t *)&v)[0] = tile[2 * lane][i];
        ((uint64_t *)&v)[1] = tile[2 * lane + 1][i];
        *(uint4 *)&dst[(j0 + 2 * lane) + NJ * (i0 + i)] = v;
    }
}

/* ---------------- single-direction pattern ceilings ----------------
 * Bound the transpose: what does the hardware give for its READ pattern
 * alone (rows of TI contiguous f64) and its WRITE pattern alone (rows of
 * TJ=64 contiguous f64 scattered at stride NJ), vs contiguous controls? */

__global__ __launch_bounds__(1024) void r_pattern(const uint64_t *__restrict__ src,
                                                  uint64_t *__restrict__ sink,
                                                  int64_t NI, int64_t NJ,
                                                  int64_t nti)
{
    /* mimic the sweep kernel's load phase: TI=128, TJ=64, NROWS=16, JC=32 */
    const int tx = threadIdx.x & 63, ty = (threadIdx.x >> 6);
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t chunk = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * 128;
    uint64_t acc = 0;
    for (int jt = 0; jt < 32; jt++) {
        const int64_t j0 = (chunk * 32 + jt) * 64;
        if (j0 >= NJ) break;
        for (int j = ty; j < 64; j += 16)
            for (int i = tx; i < 128; i += 64)
                acc ^= src[(i0 + i) + NI * (j0 + j)];
    }
    if (acc == 0xDEADBEEFCAFEBABEull) sink[0] = acc; /* keep loads live */
}

__global__ __launch_bounds__(1024) void w_pattern(uint64_t *__restrict__ dst,
                                                  int64_t NI, int64_t NJ,
                                                  int64_t nti)
{
    /* mimic the sweep kernel's store phase: rows i (TI=128), 64-elem bursts
     * at stride NJ, sweeping 32 j-tiles */
    const int tx = threadIdx.x & 63, ty = (threadIdx.x >> 6);
    const int64_t t_i = (int64_t)blockIdx.x % nti;
    const int64_t chunk = (int64_t)blockIdx.x / nti;
    const int64_t i0 = t_i * 128;
    for (int jt = 0; jt < 32; jt++) {
        const int64_t j0 = (chunk * 32 + jt) * 64;
        if (j0 >= NJ) break;
        for (int i = ty; i < 128; i += 16) {
            const int64_t row = j0 + NJ * (i0 + i);
            for (int j = tx; j < 64; j += 64)
                dst[row + j] = (uint64_t)(row + j);
        }
    }
}

__global__ __launch_bounds__(256) void r_contig(const uint64_t *__restrict__ s,
                                                uint64_t *__restrict__ sink,
                                                int64_t n)
{
    const int64_t i = ((int64_t)blockIdx.y * gridDim.x + blockIdx.x) * 256 +
                      threadIdx.x;
    if (i < n) {
        const uint64_t v = s[i];
        if (v == 0xDEADBEEFCAFEBABEull) sink[0] = v;
    }
}

__global__ __launch_bounds__(256) void w_contig(uint64_t *__restrict__ d,
                                                int64_t n)
{
    const int64_t i = ((int64_t)blockIdx.y * gridDim.x + blockIdx.x) * 256 +
                      threadIdx.x;
    if (i < n) d[i] = (uint64_t)i;
}

/* ---------------- harness ---------------- */

static double bench(void (*launch)(void *, void *, int64_t, int), void *s,
                    void *d, int64_t n, int arg, int reps, double bytes)
{
    hipEvent_t a, b;
    CHK(hipEventCreate(&a));
    CHK(hipEventCreate(&b));
    launch(s, d, n, arg); /* warmup */
    CHK(hipDeviceSynchronize());
    double best = 1e30;
    for (int r = 0; r < reps; r++) {
        CHK(hipEventRecord(a));
        launch(s, d, n, arg);
        CHK(hipEventRecord(b));
        CHK(hipEventSynchronize(b));
        float ms;
        CHK(hipEventElapsedTime(&ms, a, b));
        if (ms < best) best = ms;
    }
    CHK(hipEventDestroy(a));
    CHK(hipEventDestroy(b));
    return bytes / (best * 1e-3) / 1e9;
}

#define NELEM_BYTES (4LL << 30) /* 4 GiB payload */

int main()
{
    const int64_t bytes = NELEM_BYTES;
    const int64_t n16 = bytes / 16;
    void *s, *d;
    CHK(hipMalloc(&s, bytes));
    CHK(hipMalloc(&d, bytes));
    CHK(hipMemset(s, 0x5A, bytes));
    const double io = 2.0 * bytes;
    const int reps = 5;

    auto g = [](int64_t work, int perblk, int cap) {
        int64_t b = (work + perblk - 1) / perblk;
        if (cap && b > cap) b = cap;
        return (int)b;
    };

#define RUN(name, launch_expr)                                               \
    {                                                                        \
        auto L = +[](void *ss, void *dd, int64_t n, int arg) {               \
            (void)arg;                                                       \
            launch_expr;                                                     \
        };                                                                   \
        printf("%-28s %8.1f GB/s\n", name, bench(L, s, d, n16, 0, reps, io)); \
        fflush(stdout);                                                      \
    }

    RUN("1d gridstride cap2048", {
        hipLaunchKernelGGL(c_gridstride, dim3(2048), dim3(256), 0, 0,
                           (const uint4 *)ss, (uint4 *)dd, n);
    });
    RUN("1d gridstride cap8192", {
        hipLaunchKernelGGL(c_gridstride, dim3(8192), dim3(256), 0, 0,
                           (const uint4 *)ss, (uint4 *)dd, n);
    });
    RUN("1d direct full grid", {
        hipLaunchKernelGGL(c_direct, dim3((uint32_t)((n + 255) / 256)),
                           dim3(256), 0, 0, (const uint4 *)ss, (uint4 *)dd, n);
    });
    RUN("1d direct 512thr", {
        hipLaunchKernelGGL(c_direct512, dim3((uint32_t)((n + 511) / 512)),
                           dim3(512), 0, 0, (const uint4 *)ss, (uint4 *)dd, n);
    });
    RUN("1d nt cap2048", {
        hipLaunchKernelGGL(c_nt, dim3(2048), dim3(256), 0, 0,
                           (const v4u *)ss, (v4u *)dd, n);
    });
    RUN("1d unroll4 cap2048", {
        hipLaunchKernelGGL(c_unroll4, dim3(2048), dim3(256), 0, 0,
                           (const uint4 *)ss, (uint4 *)dd, n);
    });
    RUN("1d unroll4 cap4096", {
        hipLaunchKernelGGL(c_unroll4, dim3(4096), dim3(256), 0, 0,
                           (const uint4 *)ss, (uint4 *)dd, n);
    });
    RUN("1d nt+unroll4 cap2048", {
        hipLaunchKernelGGL(c_nt_unroll4, dim3(2048), dim3(256), 0, 0,
                           (const v4u *)ss, (v4u *)dd, n);
    });
    RUN("1d nt+unroll4 cap4096", {
        hipLaunchKernelGGL(c_nt_unroll4, dim3(4096), dim3(256), 0, 0,
                           (const v4u *)ss, (v4u *)dd, n);
    });

    /* transpose probes: 16384 x 16384 f64 (2 GiB payload x2) */
    {
        const int64_t NI = 16384, NJ = 16384;
        const double tio = 2.0 * NI * NJ * 8;
        const int64_t nti64 = NI / 64;
        const int64_t blocks64 = (NI / 64) * (NJ / 64);
#define TRUN(name, kern, TI, NR)                                             \
    {                                                                        \
        const int64_t nti = NI / TI;                                         \
        const int64_t blocks = (NI / TI) * (NJ / TI);                        \
        hipEvent_t a, b;                                                     \
        CHK(hipEventCreate(&a));                                             \
        CHK(hipEventCreate(&b));                                             \
        hipLaunchKernelGGL((kern<TI, NR>), dim3((uint32_t)blocks),           \
                           dim3(64, NR), 0, 0, (const uint64_t *)s,          \
                           (uint64_t *)d, NI, NJ, nti);                      \
        CHK(hipDeviceSynchronize());                                         \
        double best = 1e30;                                                  \
        for (int r = 0; r < reps; r++) {                                     \
            CHK(hipEventRecord(a));                                          \
            hipLaunchKernelGGL((kern<TI, NR>), dim3((uint32_t)blocks),       \
                               dim3(64, NR), 0, 0, (const uint64_t *)s,      \
                               (uint64_t *)d, NI, NJ, nti);                  \
            CHK(hipEventRecord(b));                                          \
            CHK(hipEventSynchronize(b));                                     \
            float ms;                                                        \
            CHK(hipEventElapsedTime(&ms, a, b));                             \
            if (ms < best) best = ms;                                        \
        }                                                                    \
        printf("%-28s %8.1f GB/s\n", name, tio / (best * 1e-3) / 1e9);       \
        fflush(stdout);                                                      \
        CHK(hipEventDestroy(a));                                             \
        CHK(hipEventDestroy(b));                                             \
    }
        (void)nti64;
        (void)blocks64;
        TRUN("tr scalar 64x64 r8", t_scalar, 64, 8);
        TRUN("tr scalar 64x64 r16", t_scalar, 64, 16);
        TRUN("tr scalar 32x32 r8", t_scalar, 32, 8);
        TRUN("tr scalar 128x128 r8", t_scalar, 128, 8);
        TRUN("tr vec16 64x64 r8", t_vec, 64, 8);
        TRUN("tr vec16 64x64 r4", t_vec, 64, 4);
        TRUN("tr vec16 64x64 r16", t_vec, 64, 16);
        TRUN("tr nt 64x64 r8", t_nt, 64, 8);

#define TRUN2(name, TI, TJ, NR)                                              \
    {                                                                        \
        const int64_t nti = NI / TI;                                         \
        const int64_t blocks = (NI / TI) * (NJ / TJ);                        \
        hipEvent_t a, b;                                                     \
        CHK(hipEventCreate(&a));                                             \
        CHK(hipEventCreate(&b));                                             \
        hipLaunchKernelGGL((t_rect<TI, TJ, NR>), dim3((uint32_t)blocks),     \
                           dim3(64, NR), 0, 0, (const uint64_t *)s,          \
                           (uint64_t *)d, NI, NJ, nti);                      \
        CHK(hipDeviceSynchronize());                                         \
        double best = 1e30;                                                  \
        for (int r = 0; r < reps; r++) {                                     \
            CHK(hipEventRecord(a));                                          \
            hipLaunchKernelGGL((t_rect<TI, TJ, NR>), dim3((uint32_t)blocks), \
                               dim3(64, NR), 0, 0, (const uint64_t *)s,      \
                               (uint64_t *)d, NI, NJ, nti);                  \
            CHK(hipEventRecord(b));                                          \
            CHK(hipEventSynchronize(b));                                     \
            float ms;                                                        \
            CHK(hipEventElapsedTime(&ms, a, b));                             \
            if (ms < best) best = ms;                                        \
        }                                                                    \
        printf("%-28s %8.1f GB/s\n", name, tio / (best * 1e-3) / 1e9);       \
        fflush(stdout);                                                      \
        CHK(hipEventDestroy(a));                                             \
        CHK(hipEventDestroy(b));                                             \
    }
        TRUN2("tr rect 128x64 r8", 128, 64, 8);
        TRUN2("tr rect 64x128 r8", 64, 128, 8);
        TRUN2("tr rect 128x32 r8", 128, 32, 8);
        TRUN2("tr rect 32x128 r8", 32, 128, 8);
        TRUN2("tr rect 128x64 r16", 128, 64, 16);
        TRUN2("tr rect 256x64 r8", 256, 64, 8);
        TRUN2("tr rect 64x64 r8 (ctl)", 64, 64, 8);
    }

    /* THE REAL SHAPE: the permuted x->y unpack at 1024^3 world=1 is a
     * (1024 x NJ) tall-skinny transpose (normalized desc dims (1024,
     * 1048576)); probe at NJ = 262144 (2 GiB payload). */
    {
        const int64_t NI = 1024, NJ = 262144;
        const double tio = 2.0 * NI * NJ * 8;
#define TRUNS(name, kern, TI, TJ, NR, jf)                                    \
    {                                                                        \
        const int64_t ntX = jf ? (NJ / TJ) : (NI / TI);                      \
        const int64_t blocks = (NI / TI) * (NJ / TJ);                        \
        hipEvent_t a, b;                                                     \
        CHK(hipEventCreate(&a));                                             \
        CHK(hipEventCreate(&b));                                             \
        hipLaunchKernelGGL((kern<TI, TJ, NR>), dim3((uint32_t)blocks),       \
                           dim3(64, NR), 0, 0, (const uint64_t *)s,          \
                           (uint64_t *)d, NI, NJ, ntX);                      \
        CHK(hipDeviceSynchronize());                                         \
        double best = 1e30;                                                  \
        for (int r = 0; r < reps; r++) {                                     \
            CHK(hipEventRecord(a));                                          \
            hipLaunchKernelGGL((kern<TI, TJ, NR>), dim3((uint32_t)blocks),   \
                               dim3(64, NR), 0, 0, (const uint64_t *)s,      \
                               (uint64_t *)d, NI, NJ, ntX);                  \
            CHK(hipEventRecord(b));                                          \
            CHK(hipEventSynchronize(b));                                     \
            float ms;                                                        \
            CHK(hipEventElapsedTime(&ms, a, b));                             \
            if (ms < best) best = ms;                                        \
        }                                                                    \
        printf("%-28s %8.1f GB/s\n", name, tio / (best * 1e-3) / 1e9);       \
        fflush(stdout);                                                      \
        CHK(hipEventDestroy(a));                                             \
        CHK(hipEventDestroy(b));                                             \
    }
        TRUNS("RS rect 128x64 r16 (cur)", t_rect, 128, 64, 16, 0);
        TRUNS("RS rect 64x64 r8", t_rect, 64, 64, 8, 0);
        TRUNS("RS rect 64x128 r16", t_rect, 64, 128, 16, 0);
        TRUNS("RS rect 32x256 r16", t_rect, 32, 256, 16, 0);
        TRUNS("RS jf 128x64 r16", t_rect_jf, 128, 64, 16, 1);
        TRUNS("RS jf 64x64 r8", t_rect_jf, 64, 64, 8, 1);
        TRUNS("RS jf 64x128 r16", t_rect_jf, 64, 128, 16, 1);
        TRUNS("RS jf 32x256 r16", t_rect_jf, 32, 256, 16, 1);
        TRUNS("RS jf 16x512 r16", t_rect_jf, 16, 512, 16, 1);
        TRUNS("RS rect 128x128 r16", t_rect, 128, 128, 16, 0);
        TRUNS("RS rect 128x64 r8", t_rect, 128, 64, 8, 0);
        TRUNS("RS rect 256x32 r16", t_rect, 256, 32, 16, 0);

#define TRUNW(name, TI, TJ, NR, JC)                                          \
    {                                                                        \
        const int64_t nti = NI / TI;                                         \
        const int64_t njc = (NJ / TJ + JC - 1) / JC;                         \
        const int64_t blocks = nti * njc;                                    \
        hipEvent_t a, b;                                                     \
        CHK(hipEventCreate(&a));                                             \
        CHK(hipEventCreate(&b));                                             \
        hipLaunchKernelGGL((t_sweep<TI, TJ, NR, JC>),                        \
                           dim3((uint32_t)blocks), dim3(64, NR), 0, 0,       \
                           (const uint64_t *)s, (uint64_t *)d, NI, NJ, nti); \
        CHK(hipDeviceSynchronize());                                         \
        double best = 1e30;                                                  \
        for (int r = 0; r < reps; r++) {                                     \
            CHK(hipEventRecord(a));                                          \
            hipLaunchKernelGGL((t_sweep<TI, TJ, NR, JC>),                    \
                               dim3((uint32_t)blocks), dim3(64, NR), 0, 0,   \
                               (const uint64_t *)s, (uint64_t *)d, NI, NJ,   \
                               nti);                                         \
            CHK(hipEventRecord(b));                                          \
            CHK(hipEventSynchronize(b));                                     \
            float ms;                                                        \
            CHK(hipEventElapsedTime(&ms, a, b));                             \
            if (ms < best) best = ms;                                        \
        }                                                                    \
        printf("%-28s %8.1f GB/s\n", name, tio / (best * 1e-3) / 1e9);       \
        fflush(stdout);                                                      \
        CHK(hipEventDestroy(a));                                             \
        CHK(hipEventDestroy(b));                                             \
    }
        TRUNW("RS sweep 128x64 r16 c4", 128, 64, 16, 4);
        TRUNW("RS sweep 128x64 r16 c8", 128, 64, 16, 8);
        TRUNW("RS sweep 128x64 r16 c16", 128, 64, 16, 16);
        TRUNW("RS sweep 128x64 r16 c32", 128, 64, 16, 32);
        TRUNW("RS sweep 64x64 r8 c16", 64, 64, 8, 16);
        TRUNW("RS sweep 128x128 r16 c8", 128, 128, 16, 8);

        /* within-probe interleaved A/B: rect (current) vs sweep c32/c64,
         * 12 rounds each, report per-variant best and median */
        {
            const int ROUNDS = 12;
            double t_rect_ms[ROUNDS], t_c32[ROUNDS], t_c64[ROUNDS];
            hipEvent_t a, b;
            CHK(hipEventCreate(&a));
            CHK(hipEventCreate(&b));
            const int64_t nti = NI / 128;
            const int64_t ntj_t = NJ / 64;
            const int64_t njc32 = (ntj_t + 31) / 32, njc64 = (ntj_t + 63) / 64;
#define ONE(kern_launch, arr, r)                                             \
    {                                                                        \
        CHK(hipEventRecord(a));                                              \
        kern_launch;                                                         \
        CHK(hipEventRecord(b));                                              \
        CHK(hipEventSynchronize(b));                                         \
        float ms;                                                            \
        CHK(hipEventElapsedTime(&ms, a, b));                                 \
        arr[r] = ms;                                                         \
    }
            for (int r = 0; r < ROUNDS; r++) {
                ONE(hipLaunchKernelGGL((t_rect<128, 64, 16>),
                                       dim3((uint32_t)(nti * ntj_t)),
                                       dim3(64, 16), 0, 0,
                                       (const uint64_t *)s, (uint64_t *)d, NI,
                                       NJ, nti),
                    t_rect_ms, r);
                ONE(hipLaunchKernelGGL((t_sweep<128, 64, 16, 32>),
                                       dim3((uint32_t)(nti * njc32)),
                                       dim3(64, 16), 0, 0,
                                       (const uint64_t *)s, (uint64_t *)d, NI,
                                       NJ, nti),
                    t_c32, r);
                ONE(hipLaunchKernelGGL((t_sweep<128, 64, 16, 64>),
                                       dim3((uint32_t)(nti * njc64)),
                                       dim3(64, 16), 0, 0,
                                       (const uint64_t *)s, (uint64_t *)d, NI,
                                       NJ, nti),
                    t_c64, r);
            }
            CHK(hipEventDestroy(a));
            CHK(hipEventDestroy(b));
            /* extra variants, same best-of pattern */
            {
                const int64_t njc32 = (ntj_t + 31) / 32;
                hipEvent_t a2, b2;
                CHK(hipEventCreate(&a2));
                CHK(hipEventCreate(&b2));
#define BESTOF(name, launch)                                                 \
    {                                                                        \
        launch;                                                              \
        CHK(hipDeviceSynchronize());                                         \
        double best = 1e30;                                                  \
        for (int r = 0; r < 8; r++) {                                        \
            CHK(hipEventRecord(a2));                                         \
            launch;                                                          \
            CHK(hipEventRecord(b2));                                         \
            CHK(hipEventSynchronize(b2));                                    \
            float ms;                                                        \
            CHK(hipEventElapsedTime(&ms, a2, b2));                           \
            if (ms < best) best = ms;                                        \
        }                                                                    \
        printf("AB %-10s best %8.1f GB/s\n", name,                           \
               tio / (best * 1e-3) / 1e9);                                   \
        fflush(stdout);                                                      \
    }
                BESTOF("sweepvec32",
                       hipLaunchKernelGGL((t_sweep_vec<16, 32>),
                                          dim3((uint32_t)(nti * njc32)),
                                          dim3(64, 16), 0, 0,
                                          (const uint64_t *)s, (uint64_t *)d,
                                          NI, NJ, nti));
                BESTOF("sweepdb32",
                       hipLaunchKernelGGL((t_sweep_db<16, 32>),
                                          dim3((uint32_t)(nti * njc32)),
                                          dim3(64, 16), 0, 0,
                                          (const uint64_t *)s, (uint64_t *)d,
                                          NI, NJ, nti));
                CHK(hipEventDestroy(a2));
                CHK(hipEventDestroy(b2));
            }
            /* single-direction pattern ceilings (one-direction GB/s) */
            {
                const double one = 1.0 * NI * NJ * 8;
                const int64_t njc32 = (ntj_t + 31) / 32;
                hipEvent_t a3, b3;
                CHK(hipEventCreate(&a3));
                CHK(hipEventCreate(&b3));
#define BEST1(name, launch, bytes)                                           \
    {                                                                        \
        launch;                                                              \
        CHK(hipDeviceSynchronize());                                         \
        double best = 1e30;                                                  \
        for (int r = 0; r < 8; r++) {                                        \
            CHK(hipEventRecord(a3));                                         \
            launch;                                                          \
            CHK(hipEventRecord(b3));                                         \
            CHK(hipEventSynchronize(b3));                                    \
            float ms;                                                        \
            CHK(hipEventElapsedTime(&ms, a3, b3));                           \
            if (ms < best) best = ms;                                        \
        }                                                                    \
        printf("DIR %-12s %8.1f GB/s (one direction)\n", name,               \
               bytes / (best * 1e-3) / 1e9);                                 \
        fflush(stdout);                                                      \
    }
                BEST1("read-pattern",
                      hipLaunchKernelGGL(r_pattern,
                                         dim3((uint32_t)(nti * njc32)),
                                         dim3(1024), 0, 0,
                                         (const uint64_t *)s, (uint64_t *)d,
                                         NI, NJ, nti),
                      one);
                BEST1("write-pattern",
                      hipLaunchKernelGGL(w_pattern,
                                         dim3((uint32_t)(nti * njc32)),
                                         dim3(1024), 0, 0, (uint64_t *)d, NI,
                                         NJ, nti),
                      one);
                const int64_t n64 = NI * NJ;
                BEST1("read-contig",
                      hipLaunchKernelGGL(r_contig,
                                         dim3((uint32_t)((n64 + 255) / 256)),
                                         dim3(256), 0, 0,
                                         (const uint64_t *)s, (uint64_t *)d,
                                         n64),
                      one);
                BEST1("write-contig",
                      hipLaunchKernelGGL(w_contig,
                                         dim3((uint32_t)((n64 + 255) / 256)),
                                         dim3(256), 0, 0, (uint64_t *)d, n64),
                      one);
                CHK(hipEventDestroy(a3));
                CHK(hipEventDestroy(b3));
            }
            const char *names[3] = {"rect(cur)", "sweep c32", "sweep c64"};
            double *arrs[3] = {t_rect_ms, t_c32, t_c64};
            for (int v = 0; v < 3; v++) {
                double best = 1e30, sum = 0;
                for (int r = 0; r < ROUNDS; r++) {
                    if (arrs[v][r] < best) best = arrs[v][r];
                    sum += arrs[v][r];
                }
                printf("AB %-10s best %8.1f GB/s  mean %8.1f GB/s\n",
                       names[v], tio / (best * 1e-3) / 1e9,
                       tio / ((sum / ROUNDS) * 1e-3) / 1e9);
            }
            fflush(stdout);
        }
        /* t_sweep correctness */
        {
            const int64_t ni = 256, nj = 448; /* non-multiple of chunk span */
            uint64_t *hs = (uint64_t *)malloc(ni * nj * 8);
            uint64_t *hd = (uint64_t *)malloc(ni * nj * 8);
            for (int64_t i = 0; i < ni * nj; i++) hs[i] = i * 0x9E3779B9ULL;
            CHK(hipMemcpy(s, hs, ni * nj * 8, hipMemcpyHostToDevice));
            CHK(hipMemset(d, 0xCC, ni * nj * 8));
            const int64_t nti_ = ni / 64;
            const int64_t njc_ = (nj / 64 + 3) / 4;
            hipLaunchKernelGGL((t_sweep<64, 64, 8, 4>),
                               dim3((uint32_t)(nti_ * njc_)), dim3(64, 8), 0,
                               0, (const uint64_t *)s, (uint64_t *)d, ni, nj,
                               nti_);
            CHK(hipMemcpy(hd, d, ni * nj * 8, hipMemcpyDeviceToHost));
            int64_t bad = 0;
            for (int64_t j = 0; j < nj; j++)
                for (int64_t i = 0; i < ni; i++)
                    if (hd[j + nj * i] != hs[i + ni * j]) bad++;
            printf("t_sweep correctness: %s (%lld bad)\n", bad ? "FAIL" : "OK",
                   (long long)bad);
            /* t_sweep_vec / t_sweep_db on the same pattern (ni,nj multiples
             * of 128/64) */
            {
                const int64_t ni2 = 256, nj2 = 384;
                CHK(hipMemset(d, 0xCC, ni2 * nj2 * 8));
                const int64_t nti2 = ni2 / 128, njc2 = (nj2 / 64 + 3) / 4;
                hipLaunchKernelGGL((t_sweep_vec<16, 4>),
                                   dim3((uint32_t)(nti2 * njc2)),
                                   dim3(64, 16), 0, 0, (const uint64_t *)s,
                                   (uint64_t *)d, ni2, nj2, nti2);
                CHK(hipMemcpy(hd, d, ni2 * nj2 * 8, hipMemcpyDeviceToHost));
                int64_t bad2 = 0;
                for (int64_t j = 0; j < nj2; j++)
                    for (int64_t i = 0; i < ni2; i++)
                        if (hd[j + nj2 * i] != hs[i + ni2 * j]) bad2++;
                printf("t_sweep_vec correctness: %s (%lld bad)\n",
                       bad2 ? "FAIL" : "OK", (long long)bad2);
                CHK(hipMemset(d, 0xCC, ni2 * nj2 * 8));
                hipLaunchKernelGGL((t_sweep_db<16, 4>),
                                   dim3((uint32_t)(nti2 * njc2)),
                                   dim3(64, 16), 0, 0, (const uint64_t *)s,
                                   (uint64_t *)d, ni2, nj2, nti2);
                CHK(hipMemcpy(hd, d, ni2 * nj2 * 8, hipMemcpyDeviceToHost));
                bad2 = 0;
                for (int64_t j = 0; j < nj2; j++)
                    for (int64_t i = 0; i < ni2; i++)
                        if (hd[j + nj2 * i] != hs[i + ni2 * j]) bad2++;
                printf("t_sweep_db correctness: %s (%lld bad)\n",
                       bad2 ? "FAIL" : "OK", (long long)bad2);
            }
            free(hs);
            free(hd);
        }
        /* correctness of t_rect_jf on a small pattern */
        {
            const int64_t ni = 256, nj = 384;
            uint64_t *hs = (uint64_t *)malloc(ni * nj * 8);
            uint64_t *hd = (uint64_t *)malloc(ni * nj * 8);
            for (int64_t i = 0; i < ni * nj; i++) hs[i] = i * 2654435761ULL;
            CHK(hipMemcpy(s, hs, ni * nj * 8, hipMemcpyHostToDevice));
            CHK(hipMemset(d, 0xCC, ni * nj * 8));
            hipLaunchKernelGGL((t_rect_jf<64, 64, 8>),
                               dim3((uint32_t)((ni / 64) * (nj / 64))),
                               dim3(64, 8), 0, 0, (const uint64_t *)s,
                               (uint64_t *)d, ni, nj, nj / 64);
            CHK(hipMemcpy(hd, d, ni * nj * 8, hipMemcpyDeviceToHost));
            int64_t bad = 0;
            for (int64_t j = 0; j < nj; j++)
                for (int64_t i = 0; i < ni; i++)
                    if (hd[j + nj * i] != hs[i + ni * j]) bad++;
            printf("t_rect_jf correctness: %s (%lld bad)\n",
                   bad ? "FAIL" : "OK", (long long)bad);
            free(hs);
            free(hd);
        }

        /* correctness of t_vec 64x64 on a small pattern */
        {
            const int64_t ni = 256, nj = 192;
            uint64_t *hs = (uint64_t *)malloc(ni * nj * 8);
            uint64_t *hd = (uint64_t *)malloc(ni * nj * 8);
            for (int64_t i = 0; i < ni * nj; i++) hs[i] = i * 2654435761ULL;
            CHK(hipMemcpy(s, hs, ni * nj * 8, hipMemcpyHostToDevice));
            CHK(hipMemset(d, 0xCC, ni * nj * 8));
            hipLaunchKernelGGL((t_vec<64, 8>),
                               dim3((uint32_t)((ni / 64) * (nj / 64))),
                               dim3(64, 8), 0, 0, (const uint64_t *)s,
                               (uint64_t *)d, ni, nj, ni / 64);
            CHK(hipMemcpy(hd, d, ni * nj * 8, hipMemcpyDeviceToHost));
            int64_t bad = 0;
            for (int64_t j = 0; j < nj; j++)
                for (int64_t i = 0; i < ni; i++)
                    if (hd[j + nj * i] != hs[i + ni * j]) bad++;
            printf("t_vec correctness: %s (%lld bad)\n",
                   bad ? "FAIL" : "OK", (long long)bad);
        }
    }

    CHK(hipFree(s));
    CHK(hipFree(d));
    return 0;
}
