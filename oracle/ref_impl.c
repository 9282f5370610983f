/* CPU oracle: line-faithful C restatement of PencilArrays.jl's
 * Transpositions.transpose! — TEST INFRASTRUCTURE + CPU BASELINE ONLY.
 *
 * ORACLE ONLY: the product path never links or calls this file.  Consumers:
 * tests/ (parity pin), bench.py's cpu_baseline leg, __graft_entry__.build()
 * (compiles it), __graft_entry__.smoke() (checks the GPU path against it).
 *
 * Restates, with N ranks simulated in one process and memcpy standing in for
 * MPI (BASELINE.md "CPU-baseline plan"):
 *   - split formula / axes           data_ranges.jl:4-9, 15-45
 *   - to_local                       Pencils.jl:579-587
 *   - topology rank order            MPITopologies.jl:125-131 (Cart_create
 *                                    reorder=false => row-major rank order),
 *                                    subgroup rank = coordinate (:229-242)
 *   - control flow                   Transpositions.jl:282-344
 *   - peer enumeration               Transpositions.jl:542-552
 *   - pack (copy_range!)             Transpositions.jl:554-586 (column-major
 *                                    over the block in Pi MEMORY order,
 *                                    extra dims outermost)
 *   - self block to recv tail        Transpositions.jl:394-404
 *   - exchange                       Transpositions.jl:419-428/463-479
 *                                    (memcpy here)
 *   - unpack (copy_permuted!)        Transpositions.jl:489-536, 588-667
 *     with perm = permutation(Po)/permutation(Pi)  (:506)
 *   - local-only path                Transpositions.jl:214-271
 *   - permutation operators          StaticPermutations.jl v0.3 semantics
 *                                    (not vendored by the reference; derived
 *                                    from arrays.jl:19-31 + test/permutations.jl):
 *                                    (p*t)[i]=t[p[i]]; (p/q)[i]=inv(q)[p[i]]
 *
 * Pure data movement: no floating-point arithmetic anywhere, so parity with
 * the GPU engine is exact bit equality.  Element type enters via elem_size
 * only.  All dims/perms 0-based; ranges half-open.
 *
 * Build: gcc -O3 -fopenmp -shared -fPIC ref_impl.c -o liboracle.so
 *        gcc -O3 -fopenmp ref_impl.c -DORACLE_MAIN -o oracle_bench
 */

#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#ifdef _OPENMP
#include <omp.h>
#endif

#define MAXD 10 /* max spatial + extra dims */

typedef struct {
    int64_t lo, hi; /* half-open */
} Range;

/* ---------- metadata (independent restatement) ---------------------- */

static void split_range(int64_t c, int64_t P, int64_t N, Range *r)
{
    /* data_ranges.jl:4-9 */
    r->lo = (N * c) / P;
    r->hi = (N * (c + 1)) / P;
}

static void cart_coords(int rank, int M, const int64_t *pdims, int64_t *coords)
{
    int64_t rem = rank;
    for (int i = 0; i < M; i++) {
        int64_t stride = 1;
        for (int j = i + 1; j < M; j++) stride *= pdims[j];
        coords[i] = rem / stride;
        rem %= stride;
    }
}

static int cart_rank(int M, const int64_t *pdims, const int64_t *coords)
{
    int64_t r = 0;
    for (int i = 0; i < M; i++) r = r * pdims[i] + coords[i];
    return (int)r;
}

/* region owned by process at topology coords (data_ranges.jl:30-45) */
static void axes_for_coords(int N, const int64_t *size_global, int M,
                            const int64_t *pdims, const int32_t *decomp,
                            const int64_t *coords, Range *out)
{
    for (int d = 0; d < N; d++) { out[d].lo = 0; out[d].hi = size_global[d]; }
    for (int j = 0; j < M; j++)
        split_range(coords[j], pdims[j], size_global[decomp[j]],
                    &out[decomp[j]]);
}

static void range_intersect(const Range *a, const Range *b, Range *out)
{
    int64_t lo = a->lo > b->lo ? a->lo : b->lo;
    int64_t hi = a->hi < b->hi ? a->hi : b->hi;
    out->lo = lo;
    out->hi = hi > lo ? hi : lo;
}

static void perm_inv(int N, const int32_t *p, int32_t *out)
{
    for (int i = 0; i < N; i++) out[p[i]] = i;
}

/* ---------- problem description ------------------------------------- */

typedef struct {
    int N;                    /* spatial dims */
    int M;                    /* topology dims */
    int E;                    /* extra dims */
    int64_t size_global[MAXD];
    int64_t pdims[MAXD];
    int32_t decomp_i[MAXD], decomp_o[MAXD];
    int32_t perm_i[MAXD], perm_o[MAXD]; /* perm[i] = logical dim at mem pos i */
    int64_t extra[MAXD];
    int64_t elem_size;
} Problem;

static int64_t prod_extra(const Problem *pb)
{
    int64_t p = 1;
    for (int e = 0; e < pb->E; e++) p *= pb->extra[e];
    return p;
}

static int64_t region_nelem(int N, const Range *r)
{
    int64_t n = 1;
    for (int d = 0; d < N; d++) n *= (r[d].hi - r[d].lo);
    return n;
}

/* parent local memory dims of a pencil on given coords:
 * mem axis i (fastest first) = logical dim perm[i]; extra dims appended
 * (arrays.jl:134-138) */
static void local_mem_dims(const Problem *pb, const int32_t *decomp,
                           const int32_t *perm, const int64_t *coords,
                           int64_t *mem /* N+E */)
{
    Range ax[MAXD];
    axes_for_coords(pb->N, pb->size_global, pb->M, pb->pdims, decomp, coords, ax);
    for (int i = 0; i < pb->N; i++)
        mem[i] = ax[perm[i]].hi - ax[perm[i]].lo;
    for (int e = 0; e < pb->E; e++) mem[pb->N + e] = pb->extra[e];
}

/* ---------- copy primitives ------------------------------------------ */

/* Strided N-d copy: dst[doff + sum j*dstr] = src[soff + sum j*sstr], dims K.
 * Axis 0 is the inner loop; memcpy when both inner strides are 1.
 * Strides/offsets in ELEMENTS; elem_size bytes each. */
static void copy_strided(int K, const int64_t *dims,
                         const char *src, const int64_t *sstr, int64_t soff,
                         char *dst, const int64_t *dstr, int64_t doff,
                         int64_t esz)
{
    int64_t total = 1;
    for (int a = 0; a < K; a++) total *= dims[a];
    if (total == 0) return;
    int64_t inner = dims[0];
    int64_t nouter = total / inner;
    const int contig = (K == 0) || (sstr[0] == 1 && dstr[0] == 1);

#ifdef _OPENMP
#pragma omp parallel for schedule(static) if (nouter > 1)
#endif
    for (int64_t o = 0; o < nouter; o++) {
        int64_t rem = o, so = soff, dof = doff;
        for (int a = 1; a < K; a++) {
            int64_t j = rem % dims[a];
            rem /= dims[a];
            so += j * sstr[a];
            dof += j * dstr[a];
        }
        const char *s = src + so * esz;
        char *d = dst + dof * esz;
        if (contig) {
            memcpy(d, s, (size_t)(inner * esz));
        } else if (esz == 8) {
            const int64_t ss = sstr[0], ds = dstr[0];
            const uint64_t *s8 = (const uint64_t *)s;
            uint64_t *d8 = (uint64_t *)d;
            for (int64_t j = 0; j < inner; j++) d8[j * ds] = s8[j * ss];
        } else if (esz == 4) {
            const int64_t ss = sstr[0], ds = dstr[0];
            const uint32_t *s4 = (const uint32_t *)s;
            uint32_t *d4 = (uint32_t *)d;
            for (int64_t j = 0; j < inner; j++) d4[j * ds] = s4[j * ss];
        } else {
            const int64_t ss = sstr[0], ds = dstr[0];
            for (int64_t j = 0; j < inner; j++)
                memcpy(d + j * ds * esz, s + j * ss * esz, (size_t)esz);
        }
    }
}

/* column-major strides */
static void colmajor(int K, const int64_t *dims, int64_t *str)
{
    int64_t acc = 1;
    for (int a = 0; a < K; a++) { str[a] = acc; acc *= dims[a]; }
}

/* ---------- the transpose, one rank ---------------------------------- */

/* Pack one peer block: copy_range!(buf, off, Ai, local_send_range)
 * (Transpositions.jl:554-586): iterate the block column-major over Pi MEMORY
 * axes, extra dims outermost, writing contiguously at buf+off. */
static void pack_block(const Problem *pb, const int64_t *coords,
                       const Range *srange /* global logical */,
                       const char *src_parent, char *buf, int64_t off_elems)
{
    int N = pb->N, E = pb->E, K = N + E;
    Range ax[MAXD];
    axes_for_coords(N, pb->size_global, pb->M, pb->pdims, pb->decomp_i,
                    coords, ax);
    int64_t mem[MAXD], pst[MAXD];
    local_mem_dims(pb, pb->decomp_i, pb->perm_i, coords, mem);
    colmajor(K, mem, pst);

    /* window dims + src offset in Pi memory order (to_local, Pencils.jl:579-587) */
    int64_t dims[MAXD], soff = 0;
    for (int i = 0; i < N; i++) {
        int d = pb->perm_i[i];
        dims[i] = srange[d].hi - srange[d].lo;
        soff += (srange[d].lo - ax[d].lo) * pst[i];
    }
    for (int e = 0; e < E; e++) dims[N + e] = pb->extra[e];

    int64_t dstr[MAXD];
    colmajor(K, dims, dstr);
    copy_strided(K, dims, src_parent, pst, soff, buf, dstr, off_elems,
                 pb->elem_size);
}

/* Unpack one block: copy_permuted!(Ao, o_range_iperm, buf, off, perm)
 * (Transpositions.jl:588-606): buffer is column-major over the block extents
 * in Pi memory order (+extras); scatter into the Po parent window. */
static void unpack_block(const Problem *pb, const int64_t *coords,
                         const Range *grange /* global logical */,
                         const char *buf, int64_t off_elems, char *dst_parent)
{
    int N = pb->N, E = pb->E, K = N + E;
    Range axo[MAXD];
    axes_for_coords(N, pb->size_global, pb->M, pb->pdims, pb->decomp_o,
                    coords, axo);
    int64_t mem_o[MAXD], pst_o[MAXD];
    local_mem_dims(pb, pb->decomp_o, pb->perm_o, coords, mem_o);
    colmajor(K, mem_o, pst_o);

    /* buffer axis j corresponds to logical dim perm_i[j] (j<N), extra e at N+e */
    int64_t dims[MAXD];
    for (int j = 0; j < N; j++)
        dims[j] = grange[pb->perm_i[j]].hi - grange[pb->perm_i[j]].lo;
    for (int e = 0; e < E; e++) dims[N + e] = pb->extra[e];
    int64_t sstr[MAXD];
    colmajor(K, dims, sstr);

    /* dest: mem axis i' holds logical dim perm_o[i']; it reads buffer axis
     * inv(perm_i)[perm_o[i']]  (the relative permutation, :506). */
    int32_t ipi[MAXD];
    perm_inv(N, pb->perm_i, ipi);
    int64_t dstr[MAXD], doff = 0;
    for (int ip = 0; ip < N; ip++) {
        int d = pb->perm_o[ip];
        int j = ipi[d];
        dstr[j] = pst_o[ip];
        doff += (grange[d].lo - axo[d].lo) * pst_o[ip];
    }
    for (int e = 0; e < E; e++) dstr[N + e] = pst_o[N + e];

    copy_strided(K, dims, buf, sstr, off_elems, dst_parent, dstr, doff,
                 pb->elem_size);
}

/* Full transpose for all nranks simulated in-process.
 * src_parents/dst_parents: arrays of nranks pointers to parent flats. */
void oracle_transpose_all(
    int N, const int64_t *size_global,
    int M, const int64_t *pdims,
    const int32_t *decomp_i, const int32_t *perm_i,
    const int32_t *decomp_o, const int32_t *perm_o,
    int E, const int64_t *extra, int64_t elem_size,
    const char *const *src_parents, char *const *dst_parents)
{
    Problem pb;
    pb.N = N; pb.M = M; pb.E = E; pb.elem_size = elem_size;
    memcpy(pb.size_global, size_global, N * sizeof(int64_t));
    memcpy(pb.pdims, pdims, M * sizeof(int64_t));
    memcpy(pb.decomp_i, decomp_i, M * sizeof(int32_t));
    memcpy(pb.decomp_o, decomp_o, M * sizeof(int32_t));
    memcpy(pb.perm_i, perm_i, N * sizeof(int32_t));
    memcpy(pb.perm_o, perm_o, N * sizeof(int32_t));
    for (int e = 0; e < E; e++) pb.extra[e] = extra[e];

    int nranks = 1;
    for (int j = 0; j < M; j++) nranks *= (int)pdims[j];
    int64_t pex = prod_extra(&pb);

    /* R = single differing decomposed dim (Transpositions.jl:111) */
    int R = -1;
    for (int j = 0; j < M; j++)
        if (decomp_i[j] != decomp_o[j]) { R = j; break; }

    /* staging buffers per rank (send laid out k-ascending skipping self;
     * recv likewise with the self block at the END, :394-404) */
    char **send_bufs = calloc(nranks, sizeof(char *));
    char **recv_bufs = calloc(nranks, sizeof(char *));
    int64_t *recv_total = calloc(nranks, sizeof(int64_t));

    for (int r = 0; r < nranks; r++) {
        int64_t coords[MAXD];
        cart_coords(r, M, pb.pdims, coords);
        Range axl_o[MAXD];
        axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_o, coords,
                        axl_o);
        int64_t len_o = region_nelem(N, axl_o) * pex;
        recv_total[r] = len_o; /* includes self block at tail (:317) */
        recv_bufs[r] = malloc((size_t)((len_o ? len_o : 1) * elem_size));
        Range axl_i[MAXD];
        axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_i, coords,
                        axl_i);
        int64_t len_i = region_nelem(N, axl_i) * pex;
        send_bufs[r] = malloc((size_t)((len_i ? len_i : 1) * elem_size));
    }

    if (R < 0) {
        /* local-only path (Transpositions.jl:214-271): copy or permute. */
        for (int r = 0; r < nranks; r++) {
            int64_t coords[MAXD];
            cart_coords(r, M, pb.pdims, coords);
            Range axl[MAXD];
            axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_i,
                            coords, axl);
            /* pack whole local block then unpack: equivalent to
             * permute_local! staging through recv_buf (:250-264) */
            pack_block(&pb, coords, axl, src_parents[r], recv_bufs[r], 0);
            unpack_block(&pb, coords, axl, recv_bufs[r], 0, dst_parents[r]);
        }
        goto done;
    }

    {
        int P = (int)pb.pdims[R];
        /* offsets per rank per peer; then pack, "exchange", unpack */
        for (int r = 0; r < nranks; r++) {
            int64_t coords[MAXD];
            cart_coords(r, M, pb.pdims, coords);
            Range axl_i[MAXD], axl_o[MAXD];
            axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_i,
                            coords, axl_i);
            axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_o,
                            coords, axl_o);

            /* length_self (:303-306) */
            Range self_r[MAXD];
            for (int d = 0; d < N; d++)
                range_intersect(&axl_i[d], &axl_o[d], &self_r[d]);
            int64_t len_self = region_nelem(N, self_r) * pex;
            int64_t len_recv_remote = recv_total[r] - len_self;

            int64_t isend = 0, irecv = 0;
            int myk = (int)coords[R];
            for (int k = 0; k < P; k++) {
                int64_t pc[MAXD];
                memcpy(pc, coords, M * sizeof(int64_t));
                pc[R] = k;
                Range axp_o[MAXD], axp_i[MAXD], sr[MAXD], rr[MAXD];
                axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_o,
                                pc, axp_o);
                axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_i,
                                pc, axp_i);
                for (int d = 0; d < N; d++) {
                    range_intersect(&axl_i[d], &axp_o[d], &sr[d]); /* :383 */
                    range_intersect(&axl_o[d], &axp_i[d], &rr[d]); /* :388 */
                }
                int64_t ns = region_nelem(N, sr) * pex;
                int64_t nr = region_nelem(N, rr) * pex;
                if (k == myk) {
                    /* self: copy directly into recv tail (:394-404) */
                    if (ns)
                        pack_block(&pb, coords, sr, src_parents[r],
                                   recv_bufs[r], len_recv_remote);
                } else {
                    if (ns)
                        pack_block(&pb, coords, sr, src_parents[r],
                                   send_bufs[r], isend);
                    isend += ns;
                    irecv += nr;
                }
            }
        }

        /* exchange + unpack per rank (memcpy stands in for MPI) */
        for (int r = 0; r < nranks; r++) {
            int64_t coords[MAXD];
            cart_coords(r, M, pb.pdims, coords);
            Range axl_i[MAXD], axl_o[MAXD];
            axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_i,
                            coords, axl_i);
            axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_o,
                            coords, axl_o);
            Range self_r[MAXD];
            for (int d = 0; d < N; d++)
                range_intersect(&axl_i[d], &axl_o[d], &self_r[d]);
            int64_t len_self = region_nelem(N, self_r) * pex;
            int64_t len_recv_remote = recv_total[r] - len_self;

            int myk = (int)coords[R];
            int64_t irecv = 0;
            for (int k = 0; k < P; k++) {
                int64_t pc[MAXD];
                memcpy(pc, coords, M * sizeof(int64_t));
                pc[R] = k;
                Range axp_i[MAXD], rr[MAXD];
                axes_for_coords(N, pb.size_global, M, pb.pdims, pb.decomp_i,
                                pc, axp_i);
                for (int d = 0; d < N; d++)
                    range_intersect(&axl_o[d], &axp_i[d], &rr[d]);
                int64_t nr = region_nelem(N, rr) * pex;
                int64_t off;
                if (k == myk) {
                    off = len_recv_remote;
                } else {
                    off = irecv;
                    /* fetch the matching block from the peer's send buffer:
                     * peer's send offset for me = sum over k' (skipping the
                     * peer's own k') of its send lengths before my index. */
                    int peer = cart_rank(M, pb.pdims, pc);
                    int64_t peer_isend = 0;
                    Range axpl_i[MAXD];
                    axes_for_coords(N, pb.size_global, M, pb.pdims,
                                    pb.decomp_i, pc, axpl_i);
                    for (int k2 = 0; k2 < myk; k2++) {
                        if (k2 == k) continue; /* peer's own coord along R */
                        int64_t qc[MAXD];
                        memcpy(qc, pc, M * sizeof(int64_t));
                        qc[R] = k2;
                        Range axq_o[MAXD], sr2[MAXD];
                        axes_for_coords(N, pb.size_global, M, pb.pdims,
                                        pb.decomp_o, qc, axq_o);
                        for (int d = 0; d < N; d++)
                            range_intersect(&axpl_i[d], &axq_o[d], &sr2[d]);
                        peer_isend += region_nelem(N, sr2) * pex;
                    }
                    if (nr)
                        memcpy(recv_bufs[r] + off * pb.elem_size,
                               send_bufs[peer] + peer_isend * pb.elem_size,
                               (size_t)(nr * pb.elem_size));
                    irecv += nr;
                }
                if (nr)
                    unpack_block(&pb, coords, rr, recv_bufs[r], off,
                                 dst_parents[r]);
            }
        }
    }

done:
    for (int r = 0; r < nranks; r++) { free(send_bufs[r]); free(recv_bufs[r]); }
    free(send_bufs); free(recv_bufs); free(recv_total);
}

/* local parent length helper for hosts */
int64_t oracle_local_len(int N, const int64_t *size_global, int M,
                         const int64_t *pdims, const int32_t *decomp,
                         int E, const int64_t *extra, int rank)
{
    int64_t coords[MAXD];
    cart_coords(rank, M, pdims, coords);
    Range ax[MAXD];
    axes_for_coords(N, size_global, M, pdims, decomp, coords, ax);
    int64_t n = region_nelem(N, ax);
    for (int e = 0; e < E; e++) n *= extra[e];
    return n;
}

#ifdef ORACLE_MAIN
#include <time.h>
static double now_s(void)
{
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec + 1e-9 * ts.tv_nsec;
}

/* Timing main: argv = [Nx Ny Nz P1 P2 reps]
 * x->y transpose: decomp (1,2)->(0,2), no permutation, Float64.
 * Prints effective GiB/s = global bytes / wall per transpose. */
int main(int argc, char **argv)
{
    int64_t nx = argc > 1 ? atoll(argv[1]) : 512;
    int64_t ny = argc > 2 ? atoll(argv[2]) : 512;
    int64_t nz = argc > 3 ? atoll(argv[3]) : 512;
    int64_t p1 = argc > 4 ? atoll(argv[4]) : 1;
    int64_t p2 = argc > 5 ? atoll(argv[5]) : 1;
    int reps = argc > 6 ? atoi(argv[6]) : 3;

    int64_t size_global[3] = { nx, ny, nz };
    int64_t pdims[2] = { p1, p2 };
    int32_t decomp_i[2] = { 1, 2 }, decomp_o[2] = { 0, 2 };
    int32_t perm[3] = { 0, 1, 2 };
    int nranks = (int)(p1 * p2);

    char **src = malloc(nranks * sizeof(char *));
    char **dst = malloc(nranks * sizeof(char *));
    for (int r = 0; r < nranks; r++) {
        int64_t li = oracle_local_len(3, size_global, 2, pdims, decomp_i, 0,
                                      NULL, r);
        int64_t lo = oracle_local_len(3, size_global, 2, pdims, decomp_o, 0,
                                      NULL, r);
        src[r] = malloc((size_t)(li * 8));
        dst[r] = malloc((size_t)(lo * 8));
        uint64_t s = 0xC0FFEE + (uint64_t)r;
        uint64_t *p = (uint64_t *)src[r];
        for (int64_t i = 0; i < li; i++) { /* xorshift64 fill */
            s ^= s << 13; s ^= s >> 7; s ^= s << 17;
            p[i] = s;
        }
    }

    int nthreads = 1;
#ifdef _OPENMP
    nthreads = omp_get_max_threads();
#endif
    double best = 1e30;
    for (int it = 0; it < reps; it++) {
        double t0 = now_s();
        oracle_transpose_all(3, size_global, 2, pdims, decomp_i, perm,
                             decomp_o, perm, 0, NULL, 8,
                             (const char *const *)src, dst);
        double dt = now_s() - t0;
        if (dt < best) best = dt;
    }
    double bytes = (double)nx * ny * nz * 8.0;
    printf("{\"impl\": \"oracle_c\", \"threads\": %d, \"dims\": [%lld,%lld,%lld], "
           "\"grid\": [%lld,%lld], \"seconds\": %.6f, \"gib_per_s\": %.3f}\n",
           nthreads, (long long)nx, (long long)ny, (long long)nz,
           (long long)p1, (long long)p2, best,
           bytes / best / (1024.0 * 1024.0 * 1024.0));
    for (int r = 0; r < nranks; r++) { free(src[r]); free(dst[r]); }
    free(src); free(dst);
    return 0;
}
#endif
