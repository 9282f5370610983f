/* pencilhip.h — C ABI of the MI355X-native global-transpose engine.
 *
 * This is the drop-in boundary replacing the engine underneath
 * PencilArrays.jl's Transpositions.transpose! hot path.  The reference has no
 * FFI of its own (pure Julia); each entry point below names the reference
 * function whose role it takes, so a Julia host can bind 1:1 via ccall
 * (see INTEGRATION.md for the ccall stubs), and any C/C++/Python host can
 * call it directly.  All lengths are int64 (the reference's MPI counts are
 * Cint and warn near 2^31, Pencils.jl:363-380; this ABI has no such cliff).
 *
 * Conventions: dims/permutations 0-based; ranges half-open; a permutation q
 * maps memory axes to logical dims (q[i] = logical dim at memory position i,
 * memory axis 0 fastest — the column-major Julia parent, arrays.jl:134-138).
 * Parent arrays are flat device buffers of length prod(local memory dims).
 *
 * Thread model: one process (or host thread) per GPU rank; a plan is
 * rank-local like the reference's Transposition (Transpositions.jl:70-92).
 * All execution is asynchronous on the caller's HIP stream.
 */

#ifndef PENCILHIP_H
#define PENCILHIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct pa_topology pa_topology; /* MPITopology{M}, MPITopologies.jl:72-119 */
typedef struct pa_pencil pa_pencil;     /* Pencil{N,M},    Pencils.jl:151-251    */
typedef struct pa_comm pa_comm;         /* one RCCL communicator = one 1-D
                                         * subcommunicator (MPITopologies.jl:244-251) */
typedef struct pa_plan pa_plan;         /* Transposition,  Transpositions.jl:94-119 */

typedef int pa_status; /* 0 = ok; nonzero = error, see pa_last_error() */

const char *pa_last_error(void);

/* ---- topology: process grid, rank maps ------------------------------- */
/* MPITopology(comm, pdims) with reorder=false => row-major rank order
 * (MPITopologies.jl:125-131); subgroup rank along a dim == coordinate
 * (:229-242). */
pa_status pa_topology_create(int m, const int64_t *pdims, pa_topology **out);
void pa_topology_destroy(pa_topology *t);
int pa_topology_nranks(const pa_topology *t);
pa_status pa_topology_coords(const pa_topology *t, int rank, int64_t *coords);

/* ---- pencil: decomposition metadata ---------------------------------- */
/* Pencil(topology, size_global, decomp_dims; permute) (Pencils.jl:238-251).
 * perm may be NULL (NoPermutation). */
pa_status pa_pencil_create(pa_topology *t, int n, const int64_t *size_global,
                           const int32_t *decomp_dims, const int32_t *perm,
                           pa_pencil **out);
void pa_pencil_destroy(pa_pencil *p);
/* range_local(p, rank, order) (Pencils.jl:512-514): n half-open ranges into
 * lo[n], hi[n]; memory_order != 0 permutes (axes_local_perm). */
pa_status pa_pencil_range_local(const pa_pencil *p, int rank, int memory_order,
                                int64_t *lo, int64_t *hi);
/* size_local / length_local (Pencils.jl:495,546) incl. nothing extra. */
pa_status pa_pencil_size_local(const pa_pencil *p, int rank, int memory_order,
                               int64_t *out);
int64_t pa_pencil_length_local(const pa_pencil *p, int rank);

/* ---- RCCL communicator bootstrap ------------------------------------- */
/* The host exchanges the opaque unique id out-of-band (MPI, TCP store, ...)
 * and every member of one 1-D subgroup calls pa_comm_create with the same id
 * and its subgroup rank (= its Cartesian coordinate along the transposed
 * dimension).  Replaces MPI.Cart_sub / topology.subcomms[R]
 * (MPITopologies.jl:244-251, Transpositions.jl:295-298). */
int pa_unique_id_size(void);
pa_status pa_get_unique_id(char *id /* pa_unique_id_size() bytes */);
pa_status pa_comm_create(const char *id, int nranks, int rank, pa_comm **out);
void pa_comm_destroy(pa_comm *c);

/* ---- transposition plan ---------------------------------------------- */
/* Transposition(Ao, Ai) (Transpositions.jl:94-119): validates compatibility
 * (:182-199), finds the transposed dimension R (:111), and precomputes every
 * pack/exchange/unpack block (:346-536).  elem_size in bytes (the path is
 * pure data movement; dtype enters via size only).  extra_dims (E slowest
 * axes, arrays.jl:105-106) may be NULL with e = 0.
 * Unlike the reference, which re-plans on every transpose! call (:165-167),
 * a plan is reusable and allocation-free in steady state.
 * flags bit 0 (PA_PLAN_ALIASED): src and dst parents may alias (the
 * reference's in-place / ManyPencilArray transposes, multiarrays.jl:106-143,
 * Transpositions.jl:250-264): the self block then stages through the
 * recv-buffer tail so every src read completes before any dst write.
 * Env PENCILHIP_EXCHANGE_CHUNKS=N (read at plan creation, default 1):
 * split the exchange into N chunk groups and unpack each chunk while later
 * chunks are in flight (the reference's Waitany overlap,
 * Transpositions.jl:510-517). */
#define PA_PLAN_ALIASED 1
pa_status pa_plan_create(const pa_pencil *pin, const pa_pencil *pout,
                         int64_t elem_size, int e, const int64_t *extra_dims,
                         int rank, int flags, pa_plan **out);
void pa_plan_destroy(pa_plan *p);

/* Subgroup communicator for the exchange (required when the subgroup size
 * P > 1; its nranks must equal P and its rank my coordinate along R). */
pa_status pa_plan_set_comm(pa_plan *p, pa_comm *c);

/* Staging buffers (the send_buf/recv_buf of Pencils.jl:187-189).  Either ask
 * the sizes and provide device memory, or skip: execute() allocates with
 * hipMalloc on first use. */
pa_status pa_plan_buffer_sizes(const pa_plan *p, int64_t *send_bytes,
                               int64_t *recv_bytes);
pa_status pa_plan_set_buffers(pa_plan *p, void *send_buf, void *recv_buf);

/* transpose!(t) (Transpositions.jl:161-180): pack -> RCCL grouped
 * send/recv -> fused local copy -> unpack, asynchronous on `stream`
 * (hipStream_t; pass 0 for the default stream; from PyTorch use
 * torch.cuda.current_stream().cuda_stream).  src/dst are device pointers to
 * the parent flats; they must not alias. */
pa_status pa_transpose_execute(pa_plan *p, const void *src_parent,
                               void *dst_parent, void *stream);
/* MPI.Waitall(t) (Transpositions.jl:128-131): wait for completion. */
pa_status pa_transpose_wait(pa_plan *p, void *stream);

/* ---- per-stage timing -------------------------------------------------- */
/* The TimerOutputs.@timeit_debug analogue (Transpositions.jl:173-177,327,
 * 337): opt-in HIP-event timing of the last execute's stages.  Call
 * pa_plan_stage_times after pa_transpose_wait; out = {pack_ms, local_ms,
 * exchange_ms, unpack_ms} (-1 for stages the plan does not run; exchange is
 * timed on the engine's comm stream, overlapping local_ms by design). */
pa_status pa_plan_enable_timing(pa_plan *p, int enable);
pa_status pa_plan_stage_times(pa_plan *p, double out[4]);

/* ---- plan introspection (host-side, for parity tests) ---------------- */
int pa_plan_nproc_sub(const pa_plan *p);      /* P (1 => purely local)  */
int pa_plan_r_dim(const pa_plan *p);          /* R, or -1 if same decomp */
int pa_plan_my_k(const pa_plan *p);
pa_status pa_plan_block_info(const pa_plan *p, int k, int64_t out[8]);
/* which: 0 = local fused copy, 1 = pack of peer k, 2 = unpack of peer k,
 * 3 = staged self pack (aliased mode), 4 = staged self unpack.
 * Returns the normalized strided-copy descriptor (element units):
 * nd, dims[nd], sstrides[nd], soffset, dstrides[nd], doffset. */
pa_status pa_plan_copydesc(const pa_plan *p, int which, int k, int64_t *nd,
                           int64_t *dims, int64_t *sstr, int64_t *soff,
                           int64_t *dstr, int64_t *doff);

/* ---- reductions (src/reductions.jl:9-38) ----------------------------- */
/* One ncclAllReduce over a communicator (normally the FULL topology comm):
 * the collective behind the reference's mapreduce/any/all.
 * dtype: 0=f64, 1=f32, 2=i64, 3=i32, 4=u8; op: 0=sum, 1=prod, 2=min, 3=max.
 * In-place allowed (send == recv). */
pa_status pa_allreduce(pa_comm *c, const void *sendbuf, void *recvbuf,
                       int64_t count, int dtype, int op, void *stream);

/* ---- standalone device copy (used by tests/benchmarks) --------------- */
/* Execute one strided-copy descriptor on device (same kernels the plan
 * uses).  All strides/offsets/dims in elements of elem_size bytes. */
pa_status pa_device_copy(int nd, const int64_t *dims, const int64_t *sstr,
                         int64_t soff, const int64_t *dstr, int64_t doff,
                         int64_t elem_size, const void *src, void *dst,
                         void *stream);

#ifdef __cplusplus
}
#endif

#endif /* PENCILHIP_H */
