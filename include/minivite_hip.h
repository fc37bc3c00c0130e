/* minivite_hip.h — C-ABI drop-in boundary of the MI355X-native miniVite
 * Louvain implementation.
 *
 * The reference's plugin surface is the distLouvainMethod call plus the
 * Graph accessors it consumes (caller: main.cpp:164-170; signature
 * dspl.hpp:1280-1284; Graph reads graph.hpp:140-203). The six comm-buffer
 * parameters of the reference signature are opaque scratch the caller only
 * declares (main.cpp:153-157) and are owned by the engine here. Each entry
 * point below cites the reference interface it replaces. See INTEGRATION.md
 * for the binding a maintainer would add to the reference's main.cpp.
 *
 * Process model: one process per GPU ("rank" == the reference's MPI rank;
 * "nranks" == its nprocs). Collectives ride RCCL over xGMI; rank 0 creates
 * the 128-byte RCCL unique id (mv_comm_id) and the caller transports it to
 * the other ranks (any side channel; bench.py uses a torch.distributed gloo
 * broadcast). Single-rank runs pass NULL.
 */
#ifndef MINIVITE_HIP_H
#define MINIVITE_HIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- graph construction (replaces graph.hpp plumbing, host-side) ---- */

/* Per-rank partitioned CSR (replaces class Graph, graph.hpp:85-296).
 * Host memory; tails are GLOBAL vertex ids. */
typedef struct mv_graph mv_graph;

/* GenerateRGG::generate with isLCG=true (graph.hpp:584-1213 via the -l
 * path). Produces the BIT-IDENTICAL per-rank edge set using a cell-list
 * instead of the reference's O((nv/p)^2) pair loops; needs no
 * communication (the LCG stream is leapfrogged locally). Requirements as
 * the reference: nranks a power of two, nranks | nv, 1/nranks > rn
 * (graph.hpp:610-633). random_edge_percent > 0 mirrors the -p option
 * (graph.hpp:939-1122) except that the seed is explicit instead of
 * time(0)^getpid() (graph.hpp:990) and the O(n) duplicate scan
 * (graph.hpp:1013-1018) is not reproduced (perf configs only). */
mv_graph *mv_graph_rgg(int64_t nv, int rank, int nranks,
                       int unit_edge_weight, double random_edge_percent,
                       uint64_t random_edge_seed);

/* BinaryEdgeList::read / read_balanced (graph.hpp:315-412 / 416-572).
 * Single-node local file, plain pread instead of MPI-IO. */
mv_graph *mv_graph_read_binary(const char *path, int rank, int nranks,
                               int balanced);

/* Adopt caller-owned arrays (copied). parts has nranks+1 entries
 * (graph.hpp:112-113 layout); weights NULL means unit weights. */
mv_graph *mv_graph_from_csr(int64_t nv, int rank, int nranks,
                            const int64_t *parts, int64_t lnv, int64_t lne,
                            const int64_t *xadj, const int64_t *tails,
                            const double *weights);

/* Write this rank's slice of the reference's binary format is not needed;
 * mv_graph_write_binary writes a WHOLE single-rank graph (rank count 1)
 * to the reference .bin layout (graph.hpp:342-383) so the CPU reference
 * can consume framework-generated inputs via -f. */
int mv_graph_write_binary(const mv_graph *g, const char *path);

void mv_graph_free(mv_graph *g);

/* Accessors (graph.hpp:175-178 get_lnv/get_lne/get_nv + parts). */
int64_t mv_graph_nv(const mv_graph *g);
int64_t mv_graph_lnv(const mv_graph *g);
int64_t mv_graph_lne(const mv_graph *g);
const int64_t *mv_graph_parts(const mv_graph *g); /* nranks+1 */
const int64_t *mv_graph_xadj(const mv_graph *g);  /* lnv+1 */
const int64_t *mv_graph_tails(const mv_graph *g); /* lne */
const double *mv_graph_weights(const mv_graph *g);/* lne */
/* Internal-layout hint (NULL if absent): locality_perm[k] = original local
 * vertex id at internal position k, spatially ordered. Layout metadata
 * only — results are identical with or without it. */
const int32_t *mv_graph_locality_hint(const mv_graph *g);

/* ---- the hot path (replaces distLouvainMethod, dspl.hpp:1280-1441) ---- */

typedef struct mv_engine mv_engine;

#define MV_COMM_ID_BYTES 128
/* ncclGetUniqueId wrapper (rank 0 only; returns 0 on success). */
int mv_comm_id(void *id_128_bytes);

/* Create the engine on HIP device `device`. nranks > 1 requires the 128-byte
 * id from mv_comm_id on every rank. Aborts loudly (non-zero + message on
 * stderr) when no GPU or no RCCL — there is no CPU fallback. */
mv_engine *mv_engine_create(int device, int rank, int nranks,
                            const void *comm_id_or_null);
void mv_engine_destroy(mv_engine *e);

/* ---- loopback transport (hardware-validation harness) ----
 * Runs nranks engines in ONE process on ONE device, one host thread per
 * rank; the collectives become device-to-device copies + host barriers
 * while every kernel and every byte of the multi-rank protocol stays the
 * p>1 production path. Purpose: executing the partitioned engine on a
 * single-GPU box, where RCCL refuses two ranks on one device. The product
 * multi-GPU path (bench.py, mv355) always uses RCCL. */
typedef struct mv_lb_session mv_lb_session;
mv_lb_session *mv_lb_create(int nranks);
void mv_lb_destroy(mv_lb_session *s);
/* Like mv_engine_create but exchanges ride the loopback session. */
mv_engine *mv_engine_create_lb(int device, int rank, int nranks,
                               mv_lb_session *s);

/* Upload the per-rank CSR to HBM (device layout: DESIGN.md §data layout). */
int mv_engine_load_graph(mv_engine *e, const mv_graph *g);

/* Run Louvain phase 1 on the loaded graph: the drop-in for
 * distLouvainMethod(me, nprocs, dg, ..., lower, thresh, iters)
 * (dspl.hpp:1280-1284). Includes the exchangeVertexReqs setup
 * (dspl.hpp:1112-1272) like the reference's timed span. Returns prevMod
 * (the PREVIOUS iteration's modularity — dspl.hpp:1440); *iters_out gets
 * the iteration count. Collective: all ranks must call together. */
double mv_engine_run(mv_engine *e, double lower, double thresh,
                     int *iters_out);

/* Optional parity trace: before mv_engine_run, point the engine at caller
 * buffers receiving, per iteration k (1-based), this rank's targetComm
 * (dspl.hpp:404) at target_trace[(k-1)*lnv .. k*lnv) and the global
 * modularity at mod_trace[k-1]. cap bounds the recorded iterations.
 * Pass NULLs to disable. */
void mv_engine_set_trace(mv_engine *e, int64_t *target_trace,
                         double *mod_trace, int cap);

/* Perf counters for the last mv_engine_run (HIP-event timed, per stream). */
typedef struct {
    double total_ms;        /* whole run (setup + loop) */
    double sweep_ms;        /* K4 distExecuteLouvainIteration kernel time */
    int64_t sweep_launches;
    double halo_ms;         /* RCCL exchanges + pack/unpack */
    double setup_ms;        /* exchangeVertexReqs equivalent */
    int64_t edges_local;    /* directed edges on this rank */
    int iters;
} mv_stats;
void mv_engine_get_stats(const mv_engine *e, mv_stats *out);

#ifdef __cplusplus
}
#endif

#endif /* MINIVITE_HIP_H */
