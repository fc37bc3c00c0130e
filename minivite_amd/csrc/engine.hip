// engine.hip — the MI355X-native Louvain phase-1 engine: CDNA4 HIP kernels
// + RCCL-over-xGMI halo exchange behind the C-ABI in include/minivite_hip.h.
//
// Drop-in for distLouvainMethod (dspl.hpp:1280-1441). Kernel inventory maps
// the reference's OpenMP regions: K1 vertex-degree sum
// (dspl.hpp:82-107), K2 constant (109-130), K3 comm iota (132-149), K4 THE
// sweep (276-405 with 230-274, 174-228), K5 zero (473-486), K6 cinfo apply
// (458-471), K7 modularity (407-456), K8 ghost-comm gather (559-571), K9
// cinfo reply gather (776-929), K10 ghost discovery (1112-1272). MPI call
// sites become grouped ncclSend/ncclRecv alltoallv over
// xGMI + ncclAllReduce for the 1-2 double reductions.
//
// FP discipline: built with -ffp-contract=off so the ΔQ gain expression
// (dspl.hpp:212) and all accumulations carry the same bits as the
// gcc-built reference / oracle (generic x86-64 has no FMA contraction).
// Per-vertex weight accumulation is sequential in edge order (one lane owns
// one vertex), matching dspl.hpp:240-271 exactly, so -w results align too.
//
// This is GPU-only code: engine creation fails loudly without a device.
// There is no CPU fallback anywhere in this file.

#include <algorithm>
#include <chrono>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include <hip/hip_runtime.h>
#include <hipcub/hipcub.hpp>
#include <rccl/rccl.h>

#include "../../include/minivite_hip.h"

#define HIP_CHECK(x)                                                          \
    do {                                                                      \
        hipError_t _e = (x);                                                  \
        if (_e != hipSuccess) {                                               \
            std::fprintf(stderr, "HIP error %s at %s:%d: %s\n",               \
                         hipGetErrorString(_e), __FILE__, __LINE__, #x);      \
            std::abort();                                                     \
        }                                                                     \
    } while (0)

#define NCCL_CHECK(x)                                                         \
    do {                                                                      \
        ncclResult_t _e = (x);                                                \
        if (_e != ncclSuccess) {                                              \
            std::fprintf(stderr, "RCCL error %s at %s:%d: %s\n",              \
                         ncclGetErrorString(_e), __FILE__, __LINE__, #x);     \
            std::abort();                                                     \
        }                                                                     \
    } while (0)

namespace {

using i64 = int64_t;

struct DevEdge {   // device edge record, 16 B (mirrors Edge, graph.hpp:60-66,
    i64 tidx;      // with the tail pre-translated: local i -> i, ghost ->
    double w;      // lnv + slot in the sorted ghost list)
};

struct Info16 {    // wire record for cinfo replies / delta routing: the
    i64 size;      // CommInfo payload (dspl.hpp:68-72) minus the community
    double degree; // id, which both ends know by position
};

__device__ __forceinline__ i64 dev_lower_bound(const i64 *a, i64 n, i64 key) {
    i64 lo = 0, hi = n;
    while (lo < hi) {
        i64 mid = (lo + hi) >> 1;
        if (a[mid] < key) lo = mid + 1;
        else hi = mid;
    }
    return lo;
}

__device__ __forceinline__ i64 dev_bsearch(const i64 *a, i64 n, i64 key) {
    i64 p = dev_lower_bound(a, n, key);
    return (p < n && a[p] == key) ? p : -1;
}

__device__ __forceinline__ void atomic_add_i64(i64 *p, i64 v) {
    atomicAdd(reinterpret_cast<unsigned long long *>(p),
              static_cast<unsigned long long>(v));
}

// ---- K1: vDegree + localCinfo init (dspl.hpp:82-107) ----
__global__ void k1_vertex_degree(i64 lnv, const i64 *__restrict__ xadj,
                                 const double *__restrict__ ew,
                                 double *__restrict__ vDegree,
                                 i64 *__restrict__ cinfo_size,
                                 double *__restrict__ cinfo_degree) {
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < lnv;
         i += (i64)gridDim.x * blockDim.x) {
        double tw = 0.0;
        const i64 e1 = xadj[i + 1];
        for (i64 e = xadj[i]; e < e1; e++) tw += ew[e]; // sequential edge order
        vDegree[i] = tw;
        cinfo_degree[i] = tw;
        cinfo_size[i] = 1; // dspl.hpp:104-105
    }
}

// block-partial sum for K2/K7 (deterministic: fixed block ranges, lane-
// strided partials reduced by a fixed shuffle tree; partials summed on host
// in block order)
template <typename F>
__global__ void k_partial_sum2(i64 n, F f, double *__restrict__ out2) {
    double a = 0.0, b = 0.0;
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < n;
         i += (i64)gridDim.x * blockDim.x) {
        double x, y;
        f(i, x, y);
        a += x;
        b += y;
    }
    __shared__ double sa[256], sb[256];
    sa[threadIdx.x] = a;
    sb[threadIdx.x] = b;
    __syncthreads();
    for (int s = blockDim.x / 2; s > 0; s >>= 1) {
        if (threadIdx.x < s) {
            sa[threadIdx.x] += sa[threadIdx.x + s];
            sb[threadIdx.x] += sb[threadIdx.x + s];
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        out2[2 * blockIdx.x] = sa[0];
        out2[2 * blockIdx.x + 1] = sb[0];
    }
}

// ---- K3: community iota (dspl.hpp:132-149) ----
__global__ void k3_init_comm(i64 lnv, i64 base, i64 *__restrict__ curr,
                             i64 *__restrict__ past) {
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < lnv;
         i += (i64)gridDim.x * blockDim.x) {
        curr[i] = i + base;
        past[i] = i + base;
    }
}

// ---- K10a: tag remote tails (exchangeVertexReqs, dspl.hpp:1140-1164) ----
__global__ void k_select_remote(i64 lne, const i64 *__restrict__ tails,
                                i64 base, i64 bound, i64 *__restrict__ out,
                                unsigned long long *__restrict__ count) {
    for (i64 e = blockIdx.x * (i64)blockDim.x + threadIdx.x; e < lne;
         e += (i64)gridDim.x * blockDim.x) {
        const i64 t = tails[e];
        if (t < base || t >= bound) out[atomicAdd(count, 1ull)] = t;
    }
}

// ---- K10b: translate tails into DevEdge (product layout; removes the
// reference's per-edge unordered_map lookup, dspl.hpp:253-260) ----
__global__ void k_build_edges(i64 lne, const i64 *__restrict__ tails,
                              const double *__restrict__ w, i64 base,
                              i64 bound, i64 lnv,
                              const i64 *__restrict__ ghosts, i64 nghost,
                              DevEdge *__restrict__ edges) {
    for (i64 e = blockIdx.x * (i64)blockDim.x + threadIdx.x; e < lne;
         e += (i64)gridDim.x * blockDim.x) {
        const i64 t = tails[e];
        i64 ti = (t >= base && t < bound)
                     ? t - base
                     : lnv + dev_lower_bound(ghosts, nghost, t);
        edges[e] = {ti, w[e]};
    }
}

// ---- K8: scdata gather (dspl.hpp:559-571): comm of each exported vertex ----
__global__ void k8_gather_comms(i64 n, const i64 *__restrict__ svdata,
                                i64 base, const i64 *__restrict__ currComm,
                                i64 *__restrict__ out) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x)
        out[k] = currComm[svdata[k] - base];
}

// ---- candidate remote communities (dspl.hpp:670-700): ghost comms +
// own currComm, filtered to remote owners ----
__global__ void k_filter_remote(i64 n, const i64 *__restrict__ vals, i64 base,
                                i64 bound, i64 *__restrict__ out,
                                unsigned long long *__restrict__ count) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 c = vals[k];
        if (c < base || c >= bound) out[atomicAdd(count, 1ull)] = c;
    }
}

// per-owner boundaries of a sorted id array (one thread per rank)
__global__ void k_owner_bounds(const i64 *__restrict__ sorted, i64 n,
                               const i64 *__restrict__ parts, int nranks,
                               i64 *__restrict__ bounds) {
    int r = blockIdx.x * blockDim.x + threadIdx.x;
    if (r <= nranks) bounds[r] = dev_lower_bound(sorted, n, parts[r]);
}

// ---- K9: cinfo reply gather (dspl.hpp:776-929) ----
__global__ void k9_reply_info(i64 n, const i64 *__restrict__ req_ids, i64 base,
                              const i64 *__restrict__ cinfo_size,
                              const double *__restrict__ cinfo_degree,
                              Info16 *__restrict__ out) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 c = req_ids[k] - base;
        out[k] = {cinfo_size[c], cinfo_degree[c]};
    }
}

// unpack received rc_info into SoA
__global__ void k_unpack_info(i64 n, const Info16 *__restrict__ in,
                              i64 *__restrict__ size_out,
                              double *__restrict__ deg_out) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        size_out[k] = in[k].size;
        deg_out[k] = in[k].degree;
    }
}

// pack remote-community deltas for halo #2 (updateRemoteCommunities,
// dspl.hpp:988-1004: every remoteCinfo key is sent, zeros included)
__global__ void k_pack_deltas(i64 n, const i64 *__restrict__ rcu_size,
                              const double *__restrict__ rcu_degree,
                              Info16 *__restrict__ out) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x)
        out[k] = {rcu_size[k], rcu_degree[k]};
}

// apply received deltas to owned cinfo (dspl.hpp:1089-1102; atomics because
// several senders may address one community)
__global__ void k_apply_deltas(i64 n, const i64 *__restrict__ ids, i64 base,
                               const Info16 *__restrict__ deltas,
                               i64 *__restrict__ cinfo_size,
                               double *__restrict__ cinfo_degree) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 c = ids[k] - base;
        atomic_add_i64(&cinfo_size[c], deltas[k].size);
        atomicAdd(&cinfo_degree[c], deltas[k].degree);
    }
}

// ---- K6: apply localCupdate (dspl.hpp:458-471) ----
__global__ void k6_apply_local(i64 lnv, const i64 *__restrict__ cupd_size,
                               const double *__restrict__ cupd_degree,
                               i64 *__restrict__ cinfo_size,
                               double *__restrict__ cinfo_degree) {
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < lnv;
         i += (i64)gridDim.x * blockDim.x) {
        cinfo_size[i] += cupd_size[i];
        cinfo_degree[i] += cupd_degree[i];
    }
}

// ---- K4: THE sweep (distExecuteLouvainIteration, dspl.hpp:276-405) ----
// One lane per vertex; the clmap/counter hash (dspl.hpp:230-274) lives as
// per-lane slot arrays: the first SLOTS distinct neighbor communities in
// LDS (lane-strided, conflict-free for ds b64), the tail in a per-thread
// global spill region (rare after iteration 1; L2-resident). The current
// community's accumulator is held in a register (the reference's
// counter[0], dspl.hpp:312-318). Per-lane sequential edge walk keeps -w
// accumulation in edge order.
template <int SLOTS>
__global__ __launch_bounds__(256) void k4_sweep(
    i64 lnv, i64 base, i64 bound, const i64 *__restrict__ xadj,
    const DevEdge *__restrict__ edges, const i64 *__restrict__ currComm,
    const i64 *__restrict__ ghost_comm, const double *__restrict__ vDegree,
    const i64 *__restrict__ cinfo_size, const double *__restrict__ cinfo_degree,
    i64 *__restrict__ cupd_size, double *__restrict__ cupd_degree,
    const i64 *__restrict__ rc_ids, i64 nrc, const i64 *__restrict__ rc_size,
    const double *__restrict__ rc_degree, i64 *__restrict__ rcu_size,
    double *__restrict__ rcu_degree, double constant,
    i64 *__restrict__ targetComm, double *__restrict__ clusterWeight,
    i64 *__restrict__ spill_keys, double *__restrict__ spill_acc,
    int spill_max) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    i64 *skey = reinterpret_cast<i64 *>(smem);
    double *sacc = reinterpret_cast<double *>(smem + sizeof(i64) * SLOTS * blockDim.x);
    const int tid = threadIdx.x;
    const i64 gthread = blockIdx.x * (i64)blockDim.x + threadIdx.x;
    const i64 stride = (i64)gridDim.x * blockDim.x;
    i64 *myspill_k = spill_keys + gthread * spill_max;
    double *myspill_a = spill_acc + gthread * spill_max;

    for (i64 i = gthread; i < lnv; i += stride) {
        const i64 cc = currComm[i];
        double ccDeg;
        i64 ccSize;
        if (cc >= base && cc < bound) { // dspl.hpp:296-307
            ccDeg = cinfo_degree[cc - base];
            ccSize = cinfo_size[cc - base];
        } else {
            const i64 s = dev_bsearch(rc_ids, nrc, cc);
            ccDeg = rc_degree[s];
            ccSize = rc_size[s];
        }
        const i64 e0 = xadj[i], e1 = xadj[i + 1];
        i64 target;
        if (e0 == e1) {
            target = cc; // dspl.hpp:323-324
        } else {
            double c0 = 0.0, selfLoop = 0.0;
            int ns = 0, nspill = 0;
            for (i64 e = e0; e < e1; e++) {
                const DevEdge ed = edges[e];
                if (ed.tidx == i) selfLoop += ed.w; // dspl.hpp:247-248
                const i64 tcomm = (ed.tidx < lnv) ? currComm[ed.tidx]
                                                  : ghost_comm[ed.tidx - lnv];
                if (tcomm == cc) { c0 += ed.w; continue; }
                bool found = false;
                for (int s = 0; s < ns; s++) {
                    if (skey[s * blockDim.x + tid] == tcomm) {
                        sacc[s * blockDim.x + tid] += ed.w;
                        found = true;
                        break;
                    }
                }
                if (found) continue;
                if (ns < SLOTS) {
                    skey[ns * blockDim.x + tid] = tcomm;
                    sacc[ns * blockDim.x + tid] = ed.w;
                    ns++;
                    continue;
                }
                for (int s = 0; s < nspill; s++) {
                    if (myspill_k[s] == tcomm) {
                        myspill_a[s] += ed.w;
                        found = true;
                        break;
                    }
                }
                if (!found) { // spill_max sized to the max degree: cannot overflow
                    myspill_k[nspill] = tcomm;
                    myspill_a[nspill] = ed.w;
                    nspill++;
                }
            }
            clusterWeight[i] += c0; // dspl.hpp:318

            // distGetMaxIndex (dspl.hpp:174-228)
            const double vdeg = vDegree[i];
            const double eix = c0 - selfLoop;
            const double ax = ccDeg - vdeg;
            double maxGain = 0.0;
            i64 maxIndex = cc, maxSize = ccSize;
            for (int s = 0; s < ns + nspill; s++) {
                const i64 y = (s < ns) ? skey[s * blockDim.x + tid]
                                       : myspill_k[s - ns];
                const double eiy = (s < ns) ? sacc[s * blockDim.x + tid]
                                            : myspill_a[s - ns];
                double ay;
                i64 ysz;
                if (y >= base && y < bound) {
                    ay = cinfo_degree[y - base];
                    ysz = cinfo_size[y - base];
                } else {
                    const i64 q = dev_bsearch(rc_ids, nrc, y);
                    ay = rc_degree[q];
                    ysz = rc_size[q];
                }
                const double curGain =
                    2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant; // :212
                if (curGain > maxGain ||
                    (curGain == maxGain && curGain != 0.0 && y < maxIndex)) {
                    maxGain = curGain;
                    maxIndex = y;
                    maxSize = ysz;
                }
            }
            if (maxSize == 1 && ccSize == 1 && maxIndex > cc) // :224-225
                maxIndex = cc;
            target = maxIndex;
        }

        if (target != cc) { // 4-case updates (dspl.hpp:331-399)
            const double vdeg = vDegree[i];
            if (cc >= base && cc < bound) {
                atomicAdd(&cupd_degree[cc - base], -vdeg);
                atomic_add_i64(&cupd_size[cc - base], -1);
            } else {
                const i64 s = dev_bsearch(rc_ids, nrc, cc);
                atomicAdd(&rcu_degree[s], -vdeg);
                atomic_add_i64(&rcu_size[s], -1);
            }
            if (target >= base && target < bound) {
                atomicAdd(&cupd_degree[target - base], vdeg);
                atomic_add_i64(&cupd_size[target - base], 1);
            } else {
                const i64 s = dev_bsearch(rc_ids, nrc, target);
                atomicAdd(&rcu_degree[s], vdeg);
                atomic_add_i64(&rcu_size[s], 1);
            }
        }
        targetComm[i] = target; // dspl.hpp:404
    }
}

int grid_for(i64 n, int block = 256, int cap = 2048) {
    i64 g = (n + block - 1) / block;
    return (int)std::min<i64>(std::max<i64>(g, 1), cap);
}

} // namespace

// ------------------------------------------------------------------------
struct mv_engine {
    int device = 0, rank = 0, nranks = 1;
    ncclComm_t comm = nullptr;
    hipStream_t stream = nullptr;

    // graph (device)
    i64 nv = 0, lnv = 0, lne = 0, base = 0, bound = 0;
    std::vector<i64> parts_h;
    i64 *d_parts = nullptr;
    i64 *d_xadj = nullptr;
    i64 *d_tails = nullptr;   // raw global tails (kept for setup)
    double *d_ew = nullptr;   // edge weights
    DevEdge *d_edges = nullptr;
    int unit_weights = 1;
    i64 max_degree = 0;

    // per-run state (device)
    i64 *d_curr = nullptr, *d_past = nullptr, *d_target = nullptr;
    double *d_vdeg = nullptr, *d_cw = nullptr;
    i64 *d_cinfo_size = nullptr, *d_cupd_size = nullptr;
    double *d_cinfo_deg = nullptr, *d_cupd_deg = nullptr;
    double *d_partials = nullptr; // 2 * nblocks
    double *d_red = nullptr;      // 2 doubles for allreduce

    // ghosts / halo
    i64 *d_ghosts = nullptr;     // sorted unique remote tails
    i64 nghost = 0;
    i64 *d_ghost_comm = nullptr; // per-iteration communities of ghosts
    i64 *d_svdata = nullptr;     // vertices peers want from me (global ids)
    i64 ssz = 0;
    std::vector<i64> send_off, recv_off; // per-peer offsets into svdata / ghosts
    i64 *d_scdata = nullptr;             // packed comms to export

    // remote community info (per iteration)
    i64 rc_cap = 0;
    i64 *d_cand = nullptr, *d_cand_sorted = nullptr;
    i64 *d_rc_ids = nullptr, *d_rc_size = nullptr;
    double *d_rc_degree = nullptr;
    i64 *d_rcu_size = nullptr;
    double *d_rcu_degree = nullptr;
    Info16 *d_rc_info = nullptr;
    i64 req_cap = 0;
    i64 *d_req_ids = nullptr;   // ids other ranks requested from me
    Info16 *d_req_info = nullptr;
    i64 *d_bounds = nullptr;    // nranks+1
    unsigned long long *d_count = nullptr;
    void *d_cub_tmp = nullptr;
    size_t cub_tmp_bytes = 0;

    // spill for K4
    i64 *d_spill_k = nullptr;
    double *d_spill_a = nullptr;
    int spill_max = 0;
    int sweep_grid = 0;

    // trace
    i64 *trace_target = nullptr;
    double *trace_mod = nullptr;
    int trace_cap = 0;

    mv_stats stats{};
    std::vector<hipEvent_t> ev_pool;
    int ev_used = 0;

    hipEvent_t ev_pair() {
        if (ev_used >= (int)ev_pool.size()) {
            hipEvent_t e;
            HIP_CHECK(hipEventCreate(&e));
            ev_pool.push_back(e);
        }
        return ev_pool[ev_used++];
    }
};

extern "C" {

int mv_comm_id(void *id_bytes) {
    static_assert(sizeof(ncclUniqueId) <= MV_COMM_ID_BYTES, "id size");
    ncclUniqueId id;
    if (ncclGetUniqueId(&id) != ncclSuccess) return -1;
    std::memcpy(id_bytes, &id, sizeof(id));
    return 0;
}

mv_engine *mv_engine_create(int device, int rank, int nranks,
                            const void *comm_id) {
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev <= device) {
        std::fprintf(stderr,
                     "mv_engine_create: no HIP device %d (found %d). The "
                     "MI355X engine has no CPU fallback.\n",
                     device, ndev);
        return nullptr;
    }
    auto *e = new mv_engine;
    e->device = device;
    e->rank = rank;
    e->nranks = nranks;
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipStreamCreate(&e->stream));
    if (nranks > 1) {
        if (!comm_id) {
            std::fprintf(stderr, "mv_engine_create: nranks>1 needs comm id\n");
            delete e;
            return nullptr;
        }
        ncclUniqueId id;
        std::memcpy(&id, comm_id, sizeof(id));
        NCCL_CHECK(ncclCommInitRank(&e->comm, nranks, id, rank));
    }
    return e;
}

void mv_engine_destroy(mv_engine *e) {
    if (!e) return;
    hipSetDevice(e->device);
    for (auto ev : e->ev_pool) hipEventDestroy(ev);
    if (e->comm) ncclCommDestroy(e->comm);
    // device allocations are freed with the primary context teardown at
    // process exit; free the big ones explicitly
    for (void *p : {(void *)e->d_xadj, (void *)e->d_tails, (void *)e->d_ew,
                    (void *)e->d_edges, (void *)e->d_curr, (void *)e->d_past,
                    (void *)e->d_target, (void *)e->d_vdeg, (void *)e->d_cw,
                    (void *)e->d_cinfo_size, (void *)e->d_cinfo_deg,
                    (void *)e->d_cupd_size, (void *)e->d_cupd_deg,
                    (void *)e->d_spill_k, (void *)e->d_spill_a})
        if (p) hipFree(p);
    if (e->stream) hipStreamDestroy(e->stream);
    delete e;
}

int mv_engine_load_graph(mv_engine *e, const mv_graph *g) {
    HIP_CHECK(hipSetDevice(e->device));
    e->nv = mv_graph_nv(g);
    e->lnv = mv_graph_lnv(g);
    e->lne = mv_graph_lne(g);
    e->parts_h.assign(mv_graph_parts(g), mv_graph_parts(g) + e->nranks + 1);
    e->base = e->parts_h[e->rank];
    e->bound = e->parts_h[e->rank + 1];
    const i64 lnv = e->lnv, lne = e->lne;

    const i64 *xadj = mv_graph_xadj(g);
    e->max_degree = 0;
    for (i64 i = 0; i < lnv; i++)
        e->max_degree = std::max(e->max_degree, xadj[i + 1] - xadj[i]);
    const double *w = mv_graph_weights(g);
    e->unit_weights = 1;
    for (i64 k = 0; k < lne; k++)
        if (w[k] != 1.0) { e->unit_weights = 0; break; }

    HIP_CHECK(hipMalloc(&e->d_parts, 8 * (e->nranks + 1)));
    HIP_CHECK(hipMemcpy(e->d_parts, e->parts_h.data(), 8 * (e->nranks + 1),
                        hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&e->d_xadj, 8 * (lnv + 1)));
    HIP_CHECK(hipMemcpy(e->d_xadj, xadj, 8 * (lnv + 1), hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&e->d_tails, 8 * std::max<i64>(lne, 1)));
    HIP_CHECK(hipMemcpy(e->d_tails, mv_graph_tails(g), 8 * lne,
                        hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&e->d_ew, 8 * std::max<i64>(lne, 1)));
    HIP_CHECK(hipMemcpy(e->d_ew, w, 8 * lne, hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&e->d_edges, sizeof(DevEdge) * std::max<i64>(lne, 1)));

    HIP_CHECK(hipMalloc(&e->d_curr, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_past, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_target, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_vdeg, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_cw, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_cinfo_size, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_cinfo_deg, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_cupd_size, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_cupd_deg, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_count, 8));
    HIP_CHECK(hipMalloc(&e->d_bounds, 8 * (e->nranks + 1)));
    HIP_CHECK(hipMalloc(&e->d_red, 16));
    const int nblocks = grid_for(lnv);
    HIP_CHECK(hipMalloc(&e->d_partials, 16 * nblocks));

    // K4 geometry: 256-thread blocks, 16 LDS slots/lane (64 KiB/block ->
    // 2 blocks/CU), spill region covers the max degree
    e->sweep_grid = grid_for(lnv, 256, 2048);
    e->spill_max = (int)std::max<i64>(e->max_degree - 16 + 1, 1);
    const i64 nthreads = (i64)e->sweep_grid * 256;
    HIP_CHECK(hipMalloc(&e->d_spill_k, 8 * nthreads * e->spill_max));
    HIP_CHECK(hipMalloc(&e->d_spill_a, 8 * nthreads * e->spill_max));

    e->stats = mv_stats{};
    e->stats.edges_local = lne;
    return 0;
}

void mv_engine_set_trace(mv_engine *e, int64_t *target_trace, double *mod_trace,
                         int cap) {
    e->trace_target = target_trace;
    e->trace_mod = mod_trace;
    e->trace_cap = cap;
}

void mv_engine_get_stats(const mv_engine *e, mv_stats *out) { *out = e->stats; }

// helper: alltoallv over RCCL grouped send/recv; counts/displs in elements
static void rccl_alltoallv(mv_engine *e, const void *send, const i64 *soff,
                           void *recv, const i64 *roff, size_t elem_bytes,
                           ncclDataType_t ty, size_t ty_bytes) {
    NCCL_CHECK(ncclGroupStart());
    for (int r = 0; r < e->nranks; r++) {
        if (r == e->rank) continue;
        const i64 scnt = soff[r + 1] - soff[r];
        const i64 rcnt = roff[r + 1] - roff[r];
        if (scnt > 0)
            NCCL_CHECK(ncclSend((const char *)send + soff[r] * elem_bytes,
                                scnt * elem_bytes / ty_bytes, ty, r, e->comm,
                                e->stream));
        if (rcnt > 0)
            NCCL_CHECK(ncclRecv((char *)recv + roff[r] * elem_bytes,
                                rcnt * elem_bytes / ty_bytes, ty, r, e->comm,
                                e->stream));
    }
    NCCL_CHECK(ncclGroupEnd());
}

// exchange per-peer counts: allgather of my nranks counts
static void exchange_counts(mv_engine *e, const std::vector<i64> &mine,
                            std::vector<i64> &matrix /*nranks*nranks*/) {
    i64 *d_all = nullptr;
    HIP_CHECK(hipMalloc(&d_all, 8 * e->nranks * e->nranks));
    HIP_CHECK(hipMemcpyAsync(d_all + (i64)e->rank * e->nranks, mine.data(),
                             8 * e->nranks, hipMemcpyHostToDevice, e->stream));
    NCCL_CHECK(ncclAllGather(d_all + (i64)e->rank * e->nranks, d_all,
                             e->nranks, ncclInt64, e->comm, e->stream));
    matrix.resize((size_t)e->nranks * e->nranks);
    HIP_CHECK(hipMemcpyAsync(matrix.data(), d_all, 8 * e->nranks * e->nranks,
                             hipMemcpyDeviceToHost, e->stream));
    HIP_CHECK(hipStreamSynchronize(e->stream));
    HIP_CHECK(hipFree(d_all));
}

double mv_engine_run(mv_engine *e, double lower, double thresh,
                     int *iters_out) {
    HIP_CHECK(hipSetDevice(e->device));
    const auto t_start = std::chrono::steady_clock::now();
    hipStream_t st = e->stream;
    const i64 lnv = e->lnv, lne = e->lne;
    const int p = e->nranks, me = e->rank;
    e->ev_used = 0;
    e->stats = mv_stats{};
    e->stats.edges_local = lne;

    // ---- exchangeVertexReqs equivalent (dspl.hpp:1112-1272) ----
    const auto t_setup0 = std::chrono::steady_clock::now();
    {
        HIP_CHECK(hipMemsetAsync(e->d_count, 0, 8, st));
        if (p > 1) {
            i64 *d_rem = nullptr;
            HIP_CHECK(hipMalloc(&d_rem, 8 * std::max<i64>(lne, 1)));
            k_select_remote<<<grid_for(lne), 256, 0, st>>>(
                lne, e->d_tails, e->base, e->bound, d_rem, e->d_count);
            unsigned long long nrem = 0;
            HIP_CHECK(hipMemcpyAsync(&nrem, e->d_count, 8,
                                     hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));
            // sort + unique -> ghosts
            i64 *d_sorted = nullptr;
            HIP_CHECK(hipMalloc(&d_sorted, 8 * std::max<i64>((i64)nrem, 1)));
            if (!e->d_ghosts)
                HIP_CHECK(hipMalloc(&e->d_ghosts,
                                    8 * std::max<i64>((i64)nrem, 1)));
            size_t tmp1 = 0, tmp2 = 0;
            hipcub::DeviceRadixSort::SortKeys(nullptr, tmp1, d_rem, d_sorted,
                                              (int64_t)nrem, 0, 64, st);
            i64 *d_ng = nullptr;
            HIP_CHECK(hipMalloc(&d_ng, 8));
            hipcub::DeviceSelect::Unique(nullptr, tmp2, d_sorted, e->d_ghosts,
                                         d_ng, (int64_t)nrem, st);
            size_t tmpb = std::max(tmp1, tmp2);
            void *d_tmp = nullptr;
            HIP_CHECK(hipMalloc(&d_tmp, std::max<size_t>(tmpb, 1)));
            hipcub::DeviceRadixSort::SortKeys(d_tmp, tmp1, d_rem, d_sorted,
                                              (int64_t)nrem, 0, 64, st);
            hipcub::DeviceSelect::Unique(d_tmp, tmp2, d_sorted, e->d_ghosts,
                                         d_ng, (int64_t)nrem, st);
            HIP_CHECK(hipMemcpyAsync(&e->nghost, d_ng, 8,
                                     hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));
            HIP_CHECK(hipFree(d_rem));
            HIP_CHECK(hipFree(d_sorted));
            HIP_CHECK(hipFree(d_ng));
            HIP_CHECK(hipFree(d_tmp));

            // per-owner segments of my (sorted) want list
            k_owner_bounds<<<1, p + 1, 0, st>>>(e->d_ghosts, e->nghost,
                                                e->d_parts, p, e->d_bounds);
            e->recv_off.resize(p + 1);
            HIP_CHECK(hipMemcpyAsync(e->recv_off.data(), e->d_bounds,
                                     8 * (p + 1), hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));

            // exchange want-list sizes, then the lists (dspl.hpp:1184-1252)
            std::vector<i64> want(p), matrix;
            for (int r = 0; r < p; r++)
                want[r] = e->recv_off[r + 1] - e->recv_off[r];
            exchange_counts(e, want, matrix);
            e->send_off.assign(p + 1, 0);
            for (int r = 0; r < p; r++)
                e->send_off[r + 1] =
                    e->send_off[r] + ((r == me) ? 0 : matrix[(size_t)r * p + me]);
            e->ssz = e->send_off[p];
            if (e->d_svdata) HIP_CHECK(hipFree(e->d_svdata));
            HIP_CHECK(hipMalloc(&e->d_svdata, 8 * std::max<i64>(e->ssz, 1)));
            // role swap (dspl.hpp:1255-1257): my ghost list goes OUT, the
            // peers' lists land in svdata
            rccl_alltoallv(e, e->d_ghosts, e->recv_off.data(), e->d_svdata,
                           e->send_off.data(), 8, ncclInt64, 8);
            HIP_CHECK(hipStreamSynchronize(st));

            if (e->d_ghost_comm) HIP_CHECK(hipFree(e->d_ghost_comm));
            HIP_CHECK(hipMalloc(&e->d_ghost_comm,
                                8 * std::max<i64>(e->nghost, 1)));
            if (e->d_scdata) HIP_CHECK(hipFree(e->d_scdata));
            HIP_CHECK(hipMalloc(&e->d_scdata, 8 * std::max<i64>(e->ssz, 1)));
        } else {
            e->nghost = 0;
            e->ssz = 0;
        }
        // translate tails (K10b)
        k_build_edges<<<grid_for(lne), 256, 0, st>>>(
            lne, e->d_tails, e->d_ew, e->base, e->bound, lnv, e->d_ghosts,
            e->nghost, e->d_edges);
    }

    // ---- distInitLouvain (dspl.hpp:151-172) ----
    k1_vertex_degree<<<grid_for(lnv), 256, 0, st>>>(
        lnv, e->d_xadj, e->d_ew, e->d_vdeg, e->d_cinfo_size, e->d_cinfo_deg);
    const int nblocks = grid_for(lnv);
    {
        auto f = [vd = e->d_vdeg] __device__(i64 i, double &a, double &b) {
            a = vd[i];
            b = 0.0;
        };
        k_partial_sum2<<<nblocks, 256, 0, st>>>(lnv, f, e->d_partials);
    }
    std::vector<double> partials(2 * nblocks);
    HIP_CHECK(hipMemcpyAsync(partials.data(), e->d_partials, 16 * nblocks,
                             hipMemcpyDeviceToHost, st));
    HIP_CHECK(hipStreamSynchronize(st));
    double localW = 0.0;
    for (int b = 0; b < nblocks; b++) localW += partials[2 * b];
    double totalW = localW;
    if (p > 1) { // allreduce (dspl.hpp:126)
        HIP_CHECK(hipMemcpyAsync(e->d_red, &localW, 8, hipMemcpyHostToDevice, st));
        NCCL_CHECK(ncclAllReduce(e->d_red, e->d_red, 1, ncclDouble, ncclSum,
                                 e->comm, st));
        HIP_CHECK(hipMemcpyAsync(&totalW, e->d_red, 8, hipMemcpyDeviceToHost, st));
        HIP_CHECK(hipStreamSynchronize(st));
    }
    const double constant = 1.0 / totalW; // dspl.hpp:129
    k3_init_comm<<<grid_for(lnv), 256, 0, st>>>(lnv, e->base, e->d_curr,
                                                e->d_past);
    HIP_CHECK(hipStreamSynchronize(st));
    e->stats.setup_ms =
        std::chrono::duration<double, std::milli>(
            std::chrono::steady_clock::now() - t_setup0)
            .count();

    i64 *d_curr = e->d_curr, *d_past = e->d_past, *d_target = e->d_target;
    double prevMod = lower, currMod = -1.0;
    int numIters = 0;
    std::vector<i64> rc_bounds(p + 1, 0), req_off(p + 1, 0);

    std::vector<hipEvent_t> sweep_ev;
    for (;;) {
        numIters++;

        i64 nrc = 0;
        if (p > 1) {
            const auto t_h0 = std::chrono::steady_clock::now();
            // ---- halo #1a: ghost communities (dspl.hpp:583-647) ----
            k8_gather_comms<<<grid_for(std::max<i64>(e->ssz, 1)), 256, 0, st>>>(
                e->ssz, e->d_svdata, e->base, d_curr, e->d_scdata);
            rccl_alltoallv(e, e->d_scdata, e->send_off.data(), e->d_ghost_comm,
                           e->recv_off.data(), 8, ncclInt64, 8);

            // ---- needed remote communities (dspl.hpp:670-700) ----
            const i64 cand_max = e->nghost + lnv;
            if (cand_max > e->rc_cap) {
                for (void *q : {(void *)e->d_cand, (void *)e->d_cand_sorted,
                                (void *)e->d_rc_ids, (void *)e->d_rc_size,
                                (void *)e->d_rc_degree, (void *)e->d_rcu_size,
                                (void *)e->d_rcu_degree, (void *)e->d_rc_info})
                    if (q) HIP_CHECK(hipFree(q));
                e->rc_cap = cand_max;
                HIP_CHECK(hipMalloc(&e->d_cand, 8 * cand_max));
                HIP_CHECK(hipMalloc(&e->d_cand_sorted, 8 * cand_max));
                HIP_CHECK(hipMalloc(&e->d_rc_ids, 8 * cand_max));
                HIP_CHECK(hipMalloc(&e->d_rc_size, 8 * cand_max));
                HIP_CHECK(hipMalloc(&e->d_rc_degree, 8 * cand_max));
                HIP_CHECK(hipMalloc(&e->d_rcu_size, 8 * cand_max));
                HIP_CHECK(hipMalloc(&e->d_rcu_degree, 8 * cand_max));
                HIP_CHECK(hipMalloc(&e->d_rc_info, sizeof(Info16) * cand_max));
                size_t t1 = 0, t2 = 0;
                hipcub::DeviceRadixSort::SortKeys(nullptr, t1, e->d_cand,
                                                  e->d_cand_sorted, cand_max, 0,
                                                  64, st);
                i64 *dummy = nullptr;
                hipcub::DeviceSelect::Unique(nullptr, t2, e->d_cand_sorted,
                                             e->d_rc_ids, dummy, cand_max, st);
                size_t need = std::max(t1, t2);
                if (need > e->cub_tmp_bytes) {
                    if (e->d_cub_tmp) HIP_CHECK(hipFree(e->d_cub_tmp));
                    HIP_CHECK(hipMalloc(&e->d_cub_tmp, need));
                    e->cub_tmp_bytes = need;
                }
            }
            HIP_CHECK(hipMemsetAsync(e->d_count, 0, 8, st));
            k_filter_remote<<<grid_for(std::max<i64>(e->nghost, 1)), 256, 0,
                              st>>>(e->nghost, e->d_ghost_comm, e->base,
                                    e->bound, e->d_cand, e->d_count);
            k_filter_remote<<<grid_for(lnv), 256, 0, st>>>(
                lnv, d_curr, e->base, e->bound, e->d_cand, e->d_count);
            unsigned long long ncand = 0;
            HIP_CHECK(hipMemcpyAsync(&ncand, e->d_count, 8,
                                     hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));
            size_t tb = e->cub_tmp_bytes;
            hipcub::DeviceRadixSort::SortKeys(e->d_cub_tmp, tb, e->d_cand,
                                              e->d_cand_sorted, (int64_t)ncand,
                                              0, 64, st);
            i64 *d_nrc = (i64 *)e->d_count; // reuse as output slot
            tb = e->cub_tmp_bytes;
            hipcub::DeviceSelect::Unique(e->d_cub_tmp, tb, e->d_cand_sorted,
                                         e->d_rc_ids, d_nrc, (int64_t)ncand, st);
            HIP_CHECK(hipMemcpyAsync(&nrc, d_nrc, 8, hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));

            // ---- halo #1b/#1c/#1d: request (size,degree) of those
            // communities from their owners (dspl.hpp:719-929) ----
            k_owner_bounds<<<1, p + 1, 0, st>>>(e->d_rc_ids, nrc, e->d_parts,
                                                p, e->d_bounds);
            HIP_CHECK(hipMemcpyAsync(rc_bounds.data(), e->d_bounds, 8 * (p + 1),
                                     hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));
            std::vector<i64> reqs(p), matrix;
            for (int r = 0; r < p; r++)
                reqs[r] = rc_bounds[r + 1] - rc_bounds[r];
            exchange_counts(e, reqs, matrix);
            req_off[0] = 0;
            for (int r = 0; r < p; r++)
                req_off[r + 1] =
                    req_off[r] + ((r == me) ? 0 : matrix[(size_t)r * p + me]);
            const i64 nreq = req_off[p];
            if (nreq > e->req_cap) {
                if (e->d_req_ids) HIP_CHECK(hipFree(e->d_req_ids));
                if (e->d_req_info) HIP_CHECK(hipFree(e->d_req_info));
                e->req_cap = std::max<i64>(nreq, 64);
                HIP_CHECK(hipMalloc(&e->d_req_ids, 8 * e->req_cap));
                HIP_CHECK(
                    hipMalloc(&e->d_req_info, sizeof(Info16) * e->req_cap));
            }
            rccl_alltoallv(e, e->d_rc_ids, rc_bounds.data(), e->d_req_ids,
                           req_off.data(), 8, ncclInt64, 8);
            k9_reply_info<<<grid_for(std::max<i64>(nreq, 1)), 256, 0, st>>>(
                nreq, e->d_req_ids, e->base, e->d_cinfo_size, e->d_cinfo_deg,
                e->d_req_info);
            rccl_alltoallv(e, e->d_req_info, req_off.data(), e->d_rc_info,
                           rc_bounds.data(), sizeof(Info16), ncclChar,
                           1);
            k_unpack_info<<<grid_for(std::max<i64>(nrc, 1)), 256, 0, st>>>(
                nrc, e->d_rc_info, e->d_rc_size, e->d_rc_degree);
            HIP_CHECK(hipMemsetAsync(e->d_rcu_size, 0, 8 * std::max<i64>(nrc, 1),
                                     st));
            HIP_CHECK(hipMemsetAsync(e->d_rcu_degree, 0,
                                     8 * std::max<i64>(nrc, 1), st));
            HIP_CHECK(hipStreamSynchronize(st));
            e->stats.halo_ms +=
                std::chrono::duration<double, std::milli>(
                    std::chrono::steady_clock::now() - t_h0)
                    .count();
        }

        // ---- K5 zero + K4 sweep (dspl.hpp:1371-1387) ----
        HIP_CHECK(hipMemsetAsync(e->d_cw, 0, 8 * lnv, st));
        HIP_CHECK(hipMemsetAsync(e->d_cupd_size, 0, 8 * lnv, st));
        HIP_CHECK(hipMemsetAsync(e->d_cupd_deg, 0, 8 * lnv, st));
        hipEvent_t ev0 = e->ev_pair(), ev1 = e->ev_pair();
        HIP_CHECK(hipEventRecord(ev0, st));
        constexpr int SLOTS = 16;
        k4_sweep<SLOTS><<<e->sweep_grid, 256, SLOTS * 256 * 16, st>>>(
            lnv, e->base, e->bound, e->d_xadj, e->d_edges, d_curr,
            e->d_ghost_comm, e->d_vdeg, e->d_cinfo_size, e->d_cinfo_deg,
            e->d_cupd_size, e->d_cupd_deg, e->d_rc_ids, nrc, e->d_rc_size,
            e->d_rc_degree, e->d_rcu_size, e->d_rcu_degree, constant, d_target,
            e->d_cw, e->d_spill_k, e->d_spill_a, e->spill_max);
        HIP_CHECK(hipEventRecord(ev1, st));
        sweep_ev.push_back(ev0);
        sweep_ev.push_back(ev1);
        e->stats.sweep_launches++;

        // ---- K6 (dspl.hpp:458-471) ----
        k6_apply_local<<<grid_for(lnv), 256, 0, st>>>(
            lnv, e->d_cupd_size, e->d_cupd_deg, e->d_cinfo_size,
            e->d_cinfo_deg);

        // ---- halo #2: route deltas to owners (dspl.hpp:978-1103) ----
        if (p > 1) {
            const auto t_h0 = std::chrono::steady_clock::now();
            k_pack_deltas<<<grid_for(std::max<i64>(nrc, 1)), 256, 0, st>>>(
                nrc, e->d_rcu_size, e->d_rcu_degree, e->d_rc_info);
            rccl_alltoallv(e, e->d_rc_info, rc_bounds.data(), e->d_req_info,
                           req_off.data(), sizeof(Info16), ncclChar, 1);
            k_apply_deltas<<<grid_for(std::max<i64>(req_off[p], 1)), 256, 0,
                             st>>>(req_off[p], e->d_req_ids, e->base,
                                   e->d_req_info, e->d_cinfo_size,
                                   e->d_cinfo_deg);
            HIP_CHECK(hipStreamSynchronize(st));
            e->stats.halo_ms +=
                std::chrono::duration<double, std::milli>(
                    std::chrono::steady_clock::now() - t_h0)
                    .count();
        }

        // ---- K7: modularity (dspl.hpp:407-456) ----
        {
            auto f = [cw = e->d_cw, cd = e->d_cinfo_deg] __device__(
                         i64 i, double &a, double &b) {
                a = cw[i];
                b = cd[i] * cd[i];
            };
            k_partial_sum2<<<nblocks, 256, 0, st>>>(lnv, f, e->d_partials);
        }
        HIP_CHECK(hipMemcpyAsync(partials.data(), e->d_partials, 16 * nblocks,
                                 hipMemcpyDeviceToHost, st));
        HIP_CHECK(hipStreamSynchronize(st));
        double le = 0.0, la = 0.0;
        for (int b = 0; b < nblocks; b++) {
            le += partials[2 * b];
            la += partials[2 * b + 1];
        }
        double red[2] = {le, la};
        if (p > 1) {
            HIP_CHECK(hipMemcpyAsync(e->d_red, red, 16, hipMemcpyHostToDevice,
                                     st));
            NCCL_CHECK(ncclAllReduce(e->d_red, e->d_red, 2, ncclDouble,
                                     ncclSum, e->comm, st));
            HIP_CHECK(hipMemcpyAsync(red, e->d_red, 16, hipMemcpyDeviceToHost,
                                     st));
            HIP_CHECK(hipStreamSynchronize(st));
        }
        currMod = std::fabs(red[0] * constant - red[1] * constant * constant);

        // ---- trace ----
        if (e->trace_mod && numIters <= e->trace_cap)
            e->trace_mod[numIters - 1] = currMod;
        if (e->trace_target && numIters <= e->trace_cap)
            HIP_CHECK(hipMemcpy(e->trace_target + (i64)(numIters - 1) * lnv,
                                d_target, 8 * lnv, hipMemcpyDeviceToHost));

        if (currMod - prevMod < thresh) break; // dspl.hpp:1401
        prevMod = currMod;
        if (prevMod < lower) prevMod = lower; // dspl.hpp:1404-1406
        // rotate (dspl.hpp:1417-1422): every array is fully rewritten or
        // never read, so pointer rotation is equivalent to the content swap
        i64 *tmp = d_past;
        d_past = d_curr;
        d_curr = d_target;
        d_target = tmp;
        if (numIters >= 10000) break; // safety net, never hit in practice
    }

    for (size_t k = 0; k + 1 < sweep_ev.size() + 1; k += 2) {
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, sweep_ev[k], sweep_ev[k + 1]));
        e->stats.sweep_ms += ms;
    }
    e->stats.iters = numIters;
    e->stats.total_ms = std::chrono::duration<double, std::milli>(
                            std::chrono::steady_clock::now() - t_start)
                            .count();
    *iters_out = numIters;
    return prevMod; // dspl.hpp:1440
}

} // extern "C"
