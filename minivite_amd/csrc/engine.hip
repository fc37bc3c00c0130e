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
// Device data layout:
//  * Per-vertex working arrays (currComm/pastComm/targetComm, vDegree,
//    clusterWeight) live in INTERNAL order: a spatial permutation sigma
//    (the builder's locality hint) so that a vertex's neighbors — whose
//    GLOBAL ids the RGG assigns randomly — sit nearby in memory and the
//    per-edge community gathers hit L1/L2 instead of round-tripping to the
//    Infinity Cache (measured: the label-order layout fetched 3.5x its
//    algorithmic bytes at 13% L2 hit rate, 86% wave-wait). Community
//    LABELS, the wire protocol and every result are untouched: sigma is
//    pure layout. Without a hint, internal order = degree-sorted (load
//    balance for skewed graphs).
//  * Edges: SELL-64 (sliced ELL, slice = one wave64) over internal order;
//    edge step k of a wave loads 64 consecutive int32 pre-translated tails
//    (local -> internal index, ghost -> lnv + slot in the sorted ghost
//    list) — one coalesced 256-B line pair per wave-instruction, replacing
//    the reference's 16-B AoS row walk (graph.hpp:60-66) and per-edge hash
//    lookup (dspl.hpp:253-260). Unit-weight graphs skip the weight stream
//    entirely (every w == 1.0, detected at load).
//  * Community info (localCinfo/localCupdate, dspl.hpp:61-66) is AoS
//    {int64 size; double degree} in INTERNAL order. Persistent community
//    arrays carry LABELS; sweeps run over u32 slot (p==1) / view (p>1)
//    encodings that index cinfo/rc_info directly (see k_build_view).
//
// FP discipline: built with -ffp-contract=off so the dQ gain expression
// (dspl.hpp:212) and all accumulations carry the same bits as the
// gcc-built reference / oracle (generic x86-64 has no FMA contraction).
// Per-vertex weight accumulation is sequential in edge order (one lane owns
// one vertex), matching dspl.hpp:240-271 exactly.
//
// This is GPU-only code: engine creation fails loudly without a device.
// There is no CPU fallback anywhere in this file.

#include <algorithm>
#include <chrono>
#include <condition_variable>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <type_traits>
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

struct Cinfo {     // localCinfo / localCupdate entry (Comm, dspl.hpp:61-66)
    i64 size;
    double degree;
};

struct Info16 {    // wire record for cinfo replies / delta routing: the
    i64 size;      // CommInfo payload (dspl.hpp:68-72) minus the community
    double degree; // id, which both ends know by position
};

// Speculation-safe index clamp: semantically-exclusive branches may be
// if-converted into two unconditional loads + a value select, so the
// un-taken side's address must still be dereferenceable (ghost/rc
// buffers are never null — load_graph keeps 1-element dummies — and
// indices are clamped non-negative).
__device__ __forceinline__ i64 clamp0(i64 x) { return x > 0 ? x : 0; }

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

// ---- K1: vDegree + localCinfo init (dspl.hpp:82-107); both
// internal-ordered ----
__global__ void k1_vertex_degree(i64 lnv, const unsigned *__restrict__ sigma,
                                 const i64 *__restrict__ xadj,
                                 const double *__restrict__ ew, int unit,
                                 double *__restrict__ vDegree,
                                 Cinfo *__restrict__ cinfo) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < lnv;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 v = sigma[k];
        const i64 e0 = xadj[v], e1 = xadj[v + 1];
        double tw;
        if (unit) {
            tw = (double)(e1 - e0); // unit weights: exact
        } else {
            tw = 0.0;
            for (i64 e = e0; e < e1; e++) tw += ew[e]; // sequential edge order
        }
        vDegree[k] = tw;
        cinfo[k] = {1, tw}; // dspl.hpp:104-105 (internal slot k)
    }
}

// block-partial sum for K2/K7 (deterministic: fixed block ranges, fixed
// shuffle tree; partials summed on host in block order)
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

// ---- K3: community iota (dspl.hpp:132-149): internal slot k holds the
// LABEL of the vertex it represents (multi-rank arrays carry labels) ----
__global__ void k3_init_comm(i64 lnv, i64 base,
                             const unsigned *__restrict__ sigma,
                             i64 *__restrict__ curr, i64 *__restrict__ past) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < lnv;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 l = (i64)sigma[k] + base;
        curr[k] = l;
        past[k] = l;
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

// ---- SELL build (per run, setup span) ----
__global__ void k_degrees(i64 lnv, const unsigned *__restrict__ sigma,
                          const i64 *__restrict__ xadj,
                          unsigned *__restrict__ deg) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < lnv;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 v = sigma[k];
        deg[k] = (unsigned)(xadj[v + 1] - xadj[v]);
    }
}

__global__ void k_iota32(i64 n, unsigned *__restrict__ out) {
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < n;
         i += (i64)gridDim.x * blockDim.x)
        out[i] = (unsigned)i;
}

__global__ void k_invert_perm(i64 n, const unsigned *__restrict__ perm,
                              unsigned *__restrict__ inv) {
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < n;
         i += (i64)gridDim.x * blockDim.x)
        inv[perm[i]] = (unsigned)i;
}

// slice widths: chunk c spans 64 SELL positions, width = its max degree;
// emitted as element counts (width*64)
__global__ void k_chunk_sizes(i64 nchunks, i64 lnv,
                              const unsigned *__restrict__ perm,
                              const unsigned *__restrict__ deg,
                              i64 *__restrict__ sizes) {
    for (i64 c = blockIdx.x * (i64)blockDim.x + threadIdx.x; c < nchunks;
         c += (i64)gridDim.x * blockDim.x) {
        unsigned w = 0;
        const i64 s1 = min(c * 64 + 64, lnv);
        for (i64 s = c * 64; s < s1; s++) w = max(w, deg[perm[s]]);
        sizes[c] = (i64)w * 64;
    }
}

// translate + scatter this rank's CSR rows into the SELL image
__global__ void k_fill_sell(i64 lnv, const unsigned *__restrict__ perm,
                            const unsigned *__restrict__ sigma,
                            const unsigned *__restrict__ sigma_inv,
                            const i64 *__restrict__ xadj,
                            const i64 *__restrict__ tails,
                            const double *__restrict__ w, i64 base, i64 bound,
                            const i64 *__restrict__ ghosts, i64 nghost,
                            const i64 *__restrict__ chunk_off,
                            int *__restrict__ sell_tidx,
                            double *__restrict__ sell_w) {
    for (i64 s = blockIdx.x * (i64)blockDim.x + threadIdx.x; s < lnv;
         s += (i64)gridDim.x * blockDim.x) {
        const i64 iint = perm[s];
        const i64 v = sigma[iint];
        const i64 off = chunk_off[s >> 6];
        const int l = (int)(s & 63);
        const i64 e0 = xadj[v], e1 = xadj[v + 1];
        for (i64 e = e0; e < e1; e++) {
            const i64 t = tails[e];
            const i64 ti = (t >= base && t < bound)
                               ? (i64)sigma_inv[t - base]
                               : lnv + dev_lower_bound(ghosts, nghost, t);
            const i64 slot = off + (e - e0) * 64 + l;
            sell_tidx[slot] = (int)ti;
            if (sell_w) sell_w[slot] = w[e];
        }
    }
}

__global__ void k_scatter_mark(i64 n, const unsigned *__restrict__ idx,
                               unsigned char *__restrict__ mark) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x)
        mark[idx[k]] = 1;
}

__global__ void k_gather_flags(i64 lnv, const unsigned *__restrict__ perm,
                               const unsigned char *__restrict__ mark,
                               unsigned char *__restrict__ flags,
                               int invert) {
    for (i64 s = blockIdx.x * (i64)blockDim.x + threadIdx.x; s < lnv;
         s += (i64)gridDim.x * blockDim.x)
        flags[s] = mark[perm[s]] ^ invert;
}

__global__ void k_spill_uniform(i64 nthreads, i64 per,
                                i64 *__restrict__ off) {
    for (i64 t = blockIdx.x * (i64)blockDim.x + threadIdx.x; t < nthreads;
         t += (i64)gridDim.x * blockDim.x)
        off[t] = t * per;
}

// Per-row insertion sort of the SELL image by INTERNAL tail index.
// UNIT graphs only: their per-community sums are integer-exact under any
// accumulation order and the argmax tie-break is a total order on labels,
// so edge order cannot change any result — while ascending gather
// addresses inside the sigma spatial window raise L1/L2 line reuse
// (measured +4.7% on the steady-state harness, experiments/RESULTS.md).
// -w graphs keep the reference's edge order (bit-exact accumulation).
__global__ void k_sell_sort_rows(i64 lnv, const unsigned *__restrict__ perm,
                                 const unsigned *__restrict__ deg_int,
                                 const i64 *__restrict__ chunk_off,
                                 int *__restrict__ sell_tidx) {
    for (i64 s = blockIdx.x * (i64)blockDim.x + threadIdx.x; s < lnv;
         s += (i64)gridDim.x * blockDim.x) {
        const int d = (int)deg_int[perm[s]];
        const i64 eb = chunk_off[s >> 6] + (s & 63);
        for (int k = 1; k < d; k++) {
            const int v = sell_tidx[eb + (i64)k * 64];
            int j = k - 1;
            while (j >= 0) {
                const int u = sell_tidx[eb + (i64)j * 64];
                if (u <= v) break;
                sell_tidx[eb + (i64)(j + 1) * 64] = u;
                j--;
            }
            sell_tidx[eb + (i64)(j + 1) * 64] = v;
        }
    }
}

// per-thread spill extents for skewed graphs: with the degree-DESCENDING
// SELL order, grid-stride thread t's largest vertex is its first position,
// so its spill need is max(deg_sorted[t] - min_slots, 1); offsets are the
// host-side prefix sum of these
__global__ void k_spill_need(i64 nthreads, i64 s_begin, i64 lnv,
                             const unsigned *__restrict__ perm,
                             const unsigned *__restrict__ deg, int min_slots,
                             i64 *__restrict__ need) {
    for (i64 t = blockIdx.x * (i64)blockDim.x + threadIdx.x; t < nthreads;
         t += (i64)gridDim.x * blockDim.x) {
        i64 n = 1;
        if (s_begin + t < lnv) {
            const i64 d = (i64)deg[perm[s_begin + t]] - min_slots;
            n = d > 1 ? d : 1;
        }
        need[t] = n;
    }
}

// translate the full CSR tail array into int32 internal indices (local ->
// internal slot, ghost -> lnv + ghost index) once at build: the hub wave
// kernels walk CSR rows and would otherwise pay sigma_inv[tails[e]] — two
// dependent random gathers — per edge per iteration
__global__ void k_translate_tails(i64 lne, const i64 *__restrict__ tails,
                                  i64 base, i64 bound,
                                  const unsigned *__restrict__ sigma_inv,
                                  const i64 *__restrict__ ghosts, i64 nghost,
                                  i64 lnv, int *__restrict__ out) {
    for (i64 e = blockIdx.x * (i64)blockDim.x + threadIdx.x; e < lne;
         e += (i64)gridDim.x * blockDim.x) {
        const i64 t = tails[e];
        out[e] = (t >= base && t < bound)
                     ? (int)sigma_inv[t - base]
                     : (int)(lnv + dev_lower_bound(ghosts, nghost, t));
    }
}

// translate a global-id list into internal indices (svdata, once per run)
__global__ void k_to_internal(i64 n, const i64 *__restrict__ gids, i64 base,
                              const unsigned *__restrict__ sigma_inv,
                              unsigned *__restrict__ out) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x)
        out[k] = sigma_inv[gids[k] - base];
}

// ---- the u32 community VIEW (multi-rank sweeps) ----
// Persistent community arrays carry LABELS (the reference's community ids,
// global vertex ids). Each iteration builds a u32 view: a local community
// is its home vertex's internal SLOT (< lnv, indexes cinfo/cupd directly),
// a remote one is lnv + its index in the sorted rc_ids array (indexes
// rc_info/rcu directly). The per-edge community gather halves to 4 B and
// the gain scan's per-candidate binary search disappears; only the
// tie-break needs label ORDER, fetched lazily (see k4_sweep_p1's note).
__global__ void k_build_view(i64 n, const i64 *__restrict__ labels, i64 base,
                             i64 bound, i64 lnv,
                             const unsigned *__restrict__ sigma_inv,
                             const i64 *__restrict__ rc_ids, i64 nrc,
                             unsigned *__restrict__ view) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 l = labels[k];
        view[k] = (l >= base && l < bound)
                      ? sigma_inv[l - base]
                      : (unsigned)(lnv + clamp0(dev_bsearch(rc_ids, nrc, l)));
    }
}

// targets come out of the sweep as views; persist them as labels
__global__ void k_view_to_labels(i64 s0, i64 s1,
                                 const unsigned *__restrict__ perm,
                                 const unsigned *__restrict__ vtarget,
                                 const unsigned *__restrict__ sigma, i64 base,
                                 const i64 *__restrict__ rc_ids, i64 lnv,
                                 i64 *__restrict__ out) {
    for (i64 s = s0 + blockIdx.x * (i64)blockDim.x + threadIdx.x; s < s1;
         s += (i64)gridDim.x * blockDim.x) {
        const i64 i = perm[s];
        const unsigned v = vtarget[i];
        out[i] = (v < lnv) ? base + (i64)sigma[v] : rc_ids[v - lnv];
    }
}

// ---- halo #1a delta compaction ----
// The reference resends the FULL ghost community set every iteration
// (dspl.hpp:583-647). Here each export k compares against the label LAST
// SENT and emits a (index-in-segment, label) pair into its peer segment;
// the allgathered count matrix then lets both ends of every pair decide
// full vs compact identically (compact record 12 B vs full 8 B/entry).
// last_sent is updated unconditionally: in full mode the wire carries
// scdata itself, so "last sent" == current either way.
__global__ void k_delta_compact(i64 ssz, const i64 *__restrict__ soff,
                                int nranks, const i64 *__restrict__ scdata,
                                i64 *__restrict__ last_sent,
                                unsigned *__restrict__ chg_idx,
                                i64 *__restrict__ chg_lab,
                                unsigned long long *__restrict__ cnt) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < ssz;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 c = scdata[k];
        if (last_sent[k] == c) continue;
        last_sent[k] = c;
        const int r = (int)dev_lower_bound(soff, nranks + 1, k + 1) - 1;
        const unsigned long long pos = atomicAdd(&cnt[r], 1ull);
        chg_idx[soff[r] + pos] = (unsigned)(k - soff[r]);
        chg_lab[soff[r] + pos] = c;
    }
}

__global__ void k_scatter_deltas(i64 n, const unsigned *__restrict__ idx,
                                 const i64 *__restrict__ lab, i64 seg_base,
                                 i64 *__restrict__ ghost_labels) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x)
        ghost_labels[seg_base + idx[k]] = lab[k];
}

// ---- K8: scdata gather (dspl.hpp:559-571); labels on the wire ----
__global__ void k8_gather_comms(i64 n, const unsigned *__restrict__ svdata_int,
                                const i64 *__restrict__ currComm,
                                i64 *__restrict__ out) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x)
        out[k] = currComm[svdata_int[k]];
}

// ---- candidate remote communities (dspl.hpp:670-700) ----
__global__ void k_filter_remote(i64 n, const i64 *__restrict__ vals,
                                int shift, i64 base, i64 bound,
                                i64 *__restrict__ out,
                                unsigned long long *__restrict__ count) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        const i64 c = vals[k] >> shift; // label
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
                              const unsigned *__restrict__ sigma_inv,
                              const Cinfo *__restrict__ cinfo,
                              Info16 *__restrict__ out) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        const Cinfo c = cinfo[sigma_inv[req_ids[k] - base]];
        out[k] = {c.size, c.degree};
    }
}

// apply received deltas to owned cinfo (dspl.hpp:1089-1102; atomics because
// several senders may address one community)
__global__ void k_apply_deltas(i64 n, const i64 *__restrict__ ids, i64 base,
                               const unsigned *__restrict__ sigma_inv,
                               const Info16 *__restrict__ deltas,
                               Cinfo *__restrict__ cinfo) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < n;
         k += (i64)gridDim.x * blockDim.x) {
        Cinfo *c = &cinfo[sigma_inv[ids[k] - base]];
        atomic_add_i64(&c->size, deltas[k].size);
        atomicAdd(&c->degree, deltas[k].degree);
    }
}

// ---- K6: apply localCupdate (dspl.hpp:458-471), fused with the next
// iteration's K5 zeroing of localCupdate (dspl.hpp:483-484) ----
__global__ void k6_apply_local(i64 lnv, Cinfo *__restrict__ cupd,
                               Cinfo *__restrict__ cinfo) {
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < lnv;
         i += (i64)gridDim.x * blockDim.x) {
        const Cinfo u = cupd[i];
        cinfo[i].size += u.size;
        cinfo[i].degree += u.degree;
        cupd[i] = {0, 0.0};
    }
}

// K6+K7 fused (single-rank path): apply localCupdate, zero it, and emit
// this block's modularity partials in one pass (no remote deltas exist
// between K6 and K7 when nranks == 1)
__global__ void k67_apply_and_partials(i64 lnv, Cinfo *__restrict__ cupd,
                                       Cinfo *__restrict__ cinfo,
                                       const double *__restrict__ cw,
                                       double *__restrict__ out2) {
    double a = 0.0, b = 0.0;
    for (i64 i = blockIdx.x * (i64)blockDim.x + threadIdx.x; i < lnv;
         i += (i64)gridDim.x * blockDim.x) {
        const Cinfo u = cupd[i];
        const i64 nsz = cinfo[i].size + u.size;
        const double nde = cinfo[i].degree + u.degree;
        cinfo[i] = {nsz, nde};
        cupd[i] = {0, 0.0};
        a += cw[i];
        b += nde * nde;
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

// de-permute an internal-ordered label array into original order (trace)
__global__ void k_depermute(i64 lnv, const unsigned *__restrict__ sigma_inv,
                            const i64 *__restrict__ in,
                            i64 *__restrict__ out) {
    for (i64 v = blockIdx.x * (i64)blockDim.x + threadIdx.x; v < lnv;
         v += (i64)gridDim.x * blockDim.x)
        out[v] = in[sigma_inv[v]];
}

// ---- K4: THE sweep (distExecuteLouvainIteration, dspl.hpp:276-405) ----
// One lane per vertex over the SELL image (see the layout note at the top).
// The clmap/counter hash (dspl.hpp:230-274) lives as per-lane slot arrays:
// the first SLOTS distinct neighbor communities in LDS (lane-strided,
// conflict-free for ds b64), the tail in a per-thread global spill region
// (only populated while communities are still fine-grained; L2-resident).
// The current community's accumulator is a register (the reference's
// counter[0], dspl.hpp:312-318).
// ---- K4 multi-rank sweep over the u32 VIEW (see k_build_view) ----
// Same structure as k4_sweep_p1: per-lane LDS slot arrays + spill, edges
// in chunks of 8, lazy tie-break labels. A community value < lnv is a
// local internal slot (cinfo/cupd index); >= lnv is lnv + rc index
// (rc_info/rcu index). Labels for the tie-break / singleton guard come
// from sigma (+base) or rc_ids, fetched only on gain ties (dspl.hpp:
// 174-228 order semantics preserved bit-for-bit).
template <int SLOTS, bool UNIT>
__global__ __launch_bounds__(256) void k4_sweep_mr(
    i64 s_begin, i64 s_end, i64 lnv, i64 base,
    const unsigned *__restrict__ perm,
    const unsigned *__restrict__ deg_int, const i64 *__restrict__ chunk_off,
    const int *__restrict__ sell_tidx, const double *__restrict__ sell_w,
    const unsigned *__restrict__ vcurr, const unsigned *__restrict__ vghost,
    const double *__restrict__ vDegree, const unsigned *__restrict__ sigma,
    const Cinfo *__restrict__ cinfo, Cinfo *__restrict__ cupd,
    const i64 *__restrict__ rc_ids,
    const Info16 *__restrict__ rc_info, Info16 *__restrict__ rcu,
    double constant, unsigned *__restrict__ vtarget,
    double *__restrict__ clusterWeight, unsigned *__restrict__ spill_keys,
    double *__restrict__ spill_acc, const i64 *__restrict__ spill_off) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    double *sacc = reinterpret_cast<double *>(smem);
    unsigned *skey = reinterpret_cast<unsigned *>(
        smem + sizeof(double) * SLOTS * blockDim.x);
    const int tid = threadIdx.x;
    const i64 gthread = blockIdx.x * (i64)blockDim.x + threadIdx.x;
    const i64 stride = (i64)gridDim.x * blockDim.x;
    unsigned *myspill_k = spill_keys + spill_off[gthread];
    double *myspill_a = spill_acc + spill_off[gthread];

    // lazy label fetch (tie-break order only, dspl.hpp:214-225)
    auto label_of = [&](unsigned v) -> i64 {
        return (v < lnv) ? base + (i64)sigma[v] : rc_ids[clamp0(v - lnv)];
    };

    for (i64 s = s_begin + gthread; s < s_end; s += stride) {
        const i64 i = perm[s];          // internal vertex index
        const int deg = (int)deg_int[i];
        const i64 ebase = chunk_off[s >> 6] + (s & 63);
        const unsigned cc = vcurr[i];
        double ccDeg;
        i64 ccSize;
        if (cc < lnv) { // dspl.hpp:296-307
            const Cinfo c = cinfo[cc];
            ccDeg = c.degree;
            ccSize = c.size;
        } else {
            const Info16 c = rc_info[cc - lnv];
            ccDeg = c.degree;
            ccSize = c.size;
        }
        unsigned target;
        if (deg == 0) {
            target = cc;          // dspl.hpp:323-324
            clusterWeight[i] = 0; // K5 semantics (dspl.hpp:481-482)
        } else {
            double c0 = 0.0, selfLoop = 0.0;
            int ns = 0, nspill = 0;
            // Edges in chunks of 8: issue the 8 independent tail loads and
            // the 8 dependent community gathers together (the per-edge
            // serial load->gather->probe chain was the measured bound: 80%
            // SQ_WAIT_ANY with LDS and VALU idle), then probe sequentially
            // in edge order (bit-exact -w accumulation, dspl.hpp:240-271).
            constexpr int CH = 8;
            for (int k0 = 0; k0 < deg; k0 += CH) {
                const int m = min(CH, deg - k0);
                i64 tb[CH];
                unsigned cb[CH];
                double wb[CH];
#pragma unroll
                for (int j = 0; j < CH; j++) {
                    const i64 slot =
                        (j < m) ? ebase + (i64)(k0 + j) * 64 : ebase;
                    tb[j] = sell_tidx[slot];
                    if (!UNIT) wb[j] = sell_w[slot];
                }
#pragma unroll
                for (int j = 0; j < CH; j++)
                    cb[j] = (tb[j] < lnv) ? vcurr[tb[j]]
                                          : vghost[clamp0(tb[j] - lnv)];
                for (int j = 0; j < m; j++) {
                    const i64 tidx = tb[j];
                    const double w = UNIT ? 1.0 : wb[j];
                    if (tidx == i) selfLoop += w; // dspl.hpp:247-248
                    const unsigned tcomm = cb[j];
                    if (tcomm == cc) { c0 += w; continue; }
                    bool found = false;
                    for (int t = 0; t < ns; t++) {
                        if (skey[t * blockDim.x + tid] == tcomm) {
                            sacc[t * blockDim.x + tid] += w;
                            found = true;
                            break;
                        }
                    }
                    if (found) continue;
                    if (ns < SLOTS) {
                        skey[ns * blockDim.x + tid] = tcomm;
                        sacc[ns * blockDim.x + tid] = w;
                        ns++;
                        continue;
                    }
                    for (int t = 0; t < nspill; t++) {
                        if (myspill_k[t] == tcomm) {
                            myspill_a[t] += w;
                            found = true;
                            break;
                        }
                    }
                    if (!found) { // spill_max covers the max degree
                        myspill_k[nspill] = tcomm;
                        myspill_a[nspill] = w;
                        nspill++;
                    }
                }
            }
            clusterWeight[i] = c0; // dspl.hpp:318 onto the K5-zeroed value

            // distGetMaxIndex (dspl.hpp:174-228); labels fetched lazily
            const double vdeg = vDegree[i];
            const double eix = c0 - selfLoop;
            const double ax = ccDeg - vdeg;
            double maxGain = 0.0;
            unsigned maxIndex = cc;
            i64 maxSize = ccSize;
            const int tot = ns + nspill;
            for (int t = 0; t < tot; t++) {
                const unsigned y = (t < ns) ? skey[t * blockDim.x + tid]
                                            : myspill_k[t - ns];
                const double eiy = (t < ns) ? sacc[t * blockDim.x + tid]
                                            : myspill_a[t - ns];
                double ay;
                i64 ysz;
                if (y < lnv) {
                    const Cinfo c = cinfo[y];
                    ay = c.degree;
                    ysz = c.size;
                } else {
                    const Info16 c = rc_info[y - lnv];
                    ay = c.degree;
                    ysz = c.size;
                }
                const double curGain =
                    2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant; // :212
                if (curGain > maxGain) {
                    maxGain = curGain;
                    maxIndex = y;
                    maxSize = ysz;
                } else if (curGain == maxGain && curGain != 0.0) {
                    // real branch, not a select: the label loads must not
                    // be folded into the gain-compare dataflow (ROCm 7.2
                    // if-conversion miscompile)
                    if (label_of(y) < label_of(maxIndex)) {
                        maxIndex = y;
                        maxSize = ysz;
                    }
                }
            }
            if (maxSize == 1 && ccSize == 1 && maxIndex != cc) { // :224-225
                if (label_of(maxIndex) > label_of(cc)) maxIndex = cc;
            }
            target = maxIndex;
        }

        if (target != cc) { // 4-case updates (dspl.hpp:331-399)
            const double vdeg = vDegree[i];
            if (cc < lnv) {
                Cinfo *u = &cupd[cc];
                atomicAdd(&u->degree, -vdeg);
                atomic_add_i64(&u->size, -1);
            } else {
                Info16 *u = &rcu[cc - lnv];
                atomicAdd(&u->degree, -vdeg);
                atomic_add_i64(&u->size, -1);
            }
            if (target < lnv) {
                Cinfo *u = &cupd[target];
                atomicAdd(&u->degree, vdeg);
                atomic_add_i64(&u->size, 1);
            } else {
                Info16 *u = &rcu[target - lnv];
                atomicAdd(&u->degree, vdeg);
                atomic_add_i64(&u->size, 1);
            }
        }
        vtarget[i] = target; // dspl.hpp:404 (view; persisted as a label)
    }
}

// ---- K4 high-degree path (wave-per-vertex), multi-rank VIEW mode ----
// For skewed graphs (Orkut-class hubs), one WAVE processes one vertex:
// lanes stride the CSR row (coalesced 8-B tails), aggregate into a
// per-vertex open-addressed hash in HBM (L2-resident; u32 view keys CASed
// from ~0, weights atomicAdd), then a wave-reduce argmax applies the
// reference tie-break as a total order on (gain, label<<32|view) — order-
// independent, so the reduce tree reproduces dspl.hpp:214-215 exactly.
// Unit-weight graphs only: their sums are integer-exact under any
// accumulation order; -w skewed graphs take the serial lane path instead
// (edge-order bit parity).
__global__ __launch_bounds__(256) void k4_sweep_hi_mr(
    i64 nhi, i64 lnv, i64 base, const unsigned *__restrict__ perm,
    const unsigned *__restrict__ deg_int, const unsigned *__restrict__ sigma,
    const i64 *__restrict__ xadj, const int *__restrict__ tidx32,
    const unsigned *__restrict__ vcurr, const unsigned *__restrict__ vghost,
    const double *__restrict__ vDegree, const Cinfo *__restrict__ cinfo,
    Cinfo *__restrict__ cupd, const i64 *__restrict__ rc_ids,
    const Info16 *__restrict__ rc_info, Info16 *__restrict__ rcu,
    double constant, unsigned *__restrict__ vtarget,
    double *__restrict__ clusterWeight, const i64 *__restrict__ hash_off,
    i64 *__restrict__ hkeys, double *__restrict__ hacc) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int wpb = blockDim.x >> 6;
    auto label_of = [&](unsigned v) -> i64 {
        return (v < lnv) ? base + (i64)sigma[v] : rc_ids[clamp0(v - lnv)];
    };
    for (i64 s = (i64)blockIdx.x * wpb + wid; s < nhi;
         s += (i64)gridDim.x * wpb) {
        const i64 i = perm[s];
        const i64 v = sigma[i];
        const int deg = (int)deg_int[i];
        const i64 e0 = xadj[v];
        const unsigned cc = vcurr[i];
        const i64 hoff = hash_off[s];
        const i64 cap = hash_off[s + 1] - hoff; // power of two
        double ccDeg;
        i64 ccSize;
        if (cc < lnv) {
            const Cinfo c = cinfo[cc];
            ccDeg = c.degree;
            ccSize = c.size;
        } else {
            const Info16 c = rc_info[cc - lnv];
            ccDeg = c.degree;
            ccSize = c.size;
        }
        double c0 = 0.0, selfLoop = 0.0;
        for (int k = lane; k < deg; k += 64) {
            const i64 ti = tidx32[e0 + k]; // pre-translated internal index
            const double w = 1.0;          // hub path is unit-only
            if (ti == i) selfLoop += w;
            const unsigned tcomm = (ti < lnv) ? vcurr[ti]
                                              : vghost[ti - lnv];
            if (tcomm == cc) { c0 += w; continue; }
            i64 pos = (i64)(((uint64_t)tcomm * 0x9E3779B97F4A7C15ull) >> 32) &
                      (cap - 1);
            for (;;) {
                const i64 prev = (i64)atomicCAS(
                    (unsigned long long *)&hkeys[hoff + pos],
                    (unsigned long long)(-1ll), (unsigned long long)tcomm);
                if (prev == -1 || prev == (i64)tcomm) {
                    atomicAdd(&hacc[hoff + pos], w);
                    break;
                }
                pos = (pos + 1) & (cap - 1);
            }
        }
        // wave sums (integer-exact for unit weights)
        for (int off = 32; off > 0; off >>= 1) {
            c0 += __shfl_down(c0, off, 64);
            selfLoop += __shfl_down(selfLoop, off, 64);
        }
        c0 = __shfl(c0, 0, 64);
        selfLoop = __shfl(selfLoop, 0, 64);
        const double vdeg = vDegree[i];
        const double eix = c0 - selfLoop;
        const double ax = ccDeg - vdeg;
        // per-lane best over hash slots, then wave-reduce under the total
        // order (gain desc, (label, view) asc); gain <= 0 maps to the
        // neutral element so it can never win (dspl.hpp:214-215)
        double bg = 0.0;
        i64 bl = INT64_MAX, bs = 0; // bl = (label << 32) | view
        for (i64 t = lane; t < cap; t += 64) {
            const i64 yk = hkeys[hoff + t];
            if (yk == -1) continue;
            const unsigned y = (unsigned)yk;
            const double eiy = hacc[hoff + t];
            double ay;
            i64 ysz;
            if (y < lnv) {
                const Cinfo c = cinfo[y];
                ay = c.degree;
                ysz = c.size;
            } else {
                const Info16 c = rc_info[y - lnv];
                ay = c.degree;
                ysz = c.size;
            }
            const double g =
                2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant;
            const i64 yh = (label_of(y) << 32) | (i64)y;
            if (g > bg || (g == bg && g != 0.0 && yh < bl)) {
                bg = g;
                bl = yh;
                bs = ysz;
            }
        }
        for (int off = 32; off > 0; off >>= 1) {
            const double og = __shfl_down(bg, off, 64);
            const i64 ol = __shfl_down(bl, off, 64);
            const i64 os = __shfl_down(bs, off, 64);
            if (og > bg || (og == bg && ol < bl)) {
                bg = og;
                bl = ol;
                bs = os;
            }
        }
        if (lane == 0) {
            unsigned target =
                (bl == INT64_MAX) ? cc : (unsigned)(bl & 0xffffffffu);
            const i64 tLabel =
                (bl == INT64_MAX) ? label_of(cc) : (bl >> 32);
            if (bs == 1 && ccSize == 1 && tLabel > label_of(cc))
                target = cc; // :224-225
            if (deg == 0) {
                clusterWeight[i] = 0;
                target = cc;
            } else {
                clusterWeight[i] = c0;
            }
            if (target != cc) {
                if (cc < lnv) {
                    Cinfo *u = &cupd[cc];
                    atomicAdd(&u->degree, -vdeg);
                    atomic_add_i64(&u->size, -1);
                } else {
                    Info16 *u = &rcu[cc - lnv];
                    atomicAdd(&u->degree, -vdeg);
                    atomic_add_i64(&u->size, -1);
                }
                if (target < lnv) {
                    Cinfo *u = &cupd[target];
                    atomicAdd(&u->degree, vdeg);
                    atomic_add_i64(&u->size, 1);
                } else {
                    Info16 *u = &rcu[target - lnv];
                    atomicAdd(&u->degree, vdeg);
                    atomic_add_i64(&u->size, 1);
                }
            }
            vtarget[i] = target;
        }
    }
}

// ---- K4 lane-hash path (skewed graphs, mid/high-degree band) ----
// One LANE per vertex — accumulation stays in edge order, bit-exact for
// -w (dspl.hpp:240-271) — with the clmap as an open-addressed per-VERTEX
// hash (d_hash_off regions) PLUS a compact candidate list (d_lh_list):
// O(1) expected insert/lookup instead of the LDS path's O(ncand) linear
// probe per edge, and the argmax scans exactly ncand entries instead of
// the whole hash capacity. This is the aggregation structure for
// vertices whose persistent distinct-candidate count exceeds the LDS
// slots (long-range-edge-heavy social graphs). Works for p==1 too: the
// u32 slot array IS the view there (base 0, no ghosts).
template <bool UNIT>
__global__ __launch_bounds__(256) void k4_sweep_lh(
    i64 s_begin, i64 s_end, i64 lnv, i64 base,
    const unsigned *__restrict__ perm,
    const unsigned *__restrict__ deg_int, const i64 *__restrict__ chunk_off,
    const int *__restrict__ sell_tidx, const double *__restrict__ sell_w,
    const unsigned *__restrict__ vcurr, const unsigned *__restrict__ vghost,
    const double *__restrict__ vDegree, const unsigned *__restrict__ sigma,
    const Cinfo *__restrict__ cinfo, Cinfo *__restrict__ cupd,
    const i64 *__restrict__ rc_ids, const Info16 *__restrict__ rc_info,
    Info16 *__restrict__ rcu, double constant,
    unsigned *__restrict__ vtarget, double *__restrict__ clusterWeight,
    const i64 *__restrict__ hash_off, i64 *__restrict__ hkeys,
    double *__restrict__ hacc, unsigned *__restrict__ lh_list) {
    const i64 gthread = blockIdx.x * (i64)blockDim.x + threadIdx.x;
    const i64 stride = (i64)gridDim.x * blockDim.x;
    auto label_of = [&](unsigned v) -> i64 {
        return (v < lnv) ? base + (i64)sigma[v] : rc_ids[clamp0(v - lnv)];
    };
    for (i64 s = s_begin + gthread; s < s_end; s += stride) {
        const i64 i = perm[s];
        const int deg = (int)deg_int[i];
        const i64 ebase = chunk_off[s >> 6] + (s & 63);
        const unsigned cc = vcurr[i];
        const i64 hoff = hash_off[s];
        const i64 cap = hash_off[s + 1] - hoff; // power of two
        unsigned *mylist = lh_list + (hoff >> 1); // cap/2 >= deg entries
        double ccDeg;
        i64 ccSize;
        if (cc < lnv) {
            const Cinfo c = cinfo[cc];
            ccDeg = c.degree;
            ccSize = c.size;
        } else {
            const Info16 c = rc_info[cc - lnv];
            ccDeg = c.degree;
            ccSize = c.size;
        }
        double c0 = 0.0, selfLoop = 0.0;
        int ncand = 0;
        constexpr int CH = 8;
        for (int k0 = 0; k0 < deg; k0 += CH) {
            const int m = min(CH, deg - k0);
            i64 tb[CH];
            unsigned cb[CH];
            double wb[CH];
#pragma unroll
            for (int j = 0; j < CH; j++) {
                const i64 slot = (j < m) ? ebase + (i64)(k0 + j) * 64 : ebase;
                tb[j] = sell_tidx[slot];
                if (!UNIT) wb[j] = sell_w[slot];
            }
#pragma unroll
            for (int j = 0; j < CH; j++)
                cb[j] = (tb[j] < lnv) ? vcurr[tb[j]]
                                      : vghost[clamp0(tb[j] - lnv)];
            for (int j = 0; j < m; j++) {
                const double w = UNIT ? 1.0 : wb[j];
                if (tb[j] == i) selfLoop += w; // dspl.hpp:247-248
                const unsigned tcomm = cb[j];
                if (tcomm == cc) { c0 += w; continue; }
                i64 pos = (i64)(((uint64_t)tcomm * 0x9E3779B97F4A7C15ull)
                                >> 32) & (cap - 1);
                for (;;) {
                    const i64 key = hkeys[hoff + pos];
                    if (key == (i64)tcomm) {
                        hacc[hoff + pos] += w; // edge-order: one lane owns
                        break;                 // this vertex
                    }
                    if (key == -1) {
                        hkeys[hoff + pos] = tcomm;
                        hacc[hoff + pos] = w;
                        mylist[ncand++] = (unsigned)pos;
                        break;
                    }
                    pos = (pos + 1) & (cap - 1);
                }
            }
        }
        clusterWeight[i] = c0; // dspl.hpp:318

        // distGetMaxIndex over the compact list (dspl.hpp:174-228); the
        // scan order is insertion order — fine: the argmax with the lazy
        // label tie-break is a total order, order-independent.
        const double vdeg = vDegree[i];
        const double eix = c0 - selfLoop;
        const double ax = ccDeg - vdeg;
        double maxGain = 0.0;
        unsigned maxIndex = cc;
        i64 maxSize = ccSize;
        for (int t = 0; t < ncand; t++) {
            const i64 pos = mylist[t];
            const unsigned y = (unsigned)hkeys[hoff + pos];
            const double eiy = hacc[hoff + pos];
            double ay;
            i64 ysz;
            if (y < lnv) {
                const Cinfo c = cinfo[y];
                ay = c.degree;
                ysz = c.size;
            } else {
                const Info16 c = rc_info[y - lnv];
                ay = c.degree;
                ysz = c.size;
            }
            const double curGain =
                2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant;
            if (curGain > maxGain) {
                maxGain = curGain;
                maxIndex = y;
                maxSize = ysz;
            } else if (curGain == maxGain && curGain != 0.0) {
                // real branch, not a select (DESIGN.md §4 miscompile note)
                if (label_of(y) < label_of(maxIndex)) {
                    maxIndex = y;
                    maxSize = ysz;
                }
            }
        }
        if (maxSize == 1 && ccSize == 1 && maxIndex != cc) { // :224-225
            if (label_of(maxIndex) > label_of(cc)) maxIndex = cc;
        }
        if (deg == 0) maxIndex = cc;
        if (maxIndex != cc) {
            if (cc < lnv) {
                Cinfo *u = &cupd[cc];
                atomicAdd(&u->degree, -vdeg);
                atomic_add_i64(&u->size, -1);
            } else {
                Info16 *u = &rcu[cc - lnv];
                atomicAdd(&u->degree, -vdeg);
                atomic_add_i64(&u->size, -1);
            }
            if (maxIndex < lnv) {
                Cinfo *u = &cupd[maxIndex];
                atomicAdd(&u->degree, vdeg);
                atomic_add_i64(&u->size, 1);
            } else {
                Info16 *u = &rcu[maxIndex - lnv];
                atomicAdd(&u->degree, vdeg);
                atomic_add_i64(&u->size, 1);
            }
        }
        vtarget[i] = maxIndex;
    }
}

// ---- K4 first-iteration specialization ----
// At iteration 1 currComm is the identity (dspl.hpp:145-147), so per vertex:
// counter[0] collects exactly the self-loop weight (eix = counter[0] -
// selfLoop == 0.0, dspl.hpp:183), ax = ccDegree - vDegree == 0.0 (:185),
// every community has size 1, and each distinct neighbor TAIL is its own
// candidate. With per-row tails sorted ascending (the reference CSR's
// order, graph.hpp:1145-1153 — verified at load), parallel edges are
// adjacent and candidates arrive in ascending label order, so the
// tie-break (max gain, then smallest label, dspl.hpp:214-215) reduces to a
// strict-greater streaming argmax: no clmap at all, and the candidate's
// degree ay is vDegree[tidx] — a spatially local gather instead of a
// random cinfo line. The singleton guard (dspl.hpp:224-225) is the final
// maxIndex > cc check. Gains carry the identical bits: 2.0*(eiy-0.0) and
// (ay-0.0) round exactly like the reference's expressions.
// Multi-rank iteration-1 streaming specialization, VIEW mode (see
// k4_sweep_iter1_p1's header note). currComm is the identity, so a LOCAL
// candidate's view is its tail index and its ay is vDegree[tidx]; a ghost
// candidate's view comes from vghost (already lnv + rc index — the old
// per-edge rc binary search disappears). Labels for the final singleton
// guard come batched from sigma / the sorted ghost id list.
// LBL_ASC: rows are label-ascending (the reference CSR order), so the
// strict-greater argmax implicitly keeps the smallest label on gain ties.
// With internal-sorted rows (k_sell_sort_rows) candidates stream in
// INTERNAL order instead and ties compare the batched labels explicitly.
template <bool UNIT, bool LBL_ASC>
__global__ __launch_bounds__(256) void k4_sweep_iter1_mr(
    i64 s_begin, i64 s_end, i64 lnv, i64 base,
    const unsigned *__restrict__ perm,
    const unsigned *__restrict__ deg_int, const i64 *__restrict__ chunk_off,
    const int *__restrict__ sell_tidx, const double *__restrict__ sell_w,
    const unsigned *__restrict__ vghost, const i64 *__restrict__ ghosts,
    const double *__restrict__ vDegree, const unsigned *__restrict__ sigma,
    Cinfo *__restrict__ cupd,
    const Info16 *__restrict__ rc_info, Info16 *__restrict__ rcu,
    double constant, unsigned *__restrict__ vtarget,
    double *__restrict__ clusterWeight) {
    const i64 gthread = blockIdx.x * (i64)blockDim.x + threadIdx.x;
    const i64 stride = (i64)gridDim.x * blockDim.x;
    for (i64 s = s_begin + gthread; s < s_end; s += stride) {
        const i64 i = perm[s];
        const int deg = (int)deg_int[i];
        const i64 ebase = chunk_off[s >> 6] + (s & 63);
        const unsigned cc = (unsigned)i; // own slot (identity)
        const i64 ccLabel = base + (i64)sigma[i];
        const double vdeg = vDegree[i];
        double c0 = 0.0;
        double maxGain = 0.0;
        unsigned maxIndex = cc;
        i64 maxLabel = ccLabel;
        i64 prev = INT64_MIN;
        unsigned pend_view = 0;
        i64 pend_label = 0;
        double eiy = 0.0, pend_ay = 0.0;
        bool pend = false;
        constexpr int CH = 8;
        for (int k0 = 0; k0 < deg; k0 += CH) {
            const int m = min(CH, deg - k0);
            i64 tb[CH], lb[CH];
            double wb[CH], vb[CH];
#pragma unroll
            for (int j = 0; j < CH; j++) {
                const i64 slot = (j < m) ? ebase + (i64)(k0 + j) * 64 : ebase;
                tb[j] = sell_tidx[slot];
                if (!UNIT) wb[j] = sell_w[slot];
            }
#pragma unroll
            for (int j = 0; j < CH; j++) {
                lb[j] = (tb[j] < lnv) ? base + (i64)sigma[tb[j]]
                                      : ghosts[clamp0(tb[j] - lnv)];
                vb[j] = (tb[j] < lnv) ? vDegree[tb[j]] : 0.0;
            }
            for (int j = 0; j < m; j++) {
                const i64 tidx = tb[j];
                const double w = UNIT ? 1.0 : wb[j];
                if (tidx == i) { c0 += w; continue; } // self: counter[0]
                if (tidx == prev) { eiy += w; continue; } // parallel edge
                if (pend) {
                    const double g =
                        2.0 * eiy - 2.0 * vdeg * pend_ay * constant;
                    if (g > maxGain) {
                        maxGain = g;
                        maxIndex = pend_view;
                        maxLabel = pend_label;
                    } else if (!LBL_ASC && g == maxGain && g != 0.0 &&
                               pend_label < maxLabel) {
                        maxIndex = pend_view; // dspl.hpp:214-215 tie
                        maxLabel = pend_label;
                    }
                }
                prev = tidx;
                if (tidx < lnv) {
                    pend_view = (unsigned)tidx; // candidate comm == tail
                    pend_ay = vb[j];
                } else {
                    pend_view = vghost[tidx - lnv]; // lnv + rc index
                    pend_ay = rc_info[clamp0((i64)pend_view - lnv)].degree;
                }
                pend_label = lb[j];
                eiy = w;
                pend = true;
            }
        }
        if (pend) {
            const double g = 2.0 * eiy - 2.0 * vdeg * pend_ay * constant;
            if (g > maxGain) {
                maxGain = g;
                maxIndex = pend_view;
                maxLabel = pend_label;
            } else if (!LBL_ASC && g == maxGain && g != 0.0 &&
                       pend_label < maxLabel) {
                maxIndex = pend_view;
                maxLabel = pend_label;
            }
        }
        if (maxLabel > ccLabel) maxIndex = cc; // singleton guard
        clusterWeight[i] = c0;                 // dspl.hpp:318 (eix == 0)
        if (maxIndex != cc) {                  // cc is local at iteration 1
            Cinfo *u = &cupd[i];
            atomicAdd(&u->degree, -vdeg);
            atomic_add_i64(&u->size, -1);
            if (maxIndex < lnv) {
                Cinfo *t = &cupd[maxIndex];
                atomicAdd(&t->degree, vdeg);
                atomic_add_i64(&t->size, 1);
            } else {
                Info16 *t = &rcu[maxIndex - lnv];
                atomicAdd(&t->degree, vdeg);
                atomic_add_i64(&t->size, 1);
            }
        }
        vtarget[i] = maxIndex;
    }
}


// ---- Single-rank (p==1) specializations: u32 SLOT communities ----
// At one rank every community is local (base 0, no ghosts, no remote
// paths), so a community value can be the 32-bit internal SLOT of its
// home vertex: equality probes only need community identity, and
// cinfo/cupd indexing needs exactly the slot — so the per-edge community
// gather (the sweep's dominant random traffic) halves to 4 B with NO
// added indirection on the info lookups. Only the reference's tie-break
// and singleton guard need label ORDER (dspl.hpp:214-225), and only per
// DISTINCT candidate: the label is sigma[slot], an independent
// spatially-clustered gather. (A pure-label variant was measured and
// rejected: sigma_inv[label]->cinfo adds a dependent-load chain to the
// latency-bound gain scan — 4M 104G vs 110G handles vs this design.)
__global__ void k3_init_comm32(i64 lnv, unsigned *__restrict__ curr,
                               unsigned *__restrict__ past) {
    for (i64 k = blockIdx.x * (i64)blockDim.x + threadIdx.x; k < lnv;
         k += (i64)gridDim.x * blockDim.x) {
        curr[k] = (unsigned)k; // slot k's own community (dspl.hpp:132-149)
        past[k] = (unsigned)k;
    }
}

__global__ void k_depermute32(i64 lnv, const unsigned *__restrict__ sigma,
                              const unsigned *__restrict__ sigma_inv,
                              const unsigned *__restrict__ in,
                              i64 *__restrict__ out) {
    for (i64 v = blockIdx.x * (i64)blockDim.x + threadIdx.x; v < lnv;
         v += (i64)gridDim.x * blockDim.x)
        out[v] = (i64)sigma[in[sigma_inv[v]]]; // slot -> label
}

template <int SLOTS, bool UNIT>
__global__ __launch_bounds__(256) void k4_sweep_p1(
    i64 s_begin, i64 lnv, const unsigned *__restrict__ perm,
    const unsigned *__restrict__ deg_int, const i64 *__restrict__ chunk_off,
    const int *__restrict__ sell_tidx, const double *__restrict__ sell_w,
    const unsigned *__restrict__ currComm,
    const double *__restrict__ vDegree, const unsigned *__restrict__ sigma,
    const Cinfo *__restrict__ cinfo, Cinfo *__restrict__ cupd,
    double constant, unsigned *__restrict__ targetComm,
    double *__restrict__ clusterWeight, unsigned *__restrict__ spill_keys,
    double *__restrict__ spill_acc, const i64 *__restrict__ spill_off) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    double *sacc = reinterpret_cast<double *>(smem);
    unsigned *skey = reinterpret_cast<unsigned *>(
        smem + sizeof(double) * SLOTS * blockDim.x);
    const int tid = threadIdx.x;
    const i64 gthread = blockIdx.x * (i64)blockDim.x + threadIdx.x;
    const i64 stride = (i64)gridDim.x * blockDim.x;
    unsigned *myspill_k = spill_keys + spill_off[gthread];
    double *myspill_a = spill_acc + spill_off[gthread];

    for (i64 s = s_begin + gthread; s < lnv; s += stride) {
        const i64 i = perm[s];          // internal vertex index
        const int deg = (int)deg_int[i];
        const i64 ebase = chunk_off[s >> 6] + (s & 63);
        const unsigned cc = currComm[i]; // slot
        const Cinfo ci = cinfo[cc];
        const double ccDeg = ci.degree;
        const i64 ccSize = ci.size;
        unsigned target;
        if (deg == 0) {
            target = cc;          // dspl.hpp:323-324
            clusterWeight[i] = 0; // K5 semantics (dspl.hpp:481-482)
        } else {
            double c0 = 0.0, selfLoop = 0.0;
            int ns = 0, nspill = 0;
            constexpr int CH = 8;
            for (int k0 = 0; k0 < deg; k0 += CH) {
                const int m = min(CH, deg - k0);
                i64 tb[CH];
                unsigned cb[CH];
                double wb[CH];
#pragma unroll
                for (int j = 0; j < CH; j++) {
                    const i64 slot =
                        (j < m) ? ebase + (i64)(k0 + j) * 64 : ebase;
                    tb[j] = sell_tidx[slot];
                    if (!UNIT) wb[j] = sell_w[slot];
                }
#pragma unroll
                for (int j = 0; j < CH; j++) cb[j] = currComm[tb[j]];
                for (int j = 0; j < m; j++) {
                    const i64 tidx = tb[j];
                    const double w = UNIT ? 1.0 : wb[j];
                    if (tidx == i) selfLoop += w; // dspl.hpp:247-248
                    const unsigned tcomm = cb[j];
                    if (tcomm == cc) { c0 += w; continue; }
                    bool found = false;
                    for (int t = 0; t < ns; t++) {
                        if (skey[t * blockDim.x + tid] == tcomm) {
                            sacc[t * blockDim.x + tid] += w;
                            found = true;
                            break;
                        }
                    }
                    if (found) continue;
                    if (ns < SLOTS) {
                        skey[ns * blockDim.x + tid] = tcomm;
                        sacc[ns * blockDim.x + tid] = w;
                        ns++;
                        continue;
                    }
                    for (int t = 0; t < nspill; t++) {
                        if (myspill_k[t] == tcomm) {
                            myspill_a[t] += w;
                            found = true;
                            break;
                        }
                    }
                    if (!found) {
                        myspill_k[nspill] = tcomm;
                        myspill_a[nspill] = w;
                        nspill++;
                    }
                }
            }
            clusterWeight[i] = c0; // dspl.hpp:318

            // distGetMaxIndex (dspl.hpp:174-228); order on LABELS, but
            // labels (sigma[slot]) are fetched LAZILY — only when a gain
            // tie actually needs the order — so the common strict-greater
            // path costs no label traffic at all.
            const double vdeg = vDegree[i];
            const double eix = c0 - selfLoop;
            const double ax = ccDeg - vdeg;
            double maxGain = 0.0;
            unsigned maxIndex = cc;
            i64 maxSize = ccSize;
            const int tot = ns + nspill;
            for (int t = 0; t < tot; t++) {
                const unsigned y = (t < ns) ? skey[t * blockDim.x + tid]
                                            : myspill_k[t - ns];
                const double eiy = (t < ns) ? sacc[t * blockDim.x + tid]
                                            : myspill_a[t - ns];
                const Cinfo c = cinfo[y];
                const double ay = c.degree;
                const i64 ysz = c.size;
                const double curGain =
                    2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant; // :212
                if (curGain > maxGain) {
                    maxGain = curGain;
                    maxIndex = y;
                    maxSize = ysz;
                } else if (curGain == maxGain && curGain != 0.0) {
                    // real branch, not a select: the label loads must not
                    // be folded into the gain-compare dataflow
                    if (sigma[y] < sigma[maxIndex]) {
                        maxIndex = y;
                        maxSize = ysz;
                    }
                }
            }
            if (maxSize == 1 && ccSize == 1 && maxIndex != cc) { // :224-225
                if (sigma[maxIndex] > sigma[cc]) maxIndex = cc;
            }
            target = maxIndex;
        }

        if (target != cc) { // both communities local (dspl.hpp:331-399)
            const double vdeg = vDegree[i];
            Cinfo *u = &cupd[cc];
            atomicAdd(&u->degree, -vdeg);
            atomic_add_i64(&u->size, -1);
            Cinfo *t = &cupd[target];
            atomicAdd(&t->degree, vdeg);
            atomic_add_i64(&t->size, 1);
        }
        targetComm[i] = target; // dspl.hpp:404 (slot)
    }
}

// iteration-1 streaming specialization, slot mode (see k4_sweep_iter1's
// header note). currComm is the identity permutation, so a candidate
// community IS its tail index — no community gather at all; the
// label-order stream property and ay = vDegree[tidx] carry over.
template <bool UNIT, bool LBL_ASC>
__global__ __launch_bounds__(256) void k4_sweep_iter1_p1(
    i64 s_begin, i64 lnv, const unsigned *__restrict__ perm,
    const unsigned *__restrict__ deg_int, const i64 *__restrict__ chunk_off,
    const int *__restrict__ sell_tidx, const double *__restrict__ sell_w,
    const double *__restrict__ vDegree, const unsigned *__restrict__ sigma,
    Cinfo *__restrict__ cupd, double constant,
    unsigned *__restrict__ targetComm, double *__restrict__ clusterWeight) {
    const i64 gthread = blockIdx.x * (i64)blockDim.x + threadIdx.x;
    const i64 stride = (i64)gridDim.x * blockDim.x;
    for (i64 s = s_begin + gthread; s < lnv; s += stride) {
        const i64 i = perm[s];
        const int deg = (int)deg_int[i];
        const i64 ebase = chunk_off[s >> 6] + (s & 63);
        const unsigned cc = (unsigned)i; // own slot
        const unsigned ccLabel = sigma[i];
        const double vdeg = vDegree[i];
        double c0 = 0.0;
        double maxGain = 0.0;
        unsigned maxIndex = cc;
        unsigned maxLabel = ccLabel;
        i64 prev = INT64_MIN;
        unsigned pend_slot = 0, pend_label = 0;
        double eiy = 0.0, pend_ay = 0.0;
        bool pend = false;
        constexpr int CH = 8;
        for (int k0 = 0; k0 < deg; k0 += CH) {
            const int m = min(CH, deg - k0);
            i64 tb[CH];
            unsigned lb[CH];
            double wb[CH], vb[CH];
#pragma unroll
            for (int j = 0; j < CH; j++) {
                const i64 slot = (j < m) ? ebase + (i64)(k0 + j) * 64 : ebase;
                tb[j] = sell_tidx[slot];
                if (!UNIT) wb[j] = sell_w[slot];
            }
#pragma unroll
            for (int j = 0; j < CH; j++) {
                lb[j] = sigma[tb[j]];
                vb[j] = vDegree[tb[j]];
            }
            for (int j = 0; j < m; j++) {
                const i64 tidx = tb[j];
                const double w = UNIT ? 1.0 : wb[j];
                if (tidx == i) { c0 += w; continue; } // self: counter[0]
                if (tidx == prev) { eiy += w; continue; } // parallel edge
                if (pend) {
                    const double g =
                        2.0 * eiy - 2.0 * vdeg * pend_ay * constant;
                    if (g > maxGain) {
                        maxGain = g;
                        maxIndex = pend_slot;
                        maxLabel = pend_label;
                    } else if (!LBL_ASC && g == maxGain && g != 0.0 &&
                               pend_label < maxLabel) {
                        maxIndex = pend_slot; // dspl.hpp:214-215 tie
                        maxLabel = pend_label;
                    }
                }
                prev = tidx;
                pend_slot = (unsigned)tidx; // candidate community == tail
                pend_label = lb[j];
                eiy = w;
                pend_ay = vb[j];
                pend = true;
            }
        }
        if (pend) {
            const double g = 2.0 * eiy - 2.0 * vdeg * pend_ay * constant;
            if (g > maxGain) {
                maxGain = g;
                maxIndex = pend_slot;
                maxLabel = pend_label;
            } else if (!LBL_ASC && g == maxGain && g != 0.0 &&
                       pend_label < maxLabel) {
                maxIndex = pend_slot;
                maxLabel = pend_label;
            }
        }
        if (maxLabel > ccLabel) maxIndex = cc; // singleton guard
        clusterWeight[i] = c0;                 // dspl.hpp:318 (eix == 0)
        if (maxIndex != cc) {
            Cinfo *u = &cupd[i];
            atomicAdd(&u->degree, -vdeg);
            atomic_add_i64(&u->size, -1);
            Cinfo *t = &cupd[maxIndex];
            atomicAdd(&t->degree, vdeg);
            atomic_add_i64(&t->size, 1);
        }
        targetComm[i] = maxIndex;
    }
}

// wave-per-vertex hub path, slot mode (see k4_sweep_hi's header note;
// at p==1 a global tail id IS the local vertex id). The wave-reduce
// tie-break order rides a packed (label << 32 | slot) value, so label
// order decides and the slot comes along for free.
__global__ __launch_bounds__(256) void k4_sweep_hi_p1(
    i64 nhi, i64 lnv, const unsigned *__restrict__ perm,
    const unsigned *__restrict__ deg_int, const unsigned *__restrict__ sigma,
    const i64 *__restrict__ xadj,
    const int *__restrict__ tidx32, const unsigned *__restrict__ currComm,
    const double *__restrict__ vDegree, const Cinfo *__restrict__ cinfo,
    Cinfo *__restrict__ cupd, double constant,
    unsigned *__restrict__ targetComm, double *__restrict__ clusterWeight,
    const i64 *__restrict__ hash_off, i64 *__restrict__ hkeys,
    double *__restrict__ hacc) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int wpb = blockDim.x >> 6;
    for (i64 s = (i64)blockIdx.x * wpb + wid; s < nhi;
         s += (i64)gridDim.x * wpb) {
        const i64 i = perm[s];
        const i64 v = sigma[i];
        const int deg = (int)deg_int[i];
        const i64 e0 = xadj[v];
        const unsigned cc = currComm[i]; // slot
        const unsigned ccLabel = sigma[cc];
        const i64 hoff = hash_off[s];
        const i64 cap = hash_off[s + 1] - hoff; // power of two
        const Cinfo cci = cinfo[cc];
        const double ccDeg = cci.degree;
        const i64 ccSize = cci.size;
        double c0 = 0.0, selfLoop = 0.0;
        for (int k = lane; k < deg; k += 64) {
            const i64 ti = tidx32[e0 + k]; // pre-translated internal index
            const double w = 1.0;          // hub path is unit-only
            if (ti == i) selfLoop += w;
            const i64 tcomm = (i64)currComm[ti];
            if (tcomm == (i64)cc) { c0 += w; continue; }
            i64 pos = (i64)(((uint64_t)tcomm * 0x9E3779B97F4A7C15ull) >> 32) &
                      (cap - 1);
            for (;;) {
                const i64 prev = (i64)atomicCAS(
                    (unsigned long long *)&hkeys[hoff + pos],
                    (unsigned long long)(-1ll), (unsigned long long)tcomm);
                if (prev == -1 || prev == tcomm) {
                    atomicAdd(&hacc[hoff + pos], w);
                    break;
                }
                pos = (pos + 1) & (cap - 1);
            }
        }
        for (int off = 32; off > 0; off >>= 1) {
            c0 += __shfl_down(c0, off, 64);
            selfLoop += __shfl_down(selfLoop, off, 64);
        }
        c0 = __shfl(c0, 0, 64);
        selfLoop = __shfl(selfLoop, 0, 64);
        const double vdeg = vDegree[i];
        const double eix = c0 - selfLoop;
        const double ax = ccDeg - vdeg;
        double bg = 0.0;
        i64 bl = INT64_MAX, bs = 0; // bl = (label << 32) | slot
        for (i64 t = lane; t < cap; t += 64) {
            const i64 y = hkeys[hoff + t];
            if (y == -1) continue;
            const double eiy = hacc[hoff + t];
            const Cinfo c = cinfo[y];
            const double ay = c.degree;
            const i64 ysz = c.size;
            const double g =
                2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant;
            const i64 yh = ((i64)sigma[y] << 32) | y;
            if (g > bg || (g == bg && g != 0.0 && yh < bl)) {
                bg = g;
                bl = yh;
                bs = ysz;
            }
        }
        for (int off = 32; off > 0; off >>= 1) {
            const double og = __shfl_down(bg, off, 64);
            const i64 ol = __shfl_down(bl, off, 64);
            const i64 os = __shfl_down(bs, off, 64);
            if (og > bg || (og == bg && ol < bl)) {
                bg = og;
                bl = ol;
                bs = os;
            }
        }
        if (lane == 0) {
            unsigned target =
                (bl == INT64_MAX) ? cc : (unsigned)(bl & 0xffffffffu);
            const unsigned tLabel =
                (bl == INT64_MAX) ? ccLabel : (unsigned)(bl >> 32);
            if (bs == 1 && ccSize == 1 && tLabel > ccLabel)
                target = cc; // :224-225
            if (deg == 0) {
                clusterWeight[i] = 0;
                target = cc;
            } else {
                clusterWeight[i] = c0;
            }
            if (target != cc) {
                Cinfo *u = &cupd[cc];
                atomicAdd(&u->degree, -vdeg);
                atomic_add_i64(&u->size, -1);
                Cinfo *t = &cupd[target];
                atomicAdd(&t->degree, vdeg);
                atomic_add_i64(&t->size, 1);
            }
            targetComm[i] = target;
        }
    }
}

int grid_for(i64 n, int block = 256, int cap = 2048) {
    i64 g = (n + block - 1) / block;
    return (int)std::min<i64>(std::max<i64>(g, 1), cap);
}

} // namespace

// ------------------------------------------------------------------------
// Loopback transport: nranks engines in one process on one device, one
// host thread per rank (see the header note in minivite_hip.h). Exchange
// payloads move by device-to-device hipMemcpyAsync between the engines'
// buffers; host barriers stand in for the collective rendezvous. All
// reductions run in rank order (deterministic, like the RCCL fixed-tree
// at these tiny sizes and like the oracle).
struct mv_lb_session {
    int nranks;
    std::mutex mu;
    std::condition_variable cv;
    int count = 0;
    uint64_t gen = 0;
    struct Post {
        const void *send = nullptr;
        const void *send2 = nullptr; // delta mode: chg_idx
        const void *send3 = nullptr; // delta mode: chg_lab
        const int64_t *soff = nullptr;
        size_t eb = 0;
        const double *red = nullptr;
        std::vector<int64_t> cnts;
    };
    std::vector<Post> posts;
    explicit mv_lb_session(int n) : nranks(n), posts(n) {}
    void barrier() {
        std::unique_lock<std::mutex> lk(mu);
        const uint64_t g = gen;
        if (++count == nranks) {
            count = 0;
            gen++;
            cv.notify_all();
        } else {
            cv.wait(lk, [&] { return gen != g; });
        }
    }
};

struct mv_engine {
    int device = 0, rank = 0, nranks = 1;
    ncclComm_t comm = nullptr;
    ncclComm_t comm2 = nullptr; // halo #1a's communicator (overlap mode)
    mv_lb_session *lb = nullptr;
    hipStream_t stream = nullptr;
    hipStream_t stream2 = nullptr; // halo #1a rides here, overlapped
    hipEvent_t ev_p1 = nullptr;    // sweep part 1 (exports) done [stream]
    hipEvent_t ev_p2 = nullptr;    // full sweep + writeback done [stream]
    hipEvent_t ev_cinfo = nullptr; // K6 + delta application done [stream]
    hipEvent_t ev_halo = nullptr;  // next-iteration pipeline done [stream2]
    i64 *d_cnt_all = nullptr;      // persistent p*p counts buffer
    int overlap = 0;               // this run overlaps #1a (p>1, !skewed)
    i64 nexp = 0;                  // SELL positions [0,nexp) = exported

    // graph (device)
    i64 nv = 0, lnv = 0, lne = 0, base = 0, bound = 0;
    int sort_bits = 64; // ceil(log2(nv)): radix sort width for id keys
    std::vector<i64> parts_h;
    i64 *d_parts = nullptr;
    i64 *d_xadj = nullptr;
    i64 *d_tails = nullptr;   // raw global tails (kept for setup)
    double *d_ew = nullptr;   // edge weights (CSR order)
    int unit_weights = 1;
    int rows_sorted = 1; // per-row tails ascending (reference CSR order)
    int sell_sorted = 0; // SELL rows re-sorted by internal index (unit)
    i64 max_degree = 0;

    // internal layout
    unsigned *d_sigma = nullptr;     // internal -> original local id
    unsigned *d_sigma_inv = nullptr; // original local id -> internal
    int has_hint = 0;

    // SELL-64 image (built in the per-run setup span)
    unsigned *d_deg = nullptr;       // degree per INTERNAL index
    unsigned *d_iota = nullptr, *d_perm = nullptr; // SELL pos -> internal
    i64 *d_chunk_off = nullptr;      // nchunks+1 element offsets
    i64 nchunks = 0, sell_elems = 0;
    int *d_sell_tidx = nullptr;
    double *d_sell_w = nullptr;

    // per-run state (device)
    i64 *d_curr = nullptr, *d_past = nullptr, *d_target = nullptr;
    double *d_vdeg = nullptr, *d_cw = nullptr;
    Cinfo *d_cinfo = nullptr, *d_cupd = nullptr;
    double *d_partials = nullptr; // 2 * nblocks
    double *h_partials = nullptr; // pinned mirror (pageable D2H costs
                                  // ~40 us/iteration in staging latency)
    double *d_red = nullptr;      // 2 doubles for allreduce

    // ghosts / halo
    i64 *d_ghosts = nullptr;     // sorted unique remote tails
    i64 nghost = 0;
    // u32 community views (multi-rank sweeps; rebuilt per iteration from
    // the persistent LABEL arrays once rc_ids is known)
    unsigned *d_vcurr = nullptr;   // lnv
    unsigned *d_vghost = nullptr;  // nghost
    unsigned *d_vtarget = nullptr; // lnv (persisted back as labels)
    i64 *d_svdata = nullptr;     // vertices peers want from me (global ids)
    unsigned *d_svdata_int = nullptr;
    i64 ssz = 0;
    std::vector<i64> send_off, recv_off; // per-peer offsets into svdata / ghosts
    i64 *d_scdata = nullptr;             // packed comms to export

    // halo #1a delta compaction (send side tracks the last label sent per
    // export; receive side keeps the persistent ghost label image)
    int use_delta = 0;
    i64 *d_last_sent = nullptr;            // ssz, init -1 (all changed)
    unsigned *d_chg_idx = nullptr;         // ssz, segment-based layout
    i64 *d_chg_lab = nullptr;              // ssz
    unsigned long long *d_chg_cnt = nullptr; // nranks counters
    i64 *d_cntmat = nullptr;               // nranks*nranks allgather image
    i64 *h_cntmat = nullptr;               // pinned mirror
    i64 *d_soff = nullptr;                 // send_off on device (nranks+1)
    i64 *d_ghost_labels = nullptr;         // nghost persistent labels
    unsigned *d_rchg_idx = nullptr;        // nghost recv compact indices
    i64 *d_rchg_lab = nullptr;             // nghost recv compact labels

    // remote community info (per iteration)
    i64 rc_cap = 0;
    i64 *d_cand = nullptr, *d_cand_sorted = nullptr;
    i64 *d_rc_ids = nullptr;
    Info16 *d_rc_info = nullptr; // aligned with rc_ids
    Info16 *d_rcu = nullptr;     // remoteCupdate accumulators
    i64 req_cap = 0;
    i64 *d_req_ids = nullptr;    // ids other ranks requested from me
    Info16 *d_req_info = nullptr;
    i64 *d_bounds = nullptr;     // nranks+1
    unsigned long long *d_count = nullptr;
    void *d_cub_tmp = nullptr;
    size_t cub_tmp_bytes = 0;

    // spill for K4 (per-thread extents; uniform for flat degree
    // distributions, degree-prefix-sized for skewed ones)
    i64 *d_spill_k = nullptr;
    double *d_spill_a = nullptr;
    i64 *d_spill_off = nullptr;
    i64 spill_elems = 0;
    int skewed = 0;
    int sweep_grid = 0;

    // hash-aggregation bands over the degree-sorted positions:
    // [0, nhi) = wave-per-vertex hubs (unit only), [nhi, nlh) = lane-hash
    // band, [nlh, lnv) = LDS slots + linear spill
    i64 nhi = 0;
    i64 nlh = 0;
    unsigned *d_lh_list = nullptr; // compact candidate lists (hash/2)
    int *d_tidx32 = nullptr;   // lne pre-translated tails (hub kernels)
    i64 *d_hash_off = nullptr; // nlh+1 slot offsets (per-vertex caps are powers of two)
    i64 *d_hkeys = nullptr;
    double *d_hacc = nullptr;
    i64 hash_total = 0;
    i64 hash_hub_elems = 0; // prefix the wave kernel atomics into (needs
                            // zeroed accumulators each iteration)

    // trace
    i64 *trace_target = nullptr;
    double *trace_mod = nullptr;
    int trace_cap = 0;
    i64 *d_trace_tmp = nullptr;

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

static void build_sell(mv_engine *e);
static void free_graph_state(mv_engine *e);

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
        // halo #1a rides its own communicator so RCCL can run it
        // concurrently with the epilogue's collectives (MV_NO_OVERLAP must
        // be set on ALL ranks or none — ncclCommSplit is collective)
        if (!getenv("MV_NO_OVERLAP"))
            NCCL_CHECK(ncclCommSplit(e->comm, 0, rank, &e->comm2, nullptr));
    }
    HIP_CHECK(hipStreamCreate(&e->stream2));
    HIP_CHECK(hipEventCreate(&e->ev_p1));
    HIP_CHECK(hipEventCreate(&e->ev_p2));
    HIP_CHECK(hipEventCreate(&e->ev_cinfo));
    HIP_CHECK(hipEventCreate(&e->ev_halo));
    return e;
}

mv_lb_session *mv_lb_create(int nranks) { return new mv_lb_session(nranks); }

void mv_lb_destroy(mv_lb_session *s) { delete s; }

mv_engine *mv_engine_create_lb(int device, int rank, int nranks,
                               mv_lb_session *s) {
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev <= device) {
        std::fprintf(stderr,
                     "mv_engine_create_lb: no HIP device %d (found %d)\n",
                     device, ndev);
        return nullptr;
    }
    if (!s || s->nranks != nranks) {
        std::fprintf(stderr, "mv_engine_create_lb: bad session\n");
        return nullptr;
    }
    auto *e = new mv_engine;
    e->device = device;
    e->rank = rank;
    e->nranks = nranks;
    e->lb = s;
    HIP_CHECK(hipSetDevice(device));
    HIP_CHECK(hipStreamCreate(&e->stream));
    HIP_CHECK(hipStreamCreate(&e->stream2));
    HIP_CHECK(hipEventCreate(&e->ev_p1));
    HIP_CHECK(hipEventCreate(&e->ev_p2));
    HIP_CHECK(hipEventCreate(&e->ev_cinfo));
    HIP_CHECK(hipEventCreate(&e->ev_halo));
    return e;
}

void mv_engine_destroy(mv_engine *e) {
    if (!e) return;
    (void)hipSetDevice(e->device); // teardown: best effort
    for (auto ev : e->ev_pool) (void)hipEventDestroy(ev);
    if (e->ev_p1) (void)hipEventDestroy(e->ev_p1);
    if (e->ev_p2) (void)hipEventDestroy(e->ev_p2);
    if (e->ev_cinfo) (void)hipEventDestroy(e->ev_cinfo);
    if (e->ev_halo) (void)hipEventDestroy(e->ev_halo);
    if (e->comm2) ncclCommDestroy(e->comm2);
    if (e->comm) ncclCommDestroy(e->comm);
    free_graph_state(e);
    if (e->stream2) (void)hipStreamDestroy(e->stream2);
    if (e->stream) (void)hipStreamDestroy(e->stream);
    delete e;
}

// free everything a previous load_graph/run allocated (engines may be
// re-pointed at a new graph)
static void free_graph_state(mv_engine *e) {
    for (void **p : {(void **)&e->d_parts, (void **)&e->d_xadj,
                     (void **)&e->d_tails, (void **)&e->d_ew,
                     (void **)&e->d_sigma, (void **)&e->d_sigma_inv,
                     (void **)&e->d_deg, (void **)&e->d_iota,
                     (void **)&e->d_perm, (void **)&e->d_chunk_off,
                     (void **)&e->d_sell_tidx, (void **)&e->d_sell_w,
                     (void **)&e->d_curr, (void **)&e->d_past,
                     (void **)&e->d_target, (void **)&e->d_vdeg,
                     (void **)&e->d_cw, (void **)&e->d_cinfo,
                     (void **)&e->d_cupd, (void **)&e->d_partials,
                     (void **)&e->d_red, (void **)&e->d_count,
                     (void **)&e->d_bounds, (void **)&e->d_ghosts,
                     (void **)&e->d_vcurr, (void **)&e->d_vghost,
                     (void **)&e->d_vtarget, (void **)&e->d_svdata,
                     (void **)&e->d_svdata_int, (void **)&e->d_scdata,
                     (void **)&e->d_cand, (void **)&e->d_cand_sorted,
                     (void **)&e->d_rc_ids, (void **)&e->d_rc_info,
                     (void **)&e->d_rcu, (void **)&e->d_req_ids,
                     (void **)&e->d_req_info, (void **)&e->d_cub_tmp,
                     (void **)&e->d_spill_k, (void **)&e->d_spill_a,
                     (void **)&e->d_spill_off, (void **)&e->d_hash_off,
                     (void **)&e->d_hkeys, (void **)&e->d_hacc,
                     (void **)&e->d_tidx32, (void **)&e->d_lh_list,
                     (void **)&e->d_last_sent, (void **)&e->d_chg_idx,
                     (void **)&e->d_chg_lab, (void **)&e->d_chg_cnt,
                     (void **)&e->d_cntmat, (void **)&e->d_soff,
                     (void **)&e->d_ghost_labels, (void **)&e->d_rchg_idx,
                     (void **)&e->d_rchg_lab, (void **)&e->d_cnt_all,
                     (void **)&e->d_trace_tmp}) {
        if (*p) {
            HIP_CHECK(hipFree(*p));
            *p = nullptr;
        }
    }
    if (e->h_partials) {
        HIP_CHECK(hipHostFree(e->h_partials));
        e->h_partials = nullptr;
    }
    if (e->h_cntmat) {
        HIP_CHECK(hipHostFree(e->h_cntmat));
        e->h_cntmat = nullptr;
    }
    // a stale trace against a freed d_trace_tmp (or a new lnv) would fault:
    // re-arm via mv_engine_set_trace after every load
    e->trace_target = nullptr;
    e->trace_mod = nullptr;
    e->trace_cap = 0;
    e->sell_elems = 0;
    e->spill_elems = 0;
    e->rc_cap = 0;
    e->req_cap = 0;
    e->hash_total = 0;
    e->nghost = 0;
    e->ssz = 0;
    e->nhi = 0;
    e->nlh = 0;
    e->hash_hub_elems = 0;
    e->cub_tmp_bytes = 0;
}

int mv_engine_load_graph(mv_engine *e, const mv_graph *g) {
    HIP_CHECK(hipSetDevice(e->device));
    if (e->d_xadj) free_graph_state(e); // re-load on a live engine
    e->nv = mv_graph_nv(g);
    if (e->nv >= (1ll << 31)) { // u32 slot/view encodings + int32 SELL
                                // tails need 31-bit ids
        std::fprintf(stderr, "mv_engine_load_graph: nv >= 2^31 unsupported\n");
        return -1;
    }
    e->lnv = mv_graph_lnv(g);
    e->lne = mv_graph_lne(g);
    e->sort_bits = 1;
    while ((1ll << e->sort_bits) < e->nv) e->sort_bits++;
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
    const i64 *tails_h = mv_graph_tails(g);
    e->rows_sorted = 1;
    for (i64 i = 0; i < lnv && e->rows_sorted; i++)
        for (i64 k = xadj[i] + 1; k < xadj[i + 1]; k++)
            if (tails_h[k - 1] > tails_h[k]) { e->rows_sorted = 0; break; }

    HIP_CHECK(hipMalloc(&e->d_parts, 8 * (e->nranks + 1)));
    HIP_CHECK(hipMemcpy(e->d_parts, e->parts_h.data(), 8 * (e->nranks + 1),
                        hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&e->d_xadj, 8 * (lnv + 1)));
    HIP_CHECK(hipMemcpy(e->d_xadj, xadj, 8 * (lnv + 1), hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&e->d_tails, 8 * std::max<i64>(lne, 1)));
    HIP_CHECK(hipMemcpy(e->d_tails, mv_graph_tails(g), 8 * lne,
                        hipMemcpyHostToDevice));
    // unit graphs never read edge weights on device (K1 counts rows, the
    // SELL carries no weight stream): skip the 8*lne upload entirely
    if (e->unit_weights) {
        HIP_CHECK(hipMalloc(&e->d_ew, 16)); // never-null dummy
    } else {
        HIP_CHECK(hipMalloc(&e->d_ew, 8 * std::max<i64>(lne, 1)));
        HIP_CHECK(hipMemcpy(e->d_ew, w, 8 * lne, hipMemcpyHostToDevice));
    }

    // internal layout: the builder's spatial hint, or identity (the
    // degree-sort fallback then runs in the per-run setup)
    HIP_CHECK(hipMalloc(&e->d_sigma, 4 * std::max<i64>(lnv, 1)));
    HIP_CHECK(hipMalloc(&e->d_sigma_inv, 4 * std::max<i64>(lnv, 1)));
    const int32_t *hint = mv_graph_locality_hint(g);
    e->has_hint = hint != nullptr;
    if (hint) {
        HIP_CHECK(hipMemcpy(e->d_sigma, hint, 4 * lnv, hipMemcpyHostToDevice));
    } else {
        k_iota32<<<grid_for(lnv), 256>>>(lnv, e->d_sigma);
    }
    k_invert_perm<<<grid_for(lnv), 256>>>(lnv, e->d_sigma, e->d_sigma_inv);

    e->nchunks = (lnv + 63) / 64;
    HIP_CHECK(hipMalloc(&e->d_deg, 4 * std::max<i64>(lnv, 1)));
    HIP_CHECK(hipMalloc(&e->d_iota, 4 * std::max<i64>(lnv, 1)));
    HIP_CHECK(hipMalloc(&e->d_perm, 4 * std::max<i64>(lnv, 1)));
    HIP_CHECK(hipMalloc(&e->d_chunk_off, 8 * (e->nchunks + 1)));

    HIP_CHECK(hipMalloc(&e->d_curr, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_past, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_target, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_vdeg, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_cw, 8 * lnv));
    HIP_CHECK(hipMalloc(&e->d_cinfo, sizeof(Cinfo) * lnv));
    HIP_CHECK(hipMalloc(&e->d_cupd, sizeof(Cinfo) * lnv));
    // never-null dummies for pointers that stay unused at nranks==1 but
    // may still be address-computed by if-converted loads
    if (!e->d_vghost) HIP_CHECK(hipMalloc(&e->d_vghost, 16));
    if (!e->d_rc_ids) HIP_CHECK(hipMalloc(&e->d_rc_ids, 16));
    if (!e->d_rc_info) HIP_CHECK(hipMalloc(&e->d_rc_info, sizeof(Info16)));
    if (!e->d_rcu) HIP_CHECK(hipMalloc(&e->d_rcu, sizeof(Info16)));
    HIP_CHECK(hipMalloc(&e->d_count, 8));
    HIP_CHECK(hipMalloc(&e->d_bounds, 8 * (e->nranks + 1)));
    HIP_CHECK(hipMalloc(&e->d_red, 16));
    const int nblocks = grid_for(lnv);
    HIP_CHECK(hipMalloc(&e->d_partials, 16 * nblocks));
    HIP_CHECK(hipHostMalloc(&e->h_partials, 16 * nblocks));

    // K4 geometry: 256-thread blocks, 8 LDS slots/lane (32 KiB/block ->
    // 4-5 blocks/CU). Spill sizing: uniform per-thread max_degree extents
    // for flat distributions; for skewed graphs (max degree > 256) the
    // SELL order is forced degree-descending and extents follow the
    // per-thread first-vertex degree (k_spill_need), so the total stays
    // O(lne) instead of O(threads * max_degree).
    e->sweep_grid = grid_for(lnv, 256, 2048);
    e->skewed = e->max_degree > 256;
    const i64 nthreads = (i64)e->sweep_grid * 256;
    HIP_CHECK(hipMalloc(&e->d_spill_off, 8 * nthreads));

    if (e->nranks == 1) build_sell(e);
    HIP_CHECK(hipDeviceSynchronize());
    e->stats = mv_stats{};
    e->stats.edges_local = lne;
    return 0;
}

void mv_engine_set_trace(mv_engine *e, int64_t *target_trace, double *mod_trace,
                         int cap) {
    e->trace_target = target_trace;
    e->trace_mod = mod_trace;
    e->trace_cap = cap;
    if (target_trace && !e->d_trace_tmp)
        HIP_CHECK(hipMalloc(&e->d_trace_tmp, 8 * std::max<i64>(e->lnv, 1)));
}

void mv_engine_get_stats(const mv_engine *e, mv_stats *out) { *out = e->stats; }

// helper: alltoallv; offsets in elements. RCCL grouped send/recv on the
// product path, device-to-device copies on the loopback harness.
static void rccl_alltoallv(mv_engine *e, const void *send, const i64 *soff,
                           void *recv, const i64 *roff, size_t elem_bytes,
                           ncclDataType_t ty, size_t ty_bytes,
                           ncclComm_t comm, hipStream_t stream) {
    if (e->lb) {
        mv_lb_session *s = e->lb;
        // own kernels feeding `send` must be complete before peers copy
        HIP_CHECK(hipStreamSynchronize(stream));
        s->posts[e->rank].send = send;
        s->posts[e->rank].soff = soff;
        s->posts[e->rank].eb = elem_bytes;
        s->barrier();
        for (int r = 0; r < e->nranks; r++) {
            if (r == e->rank) continue;
            const auto &ps = s->posts[r];
            const i64 cnt = ps.soff[e->rank + 1] - ps.soff[e->rank];
            if (cnt != roff[r + 1] - roff[r]) {
                std::fprintf(stderr,
                             "loopback alltoallv count mismatch: me=%d r=%d "
                             "sender says %lld, I expect %lld (eb=%zu)\n",
                             e->rank, r, (long long)cnt,
                             (long long)(roff[r + 1] - roff[r]), elem_bytes);
                std::abort();
            }
            if (cnt > 0)
                HIP_CHECK(hipMemcpyAsync(
                    (char *)recv + roff[r] * elem_bytes,
                    (const char *)ps.send + ps.soff[e->rank] * elem_bytes,
                    cnt * elem_bytes, hipMemcpyDeviceToDevice, stream));
        }
        HIP_CHECK(hipStreamSynchronize(stream));
        s->barrier(); // senders may reuse their buffers after this
        return;
    }
    NCCL_CHECK(ncclGroupStart());
    for (int r = 0; r < e->nranks; r++) {
        if (r == e->rank) continue;
        const i64 scnt = soff[r + 1] - soff[r];
        const i64 rcnt = roff[r + 1] - roff[r];
        if (scnt > 0)
            NCCL_CHECK(ncclSend((const char *)send + soff[r] * elem_bytes,
                                scnt * elem_bytes / ty_bytes, ty, r, comm,
                                stream));
        if (rcnt > 0)
            NCCL_CHECK(ncclRecv((char *)recv + roff[r] * elem_bytes,
                                rcnt * elem_bytes / ty_bytes, ty, r, comm,
                                stream));
    }
    NCCL_CHECK(ncclGroupEnd());
}

// exchange per-peer counts: allgather of my nranks counts
static void exchange_counts(mv_engine *e, const std::vector<i64> &mine,
                            std::vector<i64> &matrix /*nranks*nranks*/,
                            ncclComm_t comm, hipStream_t stream) {
    matrix.resize((size_t)e->nranks * e->nranks);
    if (e->lb) {
        mv_lb_session *s = e->lb;
        s->posts[e->rank].cnts = mine;
        s->barrier();
        for (int r = 0; r < e->nranks; r++)
            std::copy(s->posts[r].cnts.begin(), s->posts[r].cnts.end(),
                      matrix.begin() + (size_t)r * e->nranks);
        s->barrier();
        return;
    }
    // persistent buffer: a hipMalloc/hipFree here would device-sync and
    // re-serialize the overlapped pipeline
    if (!e->d_cnt_all)
        HIP_CHECK(hipMalloc(&e->d_cnt_all, 8 * e->nranks * e->nranks));
    i64 *d_all = e->d_cnt_all;
    HIP_CHECK(hipMemcpyAsync(d_all + (i64)e->rank * e->nranks, mine.data(),
                             8 * e->nranks, hipMemcpyHostToDevice, stream));
    NCCL_CHECK(ncclAllGather(d_all + (i64)e->rank * e->nranks, d_all,
                             e->nranks, ncclInt64, comm, stream));
    HIP_CHECK(hipMemcpyAsync(matrix.data(), d_all, 8 * e->nranks * e->nranks,
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
}

// allreduce-sum of n (<= 2) host doubles in place (dspl.hpp:126, :441)
static void comm_allreduce_host(mv_engine *e, double *vals, int n) {
    if (e->lb) {
        mv_lb_session *s = e->lb;
        s->posts[e->rank].red = vals;
        s->barrier();
        double acc[2] = {0.0, 0.0};
        for (int r = 0; r < e->nranks; r++) // rank order: deterministic
            for (int k = 0; k < n; k++) acc[k] += s->posts[r].red[k];
        s->barrier(); // all ranks have read every post
        for (int k = 0; k < n; k++) vals[k] = acc[k];
        return;
    }
    HIP_CHECK(hipMemcpyAsync(e->d_red, vals, 8 * n, hipMemcpyHostToDevice,
                             e->stream));
    NCCL_CHECK(ncclAllReduce(e->d_red, e->d_red, n, ncclDouble, ncclSum,
                             e->comm, e->stream));
    HIP_CHECK(hipMemcpyAsync(vals, e->d_red, 8 * n, hipMemcpyDeviceToHost,
                             e->stream));
    HIP_CHECK(hipStreamSynchronize(e->stream));
}

// Build the SELL image, internal order, spill extents and the high-degree
// hash regions. Pure graph-layout preparation (a function of the static
// graph + the ghost list), analogous to the reference's CSR existing
// before its timed span; called at load for single-rank graphs and right
// after ghost discovery inside the span for multi-rank ones (the ghost
// slots feed the tail translation).
static void build_sell(mv_engine *e) {
    hipStream_t st = e->stream;
    const i64 lnv = e->lnv;
    // build the SELL image over the internal order (tails translated
    // inline). perm = identity with a spatial hint, degree-sorted
    // otherwise (load balance for skewed degree distributions).
    k_degrees<<<grid_for(lnv), 256, 0, st>>>(lnv, e->d_sigma, e->d_xadj,
                                             e->d_deg);
    e->nhi = 0; // set by the skewed branch below when applicable
    e->nlh = 0;
    if (e->has_hint && !e->skewed) {
        k_iota32<<<grid_for(lnv), 256, 0, st>>>(lnv, e->d_perm);
    } else {
        k_iota32<<<grid_for(lnv), 256, 0, st>>>(lnv, e->d_iota);
        unsigned *d_degs = nullptr;
        HIP_CHECK(hipMalloc(&d_degs, 4 * std::max<i64>(lnv, 1)));
        size_t tb = 0;
        (void)hipcub::DeviceRadixSort::SortPairsDescending(
            nullptr, tb, e->d_deg, d_degs, e->d_iota, e->d_perm, lnv, 0,
            32, st);
        void *d_tmp = nullptr;
        HIP_CHECK(hipMalloc(&d_tmp, std::max<size_t>(tb, 1)));
        (void)hipcub::DeviceRadixSort::SortPairsDescending(
            d_tmp, tb, e->d_deg, d_degs, e->d_iota, e->d_perm, lnv, 0, 32,
            st);
        // high-degree split + per-vertex hash regions (unit: wave-per-
        // vertex atomic hash; -w: per-lane serial hash in edge order)
        std::vector<unsigned> sdeg(lnv);
        HIP_CHECK(hipMemcpyAsync(sdeg.data(), d_degs, 4 * lnv,
                                 hipMemcpyDeviceToHost, st));
        HIP_CHECK(hipStreamSynchronize(st));
        // MV_HI_THRESH (perf only): degree bound above which a vertex
        // takes the hub path (wave hash for unit, per-lane hash for -w)
        // instead of the LDS-slot + linear-spill lane path
        static const unsigned HI_THRESH = [] {
            const char *s = getenv("MV_HI_THRESH");
            return s ? (unsigned)atoi(s) : 256u;
        }();
        // MV_LH_THRESH: degree bound above which a vertex takes the
        // per-lane hash band (below: LDS slots + linear spill). Default ==
        // HI_THRESH: the band only covers WEIGHTED hubs (where it replaces
        // the old cap-scanning kernel); extending it to the unit 24-256
        // band was measured SLOWER than 24 LDS slots + linear spill on the
        // dense social shape (11.3 -> 8.3 G, experiments/RESULTS.md) — L2
        // hash probes lose to LDS at ~25 candidates.
        static const unsigned LH_THRESH = [] {
            const char *s = getenv("MV_LH_THRESH");
            return s ? (unsigned)atoi(s) : 256u;
        }();
        i64 nhi = 0;
        while (nhi < lnv && sdeg[nhi] > HI_THRESH) nhi++;
        i64 nlh = 0;
        while (nlh < lnv && sdeg[nlh] > LH_THRESH) nlh++;
        e->nhi = (e->skewed && e->unit_weights) ? nhi : 0;
        e->nlh = e->skewed ? std::max<i64>(nlh, e->nhi) : 0;
        if (e->nlh > 0) {
            if (!e->d_tidx32)
                HIP_CHECK(hipMalloc(&e->d_tidx32,
                                    4 * std::max<i64>(e->lne, 1)));
            k_translate_tails<<<grid_for(e->lne), 256, 0, st>>>(
                e->lne, e->d_tails, e->base, e->bound, e->d_sigma_inv,
                e->d_ghosts, e->nghost, lnv, e->d_tidx32);
            std::vector<i64> hoff(e->nlh + 1);
            i64 acc = 0;
            for (i64 t = 0; t < e->nlh; t++) {
                hoff[t] = acc;
                i64 cap = 64;
                while (cap < 2 * (i64)sdeg[t]) cap <<= 1;
                acc += cap;
            }
            hoff[e->nlh] = acc;
            e->hash_hub_elems = e->nhi > 0 ? hoff[e->nhi] : 0;
            if (e->d_hash_off) HIP_CHECK(hipFree(e->d_hash_off));
            HIP_CHECK(hipMalloc(&e->d_hash_off, 8 * (e->nlh + 1)));
            HIP_CHECK(hipMemcpyAsync(e->d_hash_off, hoff.data(),
                                     8 * (e->nlh + 1),
                                     hipMemcpyHostToDevice, st));
            if (acc > e->hash_total) {
                if (e->d_hkeys) HIP_CHECK(hipFree(e->d_hkeys));
                if (e->d_hacc) HIP_CHECK(hipFree(e->d_hacc));
                if (e->d_lh_list) HIP_CHECK(hipFree(e->d_lh_list));
                e->hash_total = acc;
                HIP_CHECK(hipMalloc(&e->d_hkeys, 8 * acc));
                HIP_CHECK(hipMalloc(&e->d_hacc, 8 * acc));
                HIP_CHECK(hipMalloc(&e->d_lh_list, 4 * (acc / 2 + 1)));
            }
            HIP_CHECK(hipStreamSynchronize(st));
        }
        HIP_CHECK(hipFree(d_tmp));
        HIP_CHECK(hipFree(d_degs));
    }

    // Exports-first SELL order (halo overlap): positions [0, nexp) hold
    // the vertices some peer wants (the svdata set), so the sweep can run
    // them as a first launch and halo #1a for the NEXT iteration — which
    // only needs their targets — can overlap the interior sweep + the
    // whole epilogue on stream2/comm2 (north_star: "#1a overlapped with
    // the local sweep on a second HIP stream"). Stable partition: locality
    // inside each group is preserved. Pure layout, results identical.
    e->nexp = 0;
    e->overlap = 0;
    if (e->nranks > 1 && !e->skewed && e->ssz > 0 && (e->comm2 || e->lb) &&
        !getenv("MV_NO_OVERLAP")) {
        // device-side stable partition (two hipCUB Flagged selects): the
        // host round trip cost ~ms per run at 8-GPU sizes
        unsigned char *d_mark = nullptr, *d_flags = nullptr;
        unsigned *d_out = nullptr;
        i64 *d_num = nullptr;
        HIP_CHECK(hipMalloc(&d_mark, lnv));
        HIP_CHECK(hipMalloc(&d_flags, lnv));
        HIP_CHECK(hipMalloc(&d_out, 4 * lnv));
        HIP_CHECK(hipMalloc(&d_num, 8));
        HIP_CHECK(hipMemsetAsync(d_mark, 0, lnv, st));
        k_scatter_mark<<<grid_for(e->ssz), 256, 0, st>>>(
            e->ssz, e->d_svdata_int, d_mark);
        k_gather_flags<<<grid_for(lnv), 256, 0, st>>>(lnv, e->d_perm, d_mark,
                                                      d_flags, 0);
        size_t tsel = 0;
        (void)hipcub::DeviceSelect::Flagged(nullptr, tsel, e->d_perm,
                                            d_flags, d_out, d_num, lnv, st);
        void *d_tsel = nullptr;
        HIP_CHECK(hipMalloc(&d_tsel, std::max<size_t>(tsel, 1)));
        (void)hipcub::DeviceSelect::Flagged(d_tsel, tsel, e->d_perm, d_flags,
                                            d_out, d_num, lnv, st);
        HIP_CHECK(hipMemcpyAsync(&e->nexp, d_num, 8, hipMemcpyDeviceToHost,
                                 st));
        HIP_CHECK(hipStreamSynchronize(st));
        k_gather_flags<<<grid_for(lnv), 256, 0, st>>>(lnv, e->d_perm, d_mark,
                                                      d_flags, 1);
        (void)hipcub::DeviceSelect::Flagged(d_tsel, tsel, e->d_perm, d_flags,
                                            d_out + e->nexp, d_num, lnv, st);
        HIP_CHECK(hipMemcpyAsync(e->d_perm, d_out, 4 * lnv,
                                 hipMemcpyDeviceToDevice, st));
        HIP_CHECK(hipStreamSynchronize(st));
        HIP_CHECK(hipFree(d_mark));
        HIP_CHECK(hipFree(d_flags));
        HIP_CHECK(hipFree(d_out));
        HIP_CHECK(hipFree(d_num));
        HIP_CHECK(hipFree(d_tsel));
        e->overlap = 1;
    }

    {
        i64 *d_sizes = nullptr;
        HIP_CHECK(hipMalloc(&d_sizes, 8 * std::max<i64>(e->nchunks, 1)));
        k_chunk_sizes<<<grid_for(e->nchunks), 256, 0, st>>>(
            e->nchunks, lnv, e->d_perm, e->d_deg, d_sizes);
        HIP_CHECK(hipMemsetAsync(e->d_chunk_off, 0, 8, st));
        size_t tb2 = 0;
        (void)hipcub::DeviceScan::InclusiveSum(nullptr, tb2, d_sizes,
                                         e->d_chunk_off + 1, e->nchunks,
                                         st);
        void *d_tmp2 = nullptr;
        HIP_CHECK(hipMalloc(&d_tmp2, std::max<size_t>(tb2, 1)));
        (void)hipcub::DeviceScan::InclusiveSum(d_tmp2, tb2, d_sizes,
                                         e->d_chunk_off + 1, e->nchunks,
                                         st);
        i64 total = 0;
        HIP_CHECK(hipMemcpyAsync(&total, e->d_chunk_off + e->nchunks, 8,
                                 hipMemcpyDeviceToHost, st));
        HIP_CHECK(hipStreamSynchronize(st));
        if (total > e->sell_elems) {
            if (e->d_sell_tidx) HIP_CHECK(hipFree(e->d_sell_tidx));
            if (e->d_sell_w) HIP_CHECK(hipFree(e->d_sell_w));
            e->sell_elems = total;
            HIP_CHECK(hipMalloc(&e->d_sell_tidx,
                                4 * std::max<i64>(total, 1)));
            if (!e->unit_weights)
                HIP_CHECK(hipMalloc(&e->d_sell_w,
                                    8 * std::max<i64>(total, 1)));
        }
        k_fill_sell<<<grid_for(lnv), 256, 0, st>>>(
            lnv, e->d_perm, e->d_sigma, e->d_sigma_inv, e->d_xadj,
            e->d_tails, e->d_ew, e->base, e->bound, e->d_ghosts, e->nghost,
            e->d_chunk_off, e->d_sell_tidx,
            e->unit_weights ? nullptr : e->d_sell_w);
        // unit graphs: re-sort rows by internal index for gather locality
        // (see k_sell_sort_rows; results identical, measured +4-5%)
        e->sell_sorted = 0;
        if (e->unit_weights && !e->skewed && !getenv("MV_NO_ROWSORT")) {
            k_sell_sort_rows<<<grid_for(lnv), 256, 0, st>>>(
                lnv, e->d_perm, e->d_deg, e->d_chunk_off, e->d_sell_tidx);
            e->sell_sorted = 1;
        }
        HIP_CHECK(hipFree(d_tmp2));
        HIP_CHECK(hipFree(d_sizes));
    }

    // per-thread spill extents
    {
        const i64 nthreads = (i64)e->sweep_grid * 256;
        i64 total;
        if (!e->skewed) {
            const i64 per = std::max<i64>(e->max_degree, 1);
            k_spill_uniform<<<grid_for(nthreads), 256, 0, st>>>(
                nthreads, per, e->d_spill_off);
            total = nthreads * per;
        } else {
            i64 *d_need = nullptr;
            HIP_CHECK(hipMalloc(&d_need, 8 * nthreads));
            k_spill_need<<<grid_for(nthreads), 256, 0, st>>>(
                nthreads, e->nlh, lnv, e->d_perm, e->d_deg, 4, d_need);
            std::vector<i64> need(nthreads), off(nthreads);
            HIP_CHECK(hipMemcpyAsync(need.data(), d_need, 8 * nthreads,
                                     hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));
            i64 acc = 0;
            for (i64 t = 0; t < nthreads; t++) {
                off[t] = acc;
                acc += need[t];
            }
            total = acc;
            HIP_CHECK(hipMemcpyAsync(e->d_spill_off, off.data(),
                                     8 * nthreads, hipMemcpyHostToDevice,
                                     st));
            HIP_CHECK(hipStreamSynchronize(st));
            HIP_CHECK(hipFree(d_need));
        }
        if (total > e->spill_elems) {
            if (e->d_spill_k) HIP_CHECK(hipFree(e->d_spill_k));
            if (e->d_spill_a) HIP_CHECK(hipFree(e->d_spill_a));
            e->spill_elems = total;
            HIP_CHECK(hipMalloc(&e->d_spill_k, 8 * total));
            HIP_CHECK(hipMalloc(&e->d_spill_a, 8 * total));
        }
    }

    HIP_CHECK(hipStreamSynchronize(st));
}

// Halo #1a (dspl.hpp:583-647) on (comm, stream): gather the exports'
// communities, exchange them (full resend like the reference, or
// delta-compacted), and scatter into the persistent label image
// d_ghost_labels (the per-iteration u32 view is built from it once the
// rc set is known). In overlap mode this runs on stream2/comm2
// concurrently with the interior sweep + epilogue.
static void run_halo1a(mv_engine *e, const i64 *commArr, ncclComm_t comm,
                       hipStream_t st2) {
    const int p = e->nranks;
    k8_gather_comms<<<grid_for(std::max<i64>(e->ssz, 1)), 256, 0, st2>>>(
        e->ssz, e->d_svdata_int, commArr, e->d_scdata);
    if (!e->use_delta) {
        rccl_alltoallv(e, e->d_scdata, e->send_off.data(), e->d_ghost_labels,
                       e->recv_off.data(), 8, ncclInt64, 8, comm, st2);
    } else {
        HIP_CHECK(hipMemsetAsync(e->d_chg_cnt, 0, 8 * p, st2));
        k_delta_compact<<<grid_for(std::max<i64>(e->ssz, 1)), 256, 0, st2>>>(
            e->ssz, e->d_soff, p, e->d_scdata, e->d_last_sent, e->d_chg_idx,
            e->d_chg_lab, e->d_chg_cnt);
        // tiny count matrix; rides the same stream/comm so it overlaps the
        // interior sweep in overlap mode (the host blocks only on these
        // small ops — the interior sweep is already issued)
        if (e->lb) {
            std::vector<i64> mine(p), m;
            HIP_CHECK(hipMemcpyAsync(mine.data(), (i64 *)e->d_chg_cnt, 8 * p,
                                     hipMemcpyDeviceToHost, st2));
            HIP_CHECK(hipStreamSynchronize(st2));
            exchange_counts(e, mine, m, comm, st2);
            std::copy(m.begin(), m.end(), e->h_cntmat);
        } else {
            HIP_CHECK(hipMemcpyAsync(e->d_cntmat + (i64)e->rank * p,
                                     e->d_chg_cnt, 8 * p,
                                     hipMemcpyDeviceToDevice, st2));
            NCCL_CHECK(ncclAllGather(e->d_cntmat + (i64)e->rank * p,
                                     e->d_cntmat, p, ncclInt64, comm, st2));
            HIP_CHECK(hipMemcpyAsync(e->h_cntmat, e->d_cntmat, 8 * p * p,
                                     hipMemcpyDeviceToHost, st2));
            HIP_CHECK(hipStreamSynchronize(st2));
        }
        // mixed full/compact payload: both ends of every pair decide from
        // the same matrix entry, so the plan is symmetric by construction
        auto full_mode = [](i64 changed, i64 seg) {
            return changed * 12 >= seg * 8;
        };
        if (e->lb) {
            mv_lb_session *s = e->lb;
            HIP_CHECK(hipStreamSynchronize(st2));
            s->posts[e->rank].send = e->d_scdata;
            s->posts[e->rank].send2 = e->d_chg_idx;
            s->posts[e->rank].send3 = e->d_chg_lab;
            s->posts[e->rank].soff = e->send_off.data();
            s->barrier();
            for (int r = 0; r < p; r++) {
                if (r == e->rank) continue;
                const auto &ps = s->posts[r];
                const i64 seg = e->recv_off[r + 1] - e->recv_off[r];
                const i64 chg = e->h_cntmat[(i64)r * p + e->rank];
                const i64 sb = ps.soff[e->rank];
                if (full_mode(chg, seg)) {
                    if (seg > 0)
                        HIP_CHECK(hipMemcpyAsync(
                            e->d_ghost_labels + e->recv_off[r],
                            (const i64 *)ps.send + sb, 8 * seg,
                            hipMemcpyDeviceToDevice, st2));
                } else if (chg > 0) {
                    HIP_CHECK(hipMemcpyAsync(
                        e->d_rchg_idx + e->recv_off[r],
                        (const unsigned *)ps.send2 + sb, 4 * chg,
                        hipMemcpyDeviceToDevice, st2));
                    HIP_CHECK(hipMemcpyAsync(
                        e->d_rchg_lab + e->recv_off[r],
                        (const i64 *)ps.send3 + sb, 8 * chg,
                        hipMemcpyDeviceToDevice, st2));
                }
            }
            HIP_CHECK(hipStreamSynchronize(st2));
            s->barrier();
        } else {
            NCCL_CHECK(ncclGroupStart());
            for (int r = 0; r < p; r++) {
                if (r == e->rank) continue;
                const i64 sseg = e->send_off[r + 1] - e->send_off[r];
                const i64 schg = e->h_cntmat[(i64)e->rank * p + r];
                if (full_mode(schg, sseg)) {
                    if (sseg > 0)
                        NCCL_CHECK(ncclSend(e->d_scdata + e->send_off[r],
                                            sseg, ncclInt64, r, comm, st2));
                } else if (schg > 0) {
                    NCCL_CHECK(ncclSend(e->d_chg_idx + e->send_off[r],
                                        schg * 4, ncclChar, r, comm, st2));
                    NCCL_CHECK(ncclSend(e->d_chg_lab + e->send_off[r],
                                        schg, ncclInt64, r, comm, st2));
                }
                const i64 rseg = e->recv_off[r + 1] - e->recv_off[r];
                const i64 rchg = e->h_cntmat[(i64)r * p + e->rank];
                if (full_mode(rchg, rseg)) {
                    if (rseg > 0)
                        NCCL_CHECK(
                            ncclRecv(e->d_ghost_labels + e->recv_off[r],
                                     rseg, ncclInt64, r, comm, st2));
                } else if (rchg > 0) {
                    NCCL_CHECK(ncclRecv(e->d_rchg_idx + e->recv_off[r],
                                        rchg * 4, ncclChar, r, comm, st2));
                    NCCL_CHECK(ncclRecv(e->d_rchg_lab + e->recv_off[r],
                                        rchg, ncclInt64, r, comm, st2));
                }
            }
            NCCL_CHECK(ncclGroupEnd());
        }
        for (int r = 0; r < p; r++) { // scatter compact segments
            if (r == e->rank) continue;
            const i64 rseg = e->recv_off[r + 1] - e->recv_off[r];
            const i64 rchg = e->h_cntmat[(i64)r * p + e->rank];
            if (!full_mode(rchg, rseg) && rchg > 0)
                k_scatter_deltas<<<grid_for(rchg), 256, 0, st2>>>(
                    rchg, e->d_rchg_idx + e->recv_off[r],
                    e->d_rchg_lab + e->recv_off[r], e->recv_off[r],
                    e->d_ghost_labels);
        }
    }
}

// Candidate discovery + halo #1b/#1c/#1d + u32 view build for ONE
// iteration, on (comm, st2): collect the remote communities referenced by
// `curr` / the ghost labels, fetch their (size, degree) records from the
// owners (dspl.hpp:670-929), zero rcu, and build the sweep views. In
// overlap mode this runs on stream2/comm2 during the PREVIOUS iteration's
// epilogue; `ev_cinfo` (when given) gates the reply section on that
// iteration's cinfo updates (K6 + per-sender delta application) — the
// replies must see exactly the reference's post-update localCinfo.
static void halo_prep(mv_engine *e, const i64 *curr,
                      std::vector<i64> &rc_bounds, std::vector<i64> &req_off,
                      i64 &nrc_out, ncclComm_t comm, hipStream_t st2,
                      hipEvent_t ev_cinfo) {
    const int p = e->nranks, me = e->rank;
    const i64 lnv = e->lnv;
    // ---- needed remote communities (dspl.hpp:670-700) ----
    const i64 cand_max = e->nghost + lnv;
    if (cand_max > e->rc_cap) {
        for (void *q : {(void *)e->d_cand, (void *)e->d_cand_sorted,
                        (void *)e->d_rc_ids, (void *)e->d_rc_info,
                        (void *)e->d_rcu})
            if (q) HIP_CHECK(hipFree(q));
        e->rc_cap = cand_max;
        HIP_CHECK(hipMalloc(&e->d_cand, 8 * cand_max));
        HIP_CHECK(hipMalloc(&e->d_cand_sorted, 8 * cand_max));
        HIP_CHECK(hipMalloc(&e->d_rc_ids, 8 * cand_max));
        HIP_CHECK(hipMalloc(&e->d_rc_info, sizeof(Info16) * cand_max));
        HIP_CHECK(hipMalloc(&e->d_rcu, sizeof(Info16) * cand_max));
        size_t t1 = 0, t2 = 0;
        (void)hipcub::DeviceRadixSort::SortKeys(nullptr, t1, e->d_cand,
                                                e->d_cand_sorted, cand_max,
                                                0, e->sort_bits, st2);
        i64 *dummy = nullptr;
        (void)hipcub::DeviceSelect::Unique(nullptr, t2, e->d_cand_sorted,
                                           e->d_rc_ids, dummy, cand_max,
                                           st2);
        size_t need = std::max(t1, t2);
        if (need > e->cub_tmp_bytes) {
            if (e->d_cub_tmp) HIP_CHECK(hipFree(e->d_cub_tmp));
            HIP_CHECK(hipMalloc(&e->d_cub_tmp, need));
            e->cub_tmp_bytes = need;
        }
    }
    HIP_CHECK(hipMemsetAsync(e->d_count, 0, 8, st2));
    k_filter_remote<<<grid_for(std::max<i64>(e->nghost, 1)), 256, 0, st2>>>(
        e->nghost, e->d_ghost_labels, /*shift*/ 0, e->base, e->bound,
        e->d_cand, e->d_count);
    k_filter_remote<<<grid_for(lnv), 256, 0, st2>>>(
        lnv, curr, /*shift*/ 0, e->base, e->bound, e->d_cand, e->d_count);
    unsigned long long ncand = 0;
    HIP_CHECK(hipMemcpyAsync(&ncand, e->d_count, 8, hipMemcpyDeviceToHost,
                             st2));
    HIP_CHECK(hipStreamSynchronize(st2));
    size_t tb = e->cub_tmp_bytes;
    (void)hipcub::DeviceRadixSort::SortKeys(e->d_cub_tmp, tb, e->d_cand,
                                            e->d_cand_sorted, (int64_t)ncand,
                                            0, e->sort_bits, st2);
    i64 *d_nrc = (i64 *)e->d_count; // reuse as output slot
    tb = e->cub_tmp_bytes;
    (void)hipcub::DeviceSelect::Unique(e->d_cub_tmp, tb, e->d_cand_sorted,
                                       e->d_rc_ids, d_nrc, (int64_t)ncand,
                                       st2);
    i64 nrc = 0;
    HIP_CHECK(hipMemcpyAsync(&nrc, d_nrc, 8, hipMemcpyDeviceToHost, st2));
    HIP_CHECK(hipStreamSynchronize(st2));

    // ---- halo #1b/#1c/#1d: request (size,degree) of those communities
    // from their owners (dspl.hpp:719-929) ----
    k_owner_bounds<<<1, p + 1, 0, st2>>>(e->d_rc_ids, nrc, e->d_parts, p,
                                         e->d_bounds);
    rc_bounds.assign(p + 1, 0);
    HIP_CHECK(hipMemcpyAsync(rc_bounds.data(), e->d_bounds, 8 * (p + 1),
                             hipMemcpyDeviceToHost, st2));
    HIP_CHECK(hipStreamSynchronize(st2));
    std::vector<i64> reqs(p), matrix;
    for (int r = 0; r < p; r++) reqs[r] = rc_bounds[r + 1] - rc_bounds[r];
    exchange_counts(e, reqs, matrix, comm, st2);
    req_off.assign(p + 1, 0);
    for (int r = 0; r < p; r++)
        req_off[r + 1] =
            req_off[r] + ((r == me) ? 0 : matrix[(size_t)r * p + me]);
    const i64 nreq = req_off[p];
    if (nreq > e->req_cap) { // rare growth; device-wide hipFree sync is ok
        if (e->d_req_ids) HIP_CHECK(hipFree(e->d_req_ids));
        if (e->d_req_info) HIP_CHECK(hipFree(e->d_req_info));
        e->req_cap = std::max<i64>(nreq, 64);
        HIP_CHECK(hipMalloc(&e->d_req_ids, 8 * e->req_cap));
        HIP_CHECK(hipMalloc(&e->d_req_info, sizeof(Info16) * e->req_cap));
    }
    // the previous iteration's delta routing reads d_req_ids/d_req_info
    // and writes cinfo; #1c's recv and the replies must order after it
    if (ev_cinfo) HIP_CHECK(hipStreamWaitEvent(st2, ev_cinfo, 0));
    rccl_alltoallv(e, e->d_rc_ids, rc_bounds.data(), e->d_req_ids,
                   req_off.data(), 8, ncclInt64, 8, comm, st2);
    k9_reply_info<<<grid_for(std::max<i64>(nreq, 1)), 256, 0, st2>>>(
        nreq, e->d_req_ids, e->base, e->d_sigma_inv, e->d_cinfo,
        e->d_req_info);
    rccl_alltoallv(e, e->d_req_info, req_off.data(), e->d_rc_info,
                   rc_bounds.data(), sizeof(Info16), ncclChar, 1, comm, st2);
    HIP_CHECK(hipMemsetAsync(e->d_rcu, 0,
                             sizeof(Info16) * std::max<i64>(nrc, 1), st2));
    // u32 views for the sweep (slot / lnv + rc index encoding)
    k_build_view<<<grid_for(lnv), 256, 0, st2>>>(
        lnv, curr, e->base, e->bound, lnv, e->d_sigma_inv, e->d_rc_ids, nrc,
        e->d_vcurr);
    k_build_view<<<grid_for(std::max<i64>(e->nghost, 1)), 256, 0, st2>>>(
        e->nghost, e->d_ghost_labels, e->base, e->bound, lnv,
        e->d_sigma_inv, e->d_rc_ids, nrc, e->d_vghost);
    nrc_out = nrc;
}

#define PHASE(tag)                                                            \
    do {                                                                      \
        if (getenv("MV_PHASE_DEBUG")) {                                       \
            HIP_CHECK(hipStreamSynchronize(e->stream));                       \
            std::fprintf(stderr, "[phase] %s\n", tag);                        \
            std::fflush(stderr);                                              \
        }                                                                     \
    } while (0)

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
            (void)hipcub::DeviceRadixSort::SortKeys(nullptr, tmp1, d_rem, d_sorted,
                                              (int64_t)nrem, 0, e->sort_bits,
                                              st);
            i64 *d_ng = nullptr;
            HIP_CHECK(hipMalloc(&d_ng, 8));
            (void)hipcub::DeviceSelect::Unique(nullptr, tmp2, d_sorted, e->d_ghosts,
                                         d_ng, (int64_t)nrem, st);
            size_t tmpb = std::max(tmp1, tmp2);
            void *d_tmp = nullptr;
            HIP_CHECK(hipMalloc(&d_tmp, std::max<size_t>(tmpb, 1)));
            (void)hipcub::DeviceRadixSort::SortKeys(d_tmp, tmp1, d_rem, d_sorted,
                                              (int64_t)nrem, 0, e->sort_bits,
                                              st);
            (void)hipcub::DeviceSelect::Unique(d_tmp, tmp2, d_sorted, e->d_ghosts,
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
            exchange_counts(e, want, matrix, e->comm, e->stream);
            e->send_off.assign(p + 1, 0);
            for (int r = 0; r < p; r++)
                e->send_off[r + 1] =
                    e->send_off[r] + ((r == me) ? 0 : matrix[(size_t)r * p + me]);
            e->ssz = e->send_off[p];
            if (e->d_svdata) HIP_CHECK(hipFree(e->d_svdata));
            if (e->d_svdata_int) HIP_CHECK(hipFree(e->d_svdata_int));
            HIP_CHECK(hipMalloc(&e->d_svdata, 8 * std::max<i64>(e->ssz, 1)));
            HIP_CHECK(hipMalloc(&e->d_svdata_int,
                                4 * std::max<i64>(e->ssz, 1)));
            // role swap (dspl.hpp:1255-1257): my ghost list goes OUT, the
            // peers' lists land in svdata
            rccl_alltoallv(e, e->d_ghosts, e->recv_off.data(), e->d_svdata,
                           e->send_off.data(), 8, ncclInt64, 8, e->comm,
                           e->stream);
            k_to_internal<<<grid_for(std::max<i64>(e->ssz, 1)), 256, 0, st>>>(
                e->ssz, e->d_svdata, e->base, e->d_sigma_inv, e->d_svdata_int);
            HIP_CHECK(hipStreamSynchronize(st));

            if (e->d_vghost) HIP_CHECK(hipFree(e->d_vghost));
            HIP_CHECK(hipMalloc(&e->d_vghost,
                                4 * std::max<i64>(e->nghost, 1)));
            if (!e->d_vcurr) HIP_CHECK(hipMalloc(&e->d_vcurr, 4 * lnv));
            if (!e->d_vtarget) HIP_CHECK(hipMalloc(&e->d_vtarget, 4 * lnv));
            if (e->d_scdata) HIP_CHECK(hipFree(e->d_scdata));
            HIP_CHECK(hipMalloc(&e->d_scdata, 8 * std::max<i64>(e->ssz, 1)));

            // delta-compaction state (default on; MV_NO_DELTA reverts to
            // the reference's full resend for A/B)
            e->use_delta = !getenv("MV_NO_DELTA");
            const i64 sszc = std::max<i64>(e->ssz, 1);
            const i64 ngc = std::max<i64>(e->nghost, 1);
            for (void **q :
                 {(void **)&e->d_last_sent, (void **)&e->d_chg_idx,
                  (void **)&e->d_chg_lab, (void **)&e->d_chg_cnt,
                  (void **)&e->d_cntmat, (void **)&e->d_soff,
                  (void **)&e->d_ghost_labels, (void **)&e->d_rchg_idx,
                  (void **)&e->d_rchg_lab})
                if (*q) {
                    HIP_CHECK(hipFree(*q));
                    *q = nullptr;
                }
            HIP_CHECK(hipMalloc(&e->d_last_sent, 8 * sszc));
            HIP_CHECK(hipMemsetAsync(e->d_last_sent, 0xFF, 8 * sszc, st));
            HIP_CHECK(hipMalloc(&e->d_chg_idx, 4 * sszc));
            HIP_CHECK(hipMalloc(&e->d_chg_lab, 8 * sszc));
            HIP_CHECK(hipMalloc(&e->d_chg_cnt, 8 * p));
            HIP_CHECK(hipMalloc(&e->d_cntmat, 8 * p * p));
            HIP_CHECK(hipMalloc(&e->d_soff, 8 * (p + 1)));
            HIP_CHECK(hipMemcpyAsync(e->d_soff, e->send_off.data(),
                                     8 * (p + 1), hipMemcpyHostToDevice, st));
            HIP_CHECK(hipMalloc(&e->d_ghost_labels, 8 * ngc));
            HIP_CHECK(hipMalloc(&e->d_rchg_idx, 4 * ngc));
            HIP_CHECK(hipMalloc(&e->d_rchg_lab, 8 * ngc));
            if (!e->h_cntmat)
                HIP_CHECK(hipHostMalloc(&e->h_cntmat, 8 * p * p));
            HIP_CHECK(hipStreamSynchronize(st));
        } else {
            e->nghost = 0;
            e->ssz = 0;
        }

        if (p > 1) {
            build_sell(e); // p==1: built at load
            // overlap changes the collective sequence, so it must be a
            // GLOBAL decision: all ranks take it or none (a rank can be
            // locally skewed / export-free while others are not)
            std::vector<i64> flag(p, e->overlap ? 1 : 0), fm;
            exchange_counts(e, flag, fm, e->comm, e->stream);
            for (int r = 0; r < p; r++)
                if (fm[(size_t)r * p] == 0) e->overlap = 0;
        }
    }

    PHASE("setup-done");
    // ---- distInitLouvain (dspl.hpp:151-172) ----
    k1_vertex_degree<<<grid_for(lnv), 256, 0, st>>>(
        lnv, e->d_sigma, e->d_xadj, e->d_ew, e->unit_weights, e->d_vdeg,
        e->d_cinfo);
    PHASE("k1-done");
    const int nblocks = grid_for(lnv);
    {
        auto f = [vd = e->d_vdeg] __device__(i64 i, double &a, double &b) {
            a = vd[i];
            b = 0.0;
        };
        k_partial_sum2<<<nblocks, 256, 0, st>>>(lnv, f, e->d_partials);
    }
    double *const partials = e->h_partials;
    HIP_CHECK(hipMemcpyAsync(partials, e->d_partials, 16 * nblocks,
                             hipMemcpyDeviceToHost, st));
    HIP_CHECK(hipStreamSynchronize(st));
    double localW = 0.0;
    for (int b = 0; b < nblocks; b++) localW += partials[2 * b];
    double totalW = localW;
    if (p > 1) comm_allreduce_host(e, &totalW, 1); // allreduce (dspl.hpp:126)
    const double constant = 1.0 / totalW; // dspl.hpp:129
    PHASE("k2-done");
    if (p == 1) // slot mode: the i64 arrays alias u32 slot arrays
        k3_init_comm32<<<grid_for(lnv), 256, 0, st>>>(
            lnv, (unsigned *)e->d_curr, (unsigned *)e->d_past);
    else
        k3_init_comm<<<grid_for(lnv), 256, 0, st>>>(lnv, e->base, e->d_sigma,
                                                    e->d_curr, e->d_past);
    HIP_CHECK(hipStreamSynchronize(st));
    PHASE("k3-done");
    e->stats.setup_ms =
        std::chrono::duration<double, std::milli>(
            std::chrono::steady_clock::now() - t_setup0)
            .count();

    i64 *d_curr = e->d_curr, *d_past = e->d_past, *d_target = e->d_target;
    double prevMod = lower, currMod = -1.0;
    int numIters = 0;
    // halo state for the CURRENT iteration, and the one the overlapped
    // pipeline prepares for the NEXT (swapped at each loop top)
    std::vector<i64> rc_bounds(p + 1, 0), req_off(p + 1, 0);
    std::vector<i64> rc_bounds_nx(p + 1, 0), req_off_nx(p + 1, 0);
    i64 nrc = 0, nrc_nx = 0;

    // Overlap mode: the ENTIRE next-iteration halo pipeline — #1a ghost
    // communities off the exports' targets, candidate discovery, counts,
    // #1c/#1d info fetch, rcu zero and the view builds — runs on
    // stream2/comm2 while stream A sweeps the interior and runs the
    // epilogue; only #2 + the modularity allreduce stay on the critical
    // path. Joined via ev_halo at the next iteration's top.
    ncclComm_t c2 = e->comm2 ? e->comm2 : e->comm;
    if (p > 1 && e->overlap) {
        HIP_CHECK(hipEventRecord(e->ev_p1, st));
        HIP_CHECK(hipStreamWaitEvent(e->stream2, e->ev_p1, 0));
        run_halo1a(e, e->d_curr, c2, e->stream2);
        halo_prep(e, e->d_curr, rc_bounds, req_off, nrc, c2, e->stream2,
                  nullptr); // initial cinfo (K1) is complete: A was synced
        HIP_CHECK(hipEventRecord(e->ev_halo, e->stream2));
    }

    std::vector<hipEvent_t> sweep_ev;
    for (;;) {
        numIters++;

        if (p > 1) {
            const auto t_h0 = std::chrono::steady_clock::now();
            if (e->overlap) {
                // pipeline prepared everything during the previous
                // iteration's epilogue (or pre-loop)
                HIP_CHECK(hipStreamWaitEvent(st, e->ev_halo, 0));
                if (numIters > 1) {
                    std::swap(rc_bounds, rc_bounds_nx);
                    std::swap(req_off, req_off_nx);
                    nrc = nrc_nx;
                }
            } else {
                run_halo1a(e, d_curr, e->comm, st);
                halo_prep(e, d_curr, rc_bounds, req_off, nrc, e->comm, st,
                          nullptr);
                HIP_CHECK(hipStreamSynchronize(st));
            }
            e->stats.halo_ms +=
                std::chrono::duration<double, std::milli>(
                    std::chrono::steady_clock::now() - t_h0)
                    .count();
        }

        // ---- K4 sweep (dspl.hpp:1371-1387; K5's zeroing is fused: cupd is
        // zeroed by K6 after each apply + once before the loop, and the
        // sweep overwrites clusterWeight instead of accumulating) ----
        if (numIters == 1)
            HIP_CHECK(hipMemsetAsync(e->d_cupd, 0, sizeof(Cinfo) * lnv, st));
        hipEvent_t ev0 = e->ev_pair(), ev1 = e->ev_pair();
        HIP_CHECK(hipEventRecord(ev0, st));
        // LDS slots per lane: early iterations see ~degree distinct
        // candidate communities (every vertex its own community), later
        // ones only a handful — a larger-slot / lower-occupancy variant for
        // the first iterations avoids the spill path where it matters.
        // Tunables (perf only, results identical): MV_SLOTS, MV_SLOTS_FIRST,
        // MV_FIRST_ITERS.
        static const int slots_env = [] {
            const char *s = getenv("MV_SLOTS");
            return s ? atoi(s) : 0;
        }();
        static const int slots_first_env = [] {
            const char *s = getenv("MV_SLOTS_FIRST");
            return s ? atoi(s) : -1;
        }();
        // Degree-adaptive defaults: RGG-class graphs (deg ~10) measured
        // best at 8 slots (12 first iterations at p==1); dense social
        // shapes (deg ~70, ~25 persistent distinct candidates from long-
        // range edges) at 24 (72 KiB/block, 2 blocks/CU) — +23% on the
        // orkut_like workload, while 32+ falls off the occupancy cliff
        // (experiments/RESULTS.md).
        const bool dense = lne > 16 * lnv;
        const int slots_rest = slots_env > 0 ? slots_env : (dense ? 24 : 8);
        const int slots_first =
            slots_first_env > 0
                ? slots_first_env
                : (dense ? 24 : ((p == 1) ? 12 : slots_rest));
        static const int first_iters = [] {
            const char *s = getenv("MV_FIRST_ITERS");
            return s ? atoi(s) : 2;
        }();
        const int slots = (numIters <= first_iters) ? slots_first : slots_rest;
        auto launch_sweep = [&](auto slots_tag, auto unit_tag, i64 s0,
                                i64 s1) {
            constexpr int S = decltype(slots_tag)::value;
            if (p == 1) { // u32 slots: half-size gathers, 12 B/lane LDS
                k4_sweep_p1<S, decltype(unit_tag)::value>
                    <<<e->sweep_grid, 256, S * 256 * 12, st>>>(
                        s0, lnv, e->d_perm, e->d_deg, e->d_chunk_off,
                        e->d_sell_tidx, e->d_sell_w, (const unsigned *)d_curr,
                        e->d_vdeg, e->d_sigma, e->d_cinfo, e->d_cupd,
                        constant, (unsigned *)d_target, e->d_cw,
                        (unsigned *)e->d_spill_k, e->d_spill_a,
                        e->d_spill_off);
                return;
            }
            k4_sweep_mr<S, decltype(unit_tag)::value>
                <<<grid_for(s1 - s0, 256, 2048), 256, S * 256 * 12, st>>>(
                    s0, s1, lnv, e->base, e->d_perm, e->d_deg,
                    e->d_chunk_off, e->d_sell_tidx, e->d_sell_w, e->d_vcurr,
                    e->d_vghost, e->d_vdeg, e->d_sigma, e->d_cinfo,
                    e->d_cupd, e->d_rc_ids, e->d_rc_info, e->d_rcu,
                    constant, e->d_vtarget, e->d_cw,
                    (unsigned *)e->d_spill_k, e->d_spill_a, e->d_spill_off);
        };
        auto dispatch_slots = [&](auto unit_tag, i64 s0, i64 s1) {
            switch (slots) {
            case 4: launch_sweep(std::integral_constant<int, 4>{}, unit_tag, s0, s1); break;
            case 12: launch_sweep(std::integral_constant<int, 12>{}, unit_tag, s0, s1); break;
            case 16: launch_sweep(std::integral_constant<int, 16>{}, unit_tag, s0, s1); break;
            case 24: launch_sweep(std::integral_constant<int, 24>{}, unit_tag, s0, s1); break;
            case 32: launch_sweep(std::integral_constant<int, 32>{}, unit_tag, s0, s1); break;
            case 48: launch_sweep(std::integral_constant<int, 48>{}, unit_tag, s0, s1); break;
            default: launch_sweep(std::integral_constant<int, 8>{}, unit_tag, s0, s1); break;
            }
        };
        auto launch_iter1 = [&](auto unit_tag, auto lblasc_tag, i64 s0,
                                i64 s1) {
            if (p == 1) {
                k4_sweep_iter1_p1<decltype(unit_tag)::value,
                                  decltype(lblasc_tag)::value>
                    <<<e->sweep_grid, 256, 0, st>>>(
                        s0, lnv, e->d_perm, e->d_deg, e->d_chunk_off,
                        e->d_sell_tidx, e->d_sell_w, e->d_vdeg, e->d_sigma,
                        e->d_cupd, constant, (unsigned *)d_target, e->d_cw);
                return;
            }
            k4_sweep_iter1_mr<decltype(unit_tag)::value,
                              decltype(lblasc_tag)::value>
                <<<grid_for(s1 - s0, 256, 2048), 256, 0, st>>>(
                    s0, s1, lnv, e->base, e->d_perm, e->d_deg,
                    e->d_chunk_off, e->d_sell_tidx, e->d_sell_w, e->d_vghost,
                    e->d_ghosts, e->d_vdeg, e->d_sigma, e->d_cupd,
                    e->d_rc_info, e->d_rcu, constant, e->d_vtarget, e->d_cw);
        };
        if (e->nlh > 0) { // hash-aggregation bands (skewed graphs)
            HIP_CHECK(hipMemsetAsync(e->d_hkeys, 0xFF, 8 * e->hash_total,
                                     st));
            if (e->hash_hub_elems) // the wave kernel atomic-adds into its
                                   // prefix; the lane band writes on insert
                HIP_CHECK(hipMemsetAsync(e->d_hacc, 0,
                                         8 * e->hash_hub_elems, st));
            if (e->nlh > e->nhi) { // per-lane hash band [nhi, nlh)
                auto launch_lh = [&](auto unit_tag) {
                    k4_sweep_lh<decltype(unit_tag)::value>
                        <<<grid_for(e->nlh - e->nhi), 256, 0, st>>>(
                            e->nhi, e->nlh, lnv, e->base, e->d_perm,
                            e->d_deg, e->d_chunk_off, e->d_sell_tidx,
                            e->d_sell_w,
                            p == 1 ? (const unsigned *)d_curr : e->d_vcurr,
                            e->d_vghost, e->d_vdeg, e->d_sigma, e->d_cinfo,
                            e->d_cupd, e->d_rc_ids, e->d_rc_info, e->d_rcu,
                            constant,
                            p == 1 ? (unsigned *)d_target : e->d_vtarget,
                            e->d_cw, e->d_hash_off, e->d_hkeys, e->d_hacc,
                            e->d_lh_list);
                };
                if (e->unit_weights)
                    launch_lh(std::integral_constant<bool, true>{});
                else
                    launch_lh(std::integral_constant<bool, false>{});
            }
            if (e->nhi > 0 && p == 1)
                k4_sweep_hi_p1<<<grid_for(e->nhi * 64, 256, 2048), 256, 0,
                                 st>>>(
                    e->nhi, lnv, e->d_perm, e->d_deg, e->d_sigma,
                    e->d_xadj, e->d_tidx32,
                    (const unsigned *)d_curr, e->d_vdeg, e->d_cinfo,
                    e->d_cupd, constant, (unsigned *)d_target, e->d_cw,
                    e->d_hash_off, e->d_hkeys, e->d_hacc);
            else if (e->nhi > 0)
            k4_sweep_hi_mr<<<grid_for(e->nhi * 64, 256, 2048), 256, 0,
                             st>>>(
                e->nhi, lnv, e->base, e->d_perm, e->d_deg,
                e->d_sigma, e->d_xadj, e->d_tidx32,
                e->d_vcurr, e->d_vghost, e->d_vdeg,
                e->d_cinfo, e->d_cupd, e->d_rc_ids, e->d_rc_info,
                e->d_rcu, constant, e->d_vtarget, e->d_cw, e->d_hash_off,
                e->d_hkeys, e->d_hacc);
        }
        static const bool no_iter1 = getenv("MV_NO_ITER1") != nullptr;
        auto sweep_range = [&](i64 s0, i64 s1) {
            if (s1 <= s0) return;
            if (numIters == 1 && (e->rows_sorted || e->sell_sorted) &&
                !no_iter1) {
                if (e->sell_sorted) // internal order: explicit label ties
                    launch_iter1(std::integral_constant<bool, true>{},
                                 std::integral_constant<bool, false>{}, s0,
                                 s1);
                else if (e->unit_weights)
                    launch_iter1(std::integral_constant<bool, true>{},
                                 std::integral_constant<bool, true>{}, s0,
                                 s1);
                else
                    launch_iter1(std::integral_constant<bool, false>{},
                                 std::integral_constant<bool, true>{}, s0,
                                 s1);
            } else if (e->unit_weights) {
                dispatch_slots(std::integral_constant<bool, true>{}, s0, s1);
            } else {
                dispatch_slots(std::integral_constant<bool, false>{}, s0, s1);
            }
        };
        // persist the sweep's u32 view targets as labels (p>1 only)
        auto writeback = [&](i64 s0, i64 s1) {
            if (p == 1 || s1 <= s0) return;
            k_view_to_labels<<<grid_for(s1 - s0, 256, 2048), 256, 0, st>>>(
                s0, s1, e->d_perm, e->d_vtarget, e->d_sigma, e->base,
                e->d_rc_ids, lnv, d_target);
        };
        if (p > 1 && e->overlap) {
            // exports first (their targets feed the next iteration's #1a
            // once ev_p1 fires); the full next-iteration pipeline is
            // issued after the epilogue's delta application (ev_cinfo) so
            // it runs under the epilogue + exit check — see below
            sweep_range(0, e->nexp);
            writeback(0, e->nexp);
            HIP_CHECK(hipEventRecord(e->ev_p1, st));
            sweep_range(e->nexp, lnv);
            writeback(e->nexp, lnv);
            HIP_CHECK(hipEventRecord(e->ev_p2, st));
        } else {
            sweep_range(e->nlh, lnv);
            writeback(0, lnv);
        }
        HIP_CHECK(hipEventRecord(ev1, st));
        PHASE("sweep-done");
        sweep_ev.push_back(ev0);
        sweep_ev.push_back(ev1);
        e->stats.sweep_launches++;

        // (phase markers around K6/K7 below)
        // ---- K6 (dspl.hpp:458-471); fused with K7 at p==1 ----
        if (p == 1) {
            k67_apply_and_partials<<<nblocks, 256, 0, st>>>(
                lnv, e->d_cupd, e->d_cinfo, e->d_cw, e->d_partials);
        } else {
            k6_apply_local<<<grid_for(lnv), 256, 0, st>>>(lnv, e->d_cupd,
                                                          e->d_cinfo);
        }

        // ---- halo #2: route deltas to owners (dspl.hpp:978-1103) ----
        if (p > 1) {
            const auto t_h0 = std::chrono::steady_clock::now();
            rccl_alltoallv(e, e->d_rcu, rc_bounds.data(), e->d_req_info,
                           req_off.data(), sizeof(Info16), ncclChar, 1,
                           e->comm, e->stream);
            // apply per SENDER segment, in rank order: same-stream launches
            // serialize, and one sender's ids are unique (sorted rc_ids), so
            // cinfo degree bits are run-to-run identical at any nranks —
            // like the reference's in-order delta walk (dspl.hpp:1089-1102)
            for (int r = 0; r < p; r++) {
                const i64 c = req_off[r + 1] - req_off[r];
                if (c > 0)
                    k_apply_deltas<<<grid_for(c), 256, 0, st>>>(
                        c, e->d_req_ids + req_off[r], e->base, e->d_sigma_inv,
                        e->d_req_info + req_off[r], e->d_cinfo);
            }
            if (e->overlap) {
                // the next-iteration pipeline's replies must see THIS
                // post-update cinfo and may reuse the req buffers after it
                HIP_CHECK(hipEventRecord(e->ev_cinfo, st));
            } else {
                HIP_CHECK(hipStreamSynchronize(st));
            }
            e->stats.halo_ms +=
                std::chrono::duration<double, std::milli>(
                    std::chrono::steady_clock::now() - t_h0)
                    .count();
        }

        // ---- K7: modularity (dspl.hpp:407-456; p==1 already emitted the
        // partials in the fused kernel above) ----
        if (p > 1) {
            auto f = [cw = e->d_cw, ci = e->d_cinfo] __device__(
                         i64 i, double &a, double &b) {
                a = cw[i];
                const double d = ci[i].degree;
                b = d * d;
            };
            k_partial_sum2<<<nblocks, 256, 0, st>>>(lnv, f, e->d_partials);
        }
        PHASE("k67-done");
        HIP_CHECK(hipMemcpyAsync(partials, e->d_partials, 16 * nblocks,
                                 hipMemcpyDeviceToHost, st));
        if (p > 1 && e->overlap) {
            // issue the WHOLE next-iteration halo pipeline on stream2 now
            // (A's epilogue is queued and drains concurrently): #1a off
            // the exports (ev_p1), discovery off the full targets (ev_p2),
            // replies off the updated cinfo (ev_cinfo)
            HIP_CHECK(hipStreamWaitEvent(e->stream2, e->ev_p1, 0));
            run_halo1a(e, d_target, c2, e->stream2);
            HIP_CHECK(hipStreamWaitEvent(e->stream2, e->ev_p2, 0));
            halo_prep(e, d_target, rc_bounds_nx, req_off_nx, nrc_nx, c2,
                      e->stream2, e->ev_cinfo);
            HIP_CHECK(hipEventRecord(e->ev_halo, e->stream2));
        }
        HIP_CHECK(hipStreamSynchronize(st));
        double le = 0.0, la = 0.0;
        for (int b = 0; b < nblocks; b++) {
            le += partials[2 * b];
            la += partials[2 * b + 1];
        }
        double red[2] = {le, la};
        if (p > 1) comm_allreduce_host(e, red, 2);
        currMod = std::fabs(red[0] * constant - red[1] * constant * constant);
        if (getenv("MV_MOD_DEBUG"))
            std::fprintf(stderr, "[mod] iter=%d le=%.17g la=%.17g\n",
                         numIters, red[0], red[1]);

        // ---- trace ----
        if (e->trace_mod && numIters <= e->trace_cap)
            e->trace_mod[numIters - 1] = currMod;
        if (e->trace_target && numIters <= e->trace_cap) {
            if (p == 1)
                k_depermute32<<<grid_for(lnv), 256, 0, st>>>(
                    lnv, e->d_sigma, e->d_sigma_inv,
                    (const unsigned *)d_target, e->d_trace_tmp);
            else
                k_depermute<<<grid_for(lnv), 256, 0, st>>>(
                    lnv, e->d_sigma_inv, d_target, e->d_trace_tmp);
            HIP_CHECK(hipMemcpyAsync(e->trace_target +
                                         (i64)(numIters - 1) * lnv,
                                     e->d_trace_tmp, 8 * lnv,
                                     hipMemcpyDeviceToHost, st));
            HIP_CHECK(hipStreamSynchronize(st));
        }

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
    // overlap mode posted one speculative #1a past the exit; every rank
    // posted it symmetrically — drain it before returning
    if (e->overlap) HIP_CHECK(hipStreamSynchronize(e->stream2));

    const bool dbg = getenv("MV_SWEEP_DEBUG") != nullptr;
    for (size_t k = 0; k + 1 < sweep_ev.size() + 1; k += 2) {
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, sweep_ev[k], sweep_ev[k + 1]));
        e->stats.sweep_ms += ms;
        if (dbg)
            std::fprintf(stderr, "[sweep] iter %zu: %.3f ms\n", k / 2 + 1, ms);
    }
    e->stats.iters = numIters;
    e->stats.total_ms = std::chrono::duration<double, std::milli>(
                            std::chrono::steady_clock::now() - t_start)
                            .count();
    *iters_out = numIters;
    return prevMod; // dspl.hpp:1440
}

} // extern "C"
