// graph_build.cpp — host-side graph plumbing of the MI355X-native miniVite:
// the 1-D partitioned per-rank CSR (replaces class Graph, graph.hpp:85-296),
// the -l LCG RGG generator (replaces GenerateRGG, graph.hpp:584-1213) redone
// as an O(n) cell-list that yields the BIT-IDENTICAL edge set with no
// communication, and the binary reader/writer (replaces BinaryEdgeList,
// graph.hpp:300-580).
//
// FP discipline: compiled with -ffp-contract=off so sqrt(dx*dx+dy*dy) and
// every weight carries the same bits as the reference built for generic
// x86-64 (no FMA). Equality of the produced arrays with the reference's is
// pinned by tests/test_graph_build.py against the oracle, which is itself
// pinned against the real reference (tests/golden/pins.json).

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <random>
#include <vector>

#include "../../include/minivite_hip.h"

namespace {

// ---- LCG stream (replaces utils.hpp:118-303) -----------------------------
// The reference's 2x2 "matrix power" (utils.hpp:146-176) runs in int64 with
// no modulus, wrapping mod 2^64; because its matrices are all powers of
// [[a,0],[0,1]] with b=0, the whole parallel prefix collapses to
// a^(rank*n) mod 2^64, which we compute by fast exponentiation — provably
// the same value (mod-2^64 multiplication is associative), so the per-rank
// first element (utils.hpp:217-220, C truncated %, possibly negative) and
// the sequential tail (utils.hpp:244-266, fabs mapping) are bit-identical.
constexpr int64_t MLCG = 2147483647; // utils.hpp:54
constexpr int64_t ALCG = 16807;      // utils.hpp:55

uint32_t reseeder(uint32_t initseed) { // utils.hpp:91-98
    std::seed_seq seq({initseed});
    std::vector<uint32_t> s(1);
    seq.generate(s.begin(), s.end());
    return s[0];
}

uint64_t pow_wrap(uint64_t a, uint64_t k) {
    uint64_t r = 1;
    while (k) {
        if (k & 1) r *= a;
        a *= a;
        k >>= 1;
    }
    return r;
}

// Fill the per-rank stream of `strip` (length n_lcg = 2*(nv/p)) mapped to
// (0,1) doubles exactly as LCG::generate (utils.hpp:227-268).
void lcg_strip(uint32_t seed, int64_t n_lcg, int strip, double *out) {
    const int64_t x0 = (int64_t)reseeder(seed); // utils.hpp:133-137
    int64_t x;
    if (strip == 0) {
        x = x0;
    } else {
        uint64_t pre = pow_wrap((uint64_t)ALCG, (uint64_t)strip * (uint64_t)n_lcg);
        x = (int64_t)((uint64_t)x0 * pre) % MLCG; // utils.hpp:220
    }
    const double mult = 1.0 / (1.0 + (double)(MLCG - 1)); // utils.hpp:248
    out[0] = std::fabs((double)x) * mult;
    for (int64_t i = 1; i < n_lcg; i++) {
        x = (x * ALCG) % MLCG; // utils.hpp:244-245 (BLCG=0)
        out[i] = std::fabs((double)x) * mult;
    }
}

} // namespace

struct mv_graph {
    int64_t nv = 0;
    int rank = 0, nranks = 1;
    std::vector<int64_t> parts; // nranks+1
    std::vector<int64_t> xadj;  // lnv+1
    std::vector<int64_t> tails; // lne, global ids
    std::vector<double> weights;
    // Optional INTERNAL-layout hint: locality_perm[k] = the original local
    // vertex id placed at internal position k, ordered so that spatially
    // close vertices (whose ids the RGG assigns randomly) sit close in
    // memory. Pure layout metadata — labels, results and the wire protocol
    // are untouched; the engine uses it so neighbor-community gathers hit
    // L1/L2 instead of round-tripping to the Infinity Cache. Empty when the
    // builder has no spatial knowledge (binary reader, from_csr).
    std::vector<int32_t> locality_perm;
};

// BFS internal order over the LOCAL subgraph, re-seeded from the highest-
// degree unvisited vertex per component: a cheap locality-creating layout
// for graphs without the generator's spatial hint (social-graph neighbors
// land near each other, so the sweep's per-edge community gathers hit
// cache instead of striding a random id space). Pure layout metadata —
// the engine's sigma changes memory placement only; results are identical
// by construction and pinned by the parity suites. MV_NO_BFS_HINT
// disables (perf A/B).
static void build_bfs_hint(mv_graph *g) {
    if (getenv("MV_NO_BFS_HINT")) return;
    const int64_t lnv = (int64_t)g->xadj.size() - 1;
    if (lnv <= 1 || lnv >= (1ll << 31)) return; // int32 labels below
    const int64_t base = g->parts[g->rank], bound = g->parts[g->rank + 1];
    // 2 rounds of async label propagation over the LOCAL subgraph
    // (deterministic sequential order; timestamped counters keep it
    // O(edges)); vertices sharing a propagated label end up adjacent, so
    // a vertex's neighbors — mostly co-members once Louvain coarsens —
    // sit in the same cache blocks.
    std::vector<int32_t> label(lnv), stamp(lnv, -1), cnt(lnv, 0);
    for (int64_t i = 0; i < lnv; i++) label[i] = (int32_t)i;
    for (int round = 0; round < 2; round++) {
        for (int64_t v = 0; v < lnv; v++) {
            int32_t best = label[v];
            int32_t bestc = 0;
            for (int64_t e = g->xadj[v]; e < g->xadj[v + 1]; e++) {
                const int64_t t = g->tails[e];
                if (t < base || t >= bound) continue;
                const int32_t l = label[t - base];
                if (stamp[l] != (int32_t)v) {
                    stamp[l] = (int32_t)v;
                    cnt[l] = 0;
                }
                const int32_t c = ++cnt[l];
                if (c > bestc || (c == bestc && l < best)) {
                    bestc = c;
                    best = l;
                }
            }
            label[v] = best;
        }
        std::fill(stamp.begin(), stamp.end(), -1);
    }
    // order = (label block, id); stable so ties keep id order
    std::vector<int32_t> order(lnv);
    for (int64_t i = 0; i < lnv; i++) order[i] = (int32_t)i;
    std::stable_sort(order.begin(), order.end(),
                     [&](int32_t a, int32_t b) {
                         return label[a] < label[b];
                     });
    g->locality_perm = std::move(order);
}

extern "C" {

mv_graph *mv_graph_rgg(int64_t nv, int rank, int nranks,
                       int unit_edge_weight, double random_edge_percent,
                       uint64_t random_edge_seed) {
    if (nranks <= 0 || (nranks & (nranks - 1)) || nv % nranks || rank < 0 ||
        rank >= nranks) { // graph.hpp:610-626
        std::fprintf(stderr, "mv_graph_rgg: nranks must be 2^k dividing nv\n");
        return nullptr;
    }
    const int64_t n_ = nv / nranks; // graph.hpp:608
    // radius (graph.hpp:628-631; PI is the reference's 5-digit 3.14159,
    // utils.hpp:44)
    const double rc = std::sqrt(std::log((double)nv) / (3.14159 * (double)nv));
    const double rt = std::sqrt(2.0736 / (double)nv);
    const double rn = (rc + rt) / 2.0;
    if (!(1.0 / (double)nranks > rn)) { // graph.hpp:633
        std::fprintf(stderr, "mv_graph_rgg: strip width must exceed rn\n");
        return nullptr;
    }

    auto *g = new mv_graph;
    g->nv = nv;
    g->rank = rank;
    g->nranks = nranks;
    g->parts.resize(nranks + 1);
    for (int r = 0; r <= nranks; r++)
        g->parts[r] = (nv * (int64_t)r) / nranks; // graph.hpp:112-113

    // Coordinates of this strip and its neighbors. X in (0,1); Y confined to
    // [strip/p, (strip+1)/p) (graph.hpp:668-716, utils.hpp:272-294).
    const int s_lo = rank > 0 ? rank - 1 : rank;
    const int s_hi = rank < nranks - 1 ? rank + 1 : rank;
    const int nstrips = s_hi - s_lo + 1;
    const double range = 1.0 / (double)nranks; // utils.hpp:274
    std::vector<double> X((size_t)nstrips * n_), Y((size_t)nstrips * n_);
    {
        std::vector<double> d(2 * n_);
        for (int s = s_lo; s <= s_hi; s++) {
            lcg_strip(1, 2 * n_, s, d.data()); // seed 1: graph.hpp:708
            const double lo = (double)s * range;
            double *Xs = X.data() + (size_t)(s - s_lo) * n_;
            double *Ys = Y.data() + (size_t)(s - s_lo) * n_;
            for (int64_t i = 0; i < n_; i++) {
                Xs[i] = d[i];
                Ys[i] = lo + range * d[n_ + i]; // utils.hpp:291-292
            }
        }
    }

    // Cell list over the loaded strips. Cell side >= rn in both axes so a
    // radius-rn disc is covered by the 3x3 neighborhood.
    const int ncx = std::max<int64_t>(1, (int64_t)(1.0 / rn));
    const double inv_cs = (double)ncx; // cells are 1/ncx wide >=... (1/ncx >= rn ⟺ ncx <= 1/rn) ✓
    const double ylo_all = (double)s_lo * range;
    const double yhi_all = (double)(s_hi + 1) * range;
    const int row0 = std::max(0, (int)(ylo_all * inv_cs) - 1);
    const int row1 = std::min(ncx - 1, (int)(yhi_all * inv_cs) + 1);
    const int nrows = row1 - row0 + 1;
    const int64_t npts = (int64_t)nstrips * n_;
    const int64_t ncells = (int64_t)nrows * ncx;

    auto cell_of = [&](double x, double y) -> int64_t {
        int cx = (int)(x * inv_cs);
        if (cx >= ncx) cx = ncx - 1;
        int cy = (int)(y * inv_cs);
        if (cy >= ncx) cy = ncx - 1;
        return (int64_t)(cy - row0) * ncx + cx;
    };

    std::vector<int64_t> ccount(ncells + 1, 0);
    for (int64_t t = 0; t < npts; t++)
        ccount[cell_of(X[t], Y[t]) + 1]++;
    for (int64_t c = 0; c < ncells; c++) ccount[c + 1] += ccount[c];
    std::vector<int64_t> cpts(npts);
    {
        std::vector<int64_t> cur(ccount.begin(), ccount.end() - 1);
        for (int64_t t = 0; t < npts; t++)
            cpts[cur[cell_of(X[t], Y[t])]++] = t;
    }
    // coordinates packed in cell order: the pair loop below then streams
    // contiguous memory instead of gathering across the whole strip
    std::vector<double> Xc(npts), Yc(npts);
    for (int64_t q = 0; q < npts; q++) {
        Xc[q] = X[cpts[q]];
        Yc[q] = Y[cpts[q]];
    }

    // Row of own vertex i = every neighbor within rn in any strip:
    //  - own strip: all j != i (reference emits (i,g_j) and (j,g_i) per
    //    unordered pair, graph.hpp:759-788);
    //  - adjacent strip: all j with j != i — the reference's ghost loops
    //    start j at i+1 on BOTH sides (graph.hpp:817, :849), so a
    //    cross-rank pair with equal local indices is never tested; each
    //    side contributes the other direction via the edge shipping
    //    (graph.hpp:884-935).
    // Tails per row sorted ascending == the reference's (row, tail) sort
    // (graph.hpp:1145-1153) restricted to the row.
    const int64_t own_off = (int64_t)(rank - s_lo) * n_;

    // two passes (count, then fill + per-row insertion sort) — no
    // intermediate per-row vectors, so n=2^26 fits comfortably
    g->xadj.assign(n_ + 1, 0);
    auto for_each_neighbor = [&](int64_t i, auto &&emit) {
        const double xi = X[own_off + i], yi = Y[own_off + i];
        const int cxi = std::min((int)(xi * inv_cs), ncx - 1);
        const int cyi = std::min((int)(yi * inv_cs), ncx - 1);
        for (int dy = -1; dy <= 1; dy++) {
            const int cy = cyi + dy;
            if (cy < row0 || cy > row1) continue;
            for (int dx = -1; dx <= 1; dx++) {
                const int cx = cxi + dx;
                if (cx < 0 || cx >= ncx) continue;
                const int64_t c = (int64_t)(cy - row0) * ncx + cx;
                for (int64_t q = ccount[c]; q < ccount[c + 1]; q++) {
                    const double ddx = xi - Xc[q];
                    const double ddy = yi - Yc[q];
                    const double ed = std::sqrt(ddx * ddx + ddy * ddy);
                    if (ed > rn) continue;
                    const int64_t t = cpts[q];
                    if (t == own_off + i) continue;
                    const int strip = (int)(t / n_) + s_lo;
                    const int64_t j = t % n_;
                    if (strip != rank && j == i) continue; // the j>i quirk
                    emit((int64_t)strip * n_ + j, ed);
                }
            }
        }
    };
#pragma omp parallel for schedule(dynamic, 4096)
    for (int64_t i = 0; i < n_; i++) {
        int64_t cnt = 0;
        for_each_neighbor(i, [&](int64_t, double) { cnt++; });
        g->xadj[i + 1] = cnt;
    }
    for (int64_t i = 0; i < n_; i++) g->xadj[i + 1] += g->xadj[i];
    const int64_t lne_pre = g->xadj[n_];
    g->tails.resize(lne_pre);
    g->weights.resize(lne_pre);
#pragma omp parallel for schedule(dynamic, 4096)
    for (int64_t i = 0; i < n_; i++) {
        int64_t o = g->xadj[i];
        const int64_t o0 = o;
        for_each_neighbor(i, [&](int64_t t, double ed) {
            // insertion in ascending tail order (rows are short)
            int64_t k = o;
            const double wv = unit_edge_weight ? 1.0 : ed;
            while (k > o0 && g->tails[k - 1] > t) {
                g->tails[k] = g->tails[k - 1];
                g->weights[k] = g->weights[k - 1];
                k--;
            }
            g->tails[k] = t;
            g->weights[k] = wv;
            o++;
        });
    }

    // Optional extra random edges (mirrors -p, graph.hpp:939-1122, with an
    // explicit seed; every rank replays every rank's draws so the reverse
    // directions land without communication; the reference's O(n) duplicate
    // scan is not reproduced — perf configs only, never parity).
    if (random_edge_percent > 0.0) {
        // The reference allreduces the exact undirected count
        // (graph.hpp:941-946); to stay communication-free every rank here
        // derives the SAME deterministic estimate from the RGG's expected
        // degree (nv*pi*rn^2), so all ranks replay identical draw counts.
        // Documented deviation; -p is perf-only (its reference seed is
        // time(0)^getpid(), graph.hpp:990 — never a parity config).
        const int64_t tot_und =
            (int64_t)((double)nv * (double)nv * 3.14159 * rn * rn / 2.0);
        const int64_t nrande =
            (int64_t)(random_edge_percent * (double)tot_und) / 100;
        int64_t per = nrande / nranks;
        struct TW { int64_t t; double w; };
        std::vector<std::vector<TW>> extra(n_);
        for (int src_rank = 0; src_rank < nranks; src_rank++) {
            std::mt19937_64 re(random_edge_seed * 0x9E3779B97F4A7C15ull +
                               (uint64_t)src_rank);
            std::uniform_int_distribution<int64_t> IR(0, n_ - 1),
                JR(0, nv - 1);
            int64_t cnt = per + ((src_rank == nranks - 1) ? nrande % nranks : 0);
            for (int64_t k = 0; k < cnt; k++) {
                const int64_t i = IR(re);
                const int64_t g_j = JR(re);
                const int target = (int)std::min<int64_t>(
                    nranks - 1, g_j / n_); // uniform parts ⇒ owner = g_j/n_
                const int64_t j = g_j - g->parts[target];
                if (target == src_rank && i == j) continue;
                const int64_t g_i = g->parts[src_rank] + i;
                if (src_rank == rank) extra[i].push_back({g_j, 1.0});
                if (target == rank) extra[j].push_back({g_i, 1.0});
            }
        }
        // merge the extra edges into the CSR (rebuild with shifted rows)
        int64_t n_extra = 0;
        for (auto &v : extra) n_extra += (int64_t)v.size();
        if (n_extra) {
            std::vector<int64_t> nxadj(n_ + 1, 0);
            for (int64_t i = 0; i < n_; i++)
                nxadj[i + 1] = nxadj[i] + (g->xadj[i + 1] - g->xadj[i]) +
                               (int64_t)extra[i].size();
            std::vector<int64_t> ntails(nxadj[n_]);
            std::vector<double> nweights(nxadj[n_]);
#pragma omp parallel for schedule(static)
            for (int64_t i = 0; i < n_; i++) {
                int64_t o = nxadj[i];
                for (int64_t e = g->xadj[i]; e < g->xadj[i + 1]; e++, o++) {
                    ntails[o] = g->tails[e];
                    nweights[o] = g->weights[e];
                }
                for (auto &x : extra[i]) {
                    int64_t k = o;
                    while (k > nxadj[i] && ntails[k - 1] > x.t) {
                        ntails[k] = ntails[k - 1];
                        nweights[k] = nweights[k - 1];
                        k--;
                    }
                    ntails[k] = x.t;
                    nweights[k] = x.w;
                    o++;
                }
            }
            g->xadj = std::move(nxadj);
            g->tails = std::move(ntails);
            g->weights = std::move(nweights);
        }
    }

    // spatial layout hint: own vertices in row-major cell order (stable, so
    // ties keep id order) — see mv_graph::locality_perm
    {
        std::vector<std::pair<int64_t, int32_t>> order(n_);
        for (int64_t i = 0; i < n_; i++) {
            const double xi = X[own_off + i], yi = Y[own_off + i];
            const int cx = std::min((int)(xi * inv_cs), ncx - 1);
            const int cy = std::min((int)(yi * inv_cs), ncx - 1);
            order[i] = {(int64_t)cy * ncx + cx, (int32_t)i};
        }
        std::stable_sort(order.begin(), order.end(),
                         [](const auto &a, const auto &b) {
                             return a.first < b.first;
                         });
        g->locality_perm.resize(n_);
        for (int64_t k = 0; k < n_; k++)
            g->locality_perm[k] = order[k].second;
    }

    return g;
}

mv_graph *mv_graph_from_csr(int64_t nv, int rank, int nranks,
                            const int64_t *parts, int64_t lnv, int64_t lne,
                            const int64_t *xadj, const int64_t *tails,
                            const double *weights) {
    auto *g = new mv_graph;
    g->nv = nv;
    g->rank = rank;
    g->nranks = nranks;
    g->parts.assign(parts, parts + nranks + 1);
    if (g->parts[rank + 1] - g->parts[rank] != lnv) {
        delete g;
        std::fprintf(stderr, "mv_graph_from_csr: lnv != parts range\n");
        return nullptr;
    }
    g->xadj.assign(xadj, xadj + lnv + 1);
    g->tails.assign(tails, tails + lne);
    if (weights)
        g->weights.assign(weights, weights + lne);
    else
        g->weights.assign(lne, 1.0);
    build_bfs_hint(g);
    return g;
}

// Binary format (graph.hpp:342-383): int64 nv, int64 ne, (nv+1) int64 global
// CSR offsets, ne x {int64 tail, double weight} records.
mv_graph *mv_graph_read_binary(const char *path, int rank, int nranks,
                               int balanced) {
    FILE *f = std::fopen(path, "rb");
    if (!f) {
        std::fprintf(stderr, "mv_graph_read_binary: cannot open %s\n", path);
        return nullptr;
    }
    int64_t nv = 0, ne = 0;
    if (std::fread(&nv, 8, 1, f) != 1 || std::fread(&ne, 8, 1, f) != 1) {
        std::fclose(f);
        return nullptr;
    }
    auto *g = new mv_graph;
    g->nv = nv;
    g->rank = rank;
    g->nranks = nranks;
    g->parts.resize(nranks + 1);
    if (!balanced) {
        for (int r = 0; r <= nranks; r++)
            g->parts[r] = (nv * (int64_t)r) / nranks; // graph.hpp:344
    } else {
        // find_balanced_num_edges (graph.hpp:416-461): bin vertices so each
        // rank holds ~ne/nranks edges; excess piles on the last rank.
        std::vector<int64_t> off(nv + 1);
        if (std::fread(off.data(), 8, nv + 1, f) != (size_t)(nv + 1)) {
            std::fclose(f);
            delete g;
            return nullptr;
        }
        std::vector<int64_t> mbins(nranks + 1, 0), nbins(nranks, 0);
        const int64_t nbcap = ne / nranks;
        int p = 0;
        // The reference scans off[0..nv-1] sequentially and bins vertex m
        // by off[m] - off[m-1] (with off[-1] = 0) — i.e. the degree LAGGED
        // by one vertex (graph.hpp:437-451). Reproduced exactly: the -b
        // partition must match the reference's bit-for-bit.
        int64_t past = 0;
        for (int64_t m = 0; m < nv; m++) {
            const int64_t deg = off[m] - past;
            if (nbins[p] < nbcap || p == nranks - 1) nbins[p] += deg;
            if (nbins[p] >= nbcap && p < nranks - 1) p++;
            mbins[p + 1]++;
            past = off[m];
        }
        for (int k = 1; k <= nranks; k++) mbins[k] += mbins[k - 1];
        g->parts = mbins; // repart, graph.hpp:511
    }
    const int64_t v0 = g->parts[rank], v1 = g->parts[rank + 1];
    const int64_t lnv = v1 - v0;
    g->xadj.resize(lnv + 1);
    // offsets slice (fseeko: offsets exceed 2^31 for >2 GB files)
    fseeko(f, (off_t)(16 + v0 * 8), SEEK_SET);
    if (std::fread(g->xadj.data(), 8, lnv + 1, f) != (size_t)(lnv + 1)) {
        std::fclose(f);
        delete g;
        return nullptr;
    }
    const int64_t e0 = g->xadj[0], e1 = g->xadj[lnv];
    const int64_t lne = e1 - e0;
    for (int64_t i = 0; i <= lnv; i++) g->xadj[i] -= e0; // graph.hpp:407-409
    g->tails.resize(lne);
    g->weights.resize(lne);
    std::vector<char> buf(lne * 16);
    const int64_t edge_base = 16 + (nv + 1) * 8;
    fseeko(f, (off_t)(edge_base + e0 * 16), SEEK_SET);
    if (lne && std::fread(buf.data(), 16, lne, f) != (size_t)lne) {
        std::fclose(f);
        delete g;
        return nullptr;
    }
    for (int64_t e = 0; e < lne; e++) {
        std::memcpy(&g->tails[e], buf.data() + e * 16, 8);
        std::memcpy(&g->weights[e], buf.data() + e * 16 + 8, 8);
    }
    std::fclose(f);
    build_bfs_hint(g);
    return g;
}

int mv_graph_write_binary(const mv_graph *g, const char *path) {
    if (g->nranks != 1) {
        std::fprintf(stderr, "mv_graph_write_binary: whole graphs only\n");
        return -1;
    }
    FILE *f = std::fopen(path, "wb");
    if (!f) return -1;
    const int64_t nv = g->nv, ne = (int64_t)g->tails.size();
    std::fwrite(&nv, 8, 1, f);
    std::fwrite(&ne, 8, 1, f);
    std::fwrite(g->xadj.data(), 8, nv + 1, f);
    for (int64_t e = 0; e < ne; e++) {
        std::fwrite(&g->tails[e], 8, 1, f);
        std::fwrite(&g->weights[e], 8, 1, f);
    }
    std::fclose(f);
    return 0;
}

void mv_graph_free(mv_graph *g) { delete g; }
int64_t mv_graph_nv(const mv_graph *g) { return g->nv; }
int64_t mv_graph_lnv(const mv_graph *g) {
    return g->parts[g->rank + 1] - g->parts[g->rank];
}
int64_t mv_graph_lne(const mv_graph *g) { return (int64_t)g->tails.size(); }
const int64_t *mv_graph_parts(const mv_graph *g) { return g->parts.data(); }
const int64_t *mv_graph_xadj(const mv_graph *g) { return g->xadj.data(); }
const int64_t *mv_graph_tails(const mv_graph *g) { return g->tails.data(); }
const double *mv_graph_weights(const mv_graph *g) { return g->weights.data(); }
const int32_t *mv_graph_locality_hint(const mv_graph *g) {
    return g->locality_perm.empty() ? nullptr : g->locality_perm.data();
}

} // extern "C"
