/* ref_louvain.c — ORACLE: CPU restatement of miniVite's distributed Louvain
 * phase-1 hot path and its RGG input generator.
 *
 * TEST INFRASTRUCTURE ONLY. Only tests/, __graft_entry__.smoke() and
 * bench.py's cpu_baseline leg may load this library. The product path
 * (minivite_amd) never links, loads or calls it; the HIP extension fails
 * loudly if missing — there is no fallback through this code.
 *
 * Every function cites the reference (read-only at /root/reference) file:line
 * it restates. The restatement simulates all P ranks of the reference's MPI
 * decomposition inside one process, deterministically: all MPI exchanges
 * become in-process array reads with a fixed order. For unit edge weights all
 * floating-point accumulations are sums of integers representable in double,
 * so results are bit-exact w.r.t. the reference regardless of summation
 * order; for -w (Euclidean) weights, per-vertex accumulations follow the
 * reference's sequential edge order and cross-rank sums use rank order
 * 0..P-1 (the reference's own cross-thread atomics make its last bits
 * run-order dependent there; parity bar for -w is |dmod| < 1e-9 plus
 * community agreement).
 *
 * Pinned against the real reference (oracle/_ref, built from the sources
 * under /root/reference by oracle/Makefile) via tests/golden/pins.json:
 * bit-exact graph arrays, per-iteration modularity (hex float) and
 * per-iteration community arrays for the configs listed there.
 *
 * Compile: gcc -O2 -std=c11 -fPIC -ffp-contract=off -shared -lm
 * (-ffp-contract=off: the reference is built for generic x86-64 where gcc
 * emits no FMA; contraction would change sqrt(dx*dx+dy*dy) and gain bits.)
 */

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <math.h>
#include <stdio.h>

/* ---------------- LCG parameters (utils.hpp:54-56) ---------------- */
#define MLCG 2147483647LL /* 2^31 - 1 */
#define ALCG 16807LL      /* 7^5 */
#define BLCG 0LL

/* ------------------------------------------------------------------ */
/* std::seed_seq::generate for one 32-bit output from one seed input,
 * restated from ISO C++ [rand.util.seedseq] — used by reseeder()
 * (utils.hpp:91-98). Verified against libstdc++: orc_reseeder(1) ==
 * 1967017404. */
static uint32_t seedseq_T(uint32_t x) { return x ^ (x >> 27); }

uint32_t orc_reseeder(uint32_t initseed)
{
    /* n = 1 output, s = 1 seed */
    const uint32_t n = 1, s = 1;
    uint32_t b[1] = { 0x8b8b8b8bu };
    const uint32_t t = (n >= 623) ? 11 : (n >= 68) ? 7 : (n >= 39) ? 5
                     : (n >= 7) ? 3 : (n - 1) / 2;
    const uint32_t p = (n - t) / 2, q = p + t;
    const uint32_t m = (s + 1 > n) ? (s + 1) : n;
    uint32_t seeds[1] = { initseed };
    for (uint32_t k = 0; k < m; ++k) {
        uint32_t r1 = 1664525u * seedseq_T(b[k % n] ^ b[(k + p) % n] ^ b[(k + n - 1) % n]);
        uint32_t r2 = (k == 0) ? r1 + s
                    : (k <= s) ? r1 + (k % n) + seeds[k - 1]
                               : r1 + (k % n);
        b[(k + p) % n] += r1;
        b[(k + q) % n] += r2;
        b[k % n] = r2;
    }
    for (uint32_t k = m; k < m + n; ++k) {
        uint32_t r3 = 1566083941u * seedseq_T(b[k % n] + b[(k + p) % n] + b[(k + n - 1) % n]);
        uint32_t r4 = r3 - (uint32_t)(k % n);
        b[(k + p) % n] ^= r3;
        b[(k + q) % n] ^= r4;
        b[k % n] = r4;
    }
    return b[0];
}

/* ------------------------------------------------------------------ */
/* Parallel LCG (utils.hpp:118-303).
 *
 * The reference's 2x2 matrix power (utils.hpp:146-176) runs in plain int64
 * with NO modulus, so for realistic n it wraps mod 2^64 (signed overflow;
 * gcc wraps). rnums_[0] = (x0*prefix[0] + prefix[2]) % MLCG (utils.hpp:220)
 * then uses C truncated %, which can go NEGATIVE; generate() (utils.hpp:244-266)
 * keeps the recurrence in that signed domain and maps with fabs(). All of
 * that is restated exactly, with the wrap done in uint64.
 */
typedef struct { uint64_t m[4]; } Mat2;

static Mat2 mat_mul(Mat2 a, Mat2 b) /* utils.hpp:146-157, wrapping int64 */
{
    Mat2 c;
    for (int i = 0; i < 2; i++)
        for (int j = 0; j < 2; j++) {
            uint64_t s = 0;
            for (int k = 0; k < 2; k++)
                s += a.m[i * 2 + k] * b.m[k * 2 + j];
            c.m[i * 2 + j] = s;
        }
    return c;
}

static Mat2 mat_power(Mat2 mat, int64_t k) /* utils.hpp:167-176: k-1 sequential multiplies */
{
    Mat2 tmp = mat;
    for (int64_t p = 0; p < k - 1; p++)
        mat = mat_mul(mat, tmp);
    return mat;
}

/* Fill drand[r][0..n_lcg) for every rank r in 0..p-1, where n_lcg is the
 * per-rank stream length (utils.hpp:121-141, 183-268). drand must hold
 * p * n_lcg doubles (rank r at drand + r*n_lcg). seed is the LCG ctor seed
 * (graph.hpp:708 passes 1). */
void orc_lcg_fill(uint32_t seed, int64_t n_lcg, int p, double *drand)
{
    int64_t x0 = (int64_t)orc_reseeder(seed); /* utils.hpp:133-134, bcast :137 */

    /* parallel_prefix_op (utils.hpp:183-221) simulated for all p ranks */
    Mat2 *global_op = malloc(sizeof(Mat2) * p);
    Mat2 *prefix_op = malloc(sizeof(Mat2) * p);
    Mat2 I = { {1, 0, 0, 1} };
    Mat2 G = { {(uint64_t)ALCG, 0, (uint64_t)BLCG, 1} }; /* utils.hpp:185-189 */
    Mat2 g0 = mat_power(G, n_lcg);                        /* M^(n/p), :191 */
    for (int r = 0; r < p; r++) { global_op[r] = g0; prefix_op[r] = I; }

    int steps = (int)(log2((double)p)); /* utils.hpp:196 */
    Mat2 *recv = malloc(sizeof(Mat2) * p);
    for (int s = 0; s < steps; s++) {
        for (int r = 0; r < p; r++) recv[r] = global_op[r ^ (1 << s)];
        for (int r = 0; r < p; r++) {
            global_op[r] = mat_mul(global_op[r], recv[r]); /* :207 */
            if ((r ^ (1 << s)) < r)
                prefix_op[r] = mat_mul(prefix_op[r], recv[r]); /* :209-210 */
        }
    }
    free(recv);

    const double mult = 1.0 / (1.0 + (double)(MLCG - 1)); /* utils.hpp:248 */
    for (int r = 0; r < p; r++) {
        int64_t x; /* rnums_[0], utils.hpp:217-220 */
        if (r == 0) x = x0;
        else {
            uint64_t v = (uint64_t)x0 * prefix_op[r].m[0] + prefix_op[r].m[2];
            x = (int64_t)v % MLCG; /* C truncated %, may be negative */
        }
        double *d = drand + (int64_t)r * n_lcg;
        d[0] = fabs((double)x) * mult; /* utils.hpp:265-266 */
        for (int64_t i = 1; i < n_lcg; i++) {
            x = (x * ALCG + BLCG) % MLCG; /* utils.hpp:244-245, |x|<MLCG: no overflow */
            d[i] = fabs((double)x) * mult;
        }
    }
    free(global_op);
    free(prefix_op);
}

/* ------------------------------------------------------------------ */
/* Graph container: P per-rank CSRs with the 1-D partition (graph.hpp:85-296).
 * parts[r] = (nv*r)/p (graph.hpp:112-113) unless repart'd. Tails are GLOBAL
 * vertex ids; row i of rank r is global vertex parts[r]+i. */
typedef struct {
    int64_t nv;
    int p;
    int64_t *parts;    /* p+1 */
    int64_t **xadj;    /* per rank: lnv+1 */
    int64_t **tails;   /* per rank: lne */
    double **weights;  /* per rank: lne */
    int64_t *lne;      /* per rank */
} OrcGraph;

void *orc_graph_new(int64_t nv, int p)
{
    OrcGraph *g = calloc(1, sizeof(OrcGraph));
    g->nv = nv; g->p = p;
    g->parts = malloc(8 * (p + 1));
    for (int r = 0; r <= p; r++) g->parts[r] = (nv * (int64_t)r) / p; /* graph.hpp:112-113 */
    g->xadj = calloc(p, sizeof(void *));
    g->tails = calloc(p, sizeof(void *));
    g->weights = calloc(p, sizeof(void *));
    g->lne = calloc(p, 8);
    return g;
}

void orc_graph_set_parts(void *h, const int64_t *parts) /* repart, graph.hpp:124-125 */
{
    OrcGraph *g = h;
    memcpy(g->parts, parts, 8 * (g->p + 1));
}

void orc_graph_set_rank_csr(void *h, int r, int64_t lnv, int64_t lne,
                            const int64_t *xadj, const int64_t *tails,
                            const double *w)
{
    OrcGraph *g = h;
    free(g->xadj[r]); free(g->tails[r]); free(g->weights[r]);
    g->xadj[r] = malloc(8 * (lnv + 1));
    memcpy(g->xadj[r], xadj, 8 * (lnv + 1));
    g->tails[r] = malloc(8 * lne);
    memcpy(g->tails[r], tails, 8 * lne);
    g->weights[r] = malloc(8 * lne);
    if (w) memcpy(g->weights[r], w, 8 * lne);
    else for (int64_t i = 0; i < lne; i++) g->weights[r][i] = 1.0;
    g->lne[r] = lne;
}

int64_t orc_graph_lne(void *h, int r) { return ((OrcGraph *)h)->lne[r]; }
int64_t orc_graph_lnv(void *h, int r)
{
    OrcGraph *g = h;
    return g->parts[r + 1] - g->parts[r];
}
const int64_t *orc_graph_xadj(void *h, int r) { return ((OrcGraph *)h)->xadj[r]; }
const int64_t *orc_graph_tails(void *h, int r) { return ((OrcGraph *)h)->tails[r]; }
const double *orc_graph_weights(void *h, int r) { return ((OrcGraph *)h)->weights[r]; }

void orc_graph_free(void *h)
{
    OrcGraph *g = h;
    for (int r = 0; r < g->p; r++) {
        free(g->xadj[r]); free(g->tails[r]); free(g->weights[r]);
    }
    free(g->xadj); free(g->tails); free(g->weights);
    free(g->lne); free(g->parts); free(g);
}

/* ------------------------------------------------------------------ */
/* RGG generator (graph.hpp:584-1213), brute-force pair loops exactly as the
 * reference writes them (O((nv/p)^2) — oracle-scale inputs only).
 * Requirements mirrored: p = 2^k, p | nv (graph.hpp:610-626), 1/p > rn_
 * (:633). LCG path only (-l; graph.hpp:703-729). randomEdgePercent > 0 is
 * excluded (its seed is time(0)^getpid(), graph.hpp:990 — not reproducible).
 *
 * Edge set semantics restated:
 *  - same-rank pairs i<j within rn_: two directed edges (graph.hpp:759-788);
 *  - cross-rank pairs between adjacent ranks ONLY, and only where the two
 *    LOCAL indices satisfy j > i in each side's ghost loop
 *    (graph.hpp:816-877: both ghost loops start j at i+1) — i.e. an
 *    up/down pair with EQUAL local indices is never tested;
 *  - weight = 1.0, or the Euclidean distance when !unitEdgeWeight;
 *  - final CSR sorted by (local row, global tail) (graph.hpp:1145-1180).
 */
typedef struct { int64_t row; int64_t tail; double w; } ETup;

static int etup_cmp(const void *a, const void *b) /* graph.hpp:1145-1146 */
{
    const ETup *x = a, *y = b;
    if (x->row != y->row) return x->row < y->row ? -1 : 1;
    if (x->tail != y->tail) return x->tail < y->tail ? -1 : 1;
    return 0;
}

typedef struct { ETup *v; int64_t n, cap; } EVec;
static void evec_push(EVec *e, int64_t row, int64_t tail, double w)
{
    if (e->n == e->cap) {
        e->cap = e->cap ? e->cap * 2 : 1024;
        e->v = realloc(e->v, sizeof(ETup) * e->cap);
    }
    e->v[e->n].row = row; e->v[e->n].tail = tail; e->v[e->n].w = w;
    e->n++;
}

void *orc_rgg_generate(int64_t nv, int p, int unit_weight)
{
    /* mirror the reference's input checks (graph.hpp:610-626) */
    if (p <= 0 || (p & (p - 1)) || nv % p) return NULL;
    OrcGraph *g = orc_graph_new(nv, p);
    const int64_t n_ = nv / p;                       /* graph.hpp:608 */
    const double rc = sqrt(log((double)nv) / (3.14159 * (double)nv)); /* graph.hpp:629, PI utils.hpp:44 */
    const double rt = sqrt(2.0736 / (double)nv);      /* graph.hpp:630 */
    const double rn = (rc + rt) / 2.0;                /* graph.hpp:631 */

    /* coordinates for all ranks: X = drand[r][0..n_), Y = lo + (1/p)*drand[r][n_..2n_)
     * (graph.hpp:703-716, utils.hpp:272-294; LCG seed 1, graph.hpp:708) */
    double *drand = malloc(sizeof(double) * 2 * n_ * p);
    orc_lcg_fill(1, 2 * n_, p, drand);
    double *X = malloc(sizeof(double) * nv), *Y = malloc(sizeof(double) * nv);
    const double range = 1.0 / (double)p; /* utils.hpp:274 */
    for (int r = 0; r < p; r++) {
        const double lo = (double)r * range; /* graph.hpp:671-672 */
        for (int64_t i = 0; i < n_; i++) {
            X[r * n_ + i] = drand[(int64_t)r * 2 * n_ + i];
            Y[r * n_ + i] = lo + range * drand[(int64_t)r * 2 * n_ + n_ + i]; /* utils.hpp:291-292 */
        }
    }
    free(drand);

    EVec *ev = calloc(p, sizeof(EVec));
    /* same-rank pairs (graph.hpp:759-788) */
    for (int r = 0; r < p; r++) {
        const int64_t base = g->parts[r];
        for (int64_t i = 0; i < n_; i++)
            for (int64_t j = i + 1; j < n_; j++) {
                double dx = X[base + i] - X[base + j];
                double dy = Y[base + i] - Y[base + j];
                double ed = sqrt(dx * dx + dy * dy);
                if (ed <= rn) {
                    double w = unit_weight ? 1.0 : ed;
                    evec_push(&ev[r], i, base + j, w);
                    evec_push(&ev[r], j, base + i, w);
                }
            }
    }
    /* adjacent-rank pairs: rank r's "up" ghost loop against rank r-1
     * (graph.hpp:814-844) — i local to r, j local to r-1, j from i+1.
     * r's loop adds (i -> g_j) on r and ships (j -> g_i) to r-1
     * (recv loops graph.hpp:910-935). The symmetric "down" loop
     * (graph.hpp:846-877) is rank r-1's view of the same pair set with roles
     * swapped and covers j > i with i local to r-1 — i.e. together every
     * unordered adjacent-rank pair (a in r-1, b in r) with DIFFERENT local
     * indices is tested exactly once. */
    for (int r = 1; r < p; r++) {
        const int64_t up = r - 1;
        for (int64_t i = 0; i < n_; i++)
            for (int64_t j = i + 1; j < n_; j++) {
                /* up loop on rank r: X[i] (rank r) vs X_up[j] (rank r-1) */
                double dx = X[r * n_ + i] - X[up * n_ + j];
                double dy = Y[r * n_ + i] - Y[up * n_ + j];
                double ed = sqrt(dx * dx + dy * dy);
                if (ed <= rn) {
                    double w = unit_weight ? 1.0 : ed;
                    evec_push(&ev[r], i, up * n_ + j, w);   /* graph.hpp:828-832 */
                    evec_push(&ev[up], j, r * n_ + i, w);   /* sendup -> recvdn :916-919 */
                }
                /* down loop on rank r-1: X[i] (rank r-1) vs X_down[j] (rank r) */
                dx = X[up * n_ + i] - X[r * n_ + j];
                dy = Y[up * n_ + i] - Y[r * n_ + j];
                ed = sqrt(dx * dx + dy * dy);
                if (ed <= rn) {
                    double w = unit_weight ? 1.0 : ed;
                    evec_push(&ev[up], i, r * n_ + j, w);   /* graph.hpp:858-865 */
                    evec_push(&ev[r], j, up * n_ + i, w);   /* senddn -> recvup :929-933 */
                }
            }
    }
    free(X); free(Y);

    /* CSR finalize per rank (graph.hpp:1126-1180): sort by (row, tail) */
    for (int r = 0; r < p; r++) {
        qsort(ev[r].v, ev[r].n, sizeof(ETup), etup_cmp);
        int64_t lne = ev[r].n;
        int64_t *xadj = calloc(n_ + 1, 8);
        int64_t *tails = malloc(8 * (lne > 0 ? lne : 1));
        double *w = malloc(8 * (lne > 0 ? lne : 1));
        for (int64_t e = 0; e < lne; e++) {
            xadj[ev[r].v[e].row + 1]++;
            tails[e] = ev[r].v[e].tail;
            w[e] = ev[r].v[e].w;
        }
        for (int64_t i = 0; i < n_; i++) xadj[i + 1] += xadj[i];
        g->xadj[r] = xadj; g->tails[r] = tails; g->weights[r] = w;
        g->lne[r] = lne;
        free(ev[r].v);
    }
    free(ev);
    return g;
}

/* ------------------------------------------------------------------ */
/* Louvain phase 1 (dspl.hpp:1280-1441), all P ranks simulated in-process. */

static int owner_of(const OrcGraph *g, int64_t v) /* graph.hpp:167-173 */
{
    /* upper_bound(parts, v) - 1 */
    int lo = 0, hi = g->p; /* parts has p+1 entries; answer in [0, p-1] */
    while (lo < hi) {
        int mid = (lo + hi) / 2;
        if (g->parts[mid] <= v) lo = mid + 1; else hi = mid;
    }
    return lo - 1;
}

/* per-rank iteration state */
typedef struct {
    int64_t lnv, base, bound;
    int64_t *currComm, *pastComm, *targetComm;
    double *vDegree, *clusterWeight;
    int64_t *cinfo_size; double *cinfo_degree;   /* localCinfo */
    int64_t *cupd_size;  double *cupd_degree;    /* localCupdate */
    /* ghosts (exchangeVertexReqs, dspl.hpp:1112-1272): sorted unique remote
     * tails this rank references; order within the want-list is unspecified
     * in the reference (unordered_set) and does not affect results. */
    int64_t nghost; int64_t *ghosts;             /* sorted global ids */
    int64_t *tidx;                                /* translated tails: local i -> i, ghost -> lnv+slot */
    int64_t *ghost_comm;                          /* per-iteration: community of ghosts[slot] */
    /* remote community info for this iteration (fillRemoteCommunities,
     * dspl.hpp:497-952): sorted distinct remote community ids + their info,
     * and the matching update accumulators (remoteCupdate). */
    int64_t nrc; int64_t *rc_ids;
    int64_t *rc_size; double *rc_degree;
    int64_t *rcu_size; double *rcu_degree;
    int64_t rc_cap;
} RankState;

static int64_t bsearch_i64(const int64_t *a, int64_t n, int64_t key)
{
    int64_t lo = 0, hi = n;
    while (lo < hi) {
        int64_t mid = (lo + hi) / 2;
        if (a[mid] < key) lo = mid + 1; else hi = mid;
    }
    return (lo < n && a[lo] == key) ? lo : -1;
}

static int cmp_i64(const void *a, const void *b)
{
    int64_t x = *(const int64_t *)a, y = *(const int64_t *)b;
    return x < y ? -1 : x > y ? 1 : 0;
}

/* The per-vertex sweep (distExecuteLouvainIteration, dspl.hpp:276-405, with
 * distBuildLocalMapCounter :230-274 and distGetMaxIndex :174-228).
 * clmap/counter restated as: slot 0 = the current community's accumulator
 * (the reference seeds clmap with cc at slot 0, dspl.hpp:312-313); other
 * communities in an open-addressed hash whose per-slot sums still accumulate
 * in edge order (single pass over e0..e1). */
typedef struct { int64_t *key; double *acc; int64_t cap, mask; } CHash;

static void sweep_vertex(const OrcGraph *g, RankState *rs, int r, int64_t i,
                         double constant, RankState *all)
{
    const int64_t base = rs->base, bound = rs->bound;
    const int64_t *xadj = g->xadj[r];
    const int64_t *tails = g->tails[r];
    const double *w = g->weights[r];
    const int64_t cc = rs->currComm[i];

    /* current community info (dspl.hpp:296-307) */
    double ccDegree; int64_t ccSize;
    if (cc >= base && cc < bound) {
        ccDegree = rs->cinfo_degree[cc - base];
        ccSize = rs->cinfo_size[cc - base];
    } else {
        int64_t s = bsearch_i64(rs->rc_ids, rs->nrc, cc);
        ccDegree = rs->rc_degree[s];
        ccSize = rs->rc_size[s];
    }
    const int currLocal = (cc >= base && cc < bound);

    int64_t localTarget;
    const int64_t b0 = xadj[i], b1 = xadj[i + 1];
    int64_t maxSize = ccSize;

    if (b0 != b1) {
        /* build clmap/counter (dspl.hpp:230-274) */
        int64_t deg = b1 - b0;
        int64_t cap = 16; while (cap < 2 * (deg + 1)) cap <<= 1;
        int64_t *hk = malloc(8 * cap); double *ha = malloc(8 * cap);
        for (int64_t t = 0; t < cap; t++) hk[t] = INT64_MIN;
        double c0 = 0.0, selfLoop = 0.0;

        for (int64_t e = b0; e < b1; e++) {
            const int64_t tail = tails[e];
            const double we = w[e];
            if (tail == i + base) selfLoop += we; /* dspl.hpp:247-248 */
            int64_t ti = rs->tidx[e];
            int64_t tcomm = (ti < rs->lnv) ? rs->currComm[ti]
                                           : rs->ghost_comm[ti - rs->lnv];
            if (tcomm == cc) { c0 += we; continue; } /* counter[0] (dspl.hpp:312-318) */
            int64_t hpos = (int64_t)(((uint64_t)tcomm * 0x9E3779B97F4A7C15ull) >> 1) & (cap - 1);
            for (;;) {
                if (hk[hpos] == tcomm) { ha[hpos] += we; break; }
                if (hk[hpos] == INT64_MIN) { hk[hpos] = tcomm; ha[hpos] = we; break; }
                hpos = (hpos + 1) & (cap - 1);
            }
        }
        rs->clusterWeight[i] += c0; /* dspl.hpp:318 */

        /* distGetMaxIndex (dspl.hpp:174-228): among candidates y != cc,
         * gain = 2(eiy-eix) - 2*vDeg*(ay-ax)*constant; pick max gain > 0,
         * ties -> smallest community id (gain == 0 never wins, :214-215).
         * Scan order does not affect the result under that total order. */
        const double eix = c0 - selfLoop;
        const double ax = ccDegree - rs->vDegree[i];
        const double vdeg = rs->vDegree[i];
        double maxGain = 0.0;
        int64_t maxIndex = cc;
        for (int64_t t = 0; t < cap; t++) {
            if (hk[t] == INT64_MIN) continue;
            const int64_t y = hk[t];
            double ay; int64_t ysize;
            if (y >= base && y < bound) {
                ay = rs->cinfo_degree[y - base];
                ysize = rs->cinfo_size[y - base];
            } else {
                int64_t s = bsearch_i64(rs->rc_ids, rs->nrc, y);
                ay = rs->rc_degree[s];
                ysize = rs->rc_size[s];
            }
            const double eiy = ha[t];
            const double curGain = 2.0 * (eiy - eix) - 2.0 * vdeg * (ay - ax) * constant; /* :212 */
            if (curGain > maxGain ||
                (curGain == maxGain && curGain != 0.0 && y < maxIndex)) {
                maxGain = curGain;
                maxIndex = y;
                maxSize = ysize;
            }
        }
        /* singleton guard (dspl.hpp:224-225) */
        if (maxSize == 1 && ccSize == 1 && maxIndex > cc)
            maxIndex = cc;
        localTarget = maxIndex;
        free(hk); free(ha);
    } else {
        localTarget = cc; /* dspl.hpp:323-324 */
    }

    /* 4-case community-size/degree updates (dspl.hpp:331-399) */
    const int targetLocal = (localTarget >= base && localTarget < bound);
    if (localTarget != cc && localTarget != -1) {
        const double vdeg = rs->vDegree[i];
        if (currLocal) {
            rs->cupd_degree[cc - base] -= vdeg;
            rs->cupd_size[cc - base] -= 1;
        } else {
            int64_t s = bsearch_i64(rs->rc_ids, rs->nrc, cc);
            rs->rcu_degree[s] -= vdeg;
            rs->rcu_size[s] -= 1;
        }
        if (targetLocal) {
            rs->cupd_degree[localTarget - base] += vdeg;
            rs->cupd_size[localTarget - base] += 1;
        } else {
            int64_t s = bsearch_i64(rs->rc_ids, rs->nrc, localTarget);
            rs->rcu_degree[s] += vdeg;
            rs->rcu_size[s] += 1;
        }
    }
    rs->targetComm[i] = localTarget; /* dspl.hpp:404 */
    (void)all;
}

/* Run the full method. Returns prevMod (the reference returns the PREVIOUS
 * iteration's modularity, dspl.hpp:1440). iters_out <- iteration count.
 * If trace_target != NULL it must hold max_iters*nv int64 and receives, for
 * iteration k (1-based), targetComm of every vertex at
 * trace_target[(k-1)*nv + globalvertex]. trace_mod likewise (max_iters
 * doubles) receives each iteration's currMod. max_iters <= 0 means no cap. */
double orc_louvain(void *h, double thresh, int max_iters, int *iters_out,
                   int64_t *trace_target, double *trace_mod)
{
    OrcGraph *g = h;
    const int p = g->p;
    const int64_t nv = g->nv;
    RankState *rs = calloc(p, sizeof(RankState));

    /* distInitLouvain (dspl.hpp:151-172): vDegree/cinfo via K1
     * (distSumVertexDegree :82-107), constant via K2 (:109-130),
     * comm iota via K3 (:132-149). */
    double totalEdgeWeightTwice = 0.0;
    for (int r = 0; r < p; r++) {
        RankState *s = &rs[r];
        s->base = g->parts[r]; s->bound = g->parts[r + 1];
        s->lnv = s->bound - s->base;
        const int64_t lnv = s->lnv, lne = g->lne[r];
        s->currComm = malloc(8 * lnv); s->pastComm = malloc(8 * lnv);
        s->targetComm = malloc(8 * lnv);
        s->vDegree = malloc(8 * lnv); s->clusterWeight = malloc(8 * lnv);
        s->cinfo_size = malloc(8 * lnv); s->cinfo_degree = malloc(8 * lnv);
        s->cupd_size = malloc(8 * lnv); s->cupd_degree = malloc(8 * lnv);
        double local = 0.0;
        for (int64_t i = 0; i < lnv; i++) {
            double tw = 0.0;
            for (int64_t e = g->xadj[r][i]; e < g->xadj[r][i + 1]; e++)
                tw += g->weights[r][e];
            s->vDegree[i] = tw;
            s->cinfo_degree[i] = tw; s->cinfo_size[i] = 1; /* dspl.hpp:104-105 */
            s->currComm[i] = s->pastComm[i] = i + s->base; /* dspl.hpp:145-147 */
            local += tw;
        }
        totalEdgeWeightTwice += local; /* allreduce dspl.hpp:126 */

        /* ghost discovery + tail translation (exchangeVertexReqs
         * dspl.hpp:1140-1164; want-list order is free, we use sorted) */
        int64_t *rem = malloc(8 * (lne > 0 ? lne : 1));
        int64_t nrem = 0;
        for (int64_t e = 0; e < lne; e++) {
            int64_t t = g->tails[r][e];
            if (t < s->base || t >= s->bound) rem[nrem++] = t;
        }
        qsort(rem, nrem, sizeof(int64_t), cmp_i64);
        int64_t ng = 0;
        for (int64_t k = 0; k < nrem; k++)
            if (k == 0 || rem[k] != rem[k - 1]) rem[ng++] = rem[k];
        s->nghost = ng;
        s->ghosts = malloc(8 * (ng > 0 ? ng : 1));
        memcpy(s->ghosts, rem, 8 * ng);
        free(rem);
        s->ghost_comm = malloc(8 * (ng > 0 ? ng : 1));
        s->tidx = malloc(8 * (lne > 0 ? lne : 1));
        for (int64_t e = 0; e < lne; e++) {
            int64_t t = g->tails[r][e];
            s->tidx[e] = (t >= s->base && t < s->bound)
                       ? (t - s->base)
                       : lnv + bsearch_i64(s->ghosts, s->nghost, t);
        }
        s->rc_cap = 0; s->rc_ids = NULL; s->rc_size = NULL; s->rc_degree = NULL;
        s->rcu_size = NULL; s->rcu_degree = NULL;
    }
    const double constant = 1.0 / totalEdgeWeightTwice; /* dspl.hpp:129 */

    double prevMod = -1.0; /* lower = main.cpp:149-169 passes currMod=-1 */
    const double lower = -1.0;
    double currMod = -1.0;
    int numIters = 0;

    for (;;) {
        numIters++;
        /* ---- fillRemoteCommunities (dspl.hpp:497-952), all ranks, against
         * the iteration-start state ---- */
        for (int r = 0; r < p; r++) {
            RankState *s = &rs[r];
            /* ghost communities (halo #1a, dspl.hpp:583-647) */
            for (int64_t k = 0; k < s->nghost; k++) {
                int64_t v = s->ghosts[k];
                int o = owner_of(g, v);
                s->ghost_comm[k] = rs[o].currComm[v - g->parts[o]];
            }
            /* distinct remote communities referenced (dspl.hpp:670-700):
             * ghost communities + own currComm, keep those owned elsewhere */
            int64_t cap = s->nghost + s->lnv;
            int64_t *cand = malloc(8 * (cap > 0 ? cap : 1));
            int64_t nc = 0;
            for (int64_t k = 0; k < s->nghost; k++) {
                int64_t c = s->ghost_comm[k];
                if (c < s->base || c >= s->bound) cand[nc++] = c;
            }
            for (int64_t i = 0; i < s->lnv; i++) {
                int64_t c = s->currComm[i];
                if (c < s->base || c >= s->bound) cand[nc++] = c;
            }
            qsort(cand, nc, sizeof(int64_t), cmp_i64);
            int64_t u = 0;
            for (int64_t k = 0; k < nc; k++)
                if (k == 0 || cand[k] != cand[k - 1]) cand[u++] = cand[k];
            if (u > s->rc_cap) {
                s->rc_cap = u;
                s->rc_ids = realloc(s->rc_ids, 8 * u);
                s->rc_size = realloc(s->rc_size, 8 * u);
                s->rc_degree = realloc(s->rc_degree, 8 * u);
                s->rcu_size = realloc(s->rcu_size, 8 * u);
                s->rcu_degree = realloc(s->rcu_degree, 8 * u);
            }
            s->nrc = u;
            memcpy(s->rc_ids, cand, 8 * u);
            free(cand);
            /* owner replies with (size, degree) (dspl.hpp:776-929);
             * remoteCupdate zeroed (dspl.hpp:938-951) */
            for (int64_t k = 0; k < u; k++) {
                int64_t c = s->rc_ids[k];
                int o = owner_of(g, c);
                s->rc_size[k] = rs[o].cinfo_size[c - g->parts[o]];
                s->rc_degree[k] = rs[o].cinfo_degree[c - g->parts[o]];
                s->rcu_size[k] = 0; s->rcu_degree[k] = 0.0;
            }
        }

        /* ---- the sweep (dspl.hpp:1371-1387 -> 276-405) ---- */
        for (int r = 0; r < p; r++) {
            RankState *s = &rs[r];
            for (int64_t i = 0; i < s->lnv; i++) { /* distCleanCWandCU :473-486 */
                s->clusterWeight[i] = 0.0;
                s->cupd_size[i] = 0; s->cupd_degree[i] = 0.0;
            }
            for (int64_t i = 0; i < s->lnv; i++)
                sweep_vertex(g, s, r, i, constant, rs);
        }

        /* ---- distUpdateLocalCinfo (dspl.hpp:458-471) ---- */
        for (int r = 0; r < p; r++) {
            RankState *s = &rs[r];
            for (int64_t i = 0; i < s->lnv; i++) {
                s->cinfo_size[i] += s->cupd_size[i];
                s->cinfo_degree[i] += s->cupd_degree[i];
            }
        }
        /* ---- updateRemoteCommunities (dspl.hpp:978-1103): route each
         * rank's remote deltas to the owner; apply in sender order ---- */
        for (int r = 0; r < p; r++) {
            RankState *s = &rs[r];
            /* the reference sends every remoteCinfo key, zeros included
             * (dspl.hpp:988-1004) — keep the += 0 adds for -w bit parity */
            for (int64_t k = 0; k < s->nrc; k++) {
                int64_t c = s->rc_ids[k];
                int o = owner_of(g, c);
                rs[o].cinfo_size[c - g->parts[o]] += s->rcu_size[k];
                rs[o].cinfo_degree[c - g->parts[o]] += s->rcu_degree[k];
            }
        }

        /* ---- distComputeModularity (dspl.hpp:407-456) ---- */
        double e_xx = 0.0, a2_x = 0.0;
        for (int r = 0; r < p; r++) {
            RankState *s = &rs[r];
            double le = 0.0, la = 0.0;
            for (int64_t i = 0; i < s->lnv; i++) {
                le += s->clusterWeight[i];
                la += s->cinfo_degree[i] * s->cinfo_degree[i];
            }
            e_xx += le; a2_x += la;
        }
        currMod = fabs(e_xx * constant - a2_x * constant * constant); /* :447-448 */

        if (trace_mod) trace_mod[numIters - 1] = currMod;
        if (trace_target)
            for (int r = 0; r < p; r++)
                memcpy(trace_target + (int64_t)(numIters - 1) * nv + g->parts[r],
                       rs[r].targetComm, 8 * rs[r].lnv);

        if (currMod - prevMod < thresh) break; /* dspl.hpp:1401 */
        if (max_iters > 0 && numIters >= max_iters) break;
        prevMod = currMod;
        if (prevMod < lower) prevMod = lower; /* dspl.hpp:1404-1406 */

        for (int r = 0; r < p; r++) { /* rotate (dspl.hpp:1417-1422) */
            RankState *s = &rs[r];
            for (int64_t i = 0; i < s->lnv; i++) {
                int64_t tmp = s->pastComm[i];
                s->pastComm[i] = s->currComm[i];
                s->currComm[i] = s->targetComm[i];
                s->targetComm[i] = tmp;
            }
        }
    }

    *iters_out = numIters;
    for (int r = 0; r < p; r++) {
        RankState *s = &rs[r];
        free(s->currComm); free(s->pastComm); free(s->targetComm);
        free(s->vDegree); free(s->clusterWeight);
        free(s->cinfo_size); free(s->cinfo_degree);
        free(s->cupd_size); free(s->cupd_degree);
        free(s->ghosts); free(s->ghost_comm); free(s->tidx);
        free(s->rc_ids); free(s->rc_size); free(s->rc_degree);
        free(s->rcu_size); free(s->rcu_degree);
    }
    free(rs);
    return prevMod; /* dspl.hpp:1440 */
}
