// harness_dump.cpp — VALIDATION-ONLY harness (test infrastructure, never shipped).
//
// Compiles against the UNMODIFIED reference sources that lie under
// /root/reference (via -I; nothing is copied into this repo) and drives the
// reference's own functions (graph.hpp GenerateRGG, dspl.hpp dist* functions)
// to dump full-precision golden fixtures:
//   - the per-rank graph arrays (edge_indices_, edge_list_) so our own
//     generators can be checked bit-for-bit,
//   - per-iteration modularity (hex-float, exact) and per-iteration
//     targetComm arrays, so the oracle restatement (ref_louvain.c) and the
//     HIP engine can be checked bit-for-bit against the real reference.
//
// The iteration loop below intentionally mirrors distLouvainMethod
// (dspl.hpp:1280-1441) call-for-call, because the reference offers no hook to
// observe its internal community arrays. Build artifacts go to oracle/_ref/
// (gitignored). Only tests/fixture-generation scripts invoke this.
//
// Usage: mpiexec -n P ./harness_dump <nv> <outdir> [w]
//   generates RGG with -l (LCG) semantics, unit weights unless "w" given,
//   threshold 1e-6 (main.cpp:69), lower = -1 (main.cpp:149-169).

#include <cassert>
#include <cstdio>
#include <cstdlib>
#include <string>
#include <unistd.h>
#include <vector>

#include <mpi.h>
#include <omp.h>

#include "graph.hpp"
#include "dspl.hpp"

static void dump_i64(FILE* f, const GraphElem* p, size_t n) {
    fwrite(p, sizeof(GraphElem), n, f);
}

int main(int argc, char** argv) {
    MPI_Init(&argc, &argv);
    int me, nprocs;
    MPI_Comm_size(MPI_COMM_WORLD, &nprocs);
    MPI_Comm_rank(MPI_COMM_WORLD, &me);

    if (argc < 3) {
        if (me == 0) fprintf(stderr, "usage: %s <nv> <outdir> [w]\n", argv[0]);
        MPI_Abort(MPI_COMM_WORLD, 1);
    }
    const GraphElem nv = atol(argv[1]);
    const std::string outdir = argv[2];
    const bool unitEdgeWeight = !(argc > 3 && argv[3][0] == 'w');
    const GraphWeight thresh = 1.0E-6;   // main.cpp:69
    const GraphWeight lower  = -1.0;     // main.cpp:149-150 (currMod passed as lower)

    createCommunityMPIType();

    GenerateRGG gr(nv);
    Graph* g = gr.generate(/*isLCG*/true, unitEdgeWeight, 0.0);

    // ---- dump the per-rank graph (bit-exact fixture for generator parity) ----
    {
        char path[4096];
        snprintf(path, sizeof(path), "%s/graph_r%d.bin", outdir.c_str(), me);
        FILE* f = fopen(path, "wb");
        GraphElem lnv = g->get_lnv(), lne = g->get_lne();
        fwrite(&lnv, 8, 1, f);
        fwrite(&lne, 8, 1, f);
        dump_i64(f, g->edge_indices_.data(), lnv + 1);
        // Edge is {int64 tail_, double weight_} = 16 B (graph.hpp:60-66)
        fwrite(g->edge_list_.data(), sizeof(Edge), lne, f);
        fclose(f);
    }

    // ---- replicate distLouvainMethod's loop (dspl.hpp:1280-1441) with dumps ----
    std::vector<GraphElem> pastComm, currComm, targetComm;
    std::vector<GraphWeight> vDegree, clusterWeight;
    std::vector<Comm> localCinfo, localCupdate;
    std::unordered_map<GraphElem, GraphElem> remoteComm;
    std::map<GraphElem, Comm> remoteCinfo, remoteCupdate;

    const GraphElem lnv = g->get_lnv();
    GraphWeight constantForSecondTerm;
    GraphWeight prevMod = lower, currMod = -1.0;
    int numIters = 0;

    distInitLouvain(*g, pastComm, currComm, vDegree, clusterWeight, localCinfo,
                    localCupdate, constantForSecondTerm, me);
    targetComm.resize(lnv);

    size_t ssz = 0, rsz = 0;
    std::vector<GraphElem> ssizes, rsizes, svdata, rvdata;
    exchangeVertexReqs(*g, ssz, rsz, ssizes, rsizes, svdata, rvdata, me, nprocs);

    char mpath[4096];
    snprintf(mpath, sizeof(mpath), "%s/trace_r%d.txt", outdir.c_str(), me);
    FILE* mf = fopen(mpath, "w");
    fprintf(mf, "constant %a\n", (double)constantForSecondTerm);

    while (true) {
        numIters++;
        fillRemoteCommunities(*g, me, nprocs, ssz, rsz, ssizes, rsizes,
                              svdata, rvdata, currComm, localCinfo,
                              remoteCinfo, remoteComm, remoteCupdate);
#pragma omp parallel default(shared)
        {
            distCleanCWandCU(lnv, clusterWeight, localCupdate);
#pragma omp for schedule(guided)
            for (GraphElem i = 0; i < lnv; i++) {
                distExecuteLouvainIteration(i, *g, currComm, targetComm, vDegree,
                                            localCinfo, localCupdate, remoteComm,
                                            remoteCinfo, remoteCupdate,
                                            constantForSecondTerm, clusterWeight, me);
            }
        }
#pragma omp parallel default(none), shared(localCinfo, localCupdate)
        { distUpdateLocalCinfo(localCinfo, localCupdate); }

        updateRemoteCommunities(*g, localCinfo, remoteCupdate, me, nprocs);
        currMod = distComputeModularity(*g, localCinfo, clusterWeight,
                                        constantForSecondTerm, me);

        // dump this iteration's targetComm + exact modularity
        {
            char path[4096];
            snprintf(path, sizeof(path), "%s/target_i%d_r%d.bin", outdir.c_str(),
                     numIters, me);
            FILE* f = fopen(path, "wb");
            dump_i64(f, targetComm.data(), lnv);
            fclose(f);
            fprintf(mf, "iter %d mod %a\n", numIters, (double)currMod);
        }

        if (currMod - prevMod < thresh) break;
        prevMod = currMod;
        if (prevMod < lower) prevMod = lower;
        for (GraphElem i = 0; i < lnv; i++) {
            GraphElem tmp = pastComm[i];
            pastComm[i] = currComm[i];
            currComm[i] = targetComm[i];
            targetComm[i] = tmp;
        }
    }

    fprintf(mf, "final mod %a iters %d\n", (double)prevMod, numIters);
    fclose(mf);

    if (me == 0)
        printf("nv=%lld ne=%lld mod=%.17g iters=%d\n", (long long)g->get_nv(),
               (long long)g->get_ne(), (double)prevMod, numIters);

    delete g;
    destroyCommunityMPIType();
    MPI_Finalize();
    return 0;
}
