// main_cli.cpp — mv355: drop-in CLI for the reference's main.cpp.
//
// Same getopt surface (-f -b -r -t -n -w -l -p -s, main.cpp:212-247) and
// the same result lines (main.cpp:178-196). Differences, by design:
//  * no MPI: one process drives -g N GPUs (default 1) with one host thread
//    per GPU and RCCL comms from an in-process unique id; N plays the
//    reference's nprocs role (the RGG's LCG stream splits per rank exactly
//    like the reference, so results depend on N the same way they depend
//    on the reference's mpiexec -n).
//  * -l is implied for generated graphs (the non-LCG path's
//    std::default_random_engine stream is libstdc++-specific,
//    utils.hpp:101-114); passing -n without -l warns and proceeds with
//    the LCG path.
//  * -p takes the reference semantics but a fixed seed (graph.hpp:990's
//    time(0)^getpid() is unreproducible); set MV_RAND_SEED to vary it.

#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <thread>
#include <cmath>
#include <unistd.h>
#include <vector>

#include "../../include/minivite_hip.h"

int main(int argc, char *argv[]) {
    std::string inputFileName;
    bool readBalanced = false, showGraph = false;
    bool generateGraph = false, isUnitEdgeWeight = true;
    bool randomNumberLCG = false;
    double threshold = 1.0E-6, randomEdgePercent = 0.0;
    long nvRGG = 0;
    int ngpus = 1;
    int ret;
    while ((ret = getopt(argc, argv, "f:br:t:n:wlp:sg:")) != -1) {
        switch (ret) {
        case 'f': inputFileName = optarg; break;
        case 'b': readBalanced = true; break;
        case 'r': /* ranksPerNode: MPI-IO aggregator hint, meaningless here */ break;
        case 't': threshold = atof(optarg); break;
        case 'n':
            nvRGG = atol(optarg);
            if (nvRGG > 0) generateGraph = true;
            break;
        case 'w': isUnitEdgeWeight = false; break;
        case 'l': randomNumberLCG = true; break;
        case 'p': randomEdgePercent = atof(optarg); break;
        case 's': showGraph = true; break;
        case 'g': ngpus = atoi(optarg); break;
        default:
            std::fprintf(stderr, "Option not recognized\n");
            return 1;
        }
    }
    if (argc == 1 || (!generateGraph && inputFileName.empty())) {
        std::fprintf(stderr,
                     "usage: mv355 [-n nv -l | -f file.bin [-b]] [-t thresh] "
                     "[-w] [-p pct] [-g ngpus]\n");
        return 1;
    }
    if (generateGraph && !randomNumberLCG)
        std::fprintf(stderr,
                     "mv355: -n without -l: using the LCG path anyway (the "
                     "non-LCG RNG is stdlib-specific)\n");
    const uint64_t rseed = getenv("MV_RAND_SEED")
                               ? strtoull(getenv("MV_RAND_SEED"), nullptr, 10)
                               : 7177;

    // ---- build per-rank graphs (parallel; no communication needed) ----
    auto tg0 = std::chrono::steady_clock::now();
    std::vector<mv_graph *> graphs(ngpus, nullptr);
    {
        std::vector<std::thread> th;
        for (int r = 0; r < ngpus; r++)
            th.emplace_back([&, r] {
                graphs[r] =
                    generateGraph
                        ? mv_graph_rgg(nvRGG, r, ngpus, isUnitEdgeWeight,
                                       randomEdgePercent, rseed)
                        : mv_graph_read_binary(inputFileName.c_str(), r,
                                               ngpus, readBalanced);
            });
        for (auto &t : th) t.join();
    }
    for (int r = 0; r < ngpus; r++)
        if (!graphs[r]) return 1;
    double tgen = std::chrono::duration<double>(
                      std::chrono::steady_clock::now() - tg0)
                      .count();
    if (generateGraph)
        std::printf("Time to generate distributed graph of %ld vertices (in "
                    "s): %f\n",
                    nvRGG, tgen);
    else
        std::printf("Time to read input file and create distributed graph "
                    "(in s): %f\n",
                    tgen);
    if (getenv("MV_DIST_STATS")) {
        // print_dist_stats equivalent (graph.hpp:251-286; the reference
        // gates it behind -DPRINT_DIST_STATS)
        long long sumdeg = 0, maxdeg = 0, ne = 0;
        double sum_sq = 0;
        for (int r = 0; r < ngpus; r++) {
            const long long lne = (long long)mv_graph_lne(graphs[r]);
            sumdeg += lne;
            if (lne > maxdeg) maxdeg = lne;
            sum_sq += (double)lne * (double)lne;
            ne += lne;
        }
        const double average = (double)sumdeg / ngpus;
        const double avg_sq = sum_sq / ngpus;
        const double var = avg_sq - average * average;
        std::printf("\n-------------------------------------------------------\n");
        std::printf("Graph edge distribution characteristics\n");
        std::printf("-------------------------------------------------------\n");
        std::printf("Number of vertices: %lld\n",
                    (long long)mv_graph_nv(graphs[0]));
        std::printf("Number of edges: %lld\n", ne);
        std::printf("Maximum number of edges: %lld\n", maxdeg);
        std::printf("Average number of edges: %g\n", average);
        std::printf("Expected value of X^2: %g\n", avg_sq);
        std::printf("Variance: %g\n", var);
        std::printf("Standard deviation: %g\n", sqrt(var));
        std::printf("-------------------------------------------------------\n");
    }
    if (showGraph) {
        // graph.hpp:206-248 format: "<global head> <tail> <weight>" per
        // edge, rank blocks in order
        for (int r = 0; r < ngpus; r++) {
            std::printf("###############\nProcess #%d: \n###############\n",
                        r);
            const int64_t *xadj = mv_graph_xadj(graphs[r]);
            const int64_t *tails = mv_graph_tails(graphs[r]);
            const double *wts = mv_graph_weights(graphs[r]);
            const int64_t base = mv_graph_parts(graphs[r])[r];
            const int64_t lnv = mv_graph_lnv(graphs[r]);
            for (int64_t i = 0; i < lnv; i++)
                for (int64_t e = xadj[i]; e < xadj[i + 1]; e++)
                    std::printf("%lld %lld %g\n", (long long)(i + base),
                                (long long)tails[e], wts[e]);
        }
    }

    // ---- engines + run (one thread per GPU) ----
    unsigned char cid[MV_COMM_ID_BYTES] = {0};
    if (ngpus > 1 && mv_comm_id(cid) != 0) return 1;
    std::vector<double> mods(ngpus), times(ngpus);
    std::vector<int> iters(ngpus);
    std::atomic<int> failed{0};
    {
        std::vector<std::thread> th;
        for (int r = 0; r < ngpus; r++)
            th.emplace_back([&, r] {
                mv_engine *e = mv_engine_create(r, r, ngpus,
                                                ngpus > 1 ? cid : nullptr);
                if (!e || mv_engine_load_graph(e, graphs[r]) != 0) {
                    failed = 1;
                    return;
                }
                auto t0 = std::chrono::steady_clock::now();
                int it = 0;
                mods[r] = mv_engine_run(e, -1.0, threshold, &it);
                times[r] = std::chrono::duration<double>(
                               std::chrono::steady_clock::now() - t0)
                               .count();
                iters[r] = it;
                mv_engine_destroy(e);
            });
        for (auto &t : th) t.join();
    }
    if (failed) return 2;

    double avgt = 0;
    for (int r = 0; r < ngpus; r++) avgt += times[r];
    avgt /= ngpus;

    // result block, byte-format of main.cpp:178-196
    if (!generateGraph) {
        std::printf("-------------------------------------------------------\n");
        std::printf("File: %s\n", inputFileName.c_str());
        std::printf("-------------------------------------------------------\n");
    }
    std::printf("-------------------------------------------------------\n");
    std::printf("64-bit datatype\n");
    std::printf("-------------------------------------------------------\n");
    std::printf("Average total time (in s), #Processes: %g, %d\n", avgt,
                ngpus);
    std::printf("Modularity, #Iterations: %g, %d\n", mods[0], iters[0]);
    std::printf("MODS (final modularity * average time): %g\n",
                mods[0] * avgt);
    std::printf("-------------------------------------------------------\n");

    for (auto *g : graphs) mv_graph_free(g);
    return 0;
}
