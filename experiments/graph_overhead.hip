// graph_overhead.hip — standalone measurement of the p==1 per-iteration
// HOST overhead: 3 launches (sweep-shaped, K67-shaped, 32 KB D2H) + one
// host sync per iteration, plain stream vs hipGraph replay. Answers
// whether capturing the iteration into a hipGraph would pay (the
// remaining non-kernel time is ~27 us/iteration at n=2^22).
//
// Build: hipcc --offload-arch=gfx950 -O3 graph_overhead.hip -o graph_overhead
#include <chrono>
#include <cstdio>
#include <hip/hip_runtime.h>

#define HC(x)                                                                 \
    do {                                                                      \
        hipError_t e_ = (x);                                                  \
        if (e_ != hipSuccess) {                                               \
            std::fprintf(stderr, "HIP %s @%d\n", hipGetErrorString(e_),       \
                         __LINE__);                                           \
            std::exit(1);                                                     \
        }                                                                     \
    } while (0)

__global__ void busy(double *p, long n, int reps) {
    const long i = blockIdx.x * (long)blockDim.x + threadIdx.x;
    if (i >= n) return;
    double v = p[i];
    for (int r = 0; r < reps; r++) v = v * 1.0000001 + 1e-9;
    p[i] = v;
}

__global__ void partials(const double *in, double *out, long n) {
    __shared__ double s[256];
    double a = 0;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x)
        a += in[i];
    s[threadIdx.x] = a;
    __syncthreads();
    for (int t = 128; t > 0; t >>= 1) {
        if (threadIdx.x < t) s[threadIdx.x] += s[threadIdx.x + t];
        __syncthreads();
    }
    if (threadIdx.x == 0) out[blockIdx.x] = s[0];
}

int main() {
    const long n = 4 << 20;
    const int iters = 24, runs = 40;
    double *d, *dp, *hp;
    HC(hipMalloc(&d, n * 8));
    HC(hipMalloc(&dp, 2048 * 8));
    HC(hipHostMalloc(&hp, 2048 * 8));
    hipStream_t st;
    HC(hipStreamCreate(&st));

    auto iter_body = [&](hipStream_t s) {
        busy<<<2048, 256, 0, s>>>(d, n, 40);      // ~sweep-sized
        partials<<<2048, 256, 0, s>>>(d, dp, n);  // ~K67-sized
        HC(hipMemcpyAsync(hp, dp, 2048 * 8, hipMemcpyDeviceToHost, s));
    };

    auto bench = [&](const char *name, auto &&fn) {
        fn(); // warm
        auto t0 = std::chrono::steady_clock::now();
        for (int r = 0; r < runs; r++) fn();
        double ms = std::chrono::duration<double, std::milli>(
                        std::chrono::steady_clock::now() - t0)
                        .count() /
                    runs;
        std::printf("%-12s %8.3f ms/run  (%.1f us/iter)\n", name, ms,
                    ms * 1000 / iters);
        return ms;
    };

    double base = bench("stream", [&] {
        for (int k = 0; k < iters; k++) {
            iter_body(st);
            HC(hipStreamSynchronize(st));
            volatile double acc = 0;
            for (int b = 0; b < 2048; b++) acc += hp[b];
        }
    });

    // capture one iteration, replay per iteration
    hipGraph_t g;
    hipGraphExec_t ge;
    HC(hipStreamBeginCapture(st, hipStreamCaptureModeGlobal));
    iter_body(st);
    HC(hipStreamEndCapture(st, &g));
    HC(hipGraphInstantiate(&ge, g, nullptr, nullptr, 0));
    double graph = bench("hipGraph", [&] {
        for (int k = 0; k < iters; k++) {
            HC(hipGraphLaunch(ge, st));
            HC(hipStreamSynchronize(st));
            volatile double acc = 0;
            for (int b = 0; b < 2048; b++) acc += hp[b];
        }
    });
    std::printf("delta: %.3f ms/run = %.1f us/iteration (%.2f%% of the "
                "stream run)\n",
                base - graph, (base - graph) * 1000 / iters,
                100.0 * (base - graph) / base);
    return 0;
}
