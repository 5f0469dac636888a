// Standalone radix-sort A/B timer: sorts (slot,idx) pairs like the FFAT
// chain (15-bit keys, implicit iota) and prints hipEvent microseconds.
// Built at several -DRS8_IPT values by tools/sort_ab.sh.
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

#include "../hip/wfa_kernels.h"

#define CHK(x)                                          \
    do {                                                \
        hipError_t e = (x);                             \
        if (e != hipSuccess) {                          \
            fprintf(stderr, "HIP error %d at %d\n", e, __LINE__); \
            exit(1);                                    \
        }                                               \
    } while (0)

int main(int argc, char** argv) {
    int64_t n = argc > 1 ? atoll(argv[1]) : 8388608;
    int bits = argc > 2 ? atoi(argv[2]) : 15;
    int iters = 30;
    uint32_t *ka, *va, *kt, *vt, *hist;
    CHK(hipMalloc(&ka, 4 * n));
    CHK(hipMalloc(&va, 4 * n));
    CHK(hipMalloc(&kt, 4 * n));
    CHK(hipMalloc(&vt, 4 * n));
    CHK(hipMalloc(&hist, 4 * wfa_sort_hist_u32(n)));
    std::vector<uint32_t> h(n);
    uint64_t x = 12345;
    for (int64_t i = 0; i < n; ++i) {
        x = x * 6364136223846793005ULL + 1442695040888963407ULL;
        h[i] = (uint32_t)((x >> 33) & ((1u << bits) - 1)) << 16;  // VIK layout
    }
    std::vector<uint32_t> keys = h;
    hipEvent_t e0, e1;
    CHK(hipEventCreate(&e0));
    CHK(hipEventCreate(&e1));
    double acc = 0;
    for (int it = -3; it < iters; ++it) {
        CHK(hipMemcpy(ka, keys.data(), 4 * n, hipMemcpyHostToDevice));
        uint32_t *ok, *ov;
        CHK(hipEventRecord(e0, 0));
        wfa_sort_pairs2(nullptr, ka, va, kt, vt, nullptr, nullptr, hist, n,
                        bits, &ok, &ov, nullptr, /*iota*/ 1, /*base_shift*/ 16);
        CHK(hipEventRecord(e1, 0));
        CHK(hipDeviceSynchronize());
        float ms;
        CHK(hipEventElapsedTime(&ms, e0, e1));
        if (it >= 0) acc += ms * 1000.0;
    }
    printf("n=%lld bits=%d RS8_IPT=%d: %.1f us\n", (long long)n, bits, RS8_IPT,
           acc / iters);
    return 0;
}
