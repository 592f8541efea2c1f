// Standalone fused-BN kernel probe: per-kernel µs + effective TB/s on
// the ResNet-50 bs256 layer shapes, vs a float4 copy floor.
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 csrc/probe_bn.hip -o probe_bn
#include <cstdio>
#include <cstdlib>
#include <vector>
#include "fusedbn_kernels.h"

#define CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP err %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

using bf16 = __hip_bfloat16;

__global__ __launch_bounds__(256) void copy_floor(const float4* a, float4* b,
                                                  long long n4) {
    const long long stride = (long long)gridDim.x * 256;
    for (long long i = (long long)blockIdx.x * 256 + threadIdx.x; i < n4;
         i += stride)
        b[i] = a[i];
}

static int bn_grid(long long m, int ch_chunks) {
    long long want = m * ch_chunks;
    long long blocks = (want + BN_BLOCK - 1) / BN_BLOCK;
    if (blocks > 2048) blocks = 2048;
    long long mult = (ch_chunks + BN_BLOCK - 1) / BN_BLOCK;
    if (mult < 1) mult = 1;
    blocks = ((blocks + mult - 1) / mult) * mult;
    if (blocks < mult) blocks = mult;
    return (int)blocks;
}

int main() {
    struct Shape { long long m; int c; };
    std::vector<Shape> shapes = {
        {256LL * 112 * 112, 64}, {256LL * 56 * 56, 64},
        {256LL * 56 * 56, 256}, {256LL * 28 * 28, 512},
        {256LL * 14 * 14, 1024}, {256LL * 7 * 7, 2048}};
    hipEvent_t a, b;
    CHECK(hipEventCreate(&a));
    CHECK(hipEventCreate(&b));

    for (auto sh : shapes) {
        const long long m = sh.m;
        const int c = sh.c;
        const long long n = m * c;
        bf16 *x, *z, *y, *dy, *dx, *dz;
        unsigned char *mask;
        float *params, *partials, *sums;
        CHECK(hipMalloc(&x, n * 2));
        CHECK(hipMalloc(&z, n * 2));
        CHECK(hipMalloc(&y, n * 2));
        CHECK(hipMalloc(&dy, n * 2));
        CHECK(hipMalloc(&dx, n * 2));
        CHECK(hipMalloc(&dz, n * 2));
        CHECK(hipMalloc(&mask, n / 8 + 16));
        CHECK(hipMemset(mask, 0xaa, n / 8 + 16));
        CHECK(hipMalloc(&params, 8 * c * 4));
        CHECK(hipMemset(x, 0x3c, n * 2));  // ~0.0117 bf16 pattern
        CHECK(hipMemset(z, 0x3c, n * 2));
        CHECK(hipMemset(dy, 0x3c, n * 2));
        CHECK(hipMemset(params, 0, 8 * c * 4));
        const int ch = c / 8;
        const int grid = bn_grid(m, ch);
        CHECK(hipMalloc(&partials, (long long)grid * 2 * c * 4));
        CHECK(hipMalloc(&sums, 2 * c * 4));
        CHECK(hipMemset(sums, 0, 2 * c * 4));
        float* mean = params;
        float* rstd = params + c;   // zero-init: fine for perf probe
        float* gamma = params + 2 * c;
        float* beta = params + 3 * c;

        auto bench = [&](const char* name, double bytes, auto&& launch) {
            float best = 1e30f;
            for (int r = 0; r < 6; ++r) {
                CHECK(hipEventRecord(a));
                launch();
                CHECK(hipEventRecord(b));
                CHECK(hipEventSynchronize(b));
                float ms;
                CHECK(hipEventElapsedTime(&ms, a, b));
                if (r >= 2 && ms < best) best = ms;
            }
            printf("  %-14s %8.1f us  %6.2f TB/s\n", name, best * 1e3,
                   bytes / (best * 1e-3) / 1e12);
        };

        printf("shape m=%lld c=%d (%.0f MB/tensor) grid=%d\n", m, c,
               n * 2 / 1e6, grid);
        bench("copy_floor", n * 2 * 2.0, [&] {
            copy_floor<<<2048, 256>>>((const float4*)x, (float4*)y,
                                      n * 2 / 16); });
        bench("stats", n * 2.0, [&] {
            bn_stats_kernel<bf16><<<grid, BN_BLOCK>>>(x, m, c, partials); });
        bench("fwd(add,relu)", n * 2 * 3.0, [&] {
            bn_fwd_kernel<bf16, true, true><<<grid, BN_BLOCK>>>(
                x, z, y, mask, m, c, mean, rstd, gamma, beta); });
        bench("bwd_stats", n * 2 * 2.0 + n / 8.0, [&] {
            bn_bwd_stats_kernel<bf16, true><<<grid, BN_BLOCK>>>(
                dy, x, mask, m, c, mean, rstd, partials); });
        bench("bwd_dx", n * 2 * 4.0 + n / 8.0, [&] {
            bn_bwd_dx_kernel<bf16, true, true><<<grid, BN_BLOCK>>>(
                dy, x, mask, dx, dz, m, c, mean, rstd, gamma, sums); });
        hipFree(x); hipFree(z); hipFree(y); hipFree(dy); hipFree(dx);
        hipFree(mask);
        hipFree(dz); hipFree(params); hipFree(partials); hipFree(sums);
    }
    return 0;
}
