// Standalone ablation probe for the compact (top-k emit) kernel.
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 csrc/probe_compact.hip -o probe_compact
// Run (GPU box): ./probe_compact [n] [k]
//
// Variants:
//   v0  read-only float4 scan (memory floor)
//   v1  shipped kernel: scalar loads + 2 wave_allocs per element
//   v2  scalar + wave quick-skip ballot before any alloc
//   v3  float4 + quick-skip + per-element allocs on hit
//   v4  float4 + quick-skip + block-aggregated slot allocation
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <random>

#define BLOCK 256
#define MAX_BLOCKS 2048
#define CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP err %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

__device__ __forceinline__ unsigned mag_key(float v) {
    return (__float_as_uint(v) & 0x7fffffffu) >> 18;
}
__device__ __forceinline__ unsigned mag_key2(float v) {
    return (__float_as_uint(v) >> 5) & 8191u;
}

__device__ __forceinline__ int wave_alloc(int* counter, bool pred) {
    const unsigned long long mask = __ballot(pred);
    if (!pred) return -1;
    const int lane = threadIdx.x & 63;
    const unsigned long long lt = (1ull << lane) - 1ull;
    const int rank = __popcll(mask & lt);
    int base = 0;
    const int leader = __ffsll(mask) - 1;
    if (lane == leader) base = atomicAdd(counter, __popcll(mask));
    base = __shfl(base, leader);
    return base + rank;
}

__global__ __launch_bounds__(BLOCK) void v0_scan(const float4* src, long long n4,
                                                 float* sink) {
    const long long stride = (long long)gridDim.x * BLOCK;
    float acc = 0;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < n4;
         i += stride) {
        float4 v = src[i];
        acc += v.x + v.y + v.z + v.w;
    }
    if (acc == 1234.5f) *sink = acc;  // keep live
}

template <bool QUICK>
__global__ __launch_bounds__(BLOCK) void v1_scalar(
        float* src, long long n, const int* info, int* counters,
        float* out_val, int* out_idx, int k) {
    const int T = info[0], T2 = info[2], above = info[3];
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < n;
         i += stride) {
        const float v = src[i];
        const int key = (int)mag_key(v), key2 = (int)mag_key2(v);
        const bool definite = key > T || (key == T && key2 > T2);
        const bool tie = (key == T && key2 == T2);
        if (QUICK && !__any(definite || tie)) continue;
        int slot = wave_alloc(&counters[0], definite);
        if (definite) { out_val[slot] = v; out_idx[slot] = (int)i; src[i] = 0.f; }
        slot = wave_alloc(&counters[1], tie);
        if (tie) { slot += above; if (slot < k) {
            out_val[slot] = v; out_idx[slot] = (int)i; src[i] = 0.f; } }
    }
}

__global__ __launch_bounds__(BLOCK) void v3_vec(
        float* src, long long n, const int* info, int* counters,
        float* out_val, int* out_idx, int k) {
    const int T = info[0], T2 = info[2], above = info[3];
    const long long stride = (long long)gridDim.x * BLOCK;
    const long long n4 = n >> 2;
    float4* s4 = reinterpret_cast<float4*>(src);
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < n4;
         i += stride) {
        float4 v4 = s4[i];
        float vv[4] = {v4.x, v4.y, v4.z, v4.w};
        bool any = false;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            int key = (int)mag_key(vv[j]);
            any |= key >= T;
        }
        if (!__any(any)) continue;
        bool zeroed = false;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const float v = vv[j];
            const int key = (int)mag_key(v), key2 = (int)mag_key2(v);
            const bool definite = key > T || (key == T && key2 > T2);
            const bool tie = (key == T && key2 == T2);
            int slot = wave_alloc(&counters[0], definite);
            if (definite) { out_val[slot] = v;
                out_idx[slot] = (int)(4 * i + j); vv[j] = 0.f; zeroed = true; }
            slot = wave_alloc(&counters[1], tie);
            if (tie) { slot += above; if (slot < k) { out_val[slot] = v;
                out_idx[slot] = (int)(4 * i + j); vv[j] = 0.f; zeroed = true; } }
        }
        if (zeroed) s4[i] = make_float4(vv[0], vv[1], vv[2], vv[3]);
    }
}

// v4: block-aggregated allocation.  Uniform trip count so __syncthreads
// is legal; one atomicAdd per block per iteration (if any hit).
__global__ __launch_bounds__(BLOCK) void v4_blockagg(
        float* src, long long n, const int* info, int* counters,
        float* out_val, int* out_idx, int k) {
    const int T = info[0], T2 = info[2], above = info[3];
    __shared__ int wave_cnt[BLOCK / 64];
    __shared__ int block_base;
    const long long gsz = (long long)gridDim.x * BLOCK;
    const long long n4 = n >> 2;
    const long long iters = (n4 + gsz - 1) / gsz;
    float4* s4 = reinterpret_cast<float4*>(src);
    const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
    for (long long it = 0; it < iters; ++it) {
        const long long i = it * gsz + (long long)blockIdx.x * BLOCK
                            + threadIdx.x;
        float4 v4 = i < n4 ? s4[i] : make_float4(0, 0, 0, 0);
        float vv[4] = {v4.x, v4.y, v4.z, v4.w};
        int defmask = 0, tiemask = 0;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int key = (int)mag_key(vv[j]);
            const int key2 = (int)mag_key2(vv[j]);
            if (key > T || (key == T && key2 > T2)) defmask |= 1 << j;
            else if (key == T && key2 == T2) tiemask |= 2 << (2 * j + 4);
        }
        // fast path: whole BLOCK must agree to skip (no sync hazards:
        // ballot is wave-level, combine via LDS would need sync — so
        // skip only when this wave AND all waves skip? cheaper: skip the
        // aggregation when this wave has no hits, but still hit the
        // barrier.)
        const int cnt = __popc(defmask);
        // wave total of definite hits
        unsigned long long anyb = __ballot(cnt != 0);
        int wtotal = 0;
        if (anyb) {
            // wave-wide sum of cnt via shfl butterfly (6 steps)
            int s = cnt;
            #pragma unroll
            for (int d = 32; d; d >>= 1) s += __shfl_xor(s, d);
            wtotal = s;
        }
        if (lane == 0) wave_cnt[wid] = wtotal;
        __syncthreads();
        if (threadIdx.x == 0) {
            int tot = 0;
            #pragma unroll
            for (int w = 0; w < BLOCK / 64; ++w) {
                int c = wave_cnt[w];
                wave_cnt[w] = tot;
                tot += c;
            }
            block_base = tot ? atomicAdd(&counters[0], tot) : 0;
        }
        __syncthreads();
        // rank within wave: exclusive prefix of cnt over lanes
        // (computed by ALL lanes — shfl_up in a divergent branch is UB)
        int pre = cnt;
        #pragma unroll
        for (int d = 1; d < 64; d <<= 1) {
            int o = __shfl_up(pre, d);
            if (lane >= d) pre += o;
        }
        pre -= cnt;  // exclusive
        if (cnt) {
            int slot = block_base + wave_cnt[wid] + pre;
            bool zeroed = false;
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                if (defmask & (1 << j)) {
                    out_val[slot] = vv[j];
                    out_idx[slot] = (int)(4 * i + j);
                    vv[j] = 0.f; zeroed = true; ++slot;
                }
            }
            if (zeroed) s4[i] = make_float4(vv[0], vv[1], vv[2], vv[3]);
        }
        // ties: rare — per-element wave_alloc is fine
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const bool tie = (tiemask >> (2 * j + 4)) & 2;
            int slot = wave_alloc(&counters[1], tie);
            if (tie) { slot += above; if (slot < k) {
                out_val[slot] = vv[j]; out_idx[slot] = (int)(4 * i + j); } }
        }
        __syncthreads();
    }
}

// v5: contiguous chunk per block; winners staged in LDS; ONE global
// slot-claim atomic per flush (~1 per block) instead of one per wave.
#define CAP 2048
__global__ __launch_bounds__(BLOCK) void v5_ldsbuf(
        float* src, long long n, const int* info, int* counters,
        float* out_val, int* out_idx, int k) {
    const int T = info[0], T2 = info[2], above = info[3];
    __shared__ float lval[CAP];
    __shared__ int lidx[CAP];
    __shared__ float tval[CAP];
    __shared__ int tidx[CAP];
    __shared__ int lcnt, tcnt, gbase, tgbase;
    if (threadIdx.x == 0) { lcnt = 0; tcnt = 0; }
    __syncthreads();
    const long long n4 = n >> 2;
    const long long chunk = (n4 + gridDim.x - 1) / gridDim.x;
    const long long c0 = (long long)blockIdx.x * chunk;
    const long long c1 = c0 + chunk < n4 ? c0 + chunk : n4;
    float4* s4 = reinterpret_cast<float4*>(src);
    const long long iters = (chunk + BLOCK - 1) / BLOCK;
    for (long long it = 0; it <= iters; ++it) {
        if (it < iters) {
            const long long i = c0 + it * BLOCK + threadIdx.x;
            if (i < c1) {
                float4 v4 = s4[i];
                float vv[4] = {v4.x, v4.y, v4.z, v4.w};
                bool z = false;
                #pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const float v = vv[j];
                    const int key = (int)mag_key(v);
                    const int key2 = (int)mag_key2(v);
                    if (key > T || (key == T && key2 > T2)) {
                        int p = atomicAdd(&lcnt, 1);
                        lval[p] = v;
                        lidx[p] = (int)(4 * i + j);
                        vv[j] = 0.f;
                        z = true;
                    } else if (key == T && key2 == T2) {
                        int p = atomicAdd(&tcnt, 1);
                        tval[p] = v;
                        tidx[p] = (int)(4 * i + j);
                    }
                }
                if (z) s4[i] = make_float4(vv[0], vv[1], vv[2], vv[3]);
            }
        }
        __syncthreads();
        const bool last = (it == iters);
        if ((lcnt > CAP - 4 * BLOCK) || (last && lcnt)) {
            if (threadIdx.x == 0) gbase = atomicAdd(&counters[0], lcnt);
            __syncthreads();
            for (int p = threadIdx.x; p < lcnt; p += BLOCK) {
                out_val[gbase + p] = lval[p];
                out_idx[gbase + p] = lidx[p];
            }
            __syncthreads();
            if (threadIdx.x == 0) lcnt = 0;
        }
        if ((tcnt > CAP - 4 * BLOCK) || (last && tcnt)) {
            if (threadIdx.x == 0) tgbase = above + atomicAdd(&counters[1],
                                                             tcnt);
            __syncthreads();
            for (int p = threadIdx.x; p < tcnt; p += BLOCK) {
                const int slot = tgbase + p;
                if (slot < k) {
                    out_val[slot] = tval[p];
                    out_idx[slot] = tidx[p];
                    src[tidx[p]] = 0.f;
                }
            }
            __syncthreads();
            if (threadIdx.x == 0) tcnt = 0;
        }
        __syncthreads();
    }
}

int main(int argc, char** argv) {
    long long n = argc > 1 ? atoll(argv[1]) : 25557032LL;
    int k = argc > 2 ? atoi(argv[2]) : 25557;
    n &= ~3LL;  // multiple of 4 for simplicity
    std::vector<float> h(n);
    std::mt19937 rng(0);
    std::normal_distribution<float> nd;
    for (auto& x : h) x = nd(rng);
    float *d_src, *d_val, *d_sink;
    int *d_idx, *d_info, *d_cnt;
    CHECK(hipMalloc(&d_src, n * 4));
    CHECK(hipMalloc(&d_val, (size_t)n * 4));
    CHECK(hipMalloc(&d_idx, (size_t)n * 4));
    CHECK(hipMalloc(&d_sink, 4));
    CHECK(hipMalloc(&d_info, 16));
    CHECK(hipMalloc(&d_cnt, 8));
    CHECK(hipMemcpy(d_src, h.data(), n * 4, hipMemcpyHostToDevice));

    // host-side threshold for k-th magnitude (rough): use nth_element
    std::vector<float> mags(n);
    for (long long i = 0; i < n; ++i) mags[i] = fabsf(h[i]);
    std::nth_element(mags.begin(), mags.begin() + k - 1, mags.end(),
                     [](float a, float b) { return a > b; });
    float thr = mags[k - 1];
    unsigned bits = *(unsigned*)&thr & 0x7fffffffu;
    int T = bits >> 18, T2 = (bits >> 5) & 8191;
    // count strictly above (T,T2)
    long long above = 0;
    for (long long i = 0; i < n; ++i) {
        unsigned b = *(unsigned*)&h[i] & 0x7fffffffu;
        int kk = b >> 18, kk2 = (b >> 5) & 8191;
        if (kk > T || (kk == T && kk2 > T2)) ++above;
    }
    int info[4] = {T, (int)above, T2, (int)above};
    CHECK(hipMemcpy(d_info, info, 16, hipMemcpyHostToDevice));
    printf("n=%lld k=%d T=%d T2=%d above=%lld\n", n, k, T, T2, above);

    int blocks4 = (int)std::min((long long)MAX_BLOCKS,
                                (n / 4 + BLOCK - 1) / BLOCK);
    int blocks1 = (int)std::min((long long)MAX_BLOCKS,
                                (n + BLOCK - 1) / BLOCK);
    hipEvent_t a, b;
    CHECK(hipEventCreate(&a));
    CHECK(hipEventCreate(&b));

    auto bench = [&](const char* name, auto&& launch) {
        // restore src each round (kernels zero winners)
        float best = 1e30f;
        for (int r = 0; r < 7; ++r) {
            CHECK(hipMemcpy(d_src, h.data(), n * 4, hipMemcpyHostToDevice));
            CHECK(hipMemset(d_cnt, 0, 8));
            CHECK(hipEventRecord(a));
            launch();
            CHECK(hipEventRecord(b));
            CHECK(hipEventSynchronize(b));
            float ms;
            CHECK(hipEventElapsedTime(&ms, a, b));
            if (r >= 2 && ms < best) best = ms;
        }
        int cnt[2];
        CHECK(hipMemcpy(cnt, d_cnt, 8, hipMemcpyDeviceToHost));
        printf("%-12s %8.1f us   (def=%d tie_claims=%d)\n", name,
               best * 1000.0f, cnt[0], cnt[1]);
    };

    bench("v0_floor", [&] {
        v0_scan<<<blocks4, BLOCK>>>((float4*)d_src, n / 4, d_sink); });
    bench("v1_shipped", [&] {
        v1_scalar<false><<<blocks1, BLOCK>>>(d_src, n, d_info, d_cnt,
                                             d_val, d_idx, k); });
    bench("v2_quick", [&] {
        v1_scalar<true><<<blocks1, BLOCK>>>(d_src, n, d_info, d_cnt,
                                            d_val, d_idx, k); });
    bench("v3_vec4", [&] {
        v3_vec<<<blocks4, BLOCK>>>(d_src, n, d_info, d_cnt,
                                   d_val, d_idx, k); });
    bench("v4_blockagg", [&] {
        v4_blockagg<<<blocks4, BLOCK>>>(d_src, n, d_info, d_cnt,
                                        d_val, d_idx, k); });
    bench("v5_ldsbuf", [&] {
        v5_ldsbuf<<<blocks4, BLOCK>>>(d_src, n, d_info, d_cnt,
                                      d_val, d_idx, k); });
    // correctness spot-check for v5: def count exact, outputs unique
    {
        CHECK(hipMemcpy(d_src, h.data(), n * 4, hipMemcpyHostToDevice));
        CHECK(hipMemset(d_cnt, 0, 8));
        v5_ldsbuf<<<blocks4, BLOCK>>>(d_src, n, d_info, d_cnt, d_val,
                                      d_idx, k);
        std::vector<int> idx(k);
        CHECK(hipMemcpy(idx.data(), d_idx, (size_t)k * 4,
                        hipMemcpyDeviceToHost));
        std::vector<char> seen(n, 0);
        long long dup = 0, below = 0;
        for (int key : idx) {
            if (seen[key]) ++dup;
            seen[key] = 1;
            float av = fabsf(h[key]);
            if (av < thr) ++below;
        }
        printf("v5 check: dup=%lld below_thr=%lld (expect 0, 0)\n",
               dup, below);
    }
    return 0;
}
