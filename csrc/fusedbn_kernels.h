// Pure-HIP fused-BN kernels shared by the torch extension and the
// standalone probe (csrc/probe_bn.hip).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#ifndef BN_BLOCK
#define BN_BLOCK 256
#endif

// ---- 16-byte channel-chunk load/store -------------------------------------
template <typename T> struct VecTraits;
template <> struct VecTraits<__hip_bfloat16> {
    static constexpr int W = 8;
    using V = ushort4;  // 8 bf16 = 16 B, loaded as 4x u16x2
    __device__ static void load(const __hip_bfloat16* p, float* out) {
        const uint4 raw = *reinterpret_cast<const uint4*>(p);
        const unsigned u[4] = {raw.x, raw.y, raw.z, raw.w};
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            out[2 * i] = __bfloat162float(
                __ushort_as_bfloat16((unsigned short)(u[i] & 0xffff)));
            out[2 * i + 1] = __bfloat162float(
                __ushort_as_bfloat16((unsigned short)(u[i] >> 16)));
        }
    }
    __device__ static void store(__hip_bfloat16* p, const float* in) {
        uint4 raw;
        unsigned u[4];
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            const unsigned short lo =
                __bfloat16_as_ushort(__float2bfloat16(in[2 * i]));
            const unsigned short hi =
                __bfloat16_as_ushort(__float2bfloat16(in[2 * i + 1]));
            u[i] = (unsigned)lo | ((unsigned)hi << 16);
        }
        raw.x = u[0]; raw.y = u[1]; raw.z = u[2]; raw.w = u[3];
        *reinterpret_cast<uint4*>(p) = raw;
    }
};
template <> struct VecTraits<float> {
    static constexpr int W = 4;
    __device__ static void load(const float* p, float* out) {
        const float4 v = *reinterpret_cast<const float4*>(p);
        out[0] = v.x; out[1] = v.y; out[2] = v.z; out[3] = v.w;
    }
    __device__ static void store(float* p, const float* in) {
        *reinterpret_cast<float4*>(p) =
            make_float4(in[0], in[1], in[2], in[3]);
    }
};

// ---- stats: per-channel sum / sumsq ---------------------------------------
// Row mapping (all data kernels): each block owns a CONTIGUOUS row
// range [r_begin, r_end) — a global round-robin row stride makes every
// wave access open a fresh DRAM page (measured 0.25 TB/s); contiguous
// block ranges stream at full bandwidth.  Within a block, thread t owns
// channel-chunk t % ch_chunks and walks rows at stride BLOCK/ch_chunks
// (requires ch_chunks <= BN_BLOCK; the fp32 C=2048 case falls back to
// a global-stride loop).
#define BN_ROW_RANGE(first_stmt) \
    const long long rows_per_block = (m + gridDim.x - 1) / gridDim.x; \
    const long long r_begin = (long long)blockIdx.x * rows_per_block; \
    const long long r_end = \
        (r_begin + rows_per_block < m) ? r_begin + rows_per_block : m; \
    first_stmt

template <typename T>
__global__ __launch_bounds__(BN_BLOCK) void bn_stats_kernel(
        const T* __restrict__ x, long long m, int c,
        float* __restrict__ partials /* [gridDim][2C]; fully overwritten
                                        when ch_chunks <= BLOCK, else
                                        caller zero-inits */) {
    constexpr int W = VecTraits<T>::W;
    const int ch_chunks = c / W;
    float s[W] = {0}, q[W] = {0};
    float v[W];
    int chunk;
    if (ch_chunks <= BN_BLOCK) {
        chunk = threadIdx.x % ch_chunks;
        BN_ROW_RANGE(;)
        const int rstep = BN_BLOCK / ch_chunks;
        for (long long r = r_begin + threadIdx.x / ch_chunks; r < r_end;
             r += rstep) {
            VecTraits<T>::load(x + r * c + chunk * W, v);
            #pragma unroll
            for (int i = 0; i < W; ++i) {
                s[i] += v[i];
                q[i] += v[i] * v[i];
            }
        }
    } else {  // rare fp32 large-C fallback
        const int tid = blockIdx.x * BN_BLOCK + threadIdx.x;
        chunk = tid % ch_chunks;
        const long long rstride =
            ((long long)gridDim.x * BN_BLOCK) / ch_chunks;
        if (rstride > 0) {
            for (long long r = tid / ch_chunks; r < m; r += rstride) {
                VecTraits<T>::load(x + r * c + chunk * W, v);
                #pragma unroll
                for (int i = 0; i < W; ++i) {
                    s[i] += v[i];
                    q[i] += v[i] * v[i];
                }
            }
        }
    }
    // block tree-reduction over threads sharing a chunk, then ONE
    // non-atomic partial-row store per block (a per-channel atomicAdd
    // builds gridDim-long same-address chains — measured 100s of µs).
    float* mine = partials + (long long)blockIdx.x * 2 * c;
    const int t = threadIdx.x;
    if (ch_chunks >= BN_BLOCK) {  // unique chunk per thread in block
        #pragma unroll
        for (int i = 0; i < W; ++i) {
            mine[chunk * W + i] = s[i];
            mine[c + chunk * W + i] = q[i];
        }
        return;
    }
    __shared__ float red[BN_BLOCK];
    #pragma unroll
    for (int i = 0; i < W; ++i) {
        red[t] = s[i];
        __syncthreads();
        for (int off = BN_BLOCK >> 1; off >= ch_chunks; off >>= 1) {
            if (t < off) red[t] += red[t + off];
            __syncthreads();
        }
        if (t < ch_chunks) mine[t * W + i] = red[t];
        __syncthreads();
        red[t] = q[i];
        __syncthreads();
        for (int off = BN_BLOCK >> 1; off >= ch_chunks; off >>= 1) {
            if (t < off) red[t] += red[t + off];
            __syncthreads();
        }
        if (t < ch_chunks) mine[c + t * W + i] = red[t];
        __syncthreads();
    }
}

// ---- level-1 partial reduce: nblocks rows -> slices rows, IN PLACE -------
// Parallel over (element, block-slice): 2C alone is far too few threads
// (a 2C-thread serial loop over 2048 rows measured 474 µs — more than
// the stats pass itself).  Each (element i, slice s) thread sums rows
// {s, s+slices, …} of column i and writes the result back to
// partials[s][i]: that cell's only reader is this same thread (row s
// belongs to slice s), and its first read IS row s, so the in-place
// write is race-free and deterministic — no atomics, no zeroed
// accumulator, no extra scratch.
__global__ __launch_bounds__(BN_BLOCK) void bn_reduce_slices_kernel(
        float* __restrict__ partials, int nblocks, int c, int slices) {
    const long long idx = (long long)blockIdx.x * BN_BLOCK + threadIdx.x;
    const long long total = (long long)2 * c * slices;
    if (idx >= total) return;
    const int i = (int)(idx % (2 * c));     // consecutive threads: coalesced
    const int s = (int)(idx / (2 * c));
    float acc = 0.0f;
    for (int b = s; b < nblocks; b += slices)
        acc += partials[(long long)b * 2 * c + i];
    partials[(long long)s * 2 * c + i] = acc;
}

inline int bn_slices(int nblocks, int c) {
    // Balance the two levels' per-thread loop lengths: level 2 runs
    // BN_BLOCK/CH_PER row-slices per block, so slices = 16× that gives
    // it ≤16 iterations, and level 1 then does nblocks/slices ≤ 16
    // iterations at the 2048-block grid cap.  (The old 32768-thread
    // cap was sized for atomicAdd chain length — atomics are gone.)
    const int ch_per = c < 32 ? c : 32;
    const int s = 16 * (BN_BLOCK / ch_per);
    return nblocks < s ? nblocks : s;
}

// ---- level-2 fused reduce + finalize -------------------------------------
// Replaces the old memset(2C) + atomicAdd reduce + finalize launch trio
// (launch-latency-bound at ~30 µs per BN; ×106 BN stats ops per
// ResNet-50 step ≈ 3 ms/step of glue).  Reduces the ≤ bn_slices() rows
// left by bn_reduce_slices_kernel.  One block owns CH_PER=min(32,c)
// channels: threads form (BN_BLOCK/CH_PER) row-slices (coalesced 128 B
// groups), reduce across them in LDS, then slice 0 writes sums[ch],
// sums[c+ch] and — in FINALIZE mode (forward) — mean/rstd/running
// stats directly.  No atomics, no zeroed scratch.  Parallelism is ∝ c,
// which is why level 1 must first shrink the row count: at c=64 this
// kernel is 2 blocks, and looping it over 2048 raw partial rows was
// measured 6 ms/step SLOWER than the trio it replaced.  c is always a
// power of two here (C/W pow2 is checked at dispatch), so CH_PER
// divides c and the slice count is a power of two.
inline int bn_rf_grid(int c) {  // host-side launch grid for the kernel
    return c / (c < 32 ? c : 32);
}

template <bool FINALIZE>
__global__ __launch_bounds__(BN_BLOCK) void bn_reduce_finalize_kernel(
        const float* __restrict__ partials, int nblocks, int c,
        long long m, float eps, float momentum,
        float* __restrict__ sums, float* __restrict__ saved_mean,
        float* __restrict__ saved_rstd, float* __restrict__ running_mean,
        float* __restrict__ running_var) {
    const int ch_per = c < 32 ? c : 32;
    const int slices = BN_BLOCK / ch_per;
    const int ch_off = threadIdx.x % ch_per;
    const int slice = threadIdx.x / ch_per;
    const int ch = blockIdx.x * ch_per + ch_off;
    float s_acc = 0.0f, q_acc = 0.0f;
    for (int b = slice; b < nblocks; b += slices) {
        const float* row = partials + (long long)b * 2 * c;
        s_acc += row[ch];
        q_acc += row[c + ch];
    }
    __shared__ float red_s[BN_BLOCK];
    __shared__ float red_q[BN_BLOCK];
    red_s[threadIdx.x] = s_acc;
    red_q[threadIdx.x] = q_acc;
    __syncthreads();
    for (int off = slices >> 1; off >= 1; off >>= 1) {
        if (slice < off) {
            red_s[threadIdx.x] += red_s[threadIdx.x + off * ch_per];
            red_q[threadIdx.x] += red_q[threadIdx.x + off * ch_per];
        }
        __syncthreads();
    }
    if (slice != 0) return;
    const float s = red_s[ch_off], q = red_q[ch_off];
    sums[ch] = s;
    sums[c + ch] = q;
    if (FINALIZE) {
        const float mean = s / (float)m;
        float var = q / (float)m - mean * mean;
        if (var < 0.0f) var = 0.0f;
        saved_mean[ch] = mean;
        saved_rstd[ch] = rsqrtf(var + eps);
        if (running_mean != nullptr) {
            const float unbiased = m > 1 ? var * (float)m / (float)(m - 1)
                                         : var;
            running_mean[ch] += momentum * (mean - running_mean[ch]);
            running_var[ch] += momentum * (unbiased - running_var[ch]);
        }
    }
}

// ---- forward: normalize (+add) (+relu) ------------------------------------
// With RELU and a non-null `mask`, the kernel packs one ReLU-mask bit
// per element into mask[r*ch_chunks + chunk] (W bits per byte) so the
// backward never re-reads y (saves ~2 B/elem on two backward passes).
template <typename T, bool RELU, bool ADD>
__global__ __launch_bounds__(BN_BLOCK) void bn_fwd_kernel(
        const T* __restrict__ x, const T* __restrict__ z,
        T* __restrict__ y, unsigned char* __restrict__ mask,
        long long m, int c,
        const float* __restrict__ mean, const float* __restrict__ rstd,
        const float* __restrict__ gamma, const float* __restrict__ beta) {
    constexpr int W = VecTraits<T>::W;
    const int ch_chunks = c / W;
    long long r0, r_limit, rstep;
    int chunk;
    if (ch_chunks <= BN_BLOCK) {
        BN_ROW_RANGE(;)
        chunk = threadIdx.x % ch_chunks;
        r0 = r_begin + threadIdx.x / ch_chunks;
        r_limit = r_end;
        rstep = BN_BLOCK / ch_chunks;
    } else {
        const int tid = blockIdx.x * BN_BLOCK + threadIdx.x;
        chunk = tid % ch_chunks;
        r0 = tid / ch_chunks;
        r_limit = m;
        rstep = ((long long)gridDim.x * BN_BLOCK) / ch_chunks;
        if (rstep == 0) return;
    }
    float sc[W], sh[W];
    #pragma unroll
    for (int i = 0; i < W; ++i) {
        const int ci = chunk * W + i;
        sc[i] = gamma[ci] * rstd[ci];
        sh[i] = beta[ci] - mean[ci] * sc[i];
    }
    float v[W], zr[W];
    for (long long r = r0; r < r_limit; r += rstep) {
        VecTraits<T>::load(x + r * c + chunk * W, v);
        if (ADD) VecTraits<T>::load(z + r * c + chunk * W, zr);
        unsigned mbits = 0;
        #pragma unroll
        for (int i = 0; i < W; ++i) {
            float o = v[i] * sc[i] + sh[i];
            if (ADD) o += zr[i];
            if (RELU) {
                if (o > 0.0f) mbits |= 1u << i;
                else o = 0.0f;
            }
            v[i] = o;
        }
        VecTraits<T>::store(y + r * c + chunk * W, v);
        if (RELU && mask != nullptr)
            mask[r * ch_chunks + chunk] = (unsigned char)mbits;
    }
}

// ---- backward stats: per-channel Σdy_eff, Σdy_eff·x̂ ----------------------
template <typename T, bool RELU>
__global__ __launch_bounds__(BN_BLOCK) void bn_bwd_stats_kernel(
        const T* __restrict__ dy, const T* __restrict__ x,
        const unsigned char* __restrict__ mask, long long m, int c,
        const float* __restrict__ mean, const float* __restrict__ rstd,
        float* __restrict__ partials /* [gridDim][2C]; fully overwritten
                                        when ch_chunks <= BLOCK, else
                                        caller zero-inits */) {
    constexpr int W = VecTraits<T>::W;
    const int ch_chunks = c / W;
    long long r0, r_limit, rstep;
    int chunk;
    if (ch_chunks <= BN_BLOCK) {
        BN_ROW_RANGE(;)
        chunk = threadIdx.x % ch_chunks;
        r0 = r_begin + threadIdx.x / ch_chunks;
        r_limit = r_end;
        rstep = BN_BLOCK / ch_chunks;
    } else {
        const int tid = blockIdx.x * BN_BLOCK + threadIdx.x;
        chunk = tid % ch_chunks;
        r0 = tid / ch_chunks;
        r_limit = m;
        rstep = ((long long)gridDim.x * BN_BLOCK) / ch_chunks;
    }
    float mu[W], rs[W];
    #pragma unroll
    for (int i = 0; i < W; ++i) {
        mu[i] = mean[chunk * W + i];
        rs[i] = rstd[chunk * W + i];
    }
    float db[W] = {0}, dg[W] = {0};
    float g[W], xv[W];
    if (rstep > 0) {
        for (long long r = r0; r < r_limit; r += rstep) {
            const long long base = r * c + chunk * W;
            VecTraits<T>::load(dy + base, g);
            VecTraits<T>::load(x + base, xv);
            const unsigned mbits =
                RELU ? mask[r * ch_chunks + chunk] : 0u;
            #pragma unroll
            for (int i = 0; i < W; ++i) {
                const float d =
                    (RELU && !((mbits >> i) & 1u)) ? 0.0f : g[i];
                db[i] += d;
                dg[i] += d * (xv[i] - mu[i]) * rs[i];
            }
        }
    }
    float* mine = partials + (long long)blockIdx.x * 2 * c;
    const int t = threadIdx.x;
    if (ch_chunks >= BN_BLOCK) {
        #pragma unroll
        for (int i = 0; i < W; ++i) {
            mine[chunk * W + i] = db[i];
            mine[c + chunk * W + i] = dg[i];
        }
        return;
    }
    __shared__ float red[BN_BLOCK];
    #pragma unroll
    for (int i = 0; i < W; ++i) {
        red[t] = db[i];
        __syncthreads();
        for (int off = BN_BLOCK >> 1; off >= ch_chunks; off >>= 1) {
            if (t < off) red[t] += red[t + off];
            __syncthreads();
        }
        if (t < ch_chunks) mine[t * W + i] = red[t];
        __syncthreads();
        red[t] = dg[i];
        __syncthreads();
        for (int off = BN_BLOCK >> 1; off >= ch_chunks; off >>= 1) {
            if (t < off) red[t] += red[t + off];
            __syncthreads();
        }
        if (t < ch_chunks) mine[c + t * W + i] = red[t];
        __syncthreads();
    }
}

// ---- backward dx (+dz) ----------------------------------------------------
template <typename T, bool RELU, bool ADD>
__global__ __launch_bounds__(BN_BLOCK) void bn_bwd_dx_kernel(
        const T* __restrict__ dy, const T* __restrict__ x,
        const unsigned char* __restrict__ mask, T* __restrict__ dx,
        T* __restrict__ dz,
        long long m, int c, const float* __restrict__ mean,
        const float* __restrict__ rstd, const float* __restrict__ gamma,
        const float* __restrict__ sums /* dbeta | dgamma */) {
    constexpr int W = VecTraits<T>::W;
    const int ch_chunks = c / W;
    long long r0, r_limit, rstep;
    int chunk;
    if (ch_chunks <= BN_BLOCK) {
        BN_ROW_RANGE(;)
        chunk = threadIdx.x % ch_chunks;
        r0 = r_begin + threadIdx.x / ch_chunks;
        r_limit = r_end;
        rstep = BN_BLOCK / ch_chunks;
    } else {
        const int tid = blockIdx.x * BN_BLOCK + threadIdx.x;
        chunk = tid % ch_chunks;
        r0 = tid / ch_chunks;
        r_limit = m;
        rstep = ((long long)gridDim.x * BN_BLOCK) / ch_chunks;
        if (rstep == 0) return;
    }
    const float inv_m = 1.0f / (float)m;
    float mu[W], rs[W], gm[W], dbm[W], dgm[W];
    #pragma unroll
    for (int i = 0; i < W; ++i) {
        const int ci = chunk * W + i;
        mu[i] = mean[ci];
        rs[i] = rstd[ci];
        gm[i] = gamma[ci] * rs[i];
        dbm[i] = sums[ci] * inv_m;
        dgm[i] = sums[c + ci] * inv_m;
    }
    float g[W], xv[W], dzv[W];
    for (long long r = r0; r < r_limit; r += rstep) {
        const long long base = r * c + chunk * W;
        VecTraits<T>::load(dy + base, g);
        VecTraits<T>::load(x + base, xv);
        const unsigned mbits = RELU ? mask[r * ch_chunks + chunk] : 0u;
        #pragma unroll
        for (int i = 0; i < W; ++i) {
            const float d = (RELU && !((mbits >> i) & 1u)) ? 0.0f : g[i];
            if (ADD) dzv[i] = d;
            const float xh = (xv[i] - mu[i]) * rs[i];
            g[i] = gm[i] * (d - dbm[i] - xh * dgm[i]);
        }
        VecTraits<T>::store(dx + base, g);
        if (ADD) VecTraits<T>::store(dz + base, dzv);
    }
}

