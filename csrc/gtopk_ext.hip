// gtopkssgd_amd._C — hand-written HIP/CDNA4 (gfx950) kernels for the
// gTop-k S-SGD hot path.  SURVEY.md §2.2 K1–K8.
//
// Design notes (MI355X-first, see /opt/skills/guides/cdna_hip_programming.md):
//  * wave64 everywhere; __ballot is 64-bit; blocks are 256 threads.
//  * All streaming kernels are grid-stride, float4-vectorized, capped at
//    2048 blocks (Guideline 11).
//  * Top-k select is a TWO-LEVEL bucketed threshold select: one fused
//    pass computes `residual += grad` AND an 8192-bin LDS histogram of
//    the |value| bit-key (bits 30..18); a single-block scan finds the
//    threshold bucket; a refinement histogram over the threshold bucket
//    (bits 17..5) makes the select exact on the top 26 magnitude bits;
//    a final pass compacts winners through per-block LDS staging (one
//    slot-claim atomic per chunk) and zeroes them in the residual
//    (error feedback).  Always exactly k outputs; 26-bit ties are
//    arbitrary, like torch.topk's.
//  * fp32 global atomicAdd/atomicExch are native on CDNA4; scatter sizes
//    here are O(k) ≈ 25k elements, contention-free in practice.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

#define HIST_BITS 13
#define HIST_BINS (1 << HIST_BITS)   // 8192
#define KEY_SHIFT (31 - HIST_BITS)   // 18
#define BLOCK 256
#define MAX_BLOCKS 2048

static inline int stream_blocks(long long work_items) {
    long long b = (work_items + BLOCK - 1) / BLOCK;
    if (b > MAX_BLOCKS) b = MAX_BLOCKS;
    if (b < 1) b = 1;
    return (int)b;
}

__device__ __forceinline__ unsigned mag_key(float v) {
    unsigned bits = __float_as_uint(v) & 0x7fffffffu;
    return bits >> KEY_SHIFT;
}

// second-level key: the next 13 magnitude bits (bits 17..5) — the
// two-level select is exact on the top 26 bits of |v|.
__device__ __forceinline__ unsigned mag_key2(float v) {
    unsigned bits = __float_as_uint(v) & 0x7fffffffu;
    return (bits >> 5) & (HIST_BINS - 1);
}

// ---------------------------------------------------------------------------
// K1+K2 pass 1: residual += grad, histogram of |residual| keys
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void ef_accum_hist_kernel(
        float* __restrict__ residual, const float* __restrict__ grad,
        long long n, unsigned* __restrict__ ghist) {
    __shared__ unsigned lhist[HIST_BINS];   // 32 KiB LDS
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) lhist[j] = 0;
    __syncthreads();

    const long long stride = (long long)gridDim.x * BLOCK;
    const long long tid = (long long)blockIdx.x * BLOCK + threadIdx.x;
    const long long n4 = n >> 2;
    float4* r4 = reinterpret_cast<float4*>(residual);
    const float4* g4 = reinterpret_cast<const float4*>(grad);
    for (long long i = tid; i < n4; i += stride) {
        float4 r = r4[i];
        const float4 g = g4[i];
        r.x += g.x; r.y += g.y; r.z += g.z; r.w += g.w;
        r4[i] = r;
        atomicAdd(&lhist[mag_key(r.x)], 1u);
        atomicAdd(&lhist[mag_key(r.y)], 1u);
        atomicAdd(&lhist[mag_key(r.z)], 1u);
        atomicAdd(&lhist[mag_key(r.w)], 1u);
    }
    for (long long i = (n4 << 2) + tid; i < n; i += stride) {
        float r = residual[i] + grad[i];
        residual[i] = r;
        atomicAdd(&lhist[mag_key(r)], 1u);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) {
        unsigned c = lhist[j];
        if (c) atomicAdd(&ghist[j], c);
    }
}

// Range variant of ef_accum_hist for backward-overlap: residual[s:e) +=
// grad[s:e) and level-1 histogram of the updated values.  Buckets are
// cut at parameter boundaries (arbitrary offsets), so the float4 body
// is bracketed by scalar head/tail to honour 16 B alignment.  Additive
// across disjoint ranges: the union of bucket calls over [0, n) equals
// one ef_accum_hist_kernel call.
__global__ __launch_bounds__(BLOCK) void ef_accum_hist_range_kernel(
        float* __restrict__ residual, const float* __restrict__ grad,
        long long start, long long end, unsigned* __restrict__ ghist) {
    __shared__ unsigned lhist[HIST_BINS];
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) lhist[j] = 0;
    __syncthreads();
    long long astart = (start + 3) & ~3LL;   // first float4-aligned elem
    long long aend = end & ~3LL;             // end of float4 body
    if (aend < astart) { astart = end; aend = end; }  // tiny range
    const long long stride = (long long)gridDim.x * BLOCK;
    const long long tid = (long long)blockIdx.x * BLOCK + threadIdx.x;
    for (long long i = start + tid; i < astart; i += stride) {
        const float r = residual[i] + grad[i];
        residual[i] = r;
        atomicAdd(&lhist[mag_key(r)], 1u);
    }
    for (long long i = aend + tid; i < end; i += stride) {
        const float r = residual[i] + grad[i];
        residual[i] = r;
        atomicAdd(&lhist[mag_key(r)], 1u);
    }
    float4* r4 = reinterpret_cast<float4*>(residual);
    const float4* g4 = reinterpret_cast<const float4*>(grad);
    for (long long i = (astart >> 2) + tid; i < (aend >> 2); i += stride) {
        float4 r = r4[i];
        const float4 g = g4[i];
        r.x += g.x; r.y += g.y; r.z += g.z; r.w += g.w;
        r4[i] = r;
        atomicAdd(&lhist[mag_key(r.x)], 1u);
        atomicAdd(&lhist[mag_key(r.y)], 1u);
        atomicAdd(&lhist[mag_key(r.z)], 1u);
        atomicAdd(&lhist[mag_key(r.w)], 1u);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) {
        unsigned c = lhist[j];
        if (c) atomicAdd(&ghist[j], c);
    }
}

// histogram-only variant for topk_among (candidate values)
__global__ __launch_bounds__(BLOCK) void hist_vals_kernel(
        const float* __restrict__ vals, long long n,
        unsigned* __restrict__ ghist) {
    __shared__ unsigned lhist[HIST_BINS];
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) lhist[j] = 0;
    __syncthreads();
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < n;
         i += stride)
        atomicAdd(&lhist[mag_key(vals[i])], 1u);
    __syncthreads();
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) {
        unsigned c = lhist[j];
        if (c) atomicAdd(&ghist[j], c);
    }
}

// ---------------------------------------------------------------------------
// pass 2 (single block): find threshold bin T with
//   count(bins > T) < k <= count(bins >= T).
// LEVEL=0: info = [T, count_above]; LEVEL=1 (within bin T): target is
//   k - info[1]; writes info[2] = T2, info[3] = info[1] + count_above2
//   and zeroes the compact counters.
// ---------------------------------------------------------------------------
template <int LEVEL>
__global__ __launch_bounds__(BLOCK) void scan_threshold_kernel(
        const unsigned* __restrict__ hist, int k, int* __restrict__ info,
        int* __restrict__ counters) {
    __shared__ unsigned chunk_prefix[BLOCK];
    const int t = threadIdx.x;
    const int chunk = HIST_BINS / BLOCK;        // 32 bins per thread
    const int base_hi = HIST_BINS - 1 - t * chunk;
    const int target = (LEVEL == 0) ? k : (k - info[1]);
    unsigned sum = 0;
    for (int j = 0; j < chunk; ++j) sum += hist[base_hi - j];
    chunk_prefix[t] = sum;
    __syncthreads();
    if (t == 0) {                               // serial 256-add scan
        unsigned acc = 0;
        for (int c = 0; c < BLOCK; ++c) {
            unsigned s = chunk_prefix[c];
            chunk_prefix[c] = acc;
            acc += s;
        }
        if (LEVEL == 1) {
            counters[0] = 0;
            counters[1] = 0;
        }
    }
    __syncthreads();
    unsigned cum = chunk_prefix[t];             // count strictly above chunk
    for (int j = 0; j < chunk; ++j) {
        const unsigned h = hist[base_hi - j];
        if (cum < (unsigned)target && cum + h >= (unsigned)target) {
            if (LEVEL == 0) {                   // unique writer
                info[0] = base_hi - j;
                info[1] = (int)cum;
            } else {
                info[2] = base_hi - j;
                info[3] = info[1] + (int)cum;
            }
        }
        cum += h;
    }
}

// level-2 histogram over elements whose level-1 key == T
__global__ __launch_bounds__(BLOCK) void refine_hist_kernel(
        const float* __restrict__ src, long long n,
        const int* __restrict__ info, unsigned* __restrict__ ghist2) {
    __shared__ unsigned lhist[HIST_BINS];
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) lhist[j] = 0;
    __syncthreads();
    const int T = info[0];
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < n;
         i += stride) {
        const float v = src[i];
        if ((int)mag_key(v) == T) atomicAdd(&lhist[mag_key2(v)], 1u);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) {
        unsigned c = lhist[j];
        if (c) atomicAdd(&ghist2[j], c);
    }
}

// ---------------------------------------------------------------------------
// pass 3: compact winners.  Each block owns a contiguous chunk; winners
// are staged in LDS and flushed with ONE global slot-claim atomic per
// flush (~1 per block).  The earlier per-wave ballot allocator
// serialized ~k/64 atomicAdds on one address (~10 ns each — measured
// 290 µs at n=25.5M in csrc/probe_compact.hip); this design measures
// ~25 µs.  Definite winners (above the 26-bit threshold) go to slots
// [0, above); 26-bit ties fill [above, k).
// ---------------------------------------------------------------------------
#define COMPACT_CAP 2048  // LDS entries per class (2 × 2048 × 8 B = 32 KiB)

template <bool ZERO_SRC, bool HAS_IDX>
__global__ __launch_bounds__(BLOCK) void compact_kernel(
        float* __restrict__ src, const int* __restrict__ src_idx,
        long long n, const int* __restrict__ info, int* __restrict__ counters,
        float* __restrict__ out_val, int* __restrict__ out_idx, int k) {
    const int T = info[0];
    const int T2 = info[2];
    const int above = info[3];   // count with (key,key2) strictly above (T,T2)
    __shared__ float lval[COMPACT_CAP];
    __shared__ int lidx[COMPACT_CAP];
    __shared__ float tval[COMPACT_CAP];
    __shared__ int tidx[COMPACT_CAP];
    __shared__ int lcnt, tcnt, gbase, tgbase;
    if (threadIdx.x == 0) { lcnt = 0; tcnt = 0; }
    __syncthreads();
    const long long n4 = n >> 2;
    const long long chunk = (n4 + gridDim.x - 1) / gridDim.x;
    const long long c0 = (long long)blockIdx.x * chunk;
    const long long c1 = c0 + chunk < n4 ? c0 + chunk : n4;
    float4* s4 = reinterpret_cast<float4*>(src);
    const int4* i4 = reinterpret_cast<const int4*>(src_idx);
    const long long iters = (chunk + BLOCK - 1) / BLOCK;
    for (long long it = 0; it <= iters; ++it) {
        if (it < iters) {
            const long long i = c0 + it * BLOCK + threadIdx.x;
            if (i < c1) {
                float4 v4 = s4[i];
                int4 x4;
                if (HAS_IDX) x4 = i4[i];
                float vv[4] = {v4.x, v4.y, v4.z, v4.w};
                int xx[4];
                if (HAS_IDX) {
                    xx[0] = x4.x; xx[1] = x4.y; xx[2] = x4.z; xx[3] = x4.w;
                } else {
                    xx[0] = (int)(4 * i); xx[1] = xx[0] + 1;
                    xx[2] = xx[0] + 2; xx[3] = xx[0] + 3;
                }
                bool z = false;
                #pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const float v = vv[j];
                    const int key = (int)mag_key(v);
                    const int key2 = (int)mag_key2(v);
                    if (key > T || (key == T && key2 > T2)) {
                        const int p = atomicAdd(&lcnt, 1);
                        lval[p] = v;
                        lidx[p] = xx[j];
                        if (ZERO_SRC) { vv[j] = 0.0f; z = true; }
                    } else if (key == T && key2 == T2) {
                        const int p = atomicAdd(&tcnt, 1);
                        tval[p] = v;
                        // tie zeroing needs the POSITION, which for the
                        // HAS_IDX (candidate-list) case isn't used
                        // (ZERO_SRC is false there)
                        tidx[p] = xx[j];
                    }
                }
                if (ZERO_SRC && z)
                    s4[i] = make_float4(vv[0], vv[1], vv[2], vv[3]);
            }
        } else if (blockIdx.x == gridDim.x - 1 &&
                   threadIdx.x < (int)(n - (n4 << 2))) {
            // scalar tail (n % 4 elements), handled by the last block
            const long long i = (n4 << 2) + threadIdx.x;
            const float v = src[i];
            const int key = (int)mag_key(v);
            const int key2 = (int)mag_key2(v);
            const int xi = HAS_IDX ? src_idx[i] : (int)i;
            if (key > T || (key == T && key2 > T2)) {
                const int p = atomicAdd(&lcnt, 1);
                lval[p] = v;
                lidx[p] = xi;
                if (ZERO_SRC) src[i] = 0.0f;
            } else if (key == T && key2 == T2) {
                const int p = atomicAdd(&tcnt, 1);
                tval[p] = v;
                tidx[p] = ZERO_SRC ? (int)i : xi;
            }
        }
        __syncthreads();
        const bool last = (it == iters);
        if (lcnt > COMPACT_CAP - 4 * BLOCK || (last && lcnt)) {
            if (threadIdx.x == 0) gbase = atomicAdd(&counters[0], lcnt);
            __syncthreads();
            for (int p = threadIdx.x; p < lcnt; p += BLOCK) {
                out_val[gbase + p] = lval[p];
                out_idx[gbase + p] = lidx[p];
            }
            __syncthreads();
            if (threadIdx.x == 0) lcnt = 0;
        }
        if (tcnt > COMPACT_CAP - 4 * BLOCK || (last && tcnt)) {
            if (threadIdx.x == 0)
                tgbase = above + atomicAdd(&counters[1], tcnt);
            __syncthreads();
            for (int p = threadIdx.x; p < tcnt; p += BLOCK) {
                const int slot = tgbase + p;
                if (slot < k) {
                    out_val[slot] = tval[p];
                    out_idx[slot] = tidx[p];
                    if (ZERO_SRC) src[tidx[p]] = 0.0f;
                }
            }
            __syncthreads();
            if (threadIdx.x == 0) tcnt = 0;
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Fused merge-round kernels (C1 bookkeeping, K4+K5+K8 batched).
// A merge round on the receiver is:  dense[own ∪ recv] += vals;
// candidates = gather-and-zero(dense, own ∪ recv idx); top-k of the
// candidates.  The wire format packs idx at buf[0:k] and fp32 values
// bitcast at buf[k:2k], so both kernels take the raw 2k-int32 buffers.
// Messages are ~2k ≈ 51k items: latency-bound, so the point is FEWER
// LAUNCHES per round (7 total incl. select), not per-kernel bandwidth.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void merge_scatter2_kernel(
        float* __restrict__ dense, const int* __restrict__ abuf,
        const int* __restrict__ bbuf, int k) {
    const long long stride = (long long)gridDim.x * BLOCK;
    const float* aval = reinterpret_cast<const float*>(abuf + k);
    const float* bval = reinterpret_cast<const float*>(bbuf + k);
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < 2 * k;
         i += stride) {
        if (i < k) atomicAdd(&dense[abuf[i]], aval[i]);
        else       atomicAdd(&dense[bbuf[i - k]], bval[i - k]);
    }
}

// gather-and-zero the 2k candidates, write the (val, idx) candidate
// arrays AND the level-1 histogram in the same pass
__global__ __launch_bounds__(BLOCK) void merge_gather_hist_kernel(
        float* __restrict__ dense, const int* __restrict__ abuf,
        const int* __restrict__ bbuf, int k, float* __restrict__ cand_val,
        int* __restrict__ cand_idx, unsigned* __restrict__ ghist) {
    __shared__ unsigned lhist[HIST_BINS];
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) lhist[j] = 0;
    __syncthreads();
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < 2 * k;
         i += stride) {
        const int idx = i < k ? abuf[i] : bbuf[i - k];
        const float v = atomicExch(&dense[idx], 0.0f);  // dedup: dup reads 0
        cand_val[i] = v;
        cand_idx[i] = idx;
        atomicAdd(&lhist[mag_key(v)], 1u);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < HIST_BINS; j += BLOCK) {
        unsigned c = lhist[j];
        if (c) atomicAdd(&ghist[j], c);
    }
}

// ---------------------------------------------------------------------------
// K4: scatter-add / gather-zero (merge round primitives)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void scatter_add_kernel(
        float* __restrict__ dense, const int* __restrict__ idx,
        const float* __restrict__ val, long long m) {
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < m;
         i += stride)
        atomicAdd(&dense[idx[i]], val[i]);
}

__global__ __launch_bounds__(BLOCK) void gather_zero_kernel(
        float* __restrict__ dense, const int* __restrict__ idx,
        float* __restrict__ out, long long m) {
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < m;
         i += stride)
        out[i] = atomicExch(&dense[idx[i]], 0.0f);  // dedup: dup reads 0
}

// ---------------------------------------------------------------------------
// K6: rejected-value re-insertion (marks: byte per element, reused)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void mark_set_kernel(
        unsigned char* __restrict__ marks, const int* __restrict__ idx,
        long long m, unsigned char value) {
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < m;
         i += stride)
        marks[idx[i]] = value;
}

__global__ __launch_bounds__(BLOCK) void readd_kernel(
        float* __restrict__ residual, const int* __restrict__ lidx,
        const float* __restrict__ lval, long long m,
        const unsigned char* __restrict__ marks) {
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < m;
         i += stride) {
        const int j = lidx[i];
        if (!marks[j]) residual[j] += lval[i];  // lidx unique: no atomics
    }
}

// ---------------------------------------------------------------------------
// densify: dense[idx] = val * scale, skipping zeros (dedup artifacts)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void densify_kernel(
        float* __restrict__ dense, const int* __restrict__ idx,
        const float* __restrict__ val, long long m, float scale) {
    const long long stride = (long long)gridDim.x * BLOCK;
    for (long long i = (long long)blockIdx.x * BLOCK + threadIdx.x; i < m;
         i += stride) {
        const float v = val[i];
        if (v != 0.0f) dense[idx[i]] = v * scale;
    }
}

// ---------------------------------------------------------------------------
// K7: fused flat SGD (momentum + weight decay + optional nesterov),
// reading the dense grad scratch and zeroing it in the same pass.
// Memory-bound: float4-vectorized, ~24 B/element of traffic.
// ---------------------------------------------------------------------------
__device__ __forceinline__ float sgd_one(float p, float& mbuf, float g,
                                         float lr, float mu, float wd,
                                         float damp, bool nesterov,
                                         float gscale) {
    float gg = g * gscale + wd * p;
    if (mu != 0.0f) {
        mbuf = mu * mbuf + (1.0f - damp) * gg;
        gg = nesterov ? (gg + mu * mbuf) : mbuf;
    }
    return p - lr * gg;
}

__global__ __launch_bounds__(BLOCK) void sgd_kernel(
        float* __restrict__ param, float* __restrict__ mom,
        float* __restrict__ grad, long long n, float lr, float mu, float wd,
        float damp, int nesterov, int zero_grad, float gscale) {
    const long long stride = (long long)gridDim.x * BLOCK;
    const long long tid = (long long)blockIdx.x * BLOCK + threadIdx.x;
    const long long n4 = n >> 2;
    float4* p4 = reinterpret_cast<float4*>(param);
    float4* m4 = reinterpret_cast<float4*>(mom);
    float4* g4 = reinterpret_cast<float4*>(grad);
    const bool nest = nesterov != 0;
    for (long long i = tid; i < n4; i += stride) {
        float4 p = p4[i], m = m4[i];
        const float4 g = g4[i];
        p.x = sgd_one(p.x, m.x, g.x, lr, mu, wd, damp, nest, gscale);
        p.y = sgd_one(p.y, m.y, g.y, lr, mu, wd, damp, nest, gscale);
        p.z = sgd_one(p.z, m.z, g.z, lr, mu, wd, damp, nest, gscale);
        p.w = sgd_one(p.w, m.w, g.w, lr, mu, wd, damp, nest, gscale);
        p4[i] = p;
        m4[i] = m;
        if (zero_grad) g4[i] = make_float4(0.f, 0.f, 0.f, 0.f);
    }
    for (long long i = (n4 << 2) + tid; i < n; i += stride) {
        float m = mom[i];
        param[i] = sgd_one(param[i], m, grad[i], lr, mu, wd, damp, nest,
                           gscale);
        mom[i] = m;
        if (zero_grad) grad[i] = 0.0f;
    }
}

// ===========================================================================
// Host wrappers
// ===========================================================================
#define CHECK_F32(t) TORCH_CHECK((t).is_cuda() && (t).is_contiguous() && \
    (t).scalar_type() == at::kFloat, #t " must be contiguous fp32 CUDA")
#define CHECK_I32(t) TORCH_CHECK((t).is_cuda() && (t).is_contiguous() && \
    (t).scalar_type() == at::kInt, #t " must be contiguous int32 CUDA")

static hipStream_t cur_stream() {
    return at::cuda::getCurrentCUDAStream().stream();
}

#define HIP_OK(expr) do { hipError_t _e = (expr); \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
} while (0)

static void check_launch(const char* what) {
    HIP_OK(hipGetLastError());
    (void)what;
}

void ef_compress(at::Tensor residual, at::Tensor grad, int64_t k,
                 at::Tensor hist, at::Tensor info, at::Tensor counters,
                 at::Tensor out_val, at::Tensor out_idx) {
    CHECK_F32(residual); CHECK_F32(grad); CHECK_F32(out_val);
    CHECK_I32(hist); CHECK_I32(info); CHECK_I32(counters); CHECK_I32(out_idx);
    const long long n = residual.numel();
    TORCH_CHECK(grad.numel() == n && k > 0 && k <= n);
    TORCH_CHECK(hist.numel() >= 2 * HIST_BINS && out_val.numel() >= k &&
                out_idx.numel() >= k);
    auto stream = cur_stream();
    unsigned* h1 = (unsigned*)hist.data_ptr<int>();
    unsigned* h2 = h1 + HIST_BINS;
    HIP_OK(hipMemsetAsync(h1, 0, 2 * HIST_BINS * sizeof(int), stream));
    const int blocks = stream_blocks((n + 3) / 4);
    ef_accum_hist_kernel<<<blocks, BLOCK, 0, stream>>>(
        residual.data_ptr<float>(), grad.data_ptr<float>(), n, h1);
    scan_threshold_kernel<0><<<1, BLOCK, 0, stream>>>(
        h1, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    refine_hist_kernel<<<stream_blocks(n), BLOCK, 0, stream>>>(
        residual.data_ptr<float>(), n, info.data_ptr<int>(), h2);
    scan_threshold_kernel<1><<<1, BLOCK, 0, stream>>>(
        h2, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    compact_kernel<true, false><<<stream_blocks(n), BLOCK, 0, stream>>>(
        residual.data_ptr<float>(), nullptr, n, info.data_ptr<int>(),
        counters.data_ptr<int>(), out_val.data_ptr<float>(),
        out_idx.data_ptr<int>(), (int)k);
    check_launch("gtopk");
}

// --- backward-overlap select: zero hist, accumulate per-bucket ranges
// during backward, finish (scan+refine+compact) at synchronize() ---

void ef_hist_zero(at::Tensor hist) {
    CHECK_I32(hist);
    TORCH_CHECK(hist.numel() >= 2 * HIST_BINS);
    HIP_OK(hipMemsetAsync(hist.data_ptr<int>(), 0,
                          2 * HIST_BINS * sizeof(int), cur_stream()));
}

void ef_accum_hist_range(at::Tensor residual, at::Tensor grad,
                         int64_t start, int64_t end, at::Tensor hist) {
    CHECK_F32(residual); CHECK_F32(grad); CHECK_I32(hist);
    const long long n = residual.numel();
    TORCH_CHECK(grad.numel() == n && 0 <= start && start <= end && end <= n);
    if (start == end) return;
    const int blocks = stream_blocks((end - start + 3) / 4);
    ef_accum_hist_range_kernel<<<blocks, BLOCK, 0, cur_stream()>>>(
        residual.data_ptr<float>(), grad.data_ptr<float>(), start, end,
        (unsigned*)hist.data_ptr<int>());
    check_launch("gtopk");
}

void ef_compress_finish(at::Tensor residual, int64_t k, at::Tensor hist,
                        at::Tensor info, at::Tensor counters,
                        at::Tensor out_val, at::Tensor out_idx) {
    CHECK_F32(residual); CHECK_F32(out_val);
    CHECK_I32(hist); CHECK_I32(info); CHECK_I32(counters); CHECK_I32(out_idx);
    const long long n = residual.numel();
    TORCH_CHECK(k > 0 && k <= n && hist.numel() >= 2 * HIST_BINS &&
                out_val.numel() >= k && out_idx.numel() >= k);
    auto stream = cur_stream();
    unsigned* h1 = (unsigned*)hist.data_ptr<int>();
    unsigned* h2 = h1 + HIST_BINS;
    scan_threshold_kernel<0><<<1, BLOCK, 0, stream>>>(
        h1, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    refine_hist_kernel<<<stream_blocks(n), BLOCK, 0, stream>>>(
        residual.data_ptr<float>(), n, info.data_ptr<int>(), h2);
    scan_threshold_kernel<1><<<1, BLOCK, 0, stream>>>(
        h2, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    compact_kernel<true, false><<<stream_blocks(n), BLOCK, 0, stream>>>(
        residual.data_ptr<float>(), nullptr, n, info.data_ptr<int>(),
        counters.data_ptr<int>(), out_val.data_ptr<float>(),
        out_idx.data_ptr<int>(), (int)k);
    check_launch("gtopk");
}

// One fused merge round: dense[a ∪ b] += vals; gather-zero candidates
// (+ level-1 hist); two-level select of k among 2k.  The merged top-k
// is written into out_val/out_idx — pass views of the send buffer so
// the result is wire-ready with no extra copy (K8).
void merge_round(at::Tensor dense, at::Tensor send_buf, at::Tensor recv_buf,
                 int64_t k, at::Tensor hist, at::Tensor info,
                 at::Tensor counters, at::Tensor cand_val,
                 at::Tensor cand_idx, at::Tensor out_val, at::Tensor out_idx) {
    CHECK_F32(dense); CHECK_I32(send_buf); CHECK_I32(recv_buf);
    CHECK_I32(hist); CHECK_I32(info); CHECK_I32(counters);
    CHECK_F32(cand_val); CHECK_I32(cand_idx); CHECK_F32(out_val);
    CHECK_I32(out_idx);
    TORCH_CHECK(send_buf.numel() >= 2 * k && recv_buf.numel() >= 2 * k &&
                cand_val.numel() >= 2 * k && cand_idx.numel() >= 2 * k &&
                out_val.numel() >= k && out_idx.numel() >= k &&
                hist.numel() >= 2 * HIST_BINS);
    auto stream = cur_stream();
    unsigned* h1 = (unsigned*)hist.data_ptr<int>();
    unsigned* h2 = h1 + HIST_BINS;
    HIP_OK(hipMemsetAsync(h1, 0, 2 * HIST_BINS * sizeof(int), stream));
    const long long m = 2 * k;
    const int blocks = stream_blocks(m);
    merge_scatter2_kernel<<<blocks, BLOCK, 0, stream>>>(
        dense.data_ptr<float>(), send_buf.data_ptr<int>(),
        recv_buf.data_ptr<int>(), (int)k);
    merge_gather_hist_kernel<<<blocks, BLOCK, 0, stream>>>(
        dense.data_ptr<float>(), send_buf.data_ptr<int>(),
        recv_buf.data_ptr<int>(), (int)k, cand_val.data_ptr<float>(),
        cand_idx.data_ptr<int>(), h1);
    scan_threshold_kernel<0><<<1, BLOCK, 0, stream>>>(
        h1, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    refine_hist_kernel<<<blocks, BLOCK, 0, stream>>>(
        cand_val.data_ptr<float>(), m, info.data_ptr<int>(), h2);
    scan_threshold_kernel<1><<<1, BLOCK, 0, stream>>>(
        h2, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    compact_kernel<false, true><<<blocks, BLOCK, 0, stream>>>(
        cand_val.data_ptr<float>(), cand_idx.data_ptr<int>(), m,
        info.data_ptr<int>(), counters.data_ptr<int>(),
        out_val.data_ptr<float>(), out_idx.data_ptr<int>(), (int)k);
    check_launch("gtopk");
}

void topk_among(at::Tensor values, at::Tensor indexes, int64_t k,
                at::Tensor hist, at::Tensor info, at::Tensor counters,
                at::Tensor out_val, at::Tensor out_idx) {
    CHECK_F32(values); CHECK_F32(out_val);
    CHECK_I32(indexes); CHECK_I32(hist); CHECK_I32(info);
    CHECK_I32(counters); CHECK_I32(out_idx);
    const long long m = values.numel();
    TORCH_CHECK(indexes.numel() == m && k > 0 && k <= m);
    TORCH_CHECK(hist.numel() >= 2 * HIST_BINS);
    auto stream = cur_stream();
    unsigned* h1 = (unsigned*)hist.data_ptr<int>();
    unsigned* h2 = h1 + HIST_BINS;
    HIP_OK(hipMemsetAsync(h1, 0, 2 * HIST_BINS * sizeof(int), stream));
    const int blocks = stream_blocks(m);
    hist_vals_kernel<<<blocks, BLOCK, 0, stream>>>(
        values.data_ptr<float>(), m, h1);
    scan_threshold_kernel<0><<<1, BLOCK, 0, stream>>>(
        h1, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    refine_hist_kernel<<<blocks, BLOCK, 0, stream>>>(
        values.data_ptr<float>(), m, info.data_ptr<int>(), h2);
    scan_threshold_kernel<1><<<1, BLOCK, 0, stream>>>(
        h2, (int)k, info.data_ptr<int>(), counters.data_ptr<int>());
    compact_kernel<false, true><<<blocks, BLOCK, 0, stream>>>(
        values.data_ptr<float>(), indexes.data_ptr<int>(), m,
        info.data_ptr<int>(), counters.data_ptr<int>(),
        out_val.data_ptr<float>(), out_idx.data_ptr<int>(), (int)k);
    check_launch("gtopk");
}

void scatter_add(at::Tensor dense, at::Tensor idx, at::Tensor val) {
    CHECK_F32(dense); CHECK_I32(idx); CHECK_F32(val);
    const long long m = idx.numel();
    TORCH_CHECK(val.numel() == m);
    scatter_add_kernel<<<stream_blocks(m), BLOCK, 0, cur_stream()>>>(
        dense.data_ptr<float>(), idx.data_ptr<int>(), val.data_ptr<float>(),
        m);
    check_launch("gtopk");
}

void gather_zero(at::Tensor dense, at::Tensor idx, at::Tensor out) {
    CHECK_F32(dense); CHECK_I32(idx); CHECK_F32(out);
    const long long m = idx.numel();
    TORCH_CHECK(out.numel() == m);
    gather_zero_kernel<<<stream_blocks(m), BLOCK, 0, cur_stream()>>>(
        dense.data_ptr<float>(), idx.data_ptr<int>(), out.data_ptr<float>(),
        m);
    check_launch("gtopk");
}

void readd_rejected(at::Tensor residual, at::Tensor lidx, at::Tensor lval,
                    at::Tensor gidx, at::Tensor marks) {
    CHECK_F32(residual); CHECK_I32(lidx); CHECK_F32(lval); CHECK_I32(gidx);
    TORCH_CHECK(marks.is_cuda() && marks.scalar_type() == at::kByte &&
                marks.numel() == residual.numel());
    auto stream = cur_stream();
    const long long mg = gidx.numel(), ml = lidx.numel();
    TORCH_CHECK(lval.numel() == ml);
    auto* mk = marks.data_ptr<unsigned char>();
    mark_set_kernel<<<stream_blocks(mg), BLOCK, 0, stream>>>(
        mk, gidx.data_ptr<int>(), mg, 1);
    readd_kernel<<<stream_blocks(ml), BLOCK, 0, stream>>>(
        residual.data_ptr<float>(), lidx.data_ptr<int>(),
        lval.data_ptr<float>(), ml, mk);
    mark_set_kernel<<<stream_blocks(mg), BLOCK, 0, stream>>>(
        mk, gidx.data_ptr<int>(), mg, 0);
    check_launch("gtopk");
}

void densify(at::Tensor dense, at::Tensor idx, at::Tensor val, double scale) {
    CHECK_F32(dense); CHECK_I32(idx); CHECK_F32(val);
    const long long m = idx.numel();
    TORCH_CHECK(val.numel() == m);
    densify_kernel<<<stream_blocks(m), BLOCK, 0, cur_stream()>>>(
        dense.data_ptr<float>(), idx.data_ptr<int>(), val.data_ptr<float>(),
        m, (float)scale);
    check_launch("gtopk");
}

void sgd_step(at::Tensor param, at::Tensor mom, at::Tensor grad, double lr,
              double momentum, double weight_decay, double dampening,
              bool nesterov, bool zero_grad, double grad_scale) {
    CHECK_F32(param); CHECK_F32(mom); CHECK_F32(grad);
    const long long n = param.numel();
    TORCH_CHECK(mom.numel() == n && grad.numel() == n);
    sgd_kernel<<<stream_blocks((n + 3) / 4), BLOCK, 0, cur_stream()>>>(
        param.data_ptr<float>(), mom.data_ptr<float>(),
        grad.data_ptr<float>(), n, (float)lr, (float)momentum,
        (float)weight_decay, (float)dampening, nesterov ? 1 : 0,
        zero_grad ? 1 : 0, (float)grad_scale);
    check_launch("gtopk");
}

// fused BatchNorm (csrc/fusedbn.hip)
std::vector<at::Tensor> bn_fwd_train(
    at::Tensor x, c10::optional<at::Tensor> z, at::Tensor gamma,
    at::Tensor beta, at::Tensor running_mean, at::Tensor running_var,
    double momentum, double eps, bool relu, at::Tensor stats_scratch);
at::Tensor bn_fwd_eval(at::Tensor x, c10::optional<at::Tensor> z,
                       at::Tensor gamma, at::Tensor beta,
                       at::Tensor running_mean, at::Tensor running_var,
                       double eps, bool relu);
std::vector<at::Tensor> bn_bwd(
    at::Tensor dy, at::Tensor x, c10::optional<at::Tensor> y,
    at::Tensor gamma, at::Tensor saved_mean, at::Tensor saved_rstd,
    bool relu, bool need_dz, at::Tensor stats_scratch);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("bn_fwd_train", &bn_fwd_train,
          "fused NHWC BN(+add)(+ReLU) training forward (gfx950)");
    m.def("bn_fwd_eval", &bn_fwd_eval, "fused NHWC BN eval forward");
    m.def("bn_bwd", &bn_bwd, "fused NHWC BN(+add)(+ReLU) backward");
    m.def("ef_compress", &ef_compress,
          "fused residual accumulate + bucketed top-k select (gfx950)");
    m.def("ef_hist_zero", &ef_hist_zero,
          "zero the two-level histogram scratch (overlap path)");
    m.def("ef_accum_hist_range", &ef_accum_hist_range,
          "residual[s:e) += grad[s:e) + level-1 histogram (per-bucket, "
          "launched from backward hooks on the side stream)");
    m.def("ef_compress_finish", &ef_compress_finish,
          "threshold scan + refine + compact after per-bucket accumulate");
    m.def("merge_round", &merge_round,
          "one fused gtopk merge round: scatter2 + gather/hist + select");
    m.def("topk_among", &topk_among, "top-k among candidate (val,idx) pairs");
    m.def("scatter_add", &scatter_add, "dense[idx] += val");
    m.def("gather_zero", &gather_zero, "out = dense[idx]; dense[idx] = 0");
    m.def("readd_rejected", &readd_rejected,
          "residual[lidx] += lval where lidx not in gidx");
    m.def("densify", &densify, "dense[idx] = val * scale (skip zeros)");
    m.def("sgd_step", &sgd_step,
          "fused flat SGD w/ momentum+wd+nesterov, self-zeroing grad");
    m.attr("HIST_BINS") = HIST_BINS;
}
