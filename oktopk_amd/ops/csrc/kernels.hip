// CDNA4 (gfx950) kernels for the Ok-Topk sparse-allreduce engine.
//
// Replaces the torch/CUDA op call-sites of the reference implementation
// (inventory: SURVEY.md §2.4a):
//   torch.topk threshold      -> 3-level histogram radix-select (kth_abs_bits)
//   abs>tau -> nonzero/gather -> deterministic 2-pass block-scan compaction
//   result[idx] += vals       -> scatter_add (device-scope atomics)
//   residual EF update        -> fused ef_restore_snapshot (t+=r; r=t)
//   SGD / Adam steps          -> fused elementwise kernels
//   clip_grad_norm            -> l2norm partial-sum reduction
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//   * wave64: ballots are 64-bit, popcount via __popcll
//   * memory-bound streaming kernels: fp32 scalar lane loads are 256 B/wave
//     (full line); elementwise kernels use float4 when 16B-aligned
//   * grid-stride with <=2048 blocks (G11); block size 256
//   * LDS histograms (8 KiB) + one global atomicAdd per block per bin (G12)

#include <hip/hip_runtime.h>
#include <cstdint>

#define BLOCK 256
#define WAVES_PER_BLOCK (BLOCK / 64)
#define MAX_BLOCKS 2048

static inline uint32_t tau_to_bits(float tau) {
    // |x| > tau as a uint compare on abs bit patterns; tau < 0 selects ALL
    // elements (including zeros), flagged by the sentinel 0xFFFFFFFF.
    if (tau < 0.f) return 0xFFFFFFFFu;
    union { float f; uint32_t u; } c;
    c.f = tau;
    return c.u & 0x7fffffffu;
}

__device__ __forceinline__ bool sel_gt(uint32_t abits, uint32_t tau_bits) {
    return (tau_bits == 0xFFFFFFFFu) | (abits > tau_bits);
}

static inline int n_blocks(int64_t work, int per_thread = 1) {
    int64_t b = (work + (int64_t)BLOCK * per_thread - 1) / ((int64_t)BLOCK * per_thread);
    if (b < 1) b = 1;
    if (b > MAX_BLOCKS) b = MAX_BLOCKS;
    return (int)b;
}

__device__ __forceinline__ uint32_t abs_bits(float x) {
    return __float_as_uint(x) & 0x7fffffffu;
}

// ---------------------------------------------------------------------------
// count_gt: count elements with |t| > tau
// ---------------------------------------------------------------------------
__global__ void count_gt_kernel(const float* __restrict__ t, int64_t n, int64_t n4,
                                uint32_t tau_bits, unsigned long long* __restrict__ out) {
    const float4* t4 = reinterpret_cast<const float4*>(t);
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    unsigned long long cnt = 0;
    for (; i < n4; i += stride) {
        float4 v = t4[i];
        cnt += sel_gt(abs_bits(v.x), tau_bits) + sel_gt(abs_bits(v.y), tau_bits)
             + sel_gt(abs_bits(v.z), tau_bits) + sel_gt(abs_bits(v.w), tau_bits);
    }
    for (i = n4 * 4 + (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride)
        cnt += sel_gt(abs_bits(t[i]), tau_bits);
    // wave reduce
    for (int off = 32; off > 0; off >>= 1) cnt += __shfl_down(cnt, off, 64);
    __shared__ unsigned long long ws[WAVES_PER_BLOCK];
    int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    if (lane == 0) ws[wave] = cnt;
    __syncthreads();
    if (threadIdx.x == 0) {
        unsigned long long s = 0;
        for (int w = 0; w < WAVES_PER_BLOCK; ++w) s += ws[w];
        atomicAdd(out, s);
    }
}

extern "C" void launch_count_gt(const float* t, int64_t n, float tau,
                                unsigned long long* out, hipStream_t stream) {
    uint32_t tb = tau_to_bits(tau);
    int64_t n4 = ((uintptr_t)t & 15) == 0 ? n >> 2 : 0;
    hipLaunchKernelGGL(count_gt_kernel, dim3(n_blocks(n, 8)), dim3(BLOCK), 0, stream,
                       t, n, n4, tb, out);
}

// count_multi_gt: counts for up to 8 candidate thresholds in ONE pass
// (feeds the adaptive-bump loop of add2residual, VGG/compression.py:384-404,
// with a single kernel + single D2H instead of one pass per candidate).
struct TauSet { uint32_t tb[8]; int n; };

__global__ void count_multi_gt_kernel(const float* __restrict__ t, int64_t n,
                                      TauSet taus,
                                      unsigned long long* __restrict__ out) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    unsigned long long cnt[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) cnt[j] = 0;
    for (; i < n; i += stride) {
        uint32_t u = abs_bits(t[i]);
        #pragma unroll
        for (int j = 0; j < 8; ++j)
            if (j < taus.n) cnt[j] += sel_gt(u, taus.tb[j]);
    }
    __shared__ unsigned long long ws[WAVES_PER_BLOCK];
    int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    for (int j = 0; j < taus.n; ++j) {
        unsigned long long c = cnt[j];
        for (int off = 32; off > 0; off >>= 1) c += __shfl_down(c, off, 64);
        if (lane == 0) ws[wave] = c;
        __syncthreads();
        if (threadIdx.x == 0) {
            unsigned long long s = 0;
            for (int w = 0; w < WAVES_PER_BLOCK; ++w) s += ws[w];
            atomicAdd(&out[j], s);
        }
        __syncthreads();
    }
}

extern "C" void launch_count_multi_gt(const float* t, int64_t n, const float* taus,
                                      int ntau, unsigned long long* out,
                                      hipStream_t stream) {
    TauSet ts;
    ts.n = ntau;
    for (int j = 0; j < 8; ++j) ts.tb[j] = j < ntau ? tau_to_bits(taus[j]) : 0;
    hipLaunchKernelGGL(count_multi_gt_kernel, dim3(n_blocks(n, 8)), dim3(BLOCK), 0,
                       stream, t, n, ts, out);
}

// ---------------------------------------------------------------------------
// compact_gt: deterministic (index-ordered) compaction of |t| > tau.
// Pass A: per-block counts over contiguous chunks.
// Pass B: stable intra-block scan (wave ballots + LDS) writing idx+val at
//         exclusive-scanned block offsets -> output ascending by index.
// ---------------------------------------------------------------------------
#define COMPACT_VEC 8  // elements per thread per iteration (2x float4)

// Pass A: per-wave counts for up to 8 candidate thresholds in ONE pass.
// Feeding the adaptive-bump choice (add2residual, VGG/compression.py:384-404)
// from the same pass that compaction needs anyway: one read of the tensor
// replaces the reference's count-per-candidate loop AND a separate
// compact-count pass.  Count layout: [cand][block][wave].
// Both passes are WAVE-autonomOUS: each wave owns a contiguous subchunk
// (chunk / WAVES_PER_BLOCK elements), so pass B needs no LDS scan and no
// __syncthreads, and its output is globally index-sorted by construction
// (wave subchunks are ordered, lanes own consecutive 8-element runs, the
// ballot prefix orders within the run).
__global__ void compact_count_multi_kernel(const float* __restrict__ t, int64_t n,
                                           TauSet taus, int64_t chunk,
                                           int* __restrict__ wave_counts,
                                           int nblocks) {
    int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int64_t sub = chunk / WAVES_PER_BLOCK;  // multiple of 512
    int64_t start = (int64_t)blockIdx.x * chunk + (int64_t)wave * sub;
    int64_t end = start + sub;
    if (start > n) start = n;
    if (end > n) end = n;
    int cnt[8];
    #pragma unroll
    for (int c = 0; c < 8; ++c) cnt[c] = 0;
    int64_t vend = start + ((end - start) & ~3LL);
    const float4* t4 = reinterpret_cast<const float4*>(t + start);
    int64_t n4 = (vend - start) >> 2;
    for (int64_t i = lane; i < n4; i += 64) {
        float4 x = t4[i];
        uint32_t a0 = abs_bits(x.x), a1 = abs_bits(x.y), a2 = abs_bits(x.z),
                 a3 = abs_bits(x.w);
        #pragma unroll
        for (int c = 0; c < 8; ++c)
            if (c < taus.n)
                cnt[c] += sel_gt(a0, taus.tb[c]) + sel_gt(a1, taus.tb[c]) +
                          sel_gt(a2, taus.tb[c]) + sel_gt(a3, taus.tb[c]);
    }
    for (int64_t i = vend + lane; i < end; i += 64) {
        uint32_t a = abs_bits(t[i]);
        #pragma unroll
        for (int c = 0; c < 8; ++c)
            if (c < taus.n) cnt[c] += sel_gt(a, taus.tb[c]);
    }
    for (int c = 0; c < taus.n; ++c) {
        int v = cnt[c];
        for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
        if (lane == 0)
            wave_counts[((int64_t)c * nblocks + blockIdx.x) * WAVES_PER_BLOCK + wave] = v;
    }
}

__global__ void compact_write_kernel(const float* __restrict__ t, int64_t n,
                                     uint32_t tau_bits, int64_t chunk,
                                     const int* __restrict__ wave_offsets,
                                     int32_t* __restrict__ out_idx,
                                     float* __restrict__ out_val) {
    int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int64_t sub = chunk / WAVES_PER_BLOCK;
    int64_t start = (int64_t)blockIdx.x * chunk + (int64_t)wave * sub;
    int64_t end = start + sub;
    if (start > n) start = n;
    if (end > n) end = n;
    uint64_t lt_mask = ((uint64_t)1 << lane) - 1;
    int run = wave_offsets[blockIdx.x * WAVES_PER_BLOCK + wave];
    const int64_t step = 64 * COMPACT_VEC;  // 512 elems per wave-iteration

    // Explicit 2-deep software pipeline: the next iteration's float4 loads
    // issue before this iteration's ballots/stores so hipcc emits counted
    // vmcnt waits instead of a full drain per 512-element step (see the
    // waitcnt histogram note in profiles/).  Main loop covers only full
    // iterations; the ragged tail is handled scalar below.  Measured
    // (rocprof kernel time): ~90 us at 109.5M/0.1%, ~4.9 TB/s effective.
    int64_t full_end = start + ((end - start) / step) * step;
    float4 c0, c1;
    if (start < full_end) {
        const float4* src = reinterpret_cast<const float4*>(t + start + lane * COMPACT_VEC);
        c0 = src[0];
        c1 = src[1];
    }
    for (int64_t base = start; base < full_end; base += step) {
        float4 n0, n1;
        bool have_next = base + step < full_end;
        if (have_next) {
            const float4* nsrc =
                reinterpret_cast<const float4*>(t + base + step + lane * COMPACT_VEC);
            n0 = nsrc[0];
            n1 = nsrc[1];
        }
        int64_t my = base + (int64_t)lane * COMPACT_VEC;
        float v[COMPACT_VEC] = {c0.x, c0.y, c0.z, c0.w, c1.x, c1.y, c1.z, c1.w};
        bool p[COMPACT_VEC];
        #pragma unroll
        for (int j = 0; j < COMPACT_VEC; ++j)
            p[j] = sel_gt(abs_bits(v[j]), tau_bits);
        uint64_t bj[COMPACT_VEC];
        int lane_prefix = 0, wave_total = 0;
        #pragma unroll
        for (int j = 0; j < COMPACT_VEC; ++j) {
            bj[j] = __ballot(p[j]);
            lane_prefix += __popcll(bj[j] & lt_mask);
            wave_total += __popcll(bj[j]);
        }
        // selected elements are rare (0.1-2% density): skip the whole store
        // phase on hit-free iterations (wave-uniform branch), and each j's
        // exec-masked store block when its ballot is empty — the store
        // blocks otherwise cost a waitcnt drain per iteration.
        if (wave_total) {
            int pos = run + lane_prefix;
            #pragma unroll
            for (int j = 0; j < COMPACT_VEC; ++j) {
                if (bj[j] && p[j]) {
                    out_idx[pos] = (int32_t)(my + j);
                    out_val[pos] = v[j];
                    ++pos;
                }
            }
        }
        run += wave_total;
        if (have_next) {
            c0 = n0;
            c1 = n1;
        }
    }
    // ragged tail (last partial wave-iteration), scalar
    for (int64_t base = full_end; base < end; base += step) {
        int64_t my = base + (int64_t)lane * COMPACT_VEC;
        bool p[COMPACT_VEC];
        float v[COMPACT_VEC];
        #pragma unroll
        for (int j = 0; j < COMPACT_VEC; ++j) {
            int64_t i = my + j;
            bool ok = i < end;
            float x = ok ? t[i] : 0.f;
            p[j] = ok && sel_gt(abs_bits(x), tau_bits);
            v[j] = x;
        }
        int lane_prefix = 0, wave_total = 0;
        #pragma unroll
        for (int j = 0; j < COMPACT_VEC; ++j) {
            uint64_t b = __ballot(p[j]);
            lane_prefix += __popcll(b & lt_mask);
            wave_total += __popcll(b);
        }
        int pos = run + lane_prefix;
        #pragma unroll
        for (int j = 0; j < COMPACT_VEC; ++j) {
            if (p[j]) {
                out_idx[pos] = (int32_t)(my + j);
                out_val[pos] = v[j];
                ++pos;
            }
        }
        run += wave_total;
    }
}

extern "C" void launch_compact_count_multi(const float* t, int64_t n,
                                           const float* taus, int ntau,
                                           int64_t chunk, int nblocks,
                                           int* block_counts, hipStream_t stream) {
    TauSet ts;
    ts.n = ntau;
    for (int j = 0; j < 8; ++j) ts.tb[j] = j < ntau ? tau_to_bits(taus[j]) : 0;
    hipLaunchKernelGGL(compact_count_multi_kernel, dim3(nblocks), dim3(BLOCK), 0,
                       stream, t, n, ts, chunk, block_counts, nblocks);
}

extern "C" void launch_compact_write(const float* t, int64_t n, float tau,
                                     int64_t chunk, int nblocks,
                                     const int* block_offsets, int32_t* out_idx,
                                     float* out_val, hipStream_t stream) {
    uint32_t tb = tau_to_bits(tau);
    hipLaunchKernelGGL(compact_write_kernel, dim3(nblocks), dim3(BLOCK), 0, stream,
                       t, n, tb, chunk, block_offsets, out_idx, out_val);
}

// ---------------------------------------------------------------------------
// kth_abs_bits: histogram radix-select over |x| bit patterns (nonnegative
// float bits order == uint order).  Host drives 3 levels (11/11/10 bits).
// ---------------------------------------------------------------------------
__global__ void hist_kernel(const float* __restrict__ t, int64_t n,
                            uint32_t prefix_mask, uint32_t prefix_val,
                            int shift, int nbins,
                            unsigned int* __restrict__ ghist) {
    extern __shared__ unsigned int lhist[];
    for (int b = threadIdx.x; b < nbins; b += BLOCK) lhist[b] = 0;
    __syncthreads();
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < n; i += stride) {
        uint32_t u = abs_bits(t[i]);
        if ((u & prefix_mask) == prefix_val)
            atomicAdd(&lhist[(u >> shift) & (nbins - 1)], 1u);
    }
    __syncthreads();
    for (int b = threadIdx.x; b < nbins; b += BLOCK)
        if (lhist[b]) atomicAdd(&ghist[b], lhist[b]);
}

extern "C" void launch_hist(const float* t, int64_t n, uint32_t prefix_mask,
                            uint32_t prefix_val, int shift, int nbins,
                            unsigned int* ghist, hipStream_t stream) {
    hipLaunchKernelGGL(hist_kernel, dim3(n_blocks(n, 8)), dim3(BLOCK),
                       nbins * sizeof(unsigned int), stream,
                       t, n, prefix_mask, prefix_val, shift, nbins, ghist);
}

// ---------------------------------------------------------------------------
// scatter ops
// ---------------------------------------------------------------------------
__global__ void scatter_add_kernel(float* __restrict__ dest,
                                   const int32_t* __restrict__ idx,
                                   const float* __restrict__ val, int64_t m) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < m; i += stride) atomicAdd(&dest[idx[i]], val[i]);
}

extern "C" void launch_scatter_add(float* dest, const int32_t* idx, const float* val,
                                   int64_t m, hipStream_t stream) {
    hipLaunchKernelGGL(scatter_add_kernel, dim3(n_blocks(m, 4)), dim3(BLOCK), 0, stream,
                       dest, idx, val, m);
}

__global__ void scatter_set_scaled_kernel(float* __restrict__ dest,
                                          const int32_t* __restrict__ idx,
                                          const float* __restrict__ val,
                                          float scale, int64_t m) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < m; i += stride) dest[idx[i]] = val[i] * scale;
}

extern "C" void launch_scatter_set_scaled(float* dest, const int32_t* idx,
                                          const float* val, float scale, int64_t m,
                                          hipStream_t stream) {
    hipLaunchKernelGGL(scatter_set_scaled_kernel, dim3(n_blocks(m, 4)), dim3(BLOCK), 0,
                       stream, dest, idx, val, scale, m);
}

// residual credit (reference intersect1d + update_residuals,
// VGG/allreducer.py:844-845): zero residual[idx[i]] where mask[idx[i]] —
// one kernel, no masked-select round trip
__global__ void zero_at_masked_kernel(float* __restrict__ dest,
                                      const int32_t* __restrict__ idx,
                                      const bool* __restrict__ mask, int64_t m) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < m; i += stride) {
        int32_t j = idx[i];
        if (mask[j]) dest[j] = 0.f;
    }
}

extern "C" void launch_zero_at_masked(float* dest, const int32_t* idx,
                                      const bool* mask, int64_t m,
                                      hipStream_t stream) {
    hipLaunchKernelGGL(zero_at_masked_kernel, dim3(n_blocks(m, 4)), dim3(BLOCK), 0,
                       stream, dest, idx, mask, m);
}

__global__ void zero_at_kernel(float* __restrict__ dest,
                               const int32_t* __restrict__ idx, int64_t m) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < m; i += stride) dest[idx[i]] = 0.f;
}

extern "C" void launch_zero_at(float* dest, const int32_t* idx, int64_t m,
                               hipStream_t stream) {
    hipLaunchKernelGGL(zero_at_kernel, dim3(n_blocks(m, 4)), dim3(BLOCK), 0, stream,
                       dest, idx, m);
}

// scatter_gt_credit: the fused world-1 round-2 tail.  For each selected
// (idx, val) pair with |val| > tau: result[idx] = val*scale, residual[idx]
// = 0 (the residual credit — at P==1 the global selection is a subset of
// the local one), and count it.  Replaces the reference-shaped chain
// boolean-mask -> nonzero (host sync) -> two gathers -> fill_sparse ->
// masked credit (VGG/allreducer.py:853-1081's owner filter, degenerate
// P==1 case) with one pass over the ~k selection.
__global__ void scatter_gt_credit_kernel(const int32_t* __restrict__ idx,
                                         const float* __restrict__ val,
                                         int64_t m, uint32_t tau_bits,
                                         float scale,
                                         float* __restrict__ result,
                                         float* __restrict__ residual,
                                         unsigned long long* __restrict__ cnt) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    unsigned long long c = 0;
    for (; i < m; i += stride) {
        float v = val[i];
        if (sel_gt(abs_bits(v), tau_bits)) {
            int32_t j = idx[i];
            result[j] = v * scale;
            residual[j] = 0.f;
            ++c;
        }
    }
    for (int off = 32; off > 0; off >>= 1) c += __shfl_down(c, off, 64);
    __shared__ unsigned long long ws[WAVES_PER_BLOCK];
    int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    if (lane == 0) ws[wave] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        unsigned long long s = 0;
        for (int w = 0; w < WAVES_PER_BLOCK; ++w) s += ws[w];
        atomicAdd(cnt, s);
    }
}

extern "C" void launch_scatter_gt_credit(const int32_t* idx, const float* val,
                                         int64_t m, float tau, float scale,
                                         float* result, float* residual,
                                         unsigned long long* cnt,
                                         hipStream_t stream) {
    hipLaunchKernelGGL(scatter_gt_credit_kernel, dim3(n_blocks(m, 4)),
                       dim3(BLOCK), 0, stream, idx, val, m, tau_to_bits(tau),
                       scale, result, residual, cnt);
}

// ---------------------------------------------------------------------------
// isin_sorted: binary search of each a[i] in ascending b
// ---------------------------------------------------------------------------
__global__ void isin_sorted_kernel(const int32_t* __restrict__ a, int64_t m,
                                   const int32_t* __restrict__ b, int64_t nb,
                                   bool* __restrict__ out) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < m; i += stride) {
        int32_t x = a[i];
        int64_t lo = 0, hi = nb;
        while (lo < hi) {
            int64_t mid = (lo + hi) >> 1;
            if (b[mid] < x) lo = mid + 1; else hi = mid;
        }
        out[i] = (lo < nb) && (b[lo] == x);
    }
}

extern "C" void launch_isin_sorted(const int32_t* a, int64_t m, const int32_t* b,
                                   int64_t nb, bool* out, hipStream_t stream) {
    hipLaunchKernelGGL(isin_sorted_kernel, dim3(n_blocks(m, 4)), dim3(BLOCK), 0, stream,
                       a, m, b, nb, out);
}

// ---------------------------------------------------------------------------
// fused elementwise streaming kernels (float4 when 16B-aligned)
// ---------------------------------------------------------------------------
__global__ void ef_restore_vec_kernel(float4* __restrict__ t, float4* __restrict__ r,
                                      int64_t n4) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < n4; i += stride) {
        float4 a = t[i], b = r[i];
        a.x += b.x; a.y += b.y; a.z += b.z; a.w += b.w;
        t[i] = a; r[i] = a;
    }
}

__global__ void ef_restore_scalar_kernel(float* __restrict__ t, float* __restrict__ r,
                                         int64_t lo, int64_t n) {
    int64_t i = lo + (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < n; i += stride) {
        float a = t[i] + r[i];
        t[i] = a; r[i] = a;
    }
}

extern "C" void launch_ef_restore(float* t, float* r, int64_t n, hipStream_t stream) {
    bool aligned = (((uintptr_t)t | (uintptr_t)r) & 15) == 0;
    int64_t n4 = aligned ? n / 4 : 0;
    if (n4 > 0)
        hipLaunchKernelGGL(ef_restore_vec_kernel, dim3(n_blocks(n4, 4)), dim3(BLOCK), 0,
                           stream, (float4*)t, (float4*)r, n4);
    if (n4 * 4 < n)
        hipLaunchKernelGGL(ef_restore_scalar_kernel, dim3(n_blocks(n - n4 * 4, 1)),
                           dim3(BLOCK), 0, stream, t, r, n4 * 4, n);
}

// fused bf16-grad upcast + EF restore: t = float(g) + r; r = t
// (replaces a separate 660MB upcast pass + the 1.76GB ef_restore pass with
// one 1.54GB pass when the model runs pure bf16)
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;

__device__ __forceinline__ float bf16bits_to_f32(short u) {
    union { float f; uint32_t i; } c;
    c.i = ((uint32_t)(uint16_t)u) << 16;
    return c.f;
}

__global__ void ef_upcast_vec_kernel(float* __restrict__ t, float* __restrict__ r,
                                     const short* __restrict__ g, int64_t n8) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    float4* t4 = reinterpret_cast<float4*>(t);
    float4* r4 = reinterpret_cast<float4*>(r);
    const bf16x8_t* g8 = reinterpret_cast<const bf16x8_t*>(g);
    for (; i < n8; i += stride) {
        bf16x8_t gv = g8[i];
        float4 ra = r4[2 * i], rb = r4[2 * i + 1];
        float4 ta, tb;
        ta.x = bf16bits_to_f32(gv[0]) + ra.x;
        ta.y = bf16bits_to_f32(gv[1]) + ra.y;
        ta.z = bf16bits_to_f32(gv[2]) + ra.z;
        ta.w = bf16bits_to_f32(gv[3]) + ra.w;
        tb.x = bf16bits_to_f32(gv[4]) + rb.x;
        tb.y = bf16bits_to_f32(gv[5]) + rb.y;
        tb.z = bf16bits_to_f32(gv[6]) + rb.z;
        tb.w = bf16bits_to_f32(gv[7]) + rb.w;
        t4[2 * i] = ta; t4[2 * i + 1] = tb;
        r4[2 * i] = ta; r4[2 * i + 1] = tb;
    }
}

__global__ void ef_upcast_scalar_kernel(float* __restrict__ t, float* __restrict__ r,
                                        const short* __restrict__ g, int64_t lo,
                                        int64_t n) {
    int64_t i = lo + (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < n; i += stride) {
        float v = bf16bits_to_f32(g[i]) + r[i];
        t[i] = v;
        r[i] = v;
    }
}

extern "C" void launch_ef_upcast(float* t, float* r, const void* g, int64_t n,
                                 hipStream_t stream) {
    bool aligned = ((((uintptr_t)t | (uintptr_t)r | (uintptr_t)g) & 15) == 0);
    int64_t n8 = aligned ? n / 8 : 0;
    if (n8 > 0)
        hipLaunchKernelGGL(ef_upcast_vec_kernel, dim3(n_blocks(n8, 4)), dim3(BLOCK), 0,
                           stream, t, r, (const short*)g, n8);
    if (n8 * 8 < n)
        hipLaunchKernelGGL(ef_upcast_scalar_kernel, dim3(n_blocks(n - n8 * 8, 1)),
                           dim3(BLOCK), 0, stream, t, r, (const short*)g, n8 * 8, n);
}

// ---------------------------------------------------------------------------
// Device-side reduction + scan of the compact wave counts: the host needs
// ONLY the per-candidate totals (8 ints) to apply the bump rule and size the
// output; the nw-entry offset table stays on the GPU (previously a 196 KB
// readback + CPU scan + 32 KB upload per compaction).
// ---------------------------------------------------------------------------
__global__ void count_totals_kernel(const int* __restrict__ wave_counts, int nw,
                                    int ntau, int64_t* __restrict__ totals) {
    int c = blockIdx.x;  // one block per candidate
    if (c >= ntau) return;
    int64_t s = 0;
    for (int b = threadIdx.x; b < nw; b += blockDim.x)
        s += wave_counts[(int64_t)c * nw + b];
    __shared__ int64_t sh[256];
    sh[threadIdx.x] = s;
    __syncthreads();
    for (int o = 128; o > 0; o >>= 1) {
        if (threadIdx.x < o) sh[threadIdx.x] += sh[threadIdx.x + o];
        __syncthreads();
    }
    if (threadIdx.x == 0) totals[c] = sh[0];
}

// exclusive scan of one candidate row into offs (single block; nw <= 8192)
__global__ void scan_offsets_kernel(const int* __restrict__ row, int nw,
                                    int* __restrict__ offs) {
    const int PER = 8;  // 1024 threads x 8 = 8192 max entries
    __shared__ int partial[1024];
    int tid = threadIdx.x;
    int base = tid * PER;
    int loc[PER];
    int s = 0;
    #pragma unroll
    for (int j = 0; j < PER; ++j) {
        int i = base + j;
        loc[j] = s;  // exclusive within this thread's run
        s += i < nw ? row[i] : 0;
    }
    partial[tid] = s;
    __syncthreads();
    for (int d = 1; d < 1024; d <<= 1) {  // Hillis-Steele over partials
        int v = tid >= d ? partial[tid - d] : 0;
        __syncthreads();
        partial[tid] += v;
        __syncthreads();
    }
    int prefix = tid == 0 ? 0 : partial[tid - 1];
    #pragma unroll
    for (int j = 0; j < PER; ++j) {
        int i = base + j;
        if (i < nw) offs[i] = prefix + loc[j];
    }
}

extern "C" void launch_count_totals(const int* wave_counts, int nw, int ntau,
                                    int64_t* totals, hipStream_t stream) {
    hipLaunchKernelGGL(count_totals_kernel, dim3(ntau), dim3(256), 0, stream,
                       wave_counts, nw, ntau, totals);
}

extern "C" void launch_scan_offsets(const int* row, int nw, int* offs,
                                    hipStream_t stream) {
    hipLaunchKernelGGL(scan_offsets_kernel, dim3(1), dim3(1024), 0, stream, row,
                       nw, offs);
}

// ---------------------------------------------------------------------------
// fused EF restore + compact pass A: one streaming pass does
//   t = (float)g + r   (or t += r when no bf16 grad),  r = t
// AND counts |t| > tau_c for up to 8 candidate thresholds in the compact
// count layout [cand][block][wave].  The steady-state (non-exact) oktopk
// iteration then needs only this pass + the compact write pass: the separate
// count read of t (440 MB at BERT-base scale) disappears.  Valid because the
// candidate taus derive from the PREVIOUS iteration's threshold, known
// before the restore.  Uses the compact wave-subchunk grid so pass B can
// consume the counts directly.
// ---------------------------------------------------------------------------
template <bool HAS_G>
__global__ void ef_count_kernel(float* __restrict__ t, float* __restrict__ r,
                                const short* __restrict__ g, int64_t n,
                                TauSet taus, int64_t chunk,
                                int* __restrict__ wave_counts, int nblocks) {
    int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int64_t sub = chunk / WAVES_PER_BLOCK;  // multiple of 512
    int64_t start = (int64_t)blockIdx.x * chunk + (int64_t)wave * sub;
    int64_t end = start + sub;
    if (start > n) start = n;
    if (end > n) end = n;
    int cnt[8];
    #pragma unroll
    for (int c = 0; c < 8; ++c) cnt[c] = 0;

    int64_t vend = start + ((end - start) & ~7LL);
    float4* t4 = reinterpret_cast<float4*>(t + start);
    float4* r4 = reinterpret_cast<float4*>(r + start);
    const bf16x8_t* g8 = reinterpret_cast<const bf16x8_t*>(g + start);
    int64_t ng = (vend - start) >> 3;
    for (int64_t i = lane; i < ng; i += 64) {
        float4 ra = r4[2 * i], rb = r4[2 * i + 1];
        float4 ta, tb;
        if (HAS_G) {
            bf16x8_t gv = g8[i];
            ta.x = bf16bits_to_f32(gv[0]) + ra.x;
            ta.y = bf16bits_to_f32(gv[1]) + ra.y;
            ta.z = bf16bits_to_f32(gv[2]) + ra.z;
            ta.w = bf16bits_to_f32(gv[3]) + ra.w;
            tb.x = bf16bits_to_f32(gv[4]) + rb.x;
            tb.y = bf16bits_to_f32(gv[5]) + rb.y;
            tb.z = bf16bits_to_f32(gv[6]) + rb.z;
            tb.w = bf16bits_to_f32(gv[7]) + rb.w;
        } else {
            float4 ua = t4[2 * i], ub = t4[2 * i + 1];
            ta.x = ua.x + ra.x; ta.y = ua.y + ra.y;
            ta.z = ua.z + ra.z; ta.w = ua.w + ra.w;
            tb.x = ub.x + rb.x; tb.y = ub.y + rb.y;
            tb.z = ub.z + rb.z; tb.w = ub.w + rb.w;
        }
        // only the residual snapshot is written: the engine's steady
        // state never reads the restored t again (the result densify
        // zeroes it), and pass B compacts from r — one full-tensor
        // write saved per step
        r4[2 * i] = ta; r4[2 * i + 1] = tb;
        uint32_t a[8] = {abs_bits(ta.x), abs_bits(ta.y), abs_bits(ta.z),
                         abs_bits(ta.w), abs_bits(tb.x), abs_bits(tb.y),
                         abs_bits(tb.z), abs_bits(tb.w)};
        #pragma unroll
        for (int c = 0; c < 8; ++c)
            if (c < taus.n) {
                int s = 0;
                #pragma unroll
                for (int j = 0; j < 8; ++j) s += sel_gt(a[j], taus.tb[c]);
                cnt[c] += s;
            }
    }
    for (int64_t i = vend + lane; i < end; i += 64) {
        float v = (HAS_G ? bf16bits_to_f32(g[i]) : t[i]) + r[i];
        r[i] = v;
        uint32_t a = abs_bits(v);
        #pragma unroll
        for (int c = 0; c < 8; ++c)
            if (c < taus.n) cnt[c] += sel_gt(a, taus.tb[c]);
    }
    for (int c = 0; c < taus.n; ++c) {
        int v = cnt[c];
        for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
        if (lane == 0)
            wave_counts[((int64_t)c * nblocks + blockIdx.x) * WAVES_PER_BLOCK + wave] = v;
    }
}

extern "C" void launch_ef_count(float* t, float* r, const void* g, int64_t n,
                                const float* taus, int ntau, int64_t chunk,
                                int nblocks, int* wave_counts,
                                hipStream_t stream) {
    TauSet ts;
    ts.n = ntau;
    for (int j = 0; j < 8; ++j) ts.tb[j] = j < ntau ? tau_to_bits(taus[j]) : 0;
    if (g != nullptr)
        hipLaunchKernelGGL((ef_count_kernel<true>), dim3(nblocks), dim3(BLOCK), 0,
                           stream, t, r, (const short*)g, n, ts, chunk,
                           wave_counts, nblocks);
    else
        hipLaunchKernelGGL((ef_count_kernel<false>), dim3(nblocks), dim3(BLOCK), 0,
                           stream, t, r, nullptr, n, ts, chunk, wave_counts,
                           nblocks);
}

__global__ void sgd_kernel(float* __restrict__ p, const float* __restrict__ g,
                           float* __restrict__ buf, int64_t n, float lr, float mom,
                           float wd, int nesterov, int use_mom) {
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < n; i += stride) {
        float d = g[i] + wd * p[i];
        if (use_mom) {
            float b = buf[i] * mom + d;
            buf[i] = b;
            d = nesterov ? d + mom * b : b;
        }
        p[i] -= lr * d;
    }
}

extern "C" void launch_sgd(float* p, const float* g, float* buf, int64_t n, float lr,
                           float mom, float wd, int nesterov, hipStream_t stream) {
    hipLaunchKernelGGL(sgd_kernel, dim3(n_blocks(n, 4)), dim3(BLOCK), 0, stream,
                       p, g, buf, n, lr, mom, wd, nesterov, mom != 0.f);
}

__device__ __forceinline__ short adam_f2b(float f) {
    union { float f; uint32_t i; } c;
    c.f = f;
    uint32_t lsb = (c.i >> 16) & 1;
    c.i += 0x7fff + lsb;
    return (short)(c.i >> 16);
}

__device__ __forceinline__ float adam_one(float pi, float gi, float& mi,
                                          float& vi, float lr, float b1,
                                          float b2, float eps, float wd) {
    mi = mi * b1 + (1.f - b1) * gi;
    vi = vi * b2 + (1.f - b2) * gi * gi;
    float up = mi / (sqrtf(vi) + eps) + wd * pi;
    return pi - lr * up;
}

// float4-vectorized Adam: 30 B of HBM traffic per element (4 reads + 3-4
// writes) makes this purely streaming — dword4 loads/stores lift it from
// ~4.9 to the ~5.6 TB/s streaming ceiling measured by tools/kernbench.py.
// The scalar tail covers n % 4 and the (never-taken in practice — the
// optimizer pads group boundaries) unaligned-base fallback is n4 = 0.
__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            short* __restrict__ p_bf16,
                            const float* __restrict__ gscale_ptr,
                            int64_t n, int64_t n4,
                            float lr, float b1, float b2, float eps, float wd) {
    // device-side gradient-clip scale (grad_clip_scale kernel) — folding
    // the clip into the g read removes the separate full-tensor mul pass
    // AND the host .item() sync of the norm check
    const float gs = gscale_ptr ? *gscale_ptr : 1.f;
    float4* p4 = reinterpret_cast<float4*>(p);
    const float4* g4 = reinterpret_cast<const float4*>(g);
    float4* m4 = reinterpret_cast<float4*>(m);
    float4* v4 = reinterpret_cast<float4*>(v);
    short4* b4 = reinterpret_cast<short4*>(p_bf16);
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    for (; i < n4; i += stride) {
        float4 pi = p4[i], gi = g4[i], mi = m4[i], vi = v4[i];
        pi.x = adam_one(pi.x, gi.x * gs, mi.x, vi.x, lr, b1, b2, eps, wd);
        pi.y = adam_one(pi.y, gi.y * gs, mi.y, vi.y, lr, b1, b2, eps, wd);
        pi.z = adam_one(pi.z, gi.z * gs, mi.z, vi.z, lr, b1, b2, eps, wd);
        pi.w = adam_one(pi.w, gi.w * gs, mi.w, vi.w, lr, b1, b2, eps, wd);
        m4[i] = mi; v4[i] = vi; p4[i] = pi;
        if (p_bf16)
            b4[i] = make_short4(adam_f2b(pi.x), adam_f2b(pi.y),
                                adam_f2b(pi.z), adam_f2b(pi.w));
    }
    for (i = n4 * 4 + (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < n;
         i += stride) {
        float mi = m[i], vi = v[i];
        float pn = adam_one(p[i], g[i] * gs, mi, vi, lr, b1, b2, eps, wd);
        m[i] = mi; v[i] = vi; p[i] = pn;
        // fused bf16 weight-mirror write (saves the separate cast pass of
        // the pure-bf16 model path)
        if (p_bf16) p_bf16[i] = adam_f2b(pn);
    }
}

__global__ void clip_scale_kernel(const double* __restrict__ ss,
                                  float max_norm, float* __restrict__ out) {
    if (threadIdx.x == 0) {
        double gn = sqrt(*ss);
        out[0] = gn > (double)max_norm
                     ? (float)((double)max_norm / (gn + 1e-6)) : 1.f;
    }
}

extern "C" void launch_clip_scale(const double* ss, float max_norm, float* out,
                                  hipStream_t stream) {
    hipLaunchKernelGGL(clip_scale_kernel, dim3(1), dim3(64), 0, stream,
                       ss, max_norm, out);
}

extern "C" void launch_adam(float* p, const float* g, float* m, float* v,
                            void* p_bf16, const float* gscale, int64_t n,
                            float lr, float b1, float b2, float eps, float wd,
                            hipStream_t stream) {
    bool aligned = (((uintptr_t)p | (uintptr_t)g | (uintptr_t)m |
                     (uintptr_t)v) & 15) == 0 &&
                   (((uintptr_t)p_bf16) & 7) == 0;
    int64_t n4 = aligned ? n >> 2 : 0;
    hipLaunchKernelGGL(adam_kernel, dim3(n_blocks(n, 8)), dim3(BLOCK), 0, stream,
                       p, g, m, v, (short*)p_bf16, gscale, n, n4, lr, b1, b2,
                       eps, wd);
}

// ---------------------------------------------------------------------------
// l2 norm (sum of squares in double, sqrt on host)
// ---------------------------------------------------------------------------
__global__ void sumsq_kernel(const float* __restrict__ t, int64_t n, int64_t n4,
                             double* __restrict__ out) {
    const float4* t4 = reinterpret_cast<const float4*>(t);
    int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * BLOCK;
    double acc = 0.0;
    for (; i < n4; i += stride) {
        float4 v = t4[i];
        acc += (double)v.x * v.x + (double)v.y * v.y
             + (double)v.z * v.z + (double)v.w * v.w;
    }
    for (i = n4 * 4 + (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
        double x = t[i];
        acc += x * x;
    }
    for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, 64);
    __shared__ double ws[WAVES_PER_BLOCK];
    int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    if (lane == 0) ws[wave] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        double s = 0;
        for (int w = 0; w < WAVES_PER_BLOCK; ++w) s += ws[w];
        atomicAdd(out, s);
    }
}

extern "C" void launch_sumsq(const float* t, int64_t n, double* out, hipStream_t stream) {
    int64_t n4 = ((uintptr_t)t & 15) == 0 ? n >> 2 : 0;
    hipLaunchKernelGGL(sumsq_kernel, dim3(n_blocks(n, 8)), dim3(BLOCK), 0, stream,
                       t, n, n4, out);
}

// ---------------------------------------------------------------------------
// bias-grad column sum: gy bf16 [R, C] row-major -> out fp32 [C] (atomic
// accumulate over row slabs; caller zeroes out).  Replaces torch's
// per-linear reduce_kernel chain in backward (~10 us each, 9.9 us avg in
// the r02-i profile) with a coalesced strip reduction.
// ---------------------------------------------------------------------------
__device__ __forceinline__ float ks_b2f(short u) {
    union { float f; uint32_t i; } c;
    c.i = ((uint32_t)(uint16_t)u) << 16;
    return c.f;
}

__global__ void colsum_bf16_kernel(const short* __restrict__ gy, int64_t R,
                                   int64_t C, int64_t slab,
                                   float* __restrict__ out) {
    int64_t col = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
    if (col >= C) return;
    int64_t r0 = (int64_t)blockIdx.y * slab;
    int64_t r1 = r0 + slab;
    if (r1 > R) r1 = R;
    // 4 independent partials keep 4 row loads in flight per thread
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    int64_t r = r0;
    for (; r + 4 <= r1; r += 4) {
        a0 += ks_b2f(gy[r * C + col]);
        a1 += ks_b2f(gy[(r + 1) * C + col]);
        a2 += ks_b2f(gy[(r + 2) * C + col]);
        a3 += ks_b2f(gy[(r + 3) * C + col]);
    }
    for (; r < r1; ++r) a0 += ks_b2f(gy[r * C + col]);
    atomicAdd(&out[col], (a0 + a1) + (a2 + a3));
}

extern "C" void launch_colsum_bf16(const void* gy, int64_t R, int64_t C,
                                   float* out, hipStream_t stream) {
    // fill the chip: ~1024 blocks; column strips x row slabs
    int64_t xblocks = (C + BLOCK - 1) / BLOCK;
    int64_t target = 1024 / (xblocks ? xblocks : 1) + 1;
    int64_t slab = (R + target - 1) / target;
    if (slab < 8) slab = 8;
    int64_t nslab = (R + slab - 1) / slab;
    dim3 grid((unsigned)xblocks, (unsigned)nslab);
    hipLaunchKernelGGL(colsum_bf16_kernel, grid, dim3(BLOCK), 0, stream,
                       (const short*)gy, R, C, slab, out);
}

// ---------------------------------------------------------------------------
// attention-backward D: D[bh][s] = sum_d go[b,s,h*hd+d] * out[b,s,h*hd+d]
// (one wave per (b, h, s) row of head_dim 64 — lane l holds element l).
// ---------------------------------------------------------------------------
__global__ void attn_rowdot_kernel(const short* __restrict__ go,
                                   const short* __restrict__ out,
                                   int64_t B, int64_t S, int64_t NH,
                                   float* __restrict__ d) {
    int64_t row = ((int64_t)blockIdx.x * (BLOCK / 64)) + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    int64_t nrows = B * NH * S;
    if (row >= nrows) return;
    // row = (b*NH + h)*S + s
    int64_t s = row % S;
    int64_t bh = row / S;
    int64_t b = bh / NH, h = bh % NH;
    int64_t off = ((b * S + s) * NH + h) * 64 + lane;
    float v = ks_b2f(go[off]) * ks_b2f(out[off]);
    for (int o = 32; o > 0; o >>= 1) v += __shfl_down(v, o, 64);
    if (lane == 0) d[row] = v;
}

extern "C" void launch_attn_rowdot(const void* go, const void* out, int64_t B,
                                   int64_t S, int64_t NH, float* d,
                                   hipStream_t stream) {
    int64_t nrows = B * NH * S;
    int64_t blocks = (nrows + (BLOCK / 64) - 1) / (BLOCK / 64);
    hipLaunchKernelGGL(attn_rowdot_kernel, dim3((unsigned)blocks), dim3(BLOCK),
                       0, stream, (const short*)go, (const short*)out, B, S,
                       NH, d);
}
