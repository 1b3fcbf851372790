// Flash (online-softmax) self-attention forward — bf16, head_dim 64, any
// seq multiple of 128.  Generalizes attention.hip's seq=128 single-pass
// kernel: KV is processed in 128-key tiles with the
// online max/sum rescale (cdna_hip_programming.md T13 discussion — the
// textbook order: a tile's P is exponentiated only after the rescale
// decision that covers it), so nothing O(S^2) is materialised.  For the
// backward only the per-row LSE (log-sum-exp) is saved; P is recomputed
// tile-free in torch and the dropout mask is REGENERATED from the same
// philox counters (dropout_mask_mul below), so the forward writes only
// ctx + lse.
//
// Structure per 256-thread block (4 waves): 64 query rows (16 per wave),
// one (batch, head) per blockIdx.x, query block per blockIdx.y — grid
// B*NH*(S/64) blocks, 2-3 blocks/CU by LDS.  QK^T and PV are
// v_mfma_f32_16x16x32_bf16 tiles; softmax runs on the accumulator layout
// (row-quarter shuffle reduction, same idiom as attention.hip); P stages
// through padded LDS for the PV A-fragments; V transposes into LDS at
// stage time.
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) short bf16x8_f;
typedef __attribute__((ext_vector_type(4))) float f32x4_f;

struct PhiloxArgsFA {
    unsigned long long seed;
    unsigned long long offset;
    const unsigned long long* seed_ptr;    // captured (hipGraph) variant
    const unsigned long long* offset_ptr;
    unsigned int intragraph;
    int captured;
};

__device__ __forceinline__ float fa_b2f(short u) {
    union { float f; uint32_t i; } c;
    c.i = ((uint32_t)(uint16_t)u) << 16;
    return c.f;
}

__device__ __forceinline__ short fa_f2b(float f) {
    union { float f; uint32_t i; } c;
    c.f = f;
    uint32_t lsb = (c.i >> 16) & 1;
    c.i += 0x7fff + lsb;
    return (short)(c.i >> 16);
}

__device__ __forceinline__ void fa_philox4(unsigned long long seed,
                                           unsigned long long ctr,
                                           uint32_t out[4]) {
    uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
    uint32_t c0 = (uint32_t)ctr, c1 = (uint32_t)(ctr >> 32), c2 = 0, c3 = 0;
    #pragma unroll
    for (int i = 0; i < 10; ++i) {
        uint32_t hi0 = __umulhi(0xD2511F53u, c0), lo0 = 0xD2511F53u * c0;
        uint32_t hi1 = __umulhi(0xCD9E8D57u, c2), lo1 = 0xCD9E8D57u * c2;
        uint32_t n0 = hi1 ^ c1 ^ k0, n1 = lo1, n2 = hi0 ^ c3 ^ k1, n3 = lo0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += 0x9E3779B9u; k1 += 0xBB67AE85u;
    }
    out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
}

#define FA_D 64
#define FA_BM 64            // query rows per block (16 per wave)
#define FA_BT 128           // keys per KV tile
#define FA_LDK (FA_D + 8)   // K rows padded (bf16 elems)
#define FA_LDV (FA_BT + 8)  // Vt rows padded
#define FA_LDP (FA_BT + 8)  // P rows padded

// IM = 16-row query fragments per wave (1 -> 64-row blocks, 2 -> 128-row
// blocks).  IM=2 halves the per-block KV restaging traffic — the win at
// large seq where the KV sweep dominates; IM=1 gives 2x the blocks — the
// win at small seq where grid fill dominates (launcher picks by S).
template <int IM>
__global__ void __launch_bounds__(256)
attn_fwd_fa_kernel(const short* __restrict__ qkv,  // [b, s, 3, nh, hd]
                   const short* __restrict__ mask, // [b, s] additive bf16
                   short* __restrict__ out,        // [b, s, nh*hd]
                   float* __restrict__ lse,        // [b*nh, s] (or null)
                   int B, int NH, int S, float scale, float keep_prob,
                   PhiloxArgsFA rng, int apply_dropout) {
    const int bh = blockIdx.x;
    const int b = bh / NH, h = bh % NH;
    const int qblk = blockIdx.y;            // query block of IM*64 rows
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int fr = lane & 15;               // fragment row/col
    const int fk = (lane >> 4) * 8;         // k-offset within K=32 fragment

    __shared__ short k_lds[FA_BT * FA_LDK];
    __shared__ short vt_lds[FA_D * FA_LDV];
    __shared__ short p_lds[4 * 16 * IM * FA_LDP];

    unsigned long long seed = rng.seed, offset = rng.offset;
    if (rng.captured) {
        seed = *rng.seed_ptr;
        offset = *rng.offset_ptr + rng.intragraph;
    }
    const float inv_keep = 1.f / keep_prob;
    const uint32_t thresh = (uint32_t)(keep_prob * 4294967296.0);

    const int64_t qkv_row = (int64_t)3 * NH * FA_D;
    const int64_t base_b = (int64_t)b * S * qkv_row + (int64_t)h * FA_D;
    const int q_base = qblk * FA_BM * IM + wave * 16 * IM;  // wave's rows

    // Q fragments are tile-invariant: load once
    bf16x8_f qfr[IM][2];
    #pragma unroll
    for (int i2 = 0; i2 < IM; ++i2)
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks)
            qfr[i2][ks] = *reinterpret_cast<const bf16x8_f*>(
                qkv + base_b +
                (int64_t)(q_base + i2 * 16 + fr) * qkv_row + ks * 32 + fk);

    // online-softmax state: lane covers rows q_base + i2*16 + (lane>>4)*4+r
    float m_run[IM][4], l_run[IM][4];
    #pragma unroll
    for (int i2 = 0; i2 < IM; ++i2)
        #pragma unroll
        for (int r = 0; r < 4; ++r) { m_run[i2][r] = -1e30f; l_run[i2][r] = 0.f; }
    f32x4_f oacc[IM][4];
    #pragma unroll
    for (int i2 = 0; i2 < IM; ++i2)
        #pragma unroll
        for (int j = 0; j < 4; ++j) oacc[i2][j] = {0.f, 0.f, 0.f, 0.f};

    const int ntiles = S / FA_BT;
    for (int t = 0; t < ntiles; ++t) {
        const int key0 = t * FA_BT;
        // ---- stage K tile rows + V tile transposed ------------------
        {
            int tid = threadIdx.x;
            #pragma unroll
            for (int pass = 0; pass < 4; ++pass) {
                int idx = pass * 256 + tid;    // 0..1023 = 128 rows x 8 chunks
                int row = idx >> 3, d0 = (idx & 7) * 8;
                const short* src = qkv + base_b +
                    (int64_t)(key0 + row) * qkv_row + NH * FA_D + d0;
                bf16x8_f kv = *reinterpret_cast<const bf16x8_f*>(src);
                *reinterpret_cast<bf16x8_f*>(&k_lds[row * FA_LDK + d0]) = kv;
                const short* vsrc = src + NH * FA_D;  // c=2
                bf16x8_f vv = *reinterpret_cast<const bf16x8_f*>(vsrc);
                #pragma unroll
                for (int q = 0; q < 8; ++q)
                    vt_lds[(d0 + q) * FA_LDV + row] = vv[q];
            }
        }
        __syncthreads();

        // ---- QK^T for this tile: M=16*IM, N=128, K=64 ---------------
        f32x4_f acc[IM][8];
        #pragma unroll
        for (int i2 = 0; i2 < IM; ++i2)
            #pragma unroll
            for (int j = 0; j < 8; ++j) acc[i2][j] = {0.f, 0.f, 0.f, 0.f};
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            bf16x8_f bfr[8];
            #pragma unroll
            for (int j = 0; j < 8; ++j)
                bfr[j] = *reinterpret_cast<const bf16x8_f*>(
                    &k_lds[(j * 16 + fr) * FA_LDK + ks * 32 + fk]);
            #pragma unroll
            for (int i2 = 0; i2 < IM; ++i2)
                #pragma unroll
                for (int j = 0; j < 8; ++j)
                    acc[i2][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        qfr[i2][ks], bfr[j], acc[i2][j], 0, 0, 0);
        }

        // ---- scale + mask, online rescale, exponentiate -------------
        float mvals[8];
        #pragma unroll
        for (int j = 0; j < 8; ++j)
            mvals[j] = mask ? fa_b2f(mask[(int64_t)b * S + key0 + j * 16 + fr])
                            : 0.f;
        #pragma unroll
        for (int i2 = 0; i2 < IM; ++i2) {
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                float tm = -1e30f;
                #pragma unroll
                for (int j = 0; j < 8; ++j) {
                    acc[i2][j][r] = acc[i2][j][r] * scale + mvals[j];
                    tm = fmaxf(tm, acc[i2][j][r]);
                }
                #pragma unroll
                for (int off = 1; off < 16; off <<= 1)
                    tm = fmaxf(tm, __shfl_xor(tm, off, 64));
                float mnew = fmaxf(m_run[i2][r], tm);
                float alpha = __expf(m_run[i2][r] - mnew);
                float tsum = 0.f;
                #pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float e = __expf(acc[i2][j][r] - mnew);
                    acc[i2][j][r] = e;
                    tsum += e;
                }
                #pragma unroll
                for (int off = 1; off < 16; off <<= 1)
                    tsum += __shfl_xor(tsum, off, 64);
                l_run[i2][r] = l_run[i2][r] * alpha + tsum;
                m_run[i2][r] = mnew;
                // rescale O before this tile's PV lands (textbook order)
                #pragma unroll
                for (int j = 0; j < 4; ++j) oacc[i2][j][r] *= alpha;
            }
        }

        // ---- dropout (on unnormalised P — the l sum above is pre-drop,
        //      as softmax's denominator must be) + stage P -------------
        // Dropout counter mapping (shared by fwd, both bwd kernels and the
        // regen kernel): counter = (bh*(S/4) + qrow/4)*S + key, word =
        // qrow&3 — ONE philox call covers a lane's 4 query rows of one key
        // (the accumulator layout), 4x fewer philox rounds than per-element.
        short* my_p = &p_lds[wave * 16 * IM * FA_LDP];
        #pragma unroll
        for (int i2 = 0; i2 < IM; ++i2) {
            const int64_t ctr_row = (int64_t)bh * (S >> 2) +
                ((q_base + i2 * 16) >> 2) + (lane >> 4);
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                int key = j * 16 + fr;
                uint32_t rnd[4];
                if (apply_dropout)
                    fa_philox4(seed,
                               offset + (unsigned long long)(ctr_row * S + key0 + key),
                               rnd);
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int prow = i2 * 16 + (lane >> 4) * 4 + r;
                    float a = acc[i2][j][r];
                    if (apply_dropout)
                        a = (rnd[r] < thresh) ? a * inv_keep : 0.f;
                    my_p[prow * FA_LDP + key] = fa_f2b(a);
                }
            }
        }
        __syncthreads();

        // ---- PV accumulate: M=16*IM, N=64, K=128 --------------------
        #pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
            bf16x8_f bfr[4];
            #pragma unroll
            for (int j = 0; j < 4; ++j)
                bfr[j] = *reinterpret_cast<const bf16x8_f*>(
                    &vt_lds[(j * 16 + fr) * FA_LDV + ks * 32 + fk]);
            #pragma unroll
            for (int i2 = 0; i2 < IM; ++i2) {
                bf16x8_f afr = *reinterpret_cast<const bf16x8_f*>(
                    &my_p[(i2 * 16 + fr) * FA_LDP + ks * 32 + fk]);
                #pragma unroll
                for (int j = 0; j < 4; ++j)
                    oacc[i2][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        afr, bfr[j], oacc[i2][j], 0, 0, 0);
            }
        }
        __syncthreads();  // k_lds/vt_lds/p_lds free for the next tile
    }

    // ---- epilogue: normalise, write ctx + lse -----------------------
    #pragma unroll
    for (int i2 = 0; i2 < IM; ++i2) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int qrow = q_base + i2 * 16 + (lane >> 4) * 4 + r;
            float inv_l = l_run[i2][r] > 0.f ? 1.f / l_run[i2][r] : 0.f;
            int64_t row = ((int64_t)b * S + qrow) * NH * FA_D + h * FA_D;
            #pragma unroll
            for (int j = 0; j < 4; ++j)
                out[row + j * 16 + fr] = fa_f2b(oacc[i2][j][r] * inv_l);
            if (lse && fr == 0)
                lse[(int64_t)bh * S + qrow] =
                    m_run[i2][r] + (l_run[i2][r] > 0.f ? __logf(l_run[i2][r]) : 0.f);
        }
    }
}

extern "C" void launch_attn_fwd_fa(const void* qkv, const void* mask, void* out,
                                   void* lse, int B, int NH, int S, float scale,
                                   float keep_prob, unsigned long long seed,
                                   unsigned long long offset,
                                   const void* seed_ptr, const void* offset_ptr,
                                   unsigned int intragraph, int captured,
                                   int apply_dropout, hipStream_t stream) {
    PhiloxArgsFA rng;
    rng.seed = seed;
    rng.offset = offset;
    rng.seed_ptr = (const unsigned long long*)seed_ptr;
    rng.offset_ptr = (const unsigned long long*)offset_ptr;
    rng.intragraph = intragraph;
    rng.captured = captured;
    const char* im_env = getenv("OKTOPK_FA_IM");
    // measured (same-box A/B, s512 bs8h12): IM=2 LOSES — fwd 51.4 vs
    // 39.4 us (acc[2][8] costs ~64 VGPRs -> fewer waves; the KV sweep
    // was already latency-hidden).  Default stays IM=1 at every seq.
    int im = im_env ? atoi(im_env) : 1;
    if (im == 2 && S % 128 == 0) {
        // IM=2: 128-row blocks halve the per-block KV restaging sweep —
        // the dominant cost once S/64 blocks already fill the grid
        hipLaunchKernelGGL(attn_fwd_fa_kernel<2>,
                           dim3(B * NH, S / (2 * FA_BM)), dim3(256), 0,
                           stream, (const short*)qkv, (const short*)mask,
                           (short*)out, (float*)lse, B, NH, S, scale,
                           keep_prob, rng, apply_dropout);
    } else {
        hipLaunchKernelGGL(attn_fwd_fa_kernel<1>,
                           dim3(B * NH, S / FA_BM), dim3(256), 0, stream,
                           (const short*)qkv, (const short*)mask, (short*)out,
                           (float*)lse, B, NH, S, scale, keep_prob, rng,
                           apply_dropout);
    }
}

// ---------------------------------------------------------------------------
// dropout mask regeneration for the recompute backward: a[i] *= inv_keep or
// 0 with the SAME grouped philox counters the forward used (counter =
// (bh*S/4 + q/4)*S + key, word = q&3) — one work item per (bh, q-quad,
// key), four row-strided updates each.
// ---------------------------------------------------------------------------
__global__ void dropout_mask_mul_kernel(short* __restrict__ a, int64_t bh_n,
                                        int64_t S, PhiloxArgsFA rng,
                                        uint32_t thresh, float inv_keep) {
    unsigned long long seed = rng.seed, offset = rng.offset;
    if (rng.captured) {  // hipGraph: same device-side state the fwd read
        seed = *rng.seed_ptr;
        offset = *rng.offset_ptr + rng.intragraph;
    }
    // a is [BH, S, S]; grouped counter mapping (see the fwd kernel):
    // counter = (bh*(S/4) + q/4)*S + key, word = q&3 — one work item per
    // (bh, q-group-of-4, key), applying 4 row-strided updates.
    int64_t item = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t nitems = bh_n * (S >> 2) * S;
    for (; item < nitems; item += stride) {
        int64_t key = item % S;
        int64_t qg = (item / S) % (S >> 2);
        int64_t bh = item / (S * (S >> 2));
        uint32_t rnd[4];
        fa_philox4(seed,
                   offset + (unsigned long long)((bh * (S >> 2) + qg) * S + key),
                   rnd);
        int64_t base = bh * S * S + qg * 4 * S + key;
        #pragma unroll
        for (int w = 0; w < 4; ++w) {
            float v = fa_b2f(a[base + w * S]);
            v = (rnd[w] < thresh) ? v * inv_keep : 0.f;
            a[base + w * S] = fa_f2b(v);
        }
    }
}

extern "C" void launch_dropout_mask_mul(void* a, int64_t bh_n, int64_t S,
                                        unsigned long long seed,
                                        unsigned long long offset,
                                        const void* seed_ptr,
                                        const void* offset_ptr,
                                        unsigned int intragraph, int captured,
                                        float keep_prob, hipStream_t stream) {
    PhiloxArgsFA rng;
    rng.seed = seed;
    rng.offset = offset;
    rng.seed_ptr = (const unsigned long long*)seed_ptr;
    rng.offset_ptr = (const unsigned long long*)offset_ptr;
    rng.intragraph = intragraph;
    rng.captured = captured;
    uint32_t thresh = (uint32_t)(keep_prob * 4294967296.0);
    int64_t nitems = bh_n * (S >> 2) * S;
    int blocks = (int)((nitems + 255) / 256);
    if (blocks > 65535) blocks = 65535;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(dropout_mask_mul_kernel, dim3(blocks), dim3(256), 0,
                       stream, (short*)a, bh_n, S, rng, thresh,
                       1.f / keep_prob);
}

// ---------------------------------------------------------------------------
// Flash attention BACKWARD — two kernels, no O(S^2) materialisation.
//
//   dP_ij = gO_i . V_j          dS_ij = P_ij * (dP_ij - D_i)
//   P_ij  = exp(scale*Q_i.K_j + mask_j - lse_i)    D_i = gO_i . O_i
//   dQ_i  = scale * sum_j dS_ij K_j          (kernel A: query-parallel)
//   dK_j  = scale * sum_i dS_ij Q_i          (kernel B: key-parallel)
//   dV_j  = sum_i A_ij gO_i,  A = dropout(P)/keep  (kernel B)
//
// Both recompute P from the saved LSE (no softmax reduction) and
// regenerate the dropout mask from the forward's philox counters.
// Same MFMA fragment idioms as the forward; transposed LDS images where
// a B-operand's K dimension runs along rows (K^T for dQ, Q^T/gO^T for
// dK/dV), built scalar at stage time like the forward's V transpose.
// ---------------------------------------------------------------------------

#define FB_LDR (FA_D + 8)     // row-major row pitch (bf16)
#define FB_LDT (FA_BT + 8)    // kernel-A transposed / P pitch (128 keys)
#define FB_QT 64              // kernel-B query tile
#define FB_LDQ (FB_QT + 8)    // kernel-B transposed / P pitch (64 q)

__global__ void __launch_bounds__(256)
attn_bwd_dq_kernel(const short* __restrict__ qkv,
                   const short* __restrict__ go,   // [b, s, nh*hd]
                   const float* __restrict__ lse,  // [b*nh, s]
                   const float* __restrict__ dvec, // [b*nh, s]  D_i
                   const short* __restrict__ mask, // [b, s] or null
                   short* __restrict__ dqkv,       // [b, s, 3, nh, hd]
                   int B, int NH, int S, float scale, float keep_prob,
                   PhiloxArgsFA rng, int apply_dropout) {
    const int bh = blockIdx.x;
    const int b = bh / NH, h = bh % NH;
    const int qblk = blockIdx.y;
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int fr = lane & 15, fk = (lane >> 4) * 8;

    __shared__ short k_lds[FA_BT * FB_LDR];
    __shared__ short kt_lds[FA_D * FB_LDT];
    __shared__ short v_lds[FA_BT * FB_LDR];
    __shared__ short p_lds[4 * 16 * FB_LDT];

    unsigned long long seed = rng.seed, offset = rng.offset;
    if (rng.captured) {
        seed = *rng.seed_ptr;
        offset = *rng.offset_ptr + rng.intragraph;
    }
    const float inv_keep = 1.f / keep_prob;
    const uint32_t thresh = (uint32_t)(keep_prob * 4294967296.0);

    const int64_t qkv_row = (int64_t)3 * NH * FA_D;
    const int64_t base_b = (int64_t)b * S * qkv_row + (int64_t)h * FA_D;
    const int64_t go_row = (int64_t)NH * FA_D;
    const int64_t go_base = (int64_t)b * S * go_row + (int64_t)h * FA_D;
    const int q_base = qblk * FA_BM + wave * 16;

    bf16x8_f qfr[2], gofr[2];
    #pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
        qfr[ks] = *reinterpret_cast<const bf16x8_f*>(
            qkv + base_b + (int64_t)(q_base + fr) * qkv_row + ks * 32 + fk);
        gofr[ks] = *reinterpret_cast<const bf16x8_f*>(
            go + go_base + (int64_t)(q_base + fr) * go_row + ks * 32 + fk);
    }
    float lse_r[4], d_r[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        int qrow = q_base + (lane >> 4) * 4 + r;
        lse_r[r] = lse[(int64_t)bh * S + qrow];
        d_r[r] = dvec[(int64_t)bh * S + qrow];
    }
    f32x4_f dq_acc[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) dq_acc[j] = {0.f, 0.f, 0.f, 0.f};

    const int ntiles = S / FA_BT;
    for (int t = 0; t < ntiles; ++t) {
        const int key0 = t * FA_BT;
        {
            int tid = threadIdx.x;
            #pragma unroll
            for (int pass = 0; pass < 4; ++pass) {
                int idx = pass * 256 + tid;
                int row = idx >> 3, d0 = (idx & 7) * 8;
                const short* src = qkv + base_b +
                    (int64_t)(key0 + row) * qkv_row + NH * FA_D + d0;
                bf16x8_f kv = *reinterpret_cast<const bf16x8_f*>(src);
                *reinterpret_cast<bf16x8_f*>(&k_lds[row * FB_LDR + d0]) = kv;
                #pragma unroll
                for (int e = 0; e < 8; ++e)
                    kt_lds[(d0 + e) * FB_LDT + row] = kv[e];
                bf16x8_f vv = *reinterpret_cast<const bf16x8_f*>(src + NH * FA_D);
                *reinterpret_cast<bf16x8_f*>(&v_lds[row * FB_LDR + d0]) = vv;
            }
        }
        __syncthreads();

        // S and dP tiles: M=16 q, N=128 key, K=64 d
        f32x4_f accs[8], accdp[8];
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            accs[j] = {0.f, 0.f, 0.f, 0.f};
            accdp[j] = {0.f, 0.f, 0.f, 0.f};
        }
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                bf16x8_f bk = *reinterpret_cast<const bf16x8_f*>(
                    &k_lds[(j * 16 + fr) * FB_LDR + ks * 32 + fk]);
                accs[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    qfr[ks], bk, accs[j], 0, 0, 0);
                bf16x8_f bv = *reinterpret_cast<const bf16x8_f*>(
                    &v_lds[(j * 16 + fr) * FB_LDR + ks * 32 + fk]);
                accdp[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    gofr[ks], bv, accdp[j], 0, 0, 0);
            }
        }

        // dS = P * (dP - D) * scale, staged bf16
        short* my_p = &p_lds[wave * 16 * FB_LDT];
        const int64_t ctr_row =
            (int64_t)bh * (S >> 2) + (q_base >> 2) + (lane >> 4);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            int key = j * 16 + fr;
            float mv = mask ? fa_b2f(mask[(int64_t)b * S + key0 + key]) : 0.f;
            uint32_t rnd[4];
            if (apply_dropout)
                fa_philox4(seed,
                           offset + (unsigned long long)(ctr_row * S + key0 + key),
                           rnd);
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                int prow = (lane >> 4) * 4 + r;
                float p = __expf(accs[j][r] * scale + mv - lse_r[r]);
                float dp = accdp[j][r];
                if (apply_dropout)
                    dp = (rnd[r] < thresh) ? dp * inv_keep : 0.f;
                my_p[prow * FB_LDT + key] =
                    fa_f2b(p * (dp - d_r[r]) * scale);
            }
        }
        __syncthreads();

        // dQ += dS @ K: M=16 q, N=64 d, K=128 key
        #pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
            bf16x8_f afr = *reinterpret_cast<const bf16x8_f*>(
                &my_p[fr * FB_LDT + ks * 32 + fk]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                bf16x8_f bfr = *reinterpret_cast<const bf16x8_f*>(
                    &kt_lds[(j * 16 + fr) * FB_LDT + ks * 32 + fk]);
                dq_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr, bfr, dq_acc[j], 0, 0, 0);
            }
        }
        __syncthreads();
    }

    // dqkv[b, q, 0, h, d]
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        int qrow = q_base + (lane >> 4) * 4 + r;
        int64_t row = (int64_t)b * S * qkv_row + (int64_t)qrow * qkv_row +
                      (int64_t)h * FA_D;
        #pragma unroll
        for (int j = 0; j < 4; ++j)
            dqkv[row + j * 16 + fr] = fa_f2b(dq_acc[j][r]);
    }
}

__global__ void __launch_bounds__(256)
attn_bwd_dkv_kernel(const short* __restrict__ qkv,
                    const short* __restrict__ go,
                    const float* __restrict__ lse,
                    const float* __restrict__ dvec,
                    const short* __restrict__ mask,
                    short* __restrict__ dqkv,
                    int B, int NH, int S, float scale, float keep_prob,
                    PhiloxArgsFA rng, int apply_dropout) {
    const int bh = blockIdx.x;
    const int b = bh / NH, h = bh % NH;
    const int kblk = blockIdx.y;            // 64 key rows per block
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int fr = lane & 15, fk = (lane >> 4) * 8;

    __shared__ short q_lds[FB_QT * FB_LDR];
    __shared__ short qt_lds[FA_D * FB_LDQ];
    __shared__ short go_lds[FB_QT * FB_LDR];
    __shared__ short got_lds[FA_D * FB_LDQ];
    __shared__ short pa_lds[4 * 16 * FB_LDQ];   // A^T tile (for dV)
    __shared__ short pb_lds[4 * 16 * FB_LDQ];   // dS^T tile (for dK)
    // cooperative dropout mask for this (64 q x 64 key) tile: one philox
    // call covers a q-quad of one key (the shared counter mapping), built
    // by all 256 threads in 4 items each — 1024 calls for 4096 elements
    // instead of the per-element 4096 the first version paid
    __shared__ unsigned char mask_lds[FB_QT * (FB_QT + 8)];

    unsigned long long seed = rng.seed, offset = rng.offset;
    if (rng.captured) {
        seed = *rng.seed_ptr;
        offset = *rng.offset_ptr + rng.intragraph;
    }
    const float inv_keep = 1.f / keep_prob;
    const uint32_t thresh = (uint32_t)(keep_prob * 4294967296.0);

    const int64_t qkv_row = (int64_t)3 * NH * FA_D;
    const int64_t base_b = (int64_t)b * S * qkv_row + (int64_t)h * FA_D;
    const int64_t go_row = (int64_t)NH * FA_D;
    const int64_t go_base = (int64_t)b * S * go_row + (int64_t)h * FA_D;
    const int key_base = kblk * 64 + wave * 16;
    const int tid2 = threadIdx.x;

    bf16x8_f kfr[2], vfr[2];
    #pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
        const short* kr = qkv + base_b +
            (int64_t)(key_base + fr) * qkv_row + NH * FA_D;
        kfr[ks] = *reinterpret_cast<const bf16x8_f*>(kr + ks * 32 + fk);
        vfr[ks] = *reinterpret_cast<const bf16x8_f*>(kr + NH * FA_D + ks * 32 + fk);
    }
    float mask_r[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        int krow = key_base + (lane >> 4) * 4 + r;
        mask_r[r] = mask ? fa_b2f(mask[(int64_t)b * S + krow]) : 0.f;
    }
    f32x4_f dk_acc[4], dv_acc[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
        dk_acc[j] = {0.f, 0.f, 0.f, 0.f};
        dv_acc[j] = {0.f, 0.f, 0.f, 0.f};
    }

    const int ntiles = S / FB_QT;
    for (int t = 0; t < ntiles; ++t) {
        const int q0 = t * FB_QT;
        {
            int tid = threadIdx.x;
            #pragma unroll
            for (int pass = 0; pass < 2; ++pass) {
                int idx = pass * 256 + tid;   // 0..511 = 64 rows x 8 chunks
                int row = idx >> 3, d0 = (idx & 7) * 8;
                const short* qs = qkv + base_b +
                    (int64_t)(q0 + row) * qkv_row + d0;
                bf16x8_f qv = *reinterpret_cast<const bf16x8_f*>(qs);
                *reinterpret_cast<bf16x8_f*>(&q_lds[row * FB_LDR + d0]) = qv;
                #pragma unroll
                for (int e = 0; e < 8; ++e)
                    qt_lds[(d0 + e) * FB_LDQ + row] = qv[e];
                const short* gs = go + go_base + (int64_t)(q0 + row) * go_row + d0;
                bf16x8_f gv = *reinterpret_cast<const bf16x8_f*>(gs);
                *reinterpret_cast<bf16x8_f*>(&go_lds[row * FB_LDR + d0]) = gv;
                #pragma unroll
                for (int e = 0; e < 8; ++e)
                    got_lds[(d0 + e) * FB_LDQ + row] = gv[e];
            }
        }
        __syncthreads();

        if (apply_dropout) {
            #pragma unroll
            for (int i = 0; i < 4; ++i) {
                int item = i * 256 + tid2;     // 0..1023 = 16 q-quads x 64 keys
                int qg = item >> 6, key = item & 63;
                uint32_t rnd[4];
                fa_philox4(seed,
                           offset + (unsigned long long)(
                               ((int64_t)bh * (S >> 2) + ((q0 >> 2) + qg)) * S
                               + kblk * 64 + key),
                           rnd);
                #pragma unroll
                for (int w = 0; w < 4; ++w)
                    mask_lds[(qg * 4 + w) * (FB_QT + 8) + key] =
                        rnd[w] < thresh;
            }
        }

        // S^T and dP^T tiles: M=16 key, N=64 q, K=64 d
        f32x4_f accst[4], accdpt[4];
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            accst[j] = {0.f, 0.f, 0.f, 0.f};
            accdpt[j] = {0.f, 0.f, 0.f, 0.f};
        }
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                bf16x8_f bq = *reinterpret_cast<const bf16x8_f*>(
                    &q_lds[(j * 16 + fr) * FB_LDR + ks * 32 + fk]);
                accst[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    kfr[ks], bq, accst[j], 0, 0, 0);
                bf16x8_f bg = *reinterpret_cast<const bf16x8_f*>(
                    &go_lds[(j * 16 + fr) * FB_LDR + ks * 32 + fk]);
                accdpt[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    vfr[ks], bg, accdpt[j], 0, 0, 0);
            }
        }

        if (apply_dropout)
            __syncthreads();  // mask_lds written by all threads above

        // per-column (query) stats + dS^T / A^T staging; dropout keep
        // bits come from the cooperative LDS mask
        short* my_pa = &pa_lds[wave * 16 * FB_LDQ];
        short* my_pb = &pb_lds[wave * 16 * FB_LDQ];
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            int qcol = j * 16 + fr;
            float lse_c = lse[(int64_t)bh * S + q0 + qcol];
            float d_c = dvec[(int64_t)bh * S + q0 + qcol];
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                int prow = (lane >> 4) * 4 + r;
                float p = __expf(accst[j][r] * scale + mask_r[r] - lse_c);
                float dp = accdpt[j][r];
                float a = p;
                if (apply_dropout) {
                    int key_local = wave * 16 + prow;
                    int keep = mask_lds[qcol * (FB_QT + 8) + key_local];
                    a = keep ? p * inv_keep : 0.f;
                    dp = keep ? dp * inv_keep : 0.f;
                }
                my_pa[prow * FB_LDQ + qcol] = fa_f2b(a);
                my_pb[prow * FB_LDQ + qcol] = fa_f2b(p * (dp - d_c) * scale);
            }
        }
        __syncthreads();

        // dV += A^T @ gO ; dK += dS^T @ Q   (M=16 key, N=64 d, K=64 q)
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            bf16x8_f apa = *reinterpret_cast<const bf16x8_f*>(
                &my_pa[fr * FB_LDQ + ks * 32 + fk]);
            bf16x8_f apb = *reinterpret_cast<const bf16x8_f*>(
                &my_pb[fr * FB_LDQ + ks * 32 + fk]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                bf16x8_f bg = *reinterpret_cast<const bf16x8_f*>(
                    &got_lds[(j * 16 + fr) * FB_LDQ + ks * 32 + fk]);
                dv_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    apa, bg, dv_acc[j], 0, 0, 0);
                bf16x8_f bq = *reinterpret_cast<const bf16x8_f*>(
                    &qt_lds[(j * 16 + fr) * FB_LDQ + ks * 32 + fk]);
                dk_acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    apb, bq, dk_acc[j], 0, 0, 0);
            }
        }
        __syncthreads();
    }

    // dqkv[b, key, 1, h, d] = dK ; [b, key, 2, h, d] = dV
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        int krow = key_base + (lane >> 4) * 4 + r;
        int64_t row = (int64_t)b * S * qkv_row + (int64_t)krow * qkv_row +
                      (int64_t)h * FA_D;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            dqkv[row + NH * FA_D + j * 16 + fr] = fa_f2b(dk_acc[j][r]);
            dqkv[row + 2 * NH * FA_D + j * 16 + fr] = fa_f2b(dv_acc[j][r]);
        }
    }
}

extern "C" void launch_attn_bwd_fa(const void* qkv, const void* go,
                                   const void* lse, const void* dvec,
                                   const void* mask, void* dqkv, int B, int NH,
                                   int S, float scale, float keep_prob,
                                   unsigned long long seed,
                                   unsigned long long offset,
                                   const void* seed_ptr, const void* offset_ptr,
                                   unsigned int intragraph, int captured,
                                   int apply_dropout, hipStream_t stream) {
    PhiloxArgsFA rng;
    rng.seed = seed;
    rng.offset = offset;
    rng.seed_ptr = (const unsigned long long*)seed_ptr;
    rng.offset_ptr = (const unsigned long long*)offset_ptr;
    rng.intragraph = intragraph;
    rng.captured = captured;
    hipLaunchKernelGGL(attn_bwd_dq_kernel, dim3(B * NH, S / FA_BM), dim3(256),
                       0, stream, (const short*)qkv, (const short*)go,
                       (const float*)lse, (const float*)dvec,
                       (const short*)mask, (short*)dqkv, B, NH, S, scale,
                       keep_prob, rng, apply_dropout);
    hipLaunchKernelGGL(attn_bwd_dkv_kernel, dim3(B * NH, S / 64), dim3(256),
                       0, stream, (const short*)qkv, (const short*)go,
                       (const float*)lse, (const float*)dvec,
                       (const short*)mask, (short*)dqkv, B, NH, S, scale,
                       keep_prob, rng, apply_dropout);
}
