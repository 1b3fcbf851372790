// Fused BERT self-attention forward (bf16, seq multiple not required —
// specialised for the reference config seq=128, head_dim=64).
//
// One kernel computes, per (batch, head): S = QK^T/sqrt(d) + mask,
// P = softmax(S), A = dropout(P), ctx = A @ V — replacing the reference
// path's ~10 kernels/layer (2 bmm + softmax + dropout + permutes/copies).
// Reads Q/K/V directly from the fused qkv projection buffer
// (layout [b, s, 3, nh, hd] — no permute materialisation) and writes the
// context in [b, s, nh*hd] so the output reshape is free.
//
// MFMA structure (cdna_hip_programming.md §3/§5): 256-thread block = 4 waves,
// wave owns 32 query rows; QK^T = 32 x v_mfma_f32_16x16x32_bf16 per wave
// (M=32, N=128, K=64), softmax on the accumulator layout (row-quarter
// shuffle reduction), P staged through padded LDS for the PV A-fragments,
// V transposed into LDS at stage time for the PV B-fragments.
// Dropout uses torch's philox state (capture-safe: offset pointers under
// hipGraph capture), counter = element index — own philox4x32-10.
//
// P (post-softmax) and A (post-dropout) are written out (bf16) for the
// hand-written backward (torch bmms in ops/fused_attn.py).

#include <hip/hip_runtime.h>
#include <cstdint>

typedef __attribute__((ext_vector_type(8))) short bf16x8_a;
typedef __attribute__((ext_vector_type(4))) float f32x4_a;

struct PhiloxArgs {
    unsigned long long seed;
    unsigned long long offset;
    const unsigned long long* seed_ptr;    // captured (hipGraph) variant
    const unsigned long long* offset_ptr;
    unsigned int intragraph;
    int captured;
};

__device__ __forceinline__ float att_b2f(short u) {
    union { float f; uint32_t i; } c;
    c.i = ((uint32_t)(uint16_t)u) << 16;
    return c.f;
}

__device__ __forceinline__ short att_f2b(float f) {
    union { float f; uint32_t i; } c;
    c.f = f;
    uint32_t lsb = (c.i >> 16) & 1;
    c.i += 0x7fff + lsb;
    return (short)(c.i >> 16);
}

__device__ __forceinline__ void philox4(unsigned long long seed,
                                        unsigned long long ctr, uint32_t out[4]) {
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

#define ATT_S 128
#define ATT_D 64
#define ATT_LDK (ATT_D + 8)    // K rows padded (bf16 elems)
#define ATT_LDV (ATT_S + 8)    // Vt rows padded
#define ATT_LDP (ATT_S + 8)    // P rows padded

__global__ void __launch_bounds__(256)
attn_fwd_kernel(const short* __restrict__ qkv,  // [b, s, 3, nh, hd]
                const short* __restrict__ mask, // [b, s] additive bf16 (or null)
                short* __restrict__ out,        // [b, s, nh*hd]
                short* __restrict__ p_save,     // [b*nh, s, s]
                short* __restrict__ a_save,     // [b*nh, s, s]
                int B, int NH, float scale, float keep_prob,
                PhiloxArgs rng, int apply_dropout) {
    const int bh = blockIdx.x;
    const int b = bh / NH, h = bh % NH;
    const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;

    __shared__ short k_lds[ATT_S * ATT_LDK];
    __shared__ short vt_lds[ATT_D * ATT_LDV];
    __shared__ short p_lds[4 * 32 * ATT_LDP];  // per-wave 32-row P tile

    unsigned long long seed = rng.seed, offset = rng.offset;
    if (rng.captured) {
        seed = *rng.seed_ptr;
        offset = *rng.offset_ptr + rng.intragraph;
    }

    // ---- stage K rows + V transposed --------------------------------
    // qkv element (b, t, c, h, d) at ((b*S + t)*3 + c)*NH*HD + h*HD + d
    const int64_t qkv_row = (int64_t)3 * NH * ATT_D;
    const int64_t base_b = (int64_t)b * ATT_S * qkv_row + (int64_t)h * ATT_D;
    {
        // K: 128 rows x 64 d; thread loads 8 bf16: tid 0..255 -> (row, d8)
        int tid = threadIdx.x;
        #pragma unroll
        for (int pass = 0; pass < 4; ++pass) {
            int idx = pass * 256 + tid;        // 0..1023 = 128 rows * 8 chunks
            int row = idx >> 3, d0 = (idx & 7) * 8;
            const short* src = qkv + base_b + (int64_t)row * qkv_row + NH * ATT_D + d0;
            bf16x8_a v = *reinterpret_cast<const bf16x8_a*>(src);
            *reinterpret_cast<bf16x8_a*>(&k_lds[row * ATT_LDK + d0]) = v;
            // V transposed: vt[d][key]
            const short* vsrc = src + NH * ATT_D;  // c=2
            bf16x8_a vv = *reinterpret_cast<const bf16x8_a*>(vsrc);
            #pragma unroll
            for (int q = 0; q < 8; ++q)
                vt_lds[(d0 + q) * ATT_LDV + row] = vv[q];
        }
    }
    __syncthreads();

    // ---- QK^T: wave computes rows [wave*32, wave*32+32) --------------
    const int q0 = wave * 32;
    const int fr = lane & 15;          // fragment row/col index
    const int fk = (lane >> 4) * 8;    // k-offset within K=32 fragment
    f32x4_a acc[2][8];
    #pragma unroll
    for (int i = 0; i < 2; ++i)
        #pragma unroll
        for (int j = 0; j < 8; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    #pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // K=64 in two 32-chunks
        bf16x8_a afr[2], bfr[8];
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            int q = q0 + i * 16 + fr;
            const short* src = qkv + base_b + (int64_t)q * qkv_row + ks * 32 + fk;
            afr[i] = *reinterpret_cast<const bf16x8_a*>(src);
        }
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            int key = j * 16 + fr;
            bfr[j] = *reinterpret_cast<const bf16x8_a*>(
                &k_lds[key * ATT_LDK + ks * 32 + fk]);
        }
        #pragma unroll
        for (int i = 0; i < 2; ++i)
            #pragma unroll
            for (int j = 0; j < 8; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr[i], bfr[j], acc[i][j], 0, 0, 0);
    }

    // ---- scale + mask + softmax on accumulators ----------------------
    // acc[i][j][r] holds S[q0 + i*16 + (lane>>4)*4 + r][j*16 + (lane&15)]
    float mvals[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        int key = j * 16 + (lane & 15);
        mvals[j] = mask ? att_b2f(mask[(int64_t)b * ATT_S + key]) : 0.f;
    }
    const float inv_keep = 1.f / keep_prob;
    const uint32_t thresh =
        (uint32_t)(keep_prob * 4294967296.0);  // keep if rnd < thresh
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            // row value set: acc[i][0..8][r] on the 16 lanes of my quarter
            float mx = -1e30f;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                acc[i][j][r] = acc[i][j][r] * scale + mvals[j];
                mx = fmaxf(mx, acc[i][j][r]);
            }
            #pragma unroll
            for (int off = 1; off < 16; off <<= 1)
                mx = fmaxf(mx, __shfl_xor(mx, off, 64));
            float sum = 0.f;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                float e = __expf(acc[i][j][r] - mx);
                acc[i][j][r] = e;
                sum += e;
            }
            #pragma unroll
            for (int off = 1; off < 16; off <<= 1)
                sum += __shfl_xor(sum, off, 64);
            float inv = 1.f / sum;
            #pragma unroll
            for (int j = 0; j < 8; ++j) acc[i][j][r] *= inv;
        }
    }

    // ---- dropout + save P/A + stage P into LDS -----------------------
    short* my_p = &p_lds[wave * 32 * ATT_LDP];
    const int64_t ps_base = (int64_t)bh * ATT_S * ATT_S;
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int qrow = q0 + i * 16 + (lane >> 4) * 4 + r;
            int prow = i * 16 + (lane >> 4) * 4 + r;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                int key = j * 16 + (lane & 15);
                float p = acc[i][j][r];
                short pb = att_f2b(p);
                if (p_save) p_save[ps_base + (int64_t)qrow * ATT_S + key] = pb;
                float a = p;
                if (apply_dropout) {
                    int64_t elem = ps_base + (int64_t)qrow * ATT_S + key;
                    uint32_t rnd[4];
                    philox4(seed, offset + (unsigned long long)(elem >> 2), rnd);
                    uint32_t u = rnd[elem & 3];
                    a = (u < thresh) ? p * inv_keep : 0.f;
                }
                short ab = att_f2b(a);
                if (a_save) a_save[ps_base + (int64_t)qrow * ATT_S + key] = ab;
                my_p[prow * ATT_LDP + key] = ab;
            }
        }
    }
    __syncthreads();

    // ---- PV: ctx[q][d] = sum_key A[q][key] * V[key][d] ----------------
    // M=32 (q), N=64 (d), K=128 (key) -> 2x4 fragments x 4 k-steps
    f32x4_a oacc[2][4];
    #pragma unroll
    for (int i = 0; i < 2; ++i)
        #pragma unroll
        for (int j = 0; j < 4; ++j) oacc[i][j] = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
        bf16x8_a afr[2], bfr[4];
        #pragma unroll
        for (int i = 0; i < 2; ++i)
            afr[i] = *reinterpret_cast<const bf16x8_a*>(
                &my_p[(i * 16 + fr) * ATT_LDP + ks * 32 + fk]);
        #pragma unroll
        for (int j = 0; j < 4; ++j)
            bfr[j] = *reinterpret_cast<const bf16x8_a*>(
                &vt_lds[(j * 16 + fr) * ATT_LDV + ks * 32 + fk]);
        #pragma unroll
        for (int i = 0; i < 2; ++i)
            #pragma unroll
            for (int j = 0; j < 4; ++j)
                oacc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afr[i], bfr[j], oacc[i][j], 0, 0, 0);
    }

    // ---- write ctx: out[b, q, h*64 + d] ------------------------------
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            int q = q0 + i * 16 + (lane >> 4) * 4 + r;
            int64_t row = ((int64_t)b * ATT_S + q) * NH * ATT_D + h * ATT_D;
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                int d = j * 16 + (lane & 15);
                out[row + d] = att_f2b(oacc[i][j][r]);
            }
        }
    }
}

extern "C" void launch_attn_fwd(const void* qkv, const void* mask, void* out,
                                void* p_save, void* a_save, int B, int NH,
                                float scale, float keep_prob,
                                unsigned long long seed, unsigned long long offset,
                                const void* seed_ptr, const void* offset_ptr,
                                unsigned int intragraph, int captured,
                                int apply_dropout, hipStream_t stream) {
    PhiloxArgs rng;
    rng.seed = seed;
    rng.offset = offset;
    rng.seed_ptr = (const unsigned long long*)seed_ptr;
    rng.offset_ptr = (const unsigned long long*)offset_ptr;
    rng.intragraph = intragraph;
    rng.captured = captured;
    hipLaunchKernelGGL(attn_fwd_kernel, dim3(B * NH), dim3(256), 0, stream,
                       (const short*)qkv, (const short*)mask, (short*)out,
                       (short*)p_save, (short*)a_save, B, NH, scale, keep_prob,
                       rng, apply_dropout);
}
