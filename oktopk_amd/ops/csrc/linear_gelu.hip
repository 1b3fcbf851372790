// Fused Linear + bias + GELU for the BERT MLP (bf16 in, bf16 out).
//
// Capability parity with the reference's fused-gelu LinearActivation
// (/root/reference/BERT/bert/transformers/modeling.py:75) — but as a real
// hand-written CDNA4 MFMA kernel instead of a torch jit fusion:
//   y = gelu(x @ W^T + b),  x:[M,K] bf16 row-major, W:[N,K] bf16 row-major
// (nn.Linear weight layout), y:[M,N] bf16; optionally also writes the
// pre-activation z = x@W^T+b (bf16) for the backward pass.
//
// Structure (cdna_hip_programming.md §5 anatomy, correctness-first tier):
// 128x128 output tile per 256-thread block (4 waves, 64x64 each as 4x4
// fragments of 16x16), BK=32, both operands staged K-contiguous into
// padded LDS (row stride +8 bf16 breaks the 4-way ds_read_b128 bank
// conflict, §6 G4), v_mfma_f32_16x16x32_bf16 inner loop, erf-GELU epilogue
// fused into the accumulator writeback.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define BM 128
#define BN 128
#define BK 32
#define PAD 8  // bf16 elements of row padding in LDS
#define LDK (BK + PAD)

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ float bf16_to_f32(short u) {
    union { float f; uint32_t i; } c;
    c.i = ((uint32_t)(uint16_t)u) << 16;
    return c.f;
}

__device__ __forceinline__ short f32_to_bf16(float f) {
    // round-to-nearest-even
    union { float f; uint32_t i; } c;
    c.f = f;
    uint32_t lsb = (c.i >> 16) & 1;
    c.i += 0x7fff + lsb;
    return (short)(c.i >> 16);
}

__device__ __forceinline__ float gelu_erf(float x) {
    return 0.5f * x * (1.0f + erff(x * 0.70710678118654752f));
}

__global__ void __launch_bounds__(256)
linear_gelu_kernel(const short* __restrict__ x,   // [M,K] bf16 bits
                   const short* __restrict__ w,   // [N,K] bf16 bits
                   const float* __restrict__ bias,  // [N] fp32 (or null)
                   short* __restrict__ y,        // [M,N] bf16 out (gelu)
                   short* __restrict__ z,        // [M,N] bf16 pre-act (or null)
                   int M, int N, int K, int apply_gelu) {
    __shared__ short lds_a[BM * LDK];
    __shared__ short lds_b[BN * LDK];

    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 1, wc = wave & 1;  // wave tile (64x64) in block

    f32x4 acc[4][4];
    #pragma unroll
    for (int i = 0; i < 4; ++i)
        #pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    for (int k0 = 0; k0 < K; k0 += BK) {
        // stage A and B tiles: 128 rows x 32 k each; thread loads 8 bf16
        // (16 B) per pass, 2 passes per tile (256 threads x 8 x 2 = 4096)
        #pragma unroll
        for (int pass = 0; pass < 2; ++pass) {
            int idx = pass * 256 + tid;      // 0..511
            int row = idx >> 2;              // /4: 4 chunks of 8 per row
            int col = (idx & 3) * 8;
            // A
            int gm = m0 + row;
            bf16x8 va = {0, 0, 0, 0, 0, 0, 0, 0};
            if (gm < M && k0 + col < K)
                va = *reinterpret_cast<const bf16x8*>(&x[(int64_t)gm * K + k0 + col]);
            *reinterpret_cast<bf16x8*>(&lds_a[row * LDK + col]) = va;
            // B
            int gn = n0 + row;
            bf16x8 vb = {0, 0, 0, 0, 0, 0, 0, 0};
            if (gn < N && k0 + col < K)
                vb = *reinterpret_cast<const bf16x8*>(&w[(int64_t)gn * K + k0 + col]);
            *reinterpret_cast<bf16x8*>(&lds_b[row * LDK + col]) = vb;
        }
        __syncthreads();

        // fragments: lane holds row (lane&15), k = (lane>>4)*8 .. +8
        const int fr = lane & 15;
        const int fk = (lane >> 4) * 8;
        bf16x8 afrag[4], bfrag[4];
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            int am = wr * 64 + i * 16 + fr;
            afrag[i] = *reinterpret_cast<const bf16x8*>(&lds_a[am * LDK + fk]);
            int bn = wc * 64 + i * 16 + fr;
            bfrag[i] = *reinterpret_cast<const bf16x8*>(&lds_b[bn * LDK + fk]);
        }
        #pragma unroll
        for (int i = 0; i < 4; ++i)
            #pragma unroll
            for (int j = 0; j < 4; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
        __syncthreads();
    }

    // epilogue: C row = (lane>>4)*4 + reg, col = lane&15 within each 16x16
    const int crow = (lane >> 4) * 4;
    const int ccol = lane & 15;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            int gn = n0 + wc * 64 + j * 16 + ccol;
            if (gn >= N) continue;
            float b = bias ? bias[gn] : 0.f;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                int gm = m0 + wr * 64 + i * 16 + crow + r;
                if (gm >= M) continue;
                float pre = acc[i][j][r] + b;
                if (z) z[(int64_t)gm * N + gn] = f32_to_bf16(pre);
                float out = apply_gelu ? gelu_erf(pre) : pre;
                y[(int64_t)gm * N + gn] = f32_to_bf16(out);
            }
        }
    }
}

extern "C" void launch_linear_gelu(const void* x, const void* w, const float* bias,
                                   void* y, void* z, int M, int N, int K,
                                   int apply_gelu, hipStream_t stream) {
    dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
    hipLaunchKernelGGL(linear_gelu_kernel, grid, dim3(256), 0, stream,
                       (const short*)x, (const short*)w, bias, (short*)y,
                       (short*)z, M, N, K, apply_gelu);
}
