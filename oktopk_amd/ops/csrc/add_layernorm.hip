// Fused residual-add + LayerNorm (bf16), forward + backward.
//
// BERT runs y = LN(x + sub(x)) twice per layer (reference BertSelfOutput /
// BertOutput, modeling.py); unfused that is an add kernel + RowwiseMoments +
// apply kernel forward and three backward kernels per site.  Fused: one
// kernel each way (plus a trivial fp32->bf16 cast for dgamma/dbeta).
//
// Layout: one wave per row (H = hidden size, H % 128 == 0), lane l owns
// bf16x2 pairs at columns 2*(l + 64*j) — coalesced 256 B per wave
// instruction; row statistics by wave shuffle reduction.  Backward column
// sums (dgamma/dbeta) accumulate with fp32 atomics.

#include <hip/hip_runtime.h>
#include <cstdint>

#define LN_BLOCK 256

typedef __attribute__((ext_vector_type(2))) short bf16x2_t;

__device__ __forceinline__ float b2f(short u) {
    union { float f; uint32_t i; } c;
    c.i = ((uint32_t)(uint16_t)u) << 16;
    return c.f;
}

__device__ __forceinline__ short f2b(float f) {
    union { float f; uint32_t i; } c;
    c.f = f;
    uint32_t lsb = (c.i >> 16) & 1;
    c.i += 0x7fff + lsb;
    return (short)(c.i >> 16);
}

__device__ __forceinline__ float wave_sum(float v) {
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    return __shfl(v, 0, 64);
}

// fwd: s = bf16(x + r); y = (s - mean)*rstd*gamma + beta
// saves s (bf16) and mean/rstd (fp32) for backward
__global__ void add_ln_fwd_kernel(const short* __restrict__ x,
                                  const short* __restrict__ r,
                                  const short* __restrict__ gamma,
                                  const short* __restrict__ beta,
                                  short* __restrict__ y,
                                  short* __restrict__ s_out,
                                  float* __restrict__ mean_out,
                                  float* __restrict__ rstd_out,
                                  int64_t rows, int H, float eps) {
    int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    int64_t row = (int64_t)blockIdx.x * (LN_BLOCK / 64) + wave;
    if (row >= rows) return;
    const int npairs = H / 128;  // bf16x2 pairs per lane
    const bf16x2_t* xp = reinterpret_cast<const bf16x2_t*>(x + row * H);
    const bf16x2_t* rp = reinterpret_cast<const bf16x2_t*>(r + row * H);
    bf16x2_t* sp = reinterpret_cast<bf16x2_t*>(s_out + row * H);
    bf16x2_t* yp = reinterpret_cast<bf16x2_t*>(y + row * H);
    const bf16x2_t* gp = reinterpret_cast<const bf16x2_t*>(gamma);
    const bf16x2_t* bp = reinterpret_cast<const bf16x2_t*>(beta);

    float sv[16][2];  // up to H=2048
    float acc = 0.f, acc2 = 0.f;
    for (int j = 0; j < npairs; ++j) {
        int c = lane + 64 * j;
        bf16x2_t xv = xp[c], rv = rp[c];
        // match the unfused path: the sum is materialised in bf16 first
        short s0 = f2b(b2f(xv[0]) + b2f(rv[0]));
        short s1 = f2b(b2f(xv[1]) + b2f(rv[1]));
        bf16x2_t s2; s2[0] = s0; s2[1] = s1;
        sp[c] = s2;
        float f0 = b2f(s0), f1 = b2f(s1);
        sv[j][0] = f0; sv[j][1] = f1;
        acc += f0 + f1;
        acc2 += f0 * f0 + f1 * f1;
    }
    float mean = wave_sum(acc) / H;
    float var = wave_sum(acc2) / H - mean * mean;
    float rstd = rsqrtf(var + eps);
    if (lane == 0) {
        mean_out[row] = mean;
        rstd_out[row] = rstd;
    }
    for (int j = 0; j < npairs; ++j) {
        int c = lane + 64 * j;
        bf16x2_t gv = gp[c], bv = bp[c];
        bf16x2_t o;
        o[0] = f2b((sv[j][0] - mean) * rstd * b2f(gv[0]) + b2f(bv[0]));
        o[1] = f2b((sv[j][1] - mean) * rstd * b2f(gv[1]) + b2f(bv[1]));
        yp[c] = o;
    }
}

// bwd: gx = rstd * (gyh - mean(gyh) - h * mean(gyh*h)), h = (s-mean)*rstd,
// gyh = gy*gamma; dgamma += gy*h, dbeta += gy (fp32 atomics)
// Per-lane caches of the whole row would be runtime-indexed arrays ->
// scratch (measured 272 B/lane, bench regression); instead each row is read
// twice (pass 1 sums, pass 2 recomputes — L1/L2-resident the second time).
// dgamma/dbeta accumulate into LDS column partials; ONE global atomic per
// column per block (a per-element global atomicAdd was ~1.5M atomics per
// call).  Grid is small + row-strided so flush contention stays low.
__global__ void add_ln_bwd_kernel(const short* __restrict__ gy,
                                  const short* __restrict__ s,
                                  const float* __restrict__ mean_in,
                                  const float* __restrict__ rstd_in,
                                  const short* __restrict__ gamma,
                                  short* __restrict__ gx,
                                  float* __restrict__ dgamma,
                                  float* __restrict__ dbeta,
                                  int64_t rows, int H) {
    __shared__ float col_g[2048];
    __shared__ float col_b[2048];
    for (int c = threadIdx.x; c < H; c += LN_BLOCK) {
        col_g[c] = 0.f;
        col_b[c] = 0.f;
    }
    __syncthreads();

    int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int npairs = H / 128;
    const int rpb = LN_BLOCK / 64;
    const bf16x2_t* gp = reinterpret_cast<const bf16x2_t*>(gamma);
    for (int64_t row = (int64_t)blockIdx.x * rpb + wave; row < rows;
         row += (int64_t)gridDim.x * rpb) {
        const bf16x2_t* gyp = reinterpret_cast<const bf16x2_t*>(gy + row * H);
        const bf16x2_t* sp = reinterpret_cast<const bf16x2_t*>(s + row * H);
        bf16x2_t* gxp = reinterpret_cast<bf16x2_t*>(gx + row * H);
        float mean = mean_in[row], rstd = rstd_in[row];

        float sum_gyh = 0.f, sum_gyh_h = 0.f;
        for (int j = 0; j < npairs; ++j) {
            int c = lane + 64 * j;
            bf16x2_t g = gyp[c], sv = sp[c], gm = gp[c];
            #pragma unroll
            for (int q = 0; q < 2; ++q) {
                float h = (b2f(sv[q]) - mean) * rstd;
                float gyh = b2f(g[q]) * b2f(gm[q]);
                sum_gyh += gyh;
                sum_gyh_h += gyh * h;
            }
        }
        float m1 = wave_sum(sum_gyh) / H;
        float m2 = wave_sum(sum_gyh_h) / H;
        for (int j = 0; j < npairs; ++j) {
            int c = lane + 64 * j;
            bf16x2_t g = gyp[c], sv = sp[c], gm = gp[c];
            bf16x2_t o;
            #pragma unroll
            for (int q = 0; q < 2; ++q) {
                float gf = b2f(g[q]);
                float h = (b2f(sv[q]) - mean) * rstd;
                float gyh = gf * b2f(gm[q]);
                o[q] = f2b(rstd * (gyh - m1 - h * m2));
                atomicAdd(&col_g[2 * c + q], gf * h);
                atomicAdd(&col_b[2 * c + q], gf);
            }
            gxp[c] = o;
        }
    }
    __syncthreads();
    for (int c = threadIdx.x; c < H; c += LN_BLOCK) {
        atomicAdd(&dgamma[c], col_g[c]);
        atomicAdd(&dbeta[c], col_b[c]);
    }
}

extern "C" void launch_add_ln_fwd(const void* x, const void* r, const void* gamma,
                                  const void* beta, void* y, void* s_out,
                                  float* mean_out, float* rstd_out, int64_t rows,
                                  int H, float eps, hipStream_t stream) {
    int rows_per_block = LN_BLOCK / 64;
    int64_t grid = (rows + rows_per_block - 1) / rows_per_block;
    hipLaunchKernelGGL(add_ln_fwd_kernel, dim3((uint32_t)grid), dim3(LN_BLOCK), 0,
                       stream, (const short*)x, (const short*)r, (const short*)gamma,
                       (const short*)beta, (short*)y, (short*)s_out, mean_out,
                       rstd_out, rows, H, eps);
}

extern "C" void launch_add_ln_bwd(const void* gy, const void* s, const float* mean,
                                  const float* rstd, const void* gamma, void* gx,
                                  float* dgamma, float* dbeta, int64_t rows, int H,
                                  hipStream_t stream) {
    int rows_per_block = LN_BLOCK / 64;
    int64_t grid = (rows + rows_per_block - 1) / rows_per_block;
    if (grid > 128) grid = 128;  // row-strided; bounds the column-flush atomics
    hipLaunchKernelGGL(add_ln_bwd_kernel, dim3((uint32_t)grid), dim3(LN_BLOCK), 0,
                       stream, (const short*)gy, (const short*)s, mean, rstd,
                       (const short*)gamma, (short*)gx, dgamma, dbeta, rows, H);
}
