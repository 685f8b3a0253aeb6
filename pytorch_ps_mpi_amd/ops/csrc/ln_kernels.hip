// ln_kernels.hip — fused row-wise LayerNorm for CDNA4 (gfx950 / MI355X).
//
// Transformer configs (ViT-B/16, GPT-2-small) spend several ms/step in
// torch's LayerNorm kernels; these run at HBM line rate with one WAVE per
// row, the whole row held in registers (D % 256 == 0, D <= 4096: each lane
// owns D/256 bf16x4 groups), fp32 statistics via wave shuffles.
//
//   forward : y = (x - mean) * rstd * gamma + beta ; saves mean/rstd [rows]
//   bwd dx  : dyg = dy*gamma; dx = rstd*(dyg - mean_r(dyg) - xhat*mean_r(dyg*xhat))
//             — single read of x,dy per row (register-resident), one pass
//   bwd dgb : dgamma = sum_rows(dy*xhat), dbeta = sum_rows(dy)
//             — column reduction, per-block partials + one atomic per column

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define LN_BLOCK 256
#define LN_MAXG 16  // max bf16x4 groups per lane -> D <= 64*4*16 = 4096

struct bnx4 { __hip_bfloat16 v[4]; };
struct bnx8 { __hip_bfloat16 v[8]; };

__device__ __forceinline__ float lb2f(__hip_bfloat16 h) { return __bfloat162float(h); }
__device__ __forceinline__ __hip_bfloat16 lf2b(float f) { return __float2bfloat16(f); }

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) v += __shfl_down(v, d, 64);
  return __shfl(v, 0, 64);
}

// one wave per row; 4 waves per block; grid-stride over rows
template <int G>
__global__ void __launch_bounds__(LN_BLOCK)
k_ln_fwd(const __hip_bfloat16* __restrict__ x, __hip_bfloat16* __restrict__ y,
         const __hip_bfloat16* __restrict__ gamma,
         const __hip_bfloat16* __restrict__ beta,
         float* __restrict__ mean_out, float* __restrict__ rstd_out,
         int64_t rows, int64_t D, float eps) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  float gw[G * 4], bw[G * 4];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const bnx4 gv = *(const bnx4*)(gamma + g * 256 + lane * 4);
    const bnx4 bv = *(const bnx4*)(beta + g * 256 + lane * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      gw[g * 4 + j] = lb2f(gv.v[j]);
      bw[g * 4 + j] = lb2f(bv.v[j]);
    }
  }
  const float invD = 1.0f / (float)D;
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (int64_t r = (int64_t)blockIdx.x * 4 + wave; r < rows; r += rstride) {
    const __hip_bfloat16* xr = x + r * D;
    float vx[G * 4];
    float s = 0.0f, ss = 0.0f;
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const bnx4 v = *(const bnx4*)(xr + g * 256 + lane * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float f = lb2f(v.v[j]);
        vx[g * 4 + j] = f;
        s += f;
        ss = fmaf(f, f, ss);
      }
    }
    s = wave_sum(s);
    ss = wave_sum(ss);
    const float mean = s * invD;
    float var = ss * invD - mean * mean;
    var = fmaxf(var, 0.0f);
    const float rstd = rsqrtf(var + eps);
    if (lane == 0 && mean_out != nullptr) {
      mean_out[r] = mean;
      rstd_out[r] = rstd;
    }
    __hip_bfloat16* yr = y + r * D;
#pragma unroll
    for (int g = 0; g < G; ++g) {
      bnx4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float xh = (vx[g * 4 + j] - mean) * rstd;
        o.v[j] = lf2b(fmaf(xh, gw[g * 4 + j], bw[g * 4 + j]));
      }
      *(bnx4*)(yr + g * 256 + lane * 4) = o;
    }
  }
}

template <int G>
__global__ void __launch_bounds__(LN_BLOCK)
k_ln_bwd_dx(const __hip_bfloat16* __restrict__ x,
            const __hip_bfloat16* __restrict__ dy,
            __hip_bfloat16* __restrict__ dx,
            const __hip_bfloat16* __restrict__ gamma,
            const float* __restrict__ mean_v, const float* __restrict__ rstd_v,
            int64_t rows, int64_t D) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  float gw[G * 4];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const bnx4 gv = *(const bnx4*)(gamma + g * 256 + lane * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) gw[g * 4 + j] = lb2f(gv.v[j]);
  }
  const float invD = 1.0f / (float)D;
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (int64_t r = (int64_t)blockIdx.x * 4 + wave; r < rows; r += rstride) {
    const float mean = mean_v[r];
    const float rstd = rstd_v[r];
    const __hip_bfloat16* xr = x + r * D;
    const __hip_bfloat16* dyr = dy + r * D;
    float xh[G * 4], dg[G * 4];
    float s1 = 0.0f, s2 = 0.0f;
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const bnx4 xv = *(const bnx4*)(xr + g * 256 + lane * 4);
      const bnx4 dv = *(const bnx4*)(dyr + g * 256 + lane * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float h = (lb2f(xv.v[j]) - mean) * rstd;
        const float d = lb2f(dv.v[j]) * gw[g * 4 + j];
        xh[g * 4 + j] = h;
        dg[g * 4 + j] = d;
        s1 += d;
        s2 = fmaf(d, h, s2);
      }
    }
    s1 = wave_sum(s1) * invD;
    s2 = wave_sum(s2) * invD;
    __hip_bfloat16* dxr = dx + r * D;
#pragma unroll
    for (int g = 0; g < G; ++g) {
      bnx4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        o.v[j] = lf2b(rstd * (dg[g * 4 + j] - s1
                              - xh[g * 4 + j] * s2));
      }
      *(bnx4*)(dxr + g * 256 + lane * 4) = o;
    }
  }
}

// column reduction for dgamma/dbeta (structure like bn reduce: each block
// owns all D columns, partial-sums a row stripe, one atomic per column)
__global__ void __launch_bounds__(LN_BLOCK)
k_ln_bwd_dgb(const __hip_bfloat16* __restrict__ x,
             const __hip_bfloat16* __restrict__ dy,
             const float* __restrict__ mean_v,
             const float* __restrict__ rstd_v,
             float* __restrict__ dgamma, float* __restrict__ dbeta,
             int64_t rows, int64_t D) {
  // bn_reduce-shaped column reduction (the first version gave each thread
  // an 8-BYTE column group and strided rows 1024 apart: half-rate loads and
  // no locality — measured 5x off roofline on GPT-2).  Here a wave owns a
  // 64-channel tile of blockIdx.x: lane = (row_sub = lane>>3, ch8 =
  // (lane&7)*8), 16B loads, 8 rows x 64 channels per wave pass; lanes 8
  // apart share channels and reduce via shfl; one atomicAdd per channel
  // per block (grid.y is capped, so atomic chains stay short).
  __shared__ float red[2][4][64];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int ch8 = (lane & 7) * 8;
  const int rsub = lane >> 3;
  const int64_t cbase = (int64_t)blockIdx.x * 64;

  float sg[8], sb[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) { sg[j] = 0.0f; sb[j] = 0.0f; }

  const int64_t rstride = (int64_t)gridDim.y * 32;
  for (int64_t r = (int64_t)blockIdx.y * 32 + wave * 8 + rsub; r < rows;
       r += rstride) {
    const float m = mean_v[r];
    const float rs = rstd_v[r];
    const int64_t off = r * D + cbase + ch8;
    const bnx8 xv = *(const bnx8*)(x + off);
    const bnx8 dv = *(const bnx8*)(dy + off);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float d = lb2f(dv.v[j]);
      sg[j] = fmaf(d, (lb2f(xv.v[j]) - m) * rs, sg[j]);
      sb[j] += d;
    }
  }
#pragma unroll
  for (int d = 8; d < 64; d <<= 1) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sg[j] += __shfl_xor(sg[j], d, 64);
      sb[j] += __shfl_xor(sb[j], d, 64);
    }
  }
  if (lane < 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      red[0][wave][ch8 + j] = sg[j];
      red[1][wave][ch8 + j] = sb[j];
    }
  }
  __syncthreads();
  if (wave == 0) {
    const float g = red[0][0][lane] + red[0][1][lane] + red[0][2][lane]
                  + red[0][3][lane];
    const float b = red[1][0][lane] + red[1][1][lane] + red[1][2][lane]
                  + red[1][3][lane];
    atomicAdd(&dgamma[cbase + lane], g);
    atomicAdd(&dbeta[cbase + lane], b);
  }
}

// plain column sum (Linear bias gradients): same shape as k_ln_bwd_dgb
__global__ void __launch_bounds__(LN_BLOCK)
k_colsum(const __hip_bfloat16* __restrict__ dy, float* __restrict__ out,
         int64_t rows, int64_t D) {
  __shared__ float red[4][64];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int ch8 = (lane & 7) * 8;
  const int rsub = lane >> 3;
  const int64_t cbase = (int64_t)blockIdx.x * 64;
  float sb[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) sb[j] = 0.0f;
  const int64_t rstride = (int64_t)gridDim.y * 32;
  for (int64_t r = (int64_t)blockIdx.y * 32 + wave * 8 + rsub; r < rows;
       r += rstride) {
    const bnx8 dv = *(const bnx8*)(dy + r * D + cbase + ch8);
#pragma unroll
    for (int j = 0; j < 8; ++j) sb[j] += lb2f(dv.v[j]);
  }
#pragma unroll
  for (int d = 8; d < 64; d <<= 1)
#pragma unroll
    for (int j = 0; j < 8; ++j) sb[j] += __shfl_xor(sb[j], d, 64);
  if (lane < 8)
#pragma unroll
    for (int j = 0; j < 8; ++j) red[wave][ch8 + j] = sb[j];
  __syncthreads();
  if (wave == 0)
    atomicAdd(&out[cbase + lane],
              red[0][lane] + red[1][lane] + red[2][lane] + red[3][lane]);
}

extern "C" {

int ps_colsum(void* stream_, const void* dy, float* out, int64_t rows,
              int64_t D) {
  hipStream_t s = (hipStream_t)stream_;
  if (D % 64 != 0) return 9100;
  hipError_t e = hipMemsetAsync(out, 0, D * sizeof(float), s);
  if (e) return (int)e;
  const int ct = (int)(D / 64);
  int64_t yb = (rows + 31) / 32;
  int64_t cap = 1024 / ct;
  if (cap < 1) cap = 1;
  if (yb > cap) yb = cap;
  hipLaunchKernelGGL(k_colsum, dim3((unsigned)ct, (unsigned)yb),
                     dim3(LN_BLOCK), 0, s, (const __hip_bfloat16*)dy, out,
                     rows, D);
  return (int)hipGetLastError();
}


int ps_ln_fwd(void* stream_, const void* x, void* y, const void* gamma,
              const void* beta, float* mean, float* rstd, int64_t rows,
              int64_t D, float eps) {
  hipStream_t s = (hipStream_t)stream_;
  if (D % 256 != 0 || D > 4096) return 9100;
  int64_t grid = (rows + 3) / 4;
  if (grid > 2048) grid = 2048;
  const int G = (int)(D / 256);
#define LN_F(GG)                                                              \
  case GG:                                                                    \
    hipLaunchKernelGGL((k_ln_fwd<GG>), dim3((unsigned)grid), dim3(LN_BLOCK),  \
                       0, s, (const __hip_bfloat16*)x, (__hip_bfloat16*)y,    \
                       (const __hip_bfloat16*)gamma,                          \
                       (const __hip_bfloat16*)beta, mean, rstd, rows, D,      \
                       eps);                                                  \
    break;
  switch (G) {
    LN_F(1) LN_F(2) LN_F(3) LN_F(4) LN_F(5) LN_F(6) LN_F(8) LN_F(12) LN_F(16)
    default: return 9101;
  }
#undef LN_F
  return (int)hipGetLastError();
}

int ps_ln_bwd_dx(void* stream_, const void* x, const void* dy, void* dx,
                 const void* gamma, const float* mean, const float* rstd,
                 int64_t rows, int64_t D) {
  hipStream_t s = (hipStream_t)stream_;
  if (D % 256 != 0 || D > 4096) return 9100;
  int64_t grid = (rows + 3) / 4;
  if (grid > 2048) grid = 2048;
  const int G = (int)(D / 256);
#define LN_B(GG)                                                              \
  case GG:                                                                    \
    hipLaunchKernelGGL((k_ln_bwd_dx<GG>), dim3((unsigned)grid),               \
                       dim3(LN_BLOCK), 0, s, (const __hip_bfloat16*)x,        \
                       (const __hip_bfloat16*)dy, (__hip_bfloat16*)dx,        \
                       (const __hip_bfloat16*)gamma, mean, rstd, rows, D);    \
    break;
  switch (G) {
    LN_B(1) LN_B(2) LN_B(3) LN_B(4) LN_B(5) LN_B(6) LN_B(8) LN_B(12) LN_B(16)
    default: return 9101;
  }
#undef LN_B
  return (int)hipGetLastError();
}

int ps_ln_bwd_dgb(void* stream_, const void* x, const void* dy,
                  const float* mean, const float* rstd, float* dgamma,
                  float* dbeta, int64_t rows, int64_t D) {
  hipStream_t s = (hipStream_t)stream_;
  hipError_t e = hipMemsetAsync(dgamma, 0, D * sizeof(float), s);
  if (e) return (int)e;
  e = hipMemsetAsync(dbeta, 0, D * sizeof(float), s);
  if (e) return (int)e;
  const int ct = (int)(D / 64);  // D % 256 == 0 checked by every LN entry
  int64_t yb = (rows + 31) / 32;
  int64_t cap = 1024 / ct;       // short per-channel atomic chains
  if (cap < 1) cap = 1;
  if (yb > cap) yb = cap;
  hipLaunchKernelGGL(k_ln_bwd_dgb, dim3((unsigned)ct, (unsigned)yb),
                     dim3(LN_BLOCK), 0, s, (const __hip_bfloat16*)x,
                     (const __hip_bfloat16*)dy, mean, rstd, dgamma, dbeta,
                     rows, D);
  return (int)hipGetLastError();
}

}  // extern "C"
