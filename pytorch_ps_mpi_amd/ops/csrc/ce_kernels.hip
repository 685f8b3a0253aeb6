// ce_kernels.hip — fused cross-entropy for CDNA4 (gfx950 / MI355X).
//
// For the GPT-2 head (BASELINE config 5): torch's path materializes a
// softmax over [tokens, vocab] in separate kernels; this computes per-row
// loss via a single online-logsumexp pass (bf16x8 reads, fp32 math) and the
// backward writes dlogits = gscale * (softmax - onehot) in one pass.
// Requires V % 8 == 0 (vocab padded to a multiple of 64 anyway for GEMMs).
//
//   fwd : losses[r] = lse_r - x[r, t_r]          (saves lse_r f32)
//   bwd : dlogits[r,i] = gscale * (exp(x - lse) - [i == t_r])

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define CE_BLOCK 256

struct cex8 { __hip_bfloat16 v[8]; };

__device__ __forceinline__ float cb2f(__hip_bfloat16 h) { return __bfloat162float(h); }

// merge two online-logsumexp partials
__device__ __forceinline__ void lse_merge(float& m, float& s, float m2, float s2) {
  const float M = fmaxf(m, m2);
  // exp(-inf - -inf) is nan; guard empty partials
  const float e1 = (m > -1e30f) ? __expf(m - M) : 0.0f;
  const float e2 = (m2 > -1e30f) ? __expf(m2 - M) : 0.0f;
  s = s * e1 + s2 * e2;
  m = M;
}

__global__ void __launch_bounds__(CE_BLOCK)
k_ce_fwd(const __hip_bfloat16* __restrict__ logits,
         const int64_t* __restrict__ targets, float* __restrict__ losses,
         float* __restrict__ lse_out, int64_t T, int64_t V) {
  __shared__ float red_m[4], red_s[4];
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int64_t Vg = V / 8;
  for (int64_t r = blockIdx.x; r < T; r += gridDim.x) {
    const __hip_bfloat16* row = logits + r * V;
    float m = -3e38f, s = 0.0f;
    for (int64_t g = t; g < Vg; g += CE_BLOCK) {
      const cex8 v = ((const cex8*)row)[g];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float x = cb2f(v.v[j]);
        if (x > m) {
          s = s * __expf(m - x) + 1.0f;
          m = x;
        } else {
          s += __expf(x - m);
        }
      }
    }
#pragma unroll
    for (int d = 32; d > 0; d >>= 1) {
      const float m2 = __shfl_down(m, d, 64);
      const float s2 = __shfl_down(s, d, 64);
      lse_merge(m, s, m2, s2);
    }
    if (lane == 0) { red_m[wave] = m; red_s[wave] = s; }
    __syncthreads();
    if (t == 0) {
      float M = red_m[0], S = red_s[0];
      for (int w = 1; w < 4; ++w) lse_merge(M, S, red_m[w], red_s[w]);
      const float lse = M + __logf(S);
      lse_out[r] = lse;
      const int64_t tgt = targets[r];
      losses[r] = lse - cb2f(row[tgt]);
    }
    __syncthreads();  // red_* reused next row
  }
}

__global__ void __launch_bounds__(CE_BLOCK)
k_ce_bwd(const __hip_bfloat16* __restrict__ logits,
         const int64_t* __restrict__ targets,
         const float* __restrict__ lse_v,
         __hip_bfloat16* __restrict__ dlogits, int64_t T, int64_t V,
         float gscale, const float* __restrict__ gout_dev) {
  const int t = threadIdx.x;
  if (gout_dev != nullptr) gscale *= *gout_dev;  // upstream grad, no host sync
  const int64_t Vg = V / 8;
  for (int64_t r = blockIdx.x; r < T; r += gridDim.x) {
    const __hip_bfloat16* row = logits + r * V;
    __hip_bfloat16* drow = dlogits + r * V;
    const float lse = lse_v[r];
    const int64_t tgt = targets[r];
    for (int64_t g = t; g < Vg; g += CE_BLOCK) {
      const cex8 v = ((const cex8*)row)[g];
      cex8 o;
      const int64_t base = g * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float d = __expf(cb2f(v.v[j]) - lse);
        if (base + j == tgt) d -= 1.0f;
        o.v[j] = __float2bfloat16(d * gscale);
      }
      ((cex8*)drow)[g] = o;
    }
  }
}

extern "C" {

int ps_ce_fwd(void* stream_, const void* logits, const int64_t* targets,
              float* losses, float* lse, int64_t T, int64_t V) {
  hipStream_t s = (hipStream_t)stream_;
  if (V % 8 != 0) return 9200;
  int64_t grid = T;
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(k_ce_fwd, dim3((unsigned)grid), dim3(CE_BLOCK), 0, s,
                     (const __hip_bfloat16*)logits, targets, losses, lse, T, V);
  return (int)hipGetLastError();
}

int ps_ce_bwd(void* stream_, const void* logits, const int64_t* targets,
              const float* lse, void* dlogits, int64_t T, int64_t V,
              float gscale, const float* gout_dev) {
  hipStream_t s = (hipStream_t)stream_;
  if (V % 8 != 0) return 9200;
  int64_t grid = T;
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(k_ce_bwd, dim3((unsigned)grid), dim3(CE_BLOCK), 0, s,
                     (const __hip_bfloat16*)logits, targets, lse,
                     (__hip_bfloat16*)dlogits, T, V, gscale, gout_dev);
  return (int)hipGetLastError();
}

}  // extern "C"
