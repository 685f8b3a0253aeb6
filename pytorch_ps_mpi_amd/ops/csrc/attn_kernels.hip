// attn_kernels.hip — flash-attention fwd+bwd for CDNA4 (gfx950).
//
// Status: the ON-GPU ORACLE for the MFMA-tiled production kernels
// (mfma_attn_kernels.hip, round 2): exact shape-generic one-wave-per-row
// flash fwd+bwd, hardware-validated against torch SDPA
// (tests/test_gpu_attn.py, PS_AMD_ATTN=ref).  Slow by design (~2 TF).
//
// Shape contract: q,k,v,o,do,dq,dk,dv are [B, H, N, D] bf16 contiguous with
// D == 64 (one lane per head dim).  Online-softmax forward saves
// lse[B*H*N] (fp32); backward uses the standard flash decomposition:
//   delta_q = do_q . o_q
//   p = exp(s*scale - lse_q);   ds = p * (dp - delta_q),  dp = do_q . v_k
//   dq_q = scale * sum_k ds * k;  dk_k = scale * sum_q ds * q
//   dv_k = sum_q p * do_q
// One wave per query row (fwd, dq) / key row (dk,dv); scores via full-wave
// shuffle reductions.  O(N) serial dot products per row: simple, exact
// shape-generic, slow — the MFMA-tiled rewrite is the round-2 task.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define AT_BLOCK 256
#define AT_D 64

__device__ __forceinline__ float ab2f(__hip_bfloat16 h) { return __bfloat162float(h); }

// sum over all 64 lanes, result in every lane
__device__ __forceinline__ float wave_allsum(float v) {
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) v += __shfl_xor(v, d, 64);
  return v;
}

__global__ void __launch_bounds__(AT_BLOCK)
k_attn_fwd(const __hip_bfloat16* __restrict__ q,
           const __hip_bfloat16* __restrict__ k,
           const __hip_bfloat16* __restrict__ v,
           __hip_bfloat16* __restrict__ o, float* __restrict__ lse,
           int64_t rows, int64_t N, float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (int64_t r = (int64_t)blockIdx.x * 4 + wave; r < rows; r += rstride) {
    const int64_t bh = r / N;
    const int64_t qi = r % N;
    const float qv = ab2f(q[r * AT_D + lane]);
    float m = -3e38f, l = 0.0f, acc = 0.0f;
    const int64_t kmax = causal ? (qi + 1) : N;
    const __hip_bfloat16* kbase = k + bh * N * AT_D;
    const __hip_bfloat16* vbase = v + bh * N * AT_D;
    for (int64_t kj = 0; kj < kmax; ++kj) {
      const float kv = ab2f(kbase[kj * AT_D + lane]);
      const float s = wave_allsum(qv * kv) * scale;
      const float mn = fmaxf(m, s);
      const float alpha = __expf(m - mn);
      const float p = __expf(s - mn);
      l = l * alpha + p;
      acc = acc * alpha + p * ab2f(vbase[kj * AT_D + lane]);
      m = mn;
    }
    o[r * AT_D + lane] = __float2bfloat16(acc / l);
    if (lane == 0) lse[r] = m + __logf(l);
  }
}

// delta[r] = do_r . o_r
__global__ void __launch_bounds__(AT_BLOCK)
k_attn_delta(const __hip_bfloat16* __restrict__ dout,
             const __hip_bfloat16* __restrict__ o,
             float* __restrict__ delta, int64_t rows) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (int64_t r = (int64_t)blockIdx.x * 4 + wave; r < rows; r += rstride) {
    const float d = wave_allsum(ab2f(dout[r * AT_D + lane])
                                * ab2f(o[r * AT_D + lane]));
    if (lane == 0) delta[r] = d;
  }
}

__global__ void __launch_bounds__(AT_BLOCK)
k_attn_bwd_dq(const __hip_bfloat16* __restrict__ q,
              const __hip_bfloat16* __restrict__ k,
              const __hip_bfloat16* __restrict__ v,
              const __hip_bfloat16* __restrict__ dout,
              const float* __restrict__ lse, const float* __restrict__ delta,
              __hip_bfloat16* __restrict__ dq, int64_t rows, int64_t N,
              float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (int64_t r = (int64_t)blockIdx.x * 4 + wave; r < rows; r += rstride) {
    const int64_t bh = r / N;
    const int64_t qi = r % N;
    const float qv = ab2f(q[r * AT_D + lane]);
    const float dov = ab2f(dout[r * AT_D + lane]);
    const float lse_r = lse[r];
    const float delta_r = delta[r];
    float dqv = 0.0f;
    const int64_t kmax = causal ? (qi + 1) : N;
    const __hip_bfloat16* kbase = k + bh * N * AT_D;
    const __hip_bfloat16* vbase = v + bh * N * AT_D;
    for (int64_t kj = 0; kj < kmax; ++kj) {
      const float kv = ab2f(kbase[kj * AT_D + lane]);
      const float s = wave_allsum(qv * kv) * scale;
      const float p = __expf(s - lse_r);
      const float dp = wave_allsum(dov * ab2f(vbase[kj * AT_D + lane]));
      const float ds = p * (dp - delta_r);
      dqv = fmaf(ds, kv, dqv);
    }
    dq[r * AT_D + lane] = __float2bfloat16(dqv * scale);
  }
}

__global__ void __launch_bounds__(AT_BLOCK)
k_attn_bwd_dkv(const __hip_bfloat16* __restrict__ q,
               const __hip_bfloat16* __restrict__ k,
               const __hip_bfloat16* __restrict__ v,
               const __hip_bfloat16* __restrict__ dout,
               const float* __restrict__ lse, const float* __restrict__ delta,
               __hip_bfloat16* __restrict__ dk,
               __hip_bfloat16* __restrict__ dv, int64_t rows, int64_t N,
               float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (int64_t r = (int64_t)blockIdx.x * 4 + wave; r < rows; r += rstride) {
    const int64_t bh = r / N;
    const int64_t kj = r % N;
    const float kv = ab2f(k[r * AT_D + lane]);
    const float vv = ab2f(v[r * AT_D + lane]);
    float dkv = 0.0f, dvv = 0.0f;
    const int64_t q0 = causal ? kj : 0;
    const __hip_bfloat16* qbase = q + bh * N * AT_D;
    const __hip_bfloat16* dobase = dout + bh * N * AT_D;
    for (int64_t qi = q0; qi < N; ++qi) {
      const float qv = ab2f(qbase[qi * AT_D + lane]);
      const float s = wave_allsum(qv * kv) * scale;
      const float p = __expf(s - lse[bh * N + qi]);
      const float dov = ab2f(dobase[qi * AT_D + lane]);
      dvv = fmaf(p, dov, dvv);
      const float dp = wave_allsum(dov * vv);
      const float ds = p * (dp - delta[bh * N + qi]);
      dkv = fmaf(ds, qv, dkv);
    }
    dk[r * AT_D + lane] = __float2bfloat16(dkv * scale);
    dv[r * AT_D + lane] = __float2bfloat16(dvv);
  }
}

extern "C" {

static inline unsigned at_grid(int64_t rows) {
  int64_t b = (rows + 3) / 4;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (unsigned)b;
}

int ps_attn_fwd(void* stream_, const void* q, const void* k, const void* v,
                void* o, float* lse, int64_t rows, int64_t N, float scale,
                int causal) {
  hipStream_t s = (hipStream_t)stream_;
  hipLaunchKernelGGL(k_attn_fwd, dim3(at_grid(rows)), dim3(AT_BLOCK), 0, s,
                     (const __hip_bfloat16*)q, (const __hip_bfloat16*)k,
                     (const __hip_bfloat16*)v, (__hip_bfloat16*)o, lse, rows,
                     N, scale, causal);
  return (int)hipGetLastError();
}

int ps_attn_bwd(void* stream_, const void* q, const void* k, const void* v,
                const void* o, const void* dout, const float* lse,
                float* delta, void* dq, void* dk, void* dv, int64_t rows,
                int64_t N, float scale, int causal) {
  hipStream_t s = (hipStream_t)stream_;
  const dim3 grid(at_grid(rows)), block(AT_BLOCK);
  hipLaunchKernelGGL(k_attn_delta, grid, block, 0, s,
                     (const __hip_bfloat16*)dout, (const __hip_bfloat16*)o,
                     delta, rows);
  hipLaunchKernelGGL(k_attn_bwd_dq, grid, block, 0, s,
                     (const __hip_bfloat16*)q, (const __hip_bfloat16*)k,
                     (const __hip_bfloat16*)v, (const __hip_bfloat16*)dout,
                     lse, delta, (__hip_bfloat16*)dq, rows, N, scale, causal);
  hipLaunchKernelGGL(k_attn_bwd_dkv, grid, block, 0, s,
                     (const __hip_bfloat16*)q, (const __hip_bfloat16*)k,
                     (const __hip_bfloat16*)v, (const __hip_bfloat16*)dout,
                     lse, delta, (__hip_bfloat16*)dk, (__hip_bfloat16*)dv,
                     rows, N, scale, causal);
  return (int)hipGetLastError();
}

}  // extern "C"
