// bn_kernels.hip — fused NHWC BatchNorm for CDNA4 (gfx950 / MI355X).
//
// Motivation (measured, profiles/resnet50_steady_r01.md): PyTorch's native
// batch_norm_*_channels_last kernels are 57% of a ResNet-50 bf16 step on
// MI355X at ~0.5 TB/s effective.  These kernels run the same math at HBM
// line rate: bf16x8 (16 B/lane) streams, fp32 accumulation, one 64-channel
// tile per wave, per-channel reduction via in-wave shuffles + one global
// atomic per block.
//
// Layout: x is NHWC-contiguous ("channels_last"), rows = N*H*W, row stride C,
// C % 64 == 0 (all ResNet/ViT widths; other C fall back to torch).
//
// Fusions (zero extra memory traffic for the masks):
//   forward:  y = bn(x)            | y = relu(bn(x))  | y = relu(bn(x) + z)
//   backward: relu mask recomputed as (bn(x) [+ z]) > 0 from x (and z),
//             using the saved per-channel scale/shift — no y tensor saved.
//
// Gradient formulas (count = rows):
//   xhat = (x - mean) * invstd
//   dgamma = sum(dy_eff * xhat); dbeta = sum(dy_eff)
//   dx = a*dy_eff + bx*x + cc   with per-channel
//        a  = gamma*invstd
//        bx = -a*invstd*dgamma/count
//        cc = -a*dbeta/count + a*invstd*dgamma*mean/count
//   dz = dy_eff (the residual branch), where dy_eff = mask ? dy : 0.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define BN_BLOCK 256
#define BN_CT 64  // channels per tile (one wave width)

struct bnx8 { __hip_bfloat16 v[8]; };

__device__ __forceinline__ float b2f(__hip_bfloat16 h) { return __bfloat162float(h); }
__device__ __forceinline__ __hip_bfloat16 f2b(float f) { return __float2bfloat16(f); }

// ---------------------------------------------------------------------------
// F1 / B1: per-channel partial reductions.
// Each block: 4 waves; a wave handles the 64-channel tile blockIdx.x, with
// lane = (row_sub = lane>>3, ch8 = lane&7): 8 rows x 64 channels per pass.
// Wave-internal reduce over rows via shfl_xor(8,16,32); cross-wave via LDS;
// one atomicAdd per channel per block.
// ---------------------------------------------------------------------------

template <bool IS_B1, bool HAS_Z>
__global__ void __launch_bounds__(BN_BLOCK)
k_bn_reduce(const __hip_bfloat16* __restrict__ x,
            const __hip_bfloat16* __restrict__ dy,
            const __hip_bfloat16* __restrict__ z,
            const float* __restrict__ scale,   // B1: bn scale (gamma*invstd)
            const float* __restrict__ shift,   // B1: bn shift
            float* __restrict__ out0,          // F1: sum(x)    B1: sum(dy_eff)
            float* __restrict__ out1,          // F1: sum(x^2)  B1: sum(dy_eff*x)
            int64_t rows, int64_t C, int relu) {
  __shared__ float red[2][4][BN_CT];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int ch8 = (lane & 7) * 8;
  const int rsub = lane >> 3;
  const int64_t cbase = (int64_t)blockIdx.x * BN_CT;

  float a0[8], a1[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) { a0[j] = 0.0f; a1[j] = 0.0f; }

  float sc[8], sh[8];
  if (IS_B1) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sc[j] = scale[cbase + ch8 + j];
      sh[j] = shift[cbase + ch8 + j];
    }
  }

  const int64_t rstride = (int64_t)gridDim.y * 32;
  for (int64_t r = (int64_t)blockIdx.y * 32 + wave * 8 + rsub; r < rows;
       r += rstride) {
    const int64_t off = r * C + cbase + ch8;
    const bnx8 xv = *(const bnx8*)(x + off);
    if (IS_B1) {
      const bnx8 dyv = *(const bnx8*)(dy + off);
      bnx8 zv;
      if (HAS_Z) zv = *(const bnx8*)(z + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xf = b2f(xv.v[j]);
        float g = b2f(dyv.v[j]);
        if (relu) {
          float y = fmaf(xf, sc[j], sh[j]);
          if (HAS_Z) y += b2f(zv.v[j]);
          if (y <= 0.0f) g = 0.0f;
        }
        a0[j] += g;
        a1[j] += g * xf;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xf = b2f(xv.v[j]);
        a0[j] += xf;
        a1[j] = fmaf(xf, xf, a1[j]);
      }
    }
  }

  // reduce the 8 row-subgroups (lanes 8 apart share channels)
#pragma unroll
  for (int d = 8; d < 64; d <<= 1) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      a0[j] += __shfl_xor(a0[j], d, 64);
      a1[j] += __shfl_xor(a1[j], d, 64);
    }
  }
  if (lane < 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      red[0][wave][ch8 + j] = a0[j];
      red[1][wave][ch8 + j] = a1[j];
    }
  }
  __syncthreads();
  if (wave == 0) {
    const float s0 = red[0][0][lane] + red[0][1][lane] + red[0][2][lane]
                   + red[0][3][lane];
    const float s1 = red[1][0][lane] + red[1][1][lane] + red[1][2][lane]
                   + red[1][3][lane];
    atomicAdd(&out0[cbase + lane], s0);
    atomicAdd(&out1[cbase + lane], s1);
  }
}

// ---------------------------------------------------------------------------
// F2: finalize statistics -> scale/shift (+ running stats update)
// one thread per channel (grid-stride).  gamma/beta/running in T (bf16|f32).
// saves mean/invstd as f32 for backward.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(BN_BLOCK)
k_bn_finalize(const float* __restrict__ psum, const float* __restrict__ psumsq,
              const T* __restrict__ gamma, const T* __restrict__ beta,
              T* __restrict__ rmean, T* __restrict__ rvar,
              float* __restrict__ mean_out, float* __restrict__ invstd_out,
              float* __restrict__ scale, float* __restrict__ shift,
              int64_t C, float count, float eps, float momentum) {
  int64_t c0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t c = c0; c < C; c += stride) {
    const float mean = psum[c] / count;
    float var = psumsq[c] / count - mean * mean;
    var = fmaxf(var, 0.0f);
    const float invstd = rsqrtf(var + eps);
    const float g = (gamma != nullptr) ? (float)gamma[c] : 1.0f;
    const float b = (beta != nullptr) ? (float)beta[c] : 0.0f;
    const float sc = g * invstd;
    mean_out[c] = mean;
    invstd_out[c] = invstd;
    scale[c] = sc;
    shift[c] = fmaf(-mean, sc, b);
    if (rmean != nullptr) {
      const float ub = (count > 1.0f) ? count / (count - 1.0f) : 1.0f;
      rmean[c] = (T)fmaf(momentum, mean - (float)rmean[c], (float)rmean[c]);
      rvar[c] = (T)fmaf(momentum, var * ub - (float)rvar[c], (float)rvar[c]);
    }
  }
}

// eval path: scale/shift from running stats
template <typename T>
__global__ void __launch_bounds__(BN_BLOCK)
k_bn_eval_coef(const T* __restrict__ gamma, const T* __restrict__ beta,
               const T* __restrict__ rmean, const T* __restrict__ rvar,
               float* __restrict__ scale, float* __restrict__ shift,
               int64_t C, float eps) {
  int64_t c0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t c = c0; c < C; c += stride) {
    const float invstd = rsqrtf((float)rvar[c] + eps);
    const float g = (gamma != nullptr) ? (float)gamma[c] : 1.0f;
    const float b = (beta != nullptr) ? (float)beta[c] : 0.0f;
    const float sc = g * invstd;
    scale[c] = sc;
    shift[c] = fmaf(-(float)rmean[c], sc, b);
  }
}

// ---------------------------------------------------------------------------
// F3: normalize (+ optional residual add, + optional relu), vectorized x8
// lane handles 8 consecutive channels; scale/shift gathered from L2.
// ---------------------------------------------------------------------------

// NOTE on the grid: the launcher rounds gridDim so that
// gridDim.x * BN_BLOCK is a multiple of C8 — then each thread's channel
// group is LOOP-INVARIANT and the per-channel coefficients are loaded once
// before the loop (per-iteration scalar coefficient gathers made the first
// version of these kernels VMEM-issue-bound: 44% of a ResNet-50 step).
template <bool HAS_Z>
__global__ void __launch_bounds__(BN_BLOCK)
k_bn_normalize(const __hip_bfloat16* __restrict__ x,
               const __hip_bfloat16* __restrict__ z,
               __hip_bfloat16* __restrict__ y,
               const float* __restrict__ scale,
               const float* __restrict__ shift,
               int64_t n8, int64_t C8, int relu) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t cb = (i0 % C8) * 8;  // loop-invariant: stride % C8 == 0
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = scale[cb + j];
    sh[j] = shift[cb + j];
  }
  for (int64_t i = i0; i < n8; i += stride) {
    const bnx8 xv = ((const bnx8*)x)[i];
    bnx8 zv;
    if (HAS_Z) zv = ((const bnx8*)z)[i];
    bnx8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = fmaf(b2f(xv.v[j]), sc[j], sh[j]);
      if (HAS_Z) v += b2f(zv.v[j]);
      if (relu) v = fmaxf(v, 0.0f);
      out.v[j] = f2b(v);
    }
    ((bnx8*)y)[i] = out;
  }
}

// ---------------------------------------------------------------------------
// B3: input gradients.  dx = a*dy_eff + bx*x + cc ; dz = dy_eff (if HAS_Z)
// coef arrays a,bx,cc are [C] f32 (computed by k_bn_bwd_coef).
// ---------------------------------------------------------------------------

template <bool HAS_Z>
__global__ void __launch_bounds__(BN_BLOCK)
k_bn_bwd_dx(const __hip_bfloat16* __restrict__ x,
            const __hip_bfloat16* __restrict__ dy,
            const __hip_bfloat16* __restrict__ z,
            __hip_bfloat16* __restrict__ dx,
            __hip_bfloat16* __restrict__ dz,
            const float* __restrict__ ca, const float* __restrict__ cbx,
            const float* __restrict__ cc,
            const float* __restrict__ scale, const float* __restrict__ shift,
            int64_t n8, int64_t C8, int relu) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t cb = (i0 % C8) * 8;  // loop-invariant: stride % C8 == 0
  float av[8], bxv[8], ccv[8], scv[8], shv[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    av[j] = ca[cb + j];
    bxv[j] = cbx[cb + j];
    ccv[j] = cc[cb + j];
    if (relu) { scv[j] = scale[cb + j]; shv[j] = shift[cb + j]; }
  }
  for (int64_t i = i0; i < n8; i += stride) {
    const bnx8 xv = ((const bnx8*)x)[i];
    const bnx8 dyv = ((const bnx8*)dy)[i];
    bnx8 zv;
    if (HAS_Z) zv = ((const bnx8*)z)[i];
    bnx8 odx, odz;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float xf = b2f(xv.v[j]);
      float g = b2f(dyv.v[j]);
      if (relu) {
        float yv = fmaf(xf, scv[j], shv[j]);
        if (HAS_Z) yv += b2f(zv.v[j]);
        if (yv <= 0.0f) g = 0.0f;
      }
      if (HAS_Z) odz.v[j] = f2b(g);
      odx.v[j] = f2b(fmaf(av[j], g, fmaf(bxv[j], xf, ccv[j])));
    }
    ((bnx8*)dx)[i] = odx;
    if (HAS_Z) ((bnx8*)dz)[i] = odz;
  }
}

// B2: finalize per-channel gradient coefficients + dgamma/dbeta (dtype T)
template <typename T>
__global__ void __launch_bounds__(BN_BLOCK)
k_bn_bwd_coef(const float* __restrict__ dsum,     // sum(dy_eff)
              const float* __restrict__ dxsum,    // sum(dy_eff * x)
              const float* __restrict__ mean, const float* __restrict__ invstd,
              const T* __restrict__ gamma,
              T* __restrict__ dgamma, T* __restrict__ dbeta,
              float* __restrict__ ca, float* __restrict__ cbx,
              float* __restrict__ cc, int64_t C, float count, int train) {
  int64_t c0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t c = c0; c < C; c += stride) {
    const float m = mean[c];
    const float is = invstd[c];
    const float db = dsum[c];
    // dgamma = sum(dy*xhat) = (sum(dy*x) - mean*sum(dy)) * invstd
    const float dg = (dxsum[c] - m * db) * is;
    const float g = (gamma != nullptr) ? (float)gamma[c] : 1.0f;
    if (dgamma != nullptr) dgamma[c] = (T)dg;
    if (dbeta != nullptr) dbeta[c] = (T)db;
    const float a = g * is;
    if (train) {
      cbx[c] = -a * is * dg / count;
      cc[c] = (-a * db + a * is * dg * m) / count;
    } else {
      cbx[c] = 0.0f;
      cc[c] = 0.0f;
    }
    ca[c] = a;
  }
}

// ---------------------------------------------------------------------------
// C ABI launchers
// ---------------------------------------------------------------------------

static inline dim3 bn_reduce_grid(int64_t rows, int64_t C) {
  const int ct = (int)(C / BN_CT);
  int64_t yb = (rows + 31) / 32;
  // cap total blocks at 1024 (4096 waves fills the chip): every y-block
  // costs one serialized atomicAdd chain per channel word, and at the old
  // 4096-block cap the per-word chain (~11ns/atomic) was ~45us per C=64
  // reduce
  int64_t cap = 1024 / ct;
  if (cap < 1) cap = 1;
  if (yb > cap) yb = cap;
  return dim3(ct, (unsigned)yb);
}

static inline int64_t bn_gcd(int64_t a, int64_t b) {
  while (b) { int64_t t = a % b; a = b; b = t; }
  return a;
}

// grid such that gridDim * BN_BLOCK is a multiple of C8 (loop-invariant
// channel groups in the elementwise kernels)
static inline int bn_elem_grid(int64_t n8, int64_t C8) {
  int64_t b = (n8 + BN_BLOCK - 1) / BN_BLOCK;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  const int64_t f = C8 / bn_gcd(C8, (int64_t)BN_BLOCK);
  b = (b + f - 1) / f * f;
  return (int)b;
}

extern "C" {

int ps_bn_fwd_stats(void* stream_, const void* x, float* psum, float* psumsq,
                    int64_t rows, int64_t C) {
  hipStream_t s = (hipStream_t)stream_;
  hipError_t e;
  if (psumsq == psum + C) {  // contiguous pair: one fill launch
    e = hipMemsetAsync(psum, 0, 2 * C * sizeof(float), s);
    if (e) return (int)e;
  } else {
    e = hipMemsetAsync(psum, 0, C * sizeof(float), s);
    if (e) return (int)e;
    e = hipMemsetAsync(psumsq, 0, C * sizeof(float), s);
    if (e) return (int)e;
  }
  hipLaunchKernelGGL((k_bn_reduce<false, false>), bn_reduce_grid(rows, C),
                     dim3(BN_BLOCK), 0, s, (const __hip_bfloat16*)x, nullptr,
                     nullptr, nullptr, nullptr, psum, psumsq, rows, C, 0);
  return (int)hipGetLastError();
}

int ps_bn_finalize(void* stream_, const float* psum, const float* psumsq,
                   const void* gamma, const void* beta, void* rmean,
                   void* rvar, float* mean, float* invstd, float* scale,
                   float* shift, int64_t C, double count, float eps,
                   float momentum, int t_is_bf16) {
  hipStream_t s = (hipStream_t)stream_;
  dim3 grid((unsigned)((C + BN_BLOCK - 1) / BN_BLOCK));
#define FIN(T)                                                                \
  hipLaunchKernelGGL((k_bn_finalize<T>), grid, dim3(BN_BLOCK), 0, s, psum,    \
                     psumsq, (const T*)gamma, (const T*)beta, (T*)rmean,      \
                     (T*)rvar, mean, invstd, scale, shift, C, (float)count,   \
                     eps, momentum)
  if (t_is_bf16) FIN(__hip_bfloat16); else FIN(float);
#undef FIN
  return (int)hipGetLastError();
}

int ps_bn_eval_coef(void* stream_, const void* gamma, const void* beta,
                    const void* rmean, const void* rvar, float* scale,
                    float* shift, int64_t C, float eps, int t_is_bf16) {
  hipStream_t s = (hipStream_t)stream_;
  dim3 grid((unsigned)((C + BN_BLOCK - 1) / BN_BLOCK));
#define EVC(T)                                                                \
  hipLaunchKernelGGL((k_bn_eval_coef<T>), grid, dim3(BN_BLOCK), 0, s,         \
                     (const T*)gamma, (const T*)beta, (const T*)rmean,        \
                     (const T*)rvar, scale, shift, C, eps)
  if (t_is_bf16) EVC(__hip_bfloat16); else EVC(float);
#undef EVC
  return (int)hipGetLastError();
}

int ps_bn_normalize(void* stream_, const void* x, const void* z, void* y,
                    const float* scale, const float* shift, int64_t rows,
                    int64_t C, int relu) {
  hipStream_t s = (hipStream_t)stream_;
  const int64_t n8 = rows * C / 8;
  dim3 grid(bn_elem_grid(n8, C / 8));
  if (z != nullptr)
    hipLaunchKernelGGL((k_bn_normalize<true>), grid, dim3(BN_BLOCK), 0, s,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)z,
                       (__hip_bfloat16*)y, scale, shift, n8, C / 8, relu);
  else
    hipLaunchKernelGGL((k_bn_normalize<false>), grid, dim3(BN_BLOCK), 0, s,
                       (const __hip_bfloat16*)x, nullptr, (__hip_bfloat16*)y,
                       scale, shift, n8, C / 8, relu);
  return (int)hipGetLastError();
}

int ps_bn_bwd_stats(void* stream_, const void* x, const void* dy,
                    const void* z, const float* scale, const float* shift,
                    float* dsum, float* dxsum, int64_t rows, int64_t C,
                    int relu) {
  hipStream_t s = (hipStream_t)stream_;
  hipError_t e;
  if (dxsum == dsum + C) {
    e = hipMemsetAsync(dsum, 0, 2 * C * sizeof(float), s);
    if (e) return (int)e;
  } else {
    e = hipMemsetAsync(dsum, 0, C * sizeof(float), s);
    if (e) return (int)e;
    e = hipMemsetAsync(dxsum, 0, C * sizeof(float), s);
    if (e) return (int)e;
  }
  if (z != nullptr)
    hipLaunchKernelGGL((k_bn_reduce<true, true>), bn_reduce_grid(rows, C),
                       dim3(BN_BLOCK), 0, s, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)dy, (const __hip_bfloat16*)z,
                       scale, shift, dsum, dxsum, rows, C, relu);
  else
    hipLaunchKernelGGL((k_bn_reduce<true, false>), bn_reduce_grid(rows, C),
                       dim3(BN_BLOCK), 0, s, (const __hip_bfloat16*)x,
                       (const __hip_bfloat16*)dy, nullptr, scale, shift,
                       dsum, dxsum, rows, C, relu);
  return (int)hipGetLastError();
}

int ps_bn_bwd_coef(void* stream_, const float* dsum, const float* dxsum,
                   const float* mean, const float* invstd, const void* gamma,
                   void* dgamma, void* dbeta, float* ca, float* cbx,
                   float* cc, int64_t C, double count, int train,
                   int t_is_bf16) {
  hipStream_t s = (hipStream_t)stream_;
  dim3 grid((unsigned)((C + BN_BLOCK - 1) / BN_BLOCK));
#define BWC(T)                                                                \
  hipLaunchKernelGGL((k_bn_bwd_coef<T>), grid, dim3(BN_BLOCK), 0, s, dsum,    \
                     dxsum, mean, invstd, (const T*)gamma, (T*)dgamma,        \
                     (T*)dbeta, ca, cbx, cc, C, (float)count, train)
  if (t_is_bf16) BWC(__hip_bfloat16); else BWC(float);
#undef BWC
  return (int)hipGetLastError();
}

int ps_bn_bwd_dx(void* stream_, const void* x, const void* dy, const void* z,
                 void* dx, void* dz, const float* ca, const float* cbx,
                 const float* cc, const float* scale, const float* shift,
                 int64_t rows, int64_t C, int relu) {
  hipStream_t s = (hipStream_t)stream_;
  const int64_t n8 = rows * C / 8;
  dim3 grid(bn_elem_grid(n8, C / 8));
  if (z != nullptr)
    hipLaunchKernelGGL((k_bn_bwd_dx<true>), grid, dim3(BN_BLOCK), 0, s,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                       (const __hip_bfloat16*)z, (__hip_bfloat16*)dx,
                       (__hip_bfloat16*)dz, ca, cbx, cc, scale, shift, n8,
                       C / 8, relu);
  else
    hipLaunchKernelGGL((k_bn_bwd_dx<false>), grid, dim3(BN_BLOCK), 0, s,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                       nullptr, (__hip_bfloat16*)dx, nullptr, ca, cbx, cc,
                       scale, shift, n8, C / 8, relu);
  return (int)hipGetLastError();
}

}  // extern "C"
