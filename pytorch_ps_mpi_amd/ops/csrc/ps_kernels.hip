// ps_kernels.hip — CDNA4 (gfx950 / MI355X) kernels for the async-PS training engine.
//
// These are the MI355X-native replacements for the reference's Python hot path
// (stsievert/pytorch_ps_mpi): the cross-rank gradient sum (ps.py:176), the SGD
// momentum update (ps.py:197-214), the Adam update (ps.py:218-261), and the
// lossy gradient codecs the reference delegated to its external `codings`
// package (ps.py:18). Everything operates on flat, device-resident buffers —
// no pickle, no host round trip.
//
// Design notes (see /opt/skills guides):
//  * wave = 64; blocks of 256 threads; grid-stride loops capped at ~2048 blocks
//    so a launch fills all 256 CUs across the 8 XCDs.
//  * All streaming kernels move 16 B/lane (float4 / 8×bf16) — scalar bf16
//    loads are ~2x slower on this chip.
//  * Aggregation accumulates in fp32 regardless of wire dtype.
//  * Determinism: multi-source reductions sum sources in index order, and the
//    top-k scatter is launched one message at a time (indices within one
//    message are unique), so replicated-mode ranks stay bitwise identical.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define PS_BLOCK 256
#define PS_MAX_BLOCKS 2048
#define PS_MAX_SRCS 8

static inline int ps_grid(int64_t work_items) {
  int64_t b = (work_items + PS_BLOCK - 1) / PS_BLOCK;
  if (b > PS_MAX_BLOCKS) b = PS_MAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}

// ---------------------------------------------------------------------------
// dtype helpers
// ---------------------------------------------------------------------------

__device__ __forceinline__ float ld_as_float(const float* p, int64_t i) { return p[i]; }
__device__ __forceinline__ float ld_as_float(const __hip_bfloat16* p, int64_t i) {
  return __bfloat162float(p[i]);
}

__device__ __forceinline__ void st_from_float(float* p, int64_t i, float v) { p[i] = v; }
__device__ __forceinline__ void st_from_float(__hip_bfloat16* p, int64_t i, float v) {
  p[i] = __float2bfloat16(v);
}

// 8-wide vector views (16 B) for bf16, 4-wide (16 B) for f32.
struct bf16x8 { __hip_bfloat16 v[8]; };
struct f32x4  { float v[4]; };

// ---------------------------------------------------------------------------
// fused SGD (momentum + weight decay + dampening + nesterov) over flat buffers
//   d   = grad_scale * g            (g summed over ranks upstream)
//   d  += wd * p
//   buf = momentum * buf + (1 - dampening) * d        (buf = d on first step)
//   d   = nesterov ? d + momentum * buf : buf         (if momentum != 0)
//   p  -= lr * d
//   param_out (bf16 model copy) = p                   (optional)
// Matches torch.optim.SGD / reference ps.py:197-214 exactly in fp32.
// ---------------------------------------------------------------------------

template <bool HAS_MOM, bool MOM_INIT, bool NESTEROV, typename OUT_T>
__global__ void __launch_bounds__(PS_BLOCK)
k_fused_sgd(float* __restrict__ p, float* __restrict__ buf,
            const float* __restrict__ g, OUT_T* __restrict__ p_out,
            int64_t n, float lr, float momentum, float dampening,
            float wd, float gscale) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < n; i += stride) {
    float d = g[i] * gscale;
    float pi = p[i];
    d = fmaf(wd, pi, d);
    if (HAS_MOM) {
      float b;
      if (MOM_INIT) {
        b = d;
      } else {
        b = fmaf(momentum, buf[i], (1.0f - dampening) * d);
      }
      buf[i] = b;
      d = NESTEROV ? fmaf(momentum, b, d) : b;
    }
    pi = fmaf(-lr, d, pi);
    p[i] = pi;
    if (p_out != nullptr) st_from_float(p_out, i, pi);
  }
}

// ---------------------------------------------------------------------------
// fused Adam incl. amsgrad — matches MODERN torch.optim.Adam exactly:
//   denom = sqrt(v)/sqrt(bc2) + eps.
// The reference (ps.py:218-261, torch-0.3-era) used sqrt(v) + eps with the
// bias correction folded into step_size, which effectively scales eps by
// sqrt(bc2): a benign O(eps) deviation from ps.py, not bit-identical to it.
// ---------------------------------------------------------------------------

template <bool AMSGRAD, typename OUT_T>
__global__ void __launch_bounds__(PS_BLOCK)
k_fused_adam(float* __restrict__ p, float* __restrict__ m1, float* __restrict__ m2,
             float* __restrict__ vmax, const float* __restrict__ g,
             OUT_T* __restrict__ p_out, int64_t n,
             float lr, float beta1, float beta2, float eps, float wd,
             float bc1, float bc2_sqrt, float gscale) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float step_size = lr / bc1;
  for (int64_t i = i0; i < n; i += stride) {
    float d = g[i] * gscale;
    float pi = p[i];
    d = fmaf(wd, pi, d);
    float a = fmaf(beta1, m1[i], (1.0f - beta1) * d);
    float v = fmaf(beta2, m2[i], (1.0f - beta2) * d * d);
    m1[i] = a;
    m2[i] = v;
    if (AMSGRAD) {
      v = fmaxf(v, vmax[i]);
      vmax[i] = v;
    }
    float denom = fmaf(sqrtf(v), 1.0f / bc2_sqrt, eps);
    pi = fmaf(-step_size, a / denom, pi);
    p[i] = pi;
    if (p_out != nullptr) st_from_float(p_out, i, pi);
  }
}

// ---------------------------------------------------------------------------
// multi-source reduce: dst(f32) = beta*dst + scale * sum_r src_r
// Sources summed in fixed index order → deterministic across ranks.
// Replaces the reference's python `sum(grads)` (ps.py:176).
// Vectorized 16 B/lane when n % 8 == 0 (bf16) / n % 4 == 0 (f32); buffers
// from the bucket allocator are always 256-element aligned so this is the
// common case.
// ---------------------------------------------------------------------------

struct PtrPack { const void* p[PS_MAX_SRCS]; };

template <typename T, int NSRC>
__global__ void __launch_bounds__(PS_BLOCK)
k_reduce_accum(float* __restrict__ dst, PtrPack pack, int64_t n,
               float scale, float beta) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < n; i += stride) {
    float acc = 0.0f;
#pragma unroll
    for (int r = 0; r < NSRC; ++r) {
      acc += ld_as_float((const T*)pack.p[r], i);
    }
    dst[i] = beta * dst[i] + scale * acc;
  }
}

// vectorized bf16 variant: each lane handles 8 bf16 (16 B) per iteration
template <int NSRC>
__global__ void __launch_bounds__(PS_BLOCK)
k_reduce_accum_bf16v(float* __restrict__ dst, PtrPack pack, int64_t n8,
                     float scale, float beta) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < n8; i += stride) {
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.0f;
#pragma unroll
    for (int r = 0; r < NSRC; ++r) {
      const bf16x8 v = ((const bf16x8*)pack.p[r])[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += __bfloat162float(v.v[j]);
    }
    float4 o0, o1;
    o0.x = beta * dst[i * 8 + 0] + scale * acc[0];
    o0.y = beta * dst[i * 8 + 1] + scale * acc[1];
    o0.z = beta * dst[i * 8 + 2] + scale * acc[2];
    o0.w = beta * dst[i * 8 + 3] + scale * acc[3];
    o1.x = beta * dst[i * 8 + 4] + scale * acc[4];
    o1.y = beta * dst[i * 8 + 5] + scale * acc[5];
    o1.z = beta * dst[i * 8 + 6] + scale * acc[6];
    o1.w = beta * dst[i * 8 + 7] + scale * acc[7];
    ((float4*)dst)[i * 2 + 0] = o0;
    ((float4*)dst)[i * 2 + 1] = o1;
  }
}

// ---------------------------------------------------------------------------
// dtype casts over flat buffers (f32 master <-> bf16 model copy)
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(PS_BLOCK)
k_f32_to_bf16(const float* __restrict__ src, __hip_bfloat16* __restrict__ dst, int64_t n) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < n; i += stride) dst[i] = __float2bfloat16(src[i]);
}

__global__ void __launch_bounds__(PS_BLOCK)
k_bf16_to_f32(const __hip_bfloat16* __restrict__ src, float* __restrict__ dst, int64_t n) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < n; i += stride) dst[i] = __bfloat162float(src[i]);
}

// ---------------------------------------------------------------------------
// int8 gradient quantization codec (BASELINE config 4)
// Wire layout for n elements, chunk = 256:
//   [ float scale[n_chunks] | pad to 16B | int8 q[n] | pad ]
// (the 16B pad before q keeps the int8 payload 8/16B-aligned for vector I/O)
//
// Encode: one WAVE owns a PAIR of chunks (512 elems) per iteration — lane l
// loads 8 consecutive elements as one 16B (bf16) / 2×16B (f32) vector load,
// the 32-lane half-wave shfl-reduces its chunk's absmax (no LDS, no
// __syncthreads), and stores 8 packed int8 as one 8B store.  The round-1
// scalar version (1 bf16 load/lane, one 256-chunk per 256-thread block) ran
// at 1.4 TB/s vs the ~6.3 TB/s streaming roofline; this shape is the same
// 16 B/lane discipline as every other streaming kernel in this file.
// Decode: dst += gscale * sum_r scale_r[c] * q_r[i], 8 elems/lane/iter
// (8B q loads + 2×16B dst read/write), deterministic source order.
// ---------------------------------------------------------------------------

#define QCHUNK 256

struct u8x8 { uint32_t lo, hi; };

template <typename T>
__global__ void __launch_bounds__(PS_BLOCK)
k_quant8_encode_v(const T* __restrict__ src, float* __restrict__ scales,
                  int8_t* __restrict__ q, int64_t n) {
  const int lane = threadIdx.x & 63;
  const int64_t wave0 = (int64_t)blockIdx.x * (PS_BLOCK / 64)
                      + (threadIdx.x >> 6);
  const int64_t nwaves = (int64_t)gridDim.x * (PS_BLOCK / 64);
  const int64_t nchunks = (n + QCHUNK - 1) / QCHUNK;
  const int64_t npairs = (nchunks + 1) / 2;
  for (int64_t p = wave0; p < npairs; p += nwaves) {
    const int64_t base = p * 512 + (int64_t)lane * 8;
    float x[8];
    if (base + 8 <= n) {
      if constexpr (sizeof(T) == 2) {
        const bf16x8 v = *(const bf16x8*)(src + base);
#pragma unroll
        for (int j = 0; j < 8; ++j) x[j] = __bfloat162float(v.v[j]);
      } else {
        const f32x4 a = ((const f32x4*)(src + base))[0];
        const f32x4 b = ((const f32x4*)(src + base))[1];
#pragma unroll
        for (int j = 0; j < 4; ++j) { x[j] = a.v[j]; x[4 + j] = b.v[j]; }
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        x[j] = (base + j < n) ? ld_as_float(src, base + j) : 0.0f;
    }
    float m = fabsf(x[0]);
#pragma unroll
    for (int j = 1; j < 8; ++j) m = fmaxf(m, fabsf(x[j]));
    // chunk absmax across the 32-lane half-wave (one chunk per half)
#pragma unroll
    for (int off = 16; off > 0; off >>= 1)
      m = fmaxf(m, __shfl_xor(m, off, 32));
    const float scale = (m > 0.0f) ? (m / 127.0f) : 1.0f;
    const float inv = 1.0f / scale;
    const int64_t c = p * 2 + (lane >> 5);
    if ((lane & 31) == 0 && c < nchunks) scales[c] = scale;
    u8x8 packed;
    uint32_t w[2] = {0u, 0u};
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float r = fminf(fmaxf(x[j] * inv, -127.0f), 127.0f);
      const int32_t qi = (int32_t)lrintf(r);
      w[j >> 2] |= ((uint32_t)(uint8_t)(int8_t)qi) << ((j & 3) * 8);
    }
    packed.lo = w[0];
    packed.hi = w[1];
    if (base + 8 <= n) {
      *(u8x8*)(q + base) = packed;  // 8B store (q base is 16B-aligned)
    } else {
      for (int j = 0; j < 8 && base + j < n; ++j)
        q[base + j] = (int8_t)(uint8_t)((j < 4 ? w[0] : w[1])
                                        >> ((j & 3) * 8));
    }
  }
}

template <int NSRC>
__global__ void __launch_bounds__(PS_BLOCK)
k_quant8_reduce_v(float* __restrict__ dst, PtrPack scale_pack, PtrPack q_pack,
                  int64_t n, float gscale, float beta) {
  const int64_t g0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t ngroups = (n + 7) / 8;
  for (int64_t g = g0; g < ngroups; g += stride) {
    const int64_t base = g * 8;
    const int64_t c = base / QCHUNK;  // 8-aligned base => one chunk per group
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.0f;
    if (base + 8 <= n) {
#pragma unroll
      for (int r = 0; r < NSRC; ++r) {
        const float s = ((const float*)scale_pack.p[r])[c];
        const u8x8 v = *(const u8x8*)((const int8_t*)q_pack.p[r] + base);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const uint32_t wv = j < 4 ? v.lo : v.hi;
          const int8_t qi = (int8_t)(uint8_t)(wv >> ((j & 3) * 8));
          acc[j] = fmaf(s, (float)qi, acc[j]);
        }
      }
      f32x4 o0, o1;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        o0.v[j] = beta * dst[base + j] + gscale * acc[j];
        o1.v[j] = beta * dst[base + 4 + j] + gscale * acc[4 + j];
      }
      ((f32x4*)(dst + base))[0] = o0;
      ((f32x4*)(dst + base))[1] = o1;
    } else {
      for (int64_t i = base; i < n; ++i) {
        float a = 0.0f;
#pragma unroll
        for (int r = 0; r < NSRC; ++r) {
          const float s = ((const float*)scale_pack.p[r])[i / QCHUNK];
          a = fmaf(s, (float)((const int8_t*)q_pack.p[r])[i], a);
        }
        dst[i] = beta * dst[i] + gscale * a;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// top-k magnitude sparsification codec (BASELINE config 3)
// Radix-style selection on the top 11 bits of |x| as float:
//   key(x) = (bits(|x|) >> 21) & 0x7FF   — monotone in |x| for finite floats
// Phase A: 2048-bin histogram (block-local LDS, one atomicAdd per bin per block)
// Phase B: single-block suffix scan → threshold key, exact take counts
// Phase C: compact (idx,val) pairs with two global counters
// Phase D: scatter-accumulate one message at a time (indices unique per msg)
// ---------------------------------------------------------------------------

#define TK_BINS 2048

__device__ __forceinline__ uint32_t tk_key(float x) {
  union { float f; uint32_t u; } cv;
  cv.f = fabsf(x);
  return cv.u >> 21;  // 11 bits: exponent(8) + mantissa top 3
}

// Wave-aggregated histogram: gradient magnitudes concentrate in a few dozen
// key bins, so per-element LDS atomics serialize on hot bins.  Instead each
// wave resolves its 64 keys into one LDS atomic PER DISTINCT KEY via a
// ballot loop (iterations = distinct keys in the wave).
template <typename T>
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_hist(const T* __restrict__ src, uint32_t* __restrict__ hist, int64_t n) {
  __shared__ uint32_t lh[TK_BINS];
  for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x) lh[b] = 0;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = i0; i < n; i += stride) {
    const uint32_t key = tk_key(ld_as_float(src, i));
    unsigned long long todo = __ballot(1);  // active lanes
    while (todo) {
      const int leader = __ffsll(todo) - 1;
      const uint32_t lkey = __shfl(key, leader, 64);
      const unsigned long long same = __ballot(key == lkey);
      if (lane == leader)
        atomicAdd(&lh[lkey], (uint32_t)__popcll(same & todo));
      todo &= ~same;
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x) {
    if (lh[b]) atomicAdd(&hist[b], lh[b]);
  }
}

// plan[0]=thr_key, plan[1]=n_above(strictly), plan[2]=need_from_thr_bin
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_plan(const uint32_t* __restrict__ hist, uint32_t* __restrict__ plan, int64_t k) {
  __shared__ uint32_t sh[TK_BINS];
  for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x) sh[b] = hist[b];
  __syncthreads();
  if (threadIdx.x == 0) {
    // host clamps k <= n, and sum(hist) == n, so the break always fires
    uint64_t above = 0;
    int thr = 0;
    for (int b = TK_BINS - 1; b >= 0; --b) {
      if (above + sh[b] >= (uint64_t)k) { thr = b; break; }
      above += sh[b];
    }
    plan[0] = (uint32_t)thr;
    plan[1] = (uint32_t)above;
    plan[2] = (uint32_t)((uint64_t)k - above);
  }
}

// Threshold (VARIABLE-k) plan: select every element within a magnitude
// factor of the bucket's peak, capped at kmax — the device-side
// variable-length wire of the codec contract (the reference's adaptive
// codecs varied payload size with content; here k_used rides in a wire
// header instead of a size exchange).  `off_keys` is the key-domain
// threshold distance below the top nonzero bin: keys quantize |x| to
// 1/8-octave steps, so off_keys = round(8 * log2(1/alpha)) selects
// |x| >= alpha * max|x| (to key granularity).
// plan[0]=thr_key, plan[1]=n_above, plan[2]=need_from_thr_bin, plan[3]=k_used
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_plan_thresh(const uint32_t* __restrict__ hist,
                   uint32_t* __restrict__ plan, int off_keys, int64_t kmax) {
  __shared__ uint32_t sh[TK_BINS];
  for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x) sh[b] = hist[b];
  __syncthreads();
  if (threadIdx.x == 0) {
    int bmax = 0;
    for (int b = TK_BINS - 1; b >= 0; --b) {
      if (sh[b]) { bmax = b; break; }
    }
    const int thr0 = max(0, bmax - off_keys);
    uint64_t cand = 0;
    for (int b = TK_BINS - 1; b >= thr0; --b) cand += sh[b];
    uint64_t k = cand < (uint64_t)kmax ? cand : (uint64_t)kmax;
    if (k < 1) k = 1;  // host guarantees n >= 1; keep at least the peak
    uint64_t above = 0;
    int thr = 0;
    for (int b = TK_BINS - 1; b >= 0; --b) {
      if (above + sh[b] >= k) { thr = b; break; }
      above += sh[b];
    }
    plan[0] = (uint32_t)thr;
    plan[1] = (uint32_t)above;
    plan[2] = (uint32_t)(k - above);
    plan[3] = (uint32_t)k;
  }
}

// publish k_used into the wire header
__global__ void k_topk_hdr(const uint32_t* __restrict__ plan,
                           int32_t* __restrict__ hdr) {
  if (threadIdx.x == 0) hdr[0] = (int32_t)plan[3];
}

// scatter with DEVICE-side k (wire header): launched for kmax, guarded.
template <typename T>
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_scatter_var(float* __restrict__ dst, const int32_t* __restrict__ hdr,
                   const int32_t* __restrict__ idx, const T* __restrict__ val,
                   int64_t kmax, float gscale) {
  const int64_t k = hdr[0];
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = i0; j < k; j += stride) {
    const int32_t d = idx[j];
    dst[d] += gscale * ld_as_float(val, j);
  }
}

// Compaction is 3-phase with NO contended global atomics (a single global
// counter word takes ~88 atomics/us on this chip; at 1% density nearly every
// wave carries a candidate, so even wave-aggregated slot allocation cost
// ~8 ms on 100M elements):
//   count:  block b owns the contiguous range [b*chunk,(b+1)*chunk) and
//           counts its above/eq candidates (ballot popcounts, LDS reduce)
//   scan:   one block turns per-block counts into exclusive offsets
//   emit:   block b re-reads its range and writes candidates at
//           offset[b] + block-local LDS-allocated slots (LDS atomics only)

#define TK_NB 2048  // compaction blocks (= per-class offset array length)

template <typename T>
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_count(const T* __restrict__ src, const uint32_t* __restrict__ plan,
             uint32_t* __restrict__ cnt_above, uint32_t* __restrict__ cnt_eq,
             int64_t n, int64_t chunk) {
  __shared__ uint32_t red[2][4];
  const uint32_t thr = plan[0];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t lo = (int64_t)blockIdx.x * chunk;
  const int64_t hi = min(n, lo + chunk);
  uint32_t ca = 0, ce = 0;
  for (int64_t i = lo + threadIdx.x; i < hi; i += PS_BLOCK) {
    const uint32_t key = tk_key(ld_as_float(src, i));
    const unsigned long long mab = __ballot(key > thr);
    const unsigned long long meq = __ballot(key == thr);
    if (lane == 0) {
      ca += (uint32_t)__popcll(mab);
      ce += (uint32_t)__popcll(meq);
    }
  }
  if (lane == 0) { red[0][wave] = ca; red[1][wave] = ce; }
  __syncthreads();
  if (threadIdx.x == 0) {
    cnt_above[blockIdx.x] = red[0][0] + red[0][1] + red[0][2] + red[0][3];
    cnt_eq[blockIdx.x] = red[1][0] + red[1][1] + red[1][2] + red[1][3];
  }
}

// in-place exclusive scan of both count arrays (nb <= TK_NB, single block).
// Parallel: each thread serial-scans an 8-entry strip in registers; the 256
// strip totals are scanned in LDS; strip offsets added back.  (A serial
// thread-0 loop over global memory was ~0.4 ms of dependent latency.)
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_scan(uint32_t* __restrict__ cnt_above, uint32_t* __restrict__ cnt_eq,
            int nb) {
  __shared__ uint32_t part[2][PS_BLOCK + 1];
  const int t = threadIdx.x;
  const int strip = (TK_NB / PS_BLOCK);  // 8
  uint32_t va[8], ve[8];
  uint32_t sa = 0, se = 0;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int b = t * strip + j;
    const uint32_t a = (b < nb) ? cnt_above[b] : 0;
    const uint32_t e = (b < nb) ? cnt_eq[b] : 0;
    va[j] = sa; ve[j] = se;  // local exclusive prefix
    sa += a; se += e;
  }
  part[0][t + 1] = sa;
  part[1][t + 1] = se;
  if (t == 0) { part[0][0] = 0; part[1][0] = 0; }
  __syncthreads();
  if (t == 0) {  // 256 serial LDS adds (~10 us worst case)
    for (int i = 1; i <= PS_BLOCK; ++i) {
      part[0][i] += part[0][i - 1];
      part[1][i] += part[1][i - 1];
    }
  }
  __syncthreads();
  const uint32_t oa = part[0][t], oe = part[1][t];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int b = t * strip + j;
    if (b < nb) {
      cnt_above[b] = oa + va[j];
      cnt_eq[b] = oe + ve[j];
    }
  }
}

template <typename T>
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_emit(const T* __restrict__ src, const uint32_t* __restrict__ plan,
            const uint32_t* __restrict__ off_above,
            const uint32_t* __restrict__ off_eq,
            int32_t* __restrict__ out_idx, T* __restrict__ out_val,
            int64_t n, int64_t chunk) {
  __shared__ uint32_t l_above, l_eq;
  const uint32_t thr = plan[0];
  const uint32_t n_above = plan[1];
  const uint32_t need = plan[2];
  const int lane = threadIdx.x & 63;
  const unsigned long long lane_lt = (1ull << lane) - 1ull;
  if (threadIdx.x == 0) { l_above = 0; l_eq = 0; }
  __syncthreads();
  const uint32_t base_above = off_above[blockIdx.x];
  const uint32_t base_eq = off_eq[blockIdx.x];
  const int64_t lo = (int64_t)blockIdx.x * chunk;
  const int64_t hi = min(n, lo + chunk);
  for (int64_t i = lo + threadIdx.x; i < hi; i += PS_BLOCK) {
    const T raw = src[i];
    const uint32_t key = tk_key(ld_as_float(src, i));
    const bool above = key > thr;
    const bool eqb = (key == thr);
    const unsigned long long mab = __ballot(above);
    const unsigned long long meq = __ballot(eqb);
    if (mab) {
      uint32_t wbase = 0;
      const int leader = __ffsll(mab) - 1;
      if (lane == leader) wbase = atomicAdd(&l_above, (uint32_t)__popcll(mab));
      wbase = __shfl(wbase, leader, 64);
      if (above) {
        const uint32_t slot = base_above + wbase
                            + (uint32_t)__popcll(mab & lane_lt);
        if (slot < n_above) {
          out_idx[slot] = (int32_t)i;
          out_val[slot] = raw;
        }
      }
    }
    if (meq && need > 0) {
      uint32_t wbase = 0;
      const int leader = __ffsll(meq) - 1;
      if (lane == leader) wbase = atomicAdd(&l_eq, (uint32_t)__popcll(meq));
      wbase = __shfl(wbase, leader, 64);
      if (eqb) {
        const uint32_t eq = base_eq + wbase
                          + (uint32_t)__popcll(meq & lane_lt);
        if (eq < need) {
          const uint32_t slot = n_above + eq;
          out_idx[slot] = (int32_t)i;
          out_val[slot] = raw;
        }
      }
    }
  }
}

// dst[idx[j]] = beta_select? ... : dst[idx[j]] + gscale*val[j]
// One message per launch → indices unique → no atomics needed.
template <typename T>
__global__ void __launch_bounds__(PS_BLOCK)
k_topk_scatter(float* __restrict__ dst, const int32_t* __restrict__ idx,
               const T* __restrict__ val, int64_t k, float gscale) {
  int64_t i0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = i0; j < k; j += stride) {
    const int32_t d = idx[j];
    dst[d] += gscale * ld_as_float(val, j);
  }
}

// ---------------------------------------------------------------------------
// C ABI launchers (bindings.cpp calls these; keeps torch headers out of the
// .hip translation unit so hipcc compiles it standalone)
// ---------------------------------------------------------------------------

#define LAUNCH_OK 0

extern "C" {

int ps_fused_sgd(void* stream_, float* p, float* buf, const float* g, void* p_out,
                 int p_out_is_bf16, int64_t n, float lr, float momentum,
                 float dampening, float wd, int nesterov, int mom_init, float gscale) {
  hipStream_t stream = (hipStream_t)stream_;
  dim3 grid(ps_grid(n)), block(PS_BLOCK);
  const bool has_mom = momentum != 0.0f;
#define SGD_CASE(HM, MI, NV)                                                      \
  do {                                                                            \
    if (p_out && p_out_is_bf16)                                                   \
      hipLaunchKernelGGL((k_fused_sgd<HM, MI, NV, __hip_bfloat16>), grid, block,  \
                         0, stream, p, buf, g, (__hip_bfloat16*)p_out, n, lr,     \
                         momentum, dampening, wd, gscale);                        \
    else                                                                          \
      hipLaunchKernelGGL((k_fused_sgd<HM, MI, NV, float>), grid, block, 0,        \
                         stream, p, buf, g, (float*)p_out, n, lr, momentum,       \
                         dampening, wd, gscale);                                  \
  } while (0)
  if (!has_mom) SGD_CASE(false, false, false);
  else if (mom_init && nesterov) SGD_CASE(true, true, true);
  else if (mom_init) SGD_CASE(true, true, false);
  else if (nesterov) SGD_CASE(true, false, true);
  else SGD_CASE(true, false, false);
#undef SGD_CASE
  return (int)hipGetLastError();
}

int ps_fused_adam(void* stream_, float* p, float* m1, float* m2, float* vmax,
                  const float* g, void* p_out, int p_out_is_bf16, int64_t n,
                  float lr, float beta1, float beta2, float eps, float wd,
                  float bc1, float bc2_sqrt, int amsgrad, float gscale) {
  hipStream_t stream = (hipStream_t)stream_;
  dim3 grid(ps_grid(n)), block(PS_BLOCK);
#define ADAM_CASE(AMS)                                                             \
  do {                                                                             \
    if (p_out && p_out_is_bf16)                                                    \
      hipLaunchKernelGGL((k_fused_adam<AMS, __hip_bfloat16>), grid, block, 0,      \
                         stream, p, m1, m2, vmax, g, (__hip_bfloat16*)p_out, n,    \
                         lr, beta1, beta2, eps, wd, bc1, bc2_sqrt, gscale);        \
    else                                                                           \
      hipLaunchKernelGGL((k_fused_adam<AMS, float>), grid, block, 0, stream, p,    \
                         m1, m2, vmax, g, (float*)p_out, n, lr, beta1, beta2,      \
                         eps, wd, bc1, bc2_sqrt, gscale);                          \
  } while (0)
  if (amsgrad) ADAM_CASE(true);
  else ADAM_CASE(false);
#undef ADAM_CASE
  return (int)hipGetLastError();
}

int ps_reduce_accum(void* stream_, float* dst, const void** srcs, int nsrc,
                    int src_is_bf16, int64_t n, float scale, float beta) {
  hipStream_t stream = (hipStream_t)stream_;
  if (nsrc < 1 || nsrc > PS_MAX_SRCS) return 9001;
  PtrPack pack;
  for (int i = 0; i < nsrc; ++i) pack.p[i] = srcs[i];
  const bool vec8 = src_is_bf16 && (n % 8 == 0);
  dim3 block(PS_BLOCK);
#define RED_CASE(NS)                                                                \
  case NS:                                                                          \
    if (vec8)                                                                       \
      hipLaunchKernelGGL((k_reduce_accum_bf16v<NS>), dim3(ps_grid(n / 8)), block,   \
                         0, stream, dst, pack, n / 8, scale, beta);                 \
    else if (src_is_bf16)                                                           \
      hipLaunchKernelGGL((k_reduce_accum<__hip_bfloat16, NS>), dim3(ps_grid(n)),    \
                         block, 0, stream, dst, pack, n, scale, beta);              \
    else                                                                            \
      hipLaunchKernelGGL((k_reduce_accum<float, NS>), dim3(ps_grid(n)), block, 0,   \
                         stream, dst, pack, n, scale, beta);                        \
    break;
  switch (nsrc) {
    RED_CASE(1) RED_CASE(2) RED_CASE(3) RED_CASE(4)
    RED_CASE(5) RED_CASE(6) RED_CASE(7) RED_CASE(8)
  }
#undef RED_CASE
  return (int)hipGetLastError();
}

int ps_f32_to_bf16(void* stream_, const float* src, void* dst, int64_t n) {
  hipStream_t stream = (hipStream_t)stream_;
  hipLaunchKernelGGL(k_f32_to_bf16, dim3(ps_grid(n)), dim3(PS_BLOCK), 0, stream,
                     src, (__hip_bfloat16*)dst, n);
  return (int)hipGetLastError();
}

int ps_bf16_to_f32(void* stream_, const void* src, float* dst, int64_t n) {
  hipStream_t stream = (hipStream_t)stream_;
  hipLaunchKernelGGL(k_bf16_to_f32, dim3(ps_grid(n)), dim3(PS_BLOCK), 0, stream,
                     (const __hip_bfloat16*)src, dst, n);
  return (int)hipGetLastError();
}

int ps_quant8_encode(void* stream_, const void* src, int src_is_bf16,
                     float* scales, int8_t* q, int64_t n) {
  hipStream_t stream = (hipStream_t)stream_;
  if (((uintptr_t)q & 7) != 0) return 9002;  // wire layout pads q to 16B
  const int64_t nchunks = (n + QCHUNK - 1) / QCHUNK;
  const int64_t npairs = (nchunks + 1) / 2;  // one wave per chunk pair
  dim3 grid(ps_grid(npairs * 64)), block(PS_BLOCK);
  if (src_is_bf16)
    hipLaunchKernelGGL(k_quant8_encode_v<__hip_bfloat16>, grid, block, 0,
                       stream, (const __hip_bfloat16*)src, scales, q, n);
  else
    hipLaunchKernelGGL(k_quant8_encode_v<float>, grid, block, 0, stream,
                       (const float*)src, scales, q, n);
  return (int)hipGetLastError();
}

int ps_quant8_reduce(void* stream_, float* dst, const void** scales,
                     const void** qs, int nsrc, int64_t n, float gscale, float beta) {
  hipStream_t stream = (hipStream_t)stream_;
  if (nsrc < 1 || nsrc > PS_MAX_SRCS) return 9001;
  PtrPack sp, qp;
  for (int i = 0; i < nsrc; ++i) {
    if (((uintptr_t)qs[i] & 7) != 0 || ((uintptr_t)dst & 15) != 0) return 9002;
    sp.p[i] = scales[i];
    qp.p[i] = qs[i];
  }
  dim3 grid(ps_grid((n + 7) / 8)), block(PS_BLOCK);
#define Q_CASE(NS)                                                           \
  case NS:                                                                   \
    hipLaunchKernelGGL((k_quant8_reduce_v<NS>), grid, block, 0, stream, dst, \
                       sp, qp, n, gscale, beta);                             \
    break;
  switch (nsrc) {
    Q_CASE(1) Q_CASE(2) Q_CASE(3) Q_CASE(4) Q_CASE(5) Q_CASE(6) Q_CASE(7) Q_CASE(8)
  }
#undef Q_CASE
  return (int)hipGetLastError();
}

// workspace layout (uint32): [hist TK_BINS | plan 3 | pad 1 | cntA TK_NB | cntE TK_NB]
int ps_topk_workspace_words(void) { return TK_BINS + 4 + 2 * TK_NB; }

int ps_topk_encode(void* stream_, const void* src, int src_is_bf16, int64_t n,
                   int64_t k, uint32_t* ws, int32_t* out_idx, void* out_val) {
  hipStream_t stream = (hipStream_t)stream_;
  uint32_t* hist = ws;
  uint32_t* plan = ws + TK_BINS;
  uint32_t* cnt_a = ws + TK_BINS + 4;
  uint32_t* cnt_e = cnt_a + TK_NB;
  hipError_t e = hipMemsetAsync(ws, 0, sizeof(uint32_t) * (TK_BINS + 4), stream);
  if (e != hipSuccess) return (int)e;
  // contiguous block partition for deterministic per-block offsets
  int nb = TK_NB;
  int64_t chunk = (n + nb - 1) / nb;
  chunk = (chunk + PS_BLOCK - 1) / PS_BLOCK * PS_BLOCK;
  nb = (int)((n + chunk - 1) / chunk);
  dim3 grid(ps_grid(n)), cgrid(nb), block(PS_BLOCK);
#define TK_RUN(T)                                                              \
  do {                                                                         \
    hipLaunchKernelGGL(k_topk_hist<T>, grid, block, 0, stream,                 \
                       (const T*)src, hist, n);                                \
    hipLaunchKernelGGL(k_topk_plan, dim3(1), block, 0, stream, hist, plan, k); \
    hipLaunchKernelGGL(k_topk_count<T>, cgrid, block, 0, stream,               \
                       (const T*)src, plan, cnt_a, cnt_e, n, chunk);           \
    hipLaunchKernelGGL(k_topk_scan, dim3(1), block, 0, stream, cnt_a, cnt_e,   \
                       nb);                                                    \
    hipLaunchKernelGGL(k_topk_emit<T>, cgrid, block, 0, stream, (const T*)src, \
                       plan, cnt_a, cnt_e, out_idx, (T*)out_val, n, chunk);    \
  } while (0)
  if (src_is_bf16) TK_RUN(__hip_bfloat16);
  else TK_RUN(float);
#undef TK_RUN
  return (int)hipGetLastError();
}

int ps_topk_encode_thresh(void* stream_, const void* src, int src_is_bf16,
                          int64_t n, int off_keys, int64_t kmax, uint32_t* ws,
                          int32_t* hdr, int32_t* out_idx, void* out_val) {
  hipStream_t stream = (hipStream_t)stream_;
  uint32_t* hist = ws;
  uint32_t* plan = ws + TK_BINS;
  uint32_t* cnt_a = ws + TK_BINS + 4;
  uint32_t* cnt_e = cnt_a + TK_NB;
  hipError_t e = hipMemsetAsync(ws, 0, sizeof(uint32_t) * (TK_BINS + 4), stream);
  if (e != hipSuccess) return (int)e;
  int nb = TK_NB;
  int64_t chunk = (n + nb - 1) / nb;
  chunk = (chunk + PS_BLOCK - 1) / PS_BLOCK * PS_BLOCK;
  nb = (int)((n + chunk - 1) / chunk);
  dim3 grid(ps_grid(n)), cgrid(nb), block(PS_BLOCK);
#define TKT_RUN(T)                                                             \
  do {                                                                         \
    hipLaunchKernelGGL(k_topk_hist<T>, grid, block, 0, stream,                 \
                       (const T*)src, hist, n);                                \
    hipLaunchKernelGGL(k_topk_plan_thresh, dim3(1), block, 0, stream, hist,    \
                       plan, off_keys, kmax);                                  \
    hipLaunchKernelGGL(k_topk_hdr, dim3(1), dim3(64), 0, stream, plan, hdr);   \
    hipLaunchKernelGGL(k_topk_count<T>, cgrid, block, 0, stream,               \
                       (const T*)src, plan, cnt_a, cnt_e, n, chunk);           \
    hipLaunchKernelGGL(k_topk_scan, dim3(1), block, 0, stream, cnt_a, cnt_e,   \
                       nb);                                                    \
    hipLaunchKernelGGL(k_topk_emit<T>, cgrid, block, 0, stream, (const T*)src, \
                       plan, cnt_a, cnt_e, out_idx, (T*)out_val, n, chunk);    \
  } while (0)
  if (src_is_bf16) TKT_RUN(__hip_bfloat16);
  else TKT_RUN(float);
#undef TKT_RUN
  return (int)hipGetLastError();
}

int ps_topk_scatter_var(void* stream_, float* dst, const int32_t* hdr,
                        const int32_t* idx, const void* val, int val_is_bf16,
                        int64_t kmax, float gscale) {
  hipStream_t stream = (hipStream_t)stream_;
  dim3 grid(ps_grid(kmax)), block(PS_BLOCK);
  if (val_is_bf16)
    hipLaunchKernelGGL(k_topk_scatter_var<__hip_bfloat16>, grid, block, 0,
                       stream, dst, hdr, idx, (const __hip_bfloat16*)val,
                       kmax, gscale);
  else
    hipLaunchKernelGGL(k_topk_scatter_var<float>, grid, block, 0, stream, dst,
                       hdr, idx, (const float*)val, kmax, gscale);
  return (int)hipGetLastError();
}

int ps_topk_scatter(void* stream_, float* dst, const int32_t* idx,
                    const void* val, int val_is_bf16, int64_t k, float gscale) {
  hipStream_t stream = (hipStream_t)stream_;
  dim3 grid(ps_grid(k)), block(PS_BLOCK);
  if (val_is_bf16)
    hipLaunchKernelGGL(k_topk_scatter<__hip_bfloat16>, grid, block, 0, stream,
                       dst, idx, (const __hip_bfloat16*)val, k, gscale);
  else
    hipLaunchKernelGGL(k_topk_scatter<float>, grid, block, 0, stream, dst, idx,
                       (const float*)val, k, gscale);
  return (int)hipGetLastError();
}

}  // extern "C"
