// mfma_attn_kernels.hip — MFMA-tiled flash attention fwd+bwd for CDNA4
// (gfx950 / MI355X).  Round-2 replacement of the one-wave-per-row
// correctness kernels (attn_kernels.hip, kept as the on-GPU oracle).
//
// Shape contract: q,k,v,o,do,dq,dk,dv are [B,H,N,D] bf16 contiguous with
// D == 64.  Same lse/delta conventions as the oracle kernels:
//   lse[r] = m + log(l) on scale-applied scores; delta[r] = dO_r . O_r.
//
// Design (see /opt/skills/guides: MFMA §3, LDS §2, T2/T14 notes):
//  * mfma_f32_16x16x32_bf16 everywhere; the contraction axis is always the
//    head dim (d=64 -> 2 mfma) or the 64-wide tile axis (2 mfma).
//  * fragment maps (verified on hardware by k_fa_selfcheck + numerics
//    tests): A: row=lane&15, k=(lane>>4)*8+j; B: col=lane&15, same k;
//    C/D: col=lane&15, row=(lane>>4)*4+reg.
//  * "B-frag = 8 consecutive elements of a row" means any operand whose
//    mfma-k axis is the head dim loads STRAIGHT from row-major HBM (Q, K,
//    dO, V-as-dP-operand); only operands contracted over the tile axis
//    (V in PV, K in dQ, Q/dO in dK/dV) need a transposed LDS image, filled
//    cooperatively once per tile (double-buffered in fwd/dq: one barrier
//    per tile).
//  * P / dS move from their C-layout registers to A-layout via a per-wave
//    private LDS round trip (32x64 bf16, no cross-wave sync).
//  * 64-key (fwd/dq) / 64-query (dkv) tiles: 32 MFMAs per wave per tile,
//    which halves the softmax-VALU + LDS-roundtrip cost per key vs a
//    32-wide tile (measured 77 -> see profiles/attn_bench).
//  * 4 waves x 32 rows = 128-row (fwd/dq) or 128-key (dkv) blocks; grid =
//    (ceil(N/128), B*H); every wave owns its 32-row strip end to end.
//  * masking: MASK = -1e30f scores (finite: exp stays exact-0 for real
//    rows, no NaNs for fully-masked rows); causal tiles above the diagonal
//    are skipped per wave; rows/keys >= N never store.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(4))) float f32x4v;

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

#define FA_D 64
#define FA_BM 256   // rows (fwd/dq) or keys (dkv) per 8-wave (512-thr) block
                    // (256 not 128: each kv/q tile is re-read once per block,
                    // so doubling the block halves the cross-block HBM/L3
                    // re-read traffic the bwd kernels are bound by in-model)
#define FA_BN 64    // fwd kv tile width
#define FA_BNB 32   // bwd tile width: the bwd kernels carry 2-3x the live
                    // state of fwd (two C-tiles + two accumulators), and at
                    // BN=64 their VGPR+AGPR total passes 256 -> 1 wave/SIMD
                    // (measured regression); 32 keeps them at 2 waves/SIMD
#define FA_MASK -1e30f

__device__ __forceinline__ bf16x8v fa_zero8() {
  bf16x8v z;
#pragma unroll
  for (int j = 0; j < 8; ++j) z[j] = (__bf16)0.0f;
  return z;
}

// XOR-swizzled LDS indexing (guide T2): every image is [row][col] bf16 with
// an 8-element (16B) granule; granule index is XORed with the row so lanes
// reading one column-range of many rows spread over banks.  Writes use
// fa_swe (element), 16B vector reads use fa_swg (granule base).  mask =
// cols/8 - 1.  Without this the scalar transpose-fill and P-roundtrip
// writes are 4-8-way bank conflicted (measured 10-13% of wave cycles).
__device__ __forceinline__ int fa_swe(int row, int col, int stride,
                                      int mask) {
  return row * stride + ((((col >> 3) ^ row) & mask) << 3) + (col & 7);
}
__device__ __forceinline__ int fa_swg(int row, int g8, int stride, int mask) {
  return row * stride + (((g8 ^ row) & mask) << 3);
}

// 8 consecutive bf16 of row `row` at element offset `off` (row-guarded).
// sN = element stride between consecutive sequence rows (64 if contiguous;
// e.g. 3*H*64 for q/k/v slices of a fused qkv projection - no .contiguous()
// copies on the model path).
__device__ __forceinline__ bf16x8v fa_ldrow8(const __hip_bfloat16* base,
                                             int64_t row, int64_t nrows,
                                             int off, int64_t sN) {
  if (row < nrows) return *(const bf16x8v*)(base + row * sN + off);
  return fa_zero8();
}

// cooperative transpose fill: src rows [r0, r0+64) x 64 cols -> swizzled
// dstT[64][64] (row stride FA_BN, mask 7).  256 threads, 16 elems each.
__device__ __forceinline__ void fa_fill_t(const __hip_bfloat16* src,
                                          int64_t r0, int64_t nrows,
                                          __hip_bfloat16* dstT, int64_t sN) {
  const int k = threadIdx.x >> 3;        // 0..63 source row (512 threads)
  const int d0 = (threadIdx.x & 7) * 8;  // 0..56
  const bf16x8v v = fa_ldrow8(src, r0 + k, nrows, d0, sN);
#pragma unroll
  for (int j = 0; j < 8; ++j)
    dstT[fa_swe(d0 + j, k, FA_BN, 7)] = (__hip_bfloat16)(float)v[j];
}

// 32-row variant for the bwd kernels: rows [r0, r0+32) -> swizzled
// dstT[64][32] (row stride FA_BNB, mask 3)
__device__ __forceinline__ void fa_fill_t32(const __hip_bfloat16* src,
                                            int64_t r0, int64_t nrows,
                                            __hip_bfloat16* dstT,
                                            int64_t sN) {
  if (threadIdx.x >= 256) return;        // 32x64 tile: first 256 threads
  const int k = threadIdx.x >> 3;        // 0..31 source row in tile
  const int d0 = (threadIdx.x & 7) * 8;  // 0..56
  const bf16x8v v = fa_ldrow8(src, r0 + k, nrows, d0, sN);
#pragma unroll
  for (int j = 0; j < 8; ++j)
    dstT[fa_swe(d0 + j, k, FA_BNB, 3)] = (__hip_bfloat16)(float)v[j];
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(512)
k_fa_fwd(const __hip_bfloat16* __restrict__ q,
         const __hip_bfloat16* __restrict__ k,
         const __hip_bfloat16* __restrict__ v,
         __hip_bfloat16* __restrict__ o, float* __restrict__ lse,
         int64_t N, float scale, int causal,
         int64_t H, int64_t sB, int64_t sH, int64_t sN) {
  // lds: vt double buffer [2][64][64] + per-wave P [8][32][64]
  __shared__ __align__(16) __hip_bfloat16 lds[2 * FA_D * FA_BN
                                              + 8 * 32 * FA_BN];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int c = lane & 15;
  const int g = lane >> 4;
  const int64_t bh = blockIdx.y;
  const int64_t m0 = (int64_t)blockIdx.x * FA_BM;
  const int64_t row0 = m0 + wv * 32;
  const __hip_bfloat16* qb = q + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* kb = k + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* vb = v + (bh / H) * sB + (bh % H) * sH;
  __hip_bfloat16* pbuf = lds + 2 * FA_D * FA_BN + wv * 32 * FA_BN;

  bf16x8v aQ[2][2];
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      aQ[rf][kk] = fa_ldrow8(qb, row0 + rf * 16 + c, N, kk * 32 + g * 8, sN);

  float m[2][4], l[2][4];
  f32x4v acc[2][4];
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m[rf][r] = FA_MASK;
      l[rf][r] = 0.0f;
    }
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int df = 0; df < 4; ++df) acc[rf][df] = (f32x4v)(0.0f);

  const int64_t kend = causal ? min(N, m0 + FA_BM) : N;
  const int64_t ntiles = (kend + FA_BN - 1) / FA_BN;
  if (ntiles <= 0) return;
  fa_fill_t(vb, 0, N, lds, sN);  // V tile 0 -> buffer 0
  for (int64_t t = 0; t < ntiles; ++t) {
    const int64_t k0 = t * FA_BN;
    __syncthreads();  // vt[t&1] filled; prior tile's reads complete
    if (t + 1 < ntiles)
      fa_fill_t(vb, (t + 1) * FA_BN, N,
                lds + ((t + 1) & 1) * FA_D * FA_BN, sN);
    const __hip_bfloat16* vt = lds + (t & 1) * FA_D * FA_BN;
    if (row0 >= N) continue;                 // whole wave past the rows
    if (causal && k0 > row0 + 31) continue;  // above this wave's diagonal

    // S = Q @ K^T  (B-frag: 8 consecutive d of key row -> direct load).
    // Load ALL 8 K fragments first so their L2 latencies overlap (a load
    // adjacent to its consuming MFMA costs one serialized round trip each).
    bf16x8v bK[4][2];
#pragma unroll
    for (int jf = 0; jf < 4; ++jf)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bK[jf][kk] = fa_ldrow8(kb, k0 + jf * 16 + c, N, kk * 32 + g * 8, sN);
    f32x4v S[2][4];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int jf = 0; jf < 4; ++jf) S[rf][jf] = (f32x4v)(0.0f);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int jf = 0; jf < 4; ++jf)
#pragma unroll
        for (int rf = 0; rf < 2; ++rf)
          S[rf][jf] = MFMA16(aQ[rf][kk], bK[jf][kk], S[rf][jf]);

    // scale + mask + online softmax
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t qrow = row0 + rf * 16 + g * 4 + r;
        const int64_t cmax = causal ? min(N, qrow + 1) : N;
#pragma unroll
        for (int jf = 0; jf < 4; ++jf) {
          float s = S[rf][jf][r] * scale;
          if (k0 + jf * 16 + c >= cmax) s = FA_MASK;
          S[rf][jf][r] = s;
        }
        float t2 = fmaxf(fmaxf(S[rf][0][r], S[rf][1][r]),
                         fmaxf(S[rf][2][r], S[rf][3][r]));
#pragma unroll
        for (int d = 1; d < 16; d <<= 1) t2 = fmaxf(t2, __shfl_xor(t2, d, 16));
        const float mn = fmaxf(m[rf][r], t2);
        const float al = __expf(m[rf][r] - mn);
        m[rf][r] = mn;
        l[rf][r] *= al;
#pragma unroll
        for (int df = 0; df < 4; ++df) acc[rf][df][r] *= al;
        float rs = 0.0f;
#pragma unroll
        for (int jf = 0; jf < 4; ++jf) {
          const float p = __expf(S[rf][jf][r] - mn);
          S[rf][jf][r] = p;
          rs += p;
        }
#pragma unroll
        for (int d = 1; d < 16; d <<= 1) rs += __shfl_xor(rs, d, 16);
        l[rf][r] += rs;
      }

    // P (C-layout) -> per-wave LDS -> A-frags
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int jf = 0; jf < 4; ++jf)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          pbuf[fa_swe(rf * 16 + g * 4 + r, jf * 16 + c, FA_BN, 7)] =
              (__hip_bfloat16)S[rf][jf][r];
    bf16x8v aP[2][2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        aP[rf][kk] = *(const bf16x8v*)(pbuf
            + fa_swg(rf * 16 + c, kk * 4 + g, FA_BN, 7));

    // O += P @ V   (B-frag from transposed V image)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int df = 0; df < 4; ++df) {
        const bf16x8v bV = *(const bf16x8v*)(vt
            + fa_swg(df * 16 + c, kk * 4 + g, FA_BN, 7));
#pragma unroll
        for (int rf = 0; rf < 2; ++rf)
          acc[rf][df] = MFMA16(aP[rf][kk], bV, acc[rf][df]);
      }
  }

  __hip_bfloat16* ob = o + bh * N * FA_D;
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t row = row0 + rf * 16 + g * 4 + r;
      if (row >= N) continue;
      const float inv = 1.0f / l[rf][r];
#pragma unroll
      for (int df = 0; df < 4; ++df)
        ob[row * FA_D + df * 16 + c] =
            (__hip_bfloat16)(acc[rf][df][r] * inv);
      if (c == 0) lse[bh * N + row] = m[rf][r] + __logf(l[rf][r]);
    }
}

// ---------------------------------------------------------------------------
// backward: delta then dq then dkv
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
k_fa_delta(const __hip_bfloat16* __restrict__ dout,
           const __hip_bfloat16* __restrict__ o, float* __restrict__ delta,
           int64_t rows) {
  // 8 rows per wave: lane owns 8 consecutive elements (one 16B load per
  // tensor) of row lane>>3; dot reduced across the 8-lane row group.
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int rsub = lane >> 3;        // 0..7 row within the wave's 8
  const int off = (lane & 7) * 8;
  const int64_t rstride = (int64_t)gridDim.x * 32;
  for (int64_t r0 = ((int64_t)blockIdx.x * 4 + wave) * 8; r0 < rows;
       r0 += rstride) {
    const int64_t r = r0 + rsub;
    float d = 0.0f;
    if (r < rows) {
      const bf16x8v a = *(const bf16x8v*)(dout + r * FA_D + off);
      const bf16x8v b = *(const bf16x8v*)(o + r * FA_D + off);
#pragma unroll
      for (int j = 0; j < 8; ++j) d += (float)a[j] * (float)b[j];
    }
#pragma unroll
    for (int s = 1; s < 8; s <<= 1) d += __shfl_xor(d, s, 8);
    if ((lane & 7) == 0 && r < rows) delta[r] = d;
  }
}

__global__ void __launch_bounds__(512)
k_fa_bwd_dq(const __hip_bfloat16* __restrict__ q,
            const __hip_bfloat16* __restrict__ k,
            const __hip_bfloat16* __restrict__ v,
            const __hip_bfloat16* __restrict__ dout,
            const float* __restrict__ lse, const float* __restrict__ delta,
            __hip_bfloat16* __restrict__ dq, int64_t N, float scale,
            int causal, int64_t H, int64_t sB, int64_t sH, int64_t sN,
            int64_t oB, int64_t oH, int64_t oN) {
  // lds: Kt double buffer [2][64][32] + per-wave dS [8][32][32]
  __shared__ __align__(16) __hip_bfloat16 lds[2 * FA_D * FA_BNB
                                              + 8 * 32 * FA_BNB];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int c = lane & 15;
  const int g = lane >> 4;
  const int64_t bh = blockIdx.y;
  const int64_t m0 = (int64_t)blockIdx.x * FA_BM;
  const int64_t row0 = m0 + wv * 32;
  const __hip_bfloat16* qb = q + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* kb = k + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* vb = v + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* dob = dout + bh * N * FA_D;  // contiguous
  __hip_bfloat16* sbuf = lds + 2 * FA_D * FA_BNB + wv * 32 * FA_BNB;

  bf16x8v aQ[2][2], aDO[2][2];
  float lse_r[2][4], dl_r[2][4];
#pragma unroll
  for (int rf = 0; rf < 2; ++rf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      aQ[rf][kk] = fa_ldrow8(qb, row0 + rf * 16 + c, N, kk * 32 + g * 8, sN);
      aDO[rf][kk] = fa_ldrow8(dob, row0 + rf * 16 + c, N, kk * 32 + g * 8, FA_D);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t row = row0 + rf * 16 + g * 4 + r;
      lse_r[rf][r] = (row < N) ? lse[bh * N + row] : 0.0f;
      dl_r[rf][r] = (row < N) ? delta[bh * N + row] : 0.0f;
    }
  }
  f32x4v acc[2][4];
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int df = 0; df < 4; ++df) acc[rf][df] = (f32x4v)(0.0f);

  const int64_t kend = causal ? min(N, m0 + FA_BM) : N;
  const int64_t ntiles = (kend + FA_BNB - 1) / FA_BNB;
  if (ntiles <= 0) return;
  fa_fill_t32(kb, 0, N, lds, sN);
  for (int64_t t = 0; t < ntiles; ++t) {
    const int64_t k0 = t * FA_BNB;
    __syncthreads();
    if (t + 1 < ntiles)
      fa_fill_t32(kb, (t + 1) * FA_BNB,
                  N, lds + ((t + 1) & 1) * FA_D * FA_BNB, sN);
    const __hip_bfloat16* kt = lds + (t & 1) * FA_D * FA_BNB;
    if (row0 >= N) continue;  // whole wave past the rows
    if (causal && k0 > row0 + 31) continue;

    // S and dP in one pass over kk (batched B-frag loads: latencies overlap)
    bf16x8v bK[2][2], bV[2][2];
#pragma unroll
    for (int jf = 0; jf < 2; ++jf)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bK[jf][kk] = fa_ldrow8(kb, k0 + jf * 16 + c, N, kk * 32 + g * 8, sN);
        bV[jf][kk] = fa_ldrow8(vb, k0 + jf * 16 + c, N, kk * 32 + g * 8, sN);
      }
    f32x4v S[2][2], dP[2][2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int jf = 0; jf < 2; ++jf) {
        S[rf][jf] = (f32x4v)(0.0f);
        dP[rf][jf] = (f32x4v)(0.0f);
      }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int jf = 0; jf < 2; ++jf)
#pragma unroll
        for (int rf = 0; rf < 2; ++rf) {
          S[rf][jf] = MFMA16(aQ[rf][kk], bK[jf][kk], S[rf][jf]);
          dP[rf][jf] = MFMA16(aDO[rf][kk], bV[jf][kk], dP[rf][jf]);
        }
    // dS = P * (dP - delta)
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t qrow = row0 + rf * 16 + g * 4 + r;
        const int64_t cmax = causal ? min(N, qrow + 1) : N;
#pragma unroll
        for (int jf = 0; jf < 2; ++jf) {
          float s = S[rf][jf][r] * scale;
          if (k0 + jf * 16 + c >= cmax) s = FA_MASK;
          const float p = __expf(s - lse_r[rf][r]);
          S[rf][jf][r] = p * (dP[rf][jf][r] - dl_r[rf][r]);
        }
      }
    // dS -> LDS -> A-frags; dQ += dS @ K (B-frag from transposed K image)
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int jf = 0; jf < 2; ++jf)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          sbuf[fa_swe(rf * 16 + g * 4 + r, jf * 16 + c, FA_BNB, 3)] =
              (__hip_bfloat16)S[rf][jf][r];
    bf16x8v aDS[2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
      aDS[rf] = *(const bf16x8v*)(sbuf + fa_swg(rf * 16 + c, g, FA_BNB, 3));
#pragma unroll
    for (int df = 0; df < 4; ++df) {
      const bf16x8v bKt =
          *(const bf16x8v*)(kt + fa_swg(df * 16 + c, g, FA_BNB, 3));
#pragma unroll
      for (int rf = 0; rf < 2; ++rf)
        acc[rf][df] = MFMA16(aDS[rf], bKt, acc[rf][df]);
    }
  }

  __hip_bfloat16* dqb = dq + (bh / H) * oB + (bh % H) * oH;
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t row = row0 + rf * 16 + g * 4 + r;
      if (row >= N) continue;
#pragma unroll
      for (int df = 0; df < 4; ++df)
        dqb[row * oN + df * 16 + c] =
            (__hip_bfloat16)(acc[rf][df][r] * scale);
    }
}

__global__ void __launch_bounds__(512)
k_fa_bwd_dkv(const __hip_bfloat16* __restrict__ q,
             const __hip_bfloat16* __restrict__ k,
             const __hip_bfloat16* __restrict__ v,
             const __hip_bfloat16* __restrict__ dout,
             const float* __restrict__ lse, const float* __restrict__ delta,
             __hip_bfloat16* __restrict__ dk, __hip_bfloat16* __restrict__ dv,
             int64_t N, float scale, int causal,
             int64_t H, int64_t sB, int64_t sH, int64_t sN,
             int64_t oB, int64_t oH, int64_t oN) {
  // lds: double-buffered {Qt,dOt} [2][2][64][32] + per-wave P/dS [8][32][32]
  __shared__ __align__(16) __hip_bfloat16 lds[4 * FA_D * FA_BNB
                                              + 8 * 32 * FA_BNB];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int c = lane & 15;
  const int g = lane >> 4;
  const int64_t bh = blockIdx.y;
  const int64_t m0 = (int64_t)blockIdx.x * FA_BM;  // first KEY of block
  const int64_t key0 = m0 + wv * 32;               // wave's first key
  const __hip_bfloat16* qb = q + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* kb = k + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* vb = v + (bh / H) * sB + (bh % H) * sH;
  const __hip_bfloat16* dob = dout + bh * N * FA_D;  // contiguous
  // buffer b: qt at lds + b*2*IMG, dot right after (IMG = 64*32)
  __hip_bfloat16* sbuf = lds + 4 * FA_D * FA_BNB + wv * 32 * FA_BNB;

  bf16x8v aK[2][2], aV[2][2];
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      aK[rf][kk] = fa_ldrow8(kb, key0 + rf * 16 + c, N, kk * 32 + g * 8, sN);
      aV[rf][kk] = fa_ldrow8(vb, key0 + rf * 16 + c, N, kk * 32 + g * 8, sN);
    }
  f32x4v dK[2][4], dV[2][4];
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int df = 0; df < 4; ++df) {
      dK[rf][df] = (f32x4v)(0.0f);
      dV[rf][df] = (f32x4v)(0.0f);
    }

  const int64_t q0_start = causal ? m0 : 0;
  const int64_t ntiles = (N - q0_start + FA_BNB - 1) / FA_BNB;
  if (ntiles <= 0) return;
  fa_fill_t32(qb, q0_start, N, lds, sN);
  fa_fill_t32(dob, q0_start, N, lds + FA_D * FA_BNB, FA_D);
  for (int64_t t = 0; t < ntiles; ++t) {
    const int64_t q0 = q0_start + t * FA_BNB;
    __syncthreads();  // buffer t&1 filled; prior tile's reads complete
    if (t + 1 < ntiles) {
      __hip_bfloat16* nb = lds + ((t + 1) & 1) * 2 * FA_D * FA_BNB;
      fa_fill_t32(qb, q0 + FA_BNB, N, nb, sN);
      fa_fill_t32(dob, q0 + FA_BNB, N, nb + FA_D * FA_BNB, FA_D);
    }
    const __hip_bfloat16* qt = lds + (t & 1) * 2 * FA_D * FA_BNB;
    const __hip_bfloat16* dot = qt + FA_D * FA_BNB;
    if (key0 >= N) continue;  // whole wave past the keys
    if (causal && q0 + FA_BNB - 1 < key0) continue;  // below diagonal

    float lse_c[2], dl_c[2];
#pragma unroll
    for (int jf = 0; jf < 2; ++jf) {
      const int64_t qcol = q0 + jf * 16 + c;
      lse_c[jf] = (qcol < N) ? lse[bh * N + qcol] : 0.0f;
      dl_c[jf] = (qcol < N) ? delta[bh * N + qcol] : 0.0f;
    }

    // S^T = K @ Q^T and dP^T = V @ dO^T (batched direct B-frag loads)
    bf16x8v bQ[2][2], bDO[2][2];
#pragma unroll
    for (int jf = 0; jf < 2; ++jf)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bQ[jf][kk] = fa_ldrow8(qb, q0 + jf * 16 + c, N, kk * 32 + g * 8, sN);
        bDO[jf][kk] = fa_ldrow8(dob, q0 + jf * 16 + c, N, kk * 32 + g * 8, FA_D);
      }
    f32x4v St[2][2], dPt[2][2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int jf = 0; jf < 2; ++jf) {
        St[rf][jf] = (f32x4v)(0.0f);
        dPt[rf][jf] = (f32x4v)(0.0f);
      }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int jf = 0; jf < 2; ++jf)
#pragma unroll
        for (int rf = 0; rf < 2; ++rf) {
          St[rf][jf] = MFMA16(aK[rf][kk], bQ[jf][kk], St[rf][jf]);
          dPt[rf][jf] = MFMA16(aV[rf][kk], bDO[jf][kk], dPt[rf][jf]);
        }

    // P^T = exp(s*scale - lse[q]); write P^T to LDS; dS^T kept in regs
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t keyrow = key0 + rf * 16 + g * 4 + r;
#pragma unroll
        for (int jf = 0; jf < 2; ++jf) {
          const int64_t qcol = q0 + jf * 16 + c;
          float s = St[rf][jf][r] * scale;
          if (qcol >= N || (causal && qcol < keyrow)) s = FA_MASK;
          const float p = __expf(s - lse_c[jf]);
          sbuf[fa_swe(rf * 16 + g * 4 + r, jf * 16 + c, FA_BNB, 3)] =
              (__hip_bfloat16)p;
          St[rf][jf][r] = p * (dPt[rf][jf][r] - dl_c[jf]);
        }
      }
    bf16x8v aPT[2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
      aPT[rf] = *(const bf16x8v*)(sbuf + fa_swg(rf * 16 + c, g, FA_BNB, 3));
    // dV += P^T @ dO (B from dOt image)
#pragma unroll
    for (int df = 0; df < 4; ++df) {
      const bf16x8v bDOt =
          *(const bf16x8v*)(dot + fa_swg(df * 16 + c, g, FA_BNB, 3));
#pragma unroll
      for (int rf = 0; rf < 2; ++rf)
        dV[rf][df] = MFMA16(aPT[rf], bDOt, dV[rf][df]);
    }

    // dS^T -> LDS -> A-frags; dK += dS^T @ Q (B from Qt image)
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
#pragma unroll
      for (int jf = 0; jf < 2; ++jf)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          sbuf[fa_swe(rf * 16 + g * 4 + r, jf * 16 + c, FA_BNB, 3)] =
              (__hip_bfloat16)St[rf][jf][r];
    bf16x8v aDST[2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
      aDST[rf] = *(const bf16x8v*)(sbuf + fa_swg(rf * 16 + c, g, FA_BNB, 3));
#pragma unroll
    for (int df = 0; df < 4; ++df) {
      const bf16x8v bQt =
          *(const bf16x8v*)(qt + fa_swg(df * 16 + c, g, FA_BNB, 3));
#pragma unroll
      for (int rf = 0; rf < 2; ++rf)
        dK[rf][df] = MFMA16(aDST[rf], bQt, dK[rf][df]);
    }
  }

  __hip_bfloat16* dkb = dk + (bh / H) * oB + (bh % H) * oH;
  __hip_bfloat16* dvb = dv + (bh / H) * oB + (bh % H) * oH;
#pragma unroll
  for (int rf = 0; rf < 2; ++rf)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t row = key0 + rf * 16 + g * 4 + r;
      if (row >= N) continue;
#pragma unroll
      for (int df = 0; df < 4; ++df) {
        dkb[row * oN + df * 16 + c] =
            (__hip_bfloat16)(dK[rf][df][r] * scale);
        dvb[row * oN + df * 16 + c] = (__hip_bfloat16)dV[rf][df][r];
      }
    }
}

// ---------------------------------------------------------------------------
// fragment-layout self-check: one wave computes C[16][16] = A[16][32] @
// B[32][16] via one mfma using the assumed lane maps.  Host compares against
// a reference matmul; a layout mismatch fails loudly.
// ---------------------------------------------------------------------------

__global__ void k_fa_selfcheck(const __hip_bfloat16* __restrict__ A,
                               const __hip_bfloat16* __restrict__ B,
                               float* __restrict__ C) {
  if (threadIdx.x >= 64) return;
  const int lane = threadIdx.x;
  const int c = lane & 15;
  const int g = lane >> 4;
  bf16x8v a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (__bf16)(float)A[c * 32 + g * 8 + j];   // A[row=c][k]
    b[j] = (__bf16)(float)B[(g * 8 + j) * 16 + c]; // B[k][col=c]
  }
  f32x4v acc = (f32x4v)(0.0f);
  acc = MFMA16(a, b, acc);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[(g * 4 + r) * 16 + c] = acc[r];
}

// ---------------------------------------------------------------------------
// C ABI launchers
// ---------------------------------------------------------------------------

extern "C" {

int ps_fa_fwd(void* stream_, const void* q, const void* k, const void* v,
              void* o, float* lse, int64_t BH, int64_t N, float scale,
              int causal, int64_t H, int64_t sB, int64_t sH, int64_t sN) {
  hipStream_t s = (hipStream_t)stream_;
  dim3 grid((unsigned)((N + FA_BM - 1) / FA_BM), (unsigned)BH);
  hipLaunchKernelGGL(k_fa_fwd, grid, dim3(512), 0, s,
                     (const __hip_bfloat16*)q, (const __hip_bfloat16*)k,
                     (const __hip_bfloat16*)v, (__hip_bfloat16*)o, lse, N,
                     scale, causal, H, sB, sH, sN);
  return (int)hipGetLastError();
}

int ps_fa_bwd(void* stream_, const void* q, const void* k, const void* v,
              const void* o, const void* dout, const float* lse, float* delta,
              void* dq, void* dk, void* dv, int64_t BH, int64_t N,
              float scale, int causal, int64_t H, int64_t sB, int64_t sH,
              int64_t sN, int64_t oB, int64_t oH, int64_t oN) {
  hipStream_t s = (hipStream_t)stream_;
  const int64_t rows = BH * N;
  int64_t db = (rows + 3) / 4;
  if (db > 2048) db = 2048;
  hipLaunchKernelGGL(k_fa_delta, dim3((unsigned)db), dim3(256), 0, s,
                     (const __hip_bfloat16*)dout, (const __hip_bfloat16*)o,
                     delta, rows);
  dim3 grid((unsigned)((N + FA_BM - 1) / FA_BM), (unsigned)BH);
  hipLaunchKernelGGL(k_fa_bwd_dq, grid, dim3(512), 0, s,
                     (const __hip_bfloat16*)q, (const __hip_bfloat16*)k,
                     (const __hip_bfloat16*)v, (const __hip_bfloat16*)dout,
                     lse, delta, (__hip_bfloat16*)dq, N, scale, causal,
                     H, sB, sH, sN, oB, oH, oN);
  hipLaunchKernelGGL(k_fa_bwd_dkv, grid, dim3(512), 0, s,
                     (const __hip_bfloat16*)q, (const __hip_bfloat16*)k,
                     (const __hip_bfloat16*)v, (const __hip_bfloat16*)dout,
                     lse, delta, (__hip_bfloat16*)dk, (__hip_bfloat16*)dv, N,
                     scale, causal, H, sB, sH, sN, oB, oH, oN);
  return (int)hipGetLastError();
}

int ps_fa_selfcheck(void* stream_, const void* a, const void* b, float* c) {
  hipStream_t s = (hipStream_t)stream_;
  hipLaunchKernelGGL(k_fa_selfcheck, dim3(1), dim3(64), 0, s,
                     (const __hip_bfloat16*)a, (const __hip_bfloat16*)b, c);
  return (int)hipGetLastError();
}

}  // extern "C"
