// Fused attention softmax (forward + backward) for the transformer's
// short windows — the glue between the two hipBLASLt bmm GEMMs of
// S = Q@K^T and O = P@V (models/transformer.py attention interface
// "srx_window").  Replaces BOTH measured losers at L<=128:
//   * aotriton flash backward (5.7x its forward at these windows —
//     profiles/trf262k_r2_kernel_stats.csv), and
//   * the eager math-SDPA elementwise chain (~6 kernels per call).
// One WAVE per attention row; a lane holds 4 consecutive j in registers
// (L <= 256), so the whole row lives in the wave: masked max, exp, sum,
// normalize, dropout and the store happen in ONE pass.
//
// Dropout follows HF eager semantics (mask AFTER normalize, scaled by
// 1/keep); the philox counter is (row << 6) | lane so the backward
// regenerates the identical mask without storing it.
// lse[r] = m + log(sum exp(scale*s - m)): exp(scale*s - lse) IS the
// normalized probability — the backward recomputes P from S + lse.
#pragma once
#include <hiprand/hiprand_kernel.h>

#include "srx_common.hip.h"

#define SRX_ATTN_MAX_L 256

__device__ __forceinline__ float srx_wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, SRX_WAVE));
  return v;
}

// S [NH, L, L] (raw Q@K^T, unscaled) -> P~ [NH, L, L] (normalized,
// dropout-masked probs, compute dtype) + lse [NH, L] fp32.
// lens [N]: valid prefix per window (N = NH / heads); rows i >= len and
// columns j >= len are masked.
template <typename T, bool DROP>
__global__ void attn_softmax_fwd_kernel(
    const T* __restrict__ S, T* __restrict__ P, float* __restrict__ lse,
    const int32_t* __restrict__ lens, long n_rows, int L, int heads,
    float scale, float keep, unsigned long long seed) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long r = wave; r < n_rows; r += nwaves) {
    const long nh = r / L;
    const int i = (int)(r % L);
    const int len = lens[nh / heads];
    T* prow = P + r * (long)L;
    const int j0 = lane * 4;
    if (i >= len) {  // padding row: zero probs, zero lse
      for (int j = j0; j < L; j += SRX_WAVE * 4)
        for (int e = 0; e < 4 && j + e < L; e++) Elem<T>::st(prow + j + e, 0.f);
      if (lane == 0) lse[r] = 0.f;
      continue;
    }
    const T* srow = S + r * (long)L;
    float v[4];
#pragma unroll
    for (int e = 0; e < 4; e++) {
      int j = j0 + e;
      v[e] = (j < len) ? scale * Elem<T>::ld(srow + j) : -1e30f;
    }
    float m = fmaxf(fmaxf(v[0], v[1]), fmaxf(v[2], v[3]));
    m = srx_wave_max(m);
    float s = 0.f;
#pragma unroll
    for (int e = 0; e < 4; e++) {
      v[e] = (j0 + e < len) ? __expf(v[e] - m) : 0.f;
      s += v[e];
    }
    s = wave_reduce_sum(s);
    float inv = 1.0f / s;
    if (DROP) {
      hiprandStatePhilox4_32_10_t st;
      hiprand_init(seed, ((unsigned long long)r << 6) | (unsigned)lane, 0, &st);
      float4 u = hiprand_uniform4(&st);
      float um[4] = {u.x, u.y, u.z, u.w};
#pragma unroll
      for (int e = 0; e < 4; e++)
        v[e] = um[e] < keep ? v[e] * inv / keep : 0.f;
    } else {
#pragma unroll
      for (int e = 0; e < 4; e++) v[e] *= inv;
    }
#pragma unroll
    for (int e = 0; e < 4; e++)
      if (j0 + e < L) Elem<T>::st(prow + j0 + e, v[e]);
    if (lane == 0) lse[r] = m + __logf(s);
  }
}

// dS (w.r.t. the RAW S) from (S, lse, dP~): P = exp(scale*S - lse);
// dP = dP~ * dropout_mask; delta_i = sum_j P*dP; dS = scale*P*(dP-delta).
template <typename T, bool DROP>
__global__ void attn_softmax_bwd_kernel(
    const T* __restrict__ S, const float* __restrict__ lse,
    const T* __restrict__ dPt, T* __restrict__ dS,
    const int32_t* __restrict__ lens, long n_rows, int L, int heads,
    float scale, float keep, unsigned long long seed) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long r = wave; r < n_rows; r += nwaves) {
    const long nh = r / L;
    const int i = (int)(r % L);
    const int len = lens[nh / heads];
    T* drow = dS + r * (long)L;
    const int j0 = lane * 4;
    if (i >= len) {
      for (int j = j0; j < L; j += SRX_WAVE * 4)
        for (int e = 0; e < 4 && j + e < L; e++) Elem<T>::st(drow + j + e, 0.f);
      continue;
    }
    const T* srow = S + r * (long)L;
    const T* dprow = dPt + r * (long)L;
    const float l = lse[r];
    float p[4], dp[4];
#pragma unroll
    for (int e = 0; e < 4; e++) {
      int j = j0 + e;
      if (j < len) {
        p[e] = __expf(scale * Elem<T>::ld(srow + j) - l);
        dp[e] = Elem<T>::ld(dprow + j);
      } else {
        p[e] = 0.f;
        dp[e] = 0.f;
      }
    }
    if (DROP) {
      hiprandStatePhilox4_32_10_t st;
      hiprand_init(seed, ((unsigned long long)r << 6) | (unsigned)lane, 0, &st);
      float4 u = hiprand_uniform4(&st);
      float um[4] = {u.x, u.y, u.z, u.w};
#pragma unroll
      for (int e = 0; e < 4; e++) dp[e] = um[e] < keep ? dp[e] / keep : 0.f;
    }
    float delta = p[0] * dp[0] + p[1] * dp[1] + p[2] * dp[2] + p[3] * dp[3];
    delta = wave_reduce_sum(delta);
#pragma unroll
    for (int e = 0; e < 4; e++)
      if (j0 + e < L)
        Elem<T>::st(drow + j0 + e, scale * p[e] * (dp[e] - delta));
  }
}

// ===================== fused flash-style attention (MFMA) ==============
// One workgroup per (window, head); S <= 96, D = 64: Q/K/V (+dO) tiles
// and the S x S probabilities live ENTIRELY in LDS — no HBM-materialized
// S/P (the bmm+softmax formulation above measured 236k vs aotriton
// flash's 294k words/s: the S/P round trips cost more than flash's slow
// backward).  v_mfma_f32_32x32x16_bf16 tiles, one wave per 32-row
// M-tile (the mwe_layer_fwd fragment/acc mapping: A row = lane&31,
// k-half = lane>>5; acc element rr -> row (rr&3)+8*(rr>>2)+4*(lane>>5),
// col lane&31).  Forward saves ONLY lse; backward recomputes P.
// Dropout: philox per element (counter (nh*L+i)*L+j), regenerated in
// the backward — no stored mask.
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short srx_attn_bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float srx_attn_f32x16;

#define SRX_ATTN_FUSED_MAX_L 96
#define SRX_ATTN_LDQ 72    // row-major [Lp][72] (D=64 + 16B-align pad)
#define SRX_ATTN_LDT 104   // transposed / prob tiles [.][104]

__device__ __forceinline__ srx_attn_bf16x8
srx_attn_frag(const bf16_t* base, int ld, int row, int k0, int lane) {
  return *(const srx_attn_bf16x8*)(base + (size_t)(row + (lane & 31)) * ld +
                                   k0 + 8 * (lane >> 5));
}

// strided fragment: A^T — element e of the fragment comes from
// src[(k0 + 8*(lane>>5) + e)][row + (lane&31)] of an [.][ld] tile
__device__ __forceinline__ srx_attn_bf16x8
srx_attn_frag_t(const bf16_t* base, int ld, int row, int k0, int lane) {
  union {
    srx_attn_bf16x8 v;
    bf16_t u[8];
  } f;
  const int k = k0 + 8 * (lane >> 5);
  const int c = row + (lane & 31);
#pragma unroll
  for (int e = 0; e < 8; e++) f.u[e] = base[(size_t)(k + e) * ld + c];
  return f.v;
}

__device__ __forceinline__ float srx_attn_row_red_max(float v) {
#pragma unroll
  for (int off = 1; off < 32; off <<= 1) v = fmaxf(v, __shfl_xor(v, off, SRX_WAVE));
  return v;
}

__device__ __forceinline__ float srx_attn_row_red_sum(float v) {
#pragma unroll
  for (int off = 1; off < 32; off <<= 1) v += __shfl_xor(v, off, SRX_WAVE);
  return v;
}

// Batched dropout multipliers for the MFMA acc layout: a lane's 16 acc
// elements per tile are 4 groups of 4 CONSECUTIVE rows at one column, so
// ONE philox draw (uniform4, counter (nh*L + j)*ceil(L/4) + r0/4) covers
// a group — 12 inits per lane instead of 48/96 (per-element philox made
// the first fused version SLOWER than aotriton under training dropout).
// dmul[t][rr] = 0 (dropped) or 1/keep; identical in fwd and bwd.
template <int NT>
__device__ __forceinline__ void srx_attn_drop_mul(
    float dmul[NT][16], unsigned long long seed, long nh, int L,
    int m0, int lane, float keep) {
  const int col = lane & 31;
  const int nq4 = (L + 3) >> 2;
  const float inv_keep = 1.0f / keep;
#pragma unroll
  for (int g = 0; g < 4; g++) {
    const int r0 = m0 + 8 * g + 4 * (lane >> 5);
#pragma unroll
    for (int t = 0; t < NT; t++) {
      const int j = 32 * t + col;
      hiprandStatePhilox4_32_10_t st;
      hiprand_init(seed,
                   ((unsigned long long)(nh * (long)L + j)) * nq4 + (r0 >> 2),
                   0, &st);
      float4 u = hiprand_uniform4(&st);
      const float uu[4] = {u.x, u.y, u.z, u.w};
#pragma unroll
      for (int c = 0; c < 4; c++)
        dmul[t][(g << 2) | c] = uu[c] < keep ? inv_keep : 0.f;
    }
  }
}

// ------------------------------------------------------------- forward
// Q/K/V: [NH, L, 64] (compute dtype = bf16 only); O same; lse [NH, L].
template <bool DROP, int NT>
__global__ __launch_bounds__(192) void attn_fused_fwd_kernel(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, const int32_t* __restrict__ lens,
    bf16_t* __restrict__ O, float* __restrict__ lse, long NH, int L,
    int heads, float scale, float keep, unsigned long long seed) {
  const int D = 64;
  constexpr int Lp = 32 * NT;  // compile-time: keeps p/dmul/acc arrays in
                               // REGISTERS (runtime-NT indexing lowered
                               // them to 640 B/lane of global scratch)
  extern __shared__ char smem[];
  bf16_t* Qt = (bf16_t*)smem;                       // [Lp][72]
  bf16_t* Kt = Qt + (size_t)Lp * SRX_ATTN_LDQ;      // [Lp][72]
  bf16_t* VtT = Kt + (size_t)Lp * SRX_ATTN_LDQ;     // [64][104] (V^T)
  bf16_t* Pt = VtT + (size_t)64 * SRX_ATTN_LDT;     // [Lp][104]
  const int tid = threadIdx.x;
  const int lane = tid & (SRX_WAVE - 1);
  const int w = tid / SRX_WAVE;  // wave = M-tile
  const int nthr = blockDim.x;
  for (long nh = blockIdx.x; nh < NH; nh += gridDim.x) {
    const int len = lens[nh / heads];
    const bf16_t* Qg = Q + nh * (size_t)L * D;
    const bf16_t* Kg = K + nh * (size_t)L * D;
    const bf16_t* Vg = V + nh * (size_t)L * D;
    __syncthreads();  // previous iteration's LDS reads complete
    for (int idx = tid; idx < Lp * D; idx += nthr) {
      int r = idx / D, d = idx - r * D;
      bf16_t qv = 0, kv = 0, vv = 0;
      if (r < L) {
        qv = Qg[(size_t)r * D + d];
        kv = Kg[(size_t)r * D + d];
        vv = Vg[(size_t)r * D + d];
      }
      Qt[(size_t)r * SRX_ATTN_LDQ + d] = qv;
      Kt[(size_t)r * SRX_ATTN_LDQ + d] = kv;
      VtT[(size_t)d * SRX_ATTN_LDT + r] = vv;
    }
    __syncthreads();
    const int m0 = 32 * w;  // this wave's M-tile row base
    if (m0 < Lp) {
      // ---- S = scale * Q@K^T over NT column tiles
      srx_attn_f32x16 sacc[NT];
#pragma unroll
      for (int t = 0; t < NT; t++)
#pragma unroll
        for (int rr = 0; rr < 16; rr++) sacc[t][rr] = 0.f;
#pragma unroll
      for (int t = 0; t < NT; t++)
#pragma unroll
        for (int k0 = 0; k0 < 64; k0 += 16) {
          srx_attn_bf16x8 a = srx_attn_frag(Qt, SRX_ATTN_LDQ, m0, k0, lane);
          srx_attn_bf16x8 b = srx_attn_frag(Kt, SRX_ATTN_LDQ, 32 * t, k0, lane);
          sacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, sacc[t], 0, 0, 0);
        }
      // ---- rowwise masked softmax across the NT tiles (rows stay
      // within one 32-lane half: xor<32 reductions)
      const int col = lane & 31;
      float p[NT][16];
      float dmul[DROP ? NT : 1][16];
      if (DROP)
        srx_attn_drop_mul<DROP ? NT : 1>(dmul, seed, nh, L, m0, lane, keep);
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        const int r = m0 + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
        float mx = -1e30f;
#pragma unroll
        for (int t = 0; t < NT; t++) {
          int j = 32 * t + col;
          float s = (j < len && r < len) ? scale * sacc[t][rr] : -1e30f;
          p[t][rr] = s;
          mx = fmaxf(mx, s);
        }
        mx = srx_attn_row_red_max(mx);
        float sum = 0.f;
#pragma unroll
        for (int t = 0; t < NT; t++) {
          float e = (p[t][rr] > -1e29f) ? __expf(p[t][rr] - mx) : 0.f;
          p[t][rr] = e;
          sum += e;
        }
        sum = srx_attn_row_red_sum(sum);
        float inv = sum > 0.f ? 1.0f / sum : 0.f;
#pragma unroll
        for (int t = 0; t < NT; t++) {
          float v = p[t][rr] * inv;
          if (DROP) v *= dmul[DROP ? t : 0][rr];
          p[t][rr] = v;
        }
        if (col == 0 && r < L) lse[nh * (size_t)L + r] = (r < len) ? mx + __logf(sum) : 0.f;
      }
      // stage P (dropped+normalized) for the second GEMM
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        const int r = m0 + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
#pragma unroll
        for (int t = 0; t < NT; t++)
          Pt[(size_t)r * SRX_ATTN_LDT + 32 * t + col] = f2bf(p[t][rr]);
      }
      // ---- O = P@V (A = Pt rows, B = V^T rows), N = 64 -> 2 tiles
      srx_attn_f32x16 oacc[2];
#pragma unroll
      for (int t = 0; t < 2; t++)
#pragma unroll
        for (int rr = 0; rr < 16; rr++) oacc[t][rr] = 0.f;
      for (int k0 = 0; k0 < Lp; k0 += 16)
#pragma unroll
        for (int t = 0; t < 2; t++) {
          srx_attn_bf16x8 a = srx_attn_frag(Pt, SRX_ATTN_LDT, m0, k0, lane);
          srx_attn_bf16x8 b = srx_attn_frag(VtT, SRX_ATTN_LDT, 32 * t, k0, lane);
          oacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, oacc[t], 0, 0, 0);
        }
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        const int r = m0 + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
        if (r < L) {
          for (int t = 0; t < 2; t++)
            O[nh * (size_t)L * D + (size_t)r * D + 32 * t + col] =
                f2bf(r < len ? oacc[t][rr] : 0.f);
        }
      }
    }
  }
}

// ------------------------------------------------------------ backward
// Recomputes P from (Q, K, lse); LDS holds row-major Q/K/V/dO, transposed
// Q^T/K^T/dO^T (for the dQ/dK/dV GEMMs' B operands) and the P / dS
// matrices.  GEMM chain per (window, head):
//   (1) S = Q@K^T -> P (regs) and P~ -> Pt (LDS, dropout applied)
//   (2) dP~ = dO@V^T;  dP = dP~*mask;  delta = rowsum(P*dP)
//       dS = scale*P*(dP - delta) -> DSt (LDS)
//   (3) dQ = dS@K      (B = K^T)
//   (4) dK = dS^T@Q    (A = DSt strided, B = Q^T)
//   (5) dV = P~^T@dO   (A = Pt strided,  B = dO^T)
template <bool DROP, int NT>
__global__ __launch_bounds__(192) void attn_fused_bwd_kernel(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, const bf16_t* __restrict__ dO,
    const float* __restrict__ lse, const int32_t* __restrict__ lens,
    bf16_t* __restrict__ dQ, bf16_t* __restrict__ dK,
    bf16_t* __restrict__ dV, long NH, int L, int heads, float scale,
    float keep, unsigned long long seed) {
  const int D = 64;
  constexpr int Lp = 32 * NT;
  extern __shared__ char smem[];
  bf16_t* Qt = (bf16_t*)smem;                        // [Lp][72]
  bf16_t* Kt = Qt + (size_t)Lp * SRX_ATTN_LDQ;       // [Lp][72]
  bf16_t* Vt = Kt + (size_t)Lp * SRX_ATTN_LDQ;       // [Lp][72]
  bf16_t* dOt = Vt + (size_t)Lp * SRX_ATTN_LDQ;      // [Lp][72]
  bf16_t* QtT = dOt + (size_t)Lp * SRX_ATTN_LDQ;     // [64][104]
  bf16_t* KtT = QtT + (size_t)64 * SRX_ATTN_LDT;     // [64][104]
  bf16_t* dOtT = KtT + (size_t)64 * SRX_ATTN_LDT;    // [64][104]
  bf16_t* Pt = dOtT + (size_t)64 * SRX_ATTN_LDT;     // [Lp][104]
  bf16_t* DSt = Pt + (size_t)Lp * SRX_ATTN_LDT;      // [Lp][104]
  const int tid = threadIdx.x;
  const int lane = tid & (SRX_WAVE - 1);
  const int w = tid / SRX_WAVE;
  const int nthr = blockDim.x;
  for (long nh = blockIdx.x; nh < NH; nh += gridDim.x) {
    const int len = lens[nh / heads];
    const size_t gbase = nh * (size_t)L * D;
    __syncthreads();
    for (int idx = tid; idx < Lp * D; idx += nthr) {
      int r = idx / D, d = idx - r * D;
      bf16_t qv = 0, kv = 0, vv = 0, dv = 0;
      if (r < L) {
        qv = Q[gbase + (size_t)r * D + d];
        kv = K[gbase + (size_t)r * D + d];
        vv = V[gbase + (size_t)r * D + d];
        dv = dO[gbase + (size_t)r * D + d];
      }
      Qt[(size_t)r * SRX_ATTN_LDQ + d] = qv;
      Kt[(size_t)r * SRX_ATTN_LDQ + d] = kv;
      Vt[(size_t)r * SRX_ATTN_LDQ + d] = vv;
      dOt[(size_t)r * SRX_ATTN_LDQ + d] = dv;
      QtT[(size_t)d * SRX_ATTN_LDT + r] = qv;
      KtT[(size_t)d * SRX_ATTN_LDT + r] = kv;
      dOtT[(size_t)d * SRX_ATTN_LDT + r] = dv;
    }
    __syncthreads();
    const int m0 = 32 * w;
    const int col = lane & 31;
    if (m0 < Lp) {
      // ---- (1) recompute P rows of this M-tile
      srx_attn_f32x16 sacc[NT];
#pragma unroll
      for (int t = 0; t < NT; t++)
#pragma unroll
        for (int rr = 0; rr < 16; rr++) sacc[t][rr] = 0.f;
#pragma unroll
      for (int t = 0; t < NT; t++)
#pragma unroll
        for (int k0 = 0; k0 < 64; k0 += 16) {
          srx_attn_bf16x8 a = srx_attn_frag(Qt, SRX_ATTN_LDQ, m0, k0, lane);
          srx_attn_bf16x8 b = srx_attn_frag(Kt, SRX_ATTN_LDQ, 32 * t, k0, lane);
          sacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, sacc[t], 0, 0, 0);
        }
      float p[NT][16];    // undropped P
      float dmul[DROP ? NT : 1][16];  // dropout multipliers (== fwd's)
      if (DROP)
        srx_attn_drop_mul<DROP ? NT : 1>(dmul, seed, nh, L, m0, lane, keep);
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        const int r = m0 + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
        const float l = (r < len) ? lse[nh * (size_t)L + r] : 0.f;
#pragma unroll
        for (int t = 0; t < NT; t++) {
          int j = 32 * t + col;
          float v = (r < len && j < len) ? __expf(scale * sacc[t][rr] - l) : 0.f;
          p[t][rr] = v;
          if (DROP) v *= dmul[DROP ? t : 0][rr];
          Pt[(size_t)r * SRX_ATTN_LDT + j] = f2bf(v);  // P~ (for dV)
        }
      }
      // ---- (2) dP~ = dO@V^T; dS
      srx_attn_f32x16 dpacc[NT];
#pragma unroll
      for (int t = 0; t < NT; t++)
#pragma unroll
        for (int rr = 0; rr < 16; rr++) dpacc[t][rr] = 0.f;
#pragma unroll
      for (int t = 0; t < NT; t++)
#pragma unroll
        for (int k0 = 0; k0 < 64; k0 += 16) {
          srx_attn_bf16x8 a = srx_attn_frag(dOt, SRX_ATTN_LDQ, m0, k0, lane);
          srx_attn_bf16x8 b = srx_attn_frag(Vt, SRX_ATTN_LDQ, 32 * t, k0, lane);
          dpacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, dpacc[t], 0, 0, 0);
        }
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        const int r = m0 + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
        float delta = 0.f;
#pragma unroll
        for (int t = 0; t < NT; t++) {
          float dp = dpacc[t][rr];
          if (DROP) dp *= dmul[DROP ? t : 0][rr];
          dpacc[t][rr] = dp;
          delta += p[t][rr] * dp;
        }
        delta = srx_attn_row_red_sum(delta);
#pragma unroll
        for (int t = 0; t < NT; t++) {
          float ds = scale * p[t][rr] * (dpacc[t][rr] - delta);
          DSt[(size_t)r * SRX_ATTN_LDT + 32 * t + col] = f2bf(ds);
        }
      }
    }
    __syncthreads();  // Pt / DSt complete across all waves
    if (m0 < Lp) {
      // ---- (3) dQ = dS@K  (A = DSt rows, B = K^T rows)
      srx_attn_f32x16 qacc[2];
#pragma unroll
      for (int t = 0; t < 2; t++)
#pragma unroll
        for (int rr = 0; rr < 16; rr++) qacc[t][rr] = 0.f;
      for (int k0 = 0; k0 < Lp; k0 += 16)
#pragma unroll
        for (int t = 0; t < 2; t++) {
          srx_attn_bf16x8 a = srx_attn_frag(DSt, SRX_ATTN_LDT, m0, k0, lane);
          srx_attn_bf16x8 b = srx_attn_frag(KtT, SRX_ATTN_LDT, 32 * t, k0, lane);
          qacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, qacc[t], 0, 0, 0);
        }
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        const int r = m0 + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
        if (r < L)
          for (int t = 0; t < 2; t++)
            dQ[gbase + (size_t)r * D + 32 * t + col] = f2bf(qacc[t][rr]);
      }
      // ---- (4) dK = dS^T@Q  and (5) dV = P~^T@dO  (j-tile = this wave)
      srx_attn_f32x16 kacc[2], vacc[2];
#pragma unroll
      for (int t = 0; t < 2; t++)
#pragma unroll
        for (int rr = 0; rr < 16; rr++) { kacc[t][rr] = 0.f; vacc[t][rr] = 0.f; }
      for (int k0 = 0; k0 < Lp; k0 += 16) {
        srx_attn_bf16x8 at_ds = srx_attn_frag_t(DSt, SRX_ATTN_LDT, m0, k0, lane);
        srx_attn_bf16x8 at_p = srx_attn_frag_t(Pt, SRX_ATTN_LDT, m0, k0, lane);
#pragma unroll
        for (int t = 0; t < 2; t++) {
          srx_attn_bf16x8 bq = srx_attn_frag(QtT, SRX_ATTN_LDT, 32 * t, k0, lane);
          srx_attn_bf16x8 bo = srx_attn_frag(dOtT, SRX_ATTN_LDT, 32 * t, k0, lane);
          kacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(at_ds, bq, kacc[t], 0, 0, 0);
          vacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(at_p, bo, vacc[t], 0, 0, 0);
        }
      }
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        const int j = m0 + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
        if (j < L)
          for (int t = 0; t < 2; t++) {
            dK[gbase + (size_t)j * D + 32 * t + col] = f2bf(kacc[t][rr]);
            dV[gbase + (size_t)j * D + 32 * t + col] = f2bf(vacc[t][rr]);
          }
      }
    }
  }
}
