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
