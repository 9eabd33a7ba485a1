// Elementwise activation family — the remainder of Thinc's CUDA kernel
// surface (thinc/backends/_custom_kernels.cu: mish/backprop_mish, swish,
// gelu, clipped_linear — SURVEY.md §2.2 N1).  The BASELINE configs are
// Maxout-based and never launch these, but models configured with
// mish/swish/gelu/relu/hard_* layers need them on the GPU path.
// Vectorized (ElemV b128 moves), fp32 math, backward recomputes from X
// (no saved intermediates — these are bandwidth-trivial shapes).
#pragma once
#include "srx_common.hip.h"

// OP codes (keep in sync with ops/api.py ACT_*)
#define SRX_ACT_MISH 0
#define SRX_ACT_SWISH 1
#define SRX_ACT_GELU 2
#define SRX_ACT_CLIPPED_LINEAR 3

__device__ __forceinline__ float srx_sigmoid(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

__device__ __forceinline__ float srx_softplus(float x) {
  // numerically stable log(1 + e^x)
  return fmaxf(x, 0.0f) + log1pf(__expf(-fabsf(x)));
}

template <int OP>
__device__ __forceinline__ float srx_act_fwd(float x, float slope,
                                             float offset, float lo,
                                             float hi) {
  if (OP == SRX_ACT_MISH) {
    return x * tanhf(srx_softplus(x));
  } else if (OP == SRX_ACT_SWISH) {
    return x * srx_sigmoid(x);
  } else if (OP == SRX_ACT_GELU) {
    return 0.5f * x * (1.0f + erff(x * 0.70710678118654752f));
  } else {  // clipped_linear: clip(slope*x + offset, lo, hi)
    return fminf(fmaxf(slope * x + offset, lo), hi);
  }
}

template <int OP>
__device__ __forceinline__ float srx_act_grad(float x, float slope,
                                              float offset, float lo,
                                              float hi) {
  if (OP == SRX_ACT_MISH) {
    float sp = srx_softplus(x);
    float t = tanhf(sp);
    float sig = srx_sigmoid(x);
    return t + x * sig * (1.0f - t * t);
  } else if (OP == SRX_ACT_SWISH) {
    float sig = srx_sigmoid(x);
    return sig * (1.0f + x * (1.0f - sig));
  } else if (OP == SRX_ACT_GELU) {
    float phi = 0.5f * (1.0f + erff(x * 0.70710678118654752f));
    float pdf = 0.39894228040143268f * __expf(-0.5f * x * x);
    return phi + x * pdf;
  } else {
    float y = slope * x + offset;
    return (y > lo && y < hi) ? slope : 0.0f;
  }
}

template <typename T, int V, int OP>
__global__ void act_fwd_kernel(const T* __restrict__ X, T* __restrict__ Y,
                               long total_chunks, float slope, float offset,
                               float lo, float hi) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total_chunks;
       i += (long)gridDim.x * blockDim.x) {
    float v[V];
    ElemV<T, V>::ld(X + i * V, v);
#pragma unroll
    for (int k = 0; k < V; k++)
      v[k] = srx_act_fwd<OP>(v[k], slope, offset, lo, hi);
    ElemV<T, V>::st(Y + i * V, v);
  }
}

template <typename T, int V, int OP>
__global__ void act_bwd_kernel(const T* __restrict__ dY,
                               const T* __restrict__ X, T* __restrict__ dX,
                               long total_chunks, float slope, float offset,
                               float lo, float hi) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total_chunks;
       i += (long)gridDim.x * blockDim.x) {
    float d[V], x[V];
    ElemV<T, V>::ld(dY + i * V, d);
    ElemV<T, V>::ld(X + i * V, x);
#pragma unroll
    for (int k = 0; k < V; k++)
      d[k] *= srx_act_grad<OP>(x[k], slope, offset, lo, hi);
    ElemV<T, V>::st(dX + i * V, d);
  }
}
