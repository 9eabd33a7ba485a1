// Bandwidth-bound kernels: seq2col, maxout, layernorm, fused Adam.
// Semantics defined by ops/torch_ref.py (the fp32 torch reference these are
// tested against).  All are memory-bound: vectorized loads (V elems/lane:
// 8 for bf16 = 16B, 4 for f32 = 16B per guide G13), grid-stride loops,
// fp32 accumulation.
#pragma once
#include "srx_common.hip.h"

// ------------------------------------------------------------- seq2col
// Y[t, s*W + w] = X[t-1+s, w] for s in {0,1,2}, zeroed across doc bounds.
// is_start/is_end: per-token doc-boundary bytes (computed once per batch).
template <typename T, int V>
__global__ void seq2col_fwd_kernel(const T* __restrict__ X, T* __restrict__ Y,
                                   const uint8_t* __restrict__ is_start,
                                   const uint8_t* __restrict__ is_end,
                                   long nT, int W) {
  const long chunks_per_sec = W / V;
  const long total = nT * 3 * chunks_per_sec;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long t = i / (3 * chunks_per_sec);
    long rem = i % (3 * chunks_per_sec);
    int s = (int)(rem / chunks_per_sec);
    int w = (int)(rem % chunks_per_sec) * V;
    long src_t = t + s - 1;
    bool zero = (s == 0 && (t == 0 || is_start[t])) ||
                (s == 2 && (t == nT - 1 || is_end[t]));
    T* out = Y + (t * 3 + s) * (long)W + w;
    float v[V];
    if (zero || src_t < 0 || src_t >= nT) {
#pragma unroll
      for (int k = 0; k < V; k++) v[k] = 0.0f;
    } else {
      ElemV<T, V>::ld(X + src_t * (long)W + w, v);
    }
    ElemV<T, V>::st(out, v);
  }
}

// dX[t,w] = dY[t, W+w] + (next's prev-slot) + (prev's next-slot)
//           [+ residual[t,w] when given — fuses the encoder block's
//            `dX += dY_upstream` residual add]
template <typename T, int V>
__global__ void seq2col_bwd_kernel(const T* __restrict__ dY, T* __restrict__ dX,
                                   const uint8_t* __restrict__ is_start,
                                   const uint8_t* __restrict__ is_end,
                                   const T* __restrict__ residual,
                                   long nT, int W) {
  const long chunks = W / V;
  const long total = nT * chunks;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long t = i / chunks;
    int w = (int)(i % chunks) * V;
    const long row = 3L * W;
    float acc[V], tmp[V];
    ElemV<T, V>::ld(dY + t * row + W + w, acc);
    if (t + 1 < nT && !is_start[t + 1]) {
      ElemV<T, V>::ld(dY + (t + 1) * row + w, tmp);
#pragma unroll
      for (int k = 0; k < V; k++) acc[k] += tmp[k];
    }
    if (t > 0 && !is_end[t - 1]) {
      ElemV<T, V>::ld(dY + (t - 1) * row + 2 * W + w, tmp);
#pragma unroll
      for (int k = 0; k < V; k++) acc[k] += tmp[k];
    }
    if (residual) {
      ElemV<T, V>::ld(residual + t * (long)W + w, tmp);
#pragma unroll
      for (int k = 0; k < V; k++) acc[k] += tmp[k];
    }
    ElemV<T, V>::st(dX + t * (long)W + w, acc);
  }
}

// ------------------------------------------------------------- maxout
// Pieces-major: X [N, P, W] -> Y [N, W], which [N, W] (uint8).
template <typename T, int V>
__global__ void maxout_fwd_kernel(const T* __restrict__ X, T* __restrict__ Y,
                                  uint8_t* __restrict__ which, long N, int P, int W) {
  const long chunks = W / V;
  const long total = N * chunks;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long n = i / chunks;
    int w = (int)(i % chunks) * V;
    float best[V], cand[V];
    uint8_t arg[V];
    const T* base = X + n * (long)P * W + w;
    ElemV<T, V>::ld(base, best);
#pragma unroll
    for (int k = 0; k < V; k++) arg[k] = 0;
    for (int p = 1; p < P; p++) {
      ElemV<T, V>::ld(base + (long)p * W, cand);
#pragma unroll
      for (int k = 0; k < V; k++) {
        if (cand[k] > best[k]) { best[k] = cand[k]; arg[k] = (uint8_t)p; }
      }
    }
    ElemV<T, V>::st(Y + n * (long)W + w, best);
    uint8_t* wh = which + n * (long)W + w;
    if (V == 8) {  // one 8-byte store for the argmax bytes
      uint32_t lo = 0, hi = 0;
#pragma unroll
      for (int k = 0; k < 4 && k < V; k++) lo |= (uint32_t)arg[k] << (8 * k);
#pragma unroll
      for (int k = 4; k < V; k++) hi |= (uint32_t)arg[k] << (8 * (k - 4));
      ((uint32_t*)wh)[0] = lo;
      ((uint32_t*)wh)[1] = hi;
    } else {
#pragma unroll
      for (int k = 0; k < V; k++) wh[k] = arg[k];
    }
  }
}

template <typename T, int V>
__global__ void maxout_bwd_kernel(const T* __restrict__ dY,
                                  const uint8_t* __restrict__ which,
                                  T* __restrict__ dX, long N, int P, int W) {
  const long chunks = W / V;
  const long total = N * (long)P * chunks;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long n = i / (P * chunks);
    long rem = i % (P * chunks);
    int p = (int)(rem / chunks);
    int w = (int)(rem % chunks) * V;
    const uint8_t* wh = which + n * (long)W + w;
    float v[V];
    ElemV<T, V>::ld(dY + n * (long)W + w, v);
    if (V == 8) {
      uint32_t lo = ((const uint32_t*)wh)[0], hi = ((const uint32_t*)wh)[1];
#pragma unroll
      for (int k = 0; k < V; k++) {
        uint8_t a = (uint8_t)(((k < 4 ? lo : hi) >> (8 * (k & 3))) & 0xff);
        if (a != p) v[k] = 0.0f;
      }
    } else {
#pragma unroll
      for (int k = 0; k < V; k++)
        if (wh[k] != p) v[k] = 0.0f;
    }
    ElemV<T, V>::st(dX + (n * (long)P + p) * W + w, v);
  }
}

// ----------------------------------------------------------- layernorm
// One wave per row (W <= a few hundred); 4 waves per block.
template <typename T>
__global__ void layernorm_fwd_kernel(const T* __restrict__ X,
                                     const T* __restrict__ g,
                                     const T* __restrict__ b,
                                     T* __restrict__ Y,
                                     float* __restrict__ mu_out,
                                     float* __restrict__ rstd_out,
                                     long N, int W, float eps) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long n = wave; n < N; n += nwaves) {
    const T* row = X + n * (long)W;
    float s = 0.f, sq = 0.f;
    for (int w = lane; w < W; w += SRX_WAVE) {
      float v = Elem<T>::ld(row + w);
      s += v;
      sq += v * v;
    }
    s = wave_reduce_sum(s);
    sq = wave_reduce_sum(sq);
    float mu = s / W;
    float var = sq / W - mu * mu;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (lane == 0) {
      mu_out[n] = mu;
      rstd_out[n] = rstd;
    }
    T* out = Y + n * (long)W;
    for (int w = lane; w < W; w += SRX_WAVE) {
      float v = (Elem<T>::ld(row + w) - mu) * rstd;
      Elem<T>::st(out + w, v * Elem<T>::ld(g + w) + Elem<T>::ld(b + w));
    }
  }
}

#define SRX_LN_MAX_W 1024

// Wide-row variants (W % 128 == 0, e.g. the trf's 768): each lane owns
// CONTIGUOUS PAIRS so every access is a full dword (the scalar kernels
// move bf16 2 bytes at a time — layernorm was 11%% of trf GPU time).
template <typename T>
__global__ void layernorm_fwd_v2_kernel(const T* __restrict__ X,
                                        const T* __restrict__ g,
                                        const T* __restrict__ b,
                                        T* __restrict__ Y,
                                        float* __restrict__ mu_out,
                                        float* __restrict__ rstd_out,
                                        long N, int W, float eps) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int npairs = W / 2;
  for (long n = wave; n < N; n += nwaves) {
    const T* row = X + n * (long)W;
    float s = 0.f, sq = 0.f;
    for (int p = lane; p < npairs; p += SRX_WAVE) {
      float v[2];
      Elem2<T>::ld(row + 2 * p, v);
      s += v[0] + v[1];
      sq += v[0] * v[0] + v[1] * v[1];
    }
    s = wave_reduce_sum(s);
    sq = wave_reduce_sum(sq);
    float mu = s / W;
    float var = sq / W - mu * mu;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (lane == 0) {
      mu_out[n] = mu;
      rstd_out[n] = rstd;
    }
    T* out = Y + n * (long)W;
    for (int p = lane; p < npairs; p += SRX_WAVE) {
      float v[2], gg[2], bb[2], o[2];
      Elem2<T>::ld(row + 2 * p, v);
      Elem2<T>::ld(g + 2 * p, gg);
      Elem2<T>::ld(b + 2 * p, bb);
      o[0] = (v[0] - mu) * rstd * gg[0] + bb[0];
      o[1] = (v[1] - mu) * rstd * gg[1] + bb[1];
      Elem2<T>::st(out + 2 * p, o);
    }
  }
}

template <typename T, bool DET = false>
__global__ void layernorm_bwd_v2_kernel(const T* __restrict__ dY,
                                        const T* __restrict__ X,
                                        const T* __restrict__ g,
                                        const float* __restrict__ mu,
                                        const float* __restrict__ rstd,
                                        T* __restrict__ dX,
                                        void* __restrict__ dg32,
                                        void* __restrict__ db32,
                                        long N, int W) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int npairs = W / 2;
  const int ncols = (npairs + SRX_WAVE - 1) / SRX_WAVE;
  float dg_loc[SRX_LN_MAX_W / SRX_WAVE], db_loc[SRX_LN_MAX_W / SRX_WAVE];
  for (int c = 0; c < 2 * ncols; c++) { dg_loc[c] = 0.f; db_loc[c] = 0.f; }
  for (long n = wave; n < N; n += nwaves) {
    const T* xrow = X + n * (long)W;
    const T* dyrow = dY + n * (long)W;
    float m = mu[n], r = rstd[n];
    float s1 = 0.f, s2 = 0.f;
    for (int c = 0; c < ncols; c++) {
      int p = lane + c * SRX_WAVE;
      if (p >= npairs) break;
      float x[2], dy[2], gg[2];
      Elem2<T>::ld(xrow + 2 * p, x);
      Elem2<T>::ld(dyrow + 2 * p, dy);
      Elem2<T>::ld(g + 2 * p, gg);
#pragma unroll
      for (int e = 0; e < 2; e++) {
        float xhat = (x[e] - m) * r;
        float dxhat = dy[e] * gg[e];
        s1 += dxhat;
        s2 += dxhat * xhat;
        dg_loc[2 * c + e] += dy[e] * xhat;
        db_loc[2 * c + e] += dy[e];
      }
    }
    s1 = wave_reduce_sum(s1) / W;
    s2 = wave_reduce_sum(s2) / W;
    T* out = dX + n * (long)W;
    for (int c = 0; c < ncols; c++) {
      int p = lane + c * SRX_WAVE;
      if (p >= npairs) break;
      float x[2], dy[2], gg[2], o[2];
      Elem2<T>::ld(xrow + 2 * p, x);
      Elem2<T>::ld(dyrow + 2 * p, dy);
      Elem2<T>::ld(g + 2 * p, gg);
#pragma unroll
      for (int e = 0; e < 2; e++) {
        float xhat = (x[e] - m) * r;
        float dxhat = dy[e] * gg[e];
        o[e] = r * (dxhat - s1 - xhat * s2);
      }
      Elem2<T>::st(out + 2 * p, o);
    }
  }
  for (int c = 0; c < ncols; c++) {
    int p = lane + c * SRX_WAVE;
    if (p >= npairs) continue;
#pragma unroll
    for (int e = 0; e < 2; e++) {
      int w = 2 * p + e;
      if (dg_loc[2 * c + e] != 0.f) srx_atomic_add<DET>(dg32, w, dg_loc[2 * c + e]);
      if (db_loc[2 * c + e] != 0.f) srx_atomic_add<DET>(db32, w, db_loc[2 * c + e]);
    }
  }
}

// dX per row; dg/db: per-wave REGISTER accumulation over all the wave's
// rows, ONE atomicAdd per column per wave at the end.  (A per-row-per-elem
// atomic version serialized on the 2*W hot addresses: 772 us/call at
// N=32k, W=96 — this shape runs in ~30 us.)  Needs W <= SRX_LN_MAX_W.
// DET=true: dg/db buffers are int64 fixed-point (bit-deterministic).
template <typename T, bool DET = false>
__global__ void layernorm_bwd_kernel(const T* __restrict__ dY,
                                     const T* __restrict__ X,
                                     const T* __restrict__ g,
                                     const float* __restrict__ mu,
                                     const float* __restrict__ rstd,
                                     T* __restrict__ dX,
                                     void* __restrict__ dg32,
                                     void* __restrict__ db32,
                                     long N, int W) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int ncols = (W + SRX_WAVE - 1) / SRX_WAVE;  // columns per lane
  float dg_loc[SRX_LN_MAX_W / SRX_WAVE];
  float db_loc[SRX_LN_MAX_W / SRX_WAVE];
  for (int c = 0; c < ncols; c++) { dg_loc[c] = 0.f; db_loc[c] = 0.f; }
  for (long n = wave; n < N; n += nwaves) {
    const T* xrow = X + n * (long)W;
    const T* dyrow = dY + n * (long)W;
    float m = mu[n], r = rstd[n];
    float s1 = 0.f, s2 = 0.f;
    for (int c = 0; c < ncols; c++) {
      int w = lane + c * SRX_WAVE;
      if (w >= W) break;
      float xhat = (Elem<T>::ld(xrow + w) - m) * r;
      float dy = Elem<T>::ld(dyrow + w);
      float dxhat = dy * Elem<T>::ld(g + w);
      s1 += dxhat;
      s2 += dxhat * xhat;
      dg_loc[c] += dy * xhat;
      db_loc[c] += dy;
    }
    s1 = wave_reduce_sum(s1) / W;
    s2 = wave_reduce_sum(s2) / W;
    T* out = dX + n * (long)W;
    for (int c = 0; c < ncols; c++) {
      int w = lane + c * SRX_WAVE;
      if (w >= W) break;
      float xhat = (Elem<T>::ld(xrow + w) - m) * r;
      float dxhat = Elem<T>::ld(dyrow + w) * Elem<T>::ld(g + w);
      Elem<T>::st(out + w, r * (dxhat - s1 - xhat * s2));
    }
  }
  for (int c = 0; c < ncols; c++) {
    int w = lane + c * SRX_WAVE;
    if (w < W) {
      srx_atomic_add<DET>(dg32, w, dg_loc[c]);
      srx_atomic_add<DET>(db32, w, db_loc[c]);
    }
  }
}

// -------------------------------------------------------- dropout mask
// Philox4x32-10 dropout mask (SURVEY §2.5 dropout_mask: hiprand device
// API): one kernel writes the scaled keep-mask directly in the compute
// dtype — replaces the torch chain rand_like(fp32) -> compare -> scale ->
// cast (3 kernels + a full fp32 tensor) used for the fused-MWE mask.
// Reproducible: (seed, offset) are the Philox counter; the python side
// derives them from torch's manual seed + a call counter.
#include <hiprand/hiprand_kernel.h>

template <typename T>
__global__ void dropout_mask_kernel(T* __restrict__ out, long n, float keep,
                                    float scale, unsigned long long seed,
                                    unsigned long long offset) {
  const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  hiprandStatePhilox4_32_10_t st;
  hiprand_init(seed, (unsigned long long)tid, offset, &st);
  for (long j = tid * 4; j < n; j += stride * 4) {
    float4 r = hiprand_uniform4(&st);
    float v[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int k = 0; k < 4; k++)
      if (j + k < n) Elem<T>::st(out + j + k, v[k] < keep ? scale : 0.f);
  }
}

// ---------------------------------------------------------- fused Adam
// One launch for the whole sharded step (SURVEY.md §2.5 fused_adam_sharded):
// grad (model dtype) + fp32 master/m/v -> updated state + model-dtype param.
// Decoupled weight decay; bias correction folded into lr/denom; clip scale
// precomputed on host from the all-reduced global norm.
template <typename T>
__global__ void adam_step_kernel(const T* __restrict__ grad,
                                 float* __restrict__ master,
                                 float* __restrict__ m,
                                 float* __restrict__ v,
                                 T* __restrict__ param_out,
                                 long n, const float* __restrict__ clip_scale_ptr,
                                 float lr,
                                 float beta1, float beta2, float eps,
                                 float wd, float bc1, float bc2) {
  const float clip_scale = clip_scale_ptr ? *clip_scale_ptr : 1.0f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float gv = Elem<T>::ld(grad + i) * clip_scale;
    float p = master[i] * (1.0f - lr * wd);
    float mi = beta1 * m[i] + (1.0f - beta1) * gv;
    float vi = beta2 * v[i] + (1.0f - beta2) * gv * gv;
    m[i] = mi;
    v[i] = vi;
    float denom = sqrtf(vi / bc2) + eps;
    p -= (lr / bc1) * (mi / denom);
    master[i] = p;
    Elem<T>::st(param_out + i, p);
  }
}

// grad-norm^2 partial reduce (one value per block -> atomicAdd)
template <typename T>
__global__ void sqnorm_kernel(const T* __restrict__ x, long n, float* __restrict__ out) {
  float acc = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float v = Elem<T>::ld(x + i);
    acc += v * v;
  }
  acc = wave_reduce_sum(acc);
  __shared__ float wsum[16];
  int wid = threadIdx.x / SRX_WAVE;
  int lane = threadIdx.x & (SRX_WAVE - 1);
  if (lane == 0) wsum[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int i = 0; i < (int)(blockDim.x / SRX_WAVE); i++) s += wsum[i];
    atomicAdd(out, s);
  }
}
