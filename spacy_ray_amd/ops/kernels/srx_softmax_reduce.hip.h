// Fused softmax+cross-entropy (tagger loss) and segmented ragged
// reductions (doc pooling).  Semantics: ops/torch_ref.py + torch CE.
#pragma once
#include "srx_common.hip.h"

// ------------------------------------------------------- softmax + CE
// One wave per row (grid-stride): out d[n,c] = softmax(scores[n]) - onehot,
// 0 for ignored rows (gold < 0); per-wave loss partials -> one atomicAdd
// per wave into loss_out[0]; n_valid counted into loss_out[1].
template <typename T>
__global__ void softmax_ce_kernel(const T* __restrict__ scores,
                                  const int64_t* __restrict__ gold,
                                  T* __restrict__ dScores,
                                  float* __restrict__ loss_out,
                                  long N, int C) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  float loss_acc = 0.f;
  float count_acc = 0.f;
  for (long n = wave; n < N; n += nwaves) {
    const T* row = scores + n * (long)C;
    T* drow = dScores + n * (long)C;
    int64_t g = gold[n];
    if (g < 0) {
      for (int c = lane; c < C; c += SRX_WAVE) Elem<T>::st(drow + c, 0.f);
      continue;
    }
    float m = -1e38f;
    for (int c = lane; c < C; c += SRX_WAVE) m = fmaxf(m, Elem<T>::ld(row + c));
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, SRX_WAVE));
    float z = 0.f;
    for (int c = lane; c < C; c += SRX_WAVE) z += __expf(Elem<T>::ld(row + c) - m);
    z = wave_reduce_sum(z);
    float logz = __logf(z) + m;
    for (int c = lane; c < C; c += SRX_WAVE) {
      float p = __expf(Elem<T>::ld(row + c) - logz);
      Elem<T>::st(drow + c, p - (c == (int)g ? 1.f : 0.f));
    }
    if (lane == 0) {
      loss_acc += logz - Elem<T>::ld(row + (int)g);
      count_acc += 1.f;
    }
  }
  if (lane == 0 && loss_acc != 0.f) {
    atomicAdd(loss_out + 0, loss_acc);
    atomicAdd(loss_out + 1, count_acc);
  }
}

// ------------------------------------------- transition-pipe fused CE
// The parser/NER loss over ALL transition steps at once (batched by the C++
// step loop): per row, softmax over the VALID actions, target = uniform
// over the min-cost (gold) actions; dScores = p - target (0 for invalid
// columns and for rows with no gold = excluded/missing supervision).
// loss_out[0] += -sum(target * log p); loss_out[1] += #supervised rows.
// Replaces the torch composition (cat + float + masked_fill + log_softmax +
// mul/sum) that was ~5 full fp32 passes over [SS, A].
// colsum_out [A] fp32 (zero-initialized by caller) additionally receives
// sum_n dScores[n, :] — the (unscaled) upper-bias gradient, fused here so
// the backward skips a full [SS, A] column-reduce pass.
// DET=true: colsum accumulates in int64 fixed-point (it feeds dUpperB; the
// loss scalar stays a float atomic — display-only, never differentiated).
template <typename T, bool DET = false>
__global__ void transition_ce_kernel(const T* __restrict__ scores,
                                     const uint8_t* __restrict__ gold,
                                     const uint8_t* __restrict__ valid,
                                     T* __restrict__ dScores,
                                     float* __restrict__ loss_out,
                                     void* __restrict__ colsum_out,
                                     long N, int A) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int ncols = (A + SRX_WAVE - 1) / SRX_WAVE;
  float loss_acc = 0.f;
  float count_acc = 0.f;
  float col_acc[4];  // A <= 256
  for (int c = 0; c < ncols; c++) col_acc[c] = 0.f;
  for (long n = wave; n < N; n += nwaves) {
    const T* row = scores + n * (long)A;
    const uint8_t* grow = gold + n * (long)A;
    const uint8_t* vrow = valid + n * (long)A;
    T* drow = dScores + n * (long)A;
    // load the row + masks ONCE into registers (A <= 256 -> 4 per lane);
    // the 3 logical passes (max, sumexp, dScores) reuse them
    float x[4];
    bool g[4], vv[4];
    float cnt = 0.f;
    float m = -1e38f;
    for (int c = 0; c < ncols; c++) {
      int a = lane + c * SRX_WAVE;
      bool in = a < A;
      g[c] = in && grow[a];
      vv[c] = in && vrow[a];
      x[c] = vv[c] ? Elem<T>::ld(row + a) : -1e38f;
      if (g[c]) cnt += 1.f;
      if (vv[c]) m = fmaxf(m, x[c]);
    }
    cnt = wave_reduce_sum(cnt);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, SRX_WAVE));
    if (cnt == 0.f) {  // no supervision for this row
      for (int a = lane; a < A; a += SRX_WAVE) Elem<T>::st(drow + a, 0.f);
      continue;
    }
    float z = 0.f;
    for (int c = 0; c < ncols; c++)
      if (vv[c]) z += __expf(x[c] - m);
    z = wave_reduce_sum(z);
    float logz = __logf(z) + m;
    float tgt = 1.f / cnt;
    float l = 0.f;
    for (int c = 0; c < ncols; c++) {
      int a = lane + c * SRX_WAVE;
      if (a >= A) continue;
      if (!vv[c]) {
        Elem<T>::st(drow + a, 0.f);
        continue;
      }
      float lp = x[c] - logz;  // log p
      float t = g[c] ? tgt : 0.f;
      float d = __expf(lp) - t;
      Elem<T>::st(drow + a, d);
      col_acc[c] += d;
      if (g[c]) l -= tgt * lp;
    }
    l = wave_reduce_sum(l);
    if (lane == 0) {
      loss_acc += l;
      count_acc += 1.f;
    }
  }
  if (lane == 0 && (loss_acc != 0.f || count_acc != 0.f)) {
    atomicAdd(loss_out + 0, loss_acc);
    atomicAdd(loss_out + 1, count_acc);
  }
  for (int c = 0; c < ncols; c++) {
    int a = lane + c * SRX_WAVE;
    if (a < A && col_acc[c] != 0.f) srx_atomic_add<DET>(colsum_out, a, col_acc[c]);
  }
}

// ------------------------------------------------- segmented reductions
// X [T, W] with doc offsets [N+1] -> out [N, W].  One wave per doc
// (grid-stride); lanes stride the W columns; rows of a doc walked serially
// (docs are tens of tokens — bandwidth-fine for pooling shapes).
template <typename T, int MODE>  // 0 = sum, 1 = mean, 2 = max
__global__ void reduce_ragged_kernel(const T* __restrict__ X,
                                     const int32_t* __restrict__ offsets,
                                     T* __restrict__ out,
                                     int32_t* __restrict__ argmax,  // MODE 2
                                     long N, int W) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long d = wave; d < N; d += nwaves) {
    int32_t s = offsets[d], e = offsets[d + 1];
    for (int w = lane; w < W; w += SRX_WAVE) {
      float acc = MODE == 2 ? -1e38f : 0.f;
      int32_t arg = s;
      for (int32_t t = s; t < e; t++) {
        float v = Elem<T>::ld(X + t * (long)W + w);
        if (MODE == 2) {
          if (v > acc) { acc = v; arg = t; }
        } else {
          acc += v;
        }
      }
      if (MODE == 1 && e > s) acc /= (e - s);
      if (MODE == 2 && e == s) acc = 0.f;
      Elem<T>::st(out + d * (long)W + w, acc);
      if (MODE == 2) argmax[d * (long)W + w] = arg;
    }
  }
}

// backward for sum/mean: dX[t] = dY[doc] (x 1/len for mean)
template <typename T, int MODE>
__global__ void reduce_ragged_bwd_kernel(const T* __restrict__ dY,
                                         const int32_t* __restrict__ doc_of,
                                         const int32_t* __restrict__ offsets,
                                         T* __restrict__ dX, long Ttot, int W) {
  const long total = Ttot * (long)W;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long t = i / W;
    int w = (int)(i % W);
    int32_t d = doc_of[t];
    float v = Elem<T>::ld(dY + d * (long)W + w);
    if (MODE == 1) {
      int32_t len = offsets[d + 1] - offsets[d];
      v /= len > 0 ? len : 1;
    }
    Elem<T>::st(dX + i, v);
  }
}

// backward for max: scatter dY to the argmax positions
template <typename T>
__global__ void reduce_max_bwd_kernel(const T* __restrict__ dY,
                                      const int32_t* __restrict__ argmax,
                                      T* __restrict__ dX, long N, int W) {
  const long total = N * (long)W;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long d = i / W;
    int w = (int)(i % W);
    float v = Elem<T>::ld(dY + i);
    // dX zero-initialized by caller; argmax positions are unique per (d,w)
    Elem<T>::st(dX + argmax[i] * (long)W + w, v);
  }
}
