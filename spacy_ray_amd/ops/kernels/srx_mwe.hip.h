// Fused MaxoutWindowEncoder layer (SURVEY.md §2.5 `mwe_layer`):
//   Y = X + LayerNorm(maxout_{P=3}(seq2col_{win=1}(X) @ Wt^T + bias))
// in ONE kernel launch — hand-written MFMA (v_mfma_f32_32x32x16_bf16) with
// LDS-tiled operands.
//
// Geometry (gfx950), W = 96 or 128:
// * block = W/32 waves (192 / 256 threads), one 64-token M-tile per block;
// * K = 3W decomposes into the 3 window sections: section s multiplies
//   A_s = X[t-1+s] (doc-boundary-zeroed) by B_s = cols [sW,(s+1)W) of the
//   weight — seq2col is never materialized;
// * wave w owns within-piece 32-col tile position w for ALL 3 pieces
//   (global n-tile p*(W/32)+w): the P=3 maxout is an in-register
//   elementwise max over the wave's own 3 accumulators, and the LayerNorm
//   column span of a wave is exactly within-piece cols [32w, 32w+32);
// * A and B are staged PER SECTION ([64][W+8] + [3W][W+8] at a time, not
//   all three sections at once): at W=96 that is 73 KB of LDS instead of
//   101 KB, which fits TWO blocks per CU (6 waves) instead of one — MFMA
//   and staging latency of one block hides under the other (the
//   single-block version measured 1.85 ms at T=1M; occupancy was the
//   bound, not bandwidth: the 166 KB weight re-read per block stays in
//   L2).  +8 bf16 row pad: the natural 48/64-dword stride would
//   multi-way-conflict ds_read_b128; 52/68 dwords pad to 2-way;
// * B section staged TRANSPOSED in LDS [3W][W+8] so a lane's 8 consecutive
//   k-elements are contiguous (ds_read_b128);
// * LayerNorm row stats: 5-step __shfl_xor column reduce per wave -> LDS
//   partials -> combine across waves; epilogue adds the residual re-read
//   from global X (hot in L2).
//
// Saved for the composed backward: maxout output M, argmax, mu, rstd
// (backward = layernorm_bwd + maxout scatter + two GEMMs + seq2col_bwd,
// all existing kernels/hipBLASLt).
#pragma once
#include "srx_common.hip.h"

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short srx_bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float srx_f32x16;

template <int W>
__global__ __launch_bounds__(2 * W) void mwe_layer_fwd_kernel(
    const bf16_t* __restrict__ X,      // [T, W]
    const bf16_t* __restrict__ Wt,     // [3W (out, pieces-major), 3W (in)] row-major
    const bf16_t* __restrict__ bias,   // [3W]
    const bf16_t* __restrict__ g,      // [W]
    const bf16_t* __restrict__ b,      // [W]
    const uint8_t* __restrict__ is_start,
    const uint8_t* __restrict__ is_end,
    const bf16_t* __restrict__ dropmask,  // [T, W] 0 or 1/keep, or nullptr
    bf16_t* __restrict__ Y,            // [T, W]
    bf16_t* __restrict__ Mout,         // [T, W] maxout output (for bwd)
    uint8_t* __restrict__ which,       // [T, W]
    float* __restrict__ mu_out,        // [T]
    float* __restrict__ rstd_out,      // [T]
    long T, float eps) {
  constexpr int WP = W + 8;       // padded LDS row stride (bf16 elements)
  constexpr int NW = W / 32;      // waves per block = within-piece tiles
  extern __shared__ bf16_t lds[];
  bf16_t* ldsA = lds;                          // 64 * WP (current section)
  bf16_t* ldsB = lds + 64 * WP;                // 3W * WP (current section)
  float* ldsP = (float*)(ldsB + 3 * W * WP);   // LN partials [2][NW][64]

  const int tid = threadIdx.x;
  const int nthreads = 64 * NW;
  const int wave = tid / SRX_WAVE;  // 0..NW-1 = within-piece tile position
  const int lane = tid % SRX_WAVE;
  const long t0 = (long)blockIdx.x * 64;

  srx_f32x16 acc[3][2];  // [piece][m-tile]
#pragma unroll
  for (int p = 0; p < 3; p++)
#pragma unroll
    for (int mi = 0; mi < 2; mi++)
#pragma unroll
      for (int rr = 0; rr < 16; rr++) acc[p][mi][rr] = 0.f;

  // ---- K loop: 3 sections x (W/16) MFMA K-steps
  for (int s = 0; s < 3; s++) {
    __syncthreads();
    // stage A_s: A_s[r] = X[t0+r-1+s] (zeroed across doc bounds)
    for (int r = tid; r < 64; r += nthreads) {
      long t = t0 + r;
      long src = t + s - 1;
      bool zero = (s == 0 && (t == 0 || is_start[t])) ||
                  (s == 2 && (t == T - 1 || is_end[t])) || src < 0 || src >= T;
      bf16_t* dst = ldsA + r * WP;
      const bf16_t* srcp = X + src * W;
      for (int c = 0; c < W; c += 8) {
        if (zero) {
          *(srx_bf16x8*)(dst + c) = srx_bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
        } else {
          *(srx_bf16x8*)(dst + c) = *(const srx_bf16x8*)(srcp + c);
        }
      }
    }
    // stage B_s transposed: ldsB[j][k] = Wt[j][s*W + k]
    for (int j = tid; j < 3 * W; j += nthreads) {
      const bf16_t* srcp = Wt + (long)j * (3 * W) + s * W;
      bf16_t* dst = ldsB + j * WP;
      for (int c = 0; c < W; c += 8)
        *(srx_bf16x8*)(dst + c) = *(const srx_bf16x8*)(srcp + c);
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < W; k += 16) {
      // A fragment (32x32x16 bf16): lane holds A[32*mi + (lane&31)]
      // [k + 8*(lane>>5) + e], e=0..7 — one ds_read_b128
      srx_bf16x8 afrag[2];
#pragma unroll
      for (int mi = 0; mi < 2; mi++) {
        const bf16_t* ap =
            ldsA + (32 * mi + (lane & 31)) * WP + k + 8 * (lane >> 5);
        afrag[mi] = *(const srx_bf16x8*)ap;
      }
#pragma unroll
      for (int p = 0; p < 3; p++) {
        const int gcol = 32 * (p * NW + wave) + (lane & 31);
        const bf16_t* bp = ldsB + gcol * WP + k + 8 * (lane >> 5);
        srx_bf16x8 bfrag = *(const srx_bf16x8*)bp;
#pragma unroll
        for (int mi = 0; mi < 2; mi++) {
          acc[p][mi] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              afrag[mi], bfrag, acc[p][mi], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: +bias, maxout over pieces, LN row stats, residual, store
  const int wpcol = 32 * wave + (lane & 31);  // within-piece column
  float bias_p[3];
#pragma unroll
  for (int p = 0; p < 3; p++) bias_p[p] = bf2f(bias[p * W + wpcol]);

  float mx[2][16];
  uint8_t arg[2][16];
  float psum[2][16], psq[2][16];
#pragma unroll
  for (int mi = 0; mi < 2; mi++) {
#pragma unroll
    for (int rr = 0; rr < 16; rr++) {
      float best = acc[0][mi][rr] + bias_p[0];
      uint8_t bp = 0;
#pragma unroll
      for (int p = 1; p < 3; p++) {
        float v = acc[p][mi][rr] + bias_p[p];
        if (v > best) { best = v; bp = (uint8_t)p; }
      }
      mx[mi][rr] = best;
      arg[mi][rr] = bp;
      float s1 = best, s2 = best * best;
#pragma unroll
      for (int off = 1; off < 32; off <<= 1) {
        s1 += __shfl_xor(s1, off, SRX_WAVE);
        s2 += __shfl_xor(s2, off, SRX_WAVE);
      }
      psum[mi][rr] = s1;  // row-partial over this wave's 32 columns
      psq[mi][rr] = s2;
    }
  }
  __syncthreads();
  if ((lane & 31) == 0) {  // lanes 0 and 32 (different row halves)
#pragma unroll
    for (int mi = 0; mi < 2; mi++)
#pragma unroll
      for (int rr = 0; rr < 16; rr++) {
        int r = 32 * mi + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
        ldsP[wave * 64 + r] = psum[mi][rr];
        ldsP[NW * 64 + wave * 64 + r] = psq[mi][rr];
      }
  }
  __syncthreads();
#pragma unroll
  for (int mi = 0; mi < 2; mi++) {
#pragma unroll
    for (int rr = 0; rr < 16; rr++) {
      int r = 32 * mi + (rr & 3) + 8 * (rr >> 2) + 4 * (lane >> 5);
      float s1 = 0.f, s2 = 0.f;
      for (int w2 = 0; w2 < NW; w2++) {
        s1 += ldsP[w2 * 64 + r];
        s2 += ldsP[NW * 64 + w2 * 64 + r];
      }
      float mu = s1 / W;
      float var = s2 / W - mu * mu;
      float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
      long t = t0 + r;
      if (wave == 0 && (lane & 31) == 0) {
        mu_out[t] = mu;
        rstd_out[t] = rstd;
      }
      float v = mx[mi][rr];
      float y = (v - mu) * rstd * bf2f(g[wpcol]) + bf2f(b[wpcol]);
      if (dropmask) y *= bf2f(dropmask[t * W + wpcol]);
      y += bf2f(X[t * W + wpcol]);  // residual (L2-hot re-read)
      Y[t * W + wpcol] = f2bf(y);
      Mout[t * W + wpcol] = f2bf(v);
      which[t * W + wpcol] = arg[mi][rr];
    }
  }
}

// ------------------------------------------- fused MWE backward stage 1
// One kernel for: dL = dY * dropmask; LayerNorm backward over the saved
// maxout output (Mout, mu, rstd); maxout scatter into the pieces-major
// [T, 3W] pre-activation gradient; the 3W bias column-sum; and the LN
// dg/db column sums — replacing 3 kernels + 2 reduce passes per encoder
// block per step.  Structure follows layernorm_bwd_kernel (one wave per
// row, per-wave register accumulation for the column sums, one atomic per
// column per wave; DET => int64 fixed-point).  W <= 256.
template <typename T, bool DET = false>
__global__ void mwe_bwd_stage1_kernel(
    const T* __restrict__ dY, const T* __restrict__ dropmask,  // mask may be null
    const T* __restrict__ Mout, const T* __restrict__ g,
    const float* __restrict__ mu, const float* __restrict__ rstd,
    const uint8_t* __restrict__ which, T* __restrict__ dPre,
    void* __restrict__ dg32, void* __restrict__ db32,
    void* __restrict__ dbias32, long N, int W) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int ncols = (W + SRX_WAVE - 1) / SRX_WAVE;
  float dg_loc[4], db_loc[4], dbias_loc[4 * 3];
  for (int c = 0; c < ncols; c++) { dg_loc[c] = 0.f; db_loc[c] = 0.f; }
  for (int k = 0; k < ncols * 3; k++) dbias_loc[k] = 0.f;
  for (long n = wave; n < N; n += nwaves) {
    const T* dyrow = dY + n * (long)W;
    const T* mrow = Mout + n * (long)W;
    const T* maskrow = dropmask ? dropmask + n * (long)W : nullptr;
    float m = mu[n], r = rstd[n];
    float s1 = 0.f, s2 = 0.f;
    float dl[4], xh[4];
    for (int c = 0; c < ncols; c++) {
      int w = lane + c * SRX_WAVE;
      if (w >= W) { dl[c] = 0.f; xh[c] = 0.f; continue; }
      float d = Elem<T>::ld(dyrow + w);
      if (maskrow) d *= Elem<T>::ld(maskrow + w);
      float xhat = (Elem<T>::ld(mrow + w) - m) * r;
      float dxhat = d * Elem<T>::ld(g + w);
      dl[c] = d;
      xh[c] = xhat;
      s1 += dxhat;
      s2 += dxhat * xhat;
      dg_loc[c] += d * xhat;
      db_loc[c] += d;
    }
    s1 = wave_reduce_sum(s1) / W;
    s2 = wave_reduce_sum(s2) / W;
    T* out = dPre + n * (long)(3 * W);
    const uint8_t* wrow = which + n * (long)W;
    for (int c = 0; c < ncols; c++) {
      int w = lane + c * SRX_WAVE;
      if (w >= W) continue;
      float dxhat = dl[c] * Elem<T>::ld(g + w);
      float dM = r * (dxhat - s1 - xh[c] * s2);
      int p = wrow[w];
      for (int q = 0; q < 3; q++)
        Elem<T>::st(out + q * W + w, q == p ? dM : 0.f);
      dbias_loc[c * 3 + p] += dM;
    }
  }
  for (int c = 0; c < ncols; c++) {
    int w = lane + c * SRX_WAVE;
    if (w >= W) continue;
    if (dg_loc[c] != 0.f) srx_atomic_add<DET>(dg32, w, dg_loc[c]);
    if (db_loc[c] != 0.f) srx_atomic_add<DET>(db32, w, db_loc[c]);
    for (int q = 0; q < 3; q++)
      if (dbias_loc[c * 3 + q] != 0.f)
        srx_atomic_add<DET>(dbias32, (long)q * W + w, dbias_loc[c * 3 + q]);
  }
}
