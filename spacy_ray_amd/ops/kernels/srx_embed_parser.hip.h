// HashEmbed (fused murmur + 4-row gather/scatter) and the parser state
// scorer (fused nF-row gather + sum + bias + maxout P=2).
// Semantics: ops/torch_ref.py (hashembed_*, parser_step_score).
#pragma once
#include <hip/hip_bf16.h>

#include "srx_common.hip.h"

// ----------------------------------------------------------- hashembed
// One wave per token: the wave's lanes cover the W embedding columns; the
// 4 murmur row ids are computed from the wave-uniform id (scalar-unit work)
// and the 4 rows are gathered and summed in fp32 (SURVEY.md §2.5
// hashembed_fwd: out[i] = sum_s E[h_s(id_i) % rows]).
// ldY: row stride of Y — lets the 4 attr tables write straight into their
// column block of the concatenated [T, 4W] embed matrix (the separate
// torch.cat copied 0.77 GB per step at 1M words).
template <typename T>
__global__ void hashembed_fwd_kernel(const T* __restrict__ table,
                                     const uint64_t* __restrict__ ids,
                                     T* __restrict__ Y,
                                     int32_t* __restrict__ rows_out,
                                     long nT, int nrows, int W, long ldY,
                                     uint32_t seed) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long t = wave; t < nT; t += nwaves) {
    uint32_t h[4];
    murmur3_hash4_u64_dev(ids[t], seed, h);
    int32_t r0 = (int32_t)(h[0] % (uint32_t)nrows);
    int32_t r1 = (int32_t)(h[1] % (uint32_t)nrows);
    int32_t r2 = (int32_t)(h[2] % (uint32_t)nrows);
    int32_t r3 = (int32_t)(h[3] % (uint32_t)nrows);
    if (lane < 4) rows_out[t * 4 + lane] = lane == 0 ? r0 : lane == 1 ? r1 : lane == 2 ? r2 : r3;
    const T* t0 = table + (long)r0 * W;
    const T* t1 = table + (long)r1 * W;
    const T* t2 = table + (long)r2 * W;
    const T* t3 = table + (long)r3 * W;
    T* out = Y + t * ldY;
    for (int w = lane; w < W; w += SRX_WAVE) {
      float acc = Elem<T>::ld(t0 + w) + Elem<T>::ld(t1 + w) +
                  Elem<T>::ld(t2 + w) + Elem<T>::ld(t3 + w);
      Elem<T>::st(out + w, acc);
    }
  }
}

// Backward: scatter-add dY into an fp32 table-grad workspace (atomics; the
// Zipf-hot rows are the contention worry — SURVEY.md §7 hard-part 2; fp32
// atomics across W addresses and many L2 channels are acceptable round 1,
// sort-by-row segmented reduction is the planned upgrade).
template <typename T>
__global__ void hashembed_bwd_kernel(const T* __restrict__ dY,
                                     const int32_t* __restrict__ rows,
                                     float* __restrict__ dT32,
                                     long nT, int W) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long t = wave; t < nT; t += nwaves) {
    int32_t r0 = rows[t * 4 + 0], r1 = rows[t * 4 + 1];
    int32_t r2 = rows[t * 4 + 2], r3 = rows[t * 4 + 3];
    const T* dy = dY + t * (long)W;
    for (int w = lane; w < W; w += SRX_WAVE) {
      float v = Elem<T>::ld(dy + w);
      atomicAdd(dT32 + (long)r0 * W + w, v);
      atomicAdd(dT32 + (long)r1 * W + w, v);
      atomicAdd(dT32 + (long)r2 * W + w, v);
      atomicAdd(dT32 + (long)r3 * W + w, v);
    }
  }
}

// ------------------------------------------------- parser step scorer
// One wave per state (SURVEY.md §2.5 parser_step_score): gather nF
// precomputed rows (pieces-major [T+1, nF, 2*H]), sum + bias, maxout P=2.
// H=64 fits one lane per hidden unit exactly.
template <typename T>
__global__ void parser_step_fwd_kernel(const T* __restrict__ pre,
                                       const int64_t* __restrict__ feats,
                                       const T* __restrict__ bias,
                                       T* __restrict__ hidden,
                                       uint8_t* __restrict__ which,
                                       long S, int nF, int H) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int HP = 2 * H;
  for (long s = wave; s < S; s += nwaves) {
    const int64_t* fs = feats + s * nF;
    for (int h = lane; h < H; h += SRX_WAVE) {
      float acc0 = Elem<T>::ld(bias + h);
      float acc1 = Elem<T>::ld(bias + H + h);
      for (int f = 0; f < nF; f++) {
        const T* row = pre + (fs[f] * (long)nF + f) * HP;
        acc0 += Elem<T>::ld(row + h);
        acc1 += Elem<T>::ld(row + H + h);
      }
      bool second = acc1 > acc0;
      Elem<T>::st(hidden + s * (long)H + h, second ? acc1 : acc0);
      which[s * (long)H + h] = (uint8_t)second;
    }
  }
}

// ------------------------------------- sorted segmented scatter-add
// OUT[dst_sorted[e]] += SRC[src_idx[e]] for e in [0, M), where dst_sorted is
// SORTED: each wave owns a chunk of entries, accumulates equal-dst runs in
// registers and pushes ONE atomic per (run x chunk) — Zipf-hot rows get
// M/chunk pushes instead of M (the sort-by-row segmented reduction planned
// for HashEmbed backward, SURVEY.md §7 hard-part 2; also used for the
// batched parser dPre scatter).
// DET=true: OUT is an int64 fixed-point buffer (srx_atomic_add) — the
// per-chunk register accumulation is already order-fixed (sorted entries),
// and the cross-chunk pushes become associative integer adds, so the
// result is bit-identical across runs (SRX_DETERMINISTIC).
// ldSRC: row stride of SRC (lets callers pass column-block VIEWS of a
// wider matrix, e.g. one attr's slice of the concatenated embed grad).
template <typename T, int CHUNK, bool DET = false>
__global__ void seg_scatter_add_kernel(const int32_t* __restrict__ dst_sorted,
                                       const int32_t* __restrict__ src_idx,
                                       const T* __restrict__ SRC,
                                       void* __restrict__ OUT,
                                       long M, int W, long ldSRC) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int ncols = (W + SRX_WAVE - 1) / SRX_WAVE;
  for (long c0 = wave * CHUNK; c0 < M; c0 += nwaves * CHUNK) {
    long e_end = min(c0 + (long)CHUNK, M);
    int32_t cur = dst_sorted[c0];
    float acc[16];  // ncols <= 16 supported (W <= 1024)
    for (int c = 0; c < ncols; c++) acc[c] = 0.f;
    for (long e = c0; e < e_end; e++) {
      int32_t d = dst_sorted[e];
      if (d != cur) {
        for (int c = 0; c < ncols; c++) {
          int w = lane + c * SRX_WAVE;
          if (w < W && acc[c] != 0.f)
            srx_atomic_add<DET>(OUT, (long)cur * W + w, acc[c]);
          acc[c] = 0.f;
        }
        cur = d;
      }
      const T* src = SRC + (long)src_idx[e] * ldSRC;
      for (int c = 0; c < ncols; c++) {
        int w = lane + c * SRX_WAVE;
        if (w < W) acc[c] += Elem<T>::ld(src + w);
      }
    }
    for (int c = 0; c < ncols; c++) {
      int w = lane + c * SRX_WAVE;
      if (w < W && acc[c] != 0.f)
        srx_atomic_add<DET>(OUT, (long)cur * W + w, acc[c]);
    }
  }
}

// ----------------------------------------------------- action selection
// One wave per state: actions[s] = argmax of scores over the is_gold
// columns if any, else over the valid columns (first-occurrence tie-break,
// matching numpy argmax).  Replaces a [S,A] score D2H + host argmax per
// transition step with a [S] int32 D2H.
template <typename T>
__global__ void action_select_kernel(const T* __restrict__ scores,
                                     const uint8_t* __restrict__ is_gold,
                                     const uint8_t* __restrict__ valid,
                                     int32_t* __restrict__ actions,
                                     long S, int A) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  for (long s = wave; s < S; s += nwaves) {
    float bg = -1e38f, bv = -1e38f;
    int ig = INT32_MAX, iv = INT32_MAX;
    const T* row = scores + s * (long)A;
    const uint8_t* grow = is_gold + s * (long)A;
    const uint8_t* vrow = valid + s * (long)A;
    for (int a = lane; a < A; a += SRX_WAVE) {
      float v = Elem<T>::ld(row + a);
      if (grow[a] && (v > bg || (v == bg && a < ig))) { bg = v; ig = a; }
      if (vrow[a] && (v > bv || (v == bv && a < iv))) { bv = v; iv = a; }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float obg = __shfl_xor(bg, off, SRX_WAVE);
      int oig = __shfl_xor(ig, off, SRX_WAVE);
      if (oig != INT32_MAX && (obg > bg || (obg == bg && oig < ig) || ig == INT32_MAX)) {
        bg = obg; ig = oig;
      }
      float obv = __shfl_xor(bv, off, SRX_WAVE);
      int oiv = __shfl_xor(iv, off, SRX_WAVE);
      if (oiv != INT32_MAX && (obv > bv || (obv == bv && oiv < iv) || iv == INT32_MAX)) {
        bv = obv; iv = oiv;
      }
    }
    if (lane == 0) {
      actions[s] = ig != INT32_MAX ? ig : (iv != INT32_MAX ? iv : -1);
    }
  }
}

// ---------------------------------------- batched dPre scatter (direct)
// dPre32[feats[s,f], f, :] += dSummed[s, :] for all (s, f) with
// feats[s,f] != pad_row.  Token-position destinations are near-uniform
// (each (t,f) slot receives ~SS/T ≈ 2 contributions), so plain fp32
// atomics don't serialize — ONLY the pad row (missing features, 30-50% of
// slots in early steps) is Zipf-hot, and the caller computes its gradient
// separately as one mask^T @ dSummed GEMM.  Replaces the sort + segmented
// reduction over SS*nF entries (argsort of ~13M int64 + 36 ms worst-case
// scatter in the r1 profile — VERDICT r1 item 3).
// One wave per state: the dSummed row is loaded once into registers, then
// scattered to the nF destinations.
// dBias32 [HP] fp32 (zero-initialized) additionally receives
// sum_s dSummed[s, :] — the lower-bias gradient, register-accumulated per
// wave (one atomic per column per wave) so the backward skips a full
// [SS, HP] column-reduce pass.  The PAD-row sums land in dPad32 [nF, HP]
// the same way (register per (f, c), one fp32 atomic per wave) — the pad
// row is Zipf-hot, so it stays out of the per-state atomics AND out of the
// lower-precision dPre accumulation; the caller writes dPad into dPre's
// pad row afterwards.
//
// MODE 0: dPre is fp32, plain atomicAdd.
// MODE 1: dPre is bf16 — gfx950 packed global_atomic_pk_add_bf16
//   (unsafeAtomicAdd on __hip_bfloat162), one pair per lane.  Halves the
//   scatter traffic AND removes the separate fp32->bf16 convert of the
//   whole [T+1, nF, HP] buffer (6.6 GB read + 3.3 GB write per pipe per
//   step).  Accumulation rounds per add, acceptable at ~2 contributions
//   per (token, slot) destination (pad excluded).
// MODE 2: deterministic — dPre/dBias/dPad are int64 fixed-point
//   (srx_atomic_add<true>); bit-identical across runs.
template <typename T, int MODE>
__global__ void dpre_scatter_kernel(const T* __restrict__ dSummed,
                                    const int64_t* __restrict__ feats,
                                    void* __restrict__ dPre,
                                    void* __restrict__ dBias,
                                    void* __restrict__ dPad,
                                    long S, int nF, int HP, long pad_row) {
  constexpr bool DET = MODE == 2;
  constexpr bool OutBF16 = MODE == 1;
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  // fp32: lane covers columns lane + c*64.  bf16: lane covers the PAIR
  // (2*lane + c*128, 2*lane+1 + c*128).
  const int ncols = OutBF16 ? (HP / 2 + SRX_WAVE - 1) / SRX_WAVE
                            : (HP + SRX_WAVE - 1) / SRX_WAVE;
  const int span = OutBF16 ? 2 : 1;
  float bias_acc[8];   // HP <= 256 (bf16 pairs: 2 floats per c)
  float pad_acc[16 * 8];  // nF <= 16
  for (int c = 0; c < ncols * span; c++) bias_acc[c] = 0.f;
  for (int k = 0; k < nF * ncols * span; k++) pad_acc[k] = 0.f;
  for (long s = wave; s < S; s += nwaves) {
    float v[8];
    const T* src = dSummed + s * (long)HP;
    for (int c = 0; c < ncols; c++) {
      for (int e = 0; e < span; e++) {
        int w = span * lane + e + c * span * SRX_WAVE;
        float x = w < HP ? Elem<T>::ld(src + w) : 0.f;
        v[c * span + e] = x;
        bias_acc[c * span + e] += x;
      }
    }
    const int64_t* fs = feats + s * nF;
    for (int f = 0; f < nF; f++) {
      int64_t t = fs[f];
      if (t == pad_row) {
        for (int c = 0; c < ncols * span; c++) pad_acc[f * ncols * span + c] += v[c];
        continue;
      }
      if (OutBF16) {
#if defined(__gfx950__) || defined(__gfx942__) || defined(__gfx90a__)
        __hip_bfloat162* dst =
            (__hip_bfloat162*)dPre + (t * (long)nF + f) * (HP / 2);
        for (int c = 0; c < ncols; c++) {
          int p = lane + c * SRX_WAVE;
          // maxout_bwd zeroes the non-argmax piece: HALF of dSummed is
          // exact zeros, so ~25% of pairs are (0,0) — skipping them cuts
          // the atomic OP count (the measured bound: ~0.34 ms per slot at
          // 1.9M rows is packed-atomic rate, not bandwidth)
          if (p < HP / 2 && (v[c * 2] != 0.f || v[c * 2 + 1] != 0.f)) {
            __hip_bfloat162 val(__float2bfloat16(v[c * 2]),
                                __float2bfloat16(v[c * 2 + 1]));
            unsafeAtomicAdd(dst + p, val);
          }
        }
#endif
      } else {
        long base = (t * (long)nF + f) * HP;
        for (int c = 0; c < ncols; c++) {
          int w = lane + c * SRX_WAVE;
          // half of dSummed is exact zeros (maxout backward) — skip
          if (w < HP && v[c] != 0.f) srx_atomic_add<DET>(dPre, base + w, v[c]);
        }
      }
    }
  }
  for (int c = 0; c < ncols * span; c++) {
    int w = span * lane + (c % span) + (c / span) * span * SRX_WAVE;
    if (w < HP && bias_acc[c] != 0.f)
      srx_atomic_add<DET>(dBias, w, bias_acc[c]);
  }
  for (int f = 0; f < nF; f++) {
    for (int c = 0; c < ncols * span; c++) {
      int w = span * lane + (c % span) + (c / span) * span * SRX_WAVE;
      float pv = pad_acc[f * ncols * span + c];
      if (w < HP && pv != 0.f)
        srx_atomic_add<DET>(dPad, (long)f * HP + w, pv);
    }
  }
}

// Backward: scatter dHidden into fp32 workspaces for dPre and dBias.
template <typename T>
__global__ void parser_step_bwd_kernel(const T* __restrict__ dHidden,
                                       const int64_t* __restrict__ feats,
                                       const uint8_t* __restrict__ which,
                                       float* __restrict__ dPre32,
                                       float* __restrict__ dBias32,
                                       long S, int nF, int H) {
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const long wave = ((long)blockIdx.x * blockDim.x + threadIdx.x) / SRX_WAVE;
  const long nwaves = ((long)gridDim.x * blockDim.x) / SRX_WAVE;
  const int HP = 2 * H;
  for (long s = wave; s < S; s += nwaves) {
    const int64_t* fs = feats + s * nF;
    for (int h = lane; h < H; h += SRX_WAVE) {
      float d = Elem<T>::ld(dHidden + s * (long)H + h);
      int slot = which[s * (long)H + h] ? H + h : h;
      atomicAdd(dBias32 + slot, d);
      for (int f = 0; f < nF; f++) {
        atomicAdd(dPre32 + (fs[f] * (long)nF + f) * HP + slot, d);
      }
    }
  }
}

// ------------------------------- batched dPre scatter (doc-major arenas)
// ATOMIC-FREE variant of dpre_scatter_kernel for the GPU-state-machine
// arenas (srx_gpustate.hip), where doc d's transition rows sit at
// [cap_mult*off[d], cap_mult*(off[d]+len[d])) and every feature of those
// rows is either the pad row or a token OF DOC d.  (t, f) destinations are
// therefore exclusive to one (doc, f) stream: one BLOCK per doc
// accumulates slot f in an LDS fp32 tile [len, HP] (ds-atomic adds across
// the block's waves) and writes it back with plain coalesced stores —
// no global atomics.  The direct-atomic kernel measured ~3.7 ms/call at
// 1.9M rows (packed-bf16 atomic OP RATE bound, ~1.6G pk-atomics); this
// formulation is plain-store bandwidth bound (~3.3 GB writes).
// dPre may be UNINITIALIZED (at::empty): every (t<T, f) row is written;
// the caller still fills the pad row from dPad afterwards.
// dBias/dPad side sums: register-accumulated, one atomic per column per
// block (same contract as dpre_scatter_kernel).  fp32 LDS accumulation =
// single rounding into T — slightly more accurate than packed-bf16
// atomics; LDS-atomic ordering is not fixed, so SRX_DETERMINISTIC keeps
// the int64 fixed-point path of the direct kernel.
template <typename T>
__global__ __launch_bounds__(256) void dpre_docmajor_kernel(
    const T* __restrict__ dSummed,      // [cap_total, HP]
    const int64_t* __restrict__ feats,  // [cap_total, nF]
    T* __restrict__ dPre,               // [Tb+1, nF, HP]
    float* __restrict__ dBias,          // [HP]
    float* __restrict__ dPad,           // [nF, HP]
    const int32_t* __restrict__ off, const int32_t* __restrict__ lens,
    long n_docs, long pad_row, int nF, int HP, int cap_mult) {
  // 2D grid: block (d, q) owns slot q of doc d — enough independent blocks
  // (n_docs * nF) that each block's short serial row chain hides under
  // others (the doc-per-block, q-serial version was latency-bound at
  // 12.8 ms vs the atomic kernel's 5.3; this shape measures by block
  // parallelism instead)
  extern __shared__ float acc[];  // [maxlen][HP]
  const int tid = threadIdx.x;
  const int lane = tid & (SRX_WAVE - 1);
  const int wslot = tid / SRX_WAVE;
  const int nw = blockDim.x / SRX_WAVE;
  const int ncols = HP / SRX_WAVE;  // HP % 64 == 0, HP <= 128
  const int q = blockIdx.y;
  float bias_acc[2] = {0.f, 0.f};
  float pad_acc[2] = {0.f, 0.f};
  for (long d = blockIdx.x; d < n_docs; d += gridDim.x) {
    const int n = lens[d];
    const long o = off[d];
    const long rbase = (long)cap_mult * o;
    const int rows = cap_mult * n;
    for (int i = tid; i < n * HP; i += blockDim.x) acc[i] = 0.f;
    __syncthreads();
    for (int r = wslot; r < rows; r += nw) {
      const long s = rbase + r;
      const long t = feats[s * (long)nF + q];
      const T* src = dSummed + s * (long)HP;
      float x[2];
      for (int c = 0; c < ncols; c++)
        x[c] = Elem<T>::ld(src + lane + c * SRX_WAVE);
      if (q == 0)  // bias = rowsum over ALL rows; q==0 blocks count once
        for (int c = 0; c < ncols; c++) bias_acc[c] += x[c];
      if (t == pad_row) {
        for (int c = 0; c < ncols; c++) pad_acc[c] += x[c];
      } else {
        float* a = acc + (size_t)(t - o) * HP;
        for (int c = 0; c < ncols; c++)
          atomicAdd(a + lane + c * SRX_WAVE, x[c]);
      }
    }
    __syncthreads();
    for (int i = tid; i < n * HP; i += blockDim.x) {
      int local = i / HP, w = i - local * HP;
      Elem<T>::st(dPre + ((o + local) * (long)nF + q) * HP + w, acc[i]);
    }
    __syncthreads();
  }
  for (int c = 0; c < ncols; c++) {
    if (q == 0 && bias_acc[c] != 0.f)
      atomicAdd(dBias + lane + c * SRX_WAVE, bias_acc[c]);
    if (pad_acc[c] != 0.f)
      atomicAdd(dPad + (long)q * HP + lane + c * SRX_WAVE, pad_acc[c]);
  }
}
