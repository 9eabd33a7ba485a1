// GPU-resident transition state machines: ONE WAVE PER DOCUMENT runs the
// entire greedy parse/NER in-kernel — stack, arcs, oracle bookkeeping and
// the per-step scorer all live in LDS/registers, so there are ZERO
// per-transition-step host round trips (the r2 C++ host loop still costs
// ~45 ms/step at 1M words in pack + event latency; this kernel replaces
// its per-step H2D/launch/D2H/advance cycle with one launch per batch).
//
// Per step, per wave (all 64 lanes in lockstep on one doc — no divergence):
//   1. features f[13|6] from the LDS state arrays,
//   2. valid flags + (training) the scalar-cost Goldberg-Nivre oracle
//      (same O(1) bookkeeping as transitions.cpp: on-stack bitmap in two
//      u64 registers, gold-children-in-buffer counters in LDS, gold-kids
//      CSR in global memory),
//   3. hidden = maxout2(sum of nF precomputed rows + bias)  (lanes = H),
//   4. scores = upperB + hidden @ upperW^T  (upperW staged row-major
//      [A][GS_HPAD] in LDS once per block; the dot runs as
//      v_dot2c_f32_bf16 over b128 vector reads; lanes = A),
//   5. masked argmax (min-cost mask first, valid fallback) via shuffles,
//   6. state advance (scalar updates mirrored in every lane).
// Training writes per-step rows (scores/gold/valid/feats/hidden/which)
// into per-doc arena slices at fixed capacity (2*len arc-eager, len
// BILUO) so the SAME batched CE + backward as the host loop runs after;
// unused capacity rows stay zeroed (masked out of the loss) with feats
// pre-filled to the pad row (skipped by the dPre scatter).
//
// Scope: doc len <= SRX_GS_MAXLEN and no BREAK action — longer docs and
// use_break parsers fall back to the host loop (pipes.py dispatch).
// Contract mirrored from ops/csrc/transitions.cpp (parity-tested).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>

#include <vector>

#include "srx_common.hip.h"

#define SRX_GS_MAXLEN 128

namespace {

constexpr float GS_KINV = 1e9f;
constexpr int GS_HPAD = 72;  // H(<=64) rounded up: 16B-aligned LDS rows,
                             // zero-padded tail so the dot can run 8-wide

typedef __attribute__((__vector_size__(2 * sizeof(__bf16)))) __bf16 gs_bf16x2;
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int
    gs_u32x4;

// dot(hid[0..64), wrow[0..64)) — both LDS-resident, H<=64 zero-padded.
// The scores phase is the ISSUE-bound inner loop of the state machine
// (~380 of ~700 wave-instructions per transition as a scalar loop): b128
// vector reads (8 LDS ops per 64-row) + v_dot2c_f32_bf16 (1 instruction
// per 2 elements) cut it to ~45.  hid is wave-uniform (same address in
// every lane -> LDS broadcast).
template <typename T>
__device__ __forceinline__ float gs_rowdot64(const T* wrow, const T* hid);

template <>
__device__ __forceinline__ float gs_rowdot64<bf16_t>(const bf16_t* wrow,
                                                     const bf16_t* hid) {
  float acc = 0.f;
#pragma unroll
  for (int h0 = 0; h0 < 64; h0 += 8) {
    gs_u32x4 wf = *(const gs_u32x4*)(wrow + h0);
    gs_u32x4 hf = *(const gs_u32x4*)(hid + h0);
#pragma unroll
    for (int e = 0; e < 4; e++)
      acc = __builtin_amdgcn_fdot2_f32_bf16(
          __builtin_bit_cast(gs_bf16x2, wf[e]),
          __builtin_bit_cast(gs_bf16x2, hf[e]), acc, false);
  }
  return acc;
}

template <>
__device__ __forceinline__ float gs_rowdot64<float>(const float* wrow,
                                                    const float* hid) {
  float acc = 0.f;
#pragma unroll
  for (int h0 = 0; h0 < 64; h0 += 4) {
    float4 wf = *(const float4*)(wrow + h0);
    float4 hf = *(const float4*)(hid + h0);
    acc += wf.x * hf.x + wf.y * hf.y + wf.z * hf.z + wf.w * hf.w;
  }
  return acc;
}

// ---------------------------------------------------------------- parser
// Actions: 0=SHIFT, 1=REDUCE, 2..2+L-1=LEFT-ARC(l), 2+L..2+2L-1=RIGHT-ARC(l)
template <typename T, bool TRAIN>
__global__ __launch_bounds__(256) void gpu_arceager_kernel(
    const T* __restrict__ pre,       // [Tb+1, nF, HP]
    const int32_t* __restrict__ off,     // [n_docs] global token base
    const int32_t* __restrict__ lens,    // [n_docs]
    const int32_t* __restrict__ gh,      // gold heads, doc-local, flat [total]
    const int32_t* __restrict__ gl,      // gold labels, flat [total]
    const int32_t* __restrict__ kids_off,  // CSR [total+1] (global idx)
    const int32_t* __restrict__ kids,      // gold children (doc-local ids)
    const T* __restrict__ lowerB, const T* __restrict__ upperW,
    const T* __restrict__ upperB,
    const int64_t* __restrict__ arena_base,  // [n_docs] arena row base
    T* __restrict__ scores_a, uint8_t* __restrict__ gold_a,
    uint8_t* __restrict__ valid_a, int64_t* __restrict__ feats_a,
    T* __restrict__ hidden_a, uint8_t* __restrict__ which_a,
    int32_t* __restrict__ head_out,   // flat [total] doc-local (-1 root)
    int32_t* __restrict__ label_out,  // flat [total]
    int32_t* __restrict__ steps_out,  // [1] total transitions (atomic)
    long n_docs, long pad_row, int nF, int H, int A, int L) {
  extern __shared__ char smem[];
  const int HP = 2 * H;
  T* Wlds = (T*)smem;                           // [A][GS_HPAD] row-major
  float* Blds = (float*)(Wlds + (size_t)A * GS_HPAD);  // [A]
  // per-wave state block
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const int wslot = threadIdx.x / SRX_WAVE;
  const int waves_per_block = blockDim.x / SRX_WAVE;
  // bias region padded to 16 B so every wave slot stays b128-aligned
  char* wbase = (char*)Blds + (((size_t)A * sizeof(float) + 15) & ~(size_t)15);
  // per-wave slot: 7 byte-arrays of MAXLEN + GS_HPAD hidden (T) + MAXLEN
  // gkb (host allocates the same)
  const size_t per_wave =
      (size_t)SRX_GS_MAXLEN * 8 + GS_HPAD * sizeof(T);
  char* my = wbase + (size_t)wslot * per_wave;
  uint8_t* stk = (uint8_t*)my;                       // [MAXLEN]
  int8_t* head = (int8_t*)(stk + SRX_GS_MAXLEN);     // [MAXLEN] -1 = none
  int8_t* labl = (int8_t*)(head + SRX_GS_MAXLEN);    // [MAXLEN]
  int8_t* l1 = (int8_t*)(labl + SRX_GS_MAXLEN);
  int8_t* l2 = l1 + SRX_GS_MAXLEN;
  int8_t* r1 = l2 + SRX_GS_MAXLEN;
  int8_t* r2 = r1 + SRX_GS_MAXLEN;
  T* my_hid = (T*)(my + (size_t)SRX_GS_MAXLEN * 7);  // [GS_HPAD]
  uint8_t* gkb = (uint8_t*)my_hid + GS_HPAD * sizeof(T);  // [MAXLEN]

  // block prologue: stage upper weights row-major (zero-padded to GS_HPAD
  // so the scores dot reads b128 vectors) + bias; zero the hidden pad tail
  for (int idx = threadIdx.x; idx < A * GS_HPAD; idx += blockDim.x) {
    int a = idx / GS_HPAD, h = idx % GS_HPAD;
    Wlds[idx] = h < H ? upperW[(size_t)a * H + h] : (T)0;
  }
  for (int a = threadIdx.x; a < A; a += blockDim.x)
    Blds[a] = Elem<T>::ld(upperB + a);
  for (int h = lane; h < GS_HPAD; h += SRX_WAVE) my_hid[h] = (T)0;
  __syncthreads();

  const long wave_id = ((long)blockIdx.x * waves_per_block) + wslot;
  const long nwaves = (long)gridDim.x * waves_per_block;
  for (long d = wave_id; d < n_docs; d += nwaves) {
    const int n = lens[d];
    const long o = off[d];
    // ---- init state (lanes cooperate)
    for (int i = lane; i < n; i += SRX_WAVE) {
      head[i] = -1; labl[i] = -1;
      l1[i] = -1; l2[i] = -1; r1[i] = -1; r2[i] = -1;
      if (TRAIN) {
        int32_t s = kids_off[o + i], e = kids_off[o + i + 1];
        gkb[i] = (uint8_t)(e - s);
      }
    }
    // wave-coherent LDS: same-wave writes are visible after waitcnt
    int buf = 0, ssize = 0;
    uint64_t on_stack0 = 0, on_stack1 = 0;  // bitmap over <=128 tokens
    const long abase = TRAIN ? arena_base[d] : 0;
    int steps = 0;
    for (int iter = 0; iter < 2 * n + 2; iter++) {
      if (buf >= n && ssize <= 1) break;  // final
      // ---- scalar state views (every lane computes identically)
      int s0 = ssize > 0 ? stk[ssize - 1] : -1;
      int s1 = ssize > 1 ? stk[ssize - 2] : -1;
      int s2v = ssize > 2 ? stk[ssize - 3] : -1;
      bool has_buf = buf < n;
      bool s0_has_head = s0 >= 0 && head[s0] != -1;
      // ---- features (13), doc-local -> batch-global (or pad)
      int f[13];
      f[0] = s0; f[1] = s1; f[2] = s2v;
      f[3] = has_buf ? buf : -1;
      f[4] = buf + 1 < n ? buf + 1 : -1;
      f[5] = buf + 2 < n ? buf + 2 : -1;
      f[6] = s0 >= 0 ? l1[s0] : -1;
      f[7] = s0 >= 0 ? l2[s0] : -1;
      f[8] = s0 >= 0 ? r1[s0] : -1;
      f[9] = s0 >= 0 ? r2[s0] : -1;
      f[10] = s1 >= 0 ? l1[s1] : -1;
      f[11] = s1 >= 0 ? r1[s1] : -1;
      f[12] = s0 >= 0 ? head[s0] : -1;
      long feats_g[13];
#pragma unroll
      for (int q = 0; q < 13; q++)
        feats_g[q] = f[q] >= 0 ? o + f[q] : pad_row;
      // ---- valid flags
      bool v_shift = has_buf;
      bool v_reduce = ssize > 0 && (s0_has_head || !has_buf);
      bool v_la = ssize > 0 && has_buf && !s0_has_head;
      bool v_ra = ssize > 0 && has_buf;
      // ---- oracle (TRAIN): 5 scalars -> min-cost mask
      float c_shift = 0, c_reduce = 0, c_la = 0, c_ra = 0;
      int la_gold = -1, ra_gold = -1;
      if (TRAIN) {
        int b = has_buf ? buf : -1;
        if (b >= 0) {
          int ghb = gh[o + b];
          bool ghb_on_stack =
              ghb >= 0 && ghb < n &&
              ((ghb < 64 ? (on_stack0 >> ghb) : (on_stack1 >> (ghb - 64))) & 1);
          if (ghb_on_stack) c_shift += 1;
          float skb = 0;
          for (int32_t k = kids_off[o + b]; k < kids_off[o + b + 1]; k++) {
            int c = kids[k];
            bool onstk =
                (c < 64 ? (on_stack0 >> c) : (on_stack1 >> (c - 64))) & 1;
            if (onstk && head[c] == -1) skb += 1;
          }
          c_shift += skb;
          if (s0 >= 0) {
            c_reduce = (float)gkb[s0];
            c_la = c_reduce;
            int ghs0 = gh[o + s0];
            if (ghs0 >= 0 && ghs0 > b) c_la += 1;
            if (ghb >= 0 && ghb != s0 && (ghb_on_stack || ghb > b)) c_ra += 1;
            c_ra += skb;
            if (ghs0 == b) la_gold = gl[o + s0];
            if (ghb == s0) ra_gold = gl[o + b];
          }
        } else if (s0 >= 0) {
          c_reduce = (float)gkb[s0];
        }
      }
      float cmin = GS_KINV;
      if (TRAIN) {
        if (v_shift) cmin = fminf(cmin, c_shift);
        if (v_reduce) cmin = fminf(cmin, c_reduce);
        if (v_la) cmin = fminf(cmin, c_la);
        if (v_ra) cmin = fminf(cmin, c_ra);
      }
      // ---- scoring: hidden (lanes = H)
      long arow = abase + steps;
      for (int h = lane; h < H; h += SRX_WAVE) {
        float acc0 = Elem<T>::ld(lowerB + h);
        float acc1 = Elem<T>::ld(lowerB + H + h);
        for (int q = 0; q < 13; q++) {
          const T* row = pre + (feats_g[q] * (long)13 + q) * HP;
          acc0 += Elem<T>::ld(row + h);
          acc1 += Elem<T>::ld(row + H + h);
        }
        bool second = acc1 > acc0;
        float hv = second ? acc1 : acc0;
        Elem<T>::st(my_hid + h, hv);  // T-rounded: matches the arena copy
        if (TRAIN) {
          Elem<T>::st(hidden_a + arow * H + h, hv);
          which_a[arow * H + h] = (uint8_t)second;
        }
      }
      if (TRAIN) {
        for (int q = lane; q < 13; q += SRX_WAVE)
          feats_a[arow * 13 + q] = feats_g[q];
      }
      // ---- scores + masks + argmax (lanes = A)
      const float eps = 1e-6f;
      float bg = -1e38f, bv = -1e38f;
      int ig = INT32_MAX, iv = INT32_MAX;
      for (int a = lane; a < A; a += SRX_WAVE) {
        float acc = Blds[a] + gs_rowdot64<T>(Wlds + (size_t)a * GS_HPAD, my_hid);
        bool valid, gold = false;
        if (a == 0) valid = v_shift;
        else if (a == 1) valid = v_reduce;
        else if (a < 2 + L) valid = v_la;
        else valid = v_ra;
        if (TRAIN && valid) {
          float cost;
          if (a == 0) cost = c_shift;
          else if (a == 1) cost = c_reduce;
          else if (a < 2 + L)
            cost = c_la + ((la_gold >= 0 && (a - 2) != la_gold) ? 1.f : 0.f);
          else
            cost = c_ra + ((ra_gold >= 0 && (a - 2 - L) != ra_gold) ? 1.f : 0.f);
          gold = cost <= cmin + eps;
        }
        if (TRAIN) {
          Elem<T>::st(scores_a + arow * A + a, acc);
          gold_a[arow * A + a] = gold ? 1 : 0;
          valid_a[arow * A + a] = valid ? 1 : 0;
        }
        bool sel = TRAIN ? gold : valid;
        if (sel && (acc > bg || (acc == bg && a < ig))) { bg = acc; ig = a; }
        if (valid && (acc > bv || (acc == bv && a < iv))) { bv = acc; iv = a; }
      }
#pragma unroll
      for (int sh = 32; sh > 0; sh >>= 1) {
        float obg = __shfl_xor(bg, sh, SRX_WAVE);
        int oig = __shfl_xor(ig, sh, SRX_WAVE);
        if (oig != INT32_MAX && (obg > bg || (obg == bg && oig < ig) || ig == INT32_MAX)) {
          bg = obg; ig = oig;
        }
        float obv = __shfl_xor(bv, sh, SRX_WAVE);
        int oiv = __shfl_xor(iv, sh, SRX_WAVE);
        if (oiv != INT32_MAX && (obv > bv || (obv == bv && oiv < iv) || iv == INT32_MAX)) {
          bv = obv; iv = oiv;
        }
      }
      int act = ig != INT32_MAX ? ig : iv;
      steps += 1;
      if (act == INT32_MAX) break;  // no valid action (defensive)
      // ---- advance (scalar updates; lane 0 writes LDS arrays, register
      // state updated in every lane)
      if (act == 0) {  // SHIFT
        if (lane == 0) stk[ssize] = (uint8_t)buf;
        if (buf < 64) on_stack0 |= 1ull << buf; else on_stack1 |= 1ull << (buf - 64);
        if (TRAIN && lane == 0) {
          int h2 = gh[o + buf];
          if (h2 >= 0 && h2 < n && gkb[h2] > 0) gkb[h2] -= 1;
        }
        ssize += 1;
        buf += 1;
      } else if (act == 1) {  // REDUCE
        int t = s0;
        if (t < 64) on_stack0 &= ~(1ull << t); else on_stack1 &= ~(1ull << (t - 64));
        ssize -= 1;
      } else if (act < 2 + L) {  // LEFT-ARC
        int l = act - 2;
        if (lane == 0) {
          head[s0] = (int8_t)buf;
          labl[s0] = (int8_t)l;
          // s0 < buf: update left children of buf
          if (l1[buf] == -1 || s0 < l1[buf]) { l2[buf] = l1[buf]; l1[buf] = (int8_t)s0; }
          else if (l2[buf] == -1 || s0 < l2[buf]) { l2[buf] = (int8_t)s0; }
        }
        if (s0 < 64) on_stack0 &= ~(1ull << s0); else on_stack1 &= ~(1ull << (s0 - 64));
        ssize -= 1;
      } else {  // RIGHT-ARC
        int l = act - 2 - L;
        if (lane == 0) {
          head[buf] = (int8_t)s0;
          labl[buf] = (int8_t)l;
          // buf > s0: update right children of s0
          if (r1[s0] == -1 || buf > r1[s0]) { r2[s0] = r1[s0]; r1[s0] = (int8_t)buf; }
          else if (r2[s0] == -1 || buf > r2[s0]) { r2[s0] = (int8_t)buf; }
          stk[ssize] = (uint8_t)buf;
          if (TRAIN) {
            int h2 = gh[o + buf];
            if (h2 >= 0 && h2 < n && gkb[h2] > 0) gkb[h2] -= 1;
          }
        }
        if (buf < 64) on_stack0 |= 1ull << buf; else on_stack1 |= 1ull << (buf - 64);
        ssize += 1;
        buf += 1;
      }
      // lane0's LDS writes must be visible to all lanes next iteration;
      // within one wave LDS ops are in program order (lockstep), but tell
      // the compiler not to sink them past the readers:
      __builtin_amdgcn_wave_barrier();
    }
    // ---- write results
    for (int i = lane; i < n; i += SRX_WAVE) {
      head_out[o + i] = head[i];
      label_out[o + i] = labl[i];
    }
    if (lane == 0) atomicAdd(steps_out, steps);
    __builtin_amdgcn_wave_barrier();
  }
}

// ------------------------------------------------------------------- NER
// Actions: 0=OUT; type t: 1+4t=B, 2+4t=I, 3+4t=L, 4+4t=U.  Gold codes per
// token (-1 = missing).  One wave per doc; state is (i, open, open_start).
template <typename T, bool TRAIN>
__global__ __launch_bounds__(256) void gpu_biluo_kernel(
    const T* __restrict__ pre, const int32_t* __restrict__ off,
    const int32_t* __restrict__ lens, const int32_t* __restrict__ gold,
    const T* __restrict__ lowerB, const T* __restrict__ upperW,
    const T* __restrict__ upperB, const int64_t* __restrict__ arena_base,
    T* __restrict__ scores_a, uint8_t* __restrict__ gold_a,
    uint8_t* __restrict__ valid_a, int64_t* __restrict__ feats_a,
    T* __restrict__ hidden_a, uint8_t* __restrict__ which_a,
    int32_t* __restrict__ tags_out, int32_t* __restrict__ steps_out,
    long n_docs, long pad_row, int nF, int H, int A, int NT) {
  extern __shared__ char smem[];
  const int HP = 2 * H;
  T* Wlds = (T*)smem;                           // [A][GS_HPAD] row-major
  float* Blds = (float*)(Wlds + (size_t)A * GS_HPAD);
  const int lane = threadIdx.x & (SRX_WAVE - 1);
  const int wslot = threadIdx.x / SRX_WAVE;
  const int waves_per_block = blockDim.x / SRX_WAVE;
  T* hid_lds =
      (T*)((char*)Blds + (((size_t)A * sizeof(float) + 15) & ~(size_t)15));
  T* my_hid = hid_lds + (size_t)wslot * GS_HPAD;
  for (int idx = threadIdx.x; idx < A * GS_HPAD; idx += blockDim.x) {
    int a = idx / GS_HPAD, h = idx % GS_HPAD;
    Wlds[idx] = h < H ? upperW[(size_t)a * H + h] : (T)0;
  }
  for (int a = threadIdx.x; a < A; a += blockDim.x)
    Blds[a] = Elem<T>::ld(upperB + a);
  for (int h = lane; h < GS_HPAD; h += SRX_WAVE) my_hid[h] = (T)0;
  __syncthreads();

  const long wave_id = ((long)blockIdx.x * waves_per_block) + wslot;
  const long nwaves = (long)gridDim.x * waves_per_block;
  for (long d = wave_id; d < n_docs; d += nwaves) {
    const int n = lens[d];
    const long o = off[d];
    int open = -1, open_start = -1;
    const long abase = TRAIN ? arena_base[d] : 0;
    for (int i = 0; i < n; i++) {
      bool last_tok = i == n - 1;
      int f[6] = {i - 2, i - 1, i, i + 1, i + 2, open_start};
      long feats_g[6];
#pragma unroll
      for (int q = 0; q < 6; q++)
        feats_g[q] = (f[q] >= 0 && f[q] < n) ? o + f[q] : pad_row;
      long arow = abase + i;
      for (int h = lane; h < H; h += SRX_WAVE) {
        float acc0 = Elem<T>::ld(lowerB + h);
        float acc1 = Elem<T>::ld(lowerB + H + h);
        for (int q = 0; q < 6; q++) {
          const T* row = pre + (feats_g[q] * (long)6 + q) * HP;
          acc0 += Elem<T>::ld(row + h);
          acc1 += Elem<T>::ld(row + H + h);
        }
        bool second = acc1 > acc0;
        float hv = second ? acc1 : acc0;
        Elem<T>::st(my_hid + h, hv);  // T-rounded: matches the arena copy
        if (TRAIN) {
          Elem<T>::st(hidden_a + arow * H + h, hv);
          which_a[arow * H + h] = (uint8_t)second;
        }
      }
      if (TRAIN) {
        for (int q = lane; q < 6; q += SRX_WAVE)
          feats_a[arow * 6 + q] = feats_g[q];
      }
      int gcode = TRAIN ? gold[o + i] : -2;
      float bg = -1e38f, bv = -1e38f;
      int ig = INT32_MAX, iv = INT32_MAX;
      for (int a = lane; a < A; a += SRX_WAVE) {
        float acc = Blds[a] + gs_rowdot64<T>(Wlds + (size_t)a * GS_HPAD, my_hid);
        bool valid;
        if (open < 0) {
          if (a == 0) valid = true;                       // OUT
          else {
            int kind = (a - 1) % 4;
            valid = (kind == 0) ? !last_tok : (kind == 3);  // B / U
          }
        } else {
          int kind = (a - 1) % 4, t = (a - 1) / 4;
          valid = t == open && ((kind == 1 && !last_tok) || kind == 2);  // I / L
        }
        bool g = false;
        if (TRAIN) {
          if (gcode == -1) g = false;  // missing: no supervision
          else if (gcode >= 0 && gcode < A) {
            // gold action valid? then one-hot; else uniform over valid
            bool gv;
            if (open < 0)
              gv = gcode == 0 ||
                   ((gcode - 1) % 4 == 0 && !last_tok) || ((gcode - 1) % 4 == 3);
            else
              gv = (gcode - 1) / 4 == open &&
                   (((gcode - 1) % 4 == 1 && !last_tok) || (gcode - 1) % 4 == 2);
            g = gv ? (a == gcode) : valid;
          } else {
            g = valid;
          }
        }
        if (TRAIN) {
          Elem<T>::st(scores_a + arow * A + a, acc);
          gold_a[arow * A + a] = g ? 1 : 0;
          valid_a[arow * A + a] = valid ? 1 : 0;
        }
        bool sel = TRAIN ? g : valid;
        if (sel && (acc > bg || (acc == bg && a < ig))) { bg = acc; ig = a; }
        if (valid && (acc > bv || (acc == bv && a < iv))) { bv = acc; iv = a; }
      }
#pragma unroll
      for (int sh = 32; sh > 0; sh >>= 1) {
        float obg = __shfl_xor(bg, sh, SRX_WAVE);
        int oig = __shfl_xor(ig, sh, SRX_WAVE);
        if (oig != INT32_MAX && (obg > bg || (obg == bg && oig < ig) || ig == INT32_MAX)) {
          bg = obg; ig = oig;
        }
        float obv = __shfl_xor(bv, sh, SRX_WAVE);
        int oiv = __shfl_xor(iv, sh, SRX_WAVE);
        if (oiv != INT32_MAX && (obv > bv || (obv == bv && oiv < iv) || iv == INT32_MAX)) {
          bv = obv; iv = oiv;
        }
      }
      int act = ig != INT32_MAX ? ig : iv;
      if (act == INT32_MAX) act = 0;
      if (lane == 0) tags_out[o + i] = act;
      if (act == 0) {
        open = -1; open_start = -1;
      } else {
        int t = (act - 1) / 4, kind = (act - 1) % 4;
        if (kind == 0) { open = t; open_start = i; }
        else if (kind != 1) { open = -1; open_start = -1; }
      }
    }
    if (lane == 0) atomicAdd(steps_out, n);
  }
}

}  // namespace

// ----------------------------------------------------------------- hosts
// Returns {scores, gold, valid, feats, which, hidden, heads, labels, steps}
// (arenas empty for decode).
std::vector<at::Tensor> srx_gpu_arceager(
    at::Tensor pre, at::Tensor off, at::Tensor lens, at::Tensor gh,
    at::Tensor gl, at::Tensor kids_off, at::Tensor kids, at::Tensor lowerB,
    at::Tensor upperW, at::Tensor upperB, int64_t total, int64_t n_labels,
    bool train) {
  TORCH_CHECK(pre.is_cuda() && pre.is_contiguous());
  long n_docs = off.size(0);
  int H = (int)pre.size(-1) / 2;
  int A = 2 + 2 * (int)n_labels;
  long pad_row = pre.size(0) - 1;
  TORCH_CHECK(H <= 64, "gpu state machine supports H <= 64");
  TORCH_CHECK((size_t)A * 72 * pre.element_size() <= 96 * 1024, "upper too large");
  auto opt = pre.options();
  auto optb = opt.dtype(at::kByte);
  auto opti = opt.dtype(at::kInt);
  // arena bases: doc d's rows at 2 * off[d] (capacity 2*len each)
  auto arena_base = (off.to(at::kLong) * 2).contiguous();
  long cap = 2 * total;
  at::Tensor scores_a, gold_a, valid_a, feats_a, hidden_a, which_a;
  if (train) {
    scores_a = at::zeros({cap, (long)A}, opt);
    gold_a = at::zeros({cap, (long)A}, optb);
    valid_a = at::zeros({cap, (long)A}, optb);
    feats_a = at::full({cap, 13L}, pad_row, opt.dtype(at::kLong));
    hidden_a = at::zeros({cap, (long)H}, opt);
    which_a = at::zeros({cap, (long)H}, optb);
  } else {
    scores_a = at::empty({0, (long)A}, opt);
    gold_a = at::empty({0, (long)A}, optb);
    valid_a = at::empty({0, (long)A}, optb);
    feats_a = at::empty({0, 13L}, opt.dtype(at::kLong));
    hidden_a = at::empty({0, (long)H}, opt);
    which_a = at::empty({0, (long)H}, optb);
  }
  auto heads = at::empty({total}, opti);
  auto labels = at::empty({total}, opti);
  auto steps = at::zeros({1}, opti);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int block = 256;
  const int waves_per_block = block / SRX_WAVE;
  // per-wave LDS: 7 arrays of MAXLEN bytes + GS_HPAD hidden (T) + MAXLEN gkb
  size_t per_wave =
      (size_t)SRX_GS_MAXLEN * 8 + 72 /*GS_HPAD*/ * pre.element_size();
  size_t lds = (size_t)A * 72 * pre.element_size() +
               (((size_t)A * sizeof(float) + 15) & ~(size_t)15) +
               waves_per_block * per_wave;
  int grid = (int)std::min<long>((n_docs + waves_per_block - 1) / waves_per_block,
                                 16384);
  grid = std::max(grid, 1);
#define LAUNCH_AE(T_, TR)                                                     \
  hipLaunchKernelGGL((gpu_arceager_kernel<T_, TR>), dim3(grid), dim3(block),  \
                     lds, stream, (const T_*)pre.data_ptr(),                  \
                     off.data_ptr<int32_t>(), lens.data_ptr<int32_t>(),       \
                     gh.numel() ? gh.data_ptr<int32_t>() : nullptr,           \
                     gl.numel() ? gl.data_ptr<int32_t>() : nullptr,           \
                     kids_off.numel() ? kids_off.data_ptr<int32_t>() : nullptr, \
                     kids.numel() ? kids.data_ptr<int32_t>() : nullptr,       \
                     (const T_*)lowerB.data_ptr(), (const T_*)upperW.data_ptr(), \
                     (const T_*)upperB.data_ptr(),                            \
                     arena_base.data_ptr<int64_t>(), (T_*)scores_a.data_ptr(), \
                     gold_a.numel() ? gold_a.data_ptr<uint8_t>() : nullptr,   \
                     valid_a.numel() ? valid_a.data_ptr<uint8_t>() : nullptr, \
                     feats_a.numel() ? feats_a.data_ptr<int64_t>() : nullptr, \
                     (T_*)hidden_a.data_ptr(),                                \
                     which_a.numel() ? which_a.data_ptr<uint8_t>() : nullptr, \
                     heads.data_ptr<int32_t>(), labels.data_ptr<int32_t>(),   \
                     steps.data_ptr<int32_t>(), n_docs, pad_row, 13, H, A,    \
                     (int)n_labels)
  static std::once_flag attr_ae;
  std::call_once(attr_ae, [&]() {
    (void)hipFuncSetAttribute((const void*)gpu_arceager_kernel<bf16_t, true>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
    (void)hipFuncSetAttribute((const void*)gpu_arceager_kernel<bf16_t, false>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
    (void)hipFuncSetAttribute((const void*)gpu_arceager_kernel<float, true>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
    (void)hipFuncSetAttribute((const void*)gpu_arceager_kernel<float, false>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
  });
  if (pre.scalar_type() == at::kBFloat16) {
    if (train) LAUNCH_AE(bf16_t, true); else LAUNCH_AE(bf16_t, false);
  } else {
    if (train) LAUNCH_AE(float, true); else LAUNCH_AE(float, false);
  }
#undef LAUNCH_AE
  return {scores_a, gold_a, valid_a, feats_a, which_a, hidden_a, heads,
          labels, steps};
}

std::vector<at::Tensor> srx_gpu_biluo(
    at::Tensor pre, at::Tensor off, at::Tensor lens, at::Tensor gold,
    at::Tensor lowerB, at::Tensor upperW, at::Tensor upperB,
    int64_t total, int64_t n_types, bool train) {
  TORCH_CHECK(pre.is_cuda() && pre.is_contiguous());
  long n_docs = off.size(0);
  int H = (int)pre.size(-1) / 2;
  int A = 1 + 4 * (int)n_types;
  long pad_row = pre.size(0) - 1;
  TORCH_CHECK(H <= 64, "gpu state machine supports H <= 64");
  TORCH_CHECK((size_t)A * 72 * pre.element_size() <= 96 * 1024, "upper too large");
  auto opt = pre.options();
  auto optb = opt.dtype(at::kByte);
  auto opti = opt.dtype(at::kInt);
  auto arena_base = off.to(at::kLong).contiguous();  // capacity = len
  long cap = total;
  at::Tensor scores_a, gold_a, valid_a, feats_a, hidden_a, which_a;
  if (train) {
    scores_a = at::zeros({cap, (long)A}, opt);
    gold_a = at::zeros({cap, (long)A}, optb);
    valid_a = at::zeros({cap, (long)A}, optb);
    feats_a = at::full({cap, 6L}, pad_row, opt.dtype(at::kLong));
    hidden_a = at::zeros({cap, (long)H}, opt);
    which_a = at::zeros({cap, (long)H}, optb);
  } else {
    scores_a = at::empty({0, (long)A}, opt);
    gold_a = at::empty({0, (long)A}, optb);
    valid_a = at::empty({0, (long)A}, optb);
    feats_a = at::empty({0, 6L}, opt.dtype(at::kLong));
    hidden_a = at::empty({0, (long)H}, opt);
    which_a = at::empty({0, (long)H}, optb);
  }
  auto tags = at::empty({total}, opti);
  auto steps = at::zeros({1}, opti);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int block = 256;
  const int waves_per_block = block / SRX_WAVE;
  size_t lds = (size_t)A * 72 /*GS_HPAD*/ * pre.element_size() +
               (((size_t)A * sizeof(float) + 15) & ~(size_t)15) +
               waves_per_block * 72 * pre.element_size();
  int grid = (int)std::min<long>((n_docs + waves_per_block - 1) / waves_per_block,
                                 16384);
  grid = std::max(grid, 1);
#define LAUNCH_BI(T_, TR)                                                     \
  hipLaunchKernelGGL((gpu_biluo_kernel<T_, TR>), dim3(grid), dim3(block),     \
                     lds, stream, (const T_*)pre.data_ptr(),                  \
                     off.data_ptr<int32_t>(), lens.data_ptr<int32_t>(),       \
                     gold.numel() ? gold.data_ptr<int32_t>() : nullptr,       \
                     (const T_*)lowerB.data_ptr(), (const T_*)upperW.data_ptr(), \
                     (const T_*)upperB.data_ptr(),                            \
                     arena_base.data_ptr<int64_t>(), (T_*)scores_a.data_ptr(), \
                     gold_a.numel() ? gold_a.data_ptr<uint8_t>() : nullptr,   \
                     valid_a.numel() ? valid_a.data_ptr<uint8_t>() : nullptr, \
                     feats_a.numel() ? feats_a.data_ptr<int64_t>() : nullptr, \
                     (T_*)hidden_a.data_ptr(),                                \
                     which_a.numel() ? which_a.data_ptr<uint8_t>() : nullptr, \
                     tags.data_ptr<int32_t>(), steps.data_ptr<int32_t>(),     \
                     n_docs, pad_row, 6, H, A, (int)n_types)
  if (pre.scalar_type() == at::kBFloat16) {
    if (train) LAUNCH_BI(bf16_t, true); else LAUNCH_BI(bf16_t, false);
  } else {
    if (train) LAUNCH_BI(float, true); else LAUNCH_BI(float, false);
  }
#undef LAUNCH_BI
  return {scores_a, gold_a, valid_a, feats_a, which_a, hidden_a, tags, steps};
}
