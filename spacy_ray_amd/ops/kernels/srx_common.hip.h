// Common device helpers for the gfx950 kernels.
// Written CDNA4-first: wave64 everywhere (no 32-wide warp idioms), bf16
// stored as raw ushort bits (torch bf16 layout), fp32 accumulation.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define SRX_WAVE 64

typedef unsigned short bf16_t;

__device__ __forceinline__ float bf2f(bf16_t b) {
  uint32_t u = ((uint32_t)b) << 16;
  return __uint_as_float(u);
}

__device__ __forceinline__ bf16_t f2bf(float f) {
  uint32_t u = __float_as_uint(f);
  uint32_t r = 0x7fffu + ((u >> 16) & 1u);  // round-to-nearest-even
  return (bf16_t)((u + r) >> 16);
}

template <typename T>
struct Elem;

// paired loads/stores (2 contiguous elements in one transaction — bf16
// pairs move as one dword instead of two 2-byte accesses)
template <typename T>
struct Elem2;

template <>
struct Elem<float> {
  static __device__ __forceinline__ float ld(const float* p) { return *p; }
  static __device__ __forceinline__ void st(float* p, float v) { *p = v; }
};

template <>
struct Elem<bf16_t> {
  static __device__ __forceinline__ float ld(const bf16_t* p) { return bf2f(*p); }
  static __device__ __forceinline__ void st(bf16_t* p, float v) { *p = f2bf(v); }
};

template <>
struct Elem2<float> {
  static __device__ __forceinline__ void ld(const float* p, float* v) {
    float2 x = *(const float2*)p;
    v[0] = x.x; v[1] = x.y;
  }
  static __device__ __forceinline__ void st(float* p, const float* v) {
    *(float2*)p = make_float2(v[0], v[1]);
  }
};

template <>
struct Elem2<bf16_t> {
  static __device__ __forceinline__ void ld(const bf16_t* p, float* v) {
    uint32_t u = *(const uint32_t*)p;
    v[0] = bf2f((bf16_t)(u & 0xffffu));
    v[1] = bf2f((bf16_t)(u >> 16));
  }
  static __device__ __forceinline__ void st(bf16_t* p, const float* v) {
    uint32_t u = (uint32_t)f2bf(v[0]) | ((uint32_t)f2bf(v[1]) << 16);
    *(uint32_t*)p = u;
  }
};

// ---- true vector load/store of V elements as ONE 16-byte instruction.
// The per-element Elem<T> loops compile to scalar global_load_ushort
// streams (hipcc does not merge them): seq2col_bwd measured 56 memory
// instructions per 8-element item instead of ~6, 8x off its bandwidth
// floor.  Pointers must be 16B-aligned (W and row strides are multiples
// of 8 elements everywhere these are used).
typedef __attribute__((__vector_size__(16))) uint32_t srx_u32x4_t;

template <typename T, int V>
struct ElemV;

template <>
struct ElemV<bf16_t, 8> {
  static __device__ __forceinline__ void ld(const bf16_t* p, float* v) {
    srx_u32x4_t u = *(const srx_u32x4_t*)p;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      v[2 * i] = bf2f((bf16_t)(u[i] & 0xffffu));
      v[2 * i + 1] = bf2f((bf16_t)(u[i] >> 16));
    }
  }
  static __device__ __forceinline__ void st(bf16_t* p, const float* v) {
    srx_u32x4_t u;
#pragma unroll
    for (int i = 0; i < 4; i++)
      u[i] = (uint32_t)f2bf(v[2 * i]) | ((uint32_t)f2bf(v[2 * i + 1]) << 16);
    *(srx_u32x4_t*)p = u;
  }
};

template <>
struct ElemV<float, 4> {
  static __device__ __forceinline__ void ld(const float* p, float* v) {
    float4 x = *(const float4*)p;
    v[0] = x.x; v[1] = x.y; v[2] = x.z; v[3] = x.w;
  }
  static __device__ __forceinline__ void st(float* p, const float* v) {
    *(float4*)p = make_float4(v[0], v[1], v[2], v[3]);
  }
};

// scalar fallbacks so the V=1 tail instantiations compile
template <>
struct ElemV<bf16_t, 1> {
  static __device__ __forceinline__ void ld(const bf16_t* p, float* v) {
    v[0] = bf2f(*p);
  }
  static __device__ __forceinline__ void st(bf16_t* p, const float* v) {
    *p = f2bf(v[0]);
  }
};

template <>
struct ElemV<float, 1> {
  static __device__ __forceinline__ void ld(const float* p, float* v) {
    v[0] = *p;
  }
  static __device__ __forceinline__ void st(float* p, const float* v) {
    *p = v[0];
  }
};

// Wave-wide (64-lane) sum reduction; result valid in all lanes.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, SRX_WAVE);
  return v;
}

// ---- deterministic accumulation (SRX_DETERMINISTIC / SURVEY §5.2)
// Float atomics make gradient sums depend on scheduling order (fp add is
// not associative).  The deterministic variants accumulate in FIXED-POINT
// int64 (value * 2^24, llrintf-rounded): integer atomic adds ARE
// associative, so the result is bit-identical across runs regardless of
// order.  Range: |sum| < 2^39 ~ 5.5e11 at 2^-24 ~ 6e-8 resolution — ample
// for gradient magnitudes.  The caller converts back with a single
// (deterministic) elementwise divide.
#define SRX_FIXED_SCALE 16777216.0f  // 2^24

template <bool DET>
__device__ __forceinline__ void srx_atomic_add(void* buf, long idx, float v) {
  if (DET) {
    long long q = (long long)llrintf(v * SRX_FIXED_SCALE);
    atomicAdd((unsigned long long*)buf + idx, (unsigned long long)q);
  } else {
    atomicAdd((float*)buf + idx, v);
  }
}

// ---- MurmurHash3 x64_128 of one 8-byte key: device twin of
// ops/csrc/murmur3.h::murmur3_hash4_u64 — MUST stay bit-identical (the
// CPU/GPU HashEmbed row-assignment contract).
__device__ __forceinline__ uint64_t srx_rotl64(uint64_t x, int r) {
  return (x << r) | (x >> (64 - r));
}

__device__ __forceinline__ uint64_t srx_fmix64(uint64_t k) {
  k ^= k >> 33;
  k *= 0xff51afd7ed558ccdULL;
  k ^= k >> 33;
  k *= 0xc4ceb9fe1a85ec53ULL;
  k ^= k >> 33;
  return k;
}

__device__ __forceinline__ void murmur3_hash4_u64_dev(uint64_t key, uint32_t seed,
                                                      uint32_t out[4]) {
  const uint64_t c1 = 0x87c37b91114253d5ULL;
  const uint64_t c2 = 0x4cf5ad432745937fULL;
  uint64_t h1 = seed, h2 = seed;
  uint64_t k1 = key;
  k1 *= c1;
  k1 = srx_rotl64(k1, 31);
  k1 *= c2;
  h1 ^= k1;
  h1 ^= 8ULL;
  h2 ^= 8ULL;
  h1 += h2;
  h2 += h1;
  h1 = srx_fmix64(h1);
  h2 = srx_fmix64(h2);
  h1 += h2;
  h2 += h1;
  out[0] = (uint32_t)(h1 & 0xffffffffULL);
  out[1] = (uint32_t)(h1 >> 32);
  out[2] = (uint32_t)(h2 & 0xffffffffULL);
  out[3] = (uint32_t)(h2 >> 32);
}
