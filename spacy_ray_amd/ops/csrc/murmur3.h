// MurmurHash3 x64_128 (public-domain algorithm by Austin Appleby),
// implemented from the algorithm description for this project.
//
// Entry points:
//   murmur3_x64_128  : general byte-string -> two uint64 lanes (16 bytes)
//   murmur3_hash4_u64: the Thinc `Ops.hash` contract — hash one 8-byte key,
//       return 4 uint32 lanes (the 16 output bytes little-endian).  Used to
//       pick the 4 HashEmbed rows per id (behavioral contract of thinc's
//       NumpyOps/CupyOps.hash `hash128_x64`).
//   hash_utf8        : 64-bit string hash for the StringStore.
//
// NOTE: the x86_128 variant is NOT equivalent here — for 8-byte keys its
// lanes 2 and 3 are structurally identical (no key material ever reaches
// h3/h4), which would halve HashEmbed's hash diversity.  x64_128 gives 4
// well-mixed lanes.  The HIP device copy in ops/kernels/ must stay
// bit-identical to this file.
#pragma once
#include <cstdint>
#include <cstring>

namespace srx {

static inline uint64_t rotl64(uint64_t x, int8_t r) {
  return (x << r) | (x >> (64 - r));
}

static inline uint64_t fmix64(uint64_t k) {
  k ^= k >> 33;
  k *= 0xff51afd7ed558ccdULL;
  k ^= k >> 33;
  k *= 0xc4ceb9fe1a85ec53ULL;
  k ^= k >> 33;
  return k;
}

// General-purpose x64_128 over arbitrary bytes.
inline void murmur3_x64_128(const void* key, int len, uint32_t seed, uint64_t out[2]) {
  const uint8_t* data = (const uint8_t*)key;
  const int nblocks = len / 16;
  uint64_t h1 = seed, h2 = seed;
  const uint64_t c1 = 0x87c37b91114253d5ULL;
  const uint64_t c2 = 0x4cf5ad432745937fULL;

  for (int i = 0; i < nblocks; i++) {
    uint64_t k1, k2;
    std::memcpy(&k1, data + i * 16 + 0, 8);
    std::memcpy(&k2, data + i * 16 + 8, 8);
    k1 *= c1; k1 = rotl64(k1, 31); k1 *= c2; h1 ^= k1;
    h1 = rotl64(h1, 27); h1 += h2; h1 = h1 * 5 + 0x52dce729ULL;
    k2 *= c2; k2 = rotl64(k2, 33); k2 *= c1; h2 ^= k2;
    h2 = rotl64(h2, 31); h2 += h1; h2 = h2 * 5 + 0x38495ab5ULL;
  }

  const uint8_t* tail = data + nblocks * 16;
  uint64_t k1 = 0, k2 = 0;
  switch (len & 15) {
    case 15: k2 ^= (uint64_t)tail[14] << 48; [[fallthrough]];
    case 14: k2 ^= (uint64_t)tail[13] << 40; [[fallthrough]];
    case 13: k2 ^= (uint64_t)tail[12] << 32; [[fallthrough]];
    case 12: k2 ^= (uint64_t)tail[11] << 24; [[fallthrough]];
    case 11: k2 ^= (uint64_t)tail[10] << 16; [[fallthrough]];
    case 10: k2 ^= (uint64_t)tail[9] << 8;   [[fallthrough]];
    case 9:  k2 ^= (uint64_t)tail[8] << 0;
             k2 *= c2; k2 = rotl64(k2, 33); k2 *= c1; h2 ^= k2; [[fallthrough]];
    case 8:  k1 ^= (uint64_t)tail[7] << 56;  [[fallthrough]];
    case 7:  k1 ^= (uint64_t)tail[6] << 48;  [[fallthrough]];
    case 6:  k1 ^= (uint64_t)tail[5] << 40;  [[fallthrough]];
    case 5:  k1 ^= (uint64_t)tail[4] << 32;  [[fallthrough]];
    case 4:  k1 ^= (uint64_t)tail[3] << 24;  [[fallthrough]];
    case 3:  k1 ^= (uint64_t)tail[2] << 16;  [[fallthrough]];
    case 2:  k1 ^= (uint64_t)tail[1] << 8;   [[fallthrough]];
    case 1:  k1 ^= (uint64_t)tail[0] << 0;
             k1 *= c1; k1 = rotl64(k1, 31); k1 *= c2; h1 ^= k1;
  }

  h1 ^= (uint64_t)len; h2 ^= (uint64_t)len;
  h1 += h2; h2 += h1;
  h1 = fmix64(h1); h2 = fmix64(h2);
  h1 += h2; h2 += h1;
  out[0] = h1; out[1] = h2;
}

// Specialized 8-byte (little-endian uint64) key.  Bit-identical to
// murmur3_x64_128(&key, 8, seed, out) on a little-endian host; written out
// so the HIP device function can mirror it instruction-for-instruction.
inline void murmur3_hash4_u64(uint64_t key, uint32_t seed, uint32_t out[4]) {
  const uint64_t c1 = 0x87c37b91114253d5ULL;
  const uint64_t c2 = 0x4cf5ad432745937fULL;
  uint64_t h1 = seed, h2 = seed;
  uint64_t k1 = key;
  k1 *= c1; k1 = rotl64(k1, 31); k1 *= c2; h1 ^= k1;
  h1 ^= 8ULL; h2 ^= 8ULL;
  h1 += h2; h2 += h1;
  h1 = fmix64(h1); h2 = fmix64(h2);
  h1 += h2; h2 += h1;
  out[0] = (uint32_t)(h1 & 0xffffffffULL);
  out[1] = (uint32_t)(h1 >> 32);
  out[2] = (uint32_t)(h2 & 0xffffffffULL);
  out[3] = (uint32_t)(h2 >> 32);
}

// 64-bit string hash for the StringStore (stable across runs/platforms):
// first 8 output bytes of murmur3_x64_128(bytes, seed=1).
inline uint64_t hash_utf8(const char* data, int len) {
  uint64_t out[2];
  murmur3_x64_128(data, len, 1u, out);
  return out[0];
}

// MurmurHash64A (MurmurHash2, 64-bit, by Austin Appleby — public-domain
// algorithm, implemented from its description).  spaCy's StringStore hash
// is murmurhash.mrmr.hash64(utf8, len, seed=1) = this function; the real
// `.spacy` DocBin format references strings by these values, so the
// DocBin reader/writer (data/docbin.py) must match them bit-for-bit
// (reference contract: spaCy strings.pyx hash_string via
// /root/reference/bin/get-data.sh's `spacy convert` output).
inline uint64_t murmur2_64a(const void* key, int len, uint64_t seed) {
  const uint64_t m = 0xc6a4a7935bd1e995ULL;
  const int r = 47;
  uint64_t h = seed ^ ((uint64_t)len * m);
  const uint8_t* data = (const uint8_t*)key;
  const uint8_t* end = data + (len & ~7);
  while (data != end) {
    uint64_t k;
    std::memcpy(&k, data, 8);
    data += 8;
    k *= m;
    k ^= k >> r;
    k *= m;
    h ^= k;
    h *= m;
  }
  switch (len & 7) {
    case 7: h ^= (uint64_t)data[6] << 48; [[fallthrough]];
    case 6: h ^= (uint64_t)data[5] << 40; [[fallthrough]];
    case 5: h ^= (uint64_t)data[4] << 32; [[fallthrough]];
    case 4: h ^= (uint64_t)data[3] << 24; [[fallthrough]];
    case 3: h ^= (uint64_t)data[2] << 16; [[fallthrough]];
    case 2: h ^= (uint64_t)data[1] << 8;  [[fallthrough]];
    case 1: h ^= (uint64_t)data[0];
            h *= m;
  }
  h ^= h >> r;
  h *= m;
  h ^= h >> r;
  return h;
}

}  // namespace srx
