// Shared xxh64 primitives — compiled for HOST (pool hashing, static tables)
// and DEVICE (batch key hashing on gfx950) from the same source so keys are
// bit-identical everywhere.  Mirrors pathway_amd/internals/api.py:xxh64 and
// the torch reference pathway_amd/engine/hashing.py (tests/test_hash.py
// checks all three agree).
//
// Reference semantics: 128-bit key = xxh64(seed=SEED_LO) || xxh64(seed=SEED_HI)
// over the canonical tagged serialization (value.rs:40-66 analog).
#pragma once
#include <stdint.h>

#if defined(__HIPCC__)
#define PW_HD __host__ __device__ __forceinline__
#else
#define PW_HD static inline
#endif

#define PW_P1 0x9E3779B185EBCA87ULL
#define PW_P2 0xC2B2AE3D27D4EB4FULL
#define PW_P3 0x165667B19E3779F9ULL
#define PW_P4 0x85EBCA77C2B2AE63ULL
#define PW_P5 0x27D4EB2F165667C5ULL

#define PW_SEED_LO 0ULL
#define PW_SEED_HI 0x9E3779B185EBCA87ULL

PW_HD uint64_t pw_rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }

PW_HD uint64_t pw_round(uint64_t acc, uint64_t inp) {
  acc += inp * PW_P2;
  acc = pw_rotl64(acc, 31);
  return acc * PW_P1;
}

PW_HD uint64_t pw_merge_round(uint64_t acc, uint64_t val) {
  val = pw_round(0, val);
  acc ^= val;
  return acc * PW_P1 + PW_P4;
}

PW_HD uint64_t pw_avalanche(uint64_t h) {
  h ^= h >> 33;
  h *= PW_P2;
  h ^= h >> 29;
  h *= PW_P3;
  h ^= h >> 32;
  return h;
}

// xxh64 over W little-endian 8-byte words held in a local array.
template <int W>
PW_HD uint64_t pw_xxh64_words(const uint64_t* w, uint64_t seed) {
  const int nbytes = W * 8;
  uint64_t h;
  int i = 0;
  if (nbytes >= 32) {
    uint64_t v1 = seed + PW_P1 + PW_P2;
    uint64_t v2 = seed + PW_P2;
    uint64_t v3 = seed;
    uint64_t v4 = seed - PW_P1;
    while (i + 4 <= W) {
      v1 = pw_round(v1, w[i]);
      v2 = pw_round(v2, w[i + 1]);
      v3 = pw_round(v3, w[i + 2]);
      v4 = pw_round(v4, w[i + 3]);
      i += 4;
    }
    h = pw_rotl64(v1, 1) + pw_rotl64(v2, 7) + pw_rotl64(v3, 12) + pw_rotl64(v4, 18);
    h = pw_merge_round(h, v1);
    h = pw_merge_round(h, v2);
    h = pw_merge_round(h, v3);
    h = pw_merge_round(h, v4);
  } else {
    h = seed + PW_P5;
  }
  h += (uint64_t)nbytes;
  for (; i < W; ++i) {
    h ^= pw_round(0, w[i]);
    h = pw_rotl64(h, 27) * PW_P1 + PW_P4;
  }
  return pw_avalanche(h);
}

// general byte-range xxh64 (for varlen string/bytes hashing)
PW_HD uint64_t pw_xxh64_bytes(const uint8_t* data, int64_t n, uint64_t seed) {
  int64_t i = 0;
  uint64_t h;
  if (n >= 32) {
    uint64_t v1 = seed + PW_P1 + PW_P2;
    uint64_t v2 = seed + PW_P2;
    uint64_t v3 = seed;
    uint64_t v4 = seed - PW_P1;
    while (i + 32 <= n) {
      uint64_t a, b, c, d;
      __builtin_memcpy(&a, data + i, 8);
      __builtin_memcpy(&b, data + i + 8, 8);
      __builtin_memcpy(&c, data + i + 16, 8);
      __builtin_memcpy(&d, data + i + 24, 8);
      v1 = pw_round(v1, a);
      v2 = pw_round(v2, b);
      v3 = pw_round(v3, c);
      v4 = pw_round(v4, d);
      i += 32;
    }
    h = pw_rotl64(v1, 1) + pw_rotl64(v2, 7) + pw_rotl64(v3, 12) + pw_rotl64(v4, 18);
    h = pw_merge_round(h, v1);
    h = pw_merge_round(h, v2);
    h = pw_merge_round(h, v3);
    h = pw_merge_round(h, v4);
  } else {
    h = seed + PW_P5;
  }
  h += (uint64_t)n;
  while (i + 8 <= n) {
    uint64_t k;
    __builtin_memcpy(&k, data + i, 8);
    h ^= pw_round(0, k);
    h = pw_rotl64(h, 27) * PW_P1 + PW_P4;
    i += 8;
  }
  while (i + 4 <= n) {
    uint32_t k;
    __builtin_memcpy(&k, data + i, 4);
    h ^= (uint64_t)k * PW_P1;
    h = pw_rotl64(h, 23) * PW_P2 + PW_P3;
    i += 4;
  }
  while (i < n) {
    h ^= (uint64_t)data[i] * PW_P5;
    h = pw_rotl64(h, 11) * PW_P1;
    i += 1;
  }
  return pw_avalanche(h);
}
