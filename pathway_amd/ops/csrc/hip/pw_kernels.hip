// pathway_amd HIP kernels for gfx950 (MI355X / CDNA4).
//
// Round-1 set:
//   * fused 128-bit row hashing (replaces ~60 chained torch int64 ops and
//     their launch overhead with one memory-bound pass)
//   * varlen byte-range hashing (strings: offsets+bytes arenas)
//   * fused consolidate-weights (segment boundary + weight segment-sum)
//   * sorted additive-state merge helper (count/sum reduce state)
//
// Design per /opt/skills/guides/cdna_hip_programming.md: 64-wide waves,
// 256-thread blocks, grid-stride loops capped near 2048 blocks for
// memory-bound kernels, coalesced 8B/lane accesses.  All launchers take the
// caller's hipStream_t (torch current stream) — no syncs inside.
#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdio.h>

#include "../xxhash_common.h"

#define PW_BLOCK 256
#define PW_MAX_GRID 2048

static inline int pw_grid(int64_t n) {
  int64_t g = (n + PW_BLOCK - 1) / PW_BLOCK;
  if (g > PW_MAX_GRID) g = PW_MAX_GRID;
  if (g < 1) g = 1;
  return (int)g;
}

// ---------------------------------------------------------------- hashing --

// up to 20 input word columns, SoA int64
struct WordPtrs {
  const uint64_t* p[20];
};

template <int W>
__global__ void k_hash128_words(WordPtrs ptrs, int64_t n, uint64_t* lo,
                                uint64_t* hi) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint64_t w[W];
#pragma unroll
    for (int j = 0; j < W; ++j) w[j] = ptrs.p[j][i];
    lo[i] = pw_xxh64_words<W>(w, PW_SEED_LO);
    hi[i] = pw_xxh64_words<W>(w, PW_SEED_HI);
  }
}

extern "C" int pw_hash128_words(const void** word_ptrs, int nwords, int64_t n,
                                void* lo, void* hi, void* stream) {
  WordPtrs ptrs;
  for (int j = 0; j < nwords && j < 20; ++j)
    ptrs.p[j] = (const uint64_t*)word_ptrs[j];
  hipStream_t s = (hipStream_t)stream;
  dim3 grid(pw_grid(n)), block(PW_BLOCK);
#define CASE(W)                                                        \
  case W:                                                              \
    hipLaunchKernelGGL((k_hash128_words<W>), grid, block, 0, s, ptrs,  \
                       n, (uint64_t*)lo, (uint64_t*)hi);               \
    break;
  switch (nwords) {
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
    CASE(9) CASE(10) CASE(11) CASE(12) CASE(13) CASE(14) CASE(15) CASE(16)
    CASE(17) CASE(18) CASE(19) CASE(20)
    default:
      return -1;
  }
#undef CASE
  return (int)hipGetLastError();
}

// tagged fixed-width value hash: hash of [tag, payload] per row
__global__ void k_value_hash(const uint64_t* payload, uint64_t tag, int64_t n,
                             uint64_t* lo, uint64_t* hi) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint64_t w[2] = {tag, payload[i]};
    lo[i] = pw_xxh64_words<2>(w, PW_SEED_LO);
    hi[i] = pw_xxh64_words<2>(w, PW_SEED_HI);
  }
}

extern "C" int pw_value_hash(const void* payload, uint64_t tag, int64_t n,
                             void* lo, void* hi, void* stream) {
  hipLaunchKernelGGL(k_value_hash, dim3(pw_grid(n)), dim3(PW_BLOCK), 0,
                     (hipStream_t)stream, (const uint64_t*)payload, tag, n,
                     (uint64_t*)lo, (uint64_t*)hi);
  return (int)hipGetLastError();
}

__device__ uint64_t pw_xxh64_long(const uint8_t* data, int64_t n, uint64_t tag,
                                  uint64_t seed);

// varlen: hash of [tag-word || bytes[start:end]] per row.
// One WAVE per row (rows are short strings; lanes cooperate on ≥8B chunks
// would need a parallel xxh64 — instead lane 0 of each 8-lane group handles
// one row: still coalesced enough for short tokens, revisit if hot).
__global__ void k_varlen_hash(const uint8_t* bytes, const int64_t* offsets,
                              uint64_t tag, int64_t n, uint64_t* lo,
                              uint64_t* hi) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t a = offsets[i], b = offsets[i + 1];
    // serialize: tag as 8-byte LE word, then raw bytes
    // compute xxh64 incrementally over the concatenation without copying:
    // tag word is 8 bytes => for total length < 32 we replicate the python
    // byte-path; for simplicity build a small local buffer when len<=56,
    // else fall back to a two-part streaming evaluation.
    uint8_t buf[64];
    int64_t len = b - a;
    if (len <= 56) {
      uint64_t t = tag;
      __builtin_memcpy(buf, &t, 8);
      for (int64_t k = 0; k < len; ++k) buf[8 + k] = bytes[a + k];
      lo[i] = pw_xxh64_bytes(buf, 8 + len, PW_SEED_LO);
      hi[i] = pw_xxh64_bytes(buf, 8 + len, PW_SEED_HI);
    } else {
      // long string: hash in the streaming formulation
      lo[i] = pw_xxh64_long(bytes + a, len, tag, PW_SEED_LO);
      hi[i] = pw_xxh64_long(bytes + a, len, tag, PW_SEED_HI);
    }
  }
}

// streaming xxh64 of (8-byte tag word || data[0:n]) for n+8 >= 32
__device__ uint64_t pw_xxh64_long(const uint8_t* data, int64_t n, uint64_t tag,
                                  uint64_t seed) {
  // total message = 8 + n bytes, guaranteed >= 64 here
  uint64_t v1 = seed + PW_P1 + PW_P2;
  uint64_t v2 = seed + PW_P2;
  uint64_t v3 = seed;
  uint64_t v4 = seed - PW_P1;
  // first stripe: tag + first 24 data bytes
  uint64_t a = tag, b, c, d;
  __builtin_memcpy(&b, data, 8);
  __builtin_memcpy(&c, data + 8, 8);
  __builtin_memcpy(&d, data + 16, 8);
  v1 = pw_round(v1, a);
  v2 = pw_round(v2, b);
  v3 = pw_round(v3, c);
  v4 = pw_round(v4, d);
  int64_t i = 24;
  while (i + 32 <= n) {
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
  uint64_t h = pw_rotl64(v1, 1) + pw_rotl64(v2, 7) + pw_rotl64(v3, 12) +
               pw_rotl64(v4, 18);
  h = pw_merge_round(h, v1);
  h = pw_merge_round(h, v2);
  h = pw_merge_round(h, v3);
  h = pw_merge_round(h, v4);
  h += (uint64_t)(n + 8);
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

extern "C" int pw_varlen_hash(const void* bytes, const void* offsets,
                              uint64_t tag, int64_t n, void* lo, void* hi,
                              void* stream) {
  hipLaunchKernelGGL(k_varlen_hash, dim3(pw_grid(n)), dim3(PW_BLOCK), 0,
                     (hipStream_t)stream, (const uint8_t*)bytes,
                     (const int64_t*)offsets, tag, n, (uint64_t*)lo,
                     (uint64_t*)hi);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------ consolidate --

// fused: mark run starts over up-to-4 sorted word columns + inclusive-scan
// segment ids will still be done by torch cumsum; this kernel fuses the
// multi-column neq reduction (saves 2W-1 elementwise launches).
struct Words4 {
  const uint64_t* p[4];
};

__global__ void k_run_starts(Words4 w, int nw, int64_t n, bool* starts) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool ne = (i == 0);
    if (i > 0) {
      for (int j = 0; j < nw; ++j) ne |= (w.p[j][i] != w.p[j][i - 1]);
    }
    starts[i] = ne;
  }
}

extern "C" int pw_run_starts(const void** word_ptrs, int nwords, int64_t n,
                             void* starts, void* stream) {
  Words4 w;
  for (int j = 0; j < nwords && j < 4; ++j) w.p[j] = (const uint64_t*)word_ptrs[j];
  hipLaunchKernelGGL(k_run_starts, dim3(pw_grid(n)), dim3(PW_BLOCK), 0,
                     (hipStream_t)stream, w, nwords, n, (bool*)starts);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------- host --

extern "C" void pw_host_hash128_bytes(const void* data, int64_t n,
                                      uint64_t* lo, uint64_t* hi) {
  *lo = pw_xxh64_bytes((const uint8_t*)data, n, PW_SEED_LO);
  *hi = pw_xxh64_bytes((const uint8_t*)data, n, PW_SEED_HI);
}

// batch host hashing of varlen byte rows (string pool fills)
extern "C" void pw_host_varlen_hash(const void* bytes, const int64_t* offsets,
                                    uint64_t tag, int64_t n, uint64_t* lo,
                                    uint64_t* hi) {
  const uint8_t* b = (const uint8_t*)bytes;
  for (int64_t i = 0; i < n; ++i) {
    int64_t a = offsets[i], e = offsets[i + 1];
    int64_t len = e - a;
    // tag word + bytes
    uint8_t stackbuf[4096];
    uint8_t* buf = stackbuf;
    if (len + 8 > (int64_t)sizeof(stackbuf)) buf = new uint8_t[len + 8];
    uint64_t t = tag;
    __builtin_memcpy(buf, &t, 8);
    __builtin_memcpy(buf + 8, b + a, len);
    lo[i] = pw_xxh64_bytes(buf, len + 8, PW_SEED_LO);
    hi[i] = pw_xxh64_bytes(buf, len + 8, PW_SEED_HI);
    if (buf != stackbuf) delete[] buf;
  }
}

// ------------------------------------------------------- sorted-key search --

// Lexicographic binary search over up-to-4 parallel sorted word columns.
// One thread per query — replaces the ~10-kernel-per-iteration torch
// binary-search loop (the dominant launch-overhead cost in the profile,
// profiles/wordcount_r01.md).
struct SWords {
  const uint64_t* s[4];
  const uint64_t* q[4];
};

__device__ __forceinline__ int pw_cmp_row(const SWords& w, int nw, int64_t si,
                                          int64_t qi) {
  // compare sorted[si] ? query[qi] as SIGNED int64 lexicographic
  for (int j = 0; j < nw; ++j) {
    int64_t a = (int64_t)w.s[j][si];
    int64_t b = (int64_t)w.q[j][qi];
    if (a < b) return -1;
    if (a > b) return 1;
  }
  return 0;
}

template <bool RIGHT>
__global__ void k_searchsorted(SWords w, int nw, int64_t m, int64_t nq,
                               int64_t* out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nq;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t lo = 0, hi = m;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      int c = pw_cmp_row(w, nw, mid, i);
      bool go_right = RIGHT ? (c <= 0) : (c < 0);
      if (go_right)
        lo = mid + 1;
      else
        hi = mid;
    }
    out[i] = lo;
  }
}

extern "C" int pw_searchsorted(const void** sorted_ptrs, const void** query_ptrs,
                               int nwords, int64_t m, int64_t nq, int right,
                               void* out, void* stream) {
  SWords w;
  for (int j = 0; j < nwords && j < 4; ++j) {
    w.s[j] = (const uint64_t*)sorted_ptrs[j];
    w.q[j] = (const uint64_t*)query_ptrs[j];
  }
  dim3 grid(pw_grid(nq)), block(PW_BLOCK);
  if (right)
    hipLaunchKernelGGL((k_searchsorted<true>), grid, block, 0,
                       (hipStream_t)stream, w, nwords, m, nq, (int64_t*)out);
  else
    hipLaunchKernelGGL((k_searchsorted<false>), grid, block, 0,
                       (hipStream_t)stream, w, nwords, m, nq, (int64_t*)out);
  return (int)hipGetLastError();
}

// lookup: left-searchsorted + equality check, fused (pos clamped, found flag)
__global__ void k_lookup(SWords w, int nw, int64_t m, int64_t nq, int64_t* pos,
                         bool* found) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nq;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t lo = 0, hi = m;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      if (pw_cmp_row(w, nw, mid, i) < 0)
        lo = mid + 1;
      else
        hi = mid;
    }
    bool f = false;
    int64_t p = lo;
    if (lo < m && pw_cmp_row(w, nw, lo, i) == 0) f = true;
    if (p >= m) p = m > 0 ? m - 1 : 0;
    pos[i] = p;
    found[i] = f;
  }
}

extern "C" int pw_lookup(const void** sorted_ptrs, const void** query_ptrs,
                         int nwords, int64_t m, int64_t nq, void* pos,
                         void* found, void* stream) {
  SWords w;
  for (int j = 0; j < nwords && j < 4; ++j) {
    w.s[j] = (const uint64_t*)sorted_ptrs[j];
    w.q[j] = (const uint64_t*)query_ptrs[j];
  }
  hipLaunchKernelGGL(k_lookup, dim3(pw_grid(nq)), dim3(PW_BLOCK), 0,
                     (hipStream_t)stream, w, nwords, m, nq, (int64_t*)pos,
                     (bool*)found);
  return (int)hipGetLastError();
}

// key_range: lo (left bound) and hi (right bound) in one kernel
__global__ void k_key_range(SWords w, int nw, int64_t m, int64_t nq,
                            int64_t* lo_out, int64_t* hi_out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nq;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t lo = 0, hi = m;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      if (pw_cmp_row(w, nw, mid, i) < 0)
        lo = mid + 1;
      else
        hi = mid;
    }
    int64_t left = lo;
    lo = left;
    hi = m;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      if (pw_cmp_row(w, nw, mid, i) <= 0)
        lo = mid + 1;
      else
        hi = mid;
    }
    lo_out[i] = left;
    hi_out[i] = lo;
  }
}

extern "C" int pw_key_range(const void** sorted_ptrs, const void** query_ptrs,
                            int nwords, int64_t m, int64_t nq, void* lo,
                            void* hi, void* stream) {
  SWords w;
  for (int j = 0; j < nwords && j < 4; ++j) {
    w.s[j] = (const uint64_t*)sorted_ptrs[j];
    w.q[j] = (const uint64_t*)query_ptrs[j];
  }
  hipLaunchKernelGGL(k_key_range, dim3(pw_grid(nq)), dim3(PW_BLOCK), 0,
                     (hipStream_t)stream, w, nwords, m, nq, (int64_t*)lo,
                     (int64_t*)hi);
  return (int)hipGetLastError();
}

// ---------------------------------------------------- pooled string hash --

// Fused dictionary-column value hash: out = pool_hash[code] blended with
// the canonical None hash for code < 0.  Replaces a clamp + 2 gathers +
// 2 wheres + mask compare torch chain with one pass.
__global__ void k_pool_hash(const int64_t* codes, const uint64_t* pool_lo,
                            const uint64_t* pool_hi, uint64_t none_lo,
                            uint64_t none_hi, int64_t n, uint64_t* lo,
                            uint64_t* hi) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t c = codes[i];
    if (c >= 0) {
      lo[i] = pool_lo[c];
      hi[i] = pool_hi[c];
    } else {
      lo[i] = none_lo;
      hi[i] = none_hi;
    }
  }
}

extern "C" int pw_pool_hash(const void* codes, const void* pool_lo,
                            const void* pool_hi, uint64_t none_lo,
                            uint64_t none_hi, int64_t n, void* lo, void* hi,
                            void* stream) {
  hipLaunchKernelGGL(k_pool_hash, dim3(pw_grid(n)), dim3(PW_BLOCK), 0,
                     (hipStream_t)stream, (const int64_t*)codes,
                     (const uint64_t*)pool_lo, (const uint64_t*)pool_hi,
                     none_lo, none_hi, n, (uint64_t*)lo, (uint64_t*)hi);
  return (int)hipGetLastError();
}

// varlen hash with explicit [start, end) per row (token parse output —
// tokens exclude separators so rows are not contiguous)
__global__ void k_varlen_hash_se(const uint8_t* bytes, const int64_t* starts,
                                 const int64_t* ends, uint64_t tag, int64_t n,
                                 uint64_t* lo, uint64_t* hi) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t a = starts[i], b = ends[i];
    uint8_t buf[64];
    int64_t len = b - a;
    if (len <= 56) {
      uint64_t t = tag;
      __builtin_memcpy(buf, &t, 8);
      for (int64_t k = 0; k < len; ++k) buf[8 + k] = bytes[a + k];
      lo[i] = pw_xxh64_bytes(buf, 8 + len, PW_SEED_LO);
      hi[i] = pw_xxh64_bytes(buf, 8 + len, PW_SEED_HI);
    } else {
      lo[i] = pw_xxh64_long(bytes + a, len, tag, PW_SEED_LO);
      hi[i] = pw_xxh64_long(bytes + a, len, tag, PW_SEED_HI);
    }
  }
}

extern "C" int pw_varlen_hash_se(const void* bytes, const void* starts,
                                 const void* ends, uint64_t tag, int64_t n,
                                 void* lo, void* hi, void* stream) {
  hipLaunchKernelGGL(k_varlen_hash_se, dim3(pw_grid(n)), dim3(PW_BLOCK), 0,
                     (hipStream_t)stream, (const uint8_t*)bytes,
                     (const int64_t*)starts, (const int64_t*)ends, tag, n,
                     (uint64_t*)lo, (uint64_t*)hi);
  return (int)hipGetLastError();
}

// ------------------------------------------------------- hash aggregation --
//
// Sort-free per-batch pre-aggregation for additive group-reduce
// (count/int-sum): one pass over the batch accumulating into an open-
// addressing hash table keyed by the 128-bit group key, instead of
// radix-sorting the whole batch and running the segment-boundary chain
// (rocprof: sort family + partition/nonzero were ~0.9 ms of a 2.9 ms
// wordcount step at 4M rows).
//
// Claim protocol (no intra-wave spinning): CAS the first key word against
// EMPTY; the winner then publishes word1 and the representative row.  A
// reader that observes word0 == its key but an unpublished word1 simply
// mismatches and claims the next slot — the same key may transiently own
// two slots, which the (tiny) sorted consolidation of the collected
// uniques merges afterwards.  EMPTY is a reserved 64-bit value; group keys
// are xxh64 outputs, so a real key equals it with probability 2^-64.

#define PW_HA_EMPTY 0x8000000000000000ULL

struct AccPtrs {
  const long long* p[8];
};
struct AccPtrsMut {
  long long* p[8];
};

__global__ void k_hash_agg(const long long* k0, const long long* k1,
                           AccPtrs contribs, int nacc, int64_t n,
                           unsigned long long* tk0, long long* tk1,
                           AccPtrsMut taccs, long long* rep, uint64_t mask,
                           unsigned int* overflow) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    unsigned long long key0 = (unsigned long long)k0[i];
    long long key1 = k1[i];
    uint64_t slot = ((uint64_t)key0) & mask;
    // probe cap: the table is sized ~2x expected uniques (L2-resident);
    // a longer chain means the estimate was wrong -> flag and let the
    // caller redo the batch on the sort path with a bigger table
    for (uint64_t tries = 0; tries < 512 && tries <= mask; ++tries) {
      // plain load first: the overwhelmingly common case is an already
      // claimed, matching slot — an L2 read beats a CAS round-trip
      unsigned long long cur = tk0[slot];
      if (cur == PW_HA_EMPTY)
        cur = atomicCAS(&tk0[slot], PW_HA_EMPTY, key0);
      if (cur == PW_HA_EMPTY) {
        // claimed: publish the rest of the key + a representative row
        tk1[slot] = key1;
        rep[slot] = (long long)i;
        __threadfence();
      } else if (cur != key0) {
        slot = (slot + 1) & mask;
        continue;
      } else if (tk1[slot] != key1) {
        // word0 matches but word1 is different OR not yet published:
        // treat as mismatch and probe on (duplicate slots merge later)
        slot = (slot + 1) & mask;
        continue;
      }
      for (int a = 0; a < nacc; ++a)
        atomicAdd((unsigned long long*)&taccs.p[a][slot],
                  (unsigned long long)contribs.p[a][i]);
      goto next_row;
    }
    atomicOr(overflow, 1u);
  next_row:;
  }
}

__global__ void k_hash_agg_init(unsigned long long* tk0, uint64_t cap) {
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < cap;
       i += (uint64_t)gridDim.x * blockDim.x)
    tk0[i] = PW_HA_EMPTY;
}

__global__ void k_hash_agg_collect(const unsigned long long* tk0,
                                   const long long* tk1, AccPtrs taccs,
                                   const long long* rep, int nacc,
                                   uint64_t cap, unsigned int* counter,
                                   long long* out_k0, long long* out_k1,
                                   AccPtrsMut out_accs, long long* out_rep) {
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < cap;
       i += (uint64_t)gridDim.x * blockDim.x) {
    unsigned long long k = tk0[i];
    if (k == PW_HA_EMPTY) continue;
    unsigned int idx = atomicAdd(counter, 1u);
    out_k0[idx] = (long long)k;
    out_k1[idx] = tk1[i];
    out_rep[idx] = rep[i];
    for (int a = 0; a < nacc; ++a) out_accs.p[a][idx] = taccs.p[a][i];
  }
}

extern "C" int pw_hash_agg(const void* k0, const void* k1,
                           const void** contrib_ptrs, int nacc, int64_t n,
                           void* tk0, void* tk1, void** tacc_ptrs, void* rep,
                           int64_t capacity, void* counter, void* out_k0,
                           void* out_k1, void** out_acc_ptrs, void* out_rep,
                           void* overflow, void* stream) {
  if (nacc > 8) return 1;
  hipStream_t s = (hipStream_t)stream;
  AccPtrs cp;
  AccPtrsMut tp, op;
  for (int a = 0; a < nacc; ++a) {
    cp.p[a] = (const long long*)contrib_ptrs[a];
    tp.p[a] = (long long*)tacc_ptrs[a];
    op.p[a] = (long long*)out_acc_ptrs[a];
  }
  AccPtrs tpc;
  for (int a = 0; a < nacc; ++a) tpc.p[a] = tp.p[a];
  uint64_t mask = (uint64_t)capacity - 1;
  dim3 block(PW_BLOCK);
  hipLaunchKernelGGL(k_hash_agg_init, dim3(pw_grid(capacity)), block, 0, s,
                     (unsigned long long*)tk0, (uint64_t)capacity);
  hipLaunchKernelGGL(k_hash_agg, dim3(pw_grid(n)), block, 0, s,
                     (const long long*)k0, (const long long*)k1, cp, nacc, n,
                     (unsigned long long*)tk0, (long long*)tk1, tp,
                     (long long*)rep, mask, (unsigned int*)overflow);
  hipLaunchKernelGGL(k_hash_agg_collect, dim3(pw_grid(capacity)), block, 0, s,
                     (const unsigned long long*)tk0, (const long long*)tk1,
                     tpc, (long long*)rep, nacc, (uint64_t)capacity,
                     (unsigned int*)counter, (long long*)out_k0,
                     (long long*)out_k1, op, (long long*)out_rep);
  return (int)hipGetLastError();
}

// ------------------------------------------------------ fused seg-reduce --
//
// Replaces the segment-boundary chain of the sorted pre-aggregation
// (run-starts → cumsum → nonzero → gathers → per-acc prefix-sum diffs,
// ~15 small launches on a 4M-row batch) with two kernels + one tiny
// cumsum.  Input: rows sorted by the 2-word key; output: compacted unique
// keys, per-segment int64 sums for each accumulator, and the first-row
// index of every segment (the carried-column representative).
//
// K2 does a wave-level segmented sum (shfl_up over 64 lanes): one
// atomicAdd per run-per-wave instead of one per row — on sorted data with
// d duplicates per key that is a ~min(64, d)× reduction in atomics.

struct SegWords {
  const int64_t* p[8];
};

__global__ void k_segred_count(SegWords w, int nw, int64_t n,
                               int* block_counts) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int flag = 0;
  if (i < n) {
    if (i == 0) flag = 1;
    else {
      for (int j = 0; j < nw; ++j)
        if (w.p[j][i] != w.p[j][i - 1]) { flag = 1; break; }
    }
  }
  // block reduction of flags (blockDim = 256)
  __shared__ int cnt[PW_BLOCK / 64];
  int wavecnt = __popcll(__ballot(flag));
  if ((threadIdx.x & 63) == 0) cnt[threadIdx.x >> 6] = wavecnt;
  __syncthreads();
  if (threadIdx.x == 0) {
    int total = 0;
    for (int j = 0; j < PW_BLOCK / 64; ++j) total += cnt[j];
    block_counts[blockIdx.x] = total;
  }
}

__global__ void k_segred_emit(SegWords w, int nw, AccPtrs contribs, int nacc,
                              int64_t n, const long long* block_bases,
                              AccPtrsMut out_words, long long* out_first,
                              AccPtrsMut out_accs) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int lane = threadIdx.x & 63;
  int flag = 0;
  if (i < n) {
    if (i == 0) flag = 1;
    else {
      for (int j = 0; j < nw; ++j)
        if (w.p[j][i] != w.p[j][i - 1]) { flag = 1; break; }
    }
  }
  // block-wide exclusive scan of flags via per-wave ballot + LDS
  uint64_t ball = __ballot(flag);
  int wave = threadIdx.x >> 6;
  __shared__ int wave_tot[PW_BLOCK / 64];
  if (lane == 0) wave_tot[wave] = __popcll(ball);
  __syncthreads();
  int wave_base = 0;
  for (int j = 0; j < wave; ++j) wave_base += wave_tot[j];
  uint64_t below = ball & ((lane == 0) ? 0ULL : ((1ULL << lane) - 1));
  int local_excl = wave_base + __popcll(below);
  long long base = block_bases[blockIdx.x];
  // global segment id of row i (segments are 0-based); padding lanes get
  // the sentinel -1 so a real row followed by padding still detects its
  // run tail (dropping this cost the last segment whenever n % 64 != 0)
  long long sid = base + local_excl + flag - 1;
  if (i < n && flag) {
    for (int j = 0; j < nw; ++j) out_words.p[j][sid] = (long long)w.p[j][i];
    out_first[sid] = i;
  }
  // wave-segmented sums per accumulator
  long long s = (i < n) ? sid : -1;
  for (int a = 0; a < nacc; ++a) {
    long long v = (i < n) ? contribs.p[a][i] : 0;
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      long long vv = __shfl_up(v, off, 64);
      long long ss = __shfl_up(s, off, 64);
      if (lane >= off && ss == s) v += vv;
    }
    long long nxt = __shfl_down(s, 1, 64);
    bool tail = (lane == 63) || (nxt != s);
    if (s >= 0 && tail)
      atomicAdd((unsigned long long*)&out_accs.p[a][s],
                (unsigned long long)v);
  }
}

extern "C" int pw_seg_reduce_count(const void** word_ptrs, int nw, int64_t n,
                                   void* block_counts, int64_t nblocks,
                                   void* stream) {
  if (nw > 8) return 1;
  SegWords w;
  for (int j = 0; j < nw; ++j) w.p[j] = (const int64_t*)word_ptrs[j];
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_segred_count, dim3((uint32_t)nblocks), dim3(PW_BLOCK),
                     0, s, w, nw, n, (int*)block_counts);
  return (int)hipGetLastError();
}

extern "C" int pw_seg_reduce_emit(const void** word_ptrs, int nw,
                                  const void** contrib_ptrs, int nacc,
                                  int64_t n, const void* block_bases,
                                  void** out_word_ptrs, void* out_first,
                                  void** out_acc_ptrs, int64_t nblocks,
                                  void* stream) {
  if (nacc > 8 || nw > 8) return 1;
  SegWords w;
  for (int j = 0; j < nw; ++j) w.p[j] = (const int64_t*)word_ptrs[j];
  AccPtrs cp;
  AccPtrsMut op, ow;
  for (int a = 0; a < nacc; ++a) {
    cp.p[a] = (const long long*)contrib_ptrs[a];
    op.p[a] = (long long*)out_acc_ptrs[a];
  }
  for (int j = 0; j < nw; ++j) ow.p[j] = (long long*)out_word_ptrs[j];
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_segred_emit, dim3((uint32_t)nblocks), dim3(PW_BLOCK),
                     0, s, w, nw, cp, nacc, n, (const long long*)block_bases,
                     ow, (long long*)out_first, op);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------- partition --
// Radix partition of rows by destination rank (the exchange shuffle pack,
// reference pact.rs:56 / operators.rs:126).  dest[i] in [0, world);
// world <= 64.  Output: perm such that rows grouped by destination
// (sorted[j] = input[perm[j]]) and per-destination counts.  Order within a
// destination is unspecified (the exchange does not require stability).
//
// Two passes over dest, each block owning a contiguous chunk:
//   1. per-block histogram -> block_hist[b][d]
//   2. single-block exclusive scan in destination-major order gives each
//      (d, b) cell its global base
//   3. scatter: per-block shared counters (atomicAdd in LDS) assign
//      positions within the block's (d) quota.

#define PW_PART_MAX_DEST 64

__global__ void k_partition_count(const int64_t* dest, int64_t n,
                                  int64_t chunk, int world,
                                  long long* block_hist) {
  __shared__ int hist[PW_PART_MAX_DEST];
  for (int d = threadIdx.x; d < world; d += blockDim.x) hist[d] = 0;
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t end = min(start + chunk, n);
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    atomicAdd(&hist[(int)dest[i]], 1);
  __syncthreads();
  for (int d = threadIdx.x; d < world; d += blockDim.x)
    block_hist[(int64_t)blockIdx.x * world + d] = hist[d];
}

// exclusive scan over cells ordered (d major, b minor); also emits counts[d]
__global__ void k_partition_scan(long long* block_hist, int64_t nblocks,
                                 int world, long long* counts) {
  // single block; simple sequential scan (cells <= 64 * 4096)
  if (threadIdx.x == 0) {
    long long run = 0;
    for (int d = 0; d < world; ++d) {
      long long dtotal = 0;
      for (int64_t b = 0; b < nblocks; ++b) {
        long long c = block_hist[b * world + d];
        block_hist[b * world + d] = run;
        run += c;
        dtotal += c;
      }
      counts[d] = dtotal;
    }
  }
}

__global__ void k_partition_scatter(const int64_t* dest, int64_t n,
                                    int64_t chunk, int world,
                                    const long long* block_bases,
                                    int64_t* perm) {
  __shared__ unsigned int offs[PW_PART_MAX_DEST];
  for (int d = threadIdx.x; d < world; d += blockDim.x) offs[d] = 0;
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t end = min(start + chunk, n);
  const long long* base = block_bases + (int64_t)blockIdx.x * world;
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    int d = (int)dest[i];
    unsigned int p = atomicAdd(&offs[d], 1u);
    perm[base[d] + p] = i;
  }
}

extern "C" int pw_partition(const void* dest, int64_t n, int world,
                            void* perm, void* counts, void* scratch,
                            int64_t nblocks, void* stream) {
  if (world > PW_PART_MAX_DEST) return 2;
  hipStream_t s = (hipStream_t)stream;
  int64_t chunk = (n + nblocks - 1) / nblocks;
  if (chunk < 1) chunk = 1;
  hipLaunchKernelGGL(k_partition_count, dim3((uint32_t)nblocks),
                     dim3(PW_BLOCK), 0, s, (const int64_t*)dest, n, chunk,
                     world, (long long*)scratch);
  hipLaunchKernelGGL(k_partition_scan, dim3(1), dim3(1), 0, s,
                     (long long*)scratch, nblocks, world,
                     (long long*)counts);
  hipLaunchKernelGGL(k_partition_scatter, dim3((uint32_t)nblocks),
                     dim3(PW_BLOCK), 0, s, (const int64_t*)dest, n, chunk,
                     world, (const long long*)scratch, (int64_t*)perm);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------ MFMA GEMM --
// Hand-written bf16 GEMM on gfx950 matrix cores with fused bias +
// activation epilogue — the embedder's projection/FFN GEMMs (north-star:
// "embedder forward on bf16 MFMA tiles").  C (M,N) = A (M,K) x B (K,N),
// row-major bf16 inputs, fp32 accumulate via mfma_f32_16x16x32_bf16,
// bf16 store.  Block = 128x128 tile, 4 waves (2x2), each wave owns a
// 64x64 sub-tile = 4x4 fragments; K staged through LDS in BK=64 steps,
// double-buffered.  act: 0 = none, 1 = exact GELU.
//
// Fragment layouts (cdna_hip_programming.md §3, verified by the gpu test
// against torch fp32 matmul with asymmetric inputs):
//   A: lane holds A[row0 + (l&15)][k0 + (l>>4)*8 + j], j=0..7
//   B: lane holds B[k0 + (l>>4)*8 + j][col0 + (l&15)]
//   C/D: lane l reg r -> row = row0 + (l>>4)*4 + r, col = col0 + (l&15)

typedef __attribute__((ext_vector_type(8))) __bf16 pw_frag8;
typedef __attribute__((ext_vector_type(4))) float pw_frag4f;

#define PW_GEMM_BM 128
#define PW_GEMM_BN 128
#define PW_GEMM_BK 64

__device__ __forceinline__ float pw_gelu(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752f));
}

// B is passed TRANSPOSED (Bt = B^T, (N, K) row-major): both A and Bt
// tiles then stage into LDS with linear 16B vector loads and both MFMA
// fragments are contiguous 8-element ds reads (weights are static, so
// the one-time transpose is free on the host side).
template <int NBUF>
__global__ __launch_bounds__(256) void k_gemm_bf16(
    const __bf16* __restrict__ A, const __bf16* __restrict__ Bt,
    const float* __restrict__ bias, __bf16* __restrict__ C, int64_t M,
    int64_t N, int64_t K, int act, int nbm) {
  // NBUF=2: classic intra-block double buffer (64 KB LDS, 2 blocks/CU).
  // NBUF=1: single buffer (32 KB LDS, ~5 blocks/CU) — latency hiding
  // from BLOCK-level parallelism instead; wins on short-K shapes where
  // the 6-iteration pipeline never fills (PMC: MfmaUtil 8.9% at NBUF=2).
  __shared__ __bf16 lasA[NBUF][PW_GEMM_BM][PW_GEMM_BK];
  // B stored TRANSPOSED in LDS ([col][k]) so each lane's 8-element
  // K-fragment is a contiguous ds_read_b128 (scattered writes happen
  // once per element; scattered reads would repeat per MFMA)
  __shared__ __bf16 lasBt[NBUF][PW_GEMM_BN][PW_GEMM_BK];
  // XCD-aware tile mapping: consecutive blockIdx tiles land on the same
  // XCD's L2 by swizzling in chunks (guide: blockIdx->tile must be
  // XCD-aware).  Simple bijective row-major with N-fastest is fine here
  // because N <= 12 tiles for the encoder shapes.
  int tile = blockIdx.x;
  int bm = tile % nbm;
  int bn = tile / nbm;
  int64_t row0 = (int64_t)bm * PW_GEMM_BM;
  int64_t col0 = (int64_t)bn * PW_GEMM_BN;
  int tid = threadIdx.x;
  int lane = tid & 63;
  int wave = tid >> 6;  // 0..3
  int wr = wave >> 1;   // wave row (0..1) -> 64-row strip
  int wc = wave & 1;    // wave col (0..1) -> 64-col strip

  pw_frag4f acc[4][4];
#pragma unroll
  for (int m = 0; m < 4; ++m)
#pragma unroll
    for (int n = 0; n < 4; ++n) acc[m][n] = {0.f, 0.f, 0.f, 0.f};

  int nk = (int)((K + PW_GEMM_BK - 1) / PW_GEMM_BK);
  bool interior_mn = (row0 + PW_GEMM_BM <= M) && (col0 + PW_GEMM_BN <= N);

  // fast staging for interior tiles: async global->LDS DMA, 16B per lane
  // (guide §5: global_load_lds dwordx4 — the compiler never auto-emits it)
  auto load_tiles_gll = [&](int kt, int buf) {
    int64_t k0 = (int64_t)kt * PW_GEMM_BK;
    // A tile: 16 KiB = 16 wave-chunks of 1 KiB; 4 chunks per wave
    typedef const __attribute__((address_space(1))) uint8_t* gptr_t;
    typedef __attribute__((address_space(3))) uint8_t* lptr_t;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int wchunk = wave * 4 + i;
      int eoff = wchunk * 512 + lane * 8;
      int r = eoff >> 6, c = eoff & 63;
      gptr_t srcA = (gptr_t)(const void*)&A[(row0 + r) * K + k0 + c];
      lptr_t dstA = (lptr_t)(void*)(&lasA[buf][0][0] + wchunk * 512);
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)srcA,
                                       (__attribute__((address_space(3))) void*)dstA,
                                       16, 0, 0);
      gptr_t srcB = (gptr_t)(const void*)&Bt[(col0 + r) * K + k0 + c];
      lptr_t dstB = (lptr_t)(void*)(&lasBt[buf][0][0] + wchunk * 512);
      __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)srcB,
                                       (__attribute__((address_space(3))) void*)dstB,
                                       16, 0, 0);
    }
  };

  // cooperative tile load: 256 threads, A tile 128x64 (8192 elems = 32/thread),
  // 8-wide vector loads -> 4 loads per thread
  auto load_tiles = [&](int kt, int buf) {
    int64_t k0 = (int64_t)kt * PW_GEMM_BK;
    // A: each thread loads 4 rows' 8-element chunks
    for (int it = tid; it < (PW_GEMM_BM * PW_GEMM_BK) / 8; it += 256) {
      int r = it / (PW_GEMM_BK / 8);
      int cchunk = it % (PW_GEMM_BK / 8);
      int64_t grow = row0 + r;
      int64_t gcol = k0 + cchunk * 8;
      __bf16* dst = &lasA[buf][r][cchunk * 8];
      if (grow < M && gcol + 8 <= K) {
        *(pw_frag8*)dst = *(const pw_frag8*)&A[grow * K + gcol];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          dst[j] = (grow < M && gcol + j < K) ? A[grow * K + gcol + j]
                                              : (__bf16)0.f;
      }
    }
    // Bt: tile 128 (N-rows) x 64 (K-cols), linear like A
    for (int it = tid; it < (PW_GEMM_BN * PW_GEMM_BK) / 8; it += 256) {
      int r = it / (PW_GEMM_BK / 8);
      int cchunk = it % (PW_GEMM_BK / 8);
      int64_t grow = col0 + r;       // N index
      int64_t gcol = k0 + cchunk * 8;  // K index
      __bf16* dst = &lasBt[buf][r][cchunk * 8];
      if (grow < N && gcol + 8 <= K) {
        *(pw_frag8*)dst = *(const pw_frag8*)&Bt[grow * K + gcol];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          dst[j] = (grow < N && gcol + j < K) ? Bt[grow * K + gcol + j]
                                              : (__bf16)0.f;
      }
    }
  };

  auto stage = [&](int kt, int buf) {
    if (interior_mn && (int64_t)(kt + 1) * PW_GEMM_BK <= K)
      load_tiles_gll(kt, buf);
    else
      load_tiles(kt, buf);
  };
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  for (int kt = 0; kt < nk; ++kt) {
    int buf = NBUF == 2 ? (kt & 1) : 0;
    if (NBUF == 2 && kt + 1 < nk) stage(kt + 1, buf ^ 1);
#pragma unroll
    for (int kk = 0; kk < PW_GEMM_BK / 32; ++kk) {
      // hoist every fragment read out of the MFMA loops: B re-read per m
      // was 4x the LDS traffic and the dominant bank-conflict source
      // (PMC: LdsBankConflict ratio 3.0 before, MfmaUtil 8.9%)
      int ak = kk * 32 + (lane >> 4) * 8;
      pw_frag8 afrag[4], bfrag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        int arow = wr * 64 + m * 16 + (lane & 15);
        afrag[m] = *(const pw_frag8*)&lasA[buf][arow][ak];
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        int bcol = wc * 64 + n * 16 + (lane & 15);
        bfrag[n] = *(const pw_frag8*)&lasBt[buf][bcol][ak];
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    }
    if (NBUF == 2) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    } else if (kt + 1 < nk) {
      __syncthreads();  // everyone done reading buf before overwrite
      stage(kt + 1, 0);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
  }

  // epilogue: bias + activation, bf16 store
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int64_t col = col0 + wc * 64 + n * 16 + (lane & 15);
      if (col >= N) continue;
      float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int64_t row = row0 + wr * 64 + m * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + bv;
        if (act == 1) v = pw_gelu(v);
        C[row * N + col] = (__bf16)v;
      }
    }
  }
}

extern "C" int pw_gemm_bf16(const void* A, const void* B, const void* bias,
                            void* C, int64_t M, int64_t N, int64_t K,
                            int act, void* stream) {
  int nbm = (int)((M + PW_GEMM_BM - 1) / PW_GEMM_BM);
  int nbn = (int)((N + PW_GEMM_BN - 1) / PW_GEMM_BN);
  hipStream_t s = (hipStream_t)stream;
  // short-K shapes (few BK iterations) run the single-buffer variant:
  // the 2-deep pipeline never fills at nk<=8, so 5-blocks/CU occupancy
  // hides latency better than intra-block double buffering
  if ((K + PW_GEMM_BK - 1) / PW_GEMM_BK <= 8)
    hipLaunchKernelGGL(k_gemm_bf16<1>, dim3((uint32_t)(nbm * nbn)), dim3(256),
                       0, s, (const __bf16*)A, (const __bf16*)B,
                       (const float*)bias, (__bf16*)C, M, N, K, act, nbm);
  else
    hipLaunchKernelGGL(k_gemm_bf16<2>, dim3((uint32_t)(nbm * nbn)), dim3(256),
                       0, s, (const __bf16*)A, (const __bf16*)B,
                       (const float*)bias, (__bf16*)C, M, N, K, act, nbm);
  return (int)hipGetLastError();
}

// ----------------------------------------------------------------- top-k --
// Per-query top-k over a (nq, m) score matrix (higher = better) — the
// KNN answer selection (north-star: "cosine top-k" hand-written;
// replaces torch.topk on the query path).  One block per query; each
// thread keeps a k-deep insertion list over its strided slice, lists are
// merged in LDS by k selection passes.  k <= 32.

#define PW_TOPK_MAXK 32

__global__ __launch_bounds__(256) void k_topk(
    const float* __restrict__ scores, int64_t m, int k,
    float* __restrict__ out_vals, int64_t* __restrict__ out_idx) {
  const float NEG = -3.4e38f;
  __shared__ float svals[256 * PW_TOPK_MAXK];
  __shared__ int sidx[256 * PW_TOPK_MAXK];
  int64_t q = blockIdx.x;
  const float* row = scores + q * m;
  int tid = threadIdx.x;
  float lv[PW_TOPK_MAXK];
  int li[PW_TOPK_MAXK];
#pragma unroll
  for (int j = 0; j < PW_TOPK_MAXK; ++j) {
    lv[j] = NEG;
    li[j] = -1;
  }
  for (int64_t i = tid; i < m; i += 256) {
    float v = row[i];
    if (v > lv[k - 1]) {
      // insertion into the local sorted-descending list
      int p = k - 1;
      while (p > 0 && lv[p - 1] < v) {
        lv[p] = lv[p - 1];
        li[p] = li[p - 1];
        --p;
      }
      lv[p] = v;
      li[p] = (int)i;
    }
  }
  for (int j = 0; j < k; ++j) {
    svals[tid * k + j] = lv[j];
    sidx[tid * k + j] = li[j];
  }
  __syncthreads();
  // selection: k passes; wave 0 finds the global max of the 256*k
  // candidates via wave reduction and invalidates it.  The barrier is
  // block-uniform (every thread executes it each pass).
  for (int sel = 0; sel < k; ++sel) {
    if (tid < 64) {
      float best = NEG;
      int bestpos = -1;
      for (int p = tid; p < 256 * k; p += 64) {
        float v = svals[p];
        if (v > best || (v == best && bestpos >= 0 && p < bestpos)) {
          best = v;
          bestpos = p;
        }
      }
      // wave argmax reduction (ties -> lower position for determinism)
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        float ov = __shfl_down(best, off);
        int op = __shfl_down(bestpos, off);
        if (ov > best || (ov == best && op >= 0 &&
                          (bestpos < 0 || op < bestpos))) {
          best = ov;
          bestpos = op;
        }
      }
      if (tid == 0) {
        if (bestpos >= 0 && best > NEG) {
          out_vals[q * k + sel] = best;
          out_idx[q * k + sel] = sidx[bestpos];
          svals[bestpos] = NEG;
        } else {
          out_vals[q * k + sel] = NEG;
          out_idx[q * k + sel] = -1;
        }
      }
    }
    __syncthreads();
  }
}

extern "C" int pw_topk(const void* scores, int64_t nq, int64_t m, int k,
                       void* out_vals, void* out_idx, void* stream) {
  if (k > PW_TOPK_MAXK || k < 1) return 2;
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_topk, dim3((uint32_t)nq), dim3(256), 0, s,
                     (const float*)scores, m, k, (float*)out_vals,
                     (int64_t*)out_idx);
  return (int)hipGetLastError();
}

// --------------------------------------------------- merge-consolidate --
// Fused LSM merge of the additive-reduce state:
// two UNIQUE lex-sorted 128-bit-key row sets (state A, delta B) with
// int64 accumulators merge into unique sorted keys, accumulators summed
// on key matches, rows whose weight slot (acc 0) sums to zero dropped.
// Replaces the searchsorted-merge + gather + segmented-reduce + nonzero
// + gather chain (~10 kernels) with count + emit over merge-path
// partitions (per-thread diagonal chunks; matched pairs are owned by
// the A side, a range starting on the B half of a match skips it).
// rep[] returns each output row's source row (j in A, or m + i in B;
// A preferred on matches) for carried-column gathers.

struct McAcc {
  const long long* p[8];
};
struct McAccMut {
  long long* p[8];
};
struct McWords {
  const int64_t* p[4];
};
struct McWordsMut {
  int64_t* p[4];
};

// compare the first nwc words (lexicographic, signed)
__device__ __forceinline__ bool mc_le(McWords A, int64_t i, McWords B,
                                      int64_t j, int nwc) {
  for (int w = 0; w < nwc; ++w) {
    if (A.p[w][i] != B.p[w][j]) return A.p[w][i] < B.p[w][j];
  }
  return true;
}

__device__ __forceinline__ bool mc_eq(McWords A, int64_t i, McWords B,
                                      int64_t j, int nwc) {
  for (int w = 0; w < nwc; ++w)
    if (A.p[w][i] != B.p[w][j]) return false;
  return true;
}

__device__ int64_t mc_merge_path(McWords A, int64_t m, McWords B, int64_t n,
                                 int64_t d, int nwc) {
  int64_t lo = d > n ? d - n : 0;
  int64_t hi = d < m ? d : m;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (mc_le(A, mid, B, d - mid - 1, nwc))
      lo = mid + 1;
    else
      hi = mid;
  }
  return lo;  // ai; bi = d - ai
}

#define PW_MC_CHUNK 8

// one walk function used by both phases: WRITE=false counts only.
// nw = words carried to the output; nwc (<= nw) = comparison prefix
// (callers guarantee the tail words are key-determined when nwc < nw).
template <bool WRITE>
__device__ void mc_walk(McWords A, McAcc accA, McWords B, McAcc accB,
                        int nw, int nwc, int nacc, int64_t m, int64_t n,
                        int64_t d0, int64_t d1, int* count_out,
                        int64_t base, McWordsMut O, McAccMut accO,
                        int64_t* rep) {
  int64_t ai = mc_merge_path(A, m, B, n, d0, nwc);
  int64_t bi = d0 - ai;
  int64_t d = d0;
  // skip rule: a range starting on the B half of a match
  if (d < d1 && bi < n && ai > 0 &&
      !(ai < m && mc_le(A, ai, B, bi, nwc))) {
    if (mc_eq(A, ai - 1, B, bi, nwc)) {
      ++bi;
      ++d;
    }
  }
  int cnt = 0;
  int64_t w = base;
  while (d < d1) {
    bool take_a = (bi >= n) || (ai < m && mc_le(A, ai, B, bi, nwc));
    if (take_a) {
      bool matched = (bi < n) && mc_eq(A, ai, B, bi, nwc);
      long long wsum = accA.p[0][ai] + (matched ? accB.p[0][bi] : 0ll);
      if (wsum != 0) {
        if (WRITE) {
          for (int k = 0; k < nw; ++k) O.p[k][w] = A.p[k][ai];
          for (int c = 0; c < nacc; ++c)
            accO.p[c][w] =
                accA.p[c][ai] + (matched ? accB.p[c][bi] : 0ll);
          rep[w] = ai;
          ++w;
        }
        ++cnt;
      }
      ++ai;
      ++d;
      if (matched) {
        ++bi;
        ++d;
      }
    } else {
      long long wsum = accB.p[0][bi];
      if (wsum != 0) {
        if (WRITE) {
          for (int k = 0; k < nw; ++k) O.p[k][w] = B.p[k][bi];
          for (int c = 0; c < nacc; ++c) accO.p[c][w] = accB.p[c][bi];
          rep[w] = m + bi;
          ++w;
        }
        ++cnt;
      }
      ++bi;
      ++d;
    }
  }
  if (!WRITE) *count_out = cnt;
}

__global__ void k_mc_count(McWords A, McAcc accA, McWords B, McAcc accB,
                           int nw, int nwc, int nacc, int64_t m, int64_t n,
                           int* thread_counts, int64_t nthreads) {
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= nthreads) return;
  int64_t total = m + n;
  int64_t d0 = t * PW_MC_CHUNK;
  int64_t d1 = min(d0 + PW_MC_CHUNK, total);
  McAccMut dummy{};
  McWordsMut wdummy{};
  mc_walk<false>(A, accA, B, accB, nw, nwc, nacc, m, n, d0, d1,
                 &thread_counts[t], 0, wdummy, dummy, nullptr);
}

__global__ void k_mc_emit(McWords A, McAcc accA, McWords B, McAcc accB,
                          int nw, int nwc, int nacc, int64_t m, int64_t n,
                          const int64_t* bases, int64_t nthreads,
                          McWordsMut O, McAccMut accO, int64_t* rep) {
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= nthreads) return;
  int64_t total = m + n;
  int64_t d0 = t * PW_MC_CHUNK;
  int64_t d1 = min(d0 + PW_MC_CHUNK, total);
  mc_walk<true>(A, accA, B, accB, nw, nwc, nacc, m, n, d0, d1, nullptr,
                bases[t], O, accO, rep);
}

extern "C" int pw_merge_consolidate_count(
    const void** wordsA, const void** accA, const void** wordsB,
    const void** accB, int nw, int nwc, int nacc, int64_t m, int64_t n,
    void* thread_counts, int64_t nthreads, void* stream) {
  if (nacc > 8 || nacc < 1 || nw > 4 || nwc > nw) return 2;
  McAcc A{}, B{};
  McWords WA{}, WB{};
  for (int c = 0; c < nacc; ++c) {
    A.p[c] = (const long long*)accA[c];
    B.p[c] = (const long long*)accB[c];
  }
  for (int k = 0; k < nw; ++k) {
    WA.p[k] = (const int64_t*)wordsA[k];
    WB.p[k] = (const int64_t*)wordsB[k];
  }
  hipStream_t s = (hipStream_t)stream;
  int64_t blocks = (nthreads + PW_BLOCK - 1) / PW_BLOCK;
  hipLaunchKernelGGL(k_mc_count, dim3((uint32_t)blocks), dim3(PW_BLOCK), 0,
                     s, WA, A, WB, B, nw, nwc, nacc, m, n,
                     (int*)thread_counts, nthreads);
  return (int)hipGetLastError();
}

extern "C" int pw_merge_consolidate_emit(
    const void** wordsA, const void** accA, const void** wordsB,
    const void** accB, int nw, int nwc, int nacc, int64_t m, int64_t n,
    const void* bases, int64_t nthreads, void** out_words, void** accO,
    void* rep, void* stream) {
  if (nacc > 8 || nacc < 1 || nw > 4 || nwc > nw) return 2;
  McAcc A{}, B{};
  McAccMut O{};
  McWords WA{}, WB{};
  McWordsMut WO{};
  for (int c = 0; c < nacc; ++c) {
    A.p[c] = (const long long*)accA[c];
    B.p[c] = (const long long*)accB[c];
    O.p[c] = (long long*)accO[c];
  }
  for (int k = 0; k < nw; ++k) {
    WA.p[k] = (const int64_t*)wordsA[k];
    WB.p[k] = (const int64_t*)wordsB[k];
    WO.p[k] = (int64_t*)out_words[k];
  }
  hipStream_t s = (hipStream_t)stream;
  int64_t blocks = (nthreads + PW_BLOCK - 1) / PW_BLOCK;
  hipLaunchKernelGGL(k_mc_emit, dim3((uint32_t)blocks), dim3(PW_BLOCK), 0,
                     s, WA, A, WB, B, nw, nwc, nacc, m, n,
                     (const int64_t*)bases, nthreads, WO, O, (int64_t*)rep);
  return (int)hipGetLastError();
}

// -------------------------------------------------------------- radix sort --
// Hand-written LSD radix sort of int64 keys with an int64 payload (the
// permutation) — the mandated sort kernel for the 128-bit key path
// (lex_sort_words' primary-word sort; the k1 tiebreak remains the
// existing duplicate-check).  Signed order = unsigned order after the
// sign bit is flipped once on input (flipped back on output).
//
// Per 8-bit digit pass:
//   k_rs_count: one wave per block, each lane owns a contiguous chunk
//     and counts serially (stable) into its LDS row; rows reduce to
//     per-(block,digit) counts, laid out digit-major for the scan.
//   <scan over (digit, block) runs in torch: one cumsum>
//   k_rs_scatter: lanes re-walk their chunks; position = scanned base
//     for (digit, block) + prefix of earlier lanes' counts (serial
//     per-digit LDS prefix) + running in-lane count.

#define PW_RS_THREADS 64
#define PW_RS_DIGITS 256

__global__ __launch_bounds__(PW_RS_THREADS) void k_rs_flip(
    const int64_t* in, uint64_t* out, int64_t n, int dir) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (uint64_t)in[i] ^ 0x8000000000000000ull;
}

__global__ __launch_bounds__(PW_RS_THREADS) void k_rs_count(
    const uint64_t* keys, int64_t n, int shift, int64_t chunk,
    int* block_digit_counts /* [DIGITS][nblocks] digit-major */,
    int64_t nblocks) {
  __shared__ int cnt[PW_RS_THREADS][PW_RS_DIGITS];
  int lane = threadIdx.x;
  for (int d = 0; d < PW_RS_DIGITS; ++d) cnt[lane][d] = 0;
  int64_t start = ((int64_t)blockIdx.x * PW_RS_THREADS + lane) * chunk;
  int64_t end = min(start + chunk, n);
  for (int64_t i = start; i < end; ++i) {
    int d = (int)((keys[i] >> shift) & 0xFF);
    ++cnt[lane][d];
  }
  __syncthreads();
  // reduce lanes per digit: lane handles digits lane, lane+64, ...
  for (int d = lane; d < PW_RS_DIGITS; d += PW_RS_THREADS) {
    int s = 0;
    for (int t = 0; t < PW_RS_THREADS; ++t) s += cnt[t][d];
    block_digit_counts[(int64_t)d * nblocks + blockIdx.x] = s;
  }
}

__global__ __launch_bounds__(PW_RS_THREADS) void k_rs_scatter(
    const uint64_t* keys, const int64_t* payload, int64_t n, int shift,
    int64_t chunk, const long long* bases /* [DIGITS][nblocks] excl */,
    int64_t nblocks, uint64_t* out_keys, int64_t* out_payload) {
  // counts accumulate into lane_base, then convert to exclusive prefix
  // in place (one LDS array keeps us under the 160 KB/CU limit)
  __shared__ long long lane_base[PW_RS_THREADS][PW_RS_DIGITS];
  int lane = threadIdx.x;
  for (int d = 0; d < PW_RS_DIGITS; ++d) lane_base[lane][d] = 0;
  int64_t start = ((int64_t)blockIdx.x * PW_RS_THREADS + lane) * chunk;
  int64_t end = min(start + chunk, n);
  for (int64_t i = start; i < end; ++i) {
    int d = (int)((keys[i] >> shift) & 0xFF);
    ++lane_base[lane][d];
  }
  __syncthreads();
  // serial exclusive prefix over lanes per digit (lane d' handles
  // digits d', d'+64, ...)
  for (int d = lane; d < PW_RS_DIGITS; d += PW_RS_THREADS) {
    long long run = bases[(int64_t)d * nblocks + blockIdx.x];
    for (int t = 0; t < PW_RS_THREADS; ++t) {
      long long c = lane_base[t][d];
      lane_base[t][d] = run;
      run += c;
    }
  }
  __syncthreads();
  for (int64_t i = start; i < end; ++i) {
    int d = (int)((keys[i] >> shift) & 0xFF);
    long long p = lane_base[lane][d]++;
    out_keys[p] = keys[i];
    out_payload[p] = payload[i];
  }
}

__global__ __launch_bounds__(PW_RS_THREADS) void k_rs_unflip(
    const uint64_t* in, int64_t* out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int64_t)(in[i] ^ 0x8000000000000000ull);
}

extern "C" int pw_radix_flip(const void* in, void* out, int64_t n,
                             void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int64_t blocks = min((int64_t)2048, (n + PW_RS_THREADS - 1) / PW_RS_THREADS);
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_rs_flip, dim3((uint32_t)blocks), dim3(PW_RS_THREADS),
                     0, s, (const int64_t*)in, (uint64_t*)out, n, 0);
  return (int)hipGetLastError();
}

extern "C" int pw_radix_unflip(const void* in, void* out, int64_t n,
                               void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int64_t blocks = min((int64_t)2048, (n + PW_RS_THREADS - 1) / PW_RS_THREADS);
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_rs_unflip, dim3((uint32_t)blocks),
                     dim3(PW_RS_THREADS), 0, s, (const uint64_t*)in,
                     (int64_t*)out, n);
  return (int)hipGetLastError();
}

extern "C" int pw_radix_count(const void* keys, int64_t n, int shift,
                              int64_t nblocks, int64_t chunk, void* counts,
                              void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_rs_count, dim3((uint32_t)nblocks),
                     dim3(PW_RS_THREADS), 0, s, (const uint64_t*)keys, n,
                     shift, chunk, (int*)counts, nblocks);
  return (int)hipGetLastError();
}

extern "C" int pw_radix_scatter(const void* keys, const void* payload,
                                int64_t n, int shift, int64_t nblocks,
                                int64_t chunk, const void* bases,
                                void* out_keys, void* out_payload,
                                void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_rs_scatter, dim3((uint32_t)nblocks),
                     dim3(PW_RS_THREADS), 0, s, (const uint64_t*)keys,
                     (const int64_t*)payload, n, shift, chunk,
                     (const long long*)bases, nblocks, (uint64_t*)out_keys,
                     (int64_t*)out_payload);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------ mask scan --
// Ordered positions of matching bytes / true mask entries — the newline
// scan of the ingest parse and the run-start extraction, replacing
// rocprim partition (torch nonzero) with a count + emit pair whose scan
// is one torch cumsum.

// 16-bit hit mask of `target` inside the 16-byte granule at byte offset g.
// Vector path requires 16B alignment of buf+g (the host wrapper rounds the
// per-block chunk to a multiple of 16 and torch allocations are 256B-aligned,
// so interior granules qualify); the tail / unaligned path is scalar.
__device__ __forceinline__ unsigned pw_granule_mask(const uint8_t* buf,
                                                    int64_t g, int64_t end,
                                                    uint8_t t, bool aligned) {
  unsigned m = 0;
  if (aligned && g + 16 <= end) {
    uint4 v = *(const uint4*)(buf + g);
    const uint32_t w[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
    for (int q = 0; q < 4; ++q)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        if (((w[q] >> (8 * j)) & 0xFFu) == t) m |= 1u << (4 * q + j);
  } else {
    for (int j = 0; j < 16 && g + j < end; ++j)
      if (buf[g + j] == t) m |= 1u << j;
  }
  return m;
}

__global__ void k_scan_count(const uint8_t* buf, int64_t n, int target,
                             int64_t chunk, long long* block_counts) {
  __shared__ int cnt;
  if (threadIdx.x == 0) cnt = 0;
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t end = min(start + chunk, n);
  const bool aligned = (((size_t)buf & 15) == 0);
  int local = 0;
  for (int64_t tile = start; tile < end; tile += (int64_t)blockDim.x * 16) {
    int64_t g = tile + (int64_t)threadIdx.x * 16;
    if (g < end)
      local += __popc(pw_granule_mask(buf, g, end, (uint8_t)target, aligned));
  }
  atomicAdd(&cnt, local);
  __syncthreads();
  if (threadIdx.x == 0) block_counts[blockIdx.x] = cnt;
}

__global__ void k_scan_emit(const uint8_t* buf, int64_t n, int target,
                            int64_t chunk, const long long* bases,
                            int64_t* out) {
  // Ordered emit: each 256-thread block walks its chunk in 4 KB tiles
  // (one 16 B granule per lane).  Per tile: wave-level __shfl_up inclusive
  // scan of per-granule hit counts + a 4-entry cross-wave prefix in LDS —
  // 2 barriers per 4 KB instead of a 16-barrier log-scan per 256 B.
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  __shared__ long long cursor;
  __shared__ int wave_tot[PW_BLOCK / 64];
  if (threadIdx.x == 0) cursor = bases[blockIdx.x];
  __syncthreads();
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t end = min(start + chunk, n);
  const bool aligned = (((size_t)buf & 15) == 0);
  for (int64_t tile = start; tile < end; tile += (int64_t)blockDim.x * 16) {
    int64_t g = tile + (int64_t)threadIdx.x * 16;
    unsigned hmask =
        (g < end) ? pw_granule_mask(buf, g, end, (uint8_t)target, aligned) : 0u;
    int hits = __popc(hmask);
    int incl = hits;
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      int v = __shfl_up(incl, off, 64);
      if (lane >= off) incl += v;
    }
    if (lane == 63) wave_tot[wave] = incl;
    __syncthreads();
    int wbase = 0;
    for (int w = 0; w < wave; ++w) wbase += wave_tot[w];
    long long my = cursor + wbase + (incl - hits);
    while (hmask) {
      int j = __ffs(hmask) - 1;
      hmask &= hmask - 1;
      out[my++] = g + j;
    }
    __syncthreads();
    if (threadIdx.x == (int)blockDim.x - 1) cursor += wbase + incl;
    __syncthreads();
  }
}

extern "C" int pw_scan_positions(const void* buf, int64_t n, int target,
                                 void* block_counts, int64_t nblocks,
                                 void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int64_t chunk = ((n + nblocks - 1) / nblocks + 15) & ~(int64_t)15;
  if (chunk < 16) chunk = 16;
  hipLaunchKernelGGL(k_scan_count, dim3((uint32_t)nblocks), dim3(PW_BLOCK),
                     0, s, (const uint8_t*)buf, n, target,
                     chunk, (long long*)block_counts);
  return (int)hipGetLastError();
}

extern "C" int pw_scan_emit(const void* buf, int64_t n, int target,
                            const void* bases, int64_t nblocks, void* out,
                            void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int64_t chunk = ((n + nblocks - 1) / nblocks + 15) & ~(int64_t)15;
  if (chunk < 16) chunk = 16;
  hipLaunchKernelGGL(k_scan_emit, dim3((uint32_t)nblocks), dim3(PW_BLOCK),
                     0, s, (const uint8_t*)buf, n, target, chunk,
                     (const long long*)bases, (int64_t*)out);
  return (int)hipGetLastError();
}

// ------------------------------------------------------- sort repair --
// After the 2-word fast-path sort by word0 only, rows in an equal-word0
// run may be out of order on word1.  Host-sync-checking for collisions
// cost 0.27 ms/step (10 syncs); instead: odd-even transposition passes
// restricted to equal-k0 pairs, entirely on device.  `passes` bounds the
// repairable run length; 64-bit hash keys make runs >8 over any real
// row count astronomically improbable (n^9 / 2^512).  Signed compare
// matches torch.sort / the multi-pass fallback.

__global__ void k_sort_repair(const long long* __restrict__ k0,
                              long long* k1, long long* perm, int64_t n,
                              int parity) {
  int64_t i = 2 * ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) + parity;
  if (i + 1 >= n) return;
  if (k0[i] != k0[i + 1]) return;
  long long a = k1[i], b = k1[i + 1];
  if (a > b) {
    k1[i] = b;
    k1[i + 1] = a;
    long long p = perm[i];
    perm[i] = perm[i + 1];
    perm[i + 1] = p;
  }
}

extern "C" int pw_sort_repair(const void* k0, void* k1, void* perm,
                              int64_t n, int passes, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int64_t pairs = n / 2 + 1;
  int64_t nblocks = (pairs + PW_BLOCK - 1) / PW_BLOCK;
  if (nblocks < 1) nblocks = 1;
  for (int p = 0; p < passes; ++p)
    hipLaunchKernelGGL(k_sort_repair, dim3((uint32_t)nblocks), dim3(PW_BLOCK),
                       0, s, (const long long*)k0, (long long*)k1,
                       (long long*)perm, n, p & 1);
  return (int)hipGetLastError();
}

// ------------------------------------------------------- fused gather --
// One launch gathers up to 8 8-byte columns through a shared index: the
// arrange/merge/consolidate paths gather (key words, weights, value
// columns) with the same permutation, and per-column index_select pays
// the index read + launch overhead k times.  Pointer table passed by
// value in the kernel args.

struct PwGather8 {
  const long long* src[8];
  long long* dst[8];
};

__global__ void k_gather_cols(const long long* __restrict__ idx, int64_t m,
                              int ncols, PwGather8 p) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < m; i += stride) {
    long long j = idx[i];
#pragma unroll 8
    for (int c = 0; c < 8; ++c) {
      if (c >= ncols) break;
      p.dst[c][i] = p.src[c][j];
    }
  }
}

// ------------------------------------------------------ hash table --
// Open-addressing (linear probe) device hash table over 128-bit keys —
// O(1) replacement for the binary-search k_lookup on hot stable
// dictionaries (the string-pool decode probes 4M tokens/step against a
// 50k vocab: ~1.2 slot loads beat a 16-level dependent-load search).
// Values are int64 >= 0; slot value -1 means empty.  Build never reads
// keys (inputs are unique), so a single atomicCAS on the value slot
// claims it.

__global__ void k_ht_build(const int64_t* __restrict__ klo,
                           const int64_t* __restrict__ khi,
                           const int64_t* __restrict__ vals, int64_t m,
                           int64_t* tab_lo, int64_t* tab_hi,
                           long long* tab_val, int64_t mask) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < m; i += stride) {
    uint64_t slot = (uint64_t)klo[i] & (uint64_t)mask;
    while (atomicCAS((unsigned long long*)&tab_val[slot],
                     (unsigned long long)(long long)-1,
                     (unsigned long long)(long long)(vals ? vals[i] : i)) !=
           (unsigned long long)(long long)-1)
      slot = (slot + 1) & (uint64_t)mask;
    tab_lo[slot] = klo[i];
    tab_hi[slot] = khi[i];
  }
}

__global__ void k_ht_probe(const int64_t* __restrict__ qlo,
                           const int64_t* __restrict__ qhi, int64_t nq,
                           const int64_t* __restrict__ tab_lo,
                           const int64_t* __restrict__ tab_hi,
                           const long long* __restrict__ tab_val,
                           int64_t mask, int64_t* out, bool* found) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nq; i += stride) {
    int64_t lo = qlo[i], hi = qhi[i];
    uint64_t slot = (uint64_t)lo & (uint64_t)mask;
    int64_t v = -1;
    for (;;) {
      long long tv = tab_val[slot];
      if (tv == -1) break;  // empty -> miss
      if (tab_lo[slot] == lo && tab_hi[slot] == hi) {
        v = (int64_t)tv;
        break;
      }
      slot = (slot + 1) & (uint64_t)mask;
    }
    out[i] = v;
    found[i] = v != -1;
  }
}

extern "C" int pw_ht_build(const void* klo, const void* khi, const void* vals,
                           int64_t m, void* tab_lo, void* tab_hi,
                           void* tab_val, int64_t nslots, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int64_t nblocks = (m + PW_BLOCK - 1) / PW_BLOCK;
  if (nblocks > 4096) nblocks = 4096;
  if (nblocks < 1) nblocks = 1;
  hipLaunchKernelGGL(k_ht_build, dim3((uint32_t)nblocks), dim3(PW_BLOCK), 0, s,
                     (const int64_t*)klo, (const int64_t*)khi,
                     (const int64_t*)vals, m, (int64_t*)tab_lo,
                     (int64_t*)tab_hi, (long long*)tab_val, nslots - 1);
  return (int)hipGetLastError();
}

extern "C" int pw_ht_probe(const void* qlo, const void* qhi, int64_t nq,
                           const void* tab_lo, const void* tab_hi,
                           const void* tab_val, int64_t nslots, void* out,
                           void* found, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int64_t nblocks = (nq + PW_BLOCK - 1) / PW_BLOCK;
  if (nblocks > 4096) nblocks = 4096;
  if (nblocks < 1) nblocks = 1;
  hipLaunchKernelGGL(k_ht_probe, dim3((uint32_t)nblocks), dim3(PW_BLOCK), 0, s,
                     (const int64_t*)qlo, (const int64_t*)qhi, nq,
                     (const int64_t*)tab_lo, (const int64_t*)tab_hi,
                     (const long long*)tab_val, nslots - 1, (int64_t*)out,
                     (bool*)found);
  return (int)hipGetLastError();
}

extern "C" int pw_gather_cols(const void* idx, int64_t m, int ncols,
                              void* const* srcs, void* const* dsts,
                              void* stream) {
  if (ncols < 1 || ncols > 8) return -2;
  hipStream_t s = (hipStream_t)stream;
  PwGather8 p;
  for (int c = 0; c < ncols; ++c) {
    p.src[c] = (const long long*)srcs[c];
    p.dst[c] = (long long*)dsts[c];
  }
  int64_t nblocks = (m + PW_BLOCK - 1) / PW_BLOCK;
  if (nblocks > 4096) nblocks = 4096;
  if (nblocks < 1) nblocks = 1;
  hipLaunchKernelGGL(k_gather_cols, dim3((uint32_t)nblocks), dim3(PW_BLOCK),
                     0, s, (const long long*)idx, m, ncols, p);
  return (int)hipGetLastError();
}
