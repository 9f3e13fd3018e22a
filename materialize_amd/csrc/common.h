// common.h — shared host/device types of the MI355X engine (product code).
//
// Device-resident arrangement layout follows the reference's RowRowSpine
// columnar batches (src/row-spine/src/lib.rs:56-135; field structure visible
// at src/compute/src/extensions/arrange.rs:378-383): keys / per-key val
// ranges / vals / per-val update ranges / times / diffs — SoA in HBM, plus
// an open-addressing hash index (the MI355X-native replacement for cursor
// seeks: one ~128B line per probe instead of pointer-chasing a trie).
#pragma once
#include <hip/hip_runtime.h>

#include <cstdint>

using u8 = uint8_t;
using u32 = uint32_t;
using u64 = uint64_t;
using i64 = int64_t;
using i128 = __int128;
using u128 = unsigned __int128;

#define HIP_CHECK(x)                                                      \
  do {                                                                    \
    hipError_t err_ = (x);                                                \
    if (err_ != hipSuccess) {                                             \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(err_), \
              __FILE__, __LINE__);                                        \
      abort();                                                            \
    }                                                                     \
  } while (0)

static constexpr u32 EMPTY_SLOT = 0xFFFFFFFFu;
static constexpr int MAX_KW = 2;
static constexpr int MAX_VB = 64;

// One sealed device batch (immutable once built).
struct DevBatch {
  u64 *keys = nullptr;    // [kw * n_keys], sorted (i64-tuple order)
  u32 *kv_off = nullptr;  // [n_keys + 1]
  u8 *vals = nullptr;     // [vb * n_vals], sorted (byte-lex) within key
  u32 *vu_off = nullptr;  // [n_vals + 1]
  u32 *v_offs = nullptr;  // [n_vals + 1] VARLEN val offsets into vals
  u32 *val_key = nullptr; // [n_vals] -> owning key index
  u64 *times = nullptr;   // [n_upds]
  i64 *diffs = nullptr;   // [n_upds]
  u32 *upd_val = nullptr; // [n_upds] -> owning val index
  // hash index: slot = { u64 k[kw]; u32 idx; } packed as kw+1 u64 words
  // (idx in the low 32 bits of the last word)
  u64 *hash = nullptr;    // [(kw + 1) * hash_slots]
  u64 n_keys = 0, n_vals = 0, n_upds = 0, hash_slots = 0;
  u64 lower = 0, upper = 0;
};

struct DevSchema {
  u32 kw;
  u32 vb;
};

// Routing/probe hash: splitmix64 over the key words — substituted for the
// reference's fixed-seed ahash (timely-util/src/hash.rs:33) identically on
// oracle and GPU. Must match oracle.cpp::orc_route_hash.
__host__ __device__ inline u64 route_hash(const u64 *kwords, u32 n) {
  u64 h = 0x9E3779B97F4A7C15ULL;
  for (u32 i = 0; i < n; i++) {
    u64 x = kwords[i] + h;
    x ^= x >> 30;
    x *= 0xBF58476D1CE4E5B9ULL;
    x ^= x >> 27;
    x *= 0x94D049BB133111EBULL;
    x ^= x >> 31;
    h = x;
  }
  return h;
}
