// bg_ahash.h — restatement of the row-hash Ballista's sort-shuffle uses for
// hash repartitioning, shared verbatim between the CPU oracle (compiled with
// gcc) and the HIP kernels (compiled with hipcc for gfx950) so that host and
// device partition assignment are bit-identical by construction.
//
// What it restates (reference citations):
//   * `compute_partition_indices` evaluates the partitioning key expressions,
//     calls DataFusion's `create_hashes(&arrays, REPARTITION_RANDOM_STATE
//     .random_state(), hash_buffer)` and assigns row -> partition
//     `(h % num_partitions)`:
//     /root/reference/ballista/core/src/execution_plans/sort_shuffle/writer.rs:1259-1279
//     (imports :51 `create_hashes`, :62 `REPARTITION_RANDOM_STATE`).
//   * DataFusion 55 (git tag 55.0.0-rc3, commit d5552342 — pinned by
//     /root/reference/Cargo.toml:71-79 + Cargo.lock) implements
//     `create_hashes` in `datafusion-common/src/hash_utils.rs` on top of
//     `ahash::RandomState` (ahash pinned at 0.8.12 in
//     /root/reference/Cargo.lock).  That crate source is NOT vendored under
//     /root/reference, so the items below are restated from the published
//     ahash 0.8.x fallback algorithm and DataFusion's published hash_utils:
//       - ahash fallback hasher (fallback_hash.rs): folded_multiply update,
//         large_update, finish.
//       - `RandomState::with_seeds(0,0,0,0)` with the PI2 xor constants.
//       - DataFusion `combine_hashes(l, r) = ((17*37 + l) * 37) + r` for
//         multi-column keys, first column hashed directly, null slots left
//         untouched in the hash buffer.
//
// PARITY STATUS: **parity unpinned** (SURVEY.md §8c).  The exact seed values
// inside DataFusion's `REPARTITION_RANDOM_STATE`, the key-word order in
// `AHasher::from_random_state`, and whether the build selects ahash's AES
// path are not observable in this container (no Rust toolchain, no vendored
// crates, no network).  Choices taken here, each to be re-verified against
// the pinned DataFusion commit when a Rust toolchain is available:
//   (1) seeds (0,0,0,0)          — DataFusion's repartition random state
//   (2) with_seeds xors PI2      — ahash 0.8.x `RandomState::with_seeds`
//   (3) buffer=k1, pad=k0        — ahash 0.8.x fallback `from_random_state`
//       initializes buffer from k1 and pad from k0 (the SWAPPED order,
//       unlike `new_with_keys`; flagged by round-1 review, changed r2)
//   (4) fallback (non-AES) path  — default rustc target has no +aes
// Everything downstream of partition IDs (query results, file layout,
// row conservation) is invariant to these choices and is pinned by the
// reference's own golden vectors (see tests/golden/).  GPU<->oracle partition
// IDs are bit-exact regardless, because both compile THIS header.

#ifndef BG_AHASH_H
#define BG_AHASH_H

#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define BG_HD __host__ __device__ __forceinline__
#else
#define BG_HD static inline
#endif

#ifdef __cplusplus
#define BG_INLINE_LINKAGE inline
#else
#define BG_INLINE_LINKAGE static inline
#endif

// ahash 0.8.x fallback constants (restated; see header comment).
#define BG_AHASH_MULTIPLE 6364136223846793005ULL
#define BG_AHASH_ROT 23
// ahash PI2 constants, xored by RandomState::with_seeds.
#define BG_AHASH_PI2_0 0x452821e638d01377ULL
#define BG_AHASH_PI2_1 0xbe5466cf34e90c6cULL
#define BG_AHASH_PI2_2 0xc0ac29b7c97c50ddULL
#define BG_AHASH_PI2_3 0x3f84d5b5b5470917ULL

// RandomState::with_seeds(0,0,0,0): k_i = 0 ^ PI2[i].
#define BG_AHASH_K0 BG_AHASH_PI2_0
#define BG_AHASH_K1 BG_AHASH_PI2_1
#define BG_AHASH_K2 BG_AHASH_PI2_2
#define BG_AHASH_K3 BG_AHASH_PI2_3

typedef struct {
  uint64_t buffer;  // = k1 at init (from_random_state's swapped order)
  uint64_t pad;     // = k0 at init
  uint64_t extra0;  // = k2
  uint64_t extra1;  // = k3
} bg_ahasher;

BG_HD uint64_t bg_rotl64(uint64_t x, unsigned r) {
  r &= 63u;
  return r ? ((x << r) | (x >> (64u - r))) : x;
}

// folded_multiply: 128-bit product, xor the halves.
BG_HD uint64_t bg_folded_multiply(uint64_t s, uint64_t by) {
#if defined(__SIZEOF_INT128__)
  unsigned __int128 r = (unsigned __int128)s * (unsigned __int128)by;
  return (uint64_t)r ^ (uint64_t)(r >> 64);
#else
#error "need __int128"
#endif
}

BG_HD bg_ahasher bg_ahasher_init(void) {
  bg_ahasher h;
  // ahash 0.8.x fallback AHasher::from_random_state: buffer <- k1,
  // pad <- k0 (swapped relative to new_with_keys).
  h.buffer = BG_AHASH_K1;
  h.pad = BG_AHASH_K0;
  h.extra0 = BG_AHASH_K2;
  h.extra1 = BG_AHASH_K3;
  return h;
}

// AHasher::update — all of write_u8/u16/u32/u64 funnel here (values
// zero-extended to u64 by the narrower write_* impls).
BG_HD void bg_ahasher_update(bg_ahasher* h, uint64_t d) {
  h->buffer = bg_folded_multiply(d ^ h->buffer, BG_AHASH_MULTIPLE);
}

// AHasher::large_update — write_u128 and the bulk of write().
BG_HD void bg_ahasher_large_update(bg_ahasher* h, uint64_t lo, uint64_t hi) {
  uint64_t combined = bg_folded_multiply(lo ^ h->extra0, hi ^ h->extra1);
  h->buffer = bg_rotl64((h->buffer + h->pad) ^ combined, BG_AHASH_ROT);
}

// AHasher::finish.
BG_HD uint64_t bg_ahasher_finish(const bg_ahasher* h) {
  unsigned rot = (unsigned)(h->buffer & 63u);
  return bg_rotl64(bg_folded_multiply(h->buffer, h->pad), rot);
}

// ---- Typed one-shot hashes, mirroring RandomState::hash_one(value) with the
// ---- Rust `Hash` impl for each Arrow native type DataFusion hashes.

// Int64 / UInt64 keys: Hash for i64 -> write_u64(bit pattern).
BG_HD uint64_t bg_hash_u64(uint64_t v) {
  bg_ahasher h = bg_ahasher_init();
  bg_ahasher_update(&h, v);
  return bg_ahasher_finish(&h);
}

// Int32 / Date32 keys: Hash for i32 -> write_u32 -> zero-extended update.
BG_HD uint64_t bg_hash_u32(uint32_t v) {
  bg_ahasher h = bg_ahasher_init();
  bg_ahasher_update(&h, (uint64_t)v);
  return bg_ahasher_finish(&h);
}

// Decimal128 keys: Hash for i128 -> write_u128 -> large_update(lo, hi).
BG_HD uint64_t bg_hash_u128(uint64_t lo, uint64_t hi) {
  bg_ahasher h = bg_ahasher_init();
  bg_ahasher_large_update(&h, lo, hi);
  return bg_ahasher_finish(&h);
}

// helper: LE load of up to 8 bytes
BG_HD uint64_t bg_load_le(const uint8_t* p, int n) {
  uint64_t v = 0;
  for (int i = 0; i < n; ++i) v |= ((uint64_t)p[i]) << (8 * i);
  return v;
}

// AHasher::write(&[u8]) — restated from ahash 0.8.x fallback.
BG_HD void bg_ahasher_write(bg_ahasher* h, const uint8_t* data, uint64_t len) {
  h->buffer = (h->buffer + len) * BG_AHASH_MULTIPLE;
  if (len > 8) {
    if (len > 16) {
      // tail 16 bytes first, then leading 16-byte blocks while len > 16
      const uint8_t* tail = data + len - 16;
      bg_ahasher_large_update(h, bg_load_le(tail, 8), bg_load_le(tail + 8, 8));
      uint64_t remaining = len;
      const uint8_t* p = data;
      while (remaining > 16) {
        bg_ahasher_large_update(h, bg_load_le(p, 8), bg_load_le(p + 8, 8));
        p += 16;
        remaining -= 16;
      }
    } else {
      bg_ahasher_large_update(h, bg_load_le(data, 8),
                              bg_load_le(data + len - 8, 8));
    }
  } else {
    uint64_t lo, hi;
    if (len >= 2) {
      if (len >= 4) {
        lo = bg_load_le(data, 4);
        hi = bg_load_le(data + len - 4, 4);
      } else {
        lo = bg_load_le(data, 2);
        hi = (uint64_t)data[len - 1];
      }
    } else if (len > 0) {
      lo = (uint64_t)data[0];
      hi = (uint64_t)data[0];
    } else {
      lo = 0;
      hi = 0;
    }
    bg_ahasher_large_update(h, lo, hi);
  }
}

// Utf8 keys: Rust `Hash for str` = write(bytes) then write_u8(0xff).
BG_HD uint64_t bg_hash_str(const uint8_t* data, uint64_t len) {
  bg_ahasher h = bg_ahasher_init();
  bg_ahasher_write(&h, data, len);
  bg_ahasher_update(&h, 0xffULL);
  return bg_ahasher_finish(&h);
}

// DataFusion combine_hashes (hash_utils.rs): multi-column keys fold column
// c's per-row hash into the running hash of columns 0..c-1.
BG_HD uint64_t bg_combine_hashes(uint64_t l, uint64_t r) {
  uint64_t hash = (uint64_t)(17u * 37u) + l;
  return hash * 37u + r;
}

#undef BG_HD
#endif  // BG_AHASH_H
