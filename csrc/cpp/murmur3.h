// Spark-compatible murmur3-32 (seed 42) — shared host/device header.
// Bit-exact with the reference's rust/lakesoul-io/src/utils/hash/
// (spark_murmur3.rs; typed rules mod.rs:43-133) and with Spark's hash().
// Used by the CPU path (module.cc) and the HIP kernel (csrc/hip/hash.hip).
#pragma once

#include <cstdint>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define LS_HD __host__ __device__
#else
#define LS_HD
#endif

namespace lakesoul {

constexpr uint32_t kHashSeed = 42;

LS_HD inline uint32_t mur_rotl(uint32_t x, int r) {
  return (x << r) | (x >> (32 - r));
}

LS_HD inline uint32_t mur_mix_k(uint32_t k) {
  k *= 0xCC9E2D51u;
  k = mur_rotl(k, 15);
  k *= 0x1B873593u;
  return k;
}

LS_HD inline uint32_t mur_mix_h(uint32_t h, uint32_t k) {
  h ^= k;
  h = mur_rotl(h, 13);
  h = h * 5u + 0xE6546B64u;
  return h;
}

LS_HD inline uint32_t mur_finish(uint32_t h, uint32_t nbytes) {
  h ^= nbytes;
  h ^= h >> 16;
  h *= 0x85EBCA6Bu;
  h ^= h >> 13;
  h *= 0xC2B2AE35u;
  h ^= h >> 16;
  return h;
}

// 4-byte value (bool/int8/int16/int32 sign-extended to 32-bit; f32 bits)
LS_HD inline uint32_t spark_hash_u32(uint32_t w, uint32_t seed) {
  return mur_finish(mur_mix_h(seed, mur_mix_k(w)), 4);
}

// 8-byte value (int64 / f64 bits), low word first (little-endian)
LS_HD inline uint32_t spark_hash_u64(uint64_t v, uint32_t seed) {
  uint32_t h = seed;
  h = mur_mix_h(h, mur_mix_k((uint32_t)(v & 0xFFFFFFFFu)));
  h = mur_mix_h(h, mur_mix_k((uint32_t)(v >> 32)));
  return mur_finish(h, 8);
}

LS_HD inline uint32_t spark_hash_f32(float f, uint32_t seed) {
  uint32_t bits;
  __builtin_memcpy(&bits, &f, 4);
  if (bits == 0x80000000u) bits = 0;  // -0.0 -> 0
  return spark_hash_u32(bits, seed);
}

LS_HD inline uint32_t spark_hash_f64(double d, uint32_t seed) {
  uint64_t bits;
  __builtin_memcpy(&bits, &d, 8);
  if (bits == 0x8000000000000000ull) bits = 0;
  return spark_hash_u64(bits, seed);
}

// byte string: full 4-byte words LE, then tail bytes zero-extended
// (spark_murmur3.rs:42-67)
LS_HD inline uint32_t spark_hash_bytes(const uint8_t* p, int64_t n,
                                       uint32_t seed) {
  uint32_t h = seed;
  int64_t nblocks = n / 4;
  for (int64_t i = 0; i < nblocks; i++) {
    uint32_t k;
    __builtin_memcpy(&k, p + 4 * i, 4);
    h = mur_mix_h(h, mur_mix_k(k));
  }
  for (int64_t i = nblocks * 4; i < n; i++) {
    h = mur_mix_h(h, mur_mix_k((uint32_t)p[i]));
  }
  return mur_finish(h, (uint32_t)n);
}

}  // namespace lakesoul
