// Core types shared by the HIP kernels, CPU reducer, KV transport and
// server (MI355X-native equivalent of reference common/common.h:88-264).
#pragma once

#include <cstddef>
#include <cstdint>

#if defined(__HIPCC__)
#define BPS_HD __host__ __device__
#else
#define BPS_HD
#endif

namespace bpsamd {

// dtype codes shared across the Python/C++ boundary (subset of the
// reference's DataType, common/common.h:58-70, that the engine moves).
enum class DType : int32_t {
  kFloat32 = 0,
  kFloat16 = 1,
  kBFloat16 = 2,
  kFloat64 = 3,
  kInt32 = 4,
  kInt64 = 5,
  kUInt8 = 6,
};

BPS_HD inline size_t dtype_size(DType t) {
  switch (t) {
    case DType::kFloat16:
    case DType::kBFloat16:
      return 2;
    case DType::kFloat32:
    case DType::kInt32:
      return 4;
    case DType::kFloat64:
    case DType::kInt64:
      return 8;
    case DType::kUInt8:
      return 1;
  }
  return 0;
}

// splitmix64 — seed-derivation for the per-lane counter-based RNG used by
// randomk / dithering.  Both the HIP kernels and the CPU (server) codecs
// derive per-element streams the same way, so worker-GPU and server-CPU
// stay bit-identical (the reference instead shared a sequential
// xorshift128+ stream, compressor/utils.h:74-113 — sequential streams
// don't parallelize across 64-wide wavefronts, so we use a counter-based
// construction).
BPS_HD inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97f4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

// xorshift128+ (public algorithm) — one step given a 2-word state; used
// counter-mode: state is derived per element via splitmix64.
BPS_HD inline uint64_t xorshift128p(uint64_t& s0, uint64_t& s1) {
  uint64_t x = s0;
  const uint64_t y = s1;
  s0 = y;
  x ^= x << 23;
  s1 = x ^ y ^ (x >> 17) ^ (y >> 26);
  return s1 + y;
}

// uniform in [0, 1) from a derived stream for element i
BPS_HD inline float uniform_at(uint64_t seed, uint64_t i) {
  uint64_t s0 = splitmix64(seed ^ (i * 0xA24BAED4963EE407ULL + 1));
  uint64_t s1 = splitmix64(s0 ^ 0x9FB21C651E98DF25ULL);
  uint64_t r = xorshift128p(s0, s1);
  return (r >> 40) * (1.0f / 16777216.0f);  // 24-bit mantissa
}

// random index in [0, n) for draw j
BPS_HD inline uint64_t rand_index(uint64_t seed, uint64_t j, uint64_t n) {
  uint64_t s0 = splitmix64(seed ^ (j * 0xD6E8FEB86659FD93ULL + 7));
  uint64_t s1 = splitmix64(s0 ^ 0xCA5A826395121157ULL);
  return xorshift128p(s0, s1) % n;
}

}  // namespace bpsamd
