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

// -- OCP fp8 e4m3fn software conversion --------------------------------------
// gfx950's native fp8 is OCP e4m3fn (NOT the MI300X fnuz variant).  The
// software conversion here is shared by the HIP kernels and the CPU
// server so wire bytes are bit-identical on both sides (exact power-of-2
// arithmetic + round-half-even everywhere).
// Layout: s EEEE MMM, bias 7; max normal 0x7E = 448; 0x7F = NaN (unused:
// we saturate); subnormals step 2^-9.

BPS_HD inline uint8_t fp8_e4m3_encode(float x) {
  uint8_t sign = x < 0.0f ? 0x80 : 0x00;
  float a = x < 0.0f ? -x : x;
  if (!(a > 0.0f)) return sign;                 // ±0 and NaN → 0
  if (a >= 448.0f) return sign | 0x7E;          // saturate to max normal
  int e = 0;
  float m = a;
  while (m >= 2.0f) { m *= 0.5f; ++e; }
  while (m < 1.0f && e > -6) { m *= 2.0f; --e; }
  if (m < 1.0f) {
    // subnormal: units of 2^-9
    float q = a * 512.0f;                       // a / 2^-9
    int qi = (int)(q + 0.5f);
    if (((float)qi - q == 0.5f) && (qi & 1)) --qi;   // ties to even
    if (qi <= 0) return sign;
    if (qi > 7) return sign | 0x08;             // rounds up to 2^-6
    return sign | (uint8_t)qi;
  }
  // normal: mant = round((m-1)*8)
  float mf = (m - 1.0f) * 8.0f;
  int mi = (int)(mf + 0.5f);
  if (((float)mi - mf == 0.5f) && (mi & 1)) --mi;
  if (mi == 8) { mi = 0; ++e; }
  if (e > 8) return sign | 0x7E;
  if (e == 8 && mi > 6) return sign | 0x7E;
  return sign | (uint8_t)(((e + 7) << 3) | mi);
}

BPS_HD inline float fp8_e4m3_decode(uint8_t b) {
  float sign = (b & 0x80) ? -1.0f : 1.0f;
  int ef = (b >> 3) & 0xF;
  int mant = b & 0x7;
  if (ef == 0) {
    // subnormal: mant * 2^-9
    return sign * (float)mant * 0.001953125f;
  }
  float m = 1.0f + (float)mant * 0.125f;
  int e = ef - 7;
  float p = 1.0f;
  if (e >= 0) {
    for (int i = 0; i < e; ++i) p *= 2.0f;
    return sign * m * p;
  }
  for (int i = 0; i < -e; ++i) p *= 0.5f;
  return sign * m * p;
}

}  // namespace bpsamd
