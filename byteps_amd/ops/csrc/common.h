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

BPS_HD inline uint32_t f32_bits(float f) {
  uint32_t u;
  __builtin_memcpy(&u, &f, 4);
  return u;
}

BPS_HD inline float bits_f32(uint32_t u) {
  float f;
  __builtin_memcpy(&f, &u, 4);
  return f;
}

// Branch-light bit-manipulation conversion — integer ops only on the
// normal path, so host and device produce identical bytes.
BPS_HD inline uint8_t fp8_e4m3_encode(float x) {
  uint32_t u = f32_bits(x);
  uint8_t sign = (uint8_t)((u >> 31) << 7);
  uint32_t au = u & 0x7FFFFFFFu;
  if (au > 0x7F800000u) return sign;            // NaN → 0
  if (au >= 0x43E00000u) return sign | 0x7E;    // |x| ≥ 448 → max normal
  if (au < 0x3C800000u) {                       // |x| < 2^-6 → subnormal
    float q = bits_f32(au) * 512.0f;            // units of 2^-9, q ∈ [0, 32)
    int qi = (int)(q + 0.5f);
    if (((float)qi - q == 0.5f) && (qi & 1)) --qi;   // ties to even
    return sign | (uint8_t)qi;                  // qi == 8 encodes 2^-6
  }
  // normal: RNE-round the f32 mantissa down to 3 bits via integer add
  uint32_t lsb = (au >> 20) & 1u;
  au += 0x0007FFFFu + lsb;
  int ef = (int)(au >> 23) - 127 + 7;
  uint32_t mant = (au >> 20) & 7u;
  if (ef > 15 || (ef == 15 && mant > 6)) return sign | 0x7E;
  return sign | (uint8_t)((ef << 3) | mant);
}

BPS_HD inline float fp8_e4m3_decode(uint8_t b) {
  const int ef = (b >> 3) & 0xF;
  const uint32_t mant = b & 0x7u;
  const uint32_t s = ((uint32_t)(b >> 7)) << 31;
  if (ef == 0)                                  // subnormal: mant × 2^-9
    return bits_f32(s | 0x3B000000u /*2^-9*/) * (float)mant;
  return bits_f32(s | ((uint32_t)(ef - 7 + 127) << 23) | (mant << 20));
}

}  // namespace bpsamd
