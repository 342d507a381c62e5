// CPU reducer + CPU codec implementations (OpenMP).
//
// Used by the PS server process (which has no GPU — reference
// common/cpu_reducer.cc:59-439, server/server.cc) and as the host-side
// golden path for the HIP codecs: the RNG construction in common.h is
// shared, so worker-GPU compression and server-CPU decompression agree
// bit-for-bit on indices and stochastic rounding decisions.

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <atomic>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

#include "common.h"

namespace bpsamd {

static inline float bf16_to_f32(uint16_t v) {
  uint32_t u = (uint32_t)v << 16;
  float f;
  std::memcpy(&f, &u, 4);
  return f;
}

static inline uint16_t f32_to_bf16(float f) {
  uint32_t u;
  std::memcpy(&u, &f, 4);
  // round-to-nearest-even
  uint32_t rounding = 0x7FFF + ((u >> 16) & 1);
  return (uint16_t)((u + rounding) >> 16);
}

extern "C" {

// dst += src  (fp32 accumulate for bf16, like the reference's F16C fp16
// path, common/cpu_reducer.cc:96-141)
int bps_cpu_sum(void* dst, const void* src, int64_t n, int dtype) {
  if (dtype == 0) {
    float* d = (float*)dst;
    const float* s = (const float*)src;
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i) d[i] += s[i];
  } else if (dtype == 2) {
    uint16_t* d = (uint16_t*)dst;
    const uint16_t* s = (const uint16_t*)src;
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i)
      d[i] = f32_to_bf16(bf16_to_f32(d[i]) + bf16_to_f32(s[i]));
  } else if (dtype == 3) {
    double* d = (double*)dst;
    const double* s = (const double*)src;
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i) d[i] += s[i];
  } else {
    return -1;
  }
  return 0;
}

// dst = src1 + alpha * src2
int bps_cpu_sum2(void* dst, const void* src1, const void* src2, int64_t n,
                 float alpha, int dtype) {
  if (dtype != 0) return -1;
  float* d = (float*)dst;
  const float* a = (const float*)src1;
  const float* b = (const float*)src2;
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i) d[i] = a[i] + alpha * b[i];
  return 0;
}

int bps_cpu_copy(void* dst, const void* src, int64_t nbytes) {
#pragma omp parallel
  {
    // split the copy across threads (reference OMP copy,
    // common/cpu_reducer.cc:426-437)
    int tid = 0, nthr = 1;
#ifdef _OPENMP
    tid = omp_get_thread_num();
    nthr = omp_get_num_threads();
#endif
    int64_t chunk = (nbytes + nthr - 1) / nthr;
    int64_t beg = tid * chunk;
    int64_t end = std::min<int64_t>(nbytes, beg + chunk);
    if (end > beg)
      std::memcpy((char*)dst + beg, (const char*)src + beg, end - beg);
  }
  return 0;
}

int bps_cpu_scale(void* x, int64_t n, float alpha, int dtype) {
  if (dtype == 0) {
    float* d = (float*)x;
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i) d[i] *= alpha;
    return 0;
  }
  return -1;
}

// -- CPU codecs (server side) ----------------------------------------------

int bps_cpu_onebit_compress(const float* x, int64_t n, uint64_t* bits,
                            float* scale_sum) {
  int64_t nwords = (n + 63) >> 6;
  double l1 = 0.0;
#pragma omp parallel for reduction(+ : l1)
  for (int64_t w = 0; w < nwords; ++w) {
    uint64_t mask = 0;
    int64_t lim = std::min<int64_t>(64, n - (w << 6));
    for (int64_t l = 0; l < lim; ++l) {
      float v = x[(w << 6) + l];
      l1 += std::fabs(v);
      if (v >= 0.0f) mask |= (1ULL << l);
    }
    bits[w] = mask;
  }
  *scale_sum = (float)l1;
  return 0;
}

int bps_cpu_onebit_decompress(const uint64_t* bits, float scale_sum, int64_t n,
                              float* out) {
  float scale = scale_sum / (float)n;
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i)
    out[i] = ((bits[i >> 6] >> (i & 63)) & 1ULL) ? scale : -scale;
  return 0;
}

// Fused decode → accumulator (one pass; replaces decompress-to-scratch +
// copy/sum — the server merge was 3 passes per push, VERDICT round-2
// item 4: the PS per-step cost lives in these CPU passes).
int bps_cpu_onebit_accumulate(const uint64_t* bits, float scale_sum,
                              int64_t n, float* acc, int first) {
  float scale = scale_sum / (float)n;
  if (first) {
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i)
      acc[i] = ((bits[i >> 6] >> (i & 63)) & 1ULL) ? scale : -scale;
  } else {
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i)
      acc[i] += ((bits[i >> 6] >> (i & 63)) & 1ULL) ? scale : -scale;
  }
  return 0;
}

// Fused server-side error feedback + sign-pack for the merged reply,
// split so only pass 1 sits on the pull critical path:
//   pass 1 (reply_pack): comp = acc + err;  bits = sign(comp);
//                        scale_sum = Σ|comp|       → reply is ready
//   pass 2 (err_update): err = comp − sign(comp)·scale   → runs AFTER
//                        the queued pulls flush; only needed next round
// (replaces the generic compensate/compress/decompress/subtract
// four-pass chain).
int bps_cpu_onebit_reply_pack(const float* acc, const float* err, int64_t n,
                              uint64_t* bits, float* scale_sum,
                              float* comp) {
  int64_t nwords = (n + 63) >> 6;
  double l1 = 0.0;
#pragma omp parallel for reduction(+ : l1)
  for (int64_t w = 0; w < nwords; ++w) {
    uint64_t mask = 0;
    int64_t lim = std::min<int64_t>(64, n - (w << 6));
    for (int64_t l = 0; l < lim; ++l) {
      int64_t i = (w << 6) + l;
      float c = acc[i] + err[i];
      comp[i] = c;
      l1 += std::fabs(c);
      if (c >= 0.0f) mask |= (1ULL << l);
    }
    bits[w] = mask;
  }
  *scale_sum = (float)l1;
  return 0;
}

int bps_cpu_onebit_err_update(const float* comp, int64_t n, float scale_sum,
                              float* err) {
  float scale = scale_sum / (float)n;
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i) {
    float c = comp[i];
    err[i] = c - (c >= 0.0f ? scale : -scale);
  }
  return 0;
}

int bps_cpu_randomk_indices(int64_t n, int64_t k, uint64_t seed,
                            int32_t* idx) {
  // regenerate the worker's draws from the seed (counter-mode)
#pragma omp parallel for
  for (int64_t j = 0; j < k; ++j)
    idx[j] = (int32_t)rand_index(seed, (uint64_t)j, (uint64_t)n);
  return 0;
}

int bps_cpu_sparse_scatter(const int32_t* idx, const float* val, int64_t k,
                           float* out) {
  for (int64_t j = 0; j < k; ++j) out[idx[j]] = val[j];
  return 0;
}

// sparse sum into a dense accumulator: acc[idx[j]] += val[j]
int bps_cpu_sparse_accumulate(const int32_t* idx, const float* val, int64_t k,
                              float* acc) {
  for (int64_t j = 0; j < k; ++j) acc[idx[j]] += val[j];
  return 0;
}

int bps_cpu_dithering_compress(const float* x, int64_t n, int s, uint64_t seed,
                               int natural, float norm, int8_t* code) {
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i) {
    float v = x[i];
    float r = (norm > 0.0f) ? std::fabs(v) / norm : 0.0f;
    if (!natural) {
      float t = r * s;
      int level = (int)t;
      float frac = t - level;
      level += (uniform_at(seed, (uint64_t)i) < frac) ? 1 : 0;
      if (level > s) level = s;
      code[i] = (int8_t)(v < 0.0f ? -level : level);
    } else {
      // s power-of-two levels; below 2^(1-s) stochastically → 0 (the
      // sparsity source) — bit-identical twin of the gfx950 kernel
      if (r <= 0.0f) {
        code[i] = 0;
        continue;
      }
      float lowest = std::ldexp(1.0f, 1 - s);
      int ebits;
      if (r < lowest) {
        float p_up = r / lowest;
        if (uniform_at(seed, (uint64_t)i) >= p_up) {
          code[i] = 0;
          continue;
        }
        ebits = 1 - s;
      } else {
        int e;
        float m = std::frexp(r, &e);
        float p_up = m * 2.0f - 1.0f;
        ebits = e - 1 + ((uniform_at(seed, (uint64_t)i) < p_up) ? 1 : 0);
        if (ebits > 0) ebits = 0;
        if (ebits < 1 - s) ebits = 1 - s;
      }
      int biased = ebits + s;
      code[i] = (int8_t)(v < 0.0f ? -biased : biased);
    }
  }
  return 0;
}

int bps_cpu_dithering_decompress(const int8_t* code, int64_t n, int s,
                                 int natural, float norm, float* out) {
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i) {
    int c = code[i];
    if (!natural) {
      out[i] = (float)c / (float)s * norm;
    } else {
      if (c == 0) {
        out[i] = 0.0f;
        continue;
      }
      int mag = c < 0 ? -c : c;
      float v = std::ldexp(1.0f, mag - s) * norm;
      out[i] = c < 0 ? -v : v;
    }
  }
  return 0;
}

int bps_cpu_fp8_compress(const float* x, int64_t n, float amax,
                         uint8_t* code) {
  const float scale = amax > 0.0f ? 448.0f / amax : 0.0f;
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i)
    code[i] = fp8_e4m3_encode(x[i] * scale);
  return 0;
}

int bps_cpu_fp8_decompress(const uint8_t* code, int64_t n, float amax,
                           float* out) {
  const float inv = amax / 448.0f;
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i)
    out[i] = fp8_e4m3_decode(code[i]) * inv;
  return 0;
}

// fused decode → accumulator (one pass, like the onebit/dithering paths)
int bps_cpu_fp8_accumulate(const uint8_t* code, int64_t n, float amax,
                           float* acc, int first) {
  const float inv = amax / 448.0f;
  if (first) {
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i) acc[i] = fp8_e4m3_decode(code[i]) * inv;
  } else {
#pragma omp parallel for
    for (int64_t i = 0; i < n; ++i)
      acc[i] += fp8_e4m3_decode(code[i]) * inv;
  }
  return 0;
}

// ---------------------------------------------------------------------------
// Elias-delta sparse wire for dithering codes (reference
// common/compressor/utils.h:115-250 BitWriter/Elias-delta +
// impl/dithering.cc:51-121 sign/position coding — re-designed as a
// CHUNKED stream so encode and decode parallelize):
//
//   [u32 chunk_elems][u32 nchunks]
//   nchunks × [u32 nbytes]                  (chunk byte sizes)
//   nchunks × chunk streams, byte aligned:
//       per nonzero, MSB-first bits: elias_delta(gap+1) ·
//       sign bit (1 ⇒ negative) · elias_delta(|code|) ; then
//       elias_delta(chunk_remainder+1) terminates the chunk.
//
// elias_delta(x ≥ 1): x = 2^(N-1)+rest; N = 2^(L-1)+… encoded as
// L-1 zeros, bits of N, then N-1 low bits of x.
// ---------------------------------------------------------------------------

namespace {

constexpr int64_t kChunkElems = 1 << 16;

// Concurrent pool threads each spawn their own OMP team; uncapped teams
// (= all cores) from 7+ threads oversubscribe catastrophically on
// 256-core boxes (same failure mode as the server engines, measured in
// profiles/MEASUREMENTS.md).  Cap the wire codec's teams.
inline int wire_threads() {
  static int n = [] {
    const char* e = getenv("BPS_WIRE_THREADS");
    int v = e ? atoi(e) : 4;
    return v < 1 ? 1 : v;
  }();
  return n;
}

struct BitWriter {
  uint8_t* p;
  uint64_t bitpos = 0;
  explicit BitWriter(uint8_t* out) : p(out) {}
  inline void put_bit(int b) {
    if (b) p[bitpos >> 3] |= (uint8_t)(0x80u >> (bitpos & 7));
    bitpos++;
  }
  inline void put_bits(uint32_t v, int nbits) {  // MSB first
    for (int i = nbits - 1; i >= 0; --i) put_bit((v >> i) & 1);
  }
  inline void elias_delta(uint32_t x) {  // x >= 1
    int nb = 31;
    while (nb > 0 && !((x >> nb) & 1)) --nb;       // nb = floor(log2 x)
    uint32_t N = (uint32_t)nb + 1;
    int lb = 31;
    while (lb > 0 && !((N >> lb) & 1)) --lb;       // lb = floor(log2 N)
    for (int i = 0; i < lb; ++i) put_bit(0);
    put_bits(N, lb + 1);
    if (nb > 0) put_bits(x & ((1u << nb) - 1), nb);
  }
  inline int64_t bytes() const { return (int64_t)((bitpos + 7) >> 3); }
};

struct BitReader {
  const uint8_t* p;
  uint64_t bitpos = 0;
  explicit BitReader(const uint8_t* in) : p(in) {}
  inline int get_bit() {
    int b = (p[bitpos >> 3] >> (7 - (bitpos & 7))) & 1;
    bitpos++;
    return b;
  }
  inline uint32_t get_bits(int nbits) {
    uint32_t v = 0;
    for (int i = 0; i < nbits; ++i) v = (v << 1) | get_bit();
    return v;
  }
  inline uint32_t elias_delta() {
    int lb = 0;
    while (!get_bit()) ++lb;
    uint32_t N = (1u << lb) | get_bits(lb);
    int nb = (int)N - 1;
    uint32_t x = 1u << nb;
    if (nb > 0) x |= get_bits(nb);
    return x;
  }
};

// encode one chunk of codes; returns bytes written
int64_t encode_chunk(const int8_t* code, int64_t n, uint8_t* out) {
  BitWriter w(out);
  int64_t prev = -1;
  for (int64_t i = 0; i < n; ++i) {
    if (code[i] == 0) continue;
    w.elias_delta((uint32_t)(i - prev));  // gap+1 ≥ 1
    w.put_bit(code[i] < 0);
    w.elias_delta((uint32_t)(code[i] < 0 ? -code[i] : code[i]));
    prev = i;
  }
  w.elias_delta((uint32_t)(n - prev));    // terminator: remainder+1
  return w.bytes();
}

}  // namespace

// Encode dithering codes → sparse wire.  Returns total bytes, or -1 if
// out_cap is too small (caller falls back to the dense int8 wire).
int bps_cpu_dither_encode(const int8_t* code, int64_t n, uint8_t* out,
                          int64_t out_cap, int64_t* out_len) {
  int64_t nchunks = (n + kChunkElems - 1) / kChunkElems;
  if (nchunks == 0) nchunks = 1;
  int64_t header = 8 + 4 * nchunks;
  if (out_cap < header) return -1;
  // worst case ~14 bits/elem + terminator
  std::vector<std::vector<uint8_t>> bufs(nchunks);
  std::vector<int64_t> sizes(nchunks, 0);
  bool overflow = false;
#pragma omp parallel for schedule(dynamic) num_threads(wire_threads())
  for (int64_t c = 0; c < nchunks; ++c) {
    int64_t lo = c * kChunkElems;
    int64_t len = std::min<int64_t>(kChunkElems, n - lo);
    bufs[c].assign((size_t)(len * 2 + 16), 0);
    sizes[c] = encode_chunk(code + lo, len, bufs[c].data());
  }
  int64_t total = header;
  for (int64_t c = 0; c < nchunks; ++c) total += sizes[c];
  if (overflow || total > out_cap) return -1;
  uint32_t ce = (uint32_t)kChunkElems, nc = (uint32_t)nchunks;
  std::memcpy(out, &ce, 4);
  std::memcpy(out + 4, &nc, 4);
  for (int64_t c = 0; c < nchunks; ++c) {
    uint32_t sz = (uint32_t)sizes[c];
    std::memcpy(out + 8 + 4 * c, &sz, 4);
  }
  int64_t off = header;
  for (int64_t c = 0; c < nchunks; ++c) {
    std::memcpy(out + off, bufs[c].data(), sizes[c]);
    off += sizes[c];
  }
  *out_len = total;
  return 0;
}

// Decode sparse wire → dense codes (zeros included).  Returns 0, or -1
// on malformed input.
int bps_cpu_dither_decode(const uint8_t* in, int64_t in_len, int64_t n,
                          int8_t* code) {
  if (in_len < 8) return -1;
  uint32_t ce, nc;
  std::memcpy(&ce, in, 4);
  std::memcpy(&nc, in + 4, 4);
  if (ce == 0 || nc == 0 || (int64_t)8 + 4 * nc > in_len) return -1;
  std::vector<int64_t> offs(nc + 1, 0);
  offs[0] = 8 + 4 * (int64_t)nc;
  for (uint32_t c = 0; c < nc; ++c) {
    uint32_t sz;
    std::memcpy(&sz, in + 8 + 4 * c, 4);
    offs[c + 1] = offs[c] + sz;
  }
  if (offs[nc] > in_len) return -1;
  std::memset(code, 0, (size_t)n);
  std::atomic<int> bad{0};
#pragma omp parallel for schedule(dynamic) num_threads(wire_threads())
  for (int64_t c = 0; c < (int64_t)nc; ++c) {
    int64_t lo = (int64_t)c * ce;
    int64_t len = std::min<int64_t>(ce, n - lo);
    if (len <= 0) continue;
    BitReader r(in + offs[c]);
    uint64_t maxbits = (uint64_t)(offs[c + 1] - offs[c]) * 8;
    int64_t i = -1;
    for (;;) {
      if (r.bitpos >= maxbits) { bad = 1; break; }
      uint32_t gap = r.elias_delta();
      i += gap;
      if (i >= len) break;           // terminator
      int neg = r.get_bit();
      uint32_t mag = r.elias_delta();
      if (mag > 127) { bad = 1; break; }
      code[lo + i] = (int8_t)(neg ? -(int)mag : (int)mag);
    }
  }
  return bad ? -1 : 0;
}

// Server reply path for dithering, two passes instead of
// compensate/norm/quantize three:
//   pass 1: comp = acc + err, accumulating the norm (L2 for natural,
//           max for linear)
//   pass 2: quantize comp with a CHEAP splitmix draw — the reply's
//           stochastic rounding needs no GPU-twin reproducibility (codes
//           ship on the wire; only the worker-side compress has a HIP
//           twin), and the 3-hash uniform_at was the reply's hot loop.
int bps_cpu_dither_compensate_norm(const float* acc, const float* err,
                                   int64_t n, int natural, float* comp,
                                   float* out_norm) {
  if (natural) {
    double ss = 0.0;
#pragma omp parallel for reduction(+ : ss)
    for (int64_t i = 0; i < n; ++i) {
      float c = err ? acc[i] + err[i] : acc[i];
      comp[i] = c;
      ss += (double)c * c;
    }
    *out_norm = (float)std::sqrt(ss);
  } else {
    float mx = 0.0f;
#pragma omp parallel for reduction(max : mx)
    for (int64_t i = 0; i < n; ++i) {
      float c = err ? acc[i] + err[i] : acc[i];
      comp[i] = c;
      float a = std::fabs(c);
      if (a > mx) mx = a;
    }
    *out_norm = mx;
  }
  return 0;
}

static inline float cheap_uniform(uint64_t seed, uint64_t i) {
  uint64_t z = seed + i * 0x9E3779B97F4A7C15ULL;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z ^= z >> 27;
  return (float)(z >> 40) * (1.0f / 16777216.0f);
}

int bps_cpu_dithering_compress_fast(const float* x, int64_t n, int s,
                                    uint64_t seed, int natural, float norm,
                                    int8_t* code) {
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i) {
    float v = x[i];
    float r = (norm > 0.0f) ? std::fabs(v) / norm : 0.0f;
    if (!natural) {
      float t = r * s;
      int level = (int)t;
      float frac = t - level;
      level += (cheap_uniform(seed, (uint64_t)i) < frac) ? 1 : 0;
      if (level > s) level = s;
      code[i] = (int8_t)(v < 0.0f ? -level : level);
    } else {
      if (r <= 0.0f) {
        code[i] = 0;
        continue;
      }
      float lowest = std::ldexp(1.0f, 1 - s);
      int ebits;
      if (r < lowest) {
        if (cheap_uniform(seed, (uint64_t)i) >= r / lowest) {
          code[i] = 0;
          continue;
        }
        ebits = 1 - s;
      } else {
        int e;
        float m = std::frexp(r, &e);
        float p_up = m * 2.0f - 1.0f;
        ebits = e - 1 + ((cheap_uniform(seed, (uint64_t)i) < p_up) ? 1 : 0);
        if (ebits > 0) ebits = 0;
        if (ebits < 1 - s) ebits = 1 - s;
      }
      code[i] = (int8_t)(v < 0.0f ? -(ebits + s) : (ebits + s));
    }
  }
  return 0;
}

// Fused dithering decode → accumulator (server push path, one pass over
// dense codes)
int bps_cpu_dithering_accumulate(const int8_t* code, int64_t n, int s,
                                 int natural, float norm, float* acc,
                                 int first) {
#pragma omp parallel for
  for (int64_t i = 0; i < n; ++i) {
    int c = code[i];
    float v;
    if (!natural) {
      v = (float)c / (float)s * norm;
    } else if (c == 0) {
      v = 0.0f;
    } else {
      int mag = c < 0 ? -c : c;
      v = std::ldexp(1.0f, mag - s) * norm;
      if (c < 0) v = -v;
    }
    if (first) acc[i] = v;
    else acc[i] += v;
  }
  return 0;
}

// Parallel top-k by |x|: per-thread min-heaps over disjoint ranges, then
// one partial_sort over the k*T candidates (k log k·T) — replaces the
// single-threaded O(n log k) partial_sort over all n that made topk the
// slowest server codec (reference used a serial heap
// too, impl/topk.cc:43-78).  Output pairs are ordered by descending |x|.
int bps_cpu_topk_select(const float* x, int64_t n, int64_t k, int32_t* idx,
                        float* val) {
  if (k <= 0) return -1;
  if (k > n) k = n;
  int nt = 1;
#ifdef _OPENMP
  nt = omp_get_max_threads();
#endif
  if ((int64_t)nt * 4 > n / (k > 0 ? k : 1) + 1) nt = 1;  // tiny inputs
  std::vector<int32_t> cand_idx((size_t)nt * k);
  std::vector<int64_t> cand_cnt(nt, 0);
#pragma omp parallel num_threads(nt)
  {
#ifdef _OPENMP
    int t = omp_get_thread_num();
#else
    int t = 0;
#endif
    int64_t lo = n * t / nt, hi = n * (t + 1) / nt;
    int32_t* heap = cand_idx.data() + (size_t)t * k;
    int64_t cnt = 0;
    auto less = [&](int32_t a, int32_t b) {  // min-heap on |x|
      return std::fabs(x[a]) > std::fabs(x[b]);
    };
    for (int64_t i = lo; i < hi; ++i) {
      if (cnt < k) {
        heap[cnt++] = (int32_t)i;
        if (cnt == k) std::make_heap(heap, heap + k, less);
      } else if (std::fabs(x[i]) > std::fabs(x[heap[0]])) {
        std::pop_heap(heap, heap + k, less);
        heap[k - 1] = (int32_t)i;
        std::push_heap(heap, heap + k, less);
      }
    }
    cand_cnt[t] = cnt;
  }
  // compact candidates (ranges shorter than k contribute fewer)
  std::vector<int32_t> all;
  all.reserve((size_t)nt * k);
  for (int t = 0; t < nt; ++t) {
    const int32_t* c = cand_idx.data() + (size_t)t * k;
    all.insert(all.end(), c, c + cand_cnt[t]);
  }
  if ((int64_t)all.size() > k) {
    std::partial_sort(all.begin(), all.begin() + k, all.end(),
                      [&](int32_t a, int32_t b) {
                        return std::fabs(x[a]) > std::fabs(x[b]);
                      });
    all.resize(k);
  } else {
    std::sort(all.begin(), all.end(), [&](int32_t a, int32_t b) {
      return std::fabs(x[a]) > std::fabs(x[b]);
    });
  }
  for (int64_t j = 0; j < (int64_t)all.size(); ++j) {
    idx[j] = all[j];
    val[j] = x[all[j]];
  }
  return 0;
}

float bps_cpu_norm(const float* x, int64_t n, int mode) {
  if (mode == 2) {
    float mx = 0.0f;
#pragma omp parallel for reduction(max : mx)
    for (int64_t i = 0; i < n; ++i) mx = std::max(mx, std::fabs(x[i]));
    return mx;
  }
  double acc = 0.0;
#pragma omp parallel for reduction(+ : acc)
  for (int64_t i = 0; i < n; ++i) {
    float e = std::fabs(x[i]);
    acc += (mode == 0) ? e : (double)e * e;
  }
  return (mode == 1) ? (float)std::sqrt(acc) : (float)acc;
}

}  // extern "C"
}  // namespace bpsamd
