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
#include <cstring>
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
      if (r <= 0.0f) {
        code[i] = 0;
        continue;
      }
      int e;
      float m = std::frexp(r, &e);
      float p_up = m * 2.0f - 1.0f;
      int ebits = e - 1 + ((uniform_at(seed, (uint64_t)i) < p_up) ? 1 : 0);
      if (ebits < -120) {
        code[i] = 0;
        continue;
      }
      if (ebits > 0) ebits = 0;
      int biased = ebits + 121;
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
      float v = std::ldexp(1.0f, mag - 121) * norm;
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
