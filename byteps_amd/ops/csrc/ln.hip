// Fused LayerNorm kernels for gfx950 — bf16 in/out, fp32 statistics.
//
// Under autocast, torch's layer_norm upcasts activations to fp32: the
// BERT-large profile (profiles/bert_large_steady_state.md) shows the LN
// kernels in fp32 plus bf16↔fp32 cast copies around every call (~4 ms of
// a 47 ms step).  These kernels keep bf16 storage with fp32 math:
// single-pass row statistics (each thread owns one 8×bf16 slot in
// registers), wave/LDS reductions, and the BN-style transposed-partial
// column reduction for dgamma/dbeta.
//
// Layout: x is [M, C] row-major, C contiguous.  Supported: C % 8 == 0
// and C/8 ≤ 256 (C ≤ 2048) with C/8 ∈ {power of two ≤ 64} ∪ {multiples
// of 64}; python falls back to torch otherwise.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define BLOCK 256
#define LN_RED_BLOCKS 1024

namespace {

using bf16 = __hip_bfloat16;
typedef short short8 __attribute__((ext_vector_type(8)));

__device__ inline float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ inline bf16 f2bf(float v) { return __float2bfloat16(v); }

__device__ inline void load8(const bf16* p, float* out) {
  short8 r = *reinterpret_cast<const short8*>(p);
  const bf16* e = reinterpret_cast<const bf16*>(&r);
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf2f(e[i]);
}

__device__ inline void store8(bf16* p, const float* in) {
  short8 r;
  bf16* e = reinterpret_cast<bf16*>(&r);
#pragma unroll
  for (int i = 0; i < 8; ++i) e[i] = f2bf(in[i]);
  *reinterpret_cast<short8*>(p) = r;
}

// Reduce `v` across the cpt threads of this thread's row group.
// cpt ≤ 64 must be a power of two (shfl segments); cpt > 64 must be a
// multiple of 64 (wave shfl + LDS across waves).  Returns the group sum
// in every lane of the group.
__device__ inline float group_sum(float v, int cpt, float* lds_row) {
  if (cpt <= 64) {
    for (int off = cpt >> 1; off > 0; off >>= 1)
      v += __shfl_xor(v, off, 64);
    return v;
  }
  // full-wave reduce first
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  // combine the cpt/64 waves of this group through LDS
  const int t = threadIdx.x;
  const int group = t / cpt;                  // row group in block
  const int wave_in_group = (t - group * cpt) >> 6;
  const int waves_per_group = cpt >> 6;
  if ((t & 63) == 0) lds_row[group * waves_per_group + wave_in_group] = v;
  __syncthreads();
  float total = 0.0f;
  for (int w = 0; w < waves_per_group; ++w)
    total += lds_row[group * waves_per_group + w];
  __syncthreads();
  return total;
}

// ---------------------------------------------------------------------------
// fwd: per row, mean/var over C; y = (x-mean)*invstd*gamma + beta
// One thread per 8-channel slot; groups of cpt threads per row.
// ---------------------------------------------------------------------------

__global__ void ln_fwd_kernel(const bf16* __restrict__ x,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              bf16* __restrict__ y, long long M, int C,
                              float eps, float* __restrict__ mean_out,
                              float* __restrict__ invstd_out) {
  __shared__ float lds[BLOCK / 64 + 8];
  const int cpt = C >> 3;                      // power of two dividing 256
  const int groups = BLOCK / cpt;              // rows per block pass
  const int t = threadIdx.x;
  const int g = t / cpt;
  const int c8 = t - g * cpt;
  const int c0 = c8 << 3;
  const float rn = 1.0f / (float)C;

  // uniform trip count: every thread runs every iteration (group_sum may
  // hit __syncthreads for cpt > 64), inactive rows just contribute 0
  for (long long base = (long long)blockIdx.x * groups; base < M;
       base += (long long)gridDim.x * groups) {
    const long long row = base + g;
    const bool active = row < M;
    float vals[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    if (active) load8(x + row * C + c0, vals);
    float s = 0.0f, q = 0.0f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      s += vals[i];
      q += vals[i] * vals[i];
    }
    s = group_sum(s, cpt, lds);
    q = group_sum(q, cpt, lds);
    float mean = s * rn;
    float var = q * rn - mean * mean;
    if (var < 0.0f) var = 0.0f;
    float invstd = rsqrtf(var + eps);
    if (active) {
      if (c8 == 0) {
        mean_out[row] = mean;
        invstd_out[row] = invstd;
      }
#pragma unroll
      for (int i = 0; i < 8; ++i)
        vals[i] = (vals[i] - mean) * invstd * gamma[c0 + i] + beta[c0 + i];
      store8(y + row * C + c0, vals);
    }
  }
}

// ---------------------------------------------------------------------------
// bwd: dx = invstd * (dyg - mean(dyg) - xhat * mean(dyg*xhat)) with
// dyg = dy*gamma; also accumulates per-block column partials for
// dbeta = Σ_rows dy and dgamma = Σ_rows dy*xhat (transposed layout
// [row][LN_RED_BLOCKS], folded by bps_bn_fold-style kernel).
// ---------------------------------------------------------------------------

__global__ void ln_bwd_kernel(const bf16* __restrict__ x,
                              const bf16* __restrict__ dy,
                              const float* __restrict__ gamma,
                              const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              bf16* __restrict__ dx, long long M, int C,
                              float* __restrict__ partial) {
  __shared__ float lds[BLOCK / 64 + 8];
  const int cpt = C >> 3;
  const int groups = BLOCK / cpt;
  const int t = threadIdx.x;
  const int g = t / cpt;
  const int c8 = t - g * cpt;
  const int c0 = c8 << 3;
  const float rn = 1.0f / (float)C;

  float db[8], dgm[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) db[i] = dgm[i] = 0.0f;

  for (long long base = (long long)blockIdx.x * groups; base < M;
       base += (long long)gridDim.x * groups) {
    const long long row = base + g;
    const bool active = row < M;
    float xv[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    float dv[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    float mu = 0.0f, is = 0.0f;
    if (active) {
      load8(x + row * C + c0, xv);
      load8(dy + row * C + c0, dv);
      mu = mean[row];
      is = invstd[row];
    }
    float s1 = 0.0f, s2 = 0.0f;
    float xhat[8], dyg[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      xhat[i] = (xv[i] - mu) * is;
      dyg[i] = dv[i] * gamma[c0 + i];
      s1 += dyg[i];
      s2 += dyg[i] * xhat[i];
      db[i] += dv[i];
      dgm[i] += dv[i] * xhat[i];
    }
    s1 = group_sum(s1, cpt, lds) * rn;
    s2 = group_sum(s2, cpt, lds) * rn;
    if (active) {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        dv[i] = is * (dyg[i] - s1 - xhat[i] * s2);
      store8(dx + row * C + c0, dv);
    }
  }

  // column partials: rows of a block share c8 slots across groups — LDS
  // reduce over groups, then one transposed write per block
  __shared__ float col[BLOCK * 8];
#pragma unroll
  for (int i = 0; i < 8; ++i) col[t * 8 + i] = db[i];
  __syncthreads();
  if (g == 0) {
    for (int gg = 1; gg < groups; ++gg)
#pragma unroll
      for (int i = 0; i < 8; ++i) db[i] += col[(gg * cpt + c8) * 8 + i];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      partial[(long long)(c0 + i) * LN_RED_BLOCKS + blockIdx.x] = db[i];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 8; ++i) col[t * 8 + i] = dgm[i];
  __syncthreads();
  if (g == 0) {
    for (int gg = 1; gg < groups; ++gg)
#pragma unroll
      for (int i = 0; i < 8; ++i) dgm[i] += col[(gg * cpt + c8) * 8 + i];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      partial[(long long)(C + c0 + i) * LN_RED_BLOCKS + blockIdx.x] =
          dgm[i];
  }
}

__device__ inline float wave_row_sum_ln(const float* __restrict__ row,
                                        int nblocks) {
  const int lane = threadIdx.x & 63;
  float acc = 0.0f;
  for (int b = lane; b < nblocks; b += 64) acc += row[b];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  return acc;
}

// adaptive grid matching bn.hip's red_grid (stride stays LN_RED_BLOCKS)
inline int ln_red_grid(long long M, int C) {
  int groups = BLOCK / (C >> 3);
  if (groups < 1) groups = 1;
  long long need = (M + groups - 1) / groups;
  // unlike BN, LN shapes here are large-M (tokens) with moderate C — the
  // deep grid wins (measured 26 vs 30-40 µs at [8192,1024]); the
  // per-block 2C scatter is small relative to the two streamed reads
  if (need < 1) need = 1;
  return (int)(need < LN_RED_BLOCKS ? need : LN_RED_BLOCKS);
}

__global__ void ln_fold_kernel(const float* __restrict__ partial, int C,
                               float* __restrict__ sums2, int nblocks) {
  int row = blockIdx.x;
  if (row >= 2 * C) return;
  float acc = wave_row_sum_ln(partial + (long long)row * LN_RED_BLOCKS,
                              nblocks);
  if (threadIdx.x == 0) sums2[row] = acc;
}

inline bool ln_supported(int C) {
  // cpt must be a power of two ≤ 256 so that row groups tile the block
  // exactly (group_sum's __syncthreads needs whole-block participation)
  if (C & 7) return false;
  int cpt = C >> 3;
  return cpt <= 256 && (cpt & (cpt - 1)) == 0;
}

}  // namespace

#define STREAM reinterpret_cast<hipStream_t>(stream)

extern "C" {

int bps_ln_supported(int C) { return ln_supported(C) ? 1 : 0; }
int bps_ln_red_blocks(void) { return LN_RED_BLOCKS; }

int bps_ln_fwd(const void* x, const void* gamma, const void* beta, void* y,
               long long M, int C, float eps, void* mean, void* invstd,
               void* stream) {
  if (!ln_supported(C)) return -1;
  int groups = BLOCK / (C >> 3);
  long long blocks = (M + groups - 1) / groups;
  int grid = (int)(blocks < 2048 ? (blocks > 0 ? blocks : 1) : 2048);
  hipLaunchKernelGGL(ln_fwd_kernel, dim3(grid), dim3(BLOCK), 0, STREAM,
                     (const bf16*)x, (const float*)gamma,
                     (const float*)beta, (bf16*)y, M, C, eps, (float*)mean,
                     (float*)invstd);
  return (int)hipGetLastError();
}

int bps_ln_bwd(const void* x, const void* dy, const void* gamma,
               const void* mean, const void* invstd, void* dx, long long M,
               int C, void* partial, void* stream) {
  if (!ln_supported(C)) return -1;
  hipLaunchKernelGGL(ln_bwd_kernel, dim3(ln_red_grid(M, C)), dim3(BLOCK),
                     0, STREAM, (const bf16*)x, (const bf16*)dy,
                     (const float*)gamma, (const float*)mean,
                     (const float*)invstd, (bf16*)dx, M, C,
                     (float*)partial);
  return (int)hipGetLastError();
}

int bps_ln_fold(const void* partial, long long M, int C, void* sums2,
                void* stream) {
  hipLaunchKernelGGL(ln_fold_kernel, dim3(2 * C), dim3(64), 0, STREAM,
                     (const float*)partial, C, (float*)sums2,
                     ln_red_grid(M, C));
  return (int)hipGetLastError();
}

}  // extern "C"
