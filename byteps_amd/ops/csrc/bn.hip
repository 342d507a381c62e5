// Fused BatchNorm(+residual)(+ReLU) training kernels for gfx950 — NHWC
// (channels_last) bf16 with fp32 statistics.
//
// Replaces MIOpen's 3-kernel fwd + 3-kernel bwd spatial batchnorm, which
// profiling showed at ~30% of a ResNet-50 bf16 step on MI355X
// (profiles/resnet50_steady_state.md).  Design per the CDNA4 guide:
// memory-bound single-pass kernels, 8×bf16 (16 B) vector accesses on the
// fastest (channel) dimension, fp32 accumulation, wave-shuffle → LDS →
// one atomic per block for the channel statistics.
//
// Layout: x is [M, C] row-major with C contiguous (NHWC), M = N*H*W.
// Requires C % 8 == 0 and C <= 4096 (python falls back to torch
// otherwise).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define BLOCK 256
#define MAX_GRID 2048

namespace {

using bf16 = __hip_bfloat16;
typedef short short8 __attribute__((ext_vector_type(8)));
typedef float float4v __attribute__((ext_vector_type(4)));

__device__ inline float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ inline bf16 f2bf(float v) { return __float2bfloat16(v); }

struct F8 {
  float v[8];
  __device__ void zero() {
#pragma unroll
    for (int i = 0; i < 8; ++i) v[i] = 0.0f;
  }
};

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

// Fixed partial-buffer depth: every reduce launches exactly RED_BLOCKS
// blocks (4096 waves — sized to saturate HBM: 256 blocks measured only 3-4.6 TB/s) writing per-block partial
// sums; the finalize kernel folds the partials.  Per-address atomicAdd
// chains from a 2048-block grid measured 431 µs/call (profile round 2) —
// partials + a folding pass run at memory speed.
#define RED_BLOCKS 1024

// ---------------------------------------------------------------------------
// fwd reduce: partial[b][c] = Σ_rows(b) x[m,c] ; partial[b][C+c] = Σ x²
// Each thread owns 8 consecutive channels (one short8 column slice) and a
// row subset; LDS-reduce across the row groups, leader writes the
// block's partial (no atomics).
// ---------------------------------------------------------------------------

__global__ void bn_reduce_kernel(const bf16* __restrict__ x, long long M,
                                 int C, float* __restrict__ partial) {
  const int cpt = C >> 3;                        // short8 slots per row
  const int groups = max(1, BLOCK / cpt);        // rows handled per pass
  const int t = threadIdx.x;
  const int g = t / cpt;
  const int c8 = t - g * cpt;
  __shared__ float lds[BLOCK * 8];

  F8 s, q;
  s.zero();
  q.zero();
  if (g < groups) {
    float vals[8];
    for (long long row = (long long)blockIdx.x * groups + g; row < M;
         row += (long long)gridDim.x * groups) {
      load8(x + row * C + (c8 << 3), vals);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s.v[i] += vals[i];
        q.v[i] += vals[i] * vals[i];
      }
    }
  }
  // LDS reduce across groups for this c8 (sum pass, then sq pass).
  // Partials are stored TRANSPOSED — partial[row][RED_BLOCKS] with
  // row ∈ [0,2C) — so the folding kernels read coalesced along blocks.
#pragma unroll
  for (int i = 0; i < 8; ++i) lds[t * 8 + i] = s.v[i];
  __syncthreads();
  if (g == 0 && c8 < cpt) {
    for (int gg = 1; gg < groups; ++gg)
#pragma unroll
      for (int i = 0; i < 8; ++i) s.v[i] += lds[(gg * cpt + c8) * 8 + i];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      partial[(long long)((c8 << 3) + i) * RED_BLOCKS + blockIdx.x] = s.v[i];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 8; ++i) lds[t * 8 + i] = q.v[i];
  __syncthreads();
  if (g == 0 && c8 < cpt) {
    for (int gg = 1; gg < groups; ++gg)
#pragma unroll
      for (int i = 0; i < 8; ++i) q.v[i] += lds[(gg * cpt + c8) * 8 + i];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      partial[(long long)(C + (c8 << 3) + i) * RED_BLOCKS + blockIdx.x] =
          q.v[i];
  }
}

// one-wave reduction of the first `nblocks` entries of a partial row
// (row stride is always RED_BLOCKS; small shapes fill fewer columns)
__device__ inline float wave_row_sum(const float* __restrict__ row,
                                     int nblocks) {
  const int lane = threadIdx.x & 63;
  float acc = 0.0f;
  for (int b = lane; b < nblocks; b += 64) acc += row[b];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  return acc;  // valid in lane 0
}

// adaptive reduce grid: enough row-blocks to saturate HBM, no more than
// the partial stride; small shapes stay shallow so the fold stays cheap
inline int red_grid(long long M, int C) {
  int groups = BLOCK / (C >> 3);
  if (groups < 1) groups = 1;
  long long need = (M + groups - 1) / groups;
  // cap by work too: each block's transposed-partial flush scatters 2C
  // 4-byte stores (stride RED_BLOCKS), so small tensors must stay shallow —
  // one block per ~48K elements keeps the scatter ≪ the streamed reads
  long long work_cap = (M * (long long)C) / 49152;
  if (work_cap < 64) work_cap = 64;
  if (need > work_cap) need = work_cap;
  if (need < 1) need = 1;
  return (int)(need < RED_BLOCKS ? need : RED_BLOCKS);
}

// ---------------------------------------------------------------------------
// finalize: mean/invstd from sums; running-stat update (1 thread/channel)
// ---------------------------------------------------------------------------

// one wave per channel: lanes stride the transposed partial rows
__global__ void bn_finalize_kernel(const float* __restrict__ partial,
                                   long long M, int C, float eps,
                                   float momentum,
                                   float* __restrict__ mean_out,
                                   float* __restrict__ invstd_out,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int update_running, int nblocks) {
  int c = blockIdx.x;
  if (c >= C) return;
  float sum = wave_row_sum(partial + (long long)c * RED_BLOCKS, nblocks);
  float sumsq = wave_row_sum(partial + (long long)(C + c) * RED_BLOCKS,
                             nblocks);
  if (threadIdx.x != 0) return;
  float n = (float)M;
  float mean = sum / n;
  float var = sumsq / n - mean * mean;
  if (var < 0.0f) var = 0.0f;
  mean_out[c] = mean;
  invstd_out[c] = rsqrtf(var + eps);
  if (update_running) {
    float unbiased = (M > 1) ? var * n / (n - 1.0f) : var;
    running_mean[c] += momentum * (mean - running_mean[c]);
    running_var[c] += momentum * (unbiased - running_var[c]);
  }
}

// ---------------------------------------------------------------------------
// fwd apply: y = [relu]( (x-mean)*invstd*gamma + beta [+ res] )
// ---------------------------------------------------------------------------

// With RELU the kernel also emits a 1-bit activation mask (one byte per
// 8-channel slot) so the backward never re-reads y — 16 B of y becomes
// 1 B of mask on the backward's critical path.
template <bool RELU, bool RES>
__global__ void bn_fwd_apply_kernel(const bf16* __restrict__ x,
                                    const bf16* __restrict__ res,
                                    bf16* __restrict__ y, long long M, int C,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta,
                                    unsigned char* __restrict__ mask) {
  const int cpt = C >> 3;
  const long long total = M * cpt;
  for (long long idx = (long long)blockIdx.x * BLOCK + threadIdx.x;
       idx < total; idx += (long long)gridDim.x * BLOCK) {
    const long long row = idx / cpt;
    const int c8 = (int)(idx - row * cpt);
    const int c0 = c8 << 3;
    const long long off = row * C + c0;
    float vals[8], rv[8];
    load8(x + off, vals);
    if (RES) load8(res + off, rv);
    unsigned char mbits = 0;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float scale = gamma[c0 + i] * invstd[c0 + i];
      float v = (vals[i] - mean[c0 + i]) * scale + beta[c0 + i];
      if (RES) v += rv[i];
      if (RELU) {
        if (v > 0.0f) mbits |= (unsigned char)(1u << i);
        else v = 0.0f;
      }
      vals[i] = v;
    }
    store8(y + off, vals);
    if (RELU) mask[idx] = mbits;
  }
}

// ---------------------------------------------------------------------------
// bwd reduce: dz = dy masked by y>0 (if RELU);
//   partial[b][c] = Σ_rows(b) dz ; partial[b][C+c] = Σ dz * xhat
// folded by bn_fold_kernel into sums2[2C].
// ---------------------------------------------------------------------------

template <bool RELU>
__global__ void bn_bwd_reduce_kernel(const bf16* __restrict__ x,
                                     const bf16* __restrict__ dy,
                                     const unsigned char* __restrict__ mask,
                                     long long M,
                                     int C, const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ partial) {
  const int cpt = C >> 3;
  const int groups = max(1, BLOCK / cpt);
  const int t = threadIdx.x;
  const int g = t / cpt;
  const int c8 = t - g * cpt;
  __shared__ float lds[BLOCK * 8];

  F8 s1, s2;
  s1.zero();
  s2.zero();
  if (g < groups) {
    float xv[8], dv[8];
    for (long long row = (long long)blockIdx.x * groups + g; row < M;
         row += (long long)gridDim.x * groups) {
      const long long off = row * C + (c8 << 3);
      load8(x + off, xv);
      load8(dy + off, dv);
      const unsigned char mbits = RELU ? mask[row * cpt + c8] : 0;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int c = (c8 << 3) + i;
        float dz = RELU ? ((mbits >> i) & 1 ? dv[i] : 0.0f) : dv[i];
        float xhat = (xv[i] - mean[c]) * invstd[c];
        s1.v[i] += dz;
        s2.v[i] += dz * xhat;
      }
    }
  }
  // transposed partial layout [row][RED_BLOCKS], same as the fwd reduce
#pragma unroll
  for (int i = 0; i < 8; ++i) lds[t * 8 + i] = s1.v[i];
  __syncthreads();
  if (g == 0 && c8 < cpt) {
    for (int gg = 1; gg < groups; ++gg)
#pragma unroll
      for (int i = 0; i < 8; ++i) s1.v[i] += lds[(gg * cpt + c8) * 8 + i];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      partial[(long long)((c8 << 3) + i) * RED_BLOCKS + blockIdx.x] =
          s1.v[i];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 8; ++i) lds[t * 8 + i] = s2.v[i];
  __syncthreads();
  if (g == 0 && c8 < cpt) {
    for (int gg = 1; gg < groups; ++gg)
#pragma unroll
      for (int i = 0; i < 8; ++i) s2.v[i] += lds[(gg * cpt + c8) * 8 + i];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      partial[(long long)(C + (c8 << 3) + i) * RED_BLOCKS + blockIdx.x] =
          s2.v[i];
  }
}

// fold the transposed partials into sums2[2C] — one wave per row
__global__ void bn_fold_kernel(const float* __restrict__ partial, int C,
                               float* __restrict__ sums2, int nblocks) {
  int row = blockIdx.x;
  if (row >= 2 * C) return;
  float acc = wave_row_sum(partial + (long long)row * RED_BLOCKS, nblocks);
  if (threadIdx.x == 0) sums2[row] = acc;
}

// ---------------------------------------------------------------------------
// bwd apply: dx = gamma*invstd*(dz - s1/M - xhat*s2/M); dres = dz
// ---------------------------------------------------------------------------

template <bool RELU, bool RES>
__global__ void bn_bwd_apply_kernel(const bf16* __restrict__ x,
                                    const bf16* __restrict__ dy,
                                    const unsigned char* __restrict__ mask,
                                    bf16* __restrict__ dx,
                                    bf16* __restrict__ dres, long long M,
                                    int C, const float* __restrict__ mean,
                                    const float* __restrict__ invstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ sums2) {
  const int cpt = C >> 3;
  const long long total = M * cpt;
  const float rn = 1.0f / (float)M;
  for (long long idx = (long long)blockIdx.x * BLOCK + threadIdx.x;
       idx < total; idx += (long long)gridDim.x * BLOCK) {
    const long long row = idx / cpt;
    const int c8 = (int)(idx - row * cpt);
    const int c0 = c8 << 3;
    const long long off = row * C + c0;
    float xv[8], dv[8], dzv[8];
    load8(x + off, xv);
    load8(dy + off, dv);
    const unsigned char mbits = RELU ? mask[idx] : 0;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int c = c0 + i;
      float dz = RELU ? ((mbits >> i) & 1 ? dv[i] : 0.0f) : dv[i];
      dzv[i] = dz;
      float xhat = (xv[i] - mean[c]) * invstd[c];
      dv[i] = gamma[c] * invstd[c] *
              (dz - sums2[c] * rn - xhat * sums2[C + c] * rn);
    }
    store8(dx + off, dv);
    if (RES) store8(dres + off, dzv);
  }
}

inline int grid_for_rows(long long M, int C) {
  int groups = max(1, BLOCK / (C >> 3));
  long long blocks = (M + groups - 1) / groups;
  return (int)(blocks < MAX_GRID ? (blocks > 0 ? blocks : 1) : MAX_GRID);
}

inline int grid_for_elems(long long elems) {
  long long blocks = (elems + BLOCK - 1) / BLOCK;
  return (int)(blocks < MAX_GRID ? (blocks > 0 ? blocks : 1) : MAX_GRID);
}

}  // namespace

#define STREAM reinterpret_cast<hipStream_t>(stream)

extern "C" {

int bps_bn_red_blocks(void) { return RED_BLOCKS; }

int bps_bn_reduce(const void* x, long long M, int C, void* partial,
                  void* stream) {
  if ((C & 7) || C > 2048) return -1;
  hipLaunchKernelGGL(bn_reduce_kernel, dim3(red_grid(M, C)), dim3(BLOCK), 0,
                     STREAM, (const bf16*)x, M, C, (float*)partial);
  return (int)hipGetLastError();
}

int bps_bn_finalize(const void* sums, long long M, int C, float eps,
                    float momentum, void* mean_out, void* invstd_out,
                    void* running_mean, void* running_var, int update_running,
                    void* stream) {
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(64), 0,
                     STREAM, (const float*)sums, M, C, eps, momentum,
                     (float*)mean_out, (float*)invstd_out,
                     (float*)running_mean, (float*)running_var,
                     update_running, red_grid(M, C));
  return (int)hipGetLastError();
}

int bps_bn_fwd_apply(const void* x, const void* res, void* y, long long M,
                     int C, const void* mean, const void* invstd,
                     const void* gamma, const void* beta, int relu,
                     void* mask, void* stream) {
  if ((C & 7) || C > 2048) return -1;
  if (relu && !mask) return -2;
  int g = grid_for_elems(M * (C >> 3));
#define LAUNCH_FWD(R, S)                                                    \
  hipLaunchKernelGGL((bn_fwd_apply_kernel<R, S>), dim3(g), dim3(BLOCK), 0,  \
                     STREAM, (const bf16*)x, (const bf16*)res, (bf16*)y, M, \
                     C, (const float*)mean, (const float*)invstd,           \
                     (const float*)gamma, (const float*)beta,               \
                     (unsigned char*)mask)
  if (relu && res) LAUNCH_FWD(true, true);
  else if (relu) LAUNCH_FWD(true, false);
  else if (res) LAUNCH_FWD(false, true);
  else LAUNCH_FWD(false, false);
#undef LAUNCH_FWD
  return (int)hipGetLastError();
}

int bps_bn_bwd_reduce(const void* x, const void* dy, const void* mask,
                      long long M, int C, const void* mean,
                      const void* invstd, void* partial, int relu,
                      void* stream) {
  if ((C & 7) || C > 2048) return -1;
  if (relu && !mask) return -2;
  int g = red_grid(M, C);
  if (relu)
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<true>), dim3(g),
                       dim3(BLOCK), 0, STREAM, (const bf16*)x,
                       (const bf16*)dy, (const unsigned char*)mask, M, C,
                       (const float*)mean, (const float*)invstd,
                       (float*)partial);
  else
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<false>), dim3(g),
                       dim3(BLOCK), 0, STREAM, (const bf16*)x,
                       (const bf16*)dy, (const unsigned char*)mask, M, C,
                       (const float*)mean, (const float*)invstd,
                       (float*)partial);
  return (int)hipGetLastError();
}

int bps_bn_fold(const void* partial, long long M, int C, void* sums2,
                void* stream) {
  hipLaunchKernelGGL(bn_fold_kernel, dim3(2 * C), dim3(64), 0, STREAM,
                     (const float*)partial, C, (float*)sums2,
                     red_grid(M, C));
  return (int)hipGetLastError();
}

int bps_bn_bwd_apply(const void* x, const void* dy, const void* mask,
                     void* dx,
                     void* dres, long long M, int C, const void* mean,
                     const void* invstd, const void* gamma, const void* sums2,
                     int relu, void* stream) {
  if ((C & 7) || C > 2048) return -1;
  if (relu && !mask) return -2;
  int g = grid_for_elems(M * (C >> 3));
#define LAUNCH_BWD(R, S)                                                     \
  hipLaunchKernelGGL((bn_bwd_apply_kernel<R, S>), dim3(g), dim3(BLOCK), 0,   \
                     STREAM, (const bf16*)x, (const bf16*)dy,                \
                     (const unsigned char*)mask, (bf16*)dx, (bf16*)dres,     \
                     M, C,                                                   \
                     (const float*)mean, (const float*)invstd,               \
                     (const float*)gamma, (const float*)sums2)
  if (relu && dres) LAUNCH_BWD(true, true);
  else if (relu) LAUNCH_BWD(true, false);
  else if (dres) LAUNCH_BWD(false, true);
  else LAUNCH_BWD(false, false);
#undef LAUNCH_BWD
  return (int)hipGetLastError();
}

}  // extern "C"
