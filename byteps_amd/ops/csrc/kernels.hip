// Elementwise / reduction HIP kernels for gfx950 (CDNA4).
//
// MI355X-native replacements for the reference's CPU SIMD reducer math
// (reference common/cpu_reducer.cc:59-424 — AVX/F16C on the host) and the
// implicit NCCL side-effect math.  All kernels:
//   - 256-thread blocks (4 wavefronts of 64),
//   - grid-stride with grid capped at 2048 blocks (256 CU × 8 blocks),
//   - vectorized 16 B/lane accesses (float4 / 8×bf16),
//   - fp32 accumulation for 16-bit dtypes.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include "common.h"

#define BLOCK 256
#define MAX_GRID 2048

namespace {

inline int grid_for(int64_t work_items) {
  int64_t blocks = (work_items + BLOCK - 1) / BLOCK;
  return (int)(blocks < MAX_GRID ? (blocks > 0 ? blocks : 1) : MAX_GRID);
}

using bf16 = __hip_bfloat16;
typedef float float4v __attribute__((ext_vector_type(4)));
typedef short short8 __attribute__((ext_vector_type(8)));

__device__ inline float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ inline bf16 f2bf(float v) { return __float2bfloat16(v); }

// ---------------------------------------------------------------------------
// scale: x *= alpha  (the gradient-averaging divide fused into one pass;
// reference divided on the framework side, torch/ops.cc:78-91)
// ---------------------------------------------------------------------------

__global__ void scale_f32_kernel(float* __restrict__ x, int64_t n, float a) {
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * BLOCK * 4;
  for (int64_t i = i0; i + 3 < n; i += stride) {
    float4v v = *reinterpret_cast<float4v*>(x + i);
    v.x *= a; v.y *= a; v.z *= a; v.w *= a;
    *reinterpret_cast<float4v*>(x + i) = v;
  }
  // tail
  int64_t tail_start = (n / 4) * 4;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (t < n && blockIdx.x * BLOCK + threadIdx.x < 4) x[t] *= a;
}

__global__ void scale_bf16_kernel(bf16* __restrict__ x, int64_t n, float a) {
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * BLOCK * 8;
  for (int64_t i = i0; i + 7 < n; i += stride) {
    short8 v = *reinterpret_cast<short8*>(x + i);
    bf16* e = reinterpret_cast<bf16*>(&v);
#pragma unroll
    for (int j = 0; j < 8; ++j) e[j] = f2bf(bf2f(e[j]) * a);
    *reinterpret_cast<short8*>(x + i) = v;
  }
  int64_t tail_start = (n / 8) * 8;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (t < n && blockIdx.x * BLOCK + threadIdx.x < 8) x[t] = f2bf(bf2f(x[t]) * a);
}

// ---------------------------------------------------------------------------
// axpy: y += a * x  (error-feedback / momentum building block; fp32
// accumulate for bf16 — the reference accumulated fp16 in fp32 on the CPU
// via F16C, common/cpu_reducer.cc:96-141)
// ---------------------------------------------------------------------------

__global__ void axpy_f32_kernel(float* __restrict__ y,
                                const float* __restrict__ x, int64_t n,
                                float a) {
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * BLOCK * 4;
  for (int64_t i = i0; i + 3 < n; i += stride) {
    float4v vy = *reinterpret_cast<float4v*>(y + i);
    const float4v vx = *reinterpret_cast<const float4v*>(x + i);
    vy.x = fmaf(a, vx.x, vy.x);
    vy.y = fmaf(a, vx.y, vy.y);
    vy.z = fmaf(a, vx.z, vy.z);
    vy.w = fmaf(a, vx.w, vy.w);
    *reinterpret_cast<float4v*>(y + i) = vy;
  }
  int64_t tail_start = (n / 4) * 4;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (t < n && blockIdx.x * BLOCK + threadIdx.x < 4) y[t] = fmaf(a, x[t], y[t]);
}

__global__ void axpy_bf16_kernel(bf16* __restrict__ y,
                                 const bf16* __restrict__ x, int64_t n,
                                 float a) {
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * BLOCK * 8;
  for (int64_t i = i0; i + 7 < n; i += stride) {
    short8 vy = *reinterpret_cast<short8*>(y + i);
    const short8 vx = *reinterpret_cast<const short8*>(x + i);
    bf16* ey = reinterpret_cast<bf16*>(&vy);
    const bf16* ex = reinterpret_cast<const bf16*>(&vx);
#pragma unroll
    for (int j = 0; j < 8; ++j) ey[j] = f2bf(fmaf(a, bf2f(ex[j]), bf2f(ey[j])));
    *reinterpret_cast<short8*>(y + i) = vy;
  }
  int64_t tail_start = (n / 8) * 8;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (t < n && blockIdx.x * BLOCK + threadIdx.x < 8)
    y[t] = f2bf(fmaf(a, bf2f(x[t]), bf2f(y[t])));
}

// ---------------------------------------------------------------------------
// nesterov momentum (fused): m = mu*m + g ; g = g + mu*m
// (reference impl/nesterov_momentum.cc:39-49 — two passes on CPU; one
// fused pass here)
// ---------------------------------------------------------------------------

__global__ void nesterov_f32_kernel(float* __restrict__ g,
                                    float* __restrict__ m, int64_t n,
                                    float mu) {
  int64_t i0 = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = i0; i < n; i += stride) {
    float mi = fmaf(mu, m[i], g[i]);
    m[i] = mi;
    g[i] = fmaf(mu, mi, g[i]);
  }
}

// ---------------------------------------------------------------------------
// sum reduce: out[0] += sum over n of |x| or x^2 or max|x| — norm helpers
// for the codecs.  Wave shuffle reduce → LDS → one atomic per block
// (guide §6 Guideline 12).
// ---------------------------------------------------------------------------

template <int MODE>  // 0 = sum|x|, 1 = sum x^2, 2 = max|x|
__global__ void norm_f32_kernel(const float* __restrict__ x, int64_t n,
                                float* __restrict__ out) {
  __shared__ float warp_part[BLOCK / 64];
  float acc = (MODE == 2) ? 0.0f : 0.0f;
  int64_t i0 = (int64_t)(blockIdx.x * BLOCK + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * BLOCK * 4;
  for (int64_t i = i0; i + 3 < n; i += stride) {
    const float4v v = *reinterpret_cast<const float4v*>(x + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float e = fabsf(((const float*)&v)[j]);
      if (MODE == 0) acc += e;
      else if (MODE == 1) acc += e * e;
      else acc = fmaxf(acc, e);
    }
  }
  int64_t tail_start = (n / 4) * 4;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (t < n && blockIdx.x * BLOCK + threadIdx.x < 4) {
    float e = fabsf(x[t]);
    if (MODE == 0) acc += e;
    else if (MODE == 1) acc += e * e;
    else acc = fmaxf(acc, e);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float o = __shfl_down(acc, off, 64);
    acc = (MODE == 2) ? fmaxf(acc, o) : acc + o;
  }
  if ((threadIdx.x & 63) == 0) warp_part[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float b = warp_part[0];
    for (int w = 1; w < BLOCK / 64; ++w)
      b = (MODE == 2) ? fmaxf(b, warp_part[w]) : b + warp_part[w];
    if (MODE == 2) {
      // atomic max on non-negative floats via int compare (monotone map)
      atomicMax(reinterpret_cast<int*>(out), __float_as_int(b));
    } else {
      atomicAdd(out, b);
    }
  }
}

// ---------------------------------------------------------------------------
// cast_scale: dst = (DST)(float(src) * a) — the post-collective averaging
// divide fused with the reduced-precision wire cast-back, over MANY
// buckets in ONE launch (replaces per-bucket copy_ + _foreach_div_ on the
// default hot path; the reference divided per-tensor on the framework
// side, torch/ops.cc:78-91).
//
// desc layout (int64, device memory): [dst_ptr x nseg][src_ptr x nseg]
// [vec_prefix x nseg+1].  Segment lengths are multiples of VEC (bucket
// alignment is lcm(64, world) elements), so there are no tails and every
// access is a full 16/8-byte vector.
// ---------------------------------------------------------------------------

using half_t = __half;

__device__ inline float to_f(float v) { return v; }
__device__ inline float to_f(bf16 v) { return __bfloat162float(v); }
__device__ inline float to_f(half_t v) { return __half2float(v); }
__device__ inline void from_f(float v, float* d) { *d = v; }
__device__ inline void from_f(float v, bf16* d) { *d = __float2bfloat16(v); }
__device__ inline void from_f(float v, half_t* d) { *d = __float2half(v); }

template <typename SRC, typename DST, int VEC>
__global__ void cast_scale_many_kernel(const int64_t* __restrict__ desc,
                                       int nseg, int64_t total_vec,
                                       float a) {
  const int64_t* dstp = desc;
  const int64_t* srcp = desc + nseg;
  const int64_t* pref = desc + 2 * nseg;  // nseg+1 entries, vec units
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; i < total_vec; i += stride) {
    int lo = 0, hi = nseg - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (pref[mid] <= i) lo = mid; else hi = mid - 1;
    }
    int64_t off = (i - pref[lo]) * VEC;
    const SRC* s = reinterpret_cast<const SRC*>(srcp[lo]) + off;
    DST* d = reinterpret_cast<DST*>(dstp[lo]) + off;
    struct __attribute__((aligned(8))) SV { SRC v[VEC]; };
    struct __attribute__((aligned(8))) DV { DST v[VEC]; };
    SV sv = *reinterpret_cast<const SV*>(s);
    DV dv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) from_f(to_f(sv.v[j]) * a, &dv.v[j]);
    *reinterpret_cast<DV*>(d) = dv;
  }
}

// single-segment variant (cross-barrier per-bucket path): no descriptor
// indirection, tail handled scalar
template <typename SRC, typename DST>
__global__ void cast_scale_kernel(DST* __restrict__ d,
                                  const SRC* __restrict__ s, int64_t n,
                                  float a) {
  int64_t i0 = ((int64_t)blockIdx.x * BLOCK + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * BLOCK * 4;
  for (int64_t i = i0; i + 3 < n; i += stride) {
    struct __attribute__((aligned(8))) SV { SRC v[4]; };
    struct __attribute__((aligned(8))) DV { DST v[4]; };
    SV sv = *reinterpret_cast<const SV*>(s + i);
    DV dv;
#pragma unroll
    for (int j = 0; j < 4; ++j) from_f(to_f(sv.v[j]) * a, &dv.v[j]);
    *reinterpret_cast<DV*>(d + i) = dv;
  }
  int64_t tail_start = (n / 4) * 4;
  int64_t t = tail_start + blockIdx.x * BLOCK + threadIdx.x;
  if (t < n && blockIdx.x * BLOCK + threadIdx.x < 4)
    from_f(to_f(s[t]) * a, &d[t]);
}

}  // namespace

// ---------------------------------------------------------------------------
// C ABI launchers (dtype: 0=f32, 2=bf16 — bpsamd::DType)
// ---------------------------------------------------------------------------

#define STREAM reinterpret_cast<hipStream_t>(stream)

extern "C" {

int bps_scale(void* x, int64_t n, float alpha, int dtype, void* stream) {
  int g = grid_for((n + 3) / 4);
  if (dtype == 0)
    hipLaunchKernelGGL(scale_f32_kernel, dim3(g), dim3(BLOCK), 0, STREAM,
                       (float*)x, n, alpha);
  else if (dtype == 2)
    hipLaunchKernelGGL(scale_bf16_kernel, dim3(grid_for((n + 7) / 8)),
                       dim3(BLOCK), 0, STREAM, (bf16*)x, n, alpha);
  else
    return -1;
  return (int)hipGetLastError();
}

int bps_axpy(void* y, const void* x, int64_t n, float alpha, int dtype,
             void* stream) {
  if (dtype == 0)
    hipLaunchKernelGGL(axpy_f32_kernel, dim3(grid_for((n + 3) / 4)),
                       dim3(BLOCK), 0, STREAM, (float*)y, (const float*)x, n,
                       alpha);
  else if (dtype == 2)
    hipLaunchKernelGGL(axpy_bf16_kernel, dim3(grid_for((n + 7) / 8)),
                       dim3(BLOCK), 0, STREAM, (bf16*)y, (const bf16*)x, n,
                       alpha);
  else
    return -1;
  return (int)hipGetLastError();
}

int bps_nesterov(void* g, void* m, int64_t n, float mu, int dtype,
                 void* stream) {
  if (dtype != 0) return -1;
  hipLaunchKernelGGL(nesterov_f32_kernel, dim3(grid_for(n)), dim3(BLOCK), 0,
                     STREAM, (float*)g, (float*)m, n, mu);
  return (int)hipGetLastError();
}

// dst/src dtype codes: 0=f32, 1=f16, 2=bf16.  total_vec = total elements
// divided by the vector width (4 for mixed/f32 pairs, 8 for 16-bit→16-bit).
int bps_cast_scale_many(const void* desc_dev, int nseg, int64_t total_vec,
                        float alpha, int src_dtype, int dst_dtype,
                        void* stream) {
  int g = grid_for(total_vec);
  const int64_t* d = (const int64_t*)desc_dev;
#define LAUNCH(S, D, V)                                                     \
  hipLaunchKernelGGL((cast_scale_many_kernel<S, D, V>), dim3(g),            \
                     dim3(BLOCK), 0, STREAM, d, nseg, total_vec, alpha)
  if (src_dtype == 0 && dst_dtype == 0) LAUNCH(float, float, 4);
  else if (src_dtype == 2 && dst_dtype == 0) LAUNCH(bf16, float, 4);
  else if (src_dtype == 1 && dst_dtype == 0) LAUNCH(half_t, float, 4);
  else if (src_dtype == 2 && dst_dtype == 2) LAUNCH(bf16, bf16, 8);
  else if (src_dtype == 1 && dst_dtype == 1) LAUNCH(half_t, half_t, 8);
  else if (src_dtype == 0 && dst_dtype == 2) LAUNCH(float, bf16, 4);
  else return -1;
#undef LAUNCH
  return (int)hipGetLastError();
}

int bps_cast_scale(void* dst, const void* src, int64_t n, float alpha,
                   int src_dtype, int dst_dtype, void* stream) {
  int g = grid_for((n + 3) / 4);
#define LAUNCH(S, D)                                                        \
  hipLaunchKernelGGL((cast_scale_kernel<S, D>), dim3(g), dim3(BLOCK), 0,    \
                     STREAM, (D*)dst, (const S*)src, n, alpha)
  if (src_dtype == 0 && dst_dtype == 0) LAUNCH(float, float);
  else if (src_dtype == 2 && dst_dtype == 0) LAUNCH(bf16, float);
  else if (src_dtype == 1 && dst_dtype == 0) LAUNCH(half_t, float);
  else if (src_dtype == 2 && dst_dtype == 2) LAUNCH(bf16, bf16);
  else if (src_dtype == 1 && dst_dtype == 1) LAUNCH(half_t, half_t);
  else if (src_dtype == 0 && dst_dtype == 2) LAUNCH(float, bf16);
  else return -1;
#undef LAUNCH
  return (int)hipGetLastError();
}

// mode: 0 = L1 (sum|x|), 1 = sum of squares, 2 = max|x|.
// out must be pre-zeroed (one float).
int bps_norm(const void* x, int64_t n, int mode, void* out, int dtype,
             void* stream) {
  if (dtype != 0) return -1;
  int g = grid_for((n + 3) / 4);
  if (mode == 0)
    hipLaunchKernelGGL((norm_f32_kernel<0>), dim3(g), dim3(BLOCK), 0, STREAM,
                       (const float*)x, n, (float*)out);
  else if (mode == 1)
    hipLaunchKernelGGL((norm_f32_kernel<1>), dim3(g), dim3(BLOCK), 0, STREAM,
                       (const float*)x, n, (float*)out);
  else
    hipLaunchKernelGGL((norm_f32_kernel<2>), dim3(g), dim3(BLOCK), 0, STREAM,
                       (const float*)x, n, (float*)out);
  return (int)hipGetLastError();
}

}  // extern "C"
