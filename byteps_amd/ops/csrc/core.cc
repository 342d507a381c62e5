// pybind11 module `byteps_amd.ops._core` — bindings for the HIP kernels,
// the CPU reducer/codecs, and (see kv.cc / server.cc) the KV transport and
// PS server (reference equivalent: the per-framework c_lib .so,
// torch/ops.cc:168-206, plus the server .so, server/server.cc:458-531).
//
// Tensors cross this boundary as raw (data_ptr, numel) pairs — the Python
// wrappers in byteps_amd/ops/__init__.py validate dtype/contiguity — so
// this module has no libtorch link dependency (keeps the build a single
// hipcc invocation and the .so loadable on GPU-less server nodes).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <stdexcept>

namespace py = pybind11;

// -- HIP kernel launchers (kernels.hip / compress.hip) ----------------------
extern "C" {
int bps_scale(void* x, int64_t n, float alpha, int dtype, void* stream);
int bps_cast_scale_many(const void* desc_dev, int nseg, int64_t total_vec,
                        float alpha, int src_dtype, int dst_dtype,
                        void* stream);
int bps_cast_scale(void* dst, const void* src, int64_t n, float alpha,
                   int src_dtype, int dst_dtype, void* stream);
int bps_lt_gemm_bias(const void* X, const void* W, const void* bias,
                     void* Y, int64_t M, int64_t N, int64_t K,
                     void* stream);
int bps_lt_gemm_dgelu(const void* dY2, const void* W2, const void* aux,
                      void* dY1, int64_t M, int64_t H, int64_t I,
                      void* stream);
int bps_lt_gemm_wgrad(const void* dY, const void* X, void* dW, void* db,
                      int64_t M, int64_t N, int64_t K, void* stream);
int bps_lt_gemm_dgrad(const void* dY, const void* W, void* dX, int64_t M,
                      int64_t N, int64_t K, void* stream);
int bps_axpy(void* y, const void* x, int64_t n, float alpha, int dtype,
             void* stream);
int bps_nesterov(void* g, void* m, int64_t n, float mu, int dtype,
                 void* stream);
int bps_norm(const void* x, int64_t n, int mode, void* out, int dtype,
             void* stream);
int bps_onebit_compress(const void* x, int64_t n, void* bits, void* scale_sum,
                        void* stream);
int bps_onebit_decompress(const void* bits, const void* scale_sum, int64_t n,
                          void* out, void* stream);
int bps_onebit_error(const void* x, const void* bits, const void* scale_sum,
                     int64_t n, void* err, void* stream);
int bps_randomk_compress(const void* x, int64_t n, int64_t k, uint64_t seed,
                         void* idx, void* val, void* stream);
int bps_sparse_scatter(const void* idx, const void* val, int64_t k, void* out,
                       void* stream);
int bps_sparse_error_zero(const void* idx, int64_t k, void* err, void* stream);
int bps_sparse_gather(const void* x, const void* idx, int64_t k, void* val,
                      void* stream);
int bps_dithering_compress(const void* x, int64_t n, int s, uint64_t seed,
                           int natural, const void* norm, void* code,
                           void* stream);
int bps_dithering_decompress(const void* code, int64_t n, int s, int natural,
                             const void* norm, void* out, void* stream);
int bps_fp8_compress(const void* x, int64_t n, const void* amax, void* code,
                     void* stream);
int bps_fp8_decompress(const void* code, int64_t n, const void* amax,
                       void* out, void* stream);

// -- fused batchnorm (bn.hip) ----------------------------------------------
int bps_bn_reduce(const void* x, long long M, int C, void* sums,
                  void* stream);
int bps_bn_finalize(const void* sums, long long M, int C, float eps,
                    float momentum, void* mean_out, void* invstd_out,
                    void* running_mean, void* running_var, int update_running,
                    void* stream);
int bps_bn_fwd_apply(const void* x, const void* res, void* y, long long M,
                     int C, const void* mean, const void* invstd,
                     const void* gamma, const void* beta, int relu,
                     void* mask, void* stream);
int bps_bn_bwd_reduce(const void* x, const void* dy, const void* mask,
                      long long M, int C, const void* mean,
                      const void* invstd, void* partial, int relu,
                      void* stream);
int bps_bn_fold(const void* partial, long long M, int C, void* sums2,
                void* stream);
int bps_bn_red_blocks(void);

// -- fused layernorm (ln.hip) ----------------------------------------------
int bps_ln_supported(int C);
int bps_ln_red_blocks(void);
int bps_ln_fwd(const void* x, const void* gamma, const void* beta, void* y,
               long long M, int C, float eps, void* mean, void* invstd,
               void* stream);
int bps_ln_bwd(const void* x, const void* dy, const void* gamma,
               const void* mean, const void* invstd, void* dx, long long M,
               int C, void* partial, void* stream);
int bps_ln_fold(const void* partial, long long M, int C, void* sums2,
                void* stream);
int bps_bn_bwd_apply(const void* x, const void* dy, const void* mask,
                     void* dx,
                     void* dres, long long M, int C, const void* mean,
                     const void* invstd, const void* gamma, const void* sums2,
                     int relu, void* stream);

// -- CPU reducer / codecs (cpu_reducer.cc) ---------------------------------
int bps_cpu_sum(void* dst, const void* src, int64_t n, int dtype);
int bps_cpu_sum2(void* dst, const void* src1, const void* src2, int64_t n,
                 float alpha, int dtype);
int bps_cpu_copy(void* dst, const void* src, int64_t nbytes);
int bps_cpu_scale(void* x, int64_t n, float alpha, int dtype);
int bps_cpu_topk_select(const float* x, int64_t n, int64_t k, int32_t* idx,
                        float* val);
int bps_cpu_dither_encode(const int8_t* code, int64_t n, uint8_t* out,
                          int64_t out_cap, int64_t* out_len);
int bps_cpu_dither_decode(const uint8_t* in, int64_t in_len, int64_t n,
                          int8_t* code);
int bps_cpu_onebit_compress(const float* x, int64_t n, uint64_t* bits,
                            float* scale_sum);
int bps_cpu_onebit_decompress(const uint64_t* bits, float scale_sum, int64_t n,
                              float* out);
int bps_cpu_randomk_indices(int64_t n, int64_t k, uint64_t seed, int32_t* idx);
int bps_cpu_sparse_scatter(const int32_t* idx, const float* val, int64_t k,
                           float* out);
int bps_cpu_sparse_accumulate(const int32_t* idx, const float* val, int64_t k,
                              float* acc);
int bps_cpu_dithering_compress(const float* x, int64_t n, int s, uint64_t seed,
                               int natural, float norm, int8_t* code);
int bps_cpu_dithering_decompress(const int8_t* code, int64_t n, int s,
                                 int natural, float norm, float* out);
int bps_cpu_fp8_compress(const float* x, int64_t n, float amax,
                         uint8_t* code);
int bps_cpu_fp8_decompress(const uint8_t* code, int64_t n, float amax,
                           float* out);
float bps_cpu_norm(const float* x, int64_t n, int mode);
}

namespace {

inline void check(int rc, const char* what) {
  if (rc != 0)
    throw std::runtime_error(std::string(what) + " failed with code " +
                             std::to_string(rc));
}

#define P(x) reinterpret_cast<void*>(static_cast<uintptr_t>(x))
#define CP(x) reinterpret_cast<const void*>(static_cast<uintptr_t>(x))

}  // namespace

void init_kv(py::module_& m);      // kv.cc
void init_server(py::module_& m);  // server.cc

PYBIND11_MODULE(_core, m) {
  m.doc() = "byteps_amd native core: HIP gfx950 kernels, CPU reducer, KV";
  m.attr("HIP_KERNELS") = true;

  // GPU kernels (stream = torch.cuda.current_stream().cuda_stream)
  m.def("scale", [](uintptr_t x, int64_t n, float a, int dt, uintptr_t s) {
    check(bps_scale(P(x), n, a, dt, P(s)), "bps_scale");
  });
  m.def("axpy",
        [](uintptr_t y, uintptr_t x, int64_t n, float a, int dt, uintptr_t s) {
          check(bps_axpy(P(y), CP(x), n, a, dt, P(s)), "bps_axpy");
        });
  m.def("cast_scale_many",
        [](uintptr_t desc, int nseg, int64_t total_vec, float a, int sdt,
           int ddt, uintptr_t s) {
          check(bps_cast_scale_many(CP(desc), nseg, total_vec, a, sdt, ddt,
                                    P(s)), "bps_cast_scale_many");
        });
  m.def("cast_scale",
        [](uintptr_t dst, uintptr_t src, int64_t n, float a, int sdt,
           int ddt, uintptr_t s) {
          check(bps_cast_scale(P(dst), CP(src), n, a, sdt, ddt, P(s)),
                "bps_cast_scale");
        });
  m.def("nesterov",
        [](uintptr_t g, uintptr_t mm, int64_t n, float mu, int dt,
           uintptr_t s) {
          check(bps_nesterov(P(g), P(mm), n, mu, dt, P(s)), "bps_nesterov");
        });

  // hipBLASLt epilogue-fused GEMMs (BERT MLP block)
  m.def("lt_gemm_bias",
        [](uintptr_t X, uintptr_t W, uintptr_t b, uintptr_t Y, int64_t M,
           int64_t N, int64_t K, uintptr_t s) {
          check(bps_lt_gemm_bias(CP(X), CP(W), CP(b), P(Y), M, N, K, P(s)),
                "bps_lt_gemm_bias");
        });
  m.def("lt_gemm_dgelu",
        [](uintptr_t dY2, uintptr_t W2, uintptr_t aux, uintptr_t dY1,
           int64_t M, int64_t H, int64_t I, uintptr_t s) {
          check(bps_lt_gemm_dgelu(CP(dY2), CP(W2), CP(aux), P(dY1), M, H,
                                  I, P(s)), "bps_lt_gemm_dgelu");
        });
  m.def("lt_gemm_wgrad",
        [](uintptr_t dY, uintptr_t X, uintptr_t dW, uintptr_t db, int64_t M,
           int64_t N, int64_t K, uintptr_t s) {
          check(bps_lt_gemm_wgrad(CP(dY), CP(X), P(dW),
                                  db ? P(db) : nullptr, M, N, K, P(s)),
                "bps_lt_gemm_wgrad");
        });
  m.def("lt_gemm_dgrad",
        [](uintptr_t dY, uintptr_t W, uintptr_t dX, int64_t M, int64_t N,
           int64_t K, uintptr_t s) {
          check(bps_lt_gemm_dgrad(CP(dY), CP(W), P(dX), M, N, K, P(s)),
                "bps_lt_gemm_dgrad");
        });
  m.def("norm",
        [](uintptr_t x, int64_t n, int mode, uintptr_t out, int dt,
           uintptr_t s) {
          check(bps_norm(CP(x), n, mode, P(out), dt, P(s)), "bps_norm");
        });
  m.def("onebit_compress",
        [](uintptr_t x, int64_t n, uintptr_t bits, uintptr_t sc, uintptr_t s) {
          check(bps_onebit_compress(CP(x), n, P(bits), P(sc), P(s)),
                "bps_onebit_compress");
        });
  m.def("onebit_decompress",
        [](uintptr_t bits, uintptr_t sc, int64_t n, uintptr_t out,
           uintptr_t s) {
          check(bps_onebit_decompress(CP(bits), CP(sc), n, P(out), P(s)),
                "bps_onebit_decompress");
        });
  m.def("onebit_error",
        [](uintptr_t x, uintptr_t bits, uintptr_t sc, int64_t n, uintptr_t err,
           uintptr_t s) {
          check(bps_onebit_error(CP(x), CP(bits), CP(sc), n, P(err), P(s)),
                "bps_onebit_error");
        });
  m.def("randomk_compress",
        [](uintptr_t x, int64_t n, int64_t k, uint64_t seed, uintptr_t idx,
           uintptr_t val, uintptr_t s) {
          check(bps_randomk_compress(CP(x), n, k, seed, P(idx), P(val), P(s)),
                "bps_randomk_compress");
        });
  m.def("sparse_scatter",
        [](uintptr_t idx, uintptr_t val, int64_t k, uintptr_t out,
           uintptr_t s) {
          check(bps_sparse_scatter(CP(idx), CP(val), k, P(out), P(s)),
                "bps_sparse_scatter");
        });
  m.def("sparse_error_zero",
        [](uintptr_t idx, int64_t k, uintptr_t err, uintptr_t s) {
          check(bps_sparse_error_zero(CP(idx), k, P(err), P(s)),
                "bps_sparse_error_zero");
        });
  m.def("sparse_gather",
        [](uintptr_t x, uintptr_t idx, int64_t k, uintptr_t val, uintptr_t s) {
          check(bps_sparse_gather(CP(x), CP(idx), k, P(val), P(s)),
                "bps_sparse_gather");
        });
  m.def("dithering_compress",
        [](uintptr_t x, int64_t n, int sv, uint64_t seed, int natural,
           uintptr_t norm, uintptr_t code, uintptr_t s) {
          check(bps_dithering_compress(CP(x), n, sv, seed, natural, CP(norm),
                                       P(code), P(s)),
                "bps_dithering_compress");
        });
  m.def("dithering_decompress",
        [](uintptr_t code, int64_t n, int sv, int natural, uintptr_t norm,
           uintptr_t out, uintptr_t s) {
          check(bps_dithering_decompress(CP(code), n, sv, natural, CP(norm),
                                         P(out), P(s)),
                "bps_dithering_decompress");
        });

  m.def("fp8_compress",
        [](uintptr_t x, int64_t n, uintptr_t amax, uintptr_t code,
           uintptr_t s) {
          check(bps_fp8_compress(CP(x), n, CP(amax), P(code), P(s)),
                "bps_fp8_compress");
        });
  m.def("fp8_decompress",
        [](uintptr_t code, int64_t n, uintptr_t amax, uintptr_t out,
           uintptr_t s) {
          check(bps_fp8_decompress(CP(code), n, CP(amax), P(out), P(s)),
                "bps_fp8_decompress");
        });
  m.def("cpu_fp8_compress",
        [](uintptr_t x, int64_t n, float amax, uintptr_t code) {
          check(bps_cpu_fp8_compress((const float*)P(x), n, amax,
                                     (uint8_t*)P(code)),
                "bps_cpu_fp8_compress");
        });
  m.def("cpu_fp8_decompress",
        [](uintptr_t code, int64_t n, float amax, uintptr_t out) {
          check(bps_cpu_fp8_decompress((const uint8_t*)P(code), n, amax,
                                       (float*)P(out)),
                "bps_cpu_fp8_decompress");
        });

  // fused batchnorm
  m.attr("BN_RED_BLOCKS") = bps_bn_red_blocks();
  m.def("bn_reduce", [](uintptr_t x, int64_t M, int C, uintptr_t sums,
                        uintptr_t s) {
    check(bps_bn_reduce(CP(x), M, C, P(sums), P(s)), "bps_bn_reduce");
  });
  m.def("bn_fold", [](uintptr_t partial, int64_t M, int C, uintptr_t sums2,
                      uintptr_t s) {
    check(bps_bn_fold(CP(partial), M, C, P(sums2), P(s)), "bps_bn_fold");
  });
  m.def("bn_finalize",
        [](uintptr_t sums, int64_t M, int C, float eps, float momentum,
           uintptr_t mean, uintptr_t invstd, uintptr_t rmean, uintptr_t rvar,
           int upd, uintptr_t s) {
          check(bps_bn_finalize(CP(sums), M, C, eps, momentum, P(mean),
                                P(invstd), P(rmean), P(rvar), upd, P(s)),
                "bps_bn_finalize");
        });
  m.def("bn_fwd_apply",
        [](uintptr_t x, uintptr_t res, uintptr_t y, int64_t M, int C,
           uintptr_t mean, uintptr_t invstd, uintptr_t gamma, uintptr_t beta,
           int relu, uintptr_t mask, uintptr_t s) {
          check(bps_bn_fwd_apply(CP(x), CP(res), P(y), M, C, CP(mean),
                                 CP(invstd), CP(gamma), CP(beta), relu,
                                 P(mask), P(s)),
                "bps_bn_fwd_apply");
        });
  m.def("bn_bwd_reduce",
        [](uintptr_t x, uintptr_t dy, uintptr_t y, int64_t M, int C,
           uintptr_t mean, uintptr_t invstd, uintptr_t sums2, int relu,
           uintptr_t s) {
          check(bps_bn_bwd_reduce(CP(x), CP(dy), CP(y), M, C, CP(mean),
                                  CP(invstd), P(sums2), relu, P(s)),
                "bps_bn_bwd_reduce");
        });
  m.def("bn_bwd_apply",
        [](uintptr_t x, uintptr_t dy, uintptr_t y, uintptr_t dx,
           uintptr_t dres, int64_t M, int C, uintptr_t mean, uintptr_t invstd,
           uintptr_t gamma, uintptr_t sums2, int relu, uintptr_t s) {
          check(bps_bn_bwd_apply(CP(x), CP(dy), CP(y), P(dx), P(dres), M, C,
                                 CP(mean), CP(invstd), CP(gamma), CP(sums2),
                                 relu, P(s)),
                "bps_bn_bwd_apply");
        });

  // fused layernorm
  m.attr("LN_RED_BLOCKS") = bps_ln_red_blocks();
  m.def("ln_supported", [](int C) { return bps_ln_supported(C) != 0; });
  m.def("ln_fwd",
        [](uintptr_t x, uintptr_t gamma, uintptr_t beta, uintptr_t y,
           int64_t M, int C, float eps, uintptr_t mean, uintptr_t invstd,
           uintptr_t s) {
          check(bps_ln_fwd(CP(x), CP(gamma), CP(beta), P(y), M, C, eps,
                           P(mean), P(invstd), P(s)),
                "bps_ln_fwd");
        });
  m.def("ln_bwd",
        [](uintptr_t x, uintptr_t dy, uintptr_t gamma, uintptr_t mean,
           uintptr_t invstd, uintptr_t dx, int64_t M, int C,
           uintptr_t partial, uintptr_t s) {
          check(bps_ln_bwd(CP(x), CP(dy), CP(gamma), CP(mean), CP(invstd),
                           P(dx), M, C, P(partial), P(s)),
                "bps_ln_bwd");
        });
  m.def("ln_fold", [](uintptr_t partial, int64_t M, int C, uintptr_t sums2,
                      uintptr_t s) {
    check(bps_ln_fold(CP(partial), M, C, P(sums2), P(s)), "bps_ln_fold");
  });

  // CPU reducer / codecs
  m.def("cpu_sum", [](uintptr_t d, uintptr_t s, int64_t n, int dt) {
    check(bps_cpu_sum(P(d), CP(s), n, dt), "bps_cpu_sum");
  }, py::call_guard<py::gil_scoped_release>());
  m.def("cpu_sum2",
        [](uintptr_t d, uintptr_t a, uintptr_t b, int64_t n, float alpha,
           int dt) {
          check(bps_cpu_sum2(P(d), CP(a), CP(b), n, alpha, dt), "bps_cpu_sum2");
        }, py::call_guard<py::gil_scoped_release>());
  m.def("cpu_copy", [](uintptr_t d, uintptr_t s, int64_t nbytes) {
    check(bps_cpu_copy(P(d), CP(s), nbytes), "bps_cpu_copy");
  }, py::call_guard<py::gil_scoped_release>());
  m.def("cpu_scale", [](uintptr_t x, int64_t n, float a, int dt) {
    check(bps_cpu_scale(P(x), n, a, dt), "bps_cpu_scale");
  }, py::call_guard<py::gil_scoped_release>());
  m.def("cpu_dither_encode",
        [](uintptr_t code, int64_t n, uintptr_t out, int64_t cap) {
          int64_t wlen = 0;
          int rc = bps_cpu_dither_encode((const int8_t*)CP(code), n,
                                         (uint8_t*)P(out), cap, &wlen);
          return rc == 0 ? wlen : (int64_t)-1;
        }, py::call_guard<py::gil_scoped_release>());
  m.def("cpu_dither_decode",
        [](uintptr_t in, int64_t in_len, int64_t n, uintptr_t code) {
          check(bps_cpu_dither_decode((const uint8_t*)CP(in), in_len, n,
                                      (int8_t*)P(code)),
                "bps_cpu_dither_decode");
        }, py::call_guard<py::gil_scoped_release>());
  m.def("cpu_topk_select",
        [](uintptr_t x, int64_t n, int64_t k, uintptr_t idx, uintptr_t val) {
          check(bps_cpu_topk_select((const float*)CP(x), n, k,
                                    (int32_t*)P(idx), (float*)P(val)),
                "bps_cpu_topk_select");
        }, py::call_guard<py::gil_scoped_release>());
  m.def("cpu_onebit_compress", [](uintptr_t x, int64_t n, uintptr_t bits) {
    float sc = 0.0f;
    check(bps_cpu_onebit_compress((const float*)P(x), n, (uint64_t*)P(bits),
                                  &sc),
          "bps_cpu_onebit_compress");
    return sc;
  });
  m.def("cpu_onebit_decompress",
        [](uintptr_t bits, float sc, int64_t n, uintptr_t out) {
          check(bps_cpu_onebit_decompress((const uint64_t*)P(bits), sc, n,
                                          (float*)P(out)),
                "bps_cpu_onebit_decompress");
        });
  m.def("cpu_randomk_indices",
        [](int64_t n, int64_t k, uint64_t seed, uintptr_t idx) {
          check(bps_cpu_randomk_indices(n, k, seed, (int32_t*)P(idx)),
                "bps_cpu_randomk_indices");
        });
  m.def("cpu_sparse_scatter",
        [](uintptr_t idx, uintptr_t val, int64_t k, uintptr_t out) {
          check(bps_cpu_sparse_scatter((const int32_t*)P(idx),
                                       (const float*)P(val), k,
                                       (float*)P(out)),
                "bps_cpu_sparse_scatter");
        });
  m.def("cpu_sparse_accumulate",
        [](uintptr_t idx, uintptr_t val, int64_t k, uintptr_t acc) {
          check(bps_cpu_sparse_accumulate((const int32_t*)P(idx),
                                          (const float*)P(val), k,
                                          (float*)P(acc)),
                "bps_cpu_sparse_accumulate");
        });
  m.def("cpu_dithering_compress",
        [](uintptr_t x, int64_t n, int sv, uint64_t seed, int natural,
           float norm, uintptr_t code) {
          check(bps_cpu_dithering_compress((const float*)P(x), n, sv, seed,
                                           natural, norm, (int8_t*)P(code)),
                "bps_cpu_dithering_compress");
        });
  m.def("cpu_dithering_decompress",
        [](uintptr_t code, int64_t n, int sv, int natural, float norm,
           uintptr_t out) {
          check(bps_cpu_dithering_decompress((const int8_t*)P(code), n, sv,
                                             natural, norm, (float*)P(out)),
                "bps_cpu_dithering_decompress");
        });
  m.def("cpu_norm", [](uintptr_t x, int64_t n, int mode) {
    return bps_cpu_norm((const float*)P(x), n, mode);
  });

  init_kv(m);
  init_server(m);
}
