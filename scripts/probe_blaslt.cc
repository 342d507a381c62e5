// Probe hipBLASLt epilogue support on this box: which epilogues have
// heuristic algos for bf16 in/out, fp32 compute, across trans combos.
#include <hip/hip_runtime_api.h>
#include <hipblaslt/hipblaslt.h>
#include <cstdio>

int main() {
  hipblasLtHandle_t h;
  if (hipblasLtCreate(&h) != HIPBLAS_STATUS_SUCCESS) { printf("create failed\n"); return 1; }
  struct E { const char* name; hipblasLtEpilogue_t e; bool bias; bool aux; } eps[] = {
    {"DEFAULT", HIPBLASLT_EPILOGUE_DEFAULT, false, false},
    {"BIAS", HIPBLASLT_EPILOGUE_BIAS, true, false},
    {"GELU_BIAS", HIPBLASLT_EPILOGUE_GELU_BIAS, true, false},
    {"GELU_AUX", HIPBLASLT_EPILOGUE_GELU_AUX, false, true},
    {"GELU_AUX_BIAS", HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, true, true},
    {"DGELU", HIPBLASLT_EPILOGUE_DGELU, false, true},
    {"DGELU_BGRAD", HIPBLASLT_EPILOGUE_DGELU_BGRAD, true, true},
    {"BGRADA", HIPBLASLT_EPILOGUE_BGRADA, true, false},
    {"BGRADB", HIPBLASLT_EPILOGUE_BGRADB, true, false},
  };
  hipblasOperation_t ops[2] = {HIPBLAS_OP_N, HIPBLAS_OP_T};
  int64_t m = 4096, n = 8192, k = 1024;
  for (auto& ep : eps) {
    for (int ia = 0; ia < 2; ia++) for (int ib = 0; ib < 2; ib++) {
      hipblasLtMatmulDesc_t d;
      hipblasLtMatmulDescCreate(&d, HIPBLAS_COMPUTE_32F, HIP_R_32F);
      int32_t ta = ops[ia], tb = ops[ib];
      hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_TRANSA, &ta, 4);
      hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_TRANSB, &tb, 4);
      hipblasLtEpilogue_t e = ep.e;
      hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_EPILOGUE, &e, sizeof(e));
      static char dummy[16];
      if (ep.bias) { void* p = (void*)dummy; hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &p, sizeof(p)); }
      if (ep.aux) {
        void* p = (void*)dummy; int64_t ld = m;
        hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &p, sizeof(p));
        hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld, sizeof(ld));
      }
      int64_t ar = ia ? k : m, ac = ia ? m : k;
      int64_t br = ib ? n : k, bc = ib ? k : n;
      hipblasLtMatrixLayout_t la, lb, lc;
      hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, ar, ac, ar);
      hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, br, bc, br);
      hipblasLtMatrixLayoutCreate(&lc, HIP_R_16BF, m, n, m);
      hipblasLtMatmulPreference_t pref;
      hipblasLtMatmulPreferenceCreate(&pref);
      uint64_t ws = 64ull << 20;
      hipblasLtMatmulPreferenceSetAttribute(pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, 8);
      hipblasLtMatmulHeuristicResult_t res[4];
      int found = 0;
      auto st = hipblasLtMatmulAlgoGetHeuristic(h, d, la, lb, lc, lc, pref, 4, res, &found);
      printf("%-14s %c%c: st=%d found=%d\n", ep.name, ia ? 'T' : 'N', ib ? 'T' : 'N', (int)st, found);
      hipblasLtMatmulPreferenceDestroy(pref);
      hipblasLtMatrixLayoutDestroy(la); hipblasLtMatrixLayoutDestroy(lb); hipblasLtMatrixLayoutDestroy(lc);
      hipblasLtMatmulDescDestroy(d);
    }
  }
  return 0;
}
