// Probe which hipBLASLt epilogues have kernels on gfx950 for a typical
// MLP shape (bf16 in, fp32 compute).
#include <hipblaslt/hipblaslt.h>
#include <hip/hip_runtime.h>
#include <cstdio>

int main() {
  hipblasLtHandle_t h; hipblasLtCreate(&h);
  const int64_t M = 65536, N = 3072, K = 768;
  void *A, *B, *D, *bias, *aux;
  hipMalloc(&A, N*K*2); hipMalloc(&B, M*K*2); hipMalloc(&D, M*N*2);
  hipMalloc(&bias, N*4); hipMalloc(&aux, M*N*2);
  struct { const char* name; int epi; bool aux, bias; } cases[] = {
    {"DEFAULT", 1, false, false}, {"BIAS", 4, false, true},
    {"GELU", 32, false, false}, {"GELU_BIAS", 36, false, true},
    {"GELU_AUX", 160, true, false}, {"GELU_AUX_BIAS", 164, true, true},
    {"DGELU", 192, true, false}, {"DGELU_BGRAD", 208, true, true},
    {"BGRADA", 256, false, true}, {"BGRADB", 512, false, true},
  };
  for (auto& c : cases) {
    hipblasLtMatmulDesc_t op;
    hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F, HIP_R_32F);
    hipblasOperation_t ta = HIPBLAS_OP_T, tb = HIPBLAS_OP_N;
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSA, &ta, sizeof(ta));
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_TRANSB, &tb, sizeof(tb));
    hipblasLtEpilogue_t e = (hipblasLtEpilogue_t)c.epi;
    hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &e, sizeof(e));
    if (c.bias)
      hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias));
    if (c.aux) {
      hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux, sizeof(aux));
      int64_t ld = N;
      hipblasLtMatmulDescSetAttribute(op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld, sizeof(ld));
    }
    hipblasLtMatrixLayout_t la, lb, ld_;
    hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, K, N, K);
    hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, K, M, K);
    hipblasLtMatrixLayoutCreate(&ld_, HIP_R_16BF, N, M, N);
    hipblasLtMatmulPreference_t pref; hipblasLtMatmulPreferenceCreate(&pref);
    size_t ws = 64u<<20;
    hipblasLtMatmulPreferenceSetAttribute(pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws));
    hipblasLtMatmulHeuristicResult_t res[4]; int n = 0;
    hipblasStatus_t st = hipblasLtMatmulAlgoGetHeuristic(h, op, la, lb, ld_, ld_, pref, 4, res, &n);
    printf("%-14s status=%d n_algos=%d\n", c.name, (int)st, n);
    hipblasLtMatmulPreferenceDestroy(pref);
    hipblasLtMatmulDescDestroy(op);
    hipblasLtMatrixLayoutDestroy(la); hipblasLtMatrixLayoutDestroy(lb); hipblasLtMatrixLayoutDestroy(ld_);
  }
  return 0;
}
