// hipBLASLt fused linear layers for MI355X — GEMM + epilogue in one kernel.
//
// Replaces the reference's cuBLAS GEMM + separate bias/GELU torch kernels
// (SURVEY.md §2.3 "cuBLAS GEMM" row; VERDICT r01 "GELU + bias-grad are
// unfused torch kernels, ~6% of step"). Epilogues used:
//   fwd:  BIAS, GELU_AUX_BIAS (stores pre-GELU z for the backward)
//   bwd:  DGELU_BGRAD (dH->dZ with db_up), BGRADB (db on the dW GEMM)
// All GEMMs bf16 in / fp32 compute. Per-(shape, epilogue) algorithms are
// picked by timing the top heuristic candidates once and cached — a
// lightweight in-process TunableOp (torch's TunableOp cannot see epilogue
// GEMMs).
//
// Matrix convention: torch row-major [M,K] x [N,K]^T -> [M,N] is computed
// column-major as D[N,M] = op(A=W[K,N], T) * op(B=x[K,M], N); the bias
// vector length N matches D's rows, broadcast over columns.

#include <hipblaslt/hipblaslt.h>

#include <mutex>
#include <unordered_map>
#include <vector>

#include "host_common.h"

namespace photon_hip {

namespace {

#define LT_CHECK(expr)                                                      \
  do {                                                                      \
    hipblasStatus_t st_ = (expr);                                           \
    TORCH_CHECK(st_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)st_, \
                " at " #expr);                                              \
  } while (0)

constexpr size_t kWorkspaceBytes = 64u << 20;

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    LT_CHECK(hipblasLtCreate(&hh));
    return hh;
  }();
  return h;
}

struct AlgoKey {
  int64_t m, n, k;
  int epi;
  bool operator==(const AlgoKey& o) const {
    return m == o.m && n == o.n && k == o.k && epi == o.epi;
  }
};
struct AlgoKeyHash {
  size_t operator()(const AlgoKey& k) const {
    return std::hash<int64_t>()(k.m * 1315423911 ^ k.n * 2654435761u ^
                                k.k * 97531 ^ k.epi);
  }
};

std::unordered_map<AlgoKey, hipblasLtMatmulAlgo_t, AlgoKeyHash> g_algo_cache;
std::mutex g_algo_mutex;

struct MatmulPlan {
  hipblasLtMatmulDesc_t op = nullptr;
  hipblasLtMatrixLayout_t la = nullptr, lb = nullptr, ld = nullptr;
  ~MatmulPlan() {
    if (op) hipblasLtMatmulDescDestroy(op);
    if (la) hipblasLtMatrixLayoutDestroy(la);
    if (lb) hipblasLtMatrixLayoutDestroy(lb);
    if (ld) hipblasLtMatrixLayoutDestroy(ld);
  }
};

// One matmul with epilogue: D[n x m] (col) = op(A) * op(B) + epilogue.
// All buffers bf16; compute fp32. aux/bias optional by epilogue.
void lt_matmul(const void* A, hipblasOperation_t opA, int64_t a_rows,
               int64_t a_cols, const void* B, hipblasOperation_t opB,
               int64_t b_rows, int64_t b_cols, void* D, int64_t d_rows,
               int64_t d_cols, hipblasLtEpilogue_t epi, const void* bias,
               void* bias_grad_out, void* aux, int64_t aux_ld,
               at::Tensor& workspace) {
  MatmulPlan p;
  LT_CHECK(hipblasLtMatmulDescCreate(&p.op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_TRANSA, &opA, sizeof(opA)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_TRANSB, &opB, sizeof(opB)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
  if (bias != nullptr || bias_grad_out != nullptr) {
    const void* bptr = bias ? bias : (const void*)bias_grad_out;
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bptr, sizeof(bptr)));
  }
  if (aux != nullptr) {
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &aux, sizeof(aux)));
    LT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.op, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld,
        sizeof(aux_ld)));
  }
  LT_CHECK(hipblasLtMatrixLayoutCreate(&p.la, HIP_R_16BF, a_rows, a_cols,
                                       a_rows));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&p.lb, HIP_R_16BF, b_rows, b_cols,
                                       b_rows));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&p.ld, HIP_R_16BF, d_rows, d_cols,
                                       d_rows));

  const float alpha = 1.f, beta = 0.f;
  hipStream_t stream = cur_stream();

  // algo: cached per (m, n, k, epilogue); first encounter times the top
  // heuristic candidates (3 reps each) and keeps the fastest
  const int64_t em = (opA == HIPBLAS_OP_T) ? a_cols : a_rows;
  const int64_t ek = (opA == HIPBLAS_OP_T) ? a_rows : a_cols;
  AlgoKey key{em, d_cols, ek,
              (int)epi | ((opA == HIPBLAS_OP_T) ? 1 << 16 : 0) |
                  ((opB == HIPBLAS_OP_T) ? 1 << 17 : 0)};
  hipblasLtMatmulAlgo_t algo;
  bool have = false;
  {
    std::lock_guard<std::mutex> lk(g_algo_mutex);
    auto it = g_algo_cache.find(key);
    if (it != g_algo_cache.end()) {
      algo = it->second;
      have = true;
    }
  }
  if (!have) {
    hipblasLtMatmulPreference_t pref;
    LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws = kWorkspaceBytes;
    LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
    constexpr int kMaxAlgos = 24;
    hipblasLtMatmulHeuristicResult_t results[kMaxAlgos];
    int n_results = 0;
    LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(lt_handle(), p.op, p.la, p.lb,
                                             p.ld, p.ld, pref, kMaxAlgos,
                                             results, &n_results));
    hipblasLtMatmulPreferenceDestroy(pref);
    TORCH_CHECK(n_results > 0, "hipblaslt: no algo for shape m=", em,
                " n=", d_cols, " k=", ek, " epi=", (int)epi);
    // time candidates (one warmup + 3 timed reps each)
    int best = 0;
    float best_ms = 1e30f;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    for (int i = 0; i < n_results; ++i) {
      auto run = [&]() {
        return hipblasLtMatmul(lt_handle(), p.op, &alpha, A, p.la, B, p.lb,
                               &beta, D, p.ld, D, p.ld, &results[i].algo,
                               workspace.data_ptr(), kWorkspaceBytes, stream);
      };
      if (run() != HIPBLAS_STATUS_SUCCESS) continue;
      (void)hipEventRecord(e0, stream);
      for (int r = 0; r < 3; ++r) (void)run();
      (void)hipEventRecord(e1, stream);
      (void)hipEventSynchronize(e1);
      float ms = 1e30f;
      (void)hipEventElapsedTime(&ms, e0, e1);
      if (ms < best_ms) {
        best_ms = ms;
        best = i;
      }
    }
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    algo = results[best].algo;
    std::lock_guard<std::mutex> lk(g_algo_mutex);
    g_algo_cache.emplace(key, algo);
  }

  LT_CHECK(hipblasLtMatmul(lt_handle(), p.op, &alpha, A, p.la, B, p.lb,
                           &beta, D, p.ld, D, p.ld, &algo,
                           workspace.data_ptr(), kWorkspaceBytes, stream));
}

at::Tensor ws_tensor(const at::Tensor& like) {
  return at::empty({(int64_t)kWorkspaceBytes},
                   like.options().dtype(at::kByte));
}

void check2d(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.dtype() == at::kBFloat16 && t.dim() == 2 &&
                  t.is_contiguous(),
              name, ": need contiguous 2-D bf16 CUDA tensor");
}

}  // namespace

// y = x @ W^T (+ bias) [+ GELU, storing pre-GELU aux]; returns (y, aux).
std::vector<at::Tensor> lt_linear_fwd(at::Tensor x, at::Tensor w,
                                      c10::optional<at::Tensor> bias,
                                      bool gelu) {
  check2d(x, "x");
  check2d(w, "w");
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "lt_linear_fwd: shape mismatch");
  auto y = at::empty({M, N}, x.options());
  at::Tensor aux;
  hipblasLtEpilogue_t epi = HIPBLASLT_EPILOGUE_DEFAULT;
  const void* bptr = nullptr;
  void* aptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->dtype() == at::kBFloat16, "bias must be bf16");
    bptr = bias->data_ptr();
    epi = gelu ? HIPBLASLT_EPILOGUE_GELU_AUX_BIAS : HIPBLASLT_EPILOGUE_BIAS;
  } else {
    epi = gelu ? HIPBLASLT_EPILOGUE_GELU_AUX : HIPBLASLT_EPILOGUE_DEFAULT;
  }
  if (gelu) {
    aux = at::empty({M, N}, x.options());
    aptr = aux.data_ptr();
  }
  auto ws = ws_tensor(x);
  // D[N,M] = W[K,N]^T * x[K,M]
  lt_matmul(w.data_ptr(), HIPBLAS_OP_T, K, N, x.data_ptr(), HIPBLAS_OP_N, K,
            M, y.data_ptr(), N, M, epi, bptr, nullptr, aptr, N, ws);
  return {y, gelu ? aux : at::Tensor()};
}

// dx = dy @ W; with aux z given: dx = dgelu(z) * (dy @ W) and db (DGELU_BGRAD).
std::vector<at::Tensor> lt_linear_bwd_dx(at::Tensor dy, at::Tensor w,
                                         c10::optional<at::Tensor> aux_z,
                                         bool want_bgrad) {
  check2d(dy, "dy");
  check2d(w, "w");
  const int64_t M = dy.size(0), N = dy.size(1), K = w.size(1);
  TORCH_CHECK(w.size(0) == N, "lt_linear_bwd_dx: shape mismatch");
  auto dx = at::empty({M, K}, dy.options());
  at::Tensor db;
  hipblasLtEpilogue_t epi = HIPBLASLT_EPILOGUE_DEFAULT;
  void* aptr = nullptr;
  void* bgrad = nullptr;
  if (aux_z.has_value()) {
    check2d(*aux_z, "aux_z");
    aptr = aux_z->data_ptr();
    if (want_bgrad) {
      db = at::empty({K}, dy.options());
      bgrad = db.data_ptr();
      epi = HIPBLASLT_EPILOGUE_DGELU_BGRAD;
    } else {
      epi = HIPBLASLT_EPILOGUE_DGELU;
    }
  }
  auto ws = ws_tensor(dy);
  // D[K,M] = W[K,N] * dy[N,M]; aux z is [M,K] row-major = [K,M] col, ld=K
  lt_matmul(w.data_ptr(), HIPBLAS_OP_N, K, N, dy.data_ptr(), HIPBLAS_OP_N, N,
            M, dx.data_ptr(), K, M, epi, nullptr, bgrad, aptr, K, ws);
  return {dx, db};
}

// dW = dy^T @ x; optional db = colsum(dy) fused via BGRADB.
std::vector<at::Tensor> lt_linear_bwd_dw(at::Tensor x, at::Tensor dy,
                                         bool want_bgrad) {
  check2d(x, "x");
  check2d(dy, "dy");
  const int64_t M = x.size(0), K = x.size(1), N = dy.size(1);
  TORCH_CHECK(dy.size(0) == M, "lt_linear_bwd_dw: shape mismatch");
  auto dw = at::empty({N, K}, x.options());
  at::Tensor db;
  void* bgrad = nullptr;
  hipblasLtEpilogue_t epi = HIPBLASLT_EPILOGUE_DEFAULT;
  if (want_bgrad) {
    db = at::empty({N}, dy.options());
    bgrad = db.data_ptr();
    epi = HIPBLASLT_EPILOGUE_BGRADB;
  }
  auto ws = ws_tensor(x);
  // D[K,N] (= dW^T col-major = dW row-major [N,K]) = x[K,M] * dy[N,M]^T
  lt_matmul(x.data_ptr(), HIPBLAS_OP_N, K, M, dy.data_ptr(), HIPBLAS_OP_T, N,
            M, dw.data_ptr(), K, N, epi, nullptr, bgrad, nullptr, 0, ws);
  return {dw, db};
}

}  // namespace photon_hip
