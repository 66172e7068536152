// hipBLASLt epilogue-fused dense MLP GEMMs for MI355X (gfx950).
//
// The captured-step trace showed the MLP's non-GEMM epilogues dominating
// their layers: forward relu (a separate elementwise pass per layer) and the
// backward bias-grad column reduce (partial + segment + final kernels,
// ~14 us/layer of pure reduce overhead). hipBLASLt fuses both INTO the GEMM:
//   lt_linear_relu_fwd : y = relu(x @ w^T + b)        (EPILOGUE_RELU_BIAS)
//   lt_wgrad_bgrad     : dW = g^T @ x, db = colsum(g) (EPILOGUE_BGRADB)
//
// Column-major mapping (hipBLASLt is col-major; row-major [R, C] tensors are
// col-major (C x R) views):
//   fwd:   Y(N x M) = op_T(w_view K x N) x op_N(x_view K x M), bias len N
//   wgrad: W'(K x N) = op_N(x_view K x M) x op_T(g_view N x M);
//          BGRADB reduces op(B)'s columns => db[j] = sum_m g[m, j].
// Algo selection via the hipBLASLt heuristic, cached per (shape, dtypes,
// epilogue); host-side setup runs once per shape, so graph capture replays
// pay nothing.

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hipblaslt/hipblaslt.h>

#include <map>
#include <mutex>
#include <tuple>

#include "common.h"

namespace trec_amd {

namespace {

inline hipStream_t lt_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

#define LT_CHECK(expr)                                                    \
  do {                                                                    \
    hipblasStatus_t st_ = (expr);                                         \
    TORCH_CHECK(st_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ",        \
                (int)st_, " at ", #expr);                                 \
  } while (0)

hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t handle = [] {
    hipblasLtHandle_t h;
    LT_CHECK(hipblasLtCreate(&h));
    return h;
  }();
  return handle;
}

constexpr size_t kLtWorkspace = 64u << 20;

hipDataType lt_dtype(at::ScalarType st) {
  switch (st) {
    case at::kFloat: return HIP_R_32F;
    case at::kHalf: return HIP_R_16F;
    case at::kBFloat16: return HIP_R_16BF;
    default: TORCH_CHECK(false, "lt_gemm: unsupported dtype");
  }
}

struct AlgoKey {
  int64_t m, n, k;
  int dt;
  int epi;
  bool operator<(const AlgoKey& o) const {
    return std::tie(m, n, k, dt, epi) < std::tie(o.m, o.n, o.k, o.dt, o.epi);
  }
};

std::mutex g_algo_mu;
std::map<AlgoKey, hipblasLtMatmulAlgo_t> g_algo_cache;

// run D = op(A) x op(B) (+ epilogue) with heuristic-selected, cached algo
void lt_matmul(hipblasLtMatmulDesc_t desc, hipblasLtMatrixLayout_t la,
               hipblasLtMatrixLayout_t lb, hipblasLtMatrixLayout_t ld,
               const void* A, const void* B, void* D, const AlgoKey& key,
               const at::Tensor& ref_for_ws) {
  auto handle = lt_handle();
  auto stream = lt_stream();
  auto ws = at::empty({(int64_t)kLtWorkspace},
                      ref_for_ws.options().dtype(at::kByte));
  hipblasLtMatmulAlgo_t algo;
  bool have = false;
  {
    std::lock_guard<std::mutex> lk(g_algo_mu);
    auto it = g_algo_cache.find(key);
    if (it != g_algo_cache.end()) {
      algo = it->second;
      have = true;
    }
  }
  if (!have) {
    // first call per shape: measure every heuristic candidate and keep the
    // fastest (the heuristic's top-1 under an epilogue constraint was 9%
    // slower end-to-end). Runs during eager warmup, never inside capture.
    hipblasLtMatmulPreference_t pref;
    LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws_sz = kLtWorkspace;
    LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws_sz, sizeof(ws_sz)));
    hipblasLtMatmulHeuristicResult_t res[16];
    int found = 0;
    LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(handle, desc, la, lb, ld, ld, pref,
                                             16, res, &found));
    LT_CHECK(hipblasLtMatmulPreferenceDestroy(pref));
    TORCH_CHECK(found > 0, "hipblaslt: no algo for epilogue-fused GEMM");
    float alpha1 = 1.f, beta1 = 0.f;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    float best_ms = 1e30f;
    int best = 0;
    for (int c = 0; c < found; ++c) {
      // one warm + three timed reps per candidate
      if (hipblasLtMatmul(handle, desc, &alpha1, A, la, B, lb, &beta1, D, ld, D,
                          ld, &res[c].algo, ws.data_ptr(), kLtWorkspace,
                          stream) != HIPBLAS_STATUS_SUCCESS)
        continue;
      (void)hipEventRecord(e0, stream);
      for (int r = 0; r < 3; ++r)
        (void)hipblasLtMatmul(handle, desc, &alpha1, A, la, B, lb, &beta1, D, ld,
                              D, ld, &res[c].algo, ws.data_ptr(), kLtWorkspace,
                              stream);
      (void)hipEventRecord(e1, stream);
      (void)hipEventSynchronize(e1);
      float ms = 1e30f;
      (void)hipEventElapsedTime(&ms, e0, e1);
      if (ms < best_ms) { best_ms = ms; best = c; }
    }
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    algo = res[best].algo;
    std::lock_guard<std::mutex> lk(g_algo_mu);
    g_algo_cache.emplace(key, algo);
  }
  float alpha = 1.f, beta = 0.f;
  LT_CHECK(hipblasLtMatmul(handle, desc, &alpha, A, la, B, lb, &beta, D, ld, D,
                           ld, &algo, ws.data_ptr(), kLtWorkspace, stream));
}

}  // namespace

// y = relu(x @ w^T + b); x [M, K], w [N, K], b [N] -> y [M, N]
at::Tensor lt_linear_relu_fwd(const at::Tensor& x, const at::Tensor& w,
                              const at::Tensor& b) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && w.dim() == 2);
  TORCH_CHECK(x.scalar_type() == w.scalar_type() &&
              x.scalar_type() == b.scalar_type());
  int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && b.numel() == N);
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  auto y = at::empty({M, N}, x.options());
  auto dt = lt_dtype(x.scalar_type());

  hipblasLtMatmulDesc_t desc;
  LT_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  int32_t opA = HIPBLAS_OP_T, opB = HIPBLAS_OP_N;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA,
                                           &opA, sizeof(opA)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB,
                                           &opB, sizeof(opB)));
  hipblasLtEpilogue_t epi = HIPBLASLT_EPILOGUE_RELU_BIAS;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                           &epi, sizeof(epi)));
  auto bc = b.contiguous();  // keep alive through the (possibly async) matmul
  const void* bias_ptr = bc.data_ptr();
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias_ptr, sizeof(bias_ptr)));
  int32_t bias_dt = (int32_t)dt;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bias_dt, sizeof(bias_dt)));

  // col-major views: A = w_view (K x N) op_T -> (N x K); B = x_view (K x M);
  // D = y_view (N x M)
  hipblasLtMatrixLayout_t la, lb, ld;
  LT_CHECK(hipblasLtMatrixLayoutCreate(&la, dt, K, N, K));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lb, dt, K, M, K));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&ld, dt, N, M, N));
  AlgoKey key{M, N, K, (int)dt, (int)epi};
  lt_matmul(desc, la, lb, ld, wc.data_ptr(), xc.data_ptr(), y.data_ptr(), key, x);
  LT_CHECK(hipblasLtMatrixLayoutDestroy(la));
  LT_CHECK(hipblasLtMatrixLayoutDestroy(lb));
  LT_CHECK(hipblasLtMatrixLayoutDestroy(ld));
  LT_CHECK(hipblasLtMatmulDescDestroy(desc));
  return y;
}

// dW = g^T @ x, db = colsum(g); g [M, N], x [M, K] -> (dW [N, K], db [N])
std::tuple<at::Tensor, at::Tensor> lt_wgrad_bgrad(const at::Tensor& g,
                                                  const at::Tensor& x) {
  TORCH_CHECK(g.is_cuda() && g.dim() == 2 && x.dim() == 2);
  TORCH_CHECK(g.size(0) == x.size(0) && g.scalar_type() == x.scalar_type());
  int64_t M = g.size(0), N = g.size(1), K = x.size(1);
  auto gc = g.contiguous();
  auto xc = x.contiguous();
  auto dW = at::empty({N, K}, g.options());
  auto db = at::empty({N}, g.options());
  auto dt = lt_dtype(g.scalar_type());

  hipblasLtMatmulDesc_t desc;
  LT_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  int32_t opA = HIPBLAS_OP_N, opB = HIPBLAS_OP_T;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA,
                                           &opA, sizeof(opA)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB,
                                           &opB, sizeof(opB)));
  hipblasLtEpilogue_t epi = HIPBLASLT_EPILOGUE_BGRADB;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                           &epi, sizeof(epi)));
  const void* bias_ptr = db.data_ptr();
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias_ptr, sizeof(bias_ptr)));
  int32_t bias_dt = (int32_t)dt;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bias_dt, sizeof(bias_dt)));

  // col-major: A = x_view (K x M) op_N; B = g_view (N x M) op_T -> (M x N);
  // D = dW_view (K x N)
  hipblasLtMatrixLayout_t la, lb, ld;
  LT_CHECK(hipblasLtMatrixLayoutCreate(&la, dt, K, M, K));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&lb, dt, N, M, N));
  LT_CHECK(hipblasLtMatrixLayoutCreate(&ld, dt, K, N, K));
  AlgoKey key{K, N, M, (int)dt, (int)epi};
  lt_matmul(desc, la, lb, ld, xc.data_ptr(), gc.data_ptr(), dW.data_ptr(), key, g);
  LT_CHECK(hipblasLtMatrixLayoutDestroy(la));
  LT_CHECK(hipblasLtMatrixLayoutDestroy(lb));
  LT_CHECK(hipblasLtMatrixLayoutDestroy(ld));
  LT_CHECK(hipblasLtMatmulDescDestroy(desc));
  return {dW, db};
}

}  // namespace trec_amd
