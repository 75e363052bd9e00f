// Weight-gradient GEMM with fp32 accumulation directly into main_grad:
//     main_grad[out,in] += grad_output[rows,out]^T @ input[rows,in]
// via hipblasLt (bf16 A/B, fp32 C/D, beta=1).
//
// Replaces the reference's apex fused_weight_gradient_mlp path
// (layers.py:404 gradient_accumulation_fusion): without it each linear
// backward pays an extra bf16 GEMM output write plus a cast-add pass over
// the fp32 grad buffer (13% of step time in profiles/r01 baseline).

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <hipblaslt/hipblaslt.h>

#include <map>
#include <mutex>
#include <stdexcept>

#define HIPBLASLT_CHECK(expr)                                              \
  do {                                                                     \
    hipblasStatus_t st_ = (expr);                                          \
    TORCH_CHECK(st_ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)st_, \
                " at ", #expr);                                            \
  } while (0)

namespace {

constexpr size_t kWorkspaceBytes = 64ull << 20;

struct LtContext {
  hipblasLtHandle_t handle{};
  void* workspace{};
  LtContext() {
    HIPBLASLT_CHECK(hipblasLtCreate(&handle));
    if (hipMalloc(&workspace, kWorkspaceBytes) != hipSuccess)
      throw std::runtime_error("hipMalloc workspace failed");
  }
};

LtContext& ctx() {
  static LtContext c;
  return c;
}

struct AlgoKey {
  int64_t m, n, k;
  bool operator<(const AlgoKey& o) const {
    return std::tie(m, n, k) < std::tie(o.m, o.n, o.k);
  }
};

// MoE expert wgrads vary K (tokens per expert) every step: sweep-benchmark
// only the first few K values per (M, N); later shapes take the heuristic
// top choice (still cached per exact shape).
std::map<std::pair<long, long>, int>& sweep_count() {
  static std::map<std::pair<long, long>, int> c;
  return c;
}

std::map<AlgoKey, hipblasLtMatmulAlgo_t>& algo_cache() {
  static std::map<AlgoKey, hipblasLtMatmulAlgo_t> c;
  return c;
}

std::mutex& mu() {
  static std::mutex m;
  return m;
}

}  // namespace

namespace {

// Shapes where the BGRADB-epilogue heuristic came back empty: remembered
// so the caller's fallback (separate colsum kernel) is taken without
// re-querying every step.
std::map<AlgoKey, bool>& bgrad_unsupported() {
  static std::map<AlgoKey, bool> c;
  return c;
}

}  // namespace

// grad2d: [rows, out] bf16 contiguous; input2d: [rows, in] bf16 contiguous;
// main_grad: [out, in] fp32 contiguous.  main_grad += grad2d^T @ input2d.
void wgrad_accum(torch::Tensor grad2d, torch::Tensor input2d,
                 torch::Tensor main_grad) {
  TORCH_CHECK(grad2d.is_cuda() && grad2d.scalar_type() == torch::kBFloat16 &&
              grad2d.is_contiguous());
  TORCH_CHECK(input2d.is_cuda() && input2d.scalar_type() == torch::kBFloat16 &&
              input2d.is_contiguous());
  TORCH_CHECK(main_grad.is_cuda() &&
              main_grad.scalar_type() == torch::kFloat32 &&
              main_grad.is_contiguous());
  const int64_t rows = grad2d.size(0);
  const int64_t out = grad2d.size(1);
  const int64_t in = input2d.size(1);
  TORCH_CHECK(input2d.size(0) == rows);
  TORCH_CHECK(main_grad.size(0) == out && main_grad.size(1) == in);

  // column-major formulation: D[in,out] = A(N: input^T_cm [in x rows])
  //                                     x B(T: grad_cm [out x rows])
  const int64_t M = in, N = out, K = rows;

  std::lock_guard<std::mutex> lock(mu());
  auto& c = ctx();

  hipblasLtMatmulDesc_t op{};
  HIPBLASLT_CHECK(hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F,
                                            HIP_R_32F));
  hipblasOperation_t opN = HIPBLAS_OP_N, opT = HIPBLAS_OP_T;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_TRANSA, &opN, sizeof(opN)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_TRANSB, &opT, sizeof(opT)));

  hipblasLtMatrixLayout_t la{}, lb{}, lc{};
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, M, K, M));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, N, K, N));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&lc, HIP_R_32F, M, N, M));

  float alpha = 1.f, beta = 1.f;
  hipStream_t stream = at::hip::getCurrentHIPStream().stream();

  AlgoKey key{M, N, K};
  auto it = algo_cache().find(key);
  if (it == algo_cache().end()) {
    hipblasLtMatmulPreference_t pref{};
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws = kWorkspaceBytes;
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
    hipblasLtMatmulHeuristicResult_t results[16];
    int found = 0;
    HIPBLASLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
        c.handle, op, la, lb, lc, lc, pref, 16, results, &found));
    hipblasLtMatmulPreferenceDestroy(pref);
    TORCH_CHECK(found > 0, "no hipblaslt algo for wgrad shape ", M, "x", N,
                "x", K);
    // one-shot autotune: time each candidate on a scratch C so the
    // beta=1 accumulation into live gradients is not corrupted
    int best = 0;
    int& sweeps = sweep_count()[{M, N}];
    if (found > 1 && sweeps < 4) {
      ++sweeps;
      void* scratch = nullptr;
      const size_t cbytes = (size_t)M * N * sizeof(float);
      if (hipMalloc(&scratch, cbytes) == hipSuccess) {
        float best_ms = 1e30f;
        hipEvent_t t0, t1;
        hipEventCreate(&t0);
        hipEventCreate(&t1);
        for (int i = 0; i < found; ++i) {
          // warm
          if (hipblasLtMatmul(c.handle, op, &alpha, input2d.data_ptr(), la,
                              grad2d.data_ptr(), lb, &beta, scratch, lc,
                              scratch, lc, &results[i].algo, c.workspace,
                              kWorkspaceBytes,
                              stream) != HIPBLAS_STATUS_SUCCESS)
            continue;
          hipEventRecord(t0, stream);
          for (int r = 0; r < 3; ++r)
            hipblasLtMatmul(c.handle, op, &alpha, input2d.data_ptr(), la,
                            grad2d.data_ptr(), lb, &beta, scratch, lc,
                            scratch, lc, &results[i].algo, c.workspace,
                            kWorkspaceBytes, stream);
          hipEventRecord(t1, stream);
          hipEventSynchronize(t1);
          float ms = 1e30f;
          hipEventElapsedTime(&ms, t0, t1);
          if (ms < best_ms) {
            best_ms = ms;
            best = i;
          }
        }
        hipEventDestroy(t0);
        hipEventDestroy(t1);
        hipFree(scratch);
      }
    }
    it = algo_cache().emplace(key, results[best].algo).first;
  }
  HIPBLASLT_CHECK(hipblasLtMatmul(
      c.handle, op, &alpha, input2d.data_ptr(), la, grad2d.data_ptr(), lb,
      &beta, main_grad.data_ptr(), lc, main_grad.data_ptr(), lc, &it->second,
      c.workspace, kWorkspaceBytes, stream));

  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatmulDescDestroy(op);
}


// wgrad_accum + bias gradient in the same hipblasLt call: the BGRADB
// epilogue reduces the B operand (grad2d) over the K (token) dimension
// while the mainloop already has those tiles in LDS, so dbias costs no
// extra HBM pass (replaces a separate colsum_accum launch — 2.2% of
// step time in profiles/r02_flash_final_stats.csv).  dbias [out] fp32 is
// OVERWRITTEN (epilogue semantics); the caller accumulates it into the
// bias main_grad.  Returns false when the heuristic has no
// epilogue-capable algo for the shape — caller falls back.
bool wgrad_accum_bgrad(torch::Tensor grad2d, torch::Tensor input2d,
                       torch::Tensor main_grad, torch::Tensor dbias) {
  TORCH_CHECK(grad2d.is_cuda() && grad2d.scalar_type() == torch::kBFloat16 &&
              grad2d.is_contiguous());
  TORCH_CHECK(input2d.is_cuda() && input2d.scalar_type() == torch::kBFloat16 &&
              input2d.is_contiguous());
  TORCH_CHECK(main_grad.is_cuda() &&
              main_grad.scalar_type() == torch::kFloat32 &&
              main_grad.is_contiguous());
  TORCH_CHECK(dbias.is_cuda() && dbias.scalar_type() == torch::kFloat32 &&
              dbias.is_contiguous());
  const int64_t rows = grad2d.size(0);
  const int64_t out = grad2d.size(1);
  const int64_t in = input2d.size(1);
  TORCH_CHECK(input2d.size(0) == rows);
  TORCH_CHECK(main_grad.size(0) == out && main_grad.size(1) == in);
  TORCH_CHECK(dbias.numel() == out);

  const int64_t M = in, N = out, K = rows;

  std::lock_guard<std::mutex> lock(mu());
  auto& c = ctx();

  AlgoKey key{M | (5LL << 48), N, K};
  if (bgrad_unsupported().count({M, N, K})) return false;

  hipblasLtMatmulDesc_t op{};
  HIPBLASLT_CHECK(hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F,
                                            HIP_R_32F));
  hipblasOperation_t opN = HIPBLAS_OP_N, opT = HIPBLAS_OP_T;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_TRANSA, &opN, sizeof(opN)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_TRANSB, &opT, sizeof(opT)));
  hipblasLtEpilogue_t epi = HIPBLASLT_EPILOGUE_BGRADB;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
  void* bias_ptr = dbias.data_ptr();
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias_ptr, sizeof(bias_ptr)));
  hipDataType bias_t = HIP_R_32F;
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bias_t, sizeof(bias_t)));

  hipblasLtMatrixLayout_t la{}, lb{}, lc{};
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, M, K, M));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, N, K, N));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&lc, HIP_R_32F, M, N, M));

  float alpha = 1.f, beta = 1.f;
  hipStream_t stream = at::hip::getCurrentHIPStream().stream();

  auto it = algo_cache().find(key);
  if (it == algo_cache().end()) {
    hipblasLtMatmulPreference_t pref{};
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws = kWorkspaceBytes;
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
    hipblasLtMatmulHeuristicResult_t results[16];
    int found = 0;
    hipblasStatus_t hst = hipblasLtMatmulAlgoGetHeuristic(
        c.handle, op, la, lb, lc, lc, pref, 16, results, &found);
    hipblasLtMatmulPreferenceDestroy(pref);
    if (hst != HIPBLAS_STATUS_SUCCESS || found == 0) {
      bgrad_unsupported()[{M, N, K}] = true;
      hipblasLtMatrixLayoutDestroy(la);
      hipblasLtMatrixLayoutDestroy(lb);
      hipblasLtMatrixLayoutDestroy(lc);
      hipblasLtMatmulDescDestroy(op);
      return false;
    }
    // sweep on a scratch C (beta=1 would corrupt live grads); the
    // epilogue writes the (identical) dbias each try, which is safe
    int best = 0;
    int& sweeps = sweep_count()[{M | (5LL << 48), N}];
    if (found > 1 && sweeps < 4) {
      ++sweeps;
      void* scratch = nullptr;
      const size_t cbytes = (size_t)M * N * sizeof(float);
      if (hipMalloc(&scratch, cbytes) == hipSuccess) {
        float best_ms = 1e30f;
        hipEvent_t t0, t1;
        (void)hipEventCreate(&t0);
        (void)hipEventCreate(&t1);
        for (int i = 0; i < found; ++i) {
          if (hipblasLtMatmul(c.handle, op, &alpha, input2d.data_ptr(), la,
                              grad2d.data_ptr(), lb, &beta, scratch, lc,
                              scratch, lc, &results[i].algo, c.workspace,
                              kWorkspaceBytes,
                              stream) != HIPBLAS_STATUS_SUCCESS)
            continue;
          (void)hipEventRecord(t0, stream);
          for (int r = 0; r < 3; ++r)
            hipblasLtMatmul(c.handle, op, &alpha, input2d.data_ptr(), la,
                            grad2d.data_ptr(), lb, &beta, scratch, lc,
                            scratch, lc, &results[i].algo, c.workspace,
                            kWorkspaceBytes, stream);
          (void)hipEventRecord(t1, stream);
          (void)hipEventSynchronize(t1);
          float ms = 1e30f;
          (void)hipEventElapsedTime(&ms, t0, t1);
          if (ms < best_ms) {
            best_ms = ms;
            best = i;
          }
        }
        (void)hipEventDestroy(t0);
        (void)hipEventDestroy(t1);
        (void)hipFree(scratch);
      }
    }
    it = algo_cache().emplace(key, results[best].algo).first;
  }
  hipblasStatus_t st = hipblasLtMatmul(
      c.handle, op, &alpha, input2d.data_ptr(), la, grad2d.data_ptr(), lb,
      &beta, main_grad.data_ptr(), lc, main_grad.data_ptr(), lc, &it->second,
      c.workspace, kWorkspaceBytes, stream);
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatmulDescDestroy(op);
  if (st != HIPBLAS_STATUS_SUCCESS) {
    // algo claimed support but failed at run time: remember and fall back
    algo_cache().erase(key);
    bgrad_unsupported()[{M, N, K}] = true;
    return false;
  }
  return true;
}

// Swept-autotune bf16 GEMM for the dense fc forward/dgrad paths:
//   nt:  out[rows, N] = a[rows, K] @ w[N, K]^T   (forward / F.linear)
//   nn:  out[rows, K] = g[rows, N] @ w[N, K]     (dgrad)
// Same sweep/cache machinery as wgrad_accum; bf16 out, beta = 0.
static void lt_gemm(const void* A, const void* B, void* D, int64_t M,
                    int64_t N, int64_t K, int64_t lda, int64_t ldb,
                    hipblasOperation_t opA, hipblasOperation_t opB,
                    long tag) {
  std::lock_guard<std::mutex> lock(mu());
  auto& c = ctx();
  hipblasLtMatmulDesc_t op{};
  HIPBLASLT_CHECK(
      hipblasLtMatmulDescCreate(&op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_TRANSA, &opA, sizeof(opA)));
  HIPBLASLT_CHECK(hipblasLtMatmulDescSetAttribute(
      op, HIPBLASLT_MATMUL_DESC_TRANSB, &opB, sizeof(opB)));
  hipblasLtMatrixLayout_t la{}, lb{}, lc{};
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(
      &la, HIP_R_16BF, opA == HIPBLAS_OP_N ? M : K,
      opA == HIPBLAS_OP_N ? K : M, lda));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(
      &lb, HIP_R_16BF, opB == HIPBLAS_OP_N ? K : N,
      opB == HIPBLAS_OP_N ? N : K, ldb));
  HIPBLASLT_CHECK(hipblasLtMatrixLayoutCreate(&lc, HIP_R_16BF, M, N, M));
  float alpha = 1.f, beta = 0.f;
  hipStream_t stream = at::hip::getCurrentHIPStream().stream();

  AlgoKey key{M | (tag << 48), N, K};
  auto it = algo_cache().find(key);
  if (it == algo_cache().end()) {
    hipblasLtMatmulPreference_t pref{};
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws = kWorkspaceBytes;
    HIPBLASLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
    hipblasLtMatmulHeuristicResult_t results[24];
    int found = 0;
    HIPBLASLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(
        c.handle, op, la, lb, lc, lc, pref, 24, results, &found));
    hipblasLtMatmulPreferenceDestroy(pref);
    TORCH_CHECK(found > 0, "no hipblaslt algo for gemm ", M, "x", N, "x", K);
    int best = 0;
    int& sweeps = sweep_count()[{M | (tag << 48), N}];
    if (found > 1 && sweeps < 4) {
      ++sweeps;
      void* scratch = nullptr;
      const size_t dbytes = (size_t)M * N * sizeof(unsigned short);
      if (hipMalloc(&scratch, dbytes) == hipSuccess) {
        float best_ms = 1e30f;
        hipEvent_t t0, t1;
        (void)hipEventCreate(&t0);
        (void)hipEventCreate(&t1);
        for (int i = 0; i < found; ++i) {
          if (hipblasLtMatmul(c.handle, op, &alpha, A, la, B, lb, &beta,
                              scratch, lc, scratch, lc, &results[i].algo,
                              c.workspace, kWorkspaceBytes,
                              stream) != HIPBLAS_STATUS_SUCCESS)
            continue;
          (void)hipEventRecord(t0, stream);
          for (int r = 0; r < 3; ++r)
            hipblasLtMatmul(c.handle, op, &alpha, A, la, B, lb, &beta,
                            scratch, lc, scratch, lc, &results[i].algo,
                            c.workspace, kWorkspaceBytes, stream);
          (void)hipEventRecord(t1, stream);
          (void)hipEventSynchronize(t1);
          float ms = 1e30f;
          (void)hipEventElapsedTime(&ms, t0, t1);
          if (ms < best_ms) {
            best_ms = ms;
            best = i;
          }
        }
        (void)hipEventDestroy(t0);
        (void)hipEventDestroy(t1);
        (void)hipFree(scratch);
      }
    }
    it = algo_cache().emplace(key, results[best].algo).first;
  }
  HIPBLASLT_CHECK(hipblasLtMatmul(c.handle, op, &alpha, A, la, B, lb, &beta,
                                  D, lc, D, lc, &it->second, c.workspace,
                                  kWorkspaceBytes, stream));
  hipblasLtMatrixLayoutDestroy(la);
  hipblasLtMatrixLayoutDestroy(lb);
  hipblasLtMatrixLayoutDestroy(lc);
  hipblasLtMatmulDescDestroy(op);
}

// out[rows, N] = a[rows, K] @ w[N, K]^T
torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor w) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 &&
              a.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
              w.is_contiguous());
  const int64_t rows = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K);
  auto out = torch::empty({rows, N}, a.options());
  // col-major: D[N, rows] = (w_cm[K, N])^T x a_cm[K, rows]
  lt_gemm(w.data_ptr(), a.data_ptr(), out.data_ptr(), N, rows, K,
          /*lda=*/K, /*ldb=*/K, HIPBLAS_OP_T, HIPBLAS_OP_N, /*tag=*/1);
  return out;
}

// out[rows, K] = g[rows, N] @ w[N, K]
torch::Tensor gemm_nn(torch::Tensor g, torch::Tensor w) {
  TORCH_CHECK(g.is_cuda() && g.scalar_type() == torch::kBFloat16 &&
              g.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
              w.is_contiguous());
  const int64_t rows = g.size(0), N = g.size(1), K = w.size(1);
  TORCH_CHECK(w.size(0) == N);
  auto out = torch::empty({rows, K}, g.options());
  // col-major: D[K, rows] = w_cm[K, N] x g_cm[N, rows]
  lt_gemm(w.data_ptr(), g.data_ptr(), out.data_ptr(), K, rows, N,
          /*lda=*/K, /*ldb=*/N, HIPBLAS_OP_N, HIPBLAS_OP_N, /*tag=*/2);
  return out;
}
