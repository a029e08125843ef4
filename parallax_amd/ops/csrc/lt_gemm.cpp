// hipBLASLt GEMM with offline per-shape algorithm tuning.
//
// Motivation (rocprofv3, DeepSeek-R1-Distill-8B decode @ batch 128): the
// default hipBLASLt heuristic picks MT32x128x128 kernels for the skinny
// decode GEMMs (qkv / o_proj / down) that run ~2.5x slower than their
// HBM-streaming speed-of-light, while an exhaustive search over the
// heuristic's own candidate list finds near-SOL kernels. torch's TunableOp
// cannot help because the selection must hold inside hipGraph capture.
//
// Contract: lt_linear(x[M,K] bf16, w[N,K] bf16) -> y[M,N] bf16 (y = x.w^T,
// fp32 accumulate). The FIRST call for a (M,N,K) shape must happen OUTSIDE
// graph capture (the engine's warmup guarantees this): it enumerates the
// heuristic's algorithms, times each on the current stream, and caches the
// winner + a persistent workspace. Later calls (including during capture)
// replay the cached algorithm deterministically.
//
// Row-major torch tensors are fed to hipblasLt in its column-major
// convention as the classic TN problem: y^T[N,M]_cm = w[N,K] * x^T[K,M]_cm
// with A = w (op T on its [K,N] column view), B = x.

#include <hipblaslt/hipblaslt.h>
#include <hipblaslt/hipblaslt-ext.hpp>

#include <torch/extension.h>

#include <algorithm>
#include <map>
#include <mutex>
#include <tuple>
#include <vector>

#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#define LT_CHECK(expr)                                                        \
  do {                                                                        \
    hipblasStatus_t st__ = (expr);                                            \
    TORCH_CHECK(st__ == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)st__, \
                " at " #expr);                                                \
  } while (0)

namespace {

constexpr size_t kWorkspaceBytes = 64ull << 20;  // 64 MiB, shared

struct Descs;
struct CachedShape;

struct LtState {
  hipblasLtHandle_t handle = nullptr;
  void* workspace = nullptr;
  std::map<std::tuple<int64_t, int64_t, int64_t, int>,
           std::unique_ptr<CachedShape>> shapes;
  std::mutex mu;
};

LtState& state() {
  static LtState s;
  if (s.handle == nullptr) {
    LT_CHECK(hipblasLtCreate(&s.handle));
    C10_HIP_CHECK(hipMalloc(&s.workspace, kWorkspaceBytes));
  }
  return s;
}

struct Descs {
  hipblasLtMatmulDesc_t op = nullptr;
  hipblasLtMatrixLayout_t a = nullptr, b = nullptr, c = nullptr;
  ~Descs() {
    if (op) hipblasLtMatmulDescDestroy(op);
    if (a) hipblasLtMatrixLayoutDestroy(a);
    if (b) hipblasLtMatrixLayoutDestroy(b);
    if (c) hipblasLtMatrixLayoutDestroy(c);
  }
};

void make_descs(Descs& d, int64_t M, int64_t N, int64_t K,
                hipDataType in_type = HIP_R_16BF) {
  LT_CHECK(hipblasLtMatmulDescCreate(&d.op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
  hipblasOperation_t ta = HIPBLAS_OP_T, tb = HIPBLAS_OP_N;
  LT_CHECK(hipblasLtMatmulDescSetAttribute(d.op, HIPBLASLT_MATMUL_DESC_TRANSA,
                                           &ta, sizeof(ta)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(d.op, HIPBLASLT_MATMUL_DESC_TRANSB,
                                           &tb, sizeof(tb)));
  // A = w: [K, N] column-major view of the row-major [N, K] buffer, op T
  LT_CHECK(hipblasLtMatrixLayoutCreate(&d.a, in_type, K, N, K));
  // B = x: [K, M] column-major view of the row-major [M, K] buffer
  LT_CHECK(hipblasLtMatrixLayoutCreate(&d.b, in_type, K, M, K));
  // C = y^T: [N, M] column-major = row-major y [M, N]
  LT_CHECK(hipblasLtMatrixLayoutCreate(&d.c, HIP_R_16BF, N, M, N));
}

struct CachedShape {
  Descs descs;
  hipblasLtMatmulAlgo_t algo;
};

hipblasLtMatmulAlgo_t tune_shape(int64_t M, int64_t N, int64_t K,
                                 const void* wp, const void* xp, void* yp,
                                 hipStream_t stream,
                                 hipDataType in_type = HIP_R_16BF) {
  LtState& s = state();
  Descs d;
  make_descs(d, M, N, K, in_type);

  hipblasLtMatmulPreference_t pref;
  LT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  size_t ws = kWorkspaceBytes;
  LT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));

  constexpr int kMaxHeuristic = 24;
  hipblasLtMatmulHeuristicResult_t heur[kMaxHeuristic];
  int n_heur = 0;
  LT_CHECK(hipblasLtMatmulAlgoGetHeuristic(s.handle, d.op, d.a, d.b, d.c, d.c,
                                           pref, kMaxHeuristic, heur,
                                           &n_heur));
  hipblasLtMatmulPreferenceDestroy(pref);
  TORCH_CHECK(n_heur > 0, "hipblaslt: no algorithms for shape ", M, "x", N,
              "x", K);
  std::vector<hipblasLtMatmulHeuristicResult_t> results(heur, heur + n_heur);

  // The heuristic's ~24 picks leave 2x on the table at decode shapes
  // (rocprofv3: M=512 GEMMs at 41-48% of what the library can do). Widen the
  // search over the FULL Tensile solution set, filtered by
  // matmulIsAlgoSupported — the exhaustive-screen counterpart of
  // hipblaslt-bench's offline tuning, run once per shape at engine warmup.
  static const bool kFullTune = [] {
    const char* e = getenv("PARALLAX_LT_TUNE_FULL");
    return e == nullptr || e[0] != '0';
  }();
  if (kFullTune) {
    std::vector<hipblasLtMatmulHeuristicResult_t> all;
    if (hipblaslt_ext::getAllAlgos(
            s.handle, hipblaslt_ext::GemmType::HIPBLASLT_GEMM, HIPBLAS_OP_T,
            HIPBLAS_OP_N, in_type, in_type, HIP_R_16BF, HIP_R_16BF,
            HIPBLAS_COMPUTE_32F, all) == HIPBLAS_STATUS_SUCCESS) {
      // the screening loop below is additionally time-budgeted, but cap the
      // supported-check pass too (it walks thousands of entries)
      static const size_t cap_env = [] {
        const char* e = getenv("PARALLAX_LT_TUNE_CAP");
        return e ? (size_t)atoll(e) : (size_t)0;
      }();
      const size_t cap = cap_env ? cap_env : (size_t)100000;  // budget-bound
      size_t kept = 0;
      const float alpha1 = 1.f, beta1 = 0.f;
      for (auto& r : all) {
        if (kept >= cap) break;
        size_t ws_needed = 0;
        if (hipblaslt_ext::matmulIsAlgoSupported(
                s.handle, d.op, &alpha1, d.a, d.b, &beta1, d.c, d.c, r.algo,
                ws_needed) == HIPBLAS_STATUS_SUCCESS &&
            ws_needed <= kWorkspaceBytes) {
          results.push_back(r);
          ++kept;
        }
      }
    }
  }
  const int n_results = (int)results.size();

  const float alpha = 1.f, beta = 0.f;
  auto run = [&](const hipblasLtMatmulAlgo_t& algo) {
    return hipblasLtMatmul(s.handle, d.op, &alpha, wp, d.a, xp, d.b, &beta, yp,
                           d.c, yp, d.c, &algo, s.workspace, kWorkspaceBytes,
                           stream);
  };

  hipEvent_t ev0, ev1;
  C10_HIP_CHECK(hipEventCreate(&ev0));
  C10_HIP_CHECK(hipEventCreate(&ev1));
  auto time_algo = [&](int i, int iters) -> float {
    if (run(results[i].algo) != HIPBLAS_STATUS_SUCCESS) return 1e30f;  // warm
    C10_HIP_CHECK(hipEventRecord(ev0, stream));
    for (int it = 0; it < iters; ++it) (void)run(results[i].algo);
    C10_HIP_CHECK(hipEventRecord(ev1, stream));
    C10_HIP_CHECK(hipEventSynchronize(ev1));
    float ms = 1e30f;
    C10_HIP_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
    return ms;
  };
  // screen every candidate with one timed iteration, then re-time the best
  // few — keeps one-off tuning of huge prefill shapes to ~tens of ms
  // Per-shape screening budget: the heuristic's own candidates (first in the
  // list) are always timed; the widened set is screened until the budget runs
  // out. Bounds the one-time tuning stall a previously-unseen shape causes
  // mid-serving (PARALLAX_LT_TUNE_MS, default 150 ms/shape).
  // read per call (tuning is rare): the engine warmup varies this to give
  // the hot shapes a deep search and minor graph buckets a quick one
  const char* be = getenv("PARALLAX_LT_TUNE_MS");
  const float kBudgetMs = be ? (float)atof(be) : 1500.f;
  // budget 0 = steady-state serving: a previously-unseen shape (arbitrary
  // prefill token counts) takes the heuristic's top pick with NO timing
  // loop — a mid-serving tuning stall is worse than a slightly sub-optimal
  // algo on a shape that appears once
  if (kBudgetMs <= 0.f) return results[0].algo;
  // tiny-M shapes (prefill lm_head row counts) appear mid-serving with
  // varying M: keep their one-time tuning stall small
  const float budget_ms = (M <= 64) ? std::min(kBudgetMs, 150.f) : kBudgetMs;
  std::vector<std::pair<float, int>> screened;
  float spent_ms = 0.f;
  for (int i = 0; i < n_results; ++i) {
    if (i >= n_heur && spent_ms > budget_ms) break;
    const float ms = time_algo(i, 1);
    if (ms < 1e29f) spent_ms += 2.f * ms;  // warm + timed run
    screened.emplace_back(ms, i);
  }
  std::sort(screened.begin(), screened.end());
  int best = -1;
  float best_ms = 1e30f;
  for (int r = 0; r < std::min<int>(8, (int)screened.size()); ++r) {
    if (screened[r].first >= 1e30f) continue;
    float ms = time_algo(screened[r].second, 6);
    if (ms < best_ms) {
      best_ms = ms;
      best = screened[r].second;
    }
  }
  C10_HIP_CHECK(hipEventDestroy(ev0));
  C10_HIP_CHECK(hipEventDestroy(ev1));
  TORCH_CHECK(best >= 0, "hipblaslt: every candidate algorithm failed for ", M,
              "x", N, "x", K);
  return results[best].algo;
}

}  // namespace

torch::Tensor lt_linear(torch::Tensor x, torch::Tensor w) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda(), "lt_linear: GPU tensors required");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                  w.scalar_type() == at::kBFloat16,
              "lt_linear: bf16 only");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "lt_linear: 2-D x and w");
  TORCH_CHECK(x.stride(1) == 1 && x.stride(0) == x.size(1),
              "lt_linear: x must be contiguous");
  TORCH_CHECK(w.stride(1) == 1 && w.stride(0) == w.size(1),
              "lt_linear: w must be contiguous");
  const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "lt_linear: K mismatch");

  auto y = at::empty({M, N}, x.options());
  hipStream_t stream = at::hip::getCurrentHIPStream().stream();
  LtState& s = state();

  CachedShape* cs = nullptr;
  {
    std::lock_guard<std::mutex> g(s.mu);
    auto key = std::make_tuple(M, N, K, 0);
    auto it = s.shapes.find(key);
    if (it == s.shapes.end()) {
      hipStreamCaptureStatus cap = hipStreamCaptureStatusNone;
      (void)hipStreamIsCapturing(stream, &cap);
      TORCH_CHECK(cap == hipStreamCaptureStatusNone,
                  "lt_linear: first call for shape ", M, "x", N, "x", K,
                  " happened during graph capture; warm this shape up first");
      auto entry = std::make_unique<CachedShape>();
      make_descs(entry->descs, M, N, K);
      entry->algo = tune_shape(M, N, K, w.data_ptr(), x.data_ptr(),
                               y.data_ptr(), stream);
      it = s.shapes.emplace(key, std::move(entry)).first;
    }
    cs = it->second.get();
  }

  const float alpha = 1.f, beta = 0.f;
  LT_CHECK(hipblasLtMatmul(s.handle, cs->descs.op, &alpha, w.data_ptr(),
                           cs->descs.a, x.data_ptr(), cs->descs.b, &beta,
                           y.data_ptr(), cs->descs.c, y.data_ptr(),
                           cs->descs.c, &cs->algo, s.workspace,
                           kWorkspaceBytes, stream));
  return y;
}

// fp8-E4M3 W8A8 GEMM (BASELINE DeepSeek-V3 "fp8 MFMA" dense route):
// y[M,N] bf16 = (x_q[M,K] * sx) . (w_q[N,K] * sw)^T with per-tensor DEVICE
// scale pointers (hipblaslt reads them at kernel time, so graph capture
// bakes only the persistent buffer addresses the caller guarantees).
torch::Tensor lt_linear_fp8(torch::Tensor x_q, torch::Tensor w_q,
                            torch::Tensor x_scale, torch::Tensor w_scale) {
  TORCH_CHECK(x_q.is_cuda() && w_q.is_cuda());
  TORCH_CHECK(x_q.scalar_type() == at::kFloat8_e4m3fn &&
              w_q.scalar_type() == at::kFloat8_e4m3fn,
              "lt_linear_fp8: fp8_e4m3fn inputs required");
  TORCH_CHECK(x_q.is_contiguous() && w_q.is_contiguous());
  TORCH_CHECK(x_scale.scalar_type() == at::kFloat && x_scale.is_cuda());
  TORCH_CHECK(w_scale.scalar_type() == at::kFloat && w_scale.is_cuda());
  const int64_t M = x_q.size(0), K = x_q.size(1), N = w_q.size(0);
  TORCH_CHECK(w_q.size(1) == K, "lt_linear_fp8: K mismatch");

  auto y = at::empty({M, N}, x_q.options().dtype(at::kBFloat16));
  hipStream_t stream = at::hip::getCurrentHIPStream().stream();
  LtState& s = state();

  CachedShape* cs = nullptr;
  {
    std::lock_guard<std::mutex> g(s.mu);
    auto key = std::make_tuple(M, N, K, 1);
    auto it = s.shapes.find(key);
    if (it == s.shapes.end()) {
      hipStreamCaptureStatus cap = hipStreamCaptureStatusNone;
      (void)hipStreamIsCapturing(stream, &cap);
      TORCH_CHECK(cap == hipStreamCaptureStatusNone,
                  "lt_linear_fp8: first call for shape ", M, "x", N, "x", K,
                  " happened during graph capture; warm this shape up first");
      auto entry = std::make_unique<CachedShape>();
      make_descs(entry->descs, M, N, K, HIP_R_8F_E4M3);
      entry->algo = tune_shape(M, N, K, w_q.data_ptr(), x_q.data_ptr(),
                               y.data_ptr(), stream, HIP_R_8F_E4M3);
      it = s.shapes.emplace(key, std::move(entry)).first;
    }
    cs = it->second.get();
  }
  // per-call device scale pointers (A = w, B = x in our TN formulation)
  const void* sa = w_scale.data_ptr();
  const void* sb = x_scale.data_ptr();
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      cs->descs.op, HIPBLASLT_MATMUL_DESC_A_SCALE_POINTER, &sa, sizeof(sa)));
  LT_CHECK(hipblasLtMatmulDescSetAttribute(
      cs->descs.op, HIPBLASLT_MATMUL_DESC_B_SCALE_POINTER, &sb, sizeof(sb)));
  const float alpha = 1.f, beta = 0.f;
  LT_CHECK(hipblasLtMatmul(s.handle, cs->descs.op, &alpha, w_q.data_ptr(),
                           cs->descs.a, x_q.data_ptr(), cs->descs.b, &beta,
                           y.data_ptr(), cs->descs.c, y.data_ptr(),
                           cs->descs.c, &cs->algo, s.workspace,
                           kWorkspaceBytes, stream));
  return y;
}
