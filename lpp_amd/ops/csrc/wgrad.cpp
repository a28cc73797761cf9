// Weight-gradient GEMM accumulating DIRECTLY into the optimizer's flat fp32
// buffer via hipBLASLt: dW_f32 += dY_bf16^T @ X_bf16  (beta = 1).
//
// Replaces the reference stack's two-pass flow (bf16 dW materialised by
// autograd, then a separate fp32 read-modify-write accumulation per
// microbatch per parameter — the DeepSpeed bf16/fp16 optimizer's grad-accum
// pass).  Profiled on MI355X, that separate pass is ~5% of
// a training step (~10 bytes/elem of pure HBM traffic per microbatch);
// fusing the accumulation into the wgrad GEMM epilogue removes it.
//
// Row-major tensors mapped onto hipBLASLt's column-major view:
//   D_rm[out,in] = dY_rm[T,out]^T * X_rm[T,in]
//   <=> D_cm[in,out] = X_cm[in,T](opA=N) * dY_cm[out,T](opB=T)
#include <hipblaslt/hipblaslt.h>

#include <mutex>
#include <unordered_map>

#include "common.h"

namespace lpp {

#define LPP_CHECK_BLASLT(expr)                                            \
  do {                                                                    \
    hipblasStatus_t _s = (expr);                                          \
    TORCH_CHECK(_s == HIPBLAS_STATUS_SUCCESS, "hipblaslt error ", (int)_s, \
                " at " #expr);                                            \
  } while (0)

constexpr size_t kWorkspaceBytes = 64u << 20;

struct LtContext {
  hipblasLtHandle_t handle{};
  LtContext() { LPP_CHECK_BLASLT(hipblasLtCreate(&handle)); }
};

static LtContext& lt_ctx() {
  static LtContext ctx;
  return ctx;
}

struct AlgoKey {
  int64_t m, n, k;
  bool operator==(const AlgoKey& o) const { return m == o.m && n == o.n && k == o.k; }
};
struct AlgoKeyHash {
  size_t operator()(const AlgoKey& k) const {
    return std::hash<int64_t>()(k.m * 1315423911 ^ k.n * 2654435761 ^ k.k);
  }
};

struct CachedPlan {
  hipblasLtMatmulDesc_t op{};
  hipblasLtMatrixLayout_t a{}, b{}, c{};
  hipblasLtMatmulAlgo_t algo{};
};

// dW (row-major [out, in], fp32) += dY(row-major [T, out], bf16)^T @ X(row-major [T, in], bf16)
void wgrad_f32_accum(at::Tensor x, at::Tensor dy, at::Tensor dw) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.dim() == 2);
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16 && dy.dim() == 2);
  TORCH_CHECK(dw.is_cuda() && dw.scalar_type() == at::kFloat && dw.dim() == 2);
  TORCH_CHECK(x.is_contiguous() && dy.is_contiguous() && dw.is_contiguous());
  const int64_t T = x.size(0), in = x.size(1), out = dy.size(1);
  TORCH_CHECK(dy.size(0) == T && dw.size(0) == out && dw.size(1) == in,
              "wgrad shapes: x[T,in] dy[T,out] dw[out,in]");

  static std::mutex mu;
  static std::unordered_map<AlgoKey, CachedPlan, AlgoKeyHash> plans;

  const AlgoKey key{in, out, T};
  CachedPlan plan;
  {
    std::lock_guard<std::mutex> lock(mu);
    auto it = plans.find(key);
    if (it != plans.end()) {
      plan = it->second;
    } else {
      LPP_CHECK_BLASLT(hipblasLtMatmulDescCreate(&plan.op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
      hipblasOperation_t opN = HIPBLAS_OP_N, opT = HIPBLAS_OP_T;
      LPP_CHECK_BLASLT(hipblasLtMatmulDescSetAttribute(
          plan.op, HIPBLASLT_MATMUL_DESC_TRANSA, &opN, sizeof(opN)));
      LPP_CHECK_BLASLT(hipblasLtMatmulDescSetAttribute(
          plan.op, HIPBLASLT_MATMUL_DESC_TRANSB, &opT, sizeof(opT)));
      // cm views: A = X [in, T] ld=in; B = dY [out, T] ld=out (opB=T); C/D = [in, out] ld=in
      LPP_CHECK_BLASLT(hipblasLtMatrixLayoutCreate(&plan.a, HIP_R_16BF, in, T, in));
      LPP_CHECK_BLASLT(hipblasLtMatrixLayoutCreate(&plan.b, HIP_R_16BF, out, T, out));
      LPP_CHECK_BLASLT(hipblasLtMatrixLayoutCreate(&plan.c, HIP_R_32F, in, out, in));

      hipblasLtMatmulPreference_t pref;
      LPP_CHECK_BLASLT(hipblasLtMatmulPreferenceCreate(&pref));
      size_t ws = kWorkspaceBytes;
      LPP_CHECK_BLASLT(hipblasLtMatmulPreferenceSetAttribute(
          pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
      hipblasLtMatmulHeuristicResult_t results[48];
      int found = 0;
      LPP_CHECK_BLASLT(hipblasLtMatmulAlgoGetHeuristic(
          lt_ctx().handle, plan.op, plan.a, plan.b, plan.c, plan.c, pref, 48, results,
          &found));
      hipblasLtMatmulPreferenceDestroy(pref);
      TORCH_CHECK(found > 0, "hipblaslt: no algo for wgrad shape [", out, ",", in,
                  "] k=", T);
      // measured pick: time each heuristic candidate once on scratch
      // operands (first call per shape only; the heuristic's first choice
      // was ~10% slower than the best for the 65B beta=1 f32-D shapes)
      plan.algo = results[0].algo;
      if (found > 1) {
        auto sx = at::empty({T, in}, x.options());
        auto sy = at::empty({T, out}, x.options());
        auto sw = at::zeros({out, in}, dw.options());
        auto wsbuf = at::empty({(int64_t)kWorkspaceBytes}, x.options().dtype(at::kByte));
        const float a1 = 1.0f, b1 = 1.0f;
        auto stream = current_stream();
        hipEvent_t ev0, ev1;
        LPP_CHECK_HIP(hipEventCreate(&ev0));
        LPP_CHECK_HIP(hipEventCreate(&ev1));
        float best = 1e30f;
        for (int i = 0; i < found; ++i) {
          // one warm + one timed run per candidate
          for (int rep = 0; rep < 2; ++rep) {
            if (rep == 1) LPP_CHECK_HIP(hipEventRecord(ev0, stream));
            hipblasStatus_t st = hipblasLtMatmul(
                lt_ctx().handle, plan.op, &a1, sx.data_ptr(), plan.a, sy.data_ptr(),
                plan.b, &b1, sw.data_ptr(), plan.c, sw.data_ptr(), plan.c,
                &results[i].algo, wsbuf.data_ptr(), kWorkspaceBytes, stream);
            if (st != HIPBLAS_STATUS_SUCCESS) { best = best; goto next_algo; }
          }
          LPP_CHECK_HIP(hipEventRecord(ev1, stream));
          LPP_CHECK_HIP(hipEventSynchronize(ev1));
          {
            float ms = 0.f;
            LPP_CHECK_HIP(hipEventElapsedTime(&ms, ev0, ev1));
            if (ms < best) { best = ms; plan.algo = results[i].algo; }
          }
        next_algo:;
        }
        LPP_CHECK_HIP(hipEventDestroy(ev0));
        LPP_CHECK_HIP(hipEventDestroy(ev1));
      }
      plans.emplace(key, plan);
    }
  }

  auto workspace = at::empty({(int64_t)kWorkspaceBytes},
                             x.options().dtype(at::kByte));
  const float alpha = 1.0f, beta = 1.0f;
  LPP_CHECK_BLASLT(hipblasLtMatmul(
      lt_ctx().handle, plan.op, &alpha, x.data_ptr(), plan.a, dy.data_ptr(), plan.b,
      &beta, dw.data_ptr(), plan.c, dw.data_ptr(), plan.c, &plan.algo,
      workspace.data_ptr(), kWorkspaceBytes, current_stream()));
}

}  // namespace lpp

void wgrad_f32_accum(at::Tensor x, at::Tensor dy, at::Tensor dw) {
  lpp::wgrad_f32_accum(x, dy, dw);
}
