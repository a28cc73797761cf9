// Fused AdamW for mixed precision: bf16/fp16/fp32 params, fp32 master +
// moments + grads. One HBM sweep per tensor: grad scale (dp-average
// remainder x clip coefficient x inv loss scale), decoupled weight decay,
// moment updates, bias-corrected step, master update, low-precision
// write-back.
//
// Replaces the DeepSpeed fused-Adam + fp16-wrapper path the reference
// configures (conf/...yaml:122-128,137-143).
// Oracle: MixedPrecisionAdamW's torch._foreach eager path (tests share it).
#include "common.h"

namespace lpp {

template <typename T, int VEC>
__global__ void adamw_kernel(T* __restrict__ p, float* __restrict__ master,
                             const float* __restrict__ g, float* __restrict__ m,
                             float* __restrict__ v, int64_t n, float lr, float beta1,
                             float beta2, float eps, float wd, float bias1, float bias2,
                             float grad_scale) {
  using PT = Pack<T, VEC>;
  using PF = Pack<float, VEC>;
  const float step_size = lr / bias1;
  const float inv_sqrt_bias2 = rsqrtf(bias2);
  const float decay = 1.f - lr * wd;
  const int64_t n_vec = n / VEC;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    PF gp = reinterpret_cast<const PF*>(g)[i];
    PF mp = reinterpret_cast<PF*>(m)[i];
    PF vp = reinterpret_cast<PF*>(v)[i];
    PF w = reinterpret_cast<PF*>(master)[i];
    PT out;
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float gk = gp.v[k] * grad_scale;
      float mk = beta1 * mp.v[k] + (1.f - beta1) * gk;
      float vk = beta2 * vp.v[k] + (1.f - beta2) * gk * gk;
      float wk = w.v[k] * decay;
      wk -= step_size * mk / (sqrtf(vk) * inv_sqrt_bias2 + eps);
      mp.v[k] = mk;
      vp.v[k] = vk;
      w.v[k] = wk;
      out.v[k] = from_f32<T>(wk);
    }
    reinterpret_cast<PF*>(m)[i] = mp;
    reinterpret_cast<PF*>(v)[i] = vp;
    reinterpret_cast<PF*>(master)[i] = w;
    reinterpret_cast<PT*>(p)[i] = out;
  }
  // scalar tail
  const int64_t tail = n_vec * VEC;
  for (int64_t i = tail + blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float gk = g[i] * grad_scale;
    float mk = beta1 * m[i] + (1.f - beta1) * gk;
    float vk = beta2 * v[i] + (1.f - beta2) * gk * gk;
    float wk = master[i] * decay;
    wk -= step_size * mk / (sqrtf(vk) * inv_sqrt_bias2 + eps);
    m[i] = mk;
    v[i] = vk;
    master[i] = wk;
    p[i] = from_f32<T>(wk);
  }
}

}  // namespace lpp

void fused_adamw(std::vector<at::Tensor> params, std::vector<at::Tensor> masters,
                 std::vector<at::Tensor> grads, std::vector<at::Tensor> exp_avg,
                 std::vector<at::Tensor> exp_avg_sq, double lr, double beta1, double beta2,
                 double eps, double weight_decay, double bias1, double bias2,
                 double grad_scale) {
  auto stream = lpp::current_stream();
  for (size_t t = 0; t < params.size(); ++t) {
    auto& p = params[t];
    const int64_t n = p.numel();
    TORCH_CHECK(p.is_cuda() && p.is_contiguous());
    LPP_DISPATCH_FLOAT(p.scalar_type(), "fused_adamw", [&] {
      constexpr int VEC = 4;  // 4 fp32 state elems = 16B; param pack 4*sizeof(T)
      const int grid = lpp::grid_for((n + VEC - 1) / VEC, 256);
      hipLaunchKernelGGL((lpp::adamw_kernel<scalar_t, VEC>), dim3(grid), dim3(256), 0,
                         stream, (scalar_t*)p.data_ptr(), masters[t].data_ptr<float>(),
                         grads[t].data_ptr<float>(), exp_avg[t].data_ptr<float>(),
                         exp_avg_sq[t].data_ptr<float>(), n, (float)lr, (float)beta1,
                         (float)beta2, (float)eps, (float)weight_decay, (float)bias1,
                         (float)bias2, (float)grad_scale);
    });
  }
  LPP_CHECK_HIP(hipGetLastError());
}
