// Fused SwiGLU: y = silu(gate) * up, forward + backward, gfx950.
//
// Pure elementwise / memory-bound: grid-stride, 16-byte packets per lane
// (8 x bf16). Backward recomputes sigmoid from the saved gate/up inputs —
// no intermediate is ever materialised (the eager path writes silu(g) to
// HBM and reads it back; SURVEY.md §2.7 MLP row).
//
// Oracle: lpp_amd.ops.swiglu_ref.
#include "common.h"

namespace lpp {

template <typename T, int VEC>
__global__ void swiglu_fwd_kernel(const T* __restrict__ g, const T* __restrict__ u,
                                  T* __restrict__ y, int64_t n_vec) {
  using PV = Pack<T, VEC>;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    PV gp = reinterpret_cast<const PV*>(g)[i];
    PV up = reinterpret_cast<const PV*>(u)[i];
    PV yp;
#pragma unroll
    for (int v = 0; v < VEC; ++v) {
      float gf = to_f32(gp.v[v]);
      float s = 1.f / (1.f + __expf(-gf));
      yp.v[v] = from_f32<T>(gf * s * to_f32(up.v[v]));
    }
    reinterpret_cast<PV*>(y)[i] = yp;
  }
}

template <typename T, int VEC>
__global__ void swiglu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ g,
                                  const T* __restrict__ u, T* __restrict__ dg,
                                  T* __restrict__ du, int64_t n_vec) {
  using PV = Pack<T, VEC>;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_vec;
       i += (int64_t)gridDim.x * blockDim.x) {
    PV dyp = reinterpret_cast<const PV*>(dy)[i];
    PV gp = reinterpret_cast<const PV*>(g)[i];
    PV up = reinterpret_cast<const PV*>(u)[i];
    PV dgp, dup;
#pragma unroll
    for (int v = 0; v < VEC; ++v) {
      float d = to_f32(dyp.v[v]);
      float gf = to_f32(gp.v[v]);
      float uf = to_f32(up.v[v]);
      float s = 1.f / (1.f + __expf(-gf));
      float silu = gf * s;
      float dsilu = s * (1.f + gf * (1.f - s));
      dgp.v[v] = from_f32<T>(d * uf * dsilu);
      dup.v[v] = from_f32<T>(d * silu);
    }
    reinterpret_cast<PV*>(dg)[i] = dgp;
    reinterpret_cast<PV*>(du)[i] = dup;
  }
}

}  // namespace lpp

at::Tensor swiglu_fwd(at::Tensor gate, at::Tensor up) {
  TORCH_CHECK(gate.is_cuda() && gate.is_contiguous() && up.is_contiguous());
  TORCH_CHECK(gate.sizes() == up.sizes() && gate.scalar_type() == up.scalar_type());
  auto y = at::empty_like(gate);
  const int64_t n = gate.numel();
  auto stream = lpp::current_stream();
  LPP_DISPATCH_FLOAT(gate.scalar_type(), "swiglu_fwd", [&] {
    constexpr int VEC = 16 / sizeof(scalar_t);
    TORCH_CHECK(n % VEC == 0, "swiglu: numel must be divisible by ", VEC);
    const int64_t n_vec = n / VEC;
    const int grid = lpp::grid_for(n_vec, 256);
    hipLaunchKernelGGL((lpp::swiglu_fwd_kernel<scalar_t, VEC>), dim3(grid), dim3(256), 0,
                       stream, (const scalar_t*)gate.data_ptr(),
                       (const scalar_t*)up.data_ptr(), (scalar_t*)y.data_ptr(), n_vec);
  });
  LPP_CHECK_HIP(hipGetLastError());
  return y;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor gate, at::Tensor up) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  auto dg = at::empty_like(gate);
  auto du = at::empty_like(up);
  const int64_t n = gate.numel();
  auto stream = lpp::current_stream();
  LPP_DISPATCH_FLOAT(gate.scalar_type(), "swiglu_bwd", [&] {
    constexpr int VEC = 16 / sizeof(scalar_t);
    const int64_t n_vec = n / VEC;
    const int grid = lpp::grid_for(n_vec, 256);
    hipLaunchKernelGGL((lpp::swiglu_bwd_kernel<scalar_t, VEC>), dim3(grid), dim3(256), 0,
                       stream, (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)gate.data_ptr(), (const scalar_t*)up.data_ptr(),
                       (scalar_t*)dg.data_ptr(), (scalar_t*)du.data_ptr(), n_vec);
  });
  LPP_CHECK_HIP(hipGetLastError());
  return {dg, du};
}
