// dst_f32 += src_bf16 — the fp32 accumulation pass behind the bf16-D wgrad
// GEMM option (ops/linear.py LPP_WGRAD_BF16D):
//
//   hipBLASLt's fp32-D (BSS) solution pool runs 15-20% below the bf16-D
//   (BBS) pool at the 65B wgrad shapes even in the k-contiguous layout
//   (profiles/r02_wgrad_algo_sweep.json vs r02_gemm_probe2.txt), and the
//   fp32 C read of beta=1 is only ~5% of that gap — so computing dW in
//   bf16-D and accumulating here is net faster.  Numerics = the reference
//   stack's own flow (autograd bf16 dW + fp32 accumulation, SURVEY.md
//   §2.5): ONE bf16 rounding of each microbatch's dW; the cross-microbatch
//   accumulation stays fp32.  LPP_WGRAD_BF16D=0 restores the exact
//   fp32-D GEMM epilogue.
#include "common.h"

namespace lpp {

__global__ __launch_bounds__(256) void accum_bf16_f32_kernel(
    float* __restrict__ dst, const __hip_bfloat16* __restrict__ src, int64_t n) {
  const int64_t i0 = ((int64_t)blockIdx.x * 256 + threadIdx.x) * 8;
  const int64_t stride = (int64_t)gridDim.x * 256 * 8;
  for (int64_t i = i0; i + 8 <= n; i += stride) {
    const Pack<__hip_bfloat16, 8> s = *reinterpret_cast<const Pack<__hip_bfloat16, 8>*>(src + i);
    Pack<float, 4> d0 = *reinterpret_cast<const Pack<float, 4>*>(dst + i);
    Pack<float, 4> d1 = *reinterpret_cast<const Pack<float, 4>*>(dst + i + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) d0.v[j] += __bfloat162float(s.v[j]);
#pragma unroll
    for (int j = 0; j < 4; ++j) d1.v[j] += __bfloat162float(s.v[4 + j]);
    *reinterpret_cast<Pack<float, 4>*>(dst + i) = d0;
    *reinterpret_cast<Pack<float, 4>*>(dst + i + 4) = d1;
  }
  // ragged tail (n % 8) handled by the first threads
  const int64_t tail = n & ~(int64_t)7;
  const int64_t t = blockIdx.x * 256 + threadIdx.x;
  if (t < (n - tail)) dst[tail + t] += __bfloat162float(src[tail + t]);
}

void accum_bf16_f32(at::Tensor dst, at::Tensor src) {
  TORCH_CHECK(dst.is_cuda() && dst.scalar_type() == at::kFloat && dst.is_contiguous());
  TORCH_CHECK(src.is_cuda() && src.scalar_type() == at::kBFloat16 && src.is_contiguous());
  TORCH_CHECK(dst.numel() == src.numel(), "accum: size mismatch");
  const int64_t n = dst.numel();
  const int grid = grid_for((n + 7) / 8, 256);
  hipLaunchKernelGGL(accum_bf16_f32_kernel, dim3(grid), dim3(256), 0, current_stream(),
                     dst.data_ptr<float>(),
                     reinterpret_cast<const __hip_bfloat16*>(src.data_ptr()), n);
  LPP_CHECK_HIP(hipGetLastError());
}

}  // namespace lpp

void accum_bf16_f32(at::Tensor dst, at::Tensor src) { lpp::accum_bf16_f32(dst, src); }
