// MFMA fragment-layout validation for gfx950 (v_mfma_f32_16x16x32_bf16).
//
// Assumed layouts (verified on hardware by tests/test_gpu_kernels.py
// ::test_mfma_layout against torch.matmul with ASYMMETRIC operands):
//   A[16m][32k]: lane l holds m = l&15, k = (l>>4)*8 + j, j=0..7  (bf16x8)
//   B[32k][16n]: lane l holds n = l&15, k = (l>>4)*8 + j
//   C[16m][16n]: lane l holds n = l&15, m = (l>>4)*4 + r, r=0..3  (f32x4)
// These are the layouts the attention kernels build their fragments for.
#include "common.h"

namespace lpp {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void mfma_16x16x32_kernel(const __hip_bfloat16* __restrict__ A,
                                     const __hip_bfloat16* __restrict__ B,
                                     float* __restrict__ C) {
  const int l = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int m = l & 15, ka = (l >> 4) * 8 + j;
    a[j] = *reinterpret_cast<const short*>(&A[m * 32 + ka]);
    const int n = l & 15, kb = (l >> 4) * 8 + j;
    b[j] = *reinterpret_cast<const short*>(&B[kb * 16 + n]);
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int n = l & 15, m = (l >> 4) * 4 + r;
    C[m * 16 + n] = c[r];
  }
}

// v_mfma_f32_32x32x16_bf16 layout (used by the 8-wave attention forward):
//   A[32m][16k]: lane l holds m = l&31, k = (l>>5)*8 + j   (bf16x8)
//   B[16k][32n]: lane l holds n = l&31, k = (l>>5)*8 + j
//   C[32m][32n]: lane l holds n = l&31, m = (r&3) + 8*(r>>2) + 4*(l>>5),
//                r = 0..15 (f32x16)
typedef __attribute__((ext_vector_type(16))) float f32x16;

__global__ void mfma_32x32x16_kernel(const __hip_bfloat16* __restrict__ A,
                                     const __hip_bfloat16* __restrict__ B,
                                     float* __restrict__ C) {
  const int l = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int m = l & 31, k = (l >> 5) * 8 + j;
    a[j] = *reinterpret_cast<const short*>(&A[m * 16 + k]);
    const int n = l & 31;
    b[j] = *reinterpret_cast<const short*>(&B[k * 32 + n]);
  }
  f32x16 c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int n = l & 31, m = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    C[m * 32 + n] = c[r];
  }
}

}  // namespace lpp

at::Tensor mfma_test_32x32x16(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}) && B.sizes() == at::IntArrayRef({16, 32}));
  auto C = at::zeros({32, 32}, A.options().dtype(at::kFloat));
  hipLaunchKernelGGL(lpp::mfma_32x32x16_kernel, dim3(1), dim3(64), 0,
                     lpp::current_stream(),
                     (const __hip_bfloat16*)A.contiguous().data_ptr(),
                     (const __hip_bfloat16*)B.contiguous().data_ptr(),
                     C.data_ptr<float>());
  LPP_CHECK_HIP(hipGetLastError());
  return C;
}

at::Tensor mfma_test_16x16x32(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({16, 32}) && B.sizes() == at::IntArrayRef({32, 16}));
  auto C = at::zeros({16, 16}, A.options().dtype(at::kFloat));
  hipLaunchKernelGGL(lpp::mfma_16x16x32_kernel, dim3(1), dim3(64), 0,
                     lpp::current_stream(),
                     (const __hip_bfloat16*)A.contiguous().data_ptr(),
                     (const __hip_bfloat16*)B.contiguous().data_ptr(),
                     C.data_ptr<float>());
  LPP_CHECK_HIP(hipGetLastError());
  return C;
}
