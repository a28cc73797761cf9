// RoPE (rotate_half convention) forward + backward for gfx950.
//
// x: [B, S, H, D] (D contiguous). cos/sin: [Smax, D/2] fp32, precomputed on
// host (guide Appendix B: on-device trig makes this VALU-bound; table loads
// keep it at the HBM roofline). Backward = rotation by -theta (sin negated),
// same kernel.
//
// Thread mapping: one thread handles 4 rotation pairs — 8-byte load from the
// first half (x[d..d+3]) + 8-byte load from the second half (x[d+D/2..]),
// two 8-byte stores. D is a multiple of 8 for every LLaMA geometry (128).
//
// Reference op: SURVEY.md §2.7 "RoPE apply"; position_ids threading at
// models/llama_ds_mp_wrap.py:25,37,148. Oracle: lpp_amd.ops.apply_rope_ref.
#include "common.h"

namespace lpp {

template <typename T, bool BWD>
__global__ void rope_kernel(const T* __restrict__ x, const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t, T* __restrict__ y,
                            int64_t total_pairs4,  // B*S*H*(D/2/4)
                            int S, int H, int D, int pos_offset) {
  const int half = D / 2;
  const int pair4_per_head = half / 4;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total_pairs4;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int p4 = idx % pair4_per_head;           // which 4-pair group in the head
    const int64_t bsh = idx / pair4_per_head;      // flattened (b, s, h)
    const int s = (int)((bsh / H) % S);
    const int d0 = p4 * 4;
    const int64_t base = bsh * D + d0;

    using P4 = Pack<T, 4>;
    P4 x1 = *reinterpret_cast<const P4*>(x + base);
    P4 x2 = *reinterpret_cast<const P4*>(x + base + half);
    P4 y1, y2;
    const float* cr = cos_t + (int64_t)(pos_offset + s) * half + d0;
    const float* sr = sin_t + (int64_t)(pos_offset + s) * half + d0;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float c = cr[k];
      float sn = BWD ? -sr[k] : sr[k];
      float a = to_f32(x1.v[k]);
      float b = to_f32(x2.v[k]);
      y1.v[k] = from_f32<T>(a * c - b * sn);
      y2.v[k] = from_f32<T>(b * c + a * sn);
    }
    *reinterpret_cast<P4*>(y + base) = y1;
    *reinterpret_cast<P4*>(y + base + half) = y2;
  }
}

template <bool BWD>
at::Tensor rope_launch(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t, int64_t pos_offset) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4, "rope: x must be [B,S,H,D]");
  const int B = x.size(0), S = x.size(1), H = x.size(2), D = x.size(3);
  TORCH_CHECK(D % 8 == 0, "rope: head_dim must be a multiple of 8");
  TORCH_CHECK(cos_t.is_contiguous() && sin_t.is_contiguous());
  TORCH_CHECK(cos_t.size(0) >= pos_offset + S, "rope cache too short");
  auto y = at::empty_like(x);
  const int64_t total = (int64_t)B * S * H * (D / 8);
  auto stream = lpp::current_stream();
  const int grid = lpp::grid_for(total, 256);
  LPP_DISPATCH_FLOAT(x.scalar_type(), "rope", [&] {
    hipLaunchKernelGGL((lpp::rope_kernel<scalar_t, BWD>), dim3(grid), dim3(256), 0, stream,
                       (const scalar_t*)x.data_ptr(), cos_t.data_ptr<float>(),
                       sin_t.data_ptr<float>(), (scalar_t*)y.data_ptr(), total, S, H, D,
                       (int)pos_offset);
  });
  LPP_CHECK_HIP(hipGetLastError());
  return y;
}

}  // namespace lpp

at::Tensor rope_fwd(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t, int64_t pos_offset) {
  return lpp::rope_launch<false>(x, cos_t, sin_t, pos_offset);
}

at::Tensor rope_bwd(at::Tensor dy, at::Tensor cos_t, at::Tensor sin_t, int64_t pos_offset) {
  return lpp::rope_launch<true>(dy, cos_t, sin_t, pos_offset);
}
