// Fast bf16 2-D transpose for the pre-transposed wgrad formulation.
//
// Why it exists: the fp32-accum wgrad GEMM in its natural layout is the
// k-strided hipBLASLt class (both operands contiguous along the OUTPUT
// dims) and tops out ~1.0-1.2 PF/s on MI355X; with both operands
// pre-transposed it becomes the k-contiguous class the forward GEMM uses
// and runs 1.33-1.67 PF/s (profiles/r02_gemm_probe2.txt).  The trade only
// pays if the transpose runs at HBM speed — torch's transpose-copy manages
// ~1 TB/s, this kernel is a tiled LDS transpose targeting >5 TB/s (r+w).
//
// Shape contract: in [R, C] row-major -> out [C, R] row-major, bf16/fp16.
// Tiles 64x64 via LDS with an odd-word row stride (66 elems = 33 words) so
// both the row-wise writes and column-wise reads are bank-conflict-free
// (32 banks x 4B on CDNA4).
#include "common.h"

namespace lpp {

constexpr int kTile = 64;
constexpr int kLdsStride = 66;  // 66*2B = 132B = 33 words (odd) per row

template <typename T>
__global__ __launch_bounds__(256) void transpose2d_kernel(
    const T* __restrict__ in, T* __restrict__ out, int R, int C) {
  __shared__ T tile[kTile * kLdsStride];

  const int c0 = blockIdx.x * kTile;
  const int r0 = blockIdx.y * kTile;

  // ---- load: 256 threads x 2 rows each, 8-elem packets along C
  // thread t handles (row = t/8 + 32*i, colpack = t%8)
  const int tcp = threadIdx.x & 7;         // which 8-elem packet in the tile row
  const int trow = threadIdx.x >> 3;       // 0..31
  const bool full = (r0 + kTile <= R) && (c0 + kTile <= C);
  if (full) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int r = trow + 32 * i;
      const Pack<T, 8> p = *reinterpret_cast<const Pack<T, 8>*>(
          in + (int64_t)(r0 + r) * C + c0 + tcp * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) tile[r * kLdsStride + tcp * 8 + e] = p.v[e];
    }
  } else {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int r = trow + 32 * i;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int rr = r0 + r, cc = c0 + tcp * 8 + e;
        tile[r * kLdsStride + tcp * 8 + e] =
            (rr < R && cc < C) ? in[(int64_t)rr * C + cc] : T{};
      }
    }
  }
  __syncthreads();

  // ---- store: out[C, R]; thread t writes row (c = t/8 + 32*i) of the
  // transposed tile, 8 contiguous elems along R
  if (full) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int c = trow + 32 * i;
      Pack<T, 8> p;
#pragma unroll
      for (int e = 0; e < 8; ++e) p.v[e] = tile[(tcp * 8 + e) * kLdsStride + c];
      *reinterpret_cast<Pack<T, 8>*>(out + (int64_t)(c0 + c) * R + r0 + tcp * 8) = p;
    }
  } else {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int c = trow + 32 * i;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int cc = c0 + c, rr = r0 + tcp * 8 + e;
        if (cc < C && rr < R)
          out[(int64_t)cc * R + rr] = tile[(tcp * 8 + e) * kLdsStride + c];
      }
    }
  }
}

at::Tensor transpose2d(at::Tensor in) {
  TORCH_CHECK(in.is_cuda() && in.dim() == 2 && in.is_contiguous());
  TORCH_CHECK(in.scalar_type() == at::kBFloat16 || in.scalar_type() == at::kHalf,
              "transpose2d: bf16/fp16 only");
  const int R = (int)in.size(0), C = (int)in.size(1);
  auto out = at::empty({C, R}, in.options());
  dim3 grid((C + kTile - 1) / kTile, (R + kTile - 1) / kTile);
  auto stream = current_stream();
  if (in.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(transpose2d_kernel<__hip_bfloat16>, grid, dim3(256), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(in.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()), R, C);
  } else {
    hipLaunchKernelGGL(transpose2d_kernel<__half>, grid, dim3(256), 0, stream,
                       reinterpret_cast<const __half*>(in.data_ptr()),
                       reinterpret_cast<__half*>(out.data_ptr()), R, C);
  }
  LPP_CHECK_HIP(hipGetLastError());
  return out;
}

}  // namespace lpp

at::Tensor transpose2d(at::Tensor in) { return lpp::transpose2d(in); }
