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

// ---------------------------------------------------------------------------
// v2 (the shape-aligned fast path): 64-row x 128-col tiles through a
// transposed XOR image (the attention kernels' 4x4 dword butterfly +
// ((d>>3)^d)&7 block swizzle) — every LDS access is b64/b128 vectorised:
// load b128 -> in-register quad transpose -> 2x ds_write_b64 -> ds_read_b128
// -> global b128 store.  v1 above stays for ragged shapes and fp16.

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(2))) int int2v_t;

__device__ __forceinline__ int t2swz(int d, int r) {
  return (d * 64 + (r ^ ((((d >> 3) ^ d) & 7) << 3))) * 2;
}

__global__ __launch_bounds__(256) void transpose2d_v2_kernel(
    const short* __restrict__ in, short* __restrict__ out, int R, int C) {
  __shared__ __attribute__((aligned(16))) char img[128 * 64 * 2];  // 16 KB

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int st_r = (lane >> 4) & 3;    // quad row 0..3
  const int st_c = 8 * (lane & 15);    // col pack 0..120

  const int r0 = blockIdx.y * 64;
  const int c0 = blockIdx.x * 128;

#pragma unroll
  for (int pass = 0; pass < 4; ++pass) {
    const int r = 4 * wid + 16 * pass + st_r;
    const bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(
        in + (int64_t)(r0 + r) * C + c0 + st_c);
    // 4x4 dword butterfly across the lane quad {l, l^16, l^32, l^48}
    int dw[4];
#pragma unroll
    for (int k = 0; k < 4; ++k) dw[k] = reinterpret_cast<const int*>(&v)[k];
    {
      int t0 = __shfl_xor(dw[(st_r & 1) ^ 1], 16);
      int t1 = __shfl_xor(dw[((st_r & 1) ^ 1) | 2], 16);
      if (st_r & 1) { dw[0] = t0; dw[2] = t1; } else { dw[1] = t0; dw[3] = t1; }
    }
    {
      int lo = (st_r & 2) ? 0 : 2;
      int t0 = __shfl_xor(dw[lo], 32);
      int t1 = __shfl_xor(dw[lo + 1], 32);
      dw[lo] = t0; dw[lo + 1] = t1;
    }
    const int rq = 4 * wid + 16 * pass;  // quad's first row
    const int d0 = st_c + 2 * st_r;
#pragma unroll
    for (int e = 0; e < 2; ++e) {
      const int d = d0 + e;
      unsigned w01 = e ? (((unsigned)dw[0] >> 16) | ((unsigned)dw[1] & 0xffff0000u))
                       : (((unsigned)dw[0] & 0xffffu) | ((unsigned)dw[1] << 16));
      unsigned w23 = e ? (((unsigned)dw[2] >> 16) | ((unsigned)dw[3] & 0xffff0000u))
                       : (((unsigned)dw[2] & 0xffffu) | ((unsigned)dw[3] << 16));
      int2v_t pair = {(int)w01, (int)w23};
      *reinterpret_cast<int2v_t*>(img + t2swz(d, rq)) = pair;
    }
  }
  __syncthreads();

  // store: out[c0+d][r0 + 8p ..] — 8 consecutive lanes cover packs 0..7 of
  // one out row (a full 128-B line per 8-lane group); b128 LDS reads are
  // one swizzled 8-block each, banks spread by p^e within the group.
#pragma unroll
  for (int pass = 0; pass < 4; ++pass) {
    const int d = 32 * pass + (tid >> 3);
    const int p = tid & 7;
    const bf16x8_t v = *reinterpret_cast<const bf16x8_t*>(img + t2swz(d, 8 * p));
    *reinterpret_cast<bf16x8_t*>(out + (int64_t)(c0 + d) * R + r0 + 8 * p) = v;
  }
}

at::Tensor transpose2d(at::Tensor in) {
  TORCH_CHECK(in.is_cuda() && in.dim() == 2 && in.is_contiguous());
  TORCH_CHECK(in.scalar_type() == at::kBFloat16 || in.scalar_type() == at::kHalf,
              "transpose2d: bf16/fp16 only");
  const int R = (int)in.size(0), C = (int)in.size(1);
  auto out = at::empty({C, R}, in.options());
  auto stream = current_stream();
  if (in.scalar_type() == at::kBFloat16 && R % 64 == 0 && C % 128 == 0) {
    dim3 grid(C / 128, R / 64);
    hipLaunchKernelGGL(transpose2d_v2_kernel, grid, dim3(256), 0, stream,
                       reinterpret_cast<const short*>(in.data_ptr()),
                       reinterpret_cast<short*>(out.data_ptr()), R, C);
    LPP_CHECK_HIP(hipGetLastError());
    return out;
  }
  dim3 grid((C + kTile - 1) / kTile, (R + kTile - 1) / kTile);
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
