// RMSNorm forward + backward for gfx950.
//
// Memory-bound: one workgroup (256 threads = 4 waves) per row, bf16 elements
// moved as 16-byte packets (8 x bf16). Forward reads x once, writes y + a
// per-row fp32 inv_rms for backward. Backward recomputes nothing: one pass
// for the per-row dot, one for dx, with dw accumulated per-block in LDS and
// flushed with one atomicAdd per element per block (contention = #blocks).
//
// Reference op: HF LlamaRMSNorm as used by the reference repo
// (models/llama_ds_mp_wrap.py:12,184-188). Oracle: lpp_amd.ops.rmsnorm_ref.
#include "common.h"

namespace lpp {

constexpr int BLOCK = 256;

template <typename T, int VEC>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x, const float* __restrict__ w,
                                   T* __restrict__ y, float* __restrict__ invrms,
                                   int64_t n_rows, int H, float eps) {
  __shared__ float red[BLOCK / kWave];
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T* xr = x + row * H;
    T* yr = y + row * H;
    using PV = Pack<T, VEC>;
    float ss = 0.f;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      PV buf = *reinterpret_cast<const PV*>(xr + i);
#pragma unroll
      for (int v = 0; v < VEC; ++v) {
        float f = to_f32(buf.v[v]);
        ss += f * f;
      }
    }
    float total = block_reduce_sum<BLOCK>(ss, red);
    float inv = rsqrtf(total / (float)H + eps);
    if (threadIdx.x == 0) invrms[row] = inv;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      PV xin = *reinterpret_cast<const PV*>(xr + i);
      PV out;
#pragma unroll
      for (int v = 0; v < VEC; ++v)
        out.v[v] = from_f32<T>(to_f32(xin.v[v]) * inv * w[i + v]);
      *reinterpret_cast<PV*>(yr + i) = out;
    }
  }
}

// Scalar fallback for H not divisible by 16 bytes.
template <typename T>
__global__ void rmsnorm_fwd_kernel_s(const T* __restrict__ x, const float* __restrict__ w,
                                     T* __restrict__ y, float* __restrict__ invrms,
                                     int64_t n_rows, int H, float eps) {
  __shared__ float red[BLOCK / kWave];
  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T* xr = x + row * H;
    T* yr = y + row * H;
    float ss = 0.f;
    for (int i = threadIdx.x; i < H; i += BLOCK) {
      float f = to_f32(xr[i]);
      ss += f * f;
    }
    float total = block_reduce_sum<BLOCK>(ss, red);
    float inv = rsqrtf(total / (float)H + eps);
    if (threadIdx.x == 0) invrms[row] = inv;
    for (int i = threadIdx.x; i < H; i += BLOCK)
      yr[i] = from_f32<T>(to_f32(xr[i]) * inv * w[i]);
  }
}

// Vectorised backward for H % VEC == 0 and H <= BLOCK*VEC*MAXIT (the
// production shapes): dy/x move as 16-byte packets, each packet is loaded
// ONCE and kept in registers for both the dot and the dx pass, and the
// per-thread dw partials accumulate in REGISTERS across the row loop
// (thread i's column set is row-invariant) and flush as one vectorised
// non-atomic store per block into dw_partial[block][H]; the host reduces
// partials with a single torch sum. Deterministic dw (no atomics).
// dx = inv*w*dy - x*inv^3/H * dot;  dw = sum_rows(dy*x*inv).
template <typename T, int VEC, int N_IT>
__global__ void rmsnorm_bwd_kernel_v(const T* __restrict__ dy, const T* __restrict__ x,
                                     const float* __restrict__ w,
                                     const float* __restrict__ invrms, T* __restrict__ dx,
                                     float* __restrict__ dw_partial, int64_t n_rows,
                                     int H) {
  __shared__ float red[BLOCK / kWave];
  using PV = Pack<T, VEC>;
  // N_IT is compile-time so the packet/partial arrays stay in registers
  // (a runtime bound would dynamic-index them into scratch).
  alignas(16) float dwacc[N_IT][VEC];
  // w is row-invariant but the block-reduce barrier stops the compiler
  // hoisting its loads out of the row loop — preload to registers once.
  alignas(16) float wreg[N_IT][VEC];
#pragma unroll
  for (int it = 0; it < N_IT; ++it) {
    const int i = threadIdx.x * VEC + it * BLOCK * VEC;
    *reinterpret_cast<Pack<float, 4>*>(&wreg[it][0]) =
        *reinterpret_cast<const Pack<float, 4>*>(w + i);
    *reinterpret_cast<Pack<float, 4>*>(&wreg[it][4]) =
        *reinterpret_cast<const Pack<float, 4>*>(w + i + 4);
#pragma unroll
    for (int v = 0; v < VEC; ++v) dwacc[it][v] = 0.f;
  }

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T* dyr = dy + row * H;
    const T* xr = x + row * H;
    T* dxr = dx + row * H;
    const float inv = invrms[row];
    PV dybuf[N_IT], xbuf[N_IT];
    float dot = 0.f;
#pragma unroll
    for (int it = 0; it < N_IT; ++it) {
      const int i = threadIdx.x * VEC + it * BLOCK * VEC;
      dybuf[it] = *reinterpret_cast<const PV*>(dyr + i);
      xbuf[it] = *reinterpret_cast<const PV*>(xr + i);
#pragma unroll
      for (int v = 0; v < VEC; ++v)
        dot += to_f32(dybuf[it].v[v]) * wreg[it][v] * to_f32(xbuf[it].v[v]);
    }
    const float total = block_reduce_sum<BLOCK>(dot, red);
    const float k = total * inv * inv * inv / (float)H;
#pragma unroll
    for (int it = 0; it < N_IT; ++it) {
      const int i = threadIdx.x * VEC + it * BLOCK * VEC;
      PV out;
#pragma unroll
      for (int v = 0; v < VEC; ++v) {
        const float d = to_f32(dybuf[it].v[v]);
        const float xi = to_f32(xbuf[it].v[v]);
        out.v[v] = from_f32<T>(d * wreg[it][v] * inv - xi * k);
        dwacc[it][v] += d * xi * inv;
      }
      *reinterpret_cast<PV*>(dxr + i) = out;
    }
  }
  float* my = dw_partial + (int64_t)blockIdx.x * H;
#pragma unroll
  for (int it = 0; it < N_IT; ++it) {
    const int i = threadIdx.x * VEC + it * BLOCK * VEC;
    *reinterpret_cast<Pack<float, 4>*>(my + i) = *reinterpret_cast<Pack<float, 4>*>(&dwacc[it][0]);
    *reinterpret_cast<Pack<float, 4>*>(my + i + 4) = *reinterpret_cast<Pack<float, 4>*>(&dwacc[it][4]);
  }
}

// dx = inv * w * dy - x * inv^3 / H * sum(dy * w * x)
// dw += sum_rows(dy * x * inv)
template <typename T>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                   const float* __restrict__ w,
                                   const float* __restrict__ invrms, T* __restrict__ dx,
                                   float* __restrict__ dw, int64_t n_rows, int H) {
  __shared__ float red[BLOCK / kWave];
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dw_loc = reinterpret_cast<float*>(smem);  // [H]
  for (int i = threadIdx.x; i < H; i += BLOCK) dw_loc[i] = 0.f;
  __syncthreads();

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T* dyr = dy + row * H;
    const T* xr = x + row * H;
    T* dxr = dx + row * H;
    const float inv = invrms[row];
    float dot = 0.f;
    for (int i = threadIdx.x; i < H; i += BLOCK) {
      float d = to_f32(dyr[i]);
      float xi = to_f32(xr[i]);
      dot += d * w[i] * xi;
      dw_loc[i] += d * xi * inv;
    }
    float total = block_reduce_sum<BLOCK>(dot, red);
    float k = total * inv * inv * inv / (float)H;
    for (int i = threadIdx.x; i < H; i += BLOCK) {
      float d = to_f32(dyr[i]);
      float xi = to_f32(xr[i]);
      dxr[i] = from_f32<T>(d * w[i] * inv - xi * k);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < H; i += BLOCK) atomicAdd(&dw[i], dw_loc[i]);
}

}  // namespace lpp

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor weight, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "rmsnorm_fwd: x must be contiguous on GPU");
  const int H = x.size(-1);
  const int64_t n_rows = x.numel() / H;
  auto y = at::empty_like(x);
  auto invrms = at::empty({n_rows}, x.options().dtype(at::kFloat));
  auto w32 = weight.to(at::kFloat).contiguous();
  auto stream = lpp::current_stream();
  const int grid = lpp::grid_for(n_rows, 1, 4096);
  LPP_DISPATCH_FLOAT(x.scalar_type(), "rmsnorm_fwd", [&] {
    constexpr int VEC = 16 / sizeof(scalar_t);
    if (H % VEC == 0) {
      hipLaunchKernelGGL((lpp::rmsnorm_fwd_kernel<scalar_t, VEC>), dim3(grid),
                         dim3(lpp::BLOCK), 0, stream, (const scalar_t*)x.data_ptr(),
                         w32.data_ptr<float>(), (scalar_t*)y.data_ptr(),
                         invrms.data_ptr<float>(), n_rows, H, (float)eps);
    } else {
      hipLaunchKernelGGL((lpp::rmsnorm_fwd_kernel_s<scalar_t>), dim3(grid),
                         dim3(lpp::BLOCK), 0, stream, (const scalar_t*)x.data_ptr(),
                         w32.data_ptr<float>(), (scalar_t*)y.data_ptr(),
                         invrms.data_ptr<float>(), n_rows, H, (float)eps);
    }
  });
  LPP_CHECK_HIP(hipGetLastError());
  return {y, invrms};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor weight,
                                    at::Tensor invrms) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const int H = x.size(-1);
  const int64_t n_rows = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw32 = at::zeros({H}, x.options().dtype(at::kFloat));
  auto w32 = weight.to(at::kFloat).contiguous();
  auto stream = lpp::current_stream();
  LPP_DISPATCH_FLOAT(x.scalar_type(), "rmsnorm_bwd", [&] {
    constexpr int VEC = 16 / sizeof(scalar_t);
    const int n_it = H / (lpp::BLOCK * VEC);
    if (VEC == 8 && H % (lpp::BLOCK * VEC) == 0 && n_it >= 1 && n_it <= 4) {
      const int grid = lpp::grid_for(n_rows, 1, 2048);
      auto partial = at::empty({grid, H}, x.options().dtype(at::kFloat));
      auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(grid), dim3(lpp::BLOCK), 0, stream,
                           (const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
                           w32.data_ptr<float>(), invrms.data_ptr<float>(),
                           (scalar_t*)dx.data_ptr(), partial.data_ptr<float>(), n_rows, H);
      };
      if (n_it == 1) launch(lpp::rmsnorm_bwd_kernel_v<scalar_t, VEC, 1>);
      else if (n_it == 2) launch(lpp::rmsnorm_bwd_kernel_v<scalar_t, VEC, 2>);
      else if (n_it == 3) launch(lpp::rmsnorm_bwd_kernel_v<scalar_t, VEC, 3>);
      else launch(lpp::rmsnorm_bwd_kernel_v<scalar_t, VEC, 4>);
      LPP_CHECK_HIP(hipGetLastError());
      at::sum_out(dw32, partial, {0});
    } else {
      const int grid = lpp::grid_for(n_rows, 1, 1024);
      const size_t lds = (size_t)H * sizeof(float);
      TORCH_CHECK(lds <= 160 * 1024 - 4096, "rmsnorm_bwd: H too large for LDS dw buffer");
      hipLaunchKernelGGL((lpp::rmsnorm_bwd_kernel<scalar_t>), dim3(grid), dim3(lpp::BLOCK),
                         lds, stream, (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)x.data_ptr(), w32.data_ptr<float>(),
                         invrms.data_ptr<float>(), (scalar_t*)dx.data_ptr(),
                         dw32.data_ptr<float>(), n_rows, H);
    }
  });
  LPP_CHECK_HIP(hipGetLastError());
  return {dx, dw32.to(weight.scalar_type())};
}
