// Flash-style causal attention FORWARD for gfx950 (CDNA4), bf16, D=128.
//
// Replaces the materialised-mask eager attention the reference is stuck
// with (README.md:141-143: flash attention "did not work" under DS-PP;
// the [B,1,S,S] fp16 mask is built host-side, data/flan.py:194-243).
// Here the causal mask is implicit — nothing S^2-shaped exists.
//
// Structure (per the CDNA4 guide's attention recipe, correctness-first):
//   - workgroup = 4 waves, 128 q rows (32 per wave); grid (ceil(S/128), B*H)
//   - Q fragments live in registers (loaded once per workgroup)
//   - K tile [32][128] in LDS, XOR-swizzled ((row&15)<<4) so the 16-lane
//     ds_read_b128 groups hit 16 distinct bank slots (the row-major D=128
//     tile is otherwise a 16-way conflict -- guide G4)
//   - V tile stored TRANSPOSED [128][32+8pad] so the PV B-fragment read is
//     8 contiguous bf16; the +8 pad (80 B row stride) makes the 16-lane
//     group conflict-free (banks 20*r mod 64 are distinct)
//   - online softmax in exp2 domain (scores pre-scaled by 1/sqrt(D)*log2e),
//     per-row running (m, l) replicated across each 16-lane group via
//     __shfl_xor reductions
//   - P routed through a per-wave LDS tile (C-layout -> A-fragment layout)
//   - outputs: O bf16 and LSE2[B,H,S] fp32 (base-2 logsumexp of the scaled
//     scores; consumed by the backward kernels)
//
// MFMA fragment layouts as validated by mfma_test.hip.
#include "common.h"

namespace lpp {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int AF_D = 128;        // head dim
constexpr int AF_KVB = 32;       // kv tile rows
constexpr int AF_QW = 32;        // q rows per wave
constexpr int AF_WAVES = 4;
constexpr int AF_QB = AF_QW * AF_WAVES;  // q rows per workgroup
constexpr int AF_VPAD = 8;       // Vt row pad (elements)

__device__ __forceinline__ float bf2f(short s) {
  union { unsigned u; float f; } cv;
  cv.u = ((unsigned)(unsigned short)s) << 16;
  return cv.f;
}
__device__ __forceinline__ short f2bf(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// swizzled byte offset inside a [rows][128] bf16 LDS tile
__device__ __forceinline__ int kswz(int row, int col_elem) {
  return row * 256 + ((col_elem * 2) ^ ((row & 15) << 4));
}

__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, short* __restrict__ O, float* __restrict__ LSE2,
    int B, int S, int H, int HKV, float c /* scale*log2e */) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = reinterpret_cast<short*>(smem);                    // [32][128] swizzled
  short* vt_lds = k_lds + AF_KVB * AF_D;                            // [128][40]
  short* p_lds = vt_lds + AF_D * (AF_KVB + AF_VPAD);                // per wave [32][40]

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int lg = lane & 15;        // lane-in-group (n / col index)
  const int hi = lane >> 4;        // 16-lane group id (0..3)

  const int qblk = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int hkv = h / (H / HKV);

  const int64_t q_base = (((int64_t)b * S) * H + h) * AF_D;          // + s*H*D
  const int64_t kv_base = (((int64_t)b * S) * HKV + hkv) * AF_D;
  const int64_t sHD = (int64_t)H * AF_D;
  const int64_t sHkvD = (int64_t)HKV * AF_D;

  const int qwg0 = qblk * AF_QB;
  const int qw0 = qwg0 + wid * AF_QW;      // this wave's first q row

  // ---- Q fragments: [2 qtiles][4 ksteps], lane row = lg, k = hi*8+j ----
  bf16x8 qf[2][4];
#pragma unroll
  for (int qi = 0; qi < 2; ++qi) {
    const int row = qw0 + qi * 16 + lg;
    const int64_t rb = q_base + (int64_t)min(row, S - 1) * sHD;
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      qf[qi][kk] = *reinterpret_cast<const bf16x8*>(&Q[rb + kk * 32 + hi * 8]);
  }

  float o_acc[2][8][4];  // [qtile][dtile][reg]
#pragma unroll
  for (int qi = 0; qi < 2; ++qi)
#pragma unroll
    for (int dt = 0; dt < 8; ++dt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[qi][dt][r] = 0.f;
  float m_run[2][4], l_run[2][4];
#pragma unroll
  for (int qi = 0; qi < 2; ++qi)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[qi][r] = -INFINITY;
      l_run[qi][r] = 0.f;
    }

  const int kv_end = min(S, qwg0 + AF_QB);  // causal upper bound for the WG

  for (int kv0 = 0; kv0 < kv_end; kv0 += AF_KVB) {
    // ---- cooperative staging: K (swizzled) + V transposed (padded) ----
    __syncthreads();
    {
      // 32*128 elems / 256 threads = 16 elems (2x bf16x8) per thread
#pragma unroll
      for (int pass = 0; pass < 2; ++pass) {
        const int i = (tid + pass * 256) * 8;  // element index
        const int r = i / AF_D, ccol = i % AF_D;
        const int krow = kv0 + r;
        bf16x8 kv8;
        if (krow < S)
          kv8 = *reinterpret_cast<const bf16x8*>(&K[kv_base + (int64_t)krow * sHkvD + ccol]);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j) kv8[j] = 0;
        *reinterpret_cast<bf16x8*>(reinterpret_cast<char*>(k_lds) + kswz(r, ccol)) = kv8;
        bf16x8 vv8;
        if (krow < S)
          vv8 = *reinterpret_cast<const bf16x8*>(&V[kv_base + (int64_t)krow * sHkvD + ccol]);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j) vv8[j] = 0;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vt_lds[(ccol + j) * (AF_KVB + AF_VPAD) + r] = vv8[j];
      }
    }
    __syncthreads();

    if (kv0 >= qw0 + AF_QW) continue;  // no work for this wave (causal)

    // ---- S = Q K^T ----
    f32x4 s_acc[2][2];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int ki = 0; ki < 2; ++ki) s_acc[qi][ki] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      bf16x8 kf[2];
#pragma unroll
      for (int ki = 0; ki < 2; ++ki)
        kf[ki] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<char*>(k_lds) + kswz(ki * 16 + lg, kk * 32 + hi * 8));
#pragma unroll
      for (int qi = 0; qi < 2; ++qi)
#pragma unroll
        for (int ki = 0; ki < 2; ++ki)
          s_acc[qi][ki] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[qi][kk], kf[ki], s_acc[qi][ki], 0, 0, 0);
    }

    // ---- scale + causal mask + online softmax ----
    const bool diag = (kv0 + AF_KVB > qw0);  // tile crosses the diagonal
    float p[2][2][4];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qw0 + qi * 16 + hi * 4 + r;
        float s0 = s_acc[qi][0][r] * c;
        float s1 = s_acc[qi][1][r] * c;
        const int col0 = kv0 + lg;
        const int col1 = kv0 + 16 + lg;
        if (diag || kv0 + AF_KVB > S) {
          if (col0 > row || col0 >= S) s0 = -INFINITY;
          if (col1 > row || col1 >= S) s1 = -INFINITY;
        }
        float t = fmaxf(s0, s1);
#pragma unroll
        for (int m = 1; m < 16; m <<= 1) t = fmaxf(t, __shfl_xor(t, m));
        const float m_new = fmaxf(m_run[qi][r], t);
        const float alpha = exp2f(m_run[qi][r] - m_new);
        m_run[qi][r] = m_new;
        const float p0 = (s0 == -INFINITY) ? 0.f : exp2f(s0 - m_new);
        const float p1 = (s1 == -INFINITY) ? 0.f : exp2f(s1 - m_new);
        p[qi][0][r] = p0;
        p[qi][1][r] = p1;
        float ps = p0 + p1;
#pragma unroll
        for (int m = 1; m < 16; m <<= 1) ps += __shfl_xor(ps, m);
        l_run[qi][r] = l_run[qi][r] * alpha + ps;
#pragma unroll
        for (int dt = 0; dt < 8; ++dt) o_acc[qi][dt][r] *= alpha;
      }
    }

    // ---- P -> per-wave LDS (C-layout -> A-fragment layout) ----
    short* pw = p_lds + wid * AF_QW * (AF_KVB + AF_VPAD);
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int ki = 0; ki < 2; ++ki)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          pw[(qi * 16 + hi * 4 + r) * (AF_KVB + AF_VPAD) + ki * 16 + lg] =
              f2bf(p[qi][ki][r]);

    // ---- O += P V ----
    bf16x8 pf[2];
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
      pf[qi] = *reinterpret_cast<const bf16x8*>(
          &pw[(qi * 16 + lg) * (AF_KVB + AF_VPAD) + hi * 8]);
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      const bf16x8 vf = *reinterpret_cast<const bf16x8*>(
          &vt_lds[(dt * 16 + lg) * (AF_KVB + AF_VPAD) + hi * 8]);
#pragma unroll
      for (int qi = 0; qi < 2; ++qi) {
        f32x4 acc = {o_acc[qi][dt][0], o_acc[qi][dt][1], o_acc[qi][dt][2], o_acc[qi][dt][3]};
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf[qi], vf, acc, 0, 0, 0);
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[qi][dt][r] = acc[r];
      }
    }
  }

  // ---- epilogue: normalise + store O, LSE2 ----
#pragma unroll
  for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qw0 + qi * 16 + hi * 4 + r;
      if (row >= S) continue;
      const float inv_l = (l_run[qi][r] > 0.f) ? 1.f / l_run[qi][r] : 0.f;
      const int64_t rb = q_base + (int64_t)row * sHD;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt)
        O[rb + dt * 16 + lg] = f2bf(o_acc[qi][dt][r] * inv_l);
      if (lg == 0)
        LSE2[((int64_t)b * H + h) * S + row] = m_run[qi][r] + log2f(l_run[qi][r]);
    }
  }
}

}  // namespace lpp

std::vector<at::Tensor> attention_fwd(at::Tensor q, at::Tensor k, at::Tensor v) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16, "attention_fwd: bf16 only");
  TORCH_CHECK(q.dim() == 4 && q.size(3) == lpp::AF_D, "attention_fwd: [B,S,H,128] required");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  const int B = q.size(0), S = q.size(1), H = q.size(2);
  const int HKV = k.size(2);
  TORCH_CHECK(H % HKV == 0);
  auto o = at::empty_like(q);
  auto lse2 = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  const float scale = 1.0f / std::sqrt((float)lpp::AF_D);
  const float c = scale * 1.4426950408889634f;  // log2(e)
  const int qblocks = (S + lpp::AF_QB - 1) / lpp::AF_QB;
  const size_t lds = (lpp::AF_KVB * lpp::AF_D + lpp::AF_D * (lpp::AF_KVB + lpp::AF_VPAD) +
                      lpp::AF_WAVES * lpp::AF_QW * (lpp::AF_KVB + lpp::AF_VPAD)) *
                     sizeof(short);
  hipLaunchKernelGGL(lpp::attn_fwd_kernel, dim3(qblocks, B * H), dim3(256), lds,
                     lpp::current_stream(), (const short*)q.data_ptr(),
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                     (short*)o.data_ptr(), lse2.data_ptr<float>(), B, S, H, HKV, c);
  LPP_CHECK_HIP(hipGetLastError());
  return {o, lse2};
}
