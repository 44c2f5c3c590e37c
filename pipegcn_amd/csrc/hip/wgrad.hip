// Fused dual weight-gradient GEMM for the GraphSAGE layer backward
// (gfx950 MFMA, exact fp32):
//
//     gw1[N,K] = g[M,N]^T @ x1[M,K]      gw2[N,K] = g^T @ x2[M,K]
//
// replaces the reference's two library GEMMs (`g.t() @ x`,
// /root/reference/module/layer.py:51 backward) with ONE kernel that
// streams each g tile from HBM once and feeds BOTH products — the two
// wgrads of a dual-linear layer share the output gradient, which rocBLAS
// reads twice. The reduction axis is M (graph nodes, 10^5..10^7), so the
// kernel splits M across blockIdx.z into a [S,N,K] fp32 workspace and a
// deterministic torch sum finishes (no float atomics => bitwise
// reproducible across runs).
//
// Uses v_mfma_f32_32x32x2_f32 (f32 in / f32 accumulate — an exact fmaf
// chain, no TF32; cdna_hip_programming.md §3). Block: 256 threads,
// 64(N) x 64(K) output tile, 4 waves each owning one 32x32 fragment per
// product; M staged through LDS 32 rows at a time (g/x tiles padded +1
// dword against bank conflicts). x2 == nullptr computes gw1 only (plain
// single-linear wgrad: tail layers, GCN).

#include "../common.h"

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

namespace {

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

constexpr int WB_N = 64;  // output rows per block (N dim of gw)
constexpr int WB_K = 64;  // output cols per block (K dim of gw)
constexpr int WB_M = 32;  // reduction rows staged per iteration

template <bool DUAL>
__global__ __launch_bounds__(256) void dual_wgrad_kernel(
    const float* __restrict__ g, const float* __restrict__ x1,
    const float* __restrict__ x2, float* __restrict__ ws1,
    float* __restrict__ ws2, int64_t M, int64_t N, int64_t K,
    int64_t chunk) {
  __shared__ float g_tl[WB_M][WB_N + 1];
  __shared__ float x1_tl[WB_M][WB_K + 1];
  __shared__ float x2_tl[WB_M][WB_K + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // 0/1: which 32-row band of the N tile
  const int wc = wave & 1;   // 0/1: which 32-col band of the K tile
  const int l31 = lane & 31;
  const int lk = lane >> 5;  // 0/1: m within the MFMA K=2 step

  const int64_t n0 = static_cast<int64_t>(blockIdx.x) * WB_N;
  const int64_t k0 = static_cast<int64_t>(blockIdx.y) * WB_K;
  const int64_t m_begin = static_cast<int64_t>(blockIdx.z) * chunk;
  const int64_t m_end = min(M, m_begin + chunk);

  f32x16 acc1 = {};
  f32x16 acc2 = {};

  // staging geometry: 32x64 tile, 8 floats per thread (one row slice)
  const int st_r = tid >> 3;        // 0..31
  const int st_c = (tid & 7) * 8;   // 0,8,..,56

  const bool nk_full = (n0 + WB_N <= N) && (k0 + WB_K <= K);

  for (int64_t m0 = m_begin; m0 < m_end; m0 += WB_M) {
    const bool full = nk_full && (m0 + WB_M <= m_end);
    if (full) {
#pragma unroll
      for (int q = 0; q < 8; ++q)
        g_tl[st_r][st_c + q] = g[(m0 + st_r) * N + n0 + st_c + q];
#pragma unroll
      for (int q = 0; q < 8; ++q)
        x1_tl[st_r][st_c + q] = x1[(m0 + st_r) * K + k0 + st_c + q];
      if (DUAL) {
#pragma unroll
        for (int q = 0; q < 8; ++q)
          x2_tl[st_r][st_c + q] = x2[(m0 + st_r) * K + k0 + st_c + q];
      }
    } else {
      const bool mrow = m0 + st_r < m_end;
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        const int64_t nn = n0 + st_c + q;
        g_tl[st_r][st_c + q] =
            (mrow && nn < N) ? g[(m0 + st_r) * N + nn] : 0.f;
        const int64_t kk = k0 + st_c + q;
        const bool kin = mrow && kk < K;
        x1_tl[st_r][st_c + q] = kin ? x1[(m0 + st_r) * K + kk] : 0.f;
        if (DUAL)
          x2_tl[st_r][st_c + q] = kin ? x2[(m0 + st_r) * K + kk] : 0.f;
      }
    }
    __syncthreads();
#pragma unroll
    for (int mm = 0; mm < WB_M; mm += 2) {
      const float a = g_tl[mm + lk][wr * 32 + l31];
      const float b1 = x1_tl[mm + lk][wc * 32 + l31];
      acc1 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b1, acc1, 0, 0, 0);
      if (DUAL) {
        const float b2 = x2_tl[mm + lk][wc * 32 + l31];
        acc2 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b2, acc2, 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: C layout for 32x32 shapes (same as dual_gemm.hip):
  // col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const int64_t col = k0 + wc * 32 + l31;
  if (col >= K) return;
  float* w1p = ws1 + static_cast<int64_t>(blockIdx.z) * N * K;
  float* w2p = DUAL ? ws2 + static_cast<int64_t>(blockIdx.z) * N * K
                    : nullptr;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int64_t row = n0 + wr * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * lk;
    if (row < N) {
      w1p[row * K + col] = acc1[reg];
      if (DUAL) w2p[row * K + col] = acc2[reg];
    }
  }
}

}  // namespace

std::vector<torch::Tensor> dual_wgrad_hip(torch::Tensor g, torch::Tensor x1,
                                          torch::Tensor x2) {
  TORCH_CHECK(g.is_cuda() && g.scalar_type() == torch::kFloat,
              "dual_wgrad: fp32 CUDA only");
  TORCH_CHECK(g.is_contiguous() && x1.is_contiguous());
  const int64_t M = g.size(0);
  const int64_t N = g.size(1);
  const int64_t K = x1.size(1);
  TORCH_CHECK(x1.size(0) == M);
  const bool dual = x2.defined() && x2.numel() > 0;
  if (dual) {
    TORCH_CHECK(x2.is_contiguous() && x2.size(0) == M && x2.size(1) == K);
  }

  const int64_t tiles =
      ((N + WB_N - 1) / WB_N) * ((K + WB_K - 1) / WB_K);
  // enough blocks to fill 256 CUs several times over, M chunks 32-aligned
  int64_t S = std::min<int64_t>((1024 + tiles - 1) / tiles, 64);
  S = std::min<int64_t>(S, (M + WB_M - 1) / WB_M);
  S = std::max<int64_t>(S, 1);
  const int64_t chunk = ((M + S - 1) / S + WB_M - 1) / WB_M * WB_M;

  auto opt = g.options();
  auto ws1 = torch::empty({S, N, K}, opt);
  auto ws2 = dual ? torch::empty({S, N, K}, opt) : torch::Tensor();
  dim3 grid((N + WB_N - 1) / WB_N, (K + WB_K - 1) / WB_K, S);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  if (dual)
    hipLaunchKernelGGL(HIP_KERNEL_NAME(dual_wgrad_kernel<true>), grid,
                       dim3(256), 0, stream, g.data_ptr<float>(),
                       x1.data_ptr<float>(), x2.data_ptr<float>(),
                       ws1.data_ptr<float>(), ws2.data_ptr<float>(), M, N, K,
                       chunk);
  else
    hipLaunchKernelGGL(HIP_KERNEL_NAME(dual_wgrad_kernel<false>), grid,
                       dim3(256), 0, stream, g.data_ptr<float>(),
                       x1.data_ptr<float>(), nullptr, ws1.data_ptr<float>(),
                       nullptr, M, N, K, chunk);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));
  if (S == 1) {
    if (dual) return {ws1.squeeze(0), ws2.squeeze(0)};
    return {ws1.squeeze(0)};
  }
  // deterministic split-M reduce (no float atomics anywhere)
  if (dual) return {ws1.sum(0), ws2.sum(0)};
  return {ws1.sum(0)};
}
