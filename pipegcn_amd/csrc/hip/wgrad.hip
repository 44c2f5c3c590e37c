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
// BNxBK output tile (64x64 default; see the host-side note on why the
// lower-traffic 128x128 variant measured SLOWER), 4 waves in a 2x2
// quadrant layout,
// FN x FK 32x32 fragments per wave per product; M staged through LDS 32
// rows at a time (+1-dword row padding against bank conflicts).
// x2 == nullptr computes gw1 only (tail linears, GCN).

#include "../common.h"

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

#include <cstdlib>

namespace {

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

template <bool DUAL, int BN, int BK>
__global__ __launch_bounds__(256) void dual_wgrad_kernel(
    const float* __restrict__ g, const float* __restrict__ x1,
    const float* __restrict__ x2, float* __restrict__ ws1,
    float* __restrict__ ws2, int64_t M, int64_t N, int64_t K,
    int64_t chunk) {
  constexpr int WB_M = 32;     // reduction rows staged per iteration
  constexpr int FN = BN / 64;  // 32x32 fragments per wave, N dim
  constexpr int FK = BK / 64;  // 32x32 fragments per wave, K dim
  __shared__ float g_tl[WB_M][BN + 1];
  __shared__ float x1_tl[WB_M][BK + 1];
  __shared__ float x2_tl[WB_M][BK + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // 0/1: which (BN/2)-row band of the N tile
  const int wc = wave & 1;   // 0/1: which (BK/2)-col band of the K tile
  const int l31 = lane & 31;
  const int lk = lane >> 5;  // 0/1: m within the MFMA K=2 step

  const int64_t n0 = static_cast<int64_t>(blockIdx.x) * BN;
  const int64_t k0 = static_cast<int64_t>(blockIdx.y) * BK;
  const int64_t m_begin = static_cast<int64_t>(blockIdx.z) * chunk;
  const int64_t m_end = min(M, m_begin + chunk);

  f32x16 acc1[FN][FK] = {};
  f32x16 acc2[FN][FK] = {};

  // staging geometry: one 32-row tile, B/8 consecutive floats per thread
  const int st_r = tid >> 3;  // 0..31

  const bool nk_full = (n0 + BN <= N) && (k0 + BK <= K);

  for (int64_t m0 = m_begin; m0 < m_end; m0 += WB_M) {
    const bool full = nk_full && (m0 + WB_M <= m_end);
    if (full) {
      {
        const int c0 = (tid & 7) * (BN / 8);
#pragma unroll
        for (int q = 0; q < BN / 8; ++q)
          g_tl[st_r][c0 + q] = g[(m0 + st_r) * N + n0 + c0 + q];
      }
      const int c0 = (tid & 7) * (BK / 8);
#pragma unroll
      for (int q = 0; q < BK / 8; ++q)
        x1_tl[st_r][c0 + q] = x1[(m0 + st_r) * K + k0 + c0 + q];
      if (DUAL) {
#pragma unroll
        for (int q = 0; q < BK / 8; ++q)
          x2_tl[st_r][c0 + q] = x2[(m0 + st_r) * K + k0 + c0 + q];
      }
    } else {
      const bool mrow = m0 + st_r < m_end;
      {
        const int c0 = (tid & 7) * (BN / 8);
#pragma unroll
        for (int q = 0; q < BN / 8; ++q) {
          const int64_t nn = n0 + c0 + q;
          g_tl[st_r][c0 + q] =
              (mrow && nn < N) ? g[(m0 + st_r) * N + nn] : 0.f;
        }
      }
      const int c0 = (tid & 7) * (BK / 8);
#pragma unroll
      for (int q = 0; q < BK / 8; ++q) {
        const int64_t kk = k0 + c0 + q;
        const bool kin = mrow && kk < K;
        x1_tl[st_r][c0 + q] = kin ? x1[(m0 + st_r) * K + kk] : 0.f;
        if (DUAL)
          x2_tl[st_r][c0 + q] = kin ? x2[(m0 + st_r) * K + kk] : 0.f;
      }
    }
    __syncthreads();
#pragma unroll
    for (int mm = 0; mm < WB_M; mm += 2) {
      float av[FN], b1v[FK], b2v[FK];
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
        av[fn] = g_tl[mm + lk][wr * (BN / 2) + fn * 32 + l31];
#pragma unroll
      for (int fk = 0; fk < FK; ++fk) {
        b1v[fk] = x1_tl[mm + lk][wc * (BK / 2) + fk * 32 + l31];
        if (DUAL) b2v[fk] = x2_tl[mm + lk][wc * (BK / 2) + fk * 32 + l31];
      }
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
#pragma unroll
        for (int fk = 0; fk < FK; ++fk) {
          acc1[fn][fk] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              av[fn], b1v[fk], acc1[fn][fk], 0, 0, 0);
          if (DUAL)
            acc2[fn][fk] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                av[fn], b2v[fk], acc2[fn][fk], 0, 0, 0);
        }
    }
    __syncthreads();
  }

  // epilogue: C layout for 32x32 shapes (same as dual_gemm.hip):
  // col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  float* w1p = ws1 + static_cast<int64_t>(blockIdx.z) * N * K;
  float* w2p = DUAL ? ws2 + static_cast<int64_t>(blockIdx.z) * N * K
                    : nullptr;
#pragma unroll
  for (int fn = 0; fn < FN; ++fn)
#pragma unroll
    for (int fk = 0; fk < FK; ++fk) {
      const int64_t col = k0 + wc * (BK / 2) + fk * 32 + l31;
      if (col >= K) continue;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int64_t row = n0 + wr * (BN / 2) + fn * 32 + (reg & 3) +
                            8 * (reg >> 2) + 4 * lk;
        if (row < N) {
          w1p[row * K + col] = acc1[fn][fk][reg];
          if (DUAL) w2p[row * K + col] = acc2[fn][fk][reg];
        }
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

  // 64x64 tiles measured BEST everywhere: the 128x128 variant halves
  // HBM re-reads but its 128 accumulator VGPRs per wave drop occupancy
  // and LOSE (66->57 TF at [233k]x[256,602], 111->100 TF at K=256) —
  // the kernel is latency-, not bandwidth-, bound at these shapes.
  // Template retained for the record; env knob for re-measurement.
  // PIPEGCN_WGRAD_CFG: 0 (default) 64x64, 1 -> 128x128, 2 -> 64x128
  const char* cfg_s = std::getenv("PIPEGCN_WGRAD_CFG");
  const int cfg = cfg_s ? std::atoi(cfg_s) : 0;
  const int BN = cfg == 1 ? 128 : 64;
  const int BK = cfg >= 1 ? 128 : 64;
  const int64_t tiles = ((N + BN - 1) / BN) * ((K + BK - 1) / BK);
  // enough blocks to fill 256 CUs several times over, M chunks 32-aligned
  int64_t S = std::min<int64_t>((1024 + tiles - 1) / tiles, 64);
  S = std::min<int64_t>(S, (M + 31) / 32);
  S = std::max<int64_t>(S, 1);
  const int64_t chunk = ((M + S - 1) / S + 31) / 32 * 32;

  auto opt = g.options();
  auto ws1 = torch::empty({S, N, K}, opt);
  auto ws2 = dual ? torch::empty({S, N, K}, opt) : torch::Tensor();
  dim3 grid((N + BN - 1) / BN, (K + BK - 1) / BK, S);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  float* gp = g.data_ptr<float>();
  float* x1p = x1.data_ptr<float>();
  float* x2p = dual ? x2.data_ptr<float>() : nullptr;
  float* w1p = ws1.data_ptr<float>();
  float* w2p = dual ? ws2.data_ptr<float>() : nullptr;
#define LAUNCH_WGRAD(D, BNv, BKv)                                        \
  hipLaunchKernelGGL(HIP_KERNEL_NAME(dual_wgrad_kernel<D, BNv, BKv>),    \
                     grid, dim3(256), 0, stream, gp, x1p, x2p, w1p, w2p, \
                     M, N, K, chunk)
  if (dual) {
    if (cfg == 1)
      LAUNCH_WGRAD(true, 128, 128);
    else if (cfg == 2)
      LAUNCH_WGRAD(true, 64, 128);
    else
      LAUNCH_WGRAD(true, 64, 64);
  } else {
    if (cfg == 1)
      LAUNCH_WGRAD(false, 128, 128);
    else if (cfg == 2)
      LAUNCH_WGRAD(false, 64, 128);
    else
      LAUNCH_WGRAD(false, 64, 64);
  }
#undef LAUNCH_WGRAD
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
