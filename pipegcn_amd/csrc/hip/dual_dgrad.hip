// Fused dual data-gradient GEMM for the GraphSAGE layer backward
// (gfx950 MFMA, exact fp32):
//
//     gx1[M,K] = g[M,N] @ w1[N,K]        gx2[M,K] = g @ w2[N,K]
//
// The two dgrads of a dual-linear layer share the output gradient g
// (reference site: the backward of `linear1(x1) + linear2(x2)`,
// /root/reference/module/layer.py:51). One kernel stages each g tile
// into LDS once and contracts it against BOTH weight tiles — vs two
// library GEMMs (or one GEMM against [w1 ‖ w2], which still streams the
// concatenated B and writes an interleaved output needing strided
// views). Completes the hand-written MFMA dense path: forward dual-GEMM
// (dual_gemm.hip), wgrad pair (wgrad.hip), dgrad pair (here).
//
// v_mfma_f32_32x32x2_f32 (f32 in / f32 accumulate — exact fmaf chain, no
// TF32). Block: 256 threads, 128(M) x 64(K) output tile, 4 waves in a
// 2x2 quadrant layout, each wave 2 M-fragments per product (64 acc
// VGPRs); the N reduction staged through LDS 32 rows at a time.

#include "../common.h"

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

namespace {

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

constexpr int DG_M = 128;  // output rows per block
constexpr int DG_K = 64;   // output cols per block
constexpr int DG_N = 32;   // reduction rows staged per iteration

__global__ __launch_bounds__(256) void dual_dgrad_kernel(
    const float* __restrict__ g, const float* __restrict__ w1,
    const float* __restrict__ w2, float* __restrict__ gx1,
    float* __restrict__ gx2, int64_t M, int64_t N, int64_t K) {
  __shared__ float a_tl[DG_M][DG_N + 1];
  __shared__ float b1_tl[DG_N][DG_K + 1];
  __shared__ float b2_tl[DG_N][DG_K + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // 0/1: which 64-row band of the M tile
  const int wc = wave & 1;   // 0/1: which 32-col band of the K tile
  const int l31 = lane & 31;
  const int lk = lane >> 5;  // 0/1: n within the MFMA K=2 step

  const int64_t m0 = static_cast<int64_t>(blockIdx.x) * DG_M;
  const int64_t k0 = static_cast<int64_t>(blockIdx.y) * DG_K;
  const bool interior = (m0 + DG_M <= M) && (k0 + DG_K <= K);

  f32x16 acc1[2] = {};
  f32x16 acc2[2] = {};

  // staging geometry: A 128x32 -> 16 floats/thread (half a row slice);
  // B 32x64 -> 8 floats/thread
  const int a_r = tid >> 1;            // 0..127
  const int a_c = (tid & 1) * 16;      // 0 / 16
  const int b_r = tid >> 3;            // 0..31
  const int b_c = (tid & 7) * 8;       // 0..56

  for (int64_t n0 = 0; n0 < N; n0 += DG_N) {
    if (interior && n0 + DG_N <= N) {
#pragma unroll
      for (int q = 0; q < 16; ++q)
        a_tl[a_r][a_c + q] = g[(m0 + a_r) * N + n0 + a_c + q];
#pragma unroll
      for (int q = 0; q < 8; ++q)
        b1_tl[b_r][b_c + q] = w1[(n0 + b_r) * K + k0 + b_c + q];
#pragma unroll
      for (int q = 0; q < 8; ++q)
        b2_tl[b_r][b_c + q] = w2[(n0 + b_r) * K + k0 + b_c + q];
    } else {
      const bool arow = m0 + a_r < M;
#pragma unroll
      for (int q = 0; q < 16; ++q) {
        const int64_t nn = n0 + a_c + q;
        a_tl[a_r][a_c + q] =
            (arow && nn < N) ? g[(m0 + a_r) * N + nn] : 0.f;
      }
      const bool brow = n0 + b_r < N;
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        const int64_t kk = k0 + b_c + q;
        const bool in = brow && kk < K;
        b1_tl[b_r][b_c + q] = in ? w1[(n0 + b_r) * K + kk] : 0.f;
        b2_tl[b_r][b_c + q] = in ? w2[(n0 + b_r) * K + kk] : 0.f;
      }
    }
    __syncthreads();
#pragma unroll
    for (int nn = 0; nn < DG_N; nn += 2) {
      const float a0 = a_tl[wr * 64 + l31][nn + lk];
      const float a1 = a_tl[wr * 64 + 32 + l31][nn + lk];
      const float b1 = b1_tl[nn + lk][wc * 32 + l31];
      const float b2 = b2_tl[nn + lk][wc * 32 + l31];
      acc1[0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc1[0],
                                                     0, 0, 0);
      acc1[1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc1[1],
                                                     0, 0, 0);
      acc2[0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b2, acc2[0],
                                                     0, 0, 0);
      acc2[1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b2, acc2[1],
                                                     0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: C layout for 32x32 shapes (same as dual_gemm.hip):
  // col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const int64_t col = k0 + wc * 32 + l31;
  if (col >= K) return;
#pragma unroll
  for (int fi = 0; fi < 2; ++fi)
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int64_t row = m0 + wr * 64 + fi * 32 + (reg & 3) +
                          8 * (reg >> 2) + 4 * lk;
      if (row < M) {
        gx1[row * K + col] = acc1[fi][reg];
        gx2[row * K + col] = acc2[fi][reg];
      }
    }
}

}  // namespace

std::vector<torch::Tensor> dual_dgrad_hip(torch::Tensor g, torch::Tensor w1,
                                          torch::Tensor w2) {
  TORCH_CHECK(g.is_cuda() && g.scalar_type() == torch::kFloat,
              "dual_dgrad: fp32 CUDA only");
  TORCH_CHECK(g.is_contiguous() && w1.is_contiguous() && w2.is_contiguous());
  const int64_t M = g.size(0);
  const int64_t N = g.size(1);
  const int64_t K = w1.size(1);
  TORCH_CHECK(w1.size(0) == N && w2.size(0) == N && w2.size(1) == K);

  auto gx1 = torch::empty({M, K}, g.options());
  auto gx2 = torch::empty({M, K}, g.options());
  dim3 grid((M + DG_M - 1) / DG_M, (K + DG_K - 1) / DG_K);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  hipLaunchKernelGGL(dual_dgrad_kernel, grid, dim3(256), 0, stream,
                     g.data_ptr<float>(), w1.data_ptr<float>(),
                     w2.data_ptr<float>(), gx1.data_ptr<float>(),
                     gx2.data_ptr<float>(), M, N, K);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));
  return {gx1, gx2};
}
