// Fused dual GEMM for the GraphSAGE layer forward (gfx950 MFMA, exact fp32):
//
//     out[M,N] = x1[M,K] @ w1^T + x2[M,K] @ w2^T + bias[N]
//
// replaces the reference's two cuBLAS GEMMs + elementwise add
// (`linear1(feat[:num_dst]) + linear2(ah)`, /root/reference/module/layer.py:51)
// with ONE kernel: the two products share the accumulator (mathematically a
// single GEMM over a 2K-long concatenated K axis, without materializing the
// concat), and the bias add is fused into the epilogue — one output pass
// instead of three.
//
// Uses v_mfma_f32_32x32x2_f32 (f32 in / f32 accumulate — bitwise an fmaf
// chain, NO TF32 anywhere; see cdna_hip_programming.md §3): 155 TF chip peak.
// Structure: 128x128 block tile, 4 waves (2x2), each wave a 64x64 tile of
// 2x2 32x32 fragments; K staged through LDS in TBK-deep tiles, +1-dword row
// padding for conflict-free ds_reads; interior blocks (the vast majority at
// M ~ 10^5..10^6 rows) take a guard-free staging fast path.
//
// Weights are in torch Linear layout [N, K] row-major; they are tiny
// (<= 616 KB) and L2-resident, so the transposed LDS fill reads them
// scattered without penalty.

#include "../common.h"

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

#include <cstdlib>

namespace {

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int PAD = 1;

template <int TBK>
__global__ __launch_bounds__(256) void sage_dual_gemm_kernel(
    const float* __restrict__ x1, const float* __restrict__ x2,
    const float* __restrict__ w1, const float* __restrict__ w2,
    const float* __restrict__ bias, float* __restrict__ out, int64_t M,
    int64_t N, int64_t K) {
  __shared__ float a_lds[BM][TBK + PAD];
  __shared__ float b_lds[TBK][BN + 4];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;       // 0..3
  const int wr = wave >> 1;        // wave row 0..1  (64 rows each)
  const int wc = wave & 1;         // wave col 0..1  (64 cols each)

  const int64_t m0 = static_cast<int64_t>(blockIdx.x) * BM;
  const int64_t n0 = static_cast<int64_t>(blockIdx.y) * BN;
  const bool interior = (m0 + BM <= M) && (n0 + BN <= N);
  // float4 loads need 16B alignment: base ptrs are 16B-aligned (torch
  // allocator) but row offsets row*K are only 4B-aligned when K%4!=0
  // (K=602 for reddit) — use 4 scalar dword loads then (still guard-free)
  const bool k16 = (K & 3) == 0;

  f32x16 acc[2][2] = {};

  const int l31 = lane & 31;
  const int lk = lane >> 5;  // 0/1 : k within the MFMA K=2 step

  // staging geometry: A: thread covers (tid>>per) rows x elems
  const int a_r = tid / (TBK / 4);          // A row per float4 (TBK/4 thr/row)
  const int a_c = (tid % (TBK / 4)) * 4;    // col within tile
  constexpr int A_RSTEP = 256 / (TBK / 4);  // rows covered per pass
  const int b_n = tid >> 1;                 // 0..127 (W row)
  const int b_k = (tid & 1) * (TBK / 2);    // k base

  for (int seg = 0; seg < 2; ++seg) {
    const float* X = seg == 0 ? x1 : x2;
    const float* W = seg == 0 ? w1 : w2;
    for (int64_t k0 = 0; k0 < K; k0 += TBK) {
      const bool kfull = (k0 + TBK <= K);
      // ---- stage A tile [BM][TBK]
      if (interior && kfull) {
#pragma unroll
        for (int rr = 0; rr < BM / A_RSTEP; ++rr) {
          const int row = a_r + rr * A_RSTEP;
          const float* p = X + (m0 + row) * K + k0 + a_c;
          float4 val;
          if (k16)
            val = *reinterpret_cast<const float4*>(p);
          else
            val = make_float4(p[0], p[1], p[2], p[3]);
          a_lds[row][a_c + 0] = val.x;
          a_lds[row][a_c + 1] = val.y;
          a_lds[row][a_c + 2] = val.z;
          a_lds[row][a_c + 3] = val.w;
        }
      } else {
#pragma unroll
        for (int rr = 0; rr < BM / A_RSTEP; ++rr) {
          const int row = a_r + rr * A_RSTEP;
          const int64_t gm = m0 + row;
          float tmp[4] = {0.f, 0.f, 0.f, 0.f};
          if (gm < M) {
            for (int q = 0; q < 4 && k0 + a_c + q < K; ++q)
              tmp[q] = X[gm * K + k0 + a_c + q];
          }
          a_lds[row][a_c + 0] = tmp[0];
          a_lds[row][a_c + 1] = tmp[1];
          a_lds[row][a_c + 2] = tmp[2];
          a_lds[row][a_c + 3] = tmp[3];
        }
      }
      // ---- stage B tile [TBK][BN] = W[n0:n0+BN][k0:k0+TBK] transposed
      if (interior && kfull) {
#pragma unroll
        for (int q = 0; q < TBK / 2; q += 4) {
          const float* p = W + (n0 + b_n) * K + k0 + b_k + q;
          float4 val;
          if (k16)
            val = *reinterpret_cast<const float4*>(p);
          else
            val = make_float4(p[0], p[1], p[2], p[3]);
          b_lds[b_k + q + 0][b_n] = val.x;
          b_lds[b_k + q + 1][b_n] = val.y;
          b_lds[b_k + q + 2][b_n] = val.z;
          b_lds[b_k + q + 3][b_n] = val.w;
        }
      } else {
        const int64_t gn = n0 + b_n;
#pragma unroll
        for (int q = 0; q < TBK / 2; q += 4) {
          float tmp[4] = {0.f, 0.f, 0.f, 0.f};
          if (gn < N) {
            for (int p = 0; p < 4 && k0 + b_k + q + p < K; ++p)
              tmp[p] = W[gn * K + k0 + b_k + q + p];
          }
          b_lds[b_k + q + 0][b_n] = tmp[0];
          b_lds[b_k + q + 1][b_n] = tmp[1];
          b_lds[b_k + q + 2][b_n] = tmp[2];
          b_lds[b_k + q + 3][b_n] = tmp[3];
        }
      }
      __syncthreads();

      // ---- MFMA inner loop: TBK/2 K-steps of 2
#pragma unroll
      for (int kk = 0; kk < TBK; kk += 2) {
        float a0 = a_lds[wr * 64 + l31][kk + lk];
        float a1 = a_lds[wr * 64 + 32 + l31][kk + lk];
        float b0 = b_lds[kk + lk][wc * 64 + l31];
        float b1 = b_lds[kk + lk][wc * 64 + 32 + l31];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0],
                                                         0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1],
                                                         0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0],
                                                         0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1],
                                                         0, 0, 0);
      }
      __syncthreads();
    }
  }

  // ---- epilogue: C/D layout for 32x32 shapes (shape-determined):
  // col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
      const int64_t col = n0 + wc * 64 + fj * 32 + l31;
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row_in_frag = (reg & 3) + 8 * (reg >> 2) + 4 * lk;
        const int64_t row = m0 + wr * 64 + fi * 32 + row_in_frag;
        if (row < M) out[row * N + col] = acc[fi][fj][reg] + bv;
      }
    }
  }
}


// Double-buffered variant (PIPEGCN_GEMM_DB=1): the single-buffer kernel
// serializes {global loads -> LDS, barrier, MFMA} per K-tile; here the
// NEXT tile's global loads issue into registers BEFORE the current
// tile's MFMAs, so VMEM latency hides under compute, and one barrier per
// tile replaces two. TBK=32 per buffer keeps total LDS at 67.6 KB =
// same 2-blocks/CU occupancy as the TBK=64 single-buffer config.
__global__ __launch_bounds__(256) void sage_dual_gemm_db_kernel(
    const float* __restrict__ x1, const float* __restrict__ x2,
    const float* __restrict__ w1, const float* __restrict__ w2,
    const float* __restrict__ bias, float* __restrict__ out, int64_t M,
    int64_t N, int64_t K) {
  constexpr int TBK = 32;
  __shared__ float a_lds[2][BM][TBK + PAD];
  __shared__ float b_lds[2][TBK][BN + 4];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  const int64_t m0 = static_cast<int64_t>(blockIdx.x) * BM;
  const int64_t n0 = static_cast<int64_t>(blockIdx.y) * BN;
  const bool interior = (m0 + BM <= M) && (n0 + BN <= N);
  const bool k16 = (K & 3) == 0;

  f32x16 acc[2][2] = {};

  const int l31 = lane & 31;
  const int lk = lane >> 5;

  // staging geometry (TBK=32): A: 8 thr/row, 4 passes of 32 rows,
  // 4 floats each; B: 2 thr/row over BN=128 W rows, 16 k's each
  const int a_r = tid >> 3;
  const int a_c = (tid & 7) * 4;
  const int b_n = tid >> 1;
  const int b_k = (tid & 1) * 16;

  const int nk = (int)((K + TBK - 1) / TBK);
  const int ntiles = 2 * nk;

  float4 areg[4];
  float4 breg[4];

  auto load_tile = [&](int t) {
    const int seg = t / nk;
    const int64_t k0 = (int64_t)(t % nk) * TBK;
    const float* X = seg == 0 ? x1 : x2;
    const float* W = seg == 0 ? w1 : w2;
    const bool kfull = (k0 + TBK <= K);
    if (interior && kfull) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const float* p = X + (m0 + a_r + rr * 32) * K + k0 + a_c;
        areg[rr] = k16 ? *reinterpret_cast<const float4*>(p)
                       : make_float4(p[0], p[1], p[2], p[3]);
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const float* p = W + (n0 + b_n) * K + k0 + b_k + q * 4;
        breg[q] = k16 ? *reinterpret_cast<const float4*>(p)
                      : make_float4(p[0], p[1], p[2], p[3]);
      }
    } else {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int64_t gm = m0 + a_r + rr * 32;
        float t4[4] = {0.f, 0.f, 0.f, 0.f};
        if (gm < M)
          for (int q = 0; q < 4 && k0 + a_c + q < K; ++q)
            t4[q] = X[gm * K + k0 + a_c + q];
        areg[rr] = make_float4(t4[0], t4[1], t4[2], t4[3]);
      }
      const int64_t gn = n0 + b_n;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        float t4[4] = {0.f, 0.f, 0.f, 0.f};
        if (gn < N)
          for (int p = 0; p < 4 && k0 + b_k + q * 4 + p < K; ++p)
            t4[p] = W[gn * K + k0 + b_k + q * 4 + p];
        breg[q] = make_float4(t4[0], t4[1], t4[2], t4[3]);
      }
    }
  };

  auto store_tile = [&](int buf) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      a_lds[buf][a_r + rr * 32][a_c + 0] = areg[rr].x;
      a_lds[buf][a_r + rr * 32][a_c + 1] = areg[rr].y;
      a_lds[buf][a_r + rr * 32][a_c + 2] = areg[rr].z;
      a_lds[buf][a_r + rr * 32][a_c + 3] = areg[rr].w;
    }
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      b_lds[buf][b_k + q * 4 + 0][b_n] = breg[q].x;
      b_lds[buf][b_k + q * 4 + 1][b_n] = breg[q].y;
      b_lds[buf][b_k + q * 4 + 2][b_n] = breg[q].z;
      b_lds[buf][b_k + q * 4 + 3][b_n] = breg[q].w;
    }
  };

  load_tile(0);
  store_tile(0);
  __syncthreads();
  for (int t = 0; t < ntiles; ++t) {
    const int cur = t & 1;
    if (t + 1 < ntiles) load_tile(t + 1);  // issue next tile's VMEM now
#pragma unroll
    for (int kk = 0; kk < TBK; kk += 2) {
      const float a0 = a_lds[cur][wr * 64 + l31][kk + lk];
      const float a1 = a_lds[cur][wr * 64 + 32 + l31][kk + lk];
      const float b0 = b_lds[cur][kk + lk][wc * 64 + l31];
      const float b1 = b_lds[cur][kk + lk][wc * 64 + 32 + l31];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0],
                                                       0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1],
                                                       0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0],
                                                       0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1],
                                                       0, 0, 0);
    }
    if (t + 1 < ntiles) store_tile(cur ^ 1);  // waits the loads, fills LDS
    __syncthreads();
  }

  // epilogue identical to the single-buffer kernel
#pragma unroll
  for (int fi = 0; fi < 2; ++fi) {
#pragma unroll
    for (int fj = 0; fj < 2; ++fj) {
      const int64_t col = n0 + wc * 64 + fj * 32 + l31;
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row_in_frag = (reg & 3) + 8 * (reg >> 2) + 4 * lk;
        const int64_t row = m0 + wr * 64 + fi * 32 + row_in_frag;
        if (row < M) out[row * N + col] = acc[fi][fj][reg] + bv;
      }
    }
  }
}

}  // namespace

void sage_dual_gemm_hip(torch::Tensor x1, torch::Tensor x2, torch::Tensor w1,
                        torch::Tensor w2, torch::Tensor bias,
                        torch::Tensor out) {
  TORCH_CHECK(x1.is_cuda() && x2.is_cuda() && w1.is_cuda() && w2.is_cuda());
  TORCH_CHECK(x1.scalar_type() == torch::kFloat, "fp32 only");
  TORCH_CHECK(x1.is_contiguous() && x2.is_contiguous() &&
              w1.is_contiguous() && w2.is_contiguous() && out.is_contiguous());
  const int64_t M = x1.size(0);
  const int64_t K = x1.size(1);
  const int64_t N = w1.size(0);
  TORCH_CHECK(x2.size(0) == M && x2.size(1) == K);
  TORCH_CHECK(w1.size(1) == K && w2.size(0) == N && w2.size(1) == K);
  TORCH_CHECK(out.size(0) == M && out.size(1) == N);
  const float* bp = nullptr;
  if (bias.defined() && bias.numel() > 0) {
    TORCH_CHECK(bias.is_contiguous() && bias.numel() == N);
    bp = bias.data_ptr<float>();
  }
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
  int bk = K > 512 ? 64 : 32;  // measured: BK=64 wins at K=602, 32 at 256
  if (const char* e = std::getenv("PIPEGCN_GEMM_BK")) bk = std::atoi(e);
  // double-buffered staging is the DEFAULT (measured 90->103 TF at
  // K=602, 100->107 at K=256, bitwise-equal accumulation order);
  // PIPEGCN_GEMM_DB=0 falls back to the single-buffer kernel
  static const bool use_db = [] {
    const char* s = std::getenv("PIPEGCN_GEMM_DB");
    return !(s && s[0] == '0');
  }();
  if (use_db) {
    hipLaunchKernelGGL(sage_dual_gemm_db_kernel, grid, dim3(256), 0, stream,
                       x1.data_ptr<float>(), x2.data_ptr<float>(),
                       w1.data_ptr<float>(), w2.data_ptr<float>(), bp,
                       out.data_ptr<float>(), M, N, K);
    hipError_t e2 = hipGetLastError();
    TORCH_CHECK(e2 == hipSuccess, "HIP error: ", hipGetErrorString(e2));
    return;
  }
  if (bk == 32)
    hipLaunchKernelGGL(HIP_KERNEL_NAME(sage_dual_gemm_kernel<32>), grid,
                       dim3(256), 0, stream, x1.data_ptr<float>(),
                       x2.data_ptr<float>(), w1.data_ptr<float>(),
                       w2.data_ptr<float>(), bp, out.data_ptr<float>(), M, N,
                       K);
  else
    hipLaunchKernelGGL(HIP_KERNEL_NAME(sage_dual_gemm_kernel<64>), grid,
                       dim3(256), 0, stream, x1.data_ptr<float>(),
                       x2.data_ptr<float>(), w1.data_ptr<float>(),
                       w2.data_ptr<float>(), bp, out.data_ptr<float>(), M, N,
                       K);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));
}
