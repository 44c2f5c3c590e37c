// Common declarations shared between the host graph core and the HIP kernel
// launchers of pipegcn_amd._C.
//
// The native layer replaces the reference's external stacks (DGL C++/CUDA SpMM,
// METIS partitioning, Gloo staging; see /root/reference SURVEY §2.2) with
// MI355X-native C++/HIP code compiled for gfx950 only.
#pragma once

#include <torch/extension.h>

#include <cstdint>
#include <vector>

// ---------------------------------------------------------------------------
// Host-side graph core (graph_core.cpp)
// ---------------------------------------------------------------------------

// Build a CSR adjacency over destination rows from a COO edge list.
//   u, v : int64/int32 CPU tensors of equal length E (edge u -> v)
//   num_rows : number of destination nodes (rows)
// Returns (indptr int64[num_rows+1], indices int32[E]) where
// indices[indptr[r]:indptr[r+1]] are the sources of row r (unsorted within row).
std::vector<torch::Tensor> build_csr(torch::Tensor u, torch::Tensor v,
                                     int64_t num_rows);

// Multi-way graph partitioner (replaces METIS; BFS-grown balanced seeds +
// boundary refinement). Operates on an undirectedized CSR of the full graph.
//   indptr int64[N+1], indices int32[E] (CSR over rows = nodes)
//   nparts : number of partitions
//   objective : 0 = edge cut, 1 = communication volume ("vol")
//   balance_slack : allowed imbalance, e.g. 0.05
// Returns int32[N] partition assignment.
torch::Tensor partition_graph_cpu(torch::Tensor indptr, torch::Tensor indices,
                                  int64_t nparts, int64_t objective,
                                  double balance_slack, int64_t n_refine_passes,
                                  int64_t seed);

// CPU CSR SpMM: out[r, :] = scale[r] * sum_{c in row r} feat[indices[c], :]
// scale may be an undefined tensor (no scaling). feat float32 [num_src, F].
torch::Tensor spmm_cpu(torch::Tensor indptr, torch::Tensor indices,
                       torch::Tensor feat, torch::Tensor dst_scale,
                       torch::Tensor src_scale, int64_t num_rows);

// ---------------------------------------------------------------------------
// HIP kernel launchers (hip/kernels.hip) — gfx950 only.
// ---------------------------------------------------------------------------

// out[r,:] = scale[r] * sum_{e in [indptr[r], indptr[r+1])} feat[indices[e],:]
// All tensors on device. feat fp32 [num_src, F], out fp32 [num_rows, F].
void spmm_csr_hip(torch::Tensor indptr, torch::Tensor indices,
                  torch::Tensor feat, torch::Tensor dst_scale,
                  torch::Tensor src_scale, torch::Tensor row_order,
                  torch::Tensor out);

// out[i,:] = src[idx[i],:]
void gather_rows_hip(torch::Tensor src, torch::Tensor idx, torch::Tensor out);

// dst[idx[i],:] += src[i,:]   (idx entries unique — no atomics needed? they
// are unique for the boundary scatter; kernel uses one row per wave so
// duplicates would race — documented contract: unique indices.)
void scatter_add_rows_hip(torch::Tensor dst, torch::Tensor idx,
                          torch::Tensor src);

// avg = momentum * avg + (1 - momentum) * x
void ema_update_hip(torch::Tensor avg, torch::Tensor x, double momentum);

// out[n] = sum_m x[m, n] — fast column reduction for bias/BN grads.
torch::Tensor colsum_hip(torch::Tensor x);

// Fused dropout with a bitpacked mask (1 bit/elem): y = mask ? x/(1-p) : 0.
// mask uint8[(n+7)/8]; seed drives a counter-based RNG (reproducible).
void dropout_fwd_hip(torch::Tensor x, torch::Tensor y, torch::Tensor mask,
                     double p, int64_t seed);
void dropout_bwd_hip(torch::Tensor dy, torch::Tensor mask, torch::Tensor dx,
                     double p);

// Fused LayerNorm [+ReLU]: y = [relu](w * xhat + b), xhat/rstd saved for
// backward (wave-per-row, fp32 statistics, F <= 1024).
void layer_norm_relu_fwd_hip(torch::Tensor x, torch::Tensor w,
                             torch::Tensor b, double eps, bool relu,
                             torch::Tensor y, torch::Tensor xhat,
                             torch::Tensor rstd);
// dw_part/db_part: fp32 [nwaves, F] per-wave partials (column-sum on host).
// fused dual data gradient: {g @ w1, g @ w2} via MFMA
// (csrc/hip/dual_dgrad.hip)
std::vector<torch::Tensor> dual_dgrad_hip(torch::Tensor g, torch::Tensor w1,
                                          torch::Tensor w2);

// fused dual weight gradient: {g^T x1 [, g^T x2]} via MFMA split-M
// (csrc/hip/wgrad.hip); pass an undefined x2 for the single-wgrad case
std::vector<torch::Tensor> dual_wgrad_hip(torch::Tensor g, torch::Tensor x1,
                                          torch::Tensor x2);

void layer_norm_relu_bwd_hip(torch::Tensor dy, torch::Tensor xhat,
                             torch::Tensor rstd, torch::Tensor w,
                             torch::Tensor b, bool relu, torch::Tensor dx,
                             torch::Tensor dw_part, torch::Tensor db_part);

// Fused dual GEMM for the GraphSAGE layer epilogue:
//   out[M,N] = x1[M,K] @ w1t[K,N] + x2[M,K] @ w2t[K,N] + b[N]
// fp32, MFMA (v_mfma_f32_16x16x4_f32). Weights pre-transposed to [K,N].
void sage_dual_gemm_hip(torch::Tensor x1, torch::Tensor x2, torch::Tensor w1t,
                        torch::Tensor w2t, torch::Tensor bias,
                        torch::Tensor out);
