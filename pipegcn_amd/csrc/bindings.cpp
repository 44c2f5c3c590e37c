// pybind bindings + device dispatch for pipegcn_amd._C.
//
// On device tensors every op goes to the hand-written gfx950 HIP kernels in
// hip/kernels.hip — there is NO silent eager fallback on GPU: if this
// extension is missing on a GPU box, the Python ops raise (see
// pipegcn_amd/ops/__init__.py). CPU tensors use the native C++ paths in
// graph_core.cpp (used by the CPU test tier and rank-0 full-graph eval).

#include "common.h"

static torch::Tensor spmm(torch::Tensor indptr, torch::Tensor indices,
                          torch::Tensor feat, torch::Tensor dst_scale,
                          torch::Tensor src_scale, torch::Tensor row_order,
                          int64_t num_rows) {
  if (feat.is_cuda()) {
    auto out = torch::empty({num_rows, feat.size(1)}, feat.options());
    spmm_csr_hip(indptr, indices, feat, dst_scale, src_scale, row_order,
                 out);
    return out;
  }
  if (feat.scalar_type() == torch::kBFloat16) {
    // CPU tier computes in fp32 (the CPU path serves tests + rank-0 eval)
    return spmm_cpu(indptr, indices, feat.to(torch::kFloat), dst_scale,
                    src_scale, num_rows)
        .to(torch::kBFloat16);
  }
  return spmm_cpu(indptr, indices, feat, dst_scale, src_scale, num_rows);
}

static torch::Tensor gather_rows(torch::Tensor src, torch::Tensor idx) {
  if (src.is_cuda()) {
    auto out = torch::empty({idx.numel(), src.size(1)}, src.options());
    gather_rows_hip(src, idx, out);
    return out;
  }
  return src.index_select(0, idx);
}

static void gather_rows_out(torch::Tensor src, torch::Tensor idx,
                            torch::Tensor out) {
  if (src.is_cuda()) {
    gather_rows_hip(src, idx, out);
  } else {
    torch::index_select_out(out, src, 0, idx);
  }
}

static void scatter_add_rows(torch::Tensor dst, torch::Tensor idx,
                             torch::Tensor src) {
  if (dst.is_cuda()) {
    scatter_add_rows_hip(dst, idx, src);
  } else {
    dst.index_add_(0, idx, src);
  }
}

static torch::Tensor sage_dual_gemm(torch::Tensor x1, torch::Tensor x2,
                                    torch::Tensor w1, torch::Tensor w2,
                                    torch::Tensor bias) {
  if (x1.is_cuda()) {
    auto out = torch::empty({x1.size(0), w1.size(0)}, x1.options());
    sage_dual_gemm_hip(x1, x2, w1, w2, bias, out);
    return out;
  }
  auto out = torch::mm(x1, w1.t());
  out.addmm_(x2, w2.t());
  if (bias.defined() && bias.numel() > 0) out.add_(bias);
  return out;
}

static torch::Tensor colsum(torch::Tensor x) {
  if (x.is_cuda()) {
    if (x.scalar_type() == torch::kBFloat16)
      return colsum_hip(x.to(torch::kFloat)).to(torch::kBFloat16);
    return colsum_hip(x);
  }
  return x.sum(0);
}

// GPU-only fused ops (the Python wrappers fall back to eager torch on CPU)

static std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p,
                                              int64_t seed) {
  TORCH_CHECK(x.is_cuda(), "fused dropout is the GPU path");
  auto y = torch::empty_like(x);
  auto mask = torch::empty({(x.numel() + 7) / 8},
                           x.options().dtype(torch::kUInt8));
  dropout_fwd_hip(x, y, mask, p, seed);
  return {y, mask};
}

static torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask,
                                 double p) {
  auto dx = torch::empty_like(dy);
  dropout_bwd_hip(dy.contiguous(), mask, dx, p);
  return dx;
}

static std::vector<torch::Tensor> layer_norm_relu_fwd(torch::Tensor x,
                                                      torch::Tensor w,
                                                      torch::Tensor b,
                                                      double eps, bool relu) {
  TORCH_CHECK(x.is_cuda(), "fused layer_norm_relu is the GPU path");
  auto y = torch::empty_like(x);
  auto xhat = torch::empty_like(x);
  auto rstd = torch::empty({x.size(0)}, x.options().dtype(torch::kFloat));
  layer_norm_relu_fwd_hip(x.contiguous(), w, b, eps, relu, y, xhat, rstd);
  return {y, xhat, rstd};
}

static std::vector<torch::Tensor> layer_norm_relu_bwd(
    torch::Tensor dy, torch::Tensor xhat, torch::Tensor rstd,
    torch::Tensor w, torch::Tensor b, bool relu) {
  auto dx = torch::empty_like(dy);
  const int64_t N = dy.size(0);
  const int64_t F = dy.size(1);
  const int64_t nblk = std::min<int64_t>(1024, (N + 3) / 4);
  auto dw_part = torch::empty({nblk * 4, F},
                              dy.options().dtype(torch::kFloat));
  auto db_part = torch::empty_like(dw_part);
  layer_norm_relu_bwd_hip(dy.contiguous(), xhat, rstd, w, b, relu, dx,
                          dw_part, db_part);
  return {dx, colsum_hip(dw_part), colsum_hip(db_part)};
}

static void ema_update(torch::Tensor avg, torch::Tensor x, double momentum) {
  if (avg.is_cuda()) {
    ema_update_hip(avg, x, momentum);
  } else {
    avg.mul_(momentum).add_(x, 1.0 - momentum);
  }
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "pipegcn_amd native core (gfx950 HIP kernels + host graph ops)";
  m.def("build_csr", &build_csr, "COO -> CSR (counting sort by dst)");
  m.def("partition_graph", &partition_graph_cpu,
        "BFS-grown + refined k-way partitioner (cut/vol objective)");
  m.def("spmm", &spmm, "CSR SpMM with fused scale epilogue (fwd & transpose)");
  m.def("gather_rows", &gather_rows, "out[i,:] = src[idx[i],:]");
  m.def("gather_rows_out", &gather_rows_out,
        "gather into a preallocated out buffer");
  m.def("scatter_add_rows", &scatter_add_rows, "dst[idx[i],:] += src[i,:]");
  m.def("ema_update", &ema_update, "avg = m*avg + (1-m)*x");
  m.def("colsum", &colsum, "out[n] = sum_m x[m,n]");
  m.def("sage_dual_gemm", &sage_dual_gemm,
        "out = x1 @ w1^T + x2 @ w2^T + bias (MFMA fp32, fused)");
  m.def("dual_dgrad", &dual_dgrad_hip,
        "gx1 = g w1, gx2 = g w2 fused MFMA dgrad");
  m.def("dual_wgrad", &dual_wgrad_hip,
        "gw1 = g^T x1 [, gw2 = g^T x2] fused MFMA split-M wgrad");
  m.def("dropout_fwd", &dropout_fwd,
        "fused dropout, bitpacked mask -> (y, mask)");
  m.def("dropout_bwd", &dropout_bwd, "dx = mask ? dy/(1-p) : 0");
  m.def("layer_norm_relu_fwd", &layer_norm_relu_fwd,
        "fused LayerNorm[+ReLU] -> (y, xhat, rstd)");
  m.def("layer_norm_relu_bwd", &layer_norm_relu_bwd,
        "-> (dx, dweight, dbias)");
  m.attr("with_hip") = true;
}
