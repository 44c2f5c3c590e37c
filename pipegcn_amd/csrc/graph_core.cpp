// Host-side graph core: CSR construction, CPU SpMM, and the multi-way graph
// partitioner.
//
// This replaces the reference's external native layers:
//  - DGL's C++ COO->CSR graph machinery (used via dgl.graph/heterograph at
//    /root/reference/train.py:206-229)
//  - METIS k-way partitioning invoked through
//    dgl.distributed.partition_graph (/root/reference/helper/utils.py:143).
// The partitioner here is our own implementation (METIS is not available in
// this environment): BFS-grown balanced regions + boundary (KL/FM-style)
// refinement with an edge-cut or communication-volume objective, matching the
// reference's `--partition-obj {cut,vol}` surface.

#include "common.h"

#include <ATen/Parallel.h>

#include <algorithm>
#include <deque>
#include <random>

namespace {

template <typename T>
inline const T* data_of(const torch::Tensor& t) {
  return t.data_ptr<T>();
}

}  // namespace

std::vector<torch::Tensor> build_csr(torch::Tensor u, torch::Tensor v,
                                     int64_t num_rows) {
  TORCH_CHECK(u.device().is_cpu() && v.device().is_cpu(),
              "build_csr expects CPU tensors");
  TORCH_CHECK(u.numel() == v.numel(), "u/v length mismatch");
  u = u.contiguous().to(torch::kLong);
  v = v.contiguous().to(torch::kLong);
  const int64_t E = u.numel();
  auto indptr = torch::zeros({num_rows + 1}, torch::kLong);
  auto indices = torch::empty({E}, torch::kInt);
  const int64_t* up = data_of<int64_t>(u);
  const int64_t* vp = data_of<int64_t>(v);
  int64_t* ip = indptr.data_ptr<int64_t>();
  int32_t* xp = indices.data_ptr<int32_t>();

  // counting sort by destination row
  for (int64_t e = 0; e < E; ++e) {
    TORCH_CHECK(vp[e] >= 0 && vp[e] < num_rows, "dst out of range");
    ip[vp[e] + 1]++;
  }
  for (int64_t r = 0; r < num_rows; ++r) ip[r + 1] += ip[r];
  std::vector<int64_t> cursor(ip, ip + num_rows);
  for (int64_t e = 0; e < E; ++e) {
    xp[cursor[vp[e]]++] = static_cast<int32_t>(up[e]);
  }
  return {indptr, indices};
}

torch::Tensor spmm_cpu(torch::Tensor indptr, torch::Tensor indices,
                       torch::Tensor feat, torch::Tensor dst_scale,
                       torch::Tensor src_scale, int64_t num_rows) {
  TORCH_CHECK(feat.dim() == 2 && feat.scalar_type() == torch::kFloat,
              "spmm_cpu: feat must be fp32 [N,F]");
  feat = feat.contiguous();
  const int64_t F = feat.size(1);
  auto out = torch::zeros({num_rows, F}, feat.options());
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int32_t* xp = indices.data_ptr<int32_t>();
  const float* fp = feat.data_ptr<float>();
  float* op = out.data_ptr<float>();
  const bool has_scale = dst_scale.defined() && dst_scale.numel() > 0;
  auto dsc = has_scale ? dst_scale.contiguous() : dst_scale;
  const float* sp = has_scale ? dsc.data_ptr<float>() : nullptr;
  const bool has_src = src_scale.defined() && src_scale.numel() > 0;
  auto ssc = has_src ? src_scale.contiguous() : src_scale;
  const float* ssp = has_src ? ssc.data_ptr<float>() : nullptr;

  at::parallel_for(0, num_rows, 64, [&](int64_t begin, int64_t end) {
    for (int64_t r = begin; r < end; ++r) {
      float* orow = op + r * F;
      for (int64_t e = ip[r]; e < ip[r + 1]; ++e) {
        const float* frow = fp + static_cast<int64_t>(xp[e]) * F;
        const float sv = ssp ? ssp[xp[e]] : 1.f;
        for (int64_t k = 0; k < F; ++k) orow[k] += sv * frow[k];
      }
      if (has_scale) {
        const float s = sp[r];
        for (int64_t k = 0; k < F; ++k) orow[k] *= s;
      }
    }
  });
  return out;
}

// The multilevel k-way partitioner lives in partitioner.cpp.
