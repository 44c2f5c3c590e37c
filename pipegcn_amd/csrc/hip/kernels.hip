// Hand-written HIP/CDNA4 (gfx950) kernels for the pipegcn_amd hot path.
//
// These replace the reference's DGL CUDA gSpMM (copy_src+sum, invoked at
// /root/reference/module/layer.py:47-49), its index gather/scatter
// (/root/reference/helper/feature_buffer.py:176,215-217) and the EMA
// smoothing correction (/root/reference/helper/feature_buffer.py:189-191).
//
// Design (MI355X-first, see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; feature dim is mapped across lanes with float4/2/1
//    vector loads so each wave issues 1 KiB coalesced reads of a neighbor row.
//  - SpMM is a bandwidth/gather-bound op: one wave owns one (dst-row, feature
//    chunk); the edge loop is unrolled 4x to keep 4 gathers in flight per
//    lane, rows are walked in descending-degree order (LPT scheduling — see
//    pick_vec/profiles), and occupancy stays high (tiny VGPR footprint) so
//    TLP hides HBM latency. The degree-divide epilogue is fused (scale
//    argument); kernels are dtype-templated (fp32, bf16-with-fp32-accumulate).
//  - The backward (transpose) SpMM is the same kernel run over the CSC of the
//    halo graph, built once at setup — no atomics anywhere on the hot path.

#include "../common.h"

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <vector>

#define HIP_CHECK(expr)                                              \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

namespace {

constexpr int kWave = 64;

using int32x4 = __attribute__((ext_vector_type(4))) int;

// element-type helpers: compute in fp32, store in T (fp32 or bf16)
__device__ __forceinline__ float to_f32(float v) { return v; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ void st_nt(float* p, float v) {
  __builtin_nontemporal_store(v, p);
}
__device__ __forceinline__ void st_nt(__hip_bfloat16* p, float v) {
  union {
    __hip_bfloat16 b;
    unsigned short u;
  } cv;
  cv.b = __float2bfloat16(v);
  __builtin_nontemporal_store(cv.u, reinterpret_cast<unsigned short*>(p));
}
__device__ __forceinline__ void st(float* p, float v) { *p = v; }
__device__ __forceinline__ void st(__hip_bfloat16* p, float v) {
  *p = __float2bfloat16(v);
}

inline hipStream_t current_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// ---------------------------------------------------------------------------
// CSR SpMM: out[r, fchunk] = scale[r] * sum_e feat[indices[e], fchunk]
// One wave per (row, chunk) pair, grid-stride. VEC in {4, 2, 1}.
// ---------------------------------------------------------------------------

// CHUNK_OUTER: iterate column panels as the SLOWEST dimension so all
// resident waves gather from one feature panel (num_src × 64·VEC cols) at a
// time — the panel stays Infinity-Cache(L3, 256 MiB)-resident and the random
// neighbor-row gathers become L3 hits instead of HBM reads. With chunk
// fastest (CHUNK_OUTER=false) every panel streams concurrently and the
// combined working set thrashes L3.  HAS_SRC_SCALE fuses a per-source-row
// scale (the transpose/backward SpMM's D^{-1} pre-scale) into the gather.
template <typename T, int VEC, bool CHUNK_OUTER, bool HAS_SRC_SCALE,
          bool U8 = false>
__global__ void spmm_csr_kernel(const int64_t* __restrict__ indptr,
                                const int32_t* __restrict__ indices,
                                const T* __restrict__ feat,
                                const float* __restrict__ dst_scale,
                                const float* __restrict__ src_scale,
                                const int32_t* __restrict__ row_order,
                                T* __restrict__ out, int64_t num_rows,
                                int64_t F, int64_t nchunks) {
  const int64_t wave_global =
      (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t nwaves =
      (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWave;
  const int64_t npairs = num_rows * nchunks;

  for (int64_t pair = wave_global; pair < npairs; pair += nwaves) {
    int64_t r, chunk;
    if (CHUNK_OUTER) {
      r = pair % num_rows;
      chunk = pair / num_rows;
    } else {
      r = pair / nchunks;
      chunk = pair % nchunks;
    }
    // LPT scheduling: rows pre-sorted by degree descending — the heavy
    // (lognormal-tail) rows start first, light rows backfill
    if (row_order) r = row_order[r];
    const int64_t f0 = chunk * (kWave * VEC) + lane * VEC;
    if (f0 >= F) continue;
    const bool full = (f0 + VEC <= F);

    float acc[VEC];
#pragma unroll
    for (int k = 0; k < VEC; ++k) acc[k] = 0.f;

    const int64_t e_begin = indptr[r];
    const int64_t e_end = indptr[r + 1];
    int64_t e = e_begin;

    if (full) {
      // peel until e is 4-aligned so the unrolled loop can read FOUR int32
      // indices with ONE dwordx4 — per 4 edges the VMEM stream is then
      // 1 index load + 4 row gathers instead of 8 instructions (the gather
      // path is instruction-rate-limited at small F; profiles/README.md).
      for (; e < e_end && (e & 3); ++e) {
        const int64_t u = indices[e];
        const float s = HAS_SRC_SCALE ? src_scale[u] : 1.f;
#pragma unroll
        for (int k = 0; k < VEC; ++k)
          acc[k] += s * to_f32(feat[u * F + f0 + k]);
      }
      if (U8) {
        // 8 outstanding gathers per wave instead of 4 (recorded
        // experiment, PIPEGCN_SPMM_U8 — see host-side note: neutral)
        for (; e + 8 <= e_end; e += 8) {
          const int32x4 ua = __builtin_nontemporal_load(
              reinterpret_cast<const int32x4*>(indices + e));
          const int32x4 ub = __builtin_nontemporal_load(
              reinterpret_cast<const int32x4*>(indices + e + 4));
          const int64_t u[8] = {ua[0], ua[1], ua[2], ua[3],
                                ub[0], ub[1], ub[2], ub[3]};
          float sc[8];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            sc[j] = HAS_SRC_SCALE ? src_scale[u[j]] : 1.f;
          float v[8][VEC];
#pragma unroll
          for (int j = 0; j < 8; ++j)
#pragma unroll
            for (int k = 0; k < VEC; ++k)
              v[j][k] = to_f32(feat[u[j] * F + f0 + k]);
#pragma unroll
          for (int j = 0; j < 8; ++j)
#pragma unroll
            for (int k = 0; k < VEC; ++k) acc[k] += sc[j] * v[j][k];
        }
      }
      for (; e + 4 <= e_end; e += 4) {
        const int32x4 uu = __builtin_nontemporal_load(
            reinterpret_cast<const int32x4*>(indices + e));
        const int64_t u0 = uu[0];
        const int64_t u1 = uu[1];
        const int64_t u2 = uu[2];
        const int64_t u3 = uu[3];
        float s0 = 1.f, s1 = 1.f, s2 = 1.f, s3 = 1.f;
        if (HAS_SRC_SCALE) {
          s0 = src_scale[u0];
          s1 = src_scale[u1];
          s2 = src_scale[u2];
          s3 = src_scale[u3];
        }
        float v0[VEC], v1[VEC], v2[VEC], v3[VEC];
#pragma unroll
        for (int k = 0; k < VEC; ++k) v0[k] = to_f32(feat[u0 * F + f0 + k]);
#pragma unroll
        for (int k = 0; k < VEC; ++k) v1[k] = to_f32(feat[u1 * F + f0 + k]);
#pragma unroll
        for (int k = 0; k < VEC; ++k) v2[k] = to_f32(feat[u2 * F + f0 + k]);
#pragma unroll
        for (int k = 0; k < VEC; ++k) v3[k] = to_f32(feat[u3 * F + f0 + k]);
        if (HAS_SRC_SCALE) {
#pragma unroll
          for (int k = 0; k < VEC; ++k)
            acc[k] += s0 * v0[k] + s1 * v1[k] + s2 * v2[k] + s3 * v3[k];
        } else {
#pragma unroll
          for (int k = 0; k < VEC; ++k)
            acc[k] += v0[k] + v1[k] + v2[k] + v3[k];
        }
      }
      for (; e < e_end; ++e) {
        const int64_t u = indices[e];
        const float s = HAS_SRC_SCALE ? src_scale[u] : 1.f;
#pragma unroll
        for (int k = 0; k < VEC; ++k)
          acc[k] += s * to_f32(feat[u * F + f0 + k]);
      }
      const float s = dst_scale ? dst_scale[r] : 1.f;
#pragma unroll
      for (int k = 0; k < VEC; ++k) st_nt(out + r * F + f0 + k, acc[k] * s);
    } else {
      // ragged tail chunk: scalar guarded
      for (; e < e_end; ++e) {
        const int64_t u = indices[e];
        const float s = HAS_SRC_SCALE ? src_scale[u] : 1.f;
        for (int k = 0; k < VEC && f0 + k < F; ++k)
          acc[k] += s * to_f32(feat[u * F + f0 + k]);
      }
      const float s = dst_scale ? dst_scale[r] : 1.f;
      for (int k = 0; k < VEC && f0 + k < F; ++k)
        st(out + r * F + f0 + k, acc[k] * s);
    }
  }
}

// Multi-chunk variant (MEASURED NEGATIVE — kept env-gated as a recorded
// experiment, PIPEGCN_SPMM_NC=8): ONE wave walks a row's edge list once and
// covers up to NC column chunks, reading the index list once and turning
// each edge's gathers into NC back-to-back requests on one contiguous
// feature row. Measured 2.0-2.2x SLOWER than the chunk-per-wave default
// (F=602: 41->80 ms; F=256: 15.5->31.5 ms, with 4-edge index lookahead):
// concentrating the same gather instructions into NC-fold fewer waves
// trades away wave-level parallelism, which on MI355X hides gather latency
// far better than intra-wave ILP does.
template <typename T, int VEC, int NC, bool HAS_SRC_SCALE>
__global__ void spmm_csr_mc_kernel(const int64_t* __restrict__ indptr,
                                   const int32_t* __restrict__ indices,
                                   const T* __restrict__ feat,
                                   const float* __restrict__ dst_scale,
                                   const float* __restrict__ src_scale,
                                   const int32_t* __restrict__ row_order,
                                   T* __restrict__ out, int64_t num_rows,
                                   int64_t F, int64_t ngroups) {
  const int64_t wave_global =
      (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t nwaves =
      (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWave;
  const int64_t npairs = num_rows * ngroups;

  for (int64_t pair = wave_global; pair < npairs; pair += nwaves) {
    int64_t r = pair / ngroups;
    const int64_t grp = pair % ngroups;
    if (row_order) r = row_order[r];
    const int64_t base =
        grp * NC * (kWave * VEC) + static_cast<int64_t>(lane) * VEC;
    if (base >= F) continue;

    float acc[NC][VEC];
#pragma unroll
    for (int c = 0; c < NC; ++c)
#pragma unroll
      for (int k = 0; k < VEC; ++k) acc[c][k] = 0.f;

    const int64_t e_end = indptr[r + 1];
    int64_t e = indptr[r];
    auto body = [&](int64_t u) {
      const float s = HAS_SRC_SCALE ? src_scale[u] : 1.f;
      const T* frow = feat + u * F;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        const int64_t f = base + c * (kWave * VEC);
        if (f + VEC <= F) {
#pragma unroll
          for (int k = 0; k < VEC; ++k)
            acc[c][k] += s * to_f32(frow[f + k]);
        } else if (f < F) {
          for (int k = 0; f + k < F; ++k)
            acc[c][k] += s * to_f32(frow[f + k]);
        }
      }
    };
    // 4-edge lookahead: one dwordx4 index load feeds 4 edges of gathers so
    // the index-load latency is off the per-edge critical path
    for (; e < e_end && (e & 3); ++e)
      body(__builtin_nontemporal_load(indices + e));
    for (; e + 4 <= e_end; e += 4) {
      const int32x4 uu = __builtin_nontemporal_load(
          reinterpret_cast<const int32x4*>(indices + e));
      body(uu[0]);
      body(uu[1]);
      body(uu[2]);
      body(uu[3]);
    }
    for (; e < e_end; ++e)
      body(__builtin_nontemporal_load(indices + e));
    const float s = dst_scale ? dst_scale[r] : 1.f;
    T* orow = out + r * F;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      const int64_t f = base + c * (kWave * VEC);
      if (f + VEC <= F) {
#pragma unroll
        for (int k = 0; k < VEC; ++k) st_nt(orow + f + k, acc[c][k] * s);
      } else if (f < F) {
        for (int k = 0; f + k < F; ++k) st(orow + f + k, acc[c][k] * s);
      }
    }
  }
}

template <typename T, int VEC>
void launch_spmm_mc(const int64_t* indptr, const int32_t* indices,
                    const T* feat, const float* dst_scale,
                    const float* src_scale, const int32_t* row_order, T* out,
                    int64_t num_rows, int64_t F, hipStream_t stream) {
  constexpr int NC = 8;
  const int64_t nchunks = (F + kWave * VEC - 1) / (kWave * VEC);
  const int64_t ngroups = (nchunks + NC - 1) / NC;
  const int threads = 256;
  int64_t blocks = (num_rows * ngroups * kWave + threads - 1) / threads;
  blocks = std::min<int64_t>(blocks, 8 * 65536);
  if (blocks == 0) blocks = 1;
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(threads), 0, stream, indptr,
                       indices, feat, dst_scale, src_scale, row_order, out,
                       num_rows, F, ngroups);
  };
  if (src_scale)
    launch(HIP_KERNEL_NAME(spmm_csr_mc_kernel<T, VEC, NC, true>));
  else
    launch(HIP_KERNEL_NAME(spmm_csr_mc_kernel<T, VEC, NC, false>));
  HIP_CHECK(hipGetLastError());
}

template <typename T, int VEC>
void launch_spmm(const int64_t* indptr, const int32_t* indices,
                 const T* feat, const float* dst_scale,
                 const float* src_scale, const int32_t* row_order, T* out,
                 int64_t num_src, int64_t num_rows, int64_t F,
                 bool chunk_outer, hipStream_t stream) {
  const int64_t nchunks = (F + kWave * VEC - 1) / (kWave * VEC);
  const int threads = 256;  // 4 waves
  const int64_t npairs = num_rows * nchunks;
  int64_t blocks = (npairs * kWave + threads - 1) / threads;
  // MI355X: 256 CUs; cap the grid, grid-stride covers the rest. For
  // chunk-outer panel locality the wave front must sweep rows in order, so
  // cap at full residency (256 CU × 8 blocks ≈ upper bound).
  blocks = std::min<int64_t>(blocks, chunk_outer ? 4096 : 8 * 65536);
  if (blocks == 0) blocks = 1;
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(threads), 0, stream, indptr,
                       indices, feat, dst_scale, src_scale, row_order, out,
                       num_rows, F, nchunks);
  };
  // PIPEGCN_SPMM_U8=1 switches to the 8-edge inner loop — measured
  // NEUTRAL-to-slightly-negative (F=602 fwd 41.2 -> 42.1 ms, F=256
  // 15.5 -> 15.4): the compiler already software-pipelines the 4-edge
  // loop's gathers, so per-wave MLP was not the limiter; the 3.8 TB/s
  // actual-HBM plateau (PMC, profiles/README.md) is DRAM efficiency on
  // random 512 B row-slice reads, not issue or in-flight depth.
  static const bool u8 = [] {
    const char* s = std::getenv("PIPEGCN_SPMM_U8");
    return s && s[0] == '1';
  }();
  if (chunk_outer) {
    if (src_scale)
      launch(u8 ? HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, true, true, true>)
                : HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, true, true>));
    else
      launch(u8 ? HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, true, false, true>)
                : HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, true, false>));
  } else {
    if (src_scale)
      launch(u8 ? HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, false, true, true>)
                : HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, false, true>));
    else
      launch(u8
                 ? HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, false, false, true>)
                 : HIP_KERNEL_NAME(spmm_csr_kernel<T, VEC, false, false>));
  }
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// Row gather / scatter-add. One wave per (row, chunk).
// ---------------------------------------------------------------------------

template <typename T, int VEC, bool SCATTER_ADD>
__global__ void rowcopy_kernel(const T* __restrict__ src,
                               const int64_t* __restrict__ idx,
                               T* __restrict__ dst, int64_t nrows,
                               int64_t F, int64_t nchunks) {
  const int64_t wave_global =
      (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t nwaves =
      (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWave;
  const int64_t npairs = nrows * nchunks;
  for (int64_t pair = wave_global; pair < npairs; pair += nwaves) {
    const int64_t i = pair / nchunks;
    const int64_t f0 = pair % nchunks * (kWave * VEC) + lane * VEC;
    if (f0 >= F) continue;
    const int64_t j = idx[i];
    if (f0 + VEC <= F) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        if (SCATTER_ADD)
          st(dst + j * F + f0 + k, to_f32(dst[j * F + f0 + k]) +
                                       to_f32(src[i * F + f0 + k]));
        else
          dst[i * F + f0 + k] = src[j * F + f0 + k];
      }
    } else {
      for (int k = 0; f0 + k < F; ++k) {
        if (SCATTER_ADD)
          st(dst + j * F + f0 + k, to_f32(dst[j * F + f0 + k]) +
                                       to_f32(src[i * F + f0 + k]));
        else
          dst[i * F + f0 + k] = src[j * F + f0 + k];
      }
    }
  }
}

template <typename T, int VEC, bool SCATTER_ADD>
void launch_rowcopy(const T* src, const int64_t* idx, T* dst,
                    int64_t nrows, int64_t F, hipStream_t stream) {
  const int64_t nchunks = (F + kWave * VEC - 1) / (kWave * VEC);
  const int threads = 256;
  int64_t blocks = (nrows * nchunks * kWave + threads - 1) / threads;
  blocks = std::min<int64_t>(blocks, 8 * 65536);
  if (blocks == 0) blocks = 1;
  hipLaunchKernelGGL(HIP_KERNEL_NAME(rowcopy_kernel<T, VEC, SCATTER_ADD>),
                     dim3(blocks), dim3(threads), 0, stream, src, idx, dst,
                     nrows, F, nchunks);
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// EMA: avg = m * avg + (1 - m) * x   (flat elementwise, float4)
// ---------------------------------------------------------------------------

__global__ void ema_kernel(float* __restrict__ avg,
                           const float* __restrict__ x, float m, int64_t n) {
  const int64_t i4 = (static_cast<int64_t>(blockIdx.x) * blockDim.x +
                      threadIdx.x) * 4;
  if (i4 + 4 <= n) {
    float4 a = *reinterpret_cast<float4*>(avg + i4);
    const float4 b = *reinterpret_cast<const float4*>(x + i4);
    a.x = m * a.x + (1.f - m) * b.x;
    a.y = m * a.y + (1.f - m) * b.y;
    a.z = m * a.z + (1.f - m) * b.z;
    a.w = m * a.w + (1.f - m) * b.w;
    *reinterpret_cast<float4*>(avg + i4) = a;
  } else {
    for (int64_t i = i4; i < n; ++i) avg[i] = m * avg[i] + (1.f - m) * x[i];
  }
}

// ---------------------------------------------------------------------------
// Column sum: out[n] = sum_m x[m, n]  (bias/BN reductions — ROCm eager
// .sum(0) runs at ~130 GB/s on tall tensors). Two-phase: blocks accumulate
// row stripes into partials [nblk, F]; a tiny second pass (torch) reduces.
// ---------------------------------------------------------------------------

template <int VEC>
__global__ void colsum_partial_kernel(const float* __restrict__ x,
                                      float* __restrict__ partials,
                                      int64_t M, int64_t F,
                                      int64_t rows_per_blk) {
  const int64_t blk = blockIdx.x;
  const int64_t r0 = blk * rows_per_blk;
  const int64_t r1 = min(r0 + rows_per_blk, M);
  // 256 threads cover F columns in VEC-wide strides
  for (int64_t f0 = static_cast<int64_t>(threadIdx.x) * VEC; f0 < F;
       f0 += 256 * VEC) {
    float acc[VEC];
#pragma unroll
    for (int k = 0; k < VEC; ++k) acc[k] = 0.f;
    if (f0 + VEC <= F) {
      for (int64_t r = r0; r < r1; ++r) {
#pragma unroll
        for (int k = 0; k < VEC; ++k) acc[k] += x[r * F + f0 + k];
      }
#pragma unroll
      for (int k = 0; k < VEC; ++k) partials[blk * F + f0 + k] = acc[k];
    } else {
      for (int64_t r = r0; r < r1; ++r)
        for (int k = 0; f0 + k < F; ++k) acc[k] += x[r * F + f0 + k];
      for (int k = 0; f0 + k < F; ++k) partials[blk * F + f0 + k] = acc[k];
    }
  }
}

int pick_vec(int64_t F, int64_t num_rows, int64_t elem_size = 4) {
  // With LPT row scheduling handling load balance, width is a pure
  // throughput choice. Measured on MI355X (profiles/README.md): 8 B lane
  // loads win at F=256 fp32 (VEC2 15.4 ms vs VEC1 15.7 / VEC4 16.0) and
  // F=256 bf16 (VEC4); 16 B is within ~3%; 4 B loses up to 40% for bf16.
  // Preference: 8 B, then 16 B, then 4 B — first width dividing F.
  if (const char* e = std::getenv("PIPEGCN_SPMM_VEC")) {
    int v = std::atoi(e);
    if ((v == 8 || v == 4 || v == 2 || v == 1) && F % v == 0 &&
        v * elem_size <= 16)
      return v;
  }
  // row count flips the 8B/16B preference: 2.45M-row graphs (products)
  // measured 16B 6% faster at F=256, 233k-row (reddit) 8B 4% faster.
  const bool big = num_rows >= (1 << 20);
  const std::initializer_list<int> pref =
      big ? std::initializer_list<int>{16, 8, 4}
          : std::initializer_list<int>{8, 16, 4};
  for (int bytes : pref) {
    if (bytes < elem_size) continue;
    const int v = bytes / static_cast<int>(elem_size);
    if (F % v == 0) return v;
  }
  return 1;
}

}  // namespace

template <typename T>
void spmm_dispatch(torch::Tensor& indptr, torch::Tensor& indices,
                   torch::Tensor& feat, const float* dsp, const float* ssp,
                   const int32_t* rop, torch::Tensor& out, int64_t num_src,
                   int64_t num_rows, int64_t F, int vec, bool chunk_outer,
                   hipStream_t stream) {
  const T* fp = reinterpret_cast<const T*>(feat.data_ptr());
  T* op = reinterpret_cast<T*>(out.data_ptr());
  const int64_t* ip = indptr.data_ptr<int64_t>();
  const int32_t* xp = indices.data_ptr<int32_t>();
  if (const char* e = std::getenv("PIPEGCN_SPMM_NC")) {
    if (std::atoi(e) == 8) {  // experimental whole-row multi-chunk variant
      switch (vec) {
        case 8:
          launch_spmm_mc<T, 8>(ip, xp, fp, dsp, ssp, rop, op, num_rows, F,
                               stream);
          return;
        case 4:
          launch_spmm_mc<T, 4>(ip, xp, fp, dsp, ssp, rop, op, num_rows, F,
                               stream);
          return;
        case 2:
          launch_spmm_mc<T, 2>(ip, xp, fp, dsp, ssp, rop, op, num_rows, F,
                               stream);
          return;
        default:
          launch_spmm_mc<T, 1>(ip, xp, fp, dsp, ssp, rop, op, num_rows, F,
                               stream);
          return;
      }
    }
  }
  switch (vec) {
    case 8:
      launch_spmm<T, 8>(ip, xp, fp, dsp, ssp, rop, op, num_src, num_rows, F,
                        chunk_outer, stream);
      break;
    case 4:
      launch_spmm<T, 4>(ip, xp, fp, dsp, ssp, rop, op, num_src, num_rows, F,
                        chunk_outer, stream);
      break;
    case 2:
      launch_spmm<T, 2>(ip, xp, fp, dsp, ssp, rop, op, num_src, num_rows, F,
                        chunk_outer, stream);
      break;
    default:
      launch_spmm<T, 1>(ip, xp, fp, dsp, ssp, rop, op, num_src, num_rows, F,
                        chunk_outer, stream);
  }
}

void spmm_csr_hip(torch::Tensor indptr, torch::Tensor indices,
                  torch::Tensor feat, torch::Tensor dst_scale,
                  torch::Tensor src_scale, torch::Tensor row_order,
                  torch::Tensor out) {
  TORCH_CHECK(feat.is_cuda() && out.is_cuda(), "spmm_csr_hip: device tensors");
  const bool bf16 = feat.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || feat.scalar_type() == torch::kFloat,
              "spmm: fp32 or bf16 only");
  TORCH_CHECK(out.scalar_type() == feat.scalar_type());
  TORCH_CHECK(feat.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(indptr.scalar_type() == torch::kLong &&
              indices.scalar_type() == torch::kInt);
  const int64_t num_rows = out.size(0);
  const int64_t num_src = feat.size(0);
  const int64_t F = feat.size(1);
  TORCH_CHECK(out.size(1) == F && indptr.numel() == num_rows + 1);
  const float* dsp = nullptr;
  if (dst_scale.defined() && dst_scale.numel() > 0) {
    TORCH_CHECK(dst_scale.is_contiguous() && dst_scale.numel() == num_rows);
    TORCH_CHECK(dst_scale.scalar_type() == torch::kFloat);
    dsp = dst_scale.data_ptr<float>();
  }
  const float* ssp = nullptr;
  if (src_scale.defined() && src_scale.numel() > 0) {
    TORCH_CHECK(src_scale.is_contiguous() && src_scale.numel() == num_src);
    TORCH_CHECK(src_scale.scalar_type() == torch::kFloat);
    ssp = src_scale.data_ptr<float>();
  }
  const int32_t* rop = nullptr;
  if (row_order.defined() && row_order.numel() > 0) {
    TORCH_CHECK(row_order.is_contiguous() && row_order.numel() == num_rows &&
                row_order.scalar_type() == torch::kInt);
    rop = row_order.data_ptr<int32_t>();
  }
  auto stream = current_stream();
  const int vec = pick_vec(F, num_rows, bf16 ? 2 : 4);
  bool chunk_outer = false;  // measured: chunk-inner wins at every shape
  if (const char* e = std::getenv("PIPEGCN_SPMM_ORDER"))
    chunk_outer = (e[0] == 'o');
  if (bf16)
    spmm_dispatch<__hip_bfloat16>(indptr, indices, feat, dsp, ssp, rop, out,
                                  num_src, num_rows, F, vec, chunk_outer,
                                  stream);
  else
    spmm_dispatch<float>(indptr, indices, feat, dsp, ssp, rop, out, num_src,
                         num_rows, F, vec, chunk_outer, stream);
}

template <typename T, bool SCATTER_ADD>
void rowcopy_dispatch(torch::Tensor& src, torch::Tensor& idx,
                      torch::Tensor& dst, int64_t n, int64_t F,
                      int vec, hipStream_t stream) {
  const T* sp = reinterpret_cast<const T*>(src.data_ptr());
  T* dp = reinterpret_cast<T*>(dst.data_ptr());
  const int64_t* ixp = idx.data_ptr<int64_t>();
  switch (vec) {
    case 8:
      launch_rowcopy<T, 8, SCATTER_ADD>(sp, ixp, dp, n, F, stream);
      break;
    case 4:
      launch_rowcopy<T, 4, SCATTER_ADD>(sp, ixp, dp, n, F, stream);
      break;
    case 2:
      launch_rowcopy<T, 2, SCATTER_ADD>(sp, ixp, dp, n, F, stream);
      break;
    default:
      launch_rowcopy<T, 1, SCATTER_ADD>(sp, ixp, dp, n, F, stream);
  }
}

void gather_rows_hip(torch::Tensor src, torch::Tensor idx, torch::Tensor out) {
  TORCH_CHECK(src.is_cuda() && idx.is_cuda() && out.is_cuda());
  const bool bf16 = src.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || src.scalar_type() == torch::kFloat);
  TORCH_CHECK(out.scalar_type() == src.scalar_type() &&
              idx.scalar_type() == torch::kLong);
  TORCH_CHECK(src.is_contiguous() && out.is_contiguous());
  const int64_t F = src.size(1);
  const int64_t n = idx.numel();
  TORCH_CHECK(out.size(0) == n && out.size(1) == F);
  auto stream = current_stream();
  const int vec = pick_vec(F, 0, bf16 ? 2 : 4);
  if (bf16)
    rowcopy_dispatch<__hip_bfloat16, false>(src, idx, out, n, F, vec, stream);
  else
    rowcopy_dispatch<float, false>(src, idx, out, n, F, vec, stream);
}

void scatter_add_rows_hip(torch::Tensor dst, torch::Tensor idx,
                          torch::Tensor src) {
  // Contract: idx entries are unique (one boundary peer at a time) — rows
  // are written by exactly one wave, so no atomics are needed.
  TORCH_CHECK(dst.is_cuda() && idx.is_cuda() && src.is_cuda());
  const bool bf16 = dst.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || dst.scalar_type() == torch::kFloat);
  TORCH_CHECK(src.scalar_type() == dst.scalar_type() &&
              idx.scalar_type() == torch::kLong);
  TORCH_CHECK(src.is_contiguous() && dst.is_contiguous());
  const int64_t F = dst.size(1);
  const int64_t n = idx.numel();
  TORCH_CHECK(src.size(0) == n && src.size(1) == F);
  auto stream = current_stream();
  const int vec = pick_vec(F, 0, bf16 ? 2 : 4);
  if (bf16)
    rowcopy_dispatch<__hip_bfloat16, true>(src, idx, dst, n, F, vec, stream);
  else
    rowcopy_dispatch<float, true>(src, idx, dst, n, F, vec, stream);
}

torch::Tensor colsum_hip(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kFloat);
  const int64_t M = x.size(0), F = x.size(1);
  // enough blocks to fill the chip; each handles a contiguous row stripe
  const int64_t nblk = std::min<int64_t>(2048, (M + 255) / 256);
  const int64_t rows_per_blk = (M + nblk - 1) / nblk;
  auto partials = torch::empty({nblk, F}, x.options());
  auto stream = current_stream();
  const int vec = F % 4 == 0 ? 4 : (F % 2 == 0 ? 2 : 1);
  if (vec == 4)
    hipLaunchKernelGGL(HIP_KERNEL_NAME(colsum_partial_kernel<4>), dim3(nblk),
                       dim3(256), 0, stream, x.data_ptr<float>(),
                       partials.data_ptr<float>(), M, F, rows_per_blk);
  else if (vec == 2)
    hipLaunchKernelGGL(HIP_KERNEL_NAME(colsum_partial_kernel<2>), dim3(nblk),
                       dim3(256), 0, stream, x.data_ptr<float>(),
                       partials.data_ptr<float>(), M, F, rows_per_blk);
  else
    hipLaunchKernelGGL(HIP_KERNEL_NAME(colsum_partial_kernel<1>), dim3(nblk),
                       dim3(256), 0, stream, x.data_ptr<float>(),
                       partials.data_ptr<float>(), M, F, rows_per_blk);
  HIP_CHECK(hipGetLastError());
  return partials.sum(0);
}

void ema_update_hip(torch::Tensor avg, torch::Tensor x, double momentum) {
  TORCH_CHECK(avg.is_cuda() && x.is_cuda());
  TORCH_CHECK(avg.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(avg.numel() == x.numel());
  const int64_t n = avg.numel();
  const int threads = 256;
  int64_t blocks = (n + 4 * threads - 1) / (4 * threads);
  if (blocks == 0) blocks = 1;
  hipLaunchKernelGGL(ema_kernel, dim3(blocks), dim3(threads), 0,
                     current_stream(), avg.data_ptr<float>(),
                     x.data_ptr<float>(), static_cast<float>(momentum), n);
  HIP_CHECK(hipGetLastError());
}
