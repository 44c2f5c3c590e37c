// Fused elementwise kernels for the inter-layer tail (gfx950):
//
//   1. dropout fwd/bwd with a BITPACKED mask (1 bit/element instead of
//      torch's 1 byte/element — 8x less mask memory, one fused pass)
//   2. LayerNorm [+ fused ReLU] fwd/bwd, wave-per-row
//
// Replaces the eager torch dropout / nn.LayerNorm / F.relu between layers
// (reference sites: /root/reference/module/model.py:47,53-56). Fusing
// LN+ReLU saves one full [N,F] pass AND one saved tensor: eager autograd
// keeps {LN input, ReLU output} (2 x [N,F]) plus a byte dropout mask; the
// fused pair keeps {xhat} (1 x [N,F]) and a bit mask. On the papers100M
// single-rank sizing (27.6M-row activations) that difference is tens of GB.
//
// LayerNorm layout: one 64-lane wave per row; per-lane partial sums reduced
// with __shfl_xor (6 steps, no LDS); stats in fp32 for both dtypes.
// Backward dweight/dbias: each wave accumulates its fixed column set in
// registers across its rows and writes ONE partial row per wave; a column
// sum over [nwaves, F] finishes on the host side (same two-phase shape as
// colsum_hip — the single-pass eager reduction was measured pathological).

#include "../common.h"

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#define HIP_CHECK(expr)                                                  \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

namespace {

__device__ inline float to_f32(float v) { return v; }
__device__ inline float to_f32(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
__device__ inline void from_f32(float v, float* p) { *p = v; }
__device__ inline void from_f32(float v, __hip_bfloat16* p) {
  *p = __float2bfloat16(v);
}

__device__ inline float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ inline uint64_t splitmix64(uint64_t z) {
  z += 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

// ---------------------------------------------------------------------------
// dropout: thread handles 8 consecutive elements -> one mask byte.
// keep decision: 16-bit threshold (p quantized to 1/65536) from two
// splitmix64 draws keyed on (seed, element-group index) — counter-based,
// stateless, reproducible given the host-passed seed.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void dropout_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   uint8_t* __restrict__ mask, uint64_t seed,
                                   uint32_t p16, float scale, int64_t n) {
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t t = static_cast<int64_t>(blockIdx.x) * blockDim.x +
                   threadIdx.x;
       t * 8 < n; t += stride) {
    const int64_t i0 = t * 8;
    const uint64_t r0 = splitmix64(seed ^ (2 * (uint64_t)t));
    const uint64_t r1 = splitmix64(seed ^ (2 * (uint64_t)t + 1));
    uint8_t m = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const uint64_t r = j < 4 ? r0 : r1;
      const uint32_t bits = (uint32_t)(r >> (16 * (j & 3))) & 0xffffu;
      if (bits >= p16) m |= (1u << j);
    }
    mask[t] = m;
    if (i0 + 8 <= n) {
      if constexpr (sizeof(T) == 4) {
        // fp32: two float4 transactions instead of 8 scalar dwords
        // (i0 = 8t so the 16B alignment always holds)
        const float4 a = *reinterpret_cast<const float4*>(
            reinterpret_cast<const float*>(x) + i0);
        const float4 b = *reinterpret_cast<const float4*>(
            reinterpret_cast<const float*>(x) + i0 + 4);
        float4 oa, ob;
        oa.x = (m & 1u) ? a.x * scale : 0.f;
        oa.y = (m & 2u) ? a.y * scale : 0.f;
        oa.z = (m & 4u) ? a.z * scale : 0.f;
        oa.w = (m & 8u) ? a.w * scale : 0.f;
        ob.x = (m & 16u) ? b.x * scale : 0.f;
        ob.y = (m & 32u) ? b.y * scale : 0.f;
        ob.z = (m & 64u) ? b.z * scale : 0.f;
        ob.w = (m & 128u) ? b.w * scale : 0.f;
        *reinterpret_cast<float4*>(reinterpret_cast<float*>(y) + i0) = oa;
        *reinterpret_cast<float4*>(reinterpret_cast<float*>(y) + i0 + 4) =
            ob;
      } else {
        // bf16: the 8 elements are ONE 16B transaction (i0 = 8t)
        const uint4 a = *reinterpret_cast<const uint4*>(x + i0);
        const T* ap = reinterpret_cast<const T*>(&a);
        uint4 o;
        T* op = reinterpret_cast<T*>(&o);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          from_f32((m >> j) & 1 ? to_f32(ap[j]) * scale : 0.f, &op[j]);
        *reinterpret_cast<uint4*>(y + i0) = o;
      }
    } else {
      for (int j = 0; i0 + j < n; ++j) {
        const float v = (m >> j) & 1 ? to_f32(x[i0 + j]) * scale : 0.f;
        from_f32(v, &y[i0 + j]);
      }
    }
  }
}

template <typename T>
__global__ void dropout_bwd_kernel(const T* __restrict__ dy,
                                   const uint8_t* __restrict__ mask,
                                   T* __restrict__ dx, float scale,
                                   int64_t n) {
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t t = static_cast<int64_t>(blockIdx.x) * blockDim.x +
                   threadIdx.x;
       t * 8 < n; t += stride) {
    const int64_t i0 = t * 8;
    const uint8_t m = mask[t];
    if (i0 + 8 <= n) {
      if constexpr (sizeof(T) == 4) {
        const float4 a = *reinterpret_cast<const float4*>(
            reinterpret_cast<const float*>(dy) + i0);
        const float4 b = *reinterpret_cast<const float4*>(
            reinterpret_cast<const float*>(dy) + i0 + 4);
        float4 oa, ob;
        oa.x = (m & 1u) ? a.x * scale : 0.f;
        oa.y = (m & 2u) ? a.y * scale : 0.f;
        oa.z = (m & 4u) ? a.z * scale : 0.f;
        oa.w = (m & 8u) ? a.w * scale : 0.f;
        ob.x = (m & 16u) ? b.x * scale : 0.f;
        ob.y = (m & 32u) ? b.y * scale : 0.f;
        ob.z = (m & 64u) ? b.z * scale : 0.f;
        ob.w = (m & 128u) ? b.w * scale : 0.f;
        *reinterpret_cast<float4*>(reinterpret_cast<float*>(dx) + i0) = oa;
        *reinterpret_cast<float4*>(reinterpret_cast<float*>(dx) + i0 + 4) =
            ob;
      } else {
        const uint4 a = *reinterpret_cast<const uint4*>(dy + i0);
        const T* ap = reinterpret_cast<const T*>(&a);
        uint4 o;
        T* op = reinterpret_cast<T*>(&o);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          from_f32((m >> j) & 1 ? to_f32(ap[j]) * scale : 0.f, &op[j]);
        *reinterpret_cast<uint4*>(dx + i0) = o;
      }
    } else {
      for (int j = 0; i0 + j < n; ++j) {
        const float v = (m >> j) & 1 ? to_f32(dy[i0 + j]) * scale : 0.f;
        from_f32(v, &dx[i0 + j]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// LayerNorm [+ReLU] forward: wave per row. Saves xhat (dtype T) + rstd (f32).
// MAXCH register chunks of VEC=4 columns per lane cover F <= 1024.
// ---------------------------------------------------------------------------

constexpr int LN_VEC = 4;
constexpr int LN_MAXCH = 4;  // F <= 64 * 4 * 4 = 1024

template <typename T, bool RELU>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const float* __restrict__ w,
                              const float* __restrict__ b,
                              T* __restrict__ y, T* __restrict__ xhat,
                              float* __restrict__ rstd_out, float eps,
                              int64_t N, int F) {
  // TWO rows per wave: the 12 cross-lane reduction steps per row are
  // latency-bound, and two independent rows interleave them (measured
  // win over one-row-per-wave at [233k,256])
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t row0 = (static_cast<int64_t>(blockIdx.x) * 4 + wave) * 2;
  if (row0 >= N) return;
  const int nrows = row0 + 1 < N ? 2 : 1;

  float v[2][LN_MAXCH][LN_VEC];
  float s[2] = {0.f, 0.f}, s2[2] = {0.f, 0.f};
  const int nch = (F + 64 * LN_VEC - 1) / (64 * LN_VEC);
#pragma unroll
  for (int r = 0; r < 2; ++r) {
    if (r >= nrows) break;
    const T* xr = x + (row0 + r) * F;
#pragma unroll
    for (int c = 0; c < LN_MAXCH; ++c) {
      if (c >= nch) break;
#pragma unroll
      for (int k = 0; k < LN_VEC; ++k) {
        const int f = (c * 64 + lane) * LN_VEC + k;
        const float u = f < F ? to_f32(xr[f]) : 0.f;
        v[r][c][k] = u;
        s[r] += u;
        s2[r] += u * u;
      }
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s[0] += __shfl_xor(s[0], off, 64);
    s2[0] += __shfl_xor(s2[0], off, 64);
    s[1] += __shfl_xor(s[1], off, 64);
    s2[1] += __shfl_xor(s2[1], off, 64);
  }
  float mean[2], rstd[2];
#pragma unroll
  for (int r = 0; r < 2; ++r) {
    if (r >= nrows) break;
    mean[r] = s[r] / F;
    rstd[r] = rsqrtf(fmaxf(s2[r] / F - mean[r] * mean[r], 0.f) + eps);
    if (lane == 0) rstd_out[row0 + r] = rstd[r];
  }

#pragma unroll
  for (int r = 0; r < 2; ++r) {
    if (r >= nrows) break;
    T* yr = y + (row0 + r) * F;
    T* hr = xhat + (row0 + r) * F;
#pragma unroll
    for (int c = 0; c < LN_MAXCH; ++c) {
      if (c >= nch) break;
#pragma unroll
      for (int k = 0; k < LN_VEC; ++k) {
        const int f = (c * 64 + lane) * LN_VEC + k;
        if (f >= F) break;
        const float h = (v[r][c][k] - mean[r]) * rstd[r];
        from_f32(h, &hr[f]);
        float o = w[f] * h + b[f];
        if (RELU) o = fmaxf(o, 0.f);
        from_f32(o, &yr[f]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// LayerNorm [+ReLU] backward. Wave per row (grid-stride over rows); each
// wave accumulates dweight/dbias for its fixed column set in registers and
// writes one partial row at the end.
//   g   = dy * relu_mask          (mask recomputed from w*xhat+b > 0)
//   dyw = g * w
//   dx  = rstd * (dyw - mean_f(dyw) - xhat * mean_f(dyw * xhat))
// ---------------------------------------------------------------------------

template <typename T, bool RELU>
__global__ void ln_bwd_kernel(const T* __restrict__ dy,
                              const T* __restrict__ xhat,
                              const float* __restrict__ rstd,
                              const float* __restrict__ w,
                              const float* __restrict__ b,
                              T* __restrict__ dx,
                              float* __restrict__ dw_part,
                              float* __restrict__ db_part, int64_t N, int F) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t wid = static_cast<int64_t>(blockIdx.x) * 4 + wave;
  const int64_t nwaves = static_cast<int64_t>(gridDim.x) * 4;
  const int nch = (F + 64 * LN_VEC - 1) / (64 * LN_VEC);

  float acc_dw[LN_MAXCH][LN_VEC] = {};
  float acc_db[LN_MAXCH][LN_VEC] = {};

  for (int64_t row = wid; row < N; row += nwaves) {
    const T* dyr = dy + row * F;
    const T* hr = xhat + row * F;
    const float rs = rstd[row];
    float g[LN_MAXCH][LN_VEC];
    float h[LN_MAXCH][LN_VEC];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int c = 0; c < LN_MAXCH; ++c) {
      if (c >= nch) break;
#pragma unroll
      for (int k = 0; k < LN_VEC; ++k) {
        const int f = (c * 64 + lane) * LN_VEC + k;
        float gg = 0.f, hh = 0.f;
        if (f < F) {
          hh = to_f32(hr[f]);
          gg = to_f32(dyr[f]);
          if (RELU && w[f] * hh + b[f] <= 0.f) gg = 0.f;
          acc_db[c][k] += gg;
          acc_dw[c][k] += gg * hh;
          gg *= w[f];  // dyw
        }
        g[c][k] = gg;
        h[c][k] = hh;
        s1 += gg;
        s2 += gg * hh;
      }
    }
    s1 = wave_sum(s1) / F;
    s2 = wave_sum(s2) / F;
    T* dxr = dx + row * F;
#pragma unroll
    for (int c = 0; c < LN_MAXCH; ++c) {
      if (c >= nch) break;
#pragma unroll
      for (int k = 0; k < LN_VEC; ++k) {
        const int f = (c * 64 + lane) * LN_VEC + k;
        if (f >= F) break;
        from_f32(rs * (g[c][k] - s1 - h[c][k] * s2), &dxr[f]);
      }
    }
  }

  // one partial row per wave
  float* dwp = dw_part + wid * F;
  float* dbp = db_part + wid * F;
#pragma unroll
  for (int c = 0; c < LN_MAXCH; ++c) {
    if (c >= nch) break;
#pragma unroll
    for (int k = 0; k < LN_VEC; ++k) {
      const int f = (c * 64 + lane) * LN_VEC + k;
      if (f >= F) break;
      dwp[f] = acc_dw[c][k];
      dbp[f] = acc_db[c][k];
    }
  }
}

inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

}  // namespace

void dropout_fwd_hip(torch::Tensor x, torch::Tensor y, torch::Tensor mask,
                     double p, int64_t seed) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(mask.scalar_type() == torch::kUInt8 && mask.is_contiguous());
  const int64_t n = x.numel();
  TORCH_CHECK(mask.numel() >= (n + 7) / 8);
  TORCH_CHECK(p >= 0.0 && p < 1.0);
  const uint32_t p16 = static_cast<uint32_t>(p * 65536.0 + 0.5);
  const float scale = 1.f / (1.f - p16 / 65536.f);
  const int64_t groups = (n + 7) / 8;
  const int threads = 256;
  const int64_t blocks =
      std::min<int64_t>((groups + threads - 1) / threads, 65535 * 8);
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat);
  if (bf16)
    hipLaunchKernelGGL(HIP_KERNEL_NAME(dropout_fwd_kernel<__hip_bfloat16>),
                       dim3(blocks), dim3(threads), 0, cur_stream(),
                       reinterpret_cast<__hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                       mask.data_ptr<uint8_t>(), (uint64_t)seed, p16, scale,
                       n);
  else
    hipLaunchKernelGGL(HIP_KERNEL_NAME(dropout_fwd_kernel<float>),
                       dim3(blocks), dim3(threads), 0, cur_stream(),
                       x.data_ptr<float>(), y.data_ptr<float>(),
                       mask.data_ptr<uint8_t>(), (uint64_t)seed, p16, scale,
                       n);
  HIP_CHECK(hipGetLastError());
}

void dropout_bwd_hip(torch::Tensor dy, torch::Tensor mask, torch::Tensor dx,
                     double p) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dx.is_contiguous());
  const int64_t n = dy.numel();
  const uint32_t p16 = static_cast<uint32_t>(p * 65536.0 + 0.5);
  const float scale = 1.f / (1.f - p16 / 65536.f);
  const int64_t groups = (n + 7) / 8;
  const int threads = 256;
  const int64_t blocks =
      std::min<int64_t>((groups + threads - 1) / threads, 65535 * 8);
  const bool bf16 = dy.scalar_type() == torch::kBFloat16;
  if (bf16)
    hipLaunchKernelGGL(HIP_KERNEL_NAME(dropout_bwd_kernel<__hip_bfloat16>),
                       dim3(blocks), dim3(threads), 0, cur_stream(),
                       reinterpret_cast<__hip_bfloat16*>(dy.data_ptr()),
                       mask.data_ptr<uint8_t>(),
                       reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()),
                       scale, n);
  else
    hipLaunchKernelGGL(HIP_KERNEL_NAME(dropout_bwd_kernel<float>),
                       dim3(blocks), dim3(threads), 0, cur_stream(),
                       dy.data_ptr<float>(), mask.data_ptr<uint8_t>(),
                       dx.data_ptr<float>(), scale, n);
  HIP_CHECK(hipGetLastError());
}

void layer_norm_relu_fwd_hip(torch::Tensor x, torch::Tensor w,
                             torch::Tensor b, double eps, bool relu,
                             torch::Tensor y, torch::Tensor xhat,
                             torch::Tensor rstd) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  const int64_t N = x.size(0);
  const int64_t F = x.size(1);
  TORCH_CHECK(F <= 64 * LN_VEC * LN_MAXCH, "LayerNorm F > 1024 unsupported");
  TORCH_CHECK(w.scalar_type() == torch::kFloat &&
              b.scalar_type() == torch::kFloat);
  TORCH_CHECK(w.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(rstd.scalar_type() == torch::kFloat && rstd.numel() == N);
  if (N == 0) return;
  const int64_t blocks = (N + 7) / 8;  // 4 waves x 2 rows per block
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat);
  auto launch = [&](auto tptr, auto kernel) {
    using TP = decltype(tptr);
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                       reinterpret_cast<TP>(x.data_ptr()),
                       w.data_ptr<float>(), b.data_ptr<float>(),
                       reinterpret_cast<TP>(y.data_ptr()),
                       reinterpret_cast<TP>(xhat.data_ptr()),
                       rstd.data_ptr<float>(), (float)eps, N, (int)F);
  };
  if (bf16 && relu)
    launch((__hip_bfloat16*)nullptr,
           HIP_KERNEL_NAME(ln_fwd_kernel<__hip_bfloat16, true>));
  else if (bf16)
    launch((__hip_bfloat16*)nullptr,
           HIP_KERNEL_NAME(ln_fwd_kernel<__hip_bfloat16, false>));
  else if (relu)
    launch((float*)nullptr, HIP_KERNEL_NAME(ln_fwd_kernel<float, true>));
  else
    launch((float*)nullptr, HIP_KERNEL_NAME(ln_fwd_kernel<float, false>));
  HIP_CHECK(hipGetLastError());
}

void layer_norm_relu_bwd_hip(torch::Tensor dy, torch::Tensor xhat,
                             torch::Tensor rstd, torch::Tensor w,
                             torch::Tensor b, bool relu, torch::Tensor dx,
                             torch::Tensor dw_part, torch::Tensor db_part) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 2 && dy.is_contiguous());
  const int64_t N = dy.size(0);
  const int64_t F = dy.size(1);
  const int64_t blocks = dw_part.size(0) / 4;
  TORCH_CHECK(dw_part.size(0) % 4 == 0 && dw_part.size(1) == F);
  TORCH_CHECK(db_part.sizes() == dw_part.sizes());
  if (N == 0) return;
  const bool bf16 = dy.scalar_type() == torch::kBFloat16;
  auto launch = [&](auto tptr, auto kernel) {
    using TP = decltype(tptr);
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                       reinterpret_cast<TP>(dy.data_ptr()),
                       reinterpret_cast<TP>(xhat.data_ptr()),
                       rstd.data_ptr<float>(), w.data_ptr<float>(),
                       b.data_ptr<float>(),
                       reinterpret_cast<TP>(dx.data_ptr()),
                       dw_part.data_ptr<float>(), db_part.data_ptr<float>(),
                       N, (int)F);
  };
  if (bf16 && relu)
    launch((__hip_bfloat16*)nullptr,
           HIP_KERNEL_NAME(ln_bwd_kernel<__hip_bfloat16, true>));
  else if (bf16)
    launch((__hip_bfloat16*)nullptr,
           HIP_KERNEL_NAME(ln_bwd_kernel<__hip_bfloat16, false>));
  else if (relu)
    launch((float*)nullptr, HIP_KERNEL_NAME(ln_bwd_kernel<float, true>));
  else
    launch((float*)nullptr, HIP_KERNEL_NAME(ln_bwd_kernel<float, false>));
  HIP_CHECK(hipGetLastError());
}
