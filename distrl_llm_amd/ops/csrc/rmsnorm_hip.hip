#include "hip/hip_runtime.h"
// RMSNorm forward + backward (dx only — norm weights are frozen in this
// framework: only LoRA trains). Replaces the fused RMSNorm the reference
// gets from vLLM/Unsloth (SURVEY.md §2.4 A+B).
//
// Memory-bound: explicit bf16x8 (16 B/lane) vector loads — hipcc does not
// auto-vectorize scalar bf16 loads (guide G13) — fp32 accumulation, one
// block per row, wave64 + LDS reduction.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

__global__ void rmsnorm_fwd_bf16(const __hip_bfloat16* __restrict__ x,
                                 const __hip_bfloat16* __restrict__ w,
                                 __hip_bfloat16* __restrict__ y,
                                 int hidden, float eps) {
  __shared__ float red[16];
  const int64_t row = blockIdx.x;
  const bf16x8* xr = reinterpret_cast<const bf16x8*>(x + row * hidden);
  const bf16x8* wr = reinterpret_cast<const bf16x8*>(w);
  bf16x8* yr = reinterpret_cast<bf16x8*>(y + row * hidden);
  const int nvec = hidden / 8;

  float ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = xr[i];
    #pragma unroll
    for (int j = 0; j < 8; ++j) { float f = bf2f(v.v[j]); ss += f * f; }
  }
  ss = block_sum(ss, red);
  const float r = rsqrtf(ss / hidden + eps);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 v = xr[i], wv = wr[i], o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = f2bf(bf2f(v.v[j]) * r * bf2f(wv.v[j]));
    yr[i] = o;
  }
}

__global__ void rmsnorm_fwd_f32(const float* __restrict__ x,
                                const float* __restrict__ w,
                                float* __restrict__ y, int hidden, float eps) {
  __shared__ float red[16];
  const int64_t row = blockIdx.x;
  const float* xr = x + row * hidden;
  float* yr = y + row * hidden;
  float ss = 0.f;
  for (int i = threadIdx.x; i < hidden; i += blockDim.x) { float f = xr[i]; ss += f * f; }
  ss = block_sum(ss, red);
  const float r = rsqrtf(ss / hidden + eps);
  for (int i = threadIdx.x; i < hidden; i += blockDim.x) yr[i] = xr[i] * r * w[i];
}

// dx = r*(dy*w) - r^3/H * x * sum(dy*w*x)
__global__ void rmsnorm_bwd_bf16(const __hip_bfloat16* __restrict__ dy,
                                 const __hip_bfloat16* __restrict__ x,
                                 const __hip_bfloat16* __restrict__ w,
                                 __hip_bfloat16* __restrict__ dx,
                                 int hidden, float eps) {
  __shared__ float red[16];
  const int64_t row = blockIdx.x;
  const bf16x8* dyr = reinterpret_cast<const bf16x8*>(dy + row * hidden);
  const bf16x8* xr = reinterpret_cast<const bf16x8*>(x + row * hidden);
  const bf16x8* wr = reinterpret_cast<const bf16x8*>(w);
  bf16x8* dxr = reinterpret_cast<bf16x8*>(dx + row * hidden);
  const int nvec = hidden / 8;

  float ss = 0.f, dot = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 xv = xr[i], dyv = dyr[i], wv = wr[i];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xf = bf2f(xv.v[j]);
      float dyw = bf2f(dyv.v[j]) * bf2f(wv.v[j]);
      ss += xf * xf;
      dot += dyw * xf;
    }
  }
  ss = block_sum(ss, red);
  dot = block_sum(dot, red);
  const float r = rsqrtf(ss / hidden + eps);
  const float c = r * r * r * dot / hidden;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 xv = xr[i], dyv = dyr[i], wv = wr[i], o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float dyw = bf2f(dyv.v[j]) * bf2f(wv.v[j]);
      o.v[j] = f2bf(dyw * r - bf2f(xv.v[j]) * c);
    }
    dxr[i] = o;
  }
}

__global__ void rmsnorm_bwd_f32(const float* __restrict__ dy,
                                const float* __restrict__ x,
                                const float* __restrict__ w,
                                float* __restrict__ dx, int hidden, float eps) {
  __shared__ float red[16];
  const int64_t row = blockIdx.x;
  const float* dyr = dy + row * hidden;
  const float* xr = x + row * hidden;
  float* dxr = dx + row * hidden;
  float ss = 0.f, dot = 0.f;
  for (int i = threadIdx.x; i < hidden; i += blockDim.x) {
    float dyw = dyr[i] * w[i];
    ss += xr[i] * xr[i];
    dot += dyw * xr[i];
  }
  ss = block_sum(ss, red);
  dot = block_sum(dot, red);
  const float r = rsqrtf(ss / hidden + eps);
  const float c = r * r * r * dot / hidden;
  for (int i = threadIdx.x; i < hidden; i += blockDim.x)
    dxr[i] = dyr[i] * w[i] * r - xr[i] * c;
}

}  // namespace

torch::Tensor rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int hidden = x.size(-1);
  const int64_t rows = x.numel() / hidden;
  auto y = torch::empty_like(x);
  if (rows == 0) return y;
  dim3 grid(rows), block(256);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (x.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(hidden % 8 == 0, "rmsnorm bf16 needs hidden % 8 == 0");
    hipLaunchKernelGGL(rmsnorm_fwd_bf16, grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                       hidden, (float)eps);
  } else {
    TORCH_CHECK(x.scalar_type() == at::kFloat, "rmsnorm: bf16 or fp32 only");
    hipLaunchKernelGGL(rmsnorm_fwd_f32, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(),
                       y.data_ptr<float>(), hidden, (float)eps);
  }
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                          double eps) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const int hidden = x.size(-1);
  const int64_t rows = x.numel() / hidden;
  auto dx = torch::empty_like(x);
  if (rows == 0) return dx;
  dim3 grid(rows), block(256);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (x.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(hidden % 8 == 0);
    hipLaunchKernelGGL(rmsnorm_bwd_bf16, grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()),
                       hidden, (float)eps);
  } else {
    TORCH_CHECK(x.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(rmsnorm_bwd_f32, grid, block, 0, stream,
                       dy.data_ptr<float>(), x.data_ptr<float>(),
                       w.data_ptr<float>(), dx.data_ptr<float>(),
                       hidden, (float)eps);
  }
  HIP_CHECK_LAST();
  return dx;
}

// ---- fused residual-add + RMSNorm (decode path): res = x + h;
// y = rmsnorm(res) * w. Saves one full elementwise pass per layer site.
namespace {

__global__ void add_rmsnorm_fwd_bf16(const __hip_bfloat16* __restrict__ x,
                                     const __hip_bfloat16* __restrict__ h,
                                     const __hip_bfloat16* __restrict__ w,
                                     __hip_bfloat16* __restrict__ y,
                                     __hip_bfloat16* __restrict__ res,
                                     int hidden, float eps) {
  __shared__ float red[16];
  const int64_t row = blockIdx.x;
  const bf16x8* xr = reinterpret_cast<const bf16x8*>(x + row * hidden);
  const bf16x8* hr = reinterpret_cast<const bf16x8*>(h + row * hidden);
  const bf16x8* wr = reinterpret_cast<const bf16x8*>(w);
  bf16x8* yr = reinterpret_cast<bf16x8*>(y + row * hidden);
  bf16x8* rr = reinterpret_cast<bf16x8*>(res + row * hidden);
  const int nvec = hidden / 8;

  float ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 xv = xr[i], hv = hr[i], s;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(xv.v[j]) + bf2f(hv.v[j]);
      s.v[j] = f2bf(f);
      ss += f * f;
    }
    rr[i] = s;
  }
  ss = block_sum(ss, red);
  const float r = rsqrtf(ss / hidden + eps);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    bf16x8 sv = rr[i], wv = wr[i], o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = f2bf(bf2f(sv.v[j]) * r * bf2f(wv.v[j]));
    yr[i] = o;
  }
}

}  // namespace

std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor x, torch::Tensor h,
                                           torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && h.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16);
  const int hidden = x.size(-1);
  TORCH_CHECK(hidden % 8 == 0);
  const int64_t rows = x.numel() / hidden;
  auto y = torch::empty_like(x);
  auto res = torch::empty_like(x);
  if (rows == 0) return {y, res};
  hipLaunchKernelGGL(add_rmsnorm_fwd_bf16, dim3(rows), dim3(256), 0,
                     at::hip::getCurrentHIPStreamMasqueradingAsCUDA(),
                     reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(h.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(res.data_ptr()),
                     hidden, (float)eps);
  HIP_CHECK_LAST();
  return {y, res};
}
