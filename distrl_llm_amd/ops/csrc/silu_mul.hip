// Fused SiLU-and-mul forward/backward (the SwiGLU activation the reference
// gets from vLLM activation kernels / Unsloth Triton — SURVEY.md §2.4).
// Memory-bound elementwise: bf16x8 vector loads, grid-stride loop.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

__global__ void silu_mul_fwd_bf16(const __hip_bfloat16* __restrict__ gate,
                                  const __hip_bfloat16* __restrict__ up,
                                  __hip_bfloat16* __restrict__ out,
                                  int64_t nvec) {
  const bf16x8* g8 = reinterpret_cast<const bf16x8*>(gate);
  const bf16x8* u8 = reinterpret_cast<const bf16x8*>(up);
  bf16x8* o8 = reinterpret_cast<bf16x8*>(out);
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 g = g8[i], u = u8[i], o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g.v[j]);
      float s = gf / (1.f + __expf(-gf));
      o.v[j] = f2bf(s * bf2f(u.v[j]));
    }
    o8[i] = o;
  }
}

__global__ void silu_mul_fwd_f32(const float* __restrict__ gate,
                                 const float* __restrict__ up,
                                 float* __restrict__ out, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float gf = gate[i];
    out[i] = gf / (1.f + __expf(-gf)) * up[i];
  }
}

// dgate = dy * up * sig(g) * (1 + g*(1-sig(g))); dup = dy * silu(g)
__global__ void silu_mul_bwd_bf16(const __hip_bfloat16* __restrict__ dy,
                                  const __hip_bfloat16* __restrict__ gate,
                                  const __hip_bfloat16* __restrict__ up,
                                  __hip_bfloat16* __restrict__ dgate,
                                  __hip_bfloat16* __restrict__ dup,
                                  int64_t nvec) {
  const bf16x8* d8 = reinterpret_cast<const bf16x8*>(dy);
  const bf16x8* g8 = reinterpret_cast<const bf16x8*>(gate);
  const bf16x8* u8 = reinterpret_cast<const bf16x8*>(up);
  bf16x8* dg8 = reinterpret_cast<bf16x8*>(dgate);
  bf16x8* du8 = reinterpret_cast<bf16x8*>(dup);
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 d = d8[i], g = g8[i], u = u8[i], og, ou;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g.v[j]);
      float df = bf2f(d.v[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      og.v[j] = f2bf(df * bf2f(u.v[j]) * sig * (1.f + gf * (1.f - sig)));
      ou.v[j] = f2bf(df * silu);
    }
    dg8[i] = og;
    du8[i] = ou;
  }
}

}  // namespace

torch::Tensor silu_mul_fwd(torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(gate.is_cuda() && gate.is_contiguous() && up.is_contiguous());
  auto out = torch::empty_like(gate);
  int64_t n = gate.numel();
  if (n == 0) return out;
  auto stream = at::cuda::getCurrentCUDAStream();
  if (gate.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(n % 8 == 0, "silu_mul bf16 needs numel % 8 == 0");
    int64_t nvec = n / 8;
    int blocks = (int)std::min<int64_t>(CDIV(nvec, 256), 2048);
    hipLaunchKernelGGL(silu_mul_fwd_bf16, dim3(blocks), dim3(256), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gate.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(up.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()), nvec);
  } else {
    TORCH_CHECK(gate.scalar_type() == at::kFloat);
    int blocks = (int)std::min<int64_t>(CDIV(n, 256), 2048);
    hipLaunchKernelGGL(silu_mul_fwd_f32, dim3(blocks), dim3(256), 0, stream,
                       gate.data_ptr<float>(), up.data_ptr<float>(),
                       out.data_ptr<float>(), n);
  }
  HIP_CHECK_LAST();
  return out;
}

std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dy, torch::Tensor gate,
                                        torch::Tensor up) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16,
              "silu_mul_bwd: bf16 only (fp32 path is CPU/eager)");
  auto dgate = torch::empty_like(gate);
  auto dup = torch::empty_like(up);
  int64_t n = gate.numel();
  TORCH_CHECK(n % 8 == 0);
  int64_t nvec = n / 8;
  int blocks = (int)std::min<int64_t>(CDIV(nvec, 256), 2048);
  hipLaunchKernelGGL(silu_mul_bwd_bf16, dim3(blocks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(gate.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(up.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(dgate.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(dup.data_ptr()), nvec);
  HIP_CHECK_LAST();
  return {dgate, dup};
}

// ---- packed variant: input (N, 2F) = [gate | up] rows straight out of
// the fused gate|up GEMM; output (N, F). Decode fast path.
namespace {
__global__ void silu_mul_packed_bf16(const __hip_bfloat16* __restrict__ gu,
                                     __hip_bfloat16* __restrict__ out,
                                     int fvec /* F/8 */) {
  const int64_t row = blockIdx.x;
  const bf16x8* g8 = reinterpret_cast<const bf16x8*>(gu + row * 2 * fvec * 8);
  const bf16x8* u8 = g8 + fvec;
  bf16x8* o8 = reinterpret_cast<bf16x8*>(out + row * fvec * 8);
  for (int i = threadIdx.x; i < fvec; i += blockDim.x) {
    bf16x8 g = g8[i], u = u8[i], o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g.v[j]);
      o.v[j] = f2bf(gf / (1.f + __expf(-gf)) * bf2f(u.v[j]));
    }
    o8[i] = o;
  }
}
}  // namespace

torch::Tensor silu_mul_packed(torch::Tensor gu) {
  TORCH_CHECK(gu.is_cuda() && gu.is_contiguous() && gu.dim() == 2);
  TORCH_CHECK(gu.scalar_type() == at::kBFloat16);
  const int64_t N = gu.size(0);
  const int64_t F = gu.size(1) / 2;
  TORCH_CHECK(F % 8 == 0);
  auto out = torch::empty({N, F}, gu.options());
  if (N == 0) return out;
  hipLaunchKernelGGL(silu_mul_packed_bf16, dim3(N), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<const __hip_bfloat16*>(gu.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                     (int)(F / 8));
  HIP_CHECK_LAST();
  return out;
}
