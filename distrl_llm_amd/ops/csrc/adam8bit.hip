// Fused blockwise 8-bit Adam (the reference's bnb.optim.Adam8bit,
// SURVEY.md §2.4-B): m/v states live as int8/uint8 with one fp32 absmax
// per 256-element block; the whole dequant -> Adam -> requant -> param
// update is one kernel. One workgroup per state block (256 threads = 256
// elements), absmax via block reduction.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

constexpr int QBLOCK = 256;

template <typename T>
__global__ __launch_bounds__(QBLOCK)
void adam8bit_kernel(T* __restrict__ p, const T* __restrict__ g,
                     int8_t* __restrict__ m_q, uint8_t* __restrict__ v_q,
                     float* __restrict__ m_absmax, float* __restrict__ v_absmax,
                     int64_t n, float lr, float b1, float b2, float eps,
                     float wd, float bias1, float bias2) {
  __shared__ float red[16];
  const int64_t blk = blockIdx.x;
  const int64_t i = blk * QBLOCK + threadIdx.x;
  const bool valid = i < n;

  float gv = 0.f, pv = 0.f, m = 0.f, v = 0.f;
  if (valid) {
    if constexpr (std::is_same<T, __hip_bfloat16>::value) {
      gv = bf2f(g[i]);
      pv = bf2f(p[i]);
    } else {
      gv = g[i];
      pv = p[i];
    }
    if (wd != 0.f) gv += wd * pv;
    m = (float)m_q[i] * (1.f / 127.f) * m_absmax[blk];
    v = (float)v_q[i] * (1.f / 255.f) * v_absmax[blk];
    m = b1 * m + (1.f - b1) * gv;
    v = b2 * v + (1.f - b2) * gv * gv;
    const float mhat = m / bias1;
    const float vhat = v / bias2;
    pv -= lr * mhat / (sqrtf(vhat) + eps);
  }

  // new per-block absmax
  float mam = block_max(fabsf(m), red);
  float vam = block_max(fabsf(v), red);
  mam = fmaxf(mam, 1e-12f);
  vam = fmaxf(vam, 1e-12f);
  if (threadIdx.x == 0) {
    m_absmax[blk] = mam;
    v_absmax[blk] = vam;
  }
  if (valid) {
    float mq = rintf(m / mam * 127.f);
    float vq = rintf(v / vam * 255.f);
    m_q[i] = (int8_t)fmaxf(-127.f, fminf(127.f, mq));
    v_q[i] = (uint8_t)fmaxf(0.f, fminf(255.f, vq));
    if constexpr (std::is_same<T, __hip_bfloat16>::value)
      p[i] = f2bf(pv);
    else
      p[i] = pv;
  }
}

}  // namespace

void adam8bit_step(torch::Tensor p, torch::Tensor g, torch::Tensor m_q,
                   torch::Tensor v_q, torch::Tensor m_absmax,
                   torch::Tensor v_absmax, double lr, double b1, double b2,
                   double eps, double wd, int64_t step) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && g.is_contiguous());
  const int64_t n = p.numel();
  const int64_t nblocks = CDIV(n, QBLOCK);
  TORCH_CHECK(m_absmax.numel() >= nblocks);
  const float bias1 = 1.f - powf((float)b1, (float)step);
  const float bias2 = 1.f - powf((float)b2, (float)step);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (p.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(adam8bit_kernel<__hip_bfloat16>, dim3(nblocks),
                       dim3(QBLOCK), 0, stream,
                       reinterpret_cast<__hip_bfloat16*>(p.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(g.data_ptr()),
                       m_q.data_ptr<int8_t>(), v_q.data_ptr<uint8_t>(),
                       m_absmax.data_ptr<float>(), v_absmax.data_ptr<float>(),
                       n, (float)lr, (float)b1, (float)b2, (float)eps,
                       (float)wd, bias1, bias2);
  } else {
    TORCH_CHECK(p.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(adam8bit_kernel<float>, dim3(nblocks), dim3(QBLOCK), 0,
                       stream, p.data_ptr<float>(), g.data_ptr<float>(),
                       m_q.data_ptr<int8_t>(), v_q.data_ptr<uint8_t>(),
                       m_absmax.data_ptr<float>(), v_absmax.data_ptr<float>(),
                       n, (float)lr, (float)b1, (float)b2, (float)eps,
                       (float)wd, bias1, bias2);
  }
  HIP_CHECK_LAST();
}
