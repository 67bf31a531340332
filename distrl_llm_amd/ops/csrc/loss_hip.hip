#include "hip/hip_runtime.h"
// Fused log-prob + PG/GRPO loss over the 152k vocab (SURVEY.md §2.4-B
// north star): forward emits per-token log p(target) + the row LSE in a
// single online-logsumexp streaming pass (never materializing a (B,T,V)
// log-prob tensor — the reference loops rows through log_softmax+gather,
// distributed_actor.py:253-260); backward emits dlogits directly:
// dlogits_v = w * (1[v=target] - softmax_v), w = mask*coef*dloss.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

template <typename T>
DEV_INLINE float ld(const T* p, int64_t i);
template <> DEV_INLINE float ld<__hip_bfloat16>(const __hip_bfloat16* p, int64_t i) { return bf2f(p[i]); }
template <> DEV_INLINE float ld<float>(const float* p, int64_t i) { return p[i]; }

// one workgroup per (b, t) row; single streaming pass, online logsumexp.
template <typename T>
__global__ __launch_bounds__(256)
void logprob_lse_kernel(const T* __restrict__ logits,
                        const int64_t* __restrict__ targets,
                        float* __restrict__ token_logp,
                        float* __restrict__ lse_out, int V) {
  __shared__ float red[16];
  __shared__ float tgt_logit_s;
  const int64_t row = blockIdx.x;
  const T* lr = logits + row * V;
  const int64_t tgt = targets[row];

  float m = -1e30f, s = 0.f;
  if constexpr (std::is_same<T, __hip_bfloat16>::value) {
    const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr);
    const int nvec = V / 8;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      bf16x8 v = l8[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float x = bf2f(v.v[j]);
        if (i * 8 + j == (int)tgt) tgt_logit_s = x;
        if (x > m) {
          s = s * __expf(m - x) + 1.f;
          m = x;
        } else {
          s += __expf(x - m);
        }
      }
    }
    for (int i = nvec * 8 + threadIdx.x; i < V; i += blockDim.x) {
      float x = ld(lr, i);
      if (i == (int)tgt) tgt_logit_s = x;
      if (x > m) { s = s * __expf(m - x) + 1.f; m = x; }
      else s += __expf(x - m);
    }
  } else {
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
      float x = ld(lr, i);
      if (i == tgt) tgt_logit_s = x;   // exactly one thread hits it
      if (x > m) {
        s = s * __expf(m - x) + 1.f;
        m = x;
      } else {
        s += __expf(x - m);
      }
    }
  }
  // combine per-thread (m, s): M = max m; S = sum s_i * exp(m_i - M)
  const float M = block_max(m, red);
  const float S = block_sum(s * __expf(m - M), red);
  if (threadIdx.x == 0) {
    const float lse = M + __logf(S);
    lse_out[row] = lse;
    token_logp[row] = tgt_logit_s - lse;
  }
}

template <typename T>
__global__ __launch_bounds__(256)
void loss_bwd_kernel(const T* __restrict__ logits,
                     const int64_t* __restrict__ targets,
                     const float* __restrict__ w,
                     const float* __restrict__ lse,
                     T* __restrict__ dlogits, int V) {
  const int64_t row = blockIdx.x;
  const T* lr = logits + row * V;
  T* dr = dlogits + row * V;
  const float wv = w[row];
  const float l = lse[row];
  const int64_t tgt = targets[row];
  if constexpr (std::is_same<T, __hip_bfloat16>::value) {
    const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr);
    bf16x8* d8 = reinterpret_cast<bf16x8*>(dr);
    const int nvec = V / 8;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      bf16x8 o;
      if (wv == 0.f) {
        #pragma unroll
        for (int j = 0; j < 8; ++j) o.v[j] = f2bf(0.f);
      } else {
        bf16x8 v = l8[i];
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          float p = __expf(bf2f(v.v[j]) - l);
          o.v[j] = f2bf(wv * ((i * 8 + j == (int)tgt ? 1.f : 0.f) - p));
        }
      }
      d8[i] = o;
    }
    for (int i = nvec * 8 + threadIdx.x; i < V; i += blockDim.x) {
      float g = 0.f;
      if (wv != 0.f)
        g = wv * ((i == (int)tgt ? 1.f : 0.f) - __expf(ld(lr, i) - l));
      dr[i] = f2bf(g);
    }
    return;
  }
  if (wv == 0.f) {
    for (int i = threadIdx.x; i < V; i += blockDim.x)
      dr[i] = (T)(0.f);
    return;
  }
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    float p = __expf(ld(lr, i) - l);
    float g = wv * ((i == tgt ? 1.f : 0.f) - p);
    dr[i] = g;
  }
}

}  // namespace

std::vector<torch::Tensor> logprob_lse_fwd(torch::Tensor logits,
                                           torch::Tensor targets) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == at::kLong);
  const int V = logits.size(-1);
  const int64_t rows = logits.numel() / V;
  auto sizes = logits.sizes().vec();
  sizes.pop_back();
  auto opts = logits.options().dtype(at::kFloat);
  auto token_logp = torch::empty(sizes, opts);
  auto lse = torch::empty(sizes, opts);
  if (rows == 0) return {token_logp, lse};
  dim3 grid(rows), block(256);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(logprob_lse_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
                       targets.contiguous().data_ptr<int64_t>(),
                       token_logp.data_ptr<float>(), lse.data_ptr<float>(), V);
  } else {
    TORCH_CHECK(logits.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(logprob_lse_kernel<float>, grid, block, 0, stream,
                       logits.data_ptr<float>(),
                       targets.contiguous().data_ptr<int64_t>(),
                       token_logp.data_ptr<float>(), lse.data_ptr<float>(), V);
  }
  HIP_CHECK_LAST();
  return {token_logp, lse};
}

torch::Tensor logprob_loss_bwd(torch::Tensor logits, torch::Tensor targets,
                               torch::Tensor w, torch::Tensor lse) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
  const int V = logits.size(-1);
  const int64_t rows = logits.numel() / V;
  auto dlogits = torch::empty_like(logits);
  if (rows == 0) return dlogits;
  dim3 grid(rows), block(256);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  auto wc = w.contiguous();
  auto lsec = lse.contiguous();
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(loss_bwd_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
                       targets.contiguous().data_ptr<int64_t>(),
                       wc.data_ptr<float>(), lsec.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(dlogits.data_ptr()), V);
  } else {
    hipLaunchKernelGGL(loss_bwd_kernel<float>, grid, block, 0, stream,
                       logits.data_ptr<float>(),
                       targets.contiguous().data_ptr<int64_t>(),
                       wc.data_ptr<float>(), lsec.data_ptr<float>(),
                       dlogits.data_ptr<float>(), V);
  }
  HIP_CHECK_LAST();
  return dlogits;
}
