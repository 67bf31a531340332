// Fused sampling: temperature -> top-k/top-p filtering -> categorical draw,
// one kernel, no sort (the reference's vLLM sampling stack sorts the
// 152k-vocab; SURVEY.md §2.4-A names fused top-k sampling a north star).
//
// Algorithm (per row, one workgroup):
//   A) max logit m (block reduce).
//   B) histogram over z = (l-m)/T in 1024 LDS bins spanning [-32, 0]:
//      per-bin sum(exp(z)) and count. Prefix-scan the bins from the top to
//      find the threshold bin where cumulative probability crosses top_p
//      (and cumulative count crosses top_k); tokens in bins above the
//      threshold are the kept set (resolution: 1/32 logit units — the
//      kept-set boundary can differ from an exact sort by tokens within
//      one bin of the cut, statistically immaterial at vocab 152k).
//   C) Gumbel-argmax over the kept set: argmax_v (z_v + g_v), g_v from a
//      counter-based hash RNG — an exact draw from the renormalized
//      categorical, fully parallel.
// Three streaming passes over the logits, no materialized probs.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

constexpr int NBINS = 1024;
constexpr float ZRANGE = 32.f;  // z below -32 -> prob < 1e-14, excluded

template <typename T>
DEV_INLINE float ld(const T* p, int64_t i);
template <> DEV_INLINE float ld<__hip_bfloat16>(const __hip_bfloat16* p, int64_t i) { return bf2f(p[i]); }
template <> DEV_INLINE float ld<float>(const float* p, int64_t i) { return p[i]; }

template <typename T>
__global__ __launch_bounds__(1024)
void sample_kernel(const T* __restrict__ logits, int V, float inv_temp,
                   float top_p, int top_k, const int64_t* __restrict__ seeds,
                   const int64_t* __restrict__ step,
                   int64_t* __restrict__ out) {
  __shared__ float red[16];
  __shared__ float bin_p[NBINS];
  __shared__ int bin_c[NBINS];
  __shared__ int thr_bin_s;
  __shared__ float best_val_s[16];
  __shared__ int64_t best_idx_s[16];

  const int row = blockIdx.x;
  const T* lr = logits + (int64_t)row * V;
  // step counter read from device memory so a hipGraph replay of this
  // kernel draws fresh randomness every steps (the graph increments *step)
  const uint64_t seed = (uint64_t)seeds[row] ^ splitmix64(0x5D21u + (uint64_t)*step);
  constexpr bool BF16 = std::is_same<T, __hip_bfloat16>::value;
  const int nvec = BF16 ? V / 8 : 0;

  // ---- pass A: max (bf16: 16 B/lane vector loads) ----
  float m = -1e30f;
  if constexpr (BF16) {
    const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr);
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      bf16x8 v = l8[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, bf2f(v.v[j]));
    }
    for (int i = nvec * 8 + threadIdx.x; i < V; i += blockDim.x)
      m = fmaxf(m, ld(lr, i));
  } else {
    for (int i = threadIdx.x; i < V; i += blockDim.x)
      m = fmaxf(m, ld(lr, i));
  }
  m = block_max(m, red);

  const bool filtering = (top_p < 1.f) || (top_k > 0 && top_k < V);
  int thr_bin = NBINS - 1;
  if (filtering) {
    // ---- pass B: histogram ----
    for (int i = threadIdx.x; i < NBINS; i += blockDim.x) {
      bin_p[i] = 0.f;
      bin_c[i] = 0;
    }
    __syncthreads();
    if constexpr (BF16) {
      const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr);
      for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
        bf16x8 v = l8[i];
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          float z = (bf2f(v.v[j]) - m) * inv_temp;
          if (z > -ZRANGE) {
            int b = min(NBINS - 1, (int)(-z * (NBINS / ZRANGE)));
            atomicAdd(&bin_p[b], __expf(z));
            atomicAdd(&bin_c[b], 1);
          }
        }
      }
      for (int i = nvec * 8 + threadIdx.x; i < V; i += blockDim.x) {
        float z = (ld(lr, i) - m) * inv_temp;
        if (z > -ZRANGE) {
          int b = min(NBINS - 1, (int)(-z * (NBINS / ZRANGE)));
          atomicAdd(&bin_p[b], __expf(z));
          atomicAdd(&bin_c[b], 1);
        }
      }
    } else {
      for (int i = threadIdx.x; i < V; i += blockDim.x) {
        float z = (ld(lr, i) - m) * inv_temp;
        if (z > -ZRANGE) {
          int b = min(NBINS - 1, (int)(-z * (NBINS / ZRANGE)));
          atomicAdd(&bin_p[b], __expf(z));
          atomicAdd(&bin_c[b], 1);
        }
      }
    }
    __syncthreads();
    // ---- serial scan of 1024 bins by thread 0 (trivial cost) ----
    if (threadIdx.x == 0) {
      float total = 0.f;
      for (int b = 0; b < NBINS; ++b) total += bin_p[b];
      const float target = top_p * total;
      float cp = 0.f;
      int cc = 0;
      int bp = NBINS - 1, bk = NBINS - 1;
      bool done_p = (top_p >= 1.f), done_k = (top_k <= 0 || top_k >= V);
      for (int b = 0; b < NBINS && !(done_p && done_k); ++b) {
        cp += bin_p[b];
        cc += bin_c[b];
        if (!done_p && cp >= target) { bp = b; done_p = true; }
        if (!done_k && cc >= top_k) { bk = b; done_k = true; }
      }
      thr_bin_s = min(bp, bk);
    }
    __syncthreads();
    thr_bin = thr_bin_s;
  }
  const float z_min = -(thr_bin + 1) * (ZRANGE / NBINS);

  // ---- pass C: Gumbel-argmax over the kept set ----
  float best = -1e30f;
  int64_t best_i = 0;
  if constexpr (BF16) {
    const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr);
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
      bf16x8 v = l8[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float z = (bf2f(v.v[j]) - m) * inv_temp;
        if (z >= z_min) {
          float u = hash_uniform(seed, (uint64_t)(i * 8 + j));
          float key = z - __logf(-__logf(u));
          if (key > best) { best = key; best_i = i * 8 + j; }
        }
      }
    }
    for (int i = nvec * 8 + threadIdx.x; i < V; i += blockDim.x) {
      float z = (ld(lr, i) - m) * inv_temp;
      if (z >= z_min) {
        float u = hash_uniform(seed, (uint64_t)i);
        float key = z - __logf(-__logf(u));
        if (key > best) { best = key; best_i = i; }
      }
    }
  } else {
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
      float z = (ld(lr, i) - m) * inv_temp;
      if (z >= z_min) {
        float u = hash_uniform(seed, (uint64_t)i);
        float key = z - __logf(-__logf(u));
        if (key > best) { best = key; best_i = i; }
      }
    }
  }
  // wave reduce (val, idx)
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, WAVE);
    int64_t oi = __shfl_xor(best_i, off, WAVE);
    if (ov > best) { best = ov; best_i = oi; }
  }
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    best_val_s[wid] = best;
    best_idx_s[wid] = best_i;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w)
      if (best_val_s[w] > best_val_s[0]) {
        best_val_s[0] = best_val_s[w];
        best_idx_s[0] = best_idx_s[w];
      }
    out[row] = best_idx_s[0];
  }
}

}  // namespace

torch::Tensor sample_tokens(torch::Tensor logits, double temperature,
                            double top_p, int64_t top_k, torch::Tensor seeds,
                            torch::Tensor step) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2);
  TORCH_CHECK(temperature > 0.0, "temperature 0 is greedy: use argmax");
  TORCH_CHECK(seeds.scalar_type() == at::kLong);
  TORCH_CHECK(step.scalar_type() == at::kLong && step.is_cuda());
  const int B = logits.size(0), V = logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(at::kLong));
  if (B == 0) return out;
  dim3 grid(B), block(1024);
  auto stream = at::cuda::getCurrentCUDAStream();
  const float inv_t = 1.f / (float)temperature;
  if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(sample_kernel<__hip_bfloat16>, grid, block, 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
                       V, inv_t, (float)top_p, (int)top_k,
                       seeds.data_ptr<int64_t>(), step.data_ptr<int64_t>(),
                       out.data_ptr<int64_t>());
  } else {
    TORCH_CHECK(logits.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(sample_kernel<float>, grid, block, 0, stream,
                       logits.data_ptr<float>(), V, inv_t, (float)top_p,
                       (int)top_k, seeds.data_ptr<int64_t>(),
                       step.data_ptr<int64_t>(), out.data_ptr<int64_t>());
  }
  HIP_CHECK_LAST();
  return out;
}
