// Two-stage fused sampler (ROADMAP #2): the single-workgroup-per-row
// sampler under-fills the chip at decode batch sizes (160 workgroups on
// 256 CUs; measured 363 us/step) and serializes three 304 KB passes per
// row. Here every pass is split over SLICES of the vocab (grid B x S),
// so the chip is full and each pass runs at streaming rate; the tiny
// combine steps are their own kernels. Sampling semantics and the
// counter-based RNG hash are IDENTICAL to sampling.hip's single-kernel
// version (same bins vs the global max, same threshold rule, same
// Gumbel keys per (seed, element)), so a draw is the same distribution —
// and stays hipGraph-replayable (the step counter is read from device
// memory).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

constexpr int NBINS2 = 1024;
constexpr float ZRANGE2 = 32.f;
constexpr int SLICES = 8;

template <typename T>
DEV_INLINE float ld2(const T* p, int64_t i);
template <> DEV_INLINE float ld2<__hip_bfloat16>(const __hip_bfloat16* p, int64_t i) { return bf2f(p[i]); }
template <> DEV_INLINE float ld2<float>(const float* p, int64_t i) { return p[i]; }

// K1: per-slice max
template <typename T>
__global__ __launch_bounds__(256)
void smax_kernel(const T* __restrict__ logits, int V,
                 float* __restrict__ ms) {
  __shared__ float red[16];
  const int row = blockIdx.x, sl = blockIdx.y;
  const T* lr = logits + (int64_t)row * V;
  const int per = (V + SLICES - 1) / SLICES;
  const int i0 = sl * per, i1 = min(V, i0 + per);
  float m = -1e30f;
  if constexpr (std::is_same<T, __hip_bfloat16>::value) {
    const int nv = (i1 - i0) / 8;
    const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr + i0);
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      bf16x8 v = l8[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, bf2f(v.v[j]));
    }
    for (int i = i0 + nv * 8 + threadIdx.x; i < i1; i += blockDim.x)
      m = fmaxf(m, ld2(lr, i));
  } else {
    for (int i = i0 + threadIdx.x; i < i1; i += blockDim.x)
      m = fmaxf(m, ld2(lr, i));
  }
  m = block_max(m, red);
  if (threadIdx.x == 0) ms[row * SLICES + sl] = m;
}

// K2: global max per row
__global__ __launch_bounds__(64)
void gmax_kernel(const float* __restrict__ ms, float* __restrict__ gmax,
                 int B) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row < B) {
    float m = -1e30f;
    #pragma unroll
    for (int s = 0; s < SLICES; ++s) m = fmaxf(m, ms[row * SLICES + s]);
    gmax[row] = m;
  }
}

// K3: per-slice LDS histogram vs the global max, accumulated into the
// row's global histogram (1024 global atomics per slice, not 152k)
template <typename T>
__global__ __launch_bounds__(256)
void hist_kernel(const T* __restrict__ logits, int V, float inv_temp,
                 const float* __restrict__ gmax,
                 float* __restrict__ hist_p, int* __restrict__ hist_c) {
  __shared__ float bin_p[NBINS2];
  __shared__ int bin_c[NBINS2];
  const int row = blockIdx.x, sl = blockIdx.y;
  const T* lr = logits + (int64_t)row * V;
  const float m = gmax[row];
  for (int i = threadIdx.x; i < NBINS2; i += blockDim.x) {
    bin_p[i] = 0.f;
    bin_c[i] = 0;
  }
  __syncthreads();
  const int per = (V + SLICES - 1) / SLICES;
  const int i0 = sl * per, i1 = min(V, i0 + per);
  if constexpr (std::is_same<T, __hip_bfloat16>::value) {
    const int nv = (i1 - i0) / 8;
    const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr + i0);
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      bf16x8 v = l8[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float z = (bf2f(v.v[j]) - m) * inv_temp;
        if (z > -ZRANGE2) {
          int b = min(NBINS2 - 1, (int)(-z * (NBINS2 / ZRANGE2)));
          atomicAdd(&bin_p[b], __expf(z));
          atomicAdd(&bin_c[b], 1);
        }
      }
    }
    for (int i = i0 + nv * 8 + threadIdx.x; i < i1; i += blockDim.x) {
      float z = (ld2(lr, i) - m) * inv_temp;
      if (z > -ZRANGE2) {
        int b = min(NBINS2 - 1, (int)(-z * (NBINS2 / ZRANGE2)));
        atomicAdd(&bin_p[b], __expf(z));
        atomicAdd(&bin_c[b], 1);
      }
    }
  } else {
    for (int i = i0 + threadIdx.x; i < i1; i += blockDim.x) {
      float z = (ld2(lr, i) - m) * inv_temp;
      if (z > -ZRANGE2) {
        int b = min(NBINS2 - 1, (int)(-z * (NBINS2 / ZRANGE2)));
        atomicAdd(&bin_p[b], __expf(z));
        atomicAdd(&bin_c[b], 1);
      }
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < NBINS2; b += blockDim.x) {
    if (bin_p[b] != 0.f) atomicAdd(&hist_p[(int64_t)row * NBINS2 + b], bin_p[b]);
    if (bin_c[b] != 0) atomicAdd(&hist_c[(int64_t)row * NBINS2 + b], bin_c[b]);
  }
}

// K4: threshold scan per row -> z_min (same rule as the single-kernel
// sampler: keep whole bins above the bin where cum-p/cum-k crosses)
__global__ __launch_bounds__(64)
void thr_kernel(const float* __restrict__ hist_p,
                const int* __restrict__ hist_c, int V, float top_p,
                int top_k, float* __restrict__ zmin, int B) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= B) return;
  const float* hp = hist_p + (int64_t)row * NBINS2;
  const int* hc = hist_c + (int64_t)row * NBINS2;
  float total = 0.f;
  for (int b = 0; b < NBINS2; ++b) total += hp[b];
  const float target = top_p * total;
  float cp = 0.f;
  int cc = 0;
  int bp = NBINS2 - 1, bk = NBINS2 - 1;
  bool done_p = (top_p >= 1.f), done_k = (top_k <= 0 || top_k >= V);
  for (int b = 0; b < NBINS2 && !(done_p && done_k); ++b) {
    cp += hp[b];
    cc += hc[b];
    if (!done_p && cp >= target) { bp = b; done_p = true; }
    if (!done_k && cc >= top_k) { bk = b; done_k = true; }
  }
  const int thr = min(bp, bk);
  zmin[row] = -(thr + 1) * (ZRANGE2 / NBINS2);
}

// K5: per-slice Gumbel-argmax over the kept set
template <typename T>
__global__ __launch_bounds__(256)
void gumbel_kernel(const T* __restrict__ logits, int V, float inv_temp,
                   const float* __restrict__ gmax,
                   const float* __restrict__ zmin,
                   const int64_t* __restrict__ seeds,
                   const int64_t* __restrict__ step,
                   float* __restrict__ pb_val, int64_t* __restrict__ pb_idx) {
  __shared__ float red_v[4];
  __shared__ int64_t red_i[4];
  const int row = blockIdx.x, sl = blockIdx.y;
  const T* lr = logits + (int64_t)row * V;
  const float m = gmax[row];
  const float zm = zmin[row];
  const uint64_t seed = (uint64_t)seeds[row]
                        ^ splitmix64(0x5D21u + (uint64_t)*step);
  const int per = (V + SLICES - 1) / SLICES;
  const int i0 = sl * per, i1 = min(V, i0 + per);
  float best = -1e30f;
  int64_t best_i = 0;
  if constexpr (std::is_same<T, __hip_bfloat16>::value) {
    const int nv = (i1 - i0) / 8;
    const bf16x8* l8 = reinterpret_cast<const bf16x8*>(lr + i0);
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      bf16x8 v = l8[i];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float z = (bf2f(v.v[j]) - m) * inv_temp;
        if (z >= zm) {
          const int64_t gi = i0 + i * 8 + j;
          float u = hash_uniform(seed, (uint64_t)gi);
          float key = z - __logf(-__logf(u));
          if (key > best) { best = key; best_i = gi; }
        }
      }
    }
    for (int i = i0 + nv * 8 + threadIdx.x; i < i1; i += blockDim.x) {
      float z = (ld2(lr, i) - m) * inv_temp;
      if (z >= zm) {
        float u = hash_uniform(seed, (uint64_t)i);
        float key = z - __logf(-__logf(u));
        if (key > best) { best = key; best_i = i; }
      }
    }
  } else {
    for (int i = i0 + threadIdx.x; i < i1; i += blockDim.x) {
      float z = (ld2(lr, i) - m) * inv_temp;
      if (z >= zm) {
        float u = hash_uniform(seed, (uint64_t)i);
        float key = z - __logf(-__logf(u));
        if (key > best) { best = key; best_i = i; }
      }
    }
  }
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, WAVE);
    int64_t oi = __shfl_xor(best_i, off, WAVE);
    if (ov > best) { best = ov; best_i = oi; }
  }
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    red_v[wid] = best;
    red_i[wid] = best_i;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w)
      if (red_v[w] > red_v[0]) { red_v[0] = red_v[w]; red_i[0] = red_i[w]; }
    pb_val[row * SLICES + sl] = red_v[0];
    pb_idx[row * SLICES + sl] = red_i[0];
  }
}

// K6: combine slice winners
__global__ __launch_bounds__(64)
void combine_kernel(const float* __restrict__ pb_val,
                    const int64_t* __restrict__ pb_idx,
                    int64_t* __restrict__ out, int B) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row < B) {
    float best = -1e30f;
    int64_t bi = 0;
    #pragma unroll
    for (int s = 0; s < SLICES; ++s) {
      const float v = pb_val[row * SLICES + s];
      if (v > best) { best = v; bi = pb_idx[row * SLICES + s]; }
    }
    out[row] = bi;
  }
}

}  // namespace

torch::Tensor sample_tokens2(torch::Tensor logits, double temperature,
                             double top_p, int64_t top_k, torch::Tensor seeds,
                             torch::Tensor step) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2);
  TORCH_CHECK(temperature > 0.0);
  const int B = logits.size(0), V = logits.size(1);
  auto opts_f = logits.options().dtype(at::kFloat);
  auto out = torch::empty({B}, logits.options().dtype(at::kLong));
  if (B == 0) return out;
  auto ms = torch::empty({B, SLICES}, opts_f);
  auto gmax = torch::empty({B}, opts_f);
  auto zmin = torch::empty({B}, opts_f);
  auto pb_val = torch::empty({B, SLICES}, opts_f);
  auto pb_idx = torch::empty({B, SLICES}, logits.options().dtype(at::kLong));
  const bool filtering = (top_p < 1.0) || (top_k > 0 && top_k < V);
  torch::Tensor hist_p, hist_c;
  auto stream = at::cuda::getCurrentCUDAStream();
  const float inv_t = 1.f / (float)temperature;
  dim3 gs(B, SLICES), bs(256), g1((B + 63) / 64), b1(64);

  #define SAMPLE2_BODY(T, PTR)                                              \
    hipLaunchKernelGGL(smax_kernel<T>, gs, bs, 0, stream, PTR, V,           \
                       ms.data_ptr<float>());                               \
    hipLaunchKernelGGL(gmax_kernel, g1, b1, 0, stream,                      \
                       ms.data_ptr<float>(), gmax.data_ptr<float>(), B);    \
    if (filtering) {                                                        \
      hist_p = torch::zeros({B, NBINS2}, opts_f);                           \
      hist_c = torch::zeros({B, NBINS2},                                    \
                            logits.options().dtype(at::kInt));              \
      hipLaunchKernelGGL(hist_kernel<T>, gs, bs, 0, stream, PTR, V, inv_t,  \
                         gmax.data_ptr<float>(), hist_p.data_ptr<float>(),  \
                         hist_c.data_ptr<int>());                           \
      hipLaunchKernelGGL(thr_kernel, g1, b1, 0, stream,                     \
                         hist_p.data_ptr<float>(), hist_c.data_ptr<int>(),  \
                         V, (float)top_p, (int)top_k,                       \
                         zmin.data_ptr<float>(), B);                        \
    } else {                                                                \
      zmin.fill_(-ZRANGE2);                                                 \
    }                                                                       \
    hipLaunchKernelGGL(gumbel_kernel<T>, gs, bs, 0, stream, PTR, V, inv_t,  \
                       gmax.data_ptr<float>(), zmin.data_ptr<float>(),      \
                       seeds.data_ptr<int64_t>(), step.data_ptr<int64_t>(), \
                       pb_val.data_ptr<float>(),                            \
                       pb_idx.data_ptr<int64_t>());                         \
    hipLaunchKernelGGL(combine_kernel, g1, b1, 0, stream,                   \
                       pb_val.data_ptr<float>(),                            \
                       pb_idx.data_ptr<int64_t>(),                          \
                       out.data_ptr<int64_t>(), B)

  if (logits.scalar_type() == at::kBFloat16) {
    SAMPLE2_BODY(__hip_bfloat16,
                 reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()));
  } else {
    TORCH_CHECK(logits.scalar_type() == at::kFloat);
    SAMPLE2_BODY(float, logits.data_ptr<float>());
  }
  #undef SAMPLE2_BODY
  HIP_CHECK_LAST();
  return out;
}
