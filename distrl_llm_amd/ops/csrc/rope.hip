// Fused in-place RoPE for the generation path (rotate_half / neox
// convention, matching HF Qwen2/Llama). Replaces vLLM's pos_encoding
// kernels. cos/sin computed once per token into LDS and
// reused across all q + kv heads.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

__global__ void rope_inplace_kernel(__hip_bfloat16* __restrict__ q,
                                    __hip_bfloat16* __restrict__ k,
                                    const int* __restrict__ positions,
                                    const float* __restrict__ inv_freq,
                                    int H, int KV, int D) {
  extern __shared__ float smem[];  // [D/2 cos][D/2 sin]
  const int half = D / 2;
  float* cs = smem;
  float* sn = smem + half;
  const int t = blockIdx.x;
  const float pos = (float)positions[t];
  for (int i = threadIdx.x; i < half; i += blockDim.x) {
    float a = pos * inv_freq[i];
    sn[i] = __sinf(a);
    cs[i] = __cosf(a);
  }
  __syncthreads();

  const int total = (H + KV) * half;
  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int head = idx / half;
    const int i = idx % half;
    __hip_bfloat16* ptr = (head < H)
        ? q + ((int64_t)t * H + head) * D
        : k + ((int64_t)t * KV + (head - H)) * D;
    float x1 = bf2f(ptr[i]);
    float x2 = bf2f(ptr[i + half]);
    float c = cs[i], s = sn[i];
    ptr[i] = f2bf(x1 * c - x2 * s);
    ptr[i + half] = f2bf(x2 * c + x1 * s);
  }
}

// ---- fused packed-QKV RoPE + paged KV scatter (decode fast path) ----
// qkv: (T, H*D + 2*KV*D) packed rows straight out of the fused qkv GEMM.
// Rotates q and k in place, then scatters the (rotated) k and v rows into
// the paged pool — one kernel replaces rope + 2 contiguous-copies +
// kv_cache_scatter.
__global__ void rope_scatter_qkv_kernel(__hip_bfloat16* __restrict__ qkv,
                                        const int* __restrict__ positions,
                                        const int64_t* __restrict__ slots,
                                        const float* __restrict__ inv_freq,
                                        __hip_bfloat16* __restrict__ key_cache,
                                        __hip_bfloat16* __restrict__ value_cache,
                                        int H, int KV, int D) {
  extern __shared__ float smem[];  // [D/2 cos][D/2 sin]
  const int half = D / 2;
  float* cs = smem;
  float* sn = smem + half;
  const int t = blockIdx.x;
  const int row_stride = (H + 2 * KV) * D;
  const float pos = (float)positions[t];
  for (int i = threadIdx.x; i < half; i += blockDim.x) {
    float a = pos * inv_freq[i];
    sn[i] = __sinf(a);
    cs[i] = __cosf(a);
  }
  __syncthreads();

  __hip_bfloat16* base = qkv + (int64_t)t * row_stride;
  const int total = (H + KV) * half;  // rotate q heads then k heads
  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int head = idx / half;
    const int i = idx % half;
    __hip_bfloat16* ptr = base + head * D;  // q heads then k heads: packed
    float x1 = bf2f(ptr[i]);
    float x2 = bf2f(ptr[i + half]);
    float c = cs[i], s = sn[i];
    ptr[i] = f2bf(x1 * c - x2 * s);
    ptr[i + half] = f2bf(x2 * c + x1 * s);
  }
  __syncthreads();

  // scatter rotated k and raw v into the pool
  const int64_t slot = slots[t];
  if (slot < 0) return;
  const int row_vecs = KV * D / 8;
  const bf16x8* ks = reinterpret_cast<const bf16x8*>(base + H * D);
  const bf16x8* vs = reinterpret_cast<const bf16x8*>(base + (H + KV) * D);
  bf16x8* kd = reinterpret_cast<bf16x8*>(key_cache) + slot * row_vecs;
  bf16x8* vd = reinterpret_cast<bf16x8*>(value_cache) + slot * row_vecs;
  for (int i = threadIdx.x; i < row_vecs; i += blockDim.x) {
    kd[i] = ks[i];
    vd[i] = vs[i];
  }
}

}  // namespace

void rope_scatter_qkv(torch::Tensor qkv, torch::Tensor positions,
                      torch::Tensor slots, torch::Tensor inv_freq,
                      torch::Tensor key_cache, torch::Tensor value_cache,
                      int64_t H, int64_t KV, int64_t D) {
  TORCH_CHECK(qkv.is_cuda() && qkv.is_contiguous());
  TORCH_CHECK(qkv.scalar_type() == at::kBFloat16);
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  TORCH_CHECK(slots.scalar_type() == at::kLong);
  const int T = qkv.size(0);
  TORCH_CHECK(qkv.size(1) == (H + 2 * KV) * D);
  TORCH_CHECK((KV * D) % 8 == 0);
  if (T == 0) return;
  const int smem = D * sizeof(float);
  hipLaunchKernelGGL(rope_scatter_qkv_kernel, dim3(T), dim3(256), smem,
                     at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<__hip_bfloat16*>(qkv.data_ptr()),
                     positions.data_ptr<int>(), slots.data_ptr<int64_t>(),
                     inv_freq.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(key_cache.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(value_cache.data_ptr()),
                     (int)H, (int)KV, (int)D);
  HIP_CHECK_LAST();
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor inv_freq) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(positions.scalar_type() == at::kInt);
  const int T = q.size(0), H = q.size(1), D = q.size(2);
  const int KV = k.size(1);
  TORCH_CHECK(k.size(0) == T && k.size(2) == D);
  if (T == 0) return;
  const int smem = D * sizeof(float);
  hipLaunchKernelGGL(rope_inplace_kernel, dim3(T), dim3(256), smem,
                     at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<__hip_bfloat16*>(q.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(k.data_ptr()),
                     positions.data_ptr<int>(), inv_freq.data_ptr<float>(),
                     H, KV, D);
  HIP_CHECK_LAST();
}
