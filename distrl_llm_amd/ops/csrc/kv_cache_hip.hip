#include "hip/hip_runtime.h"
// Paged KV-cache scatter (the reference's vLLM reshape_and_cache,
// SURVEY.md §2.4-A): write the step's new K/V rows into their pool slots.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// k,v: (T, KV, D) bf16; caches: (num_blocks, block_size, KV, D);
// slot_mapping: (T,) int32 flat slot = block*block_size + offset, -1 skips.
__global__ void kv_scatter_kernel(const bf16x8* __restrict__ k,
                                  const bf16x8* __restrict__ v,
                                  bf16x8* __restrict__ key_cache,
                                  bf16x8* __restrict__ value_cache,
                                  const int* __restrict__ slot_mapping,
                                  int row_vecs /* KV*D/8 */) {
  const int t = blockIdx.x;
  const int slot = slot_mapping[t];
  if (slot < 0) return;
  const bf16x8* ks = k + (int64_t)t * row_vecs;
  const bf16x8* vs = v + (int64_t)t * row_vecs;
  bf16x8* kd = key_cache + (int64_t)slot * row_vecs;
  bf16x8* vd = value_cache + (int64_t)slot * row_vecs;
  for (int i = threadIdx.x; i < row_vecs; i += blockDim.x) {
    kd[i] = ks[i];
    vd[i] = vs[i];
  }
}

}  // namespace

void kv_cache_scatter(torch::Tensor k, torch::Tensor v,
                      torch::Tensor key_cache, torch::Tensor value_cache,
                      torch::Tensor slot_mapping) {
  TORCH_CHECK(k.is_cuda() && k.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(key_cache.is_contiguous() && value_cache.is_contiguous());
  TORCH_CHECK(k.scalar_type() == at::kBFloat16);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kInt);
  const int T = k.size(0);
  if (T == 0) return;
  const int KV = k.size(1), D = k.size(2);
  TORCH_CHECK((KV * D) % 8 == 0);
  const int row_vecs = KV * D / 8;
  hipLaunchKernelGGL(kv_scatter_kernel, dim3(T), dim3(128), 0,
                     at::hip::getCurrentHIPStreamMasqueradingAsCUDA(),
                     reinterpret_cast<const bf16x8*>(k.data_ptr()),
                     reinterpret_cast<const bf16x8*>(v.data_ptr()),
                     reinterpret_cast<bf16x8*>(key_cache.data_ptr()),
                     reinterpret_cast<bf16x8*>(value_cache.data_ptr()),
                     slot_mapping.data_ptr<int>(), row_vecs);
  HIP_CHECK_LAST();
}
