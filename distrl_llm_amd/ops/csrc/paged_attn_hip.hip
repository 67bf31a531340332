#include "hip/hip_runtime.h"
// Paged-attention decode (GQA) over the block-paged KV pool — the decode
// hot loop of the generation engine (reference counterpart: vLLM
// paged_attention_v1/v2), written CDNA4-first:
//
//   grid = (num_seqs, n_kv_heads); one workgroup owns one (sequence,
//   kv-head) pair and its whole group of query heads (GQA 7:1 on
//   Qwen2.5-7B). 4 waves / 256 threads.
//   Phase 1: 16-lane thread groups stream K rows (one bf16x8 = 16 B per
//   lane), dot against the group's Q (staged in LDS as fp32), shfl-reduce
//   within the 16-lane group, scores -> LDS.
//   Phase 2: per-head softmax (wave-strided max/sum over the LDS scores).
//   Phase 3: V streamed through LDS in 32-token tiles, each thread
//   accumulates one (head, 8-dim) slice in registers, epilogue divides by
//   the softmax denominator and stores bf16x8.
//
// K and V are each read exactly once from HBM per step; the kernel is
// KV-bandwidth-bound as it should be.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int TPG = 16;          // threads per K-token group (D<=128: 8 B..16 B/lane)
constexpr int VTILE = 32;        // V tokens staged per LDS tile

template <int D>
__global__ __launch_bounds__(256)
void paged_decode_kernel(const __hip_bfloat16* __restrict__ q,      // (N,H,D)
                         const __hip_bfloat16* __restrict__ kcache, // (nb,bs,KV,D)
                         const __hip_bfloat16* __restrict__ vcache,
                         const int* __restrict__ block_tables,      // (N,max_nb)
                         const int* __restrict__ ctx_lens,          // (N,)
                         __hip_bfloat16* __restrict__ out,          // (N,H,D)
                         int H, int KV, int max_nb, int block_size,
                         int Lpad, float scale) {
  const int seq = blockIdx.x;
  const int kv = blockIdx.y;
  const int group = H / KV;
  const int L = ctx_lens[seq];
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* q_lds = reinterpret_cast<float*>(smem_raw);          // group*D
  float* denom = q_lds + group * D;                           // group
  float* scores = denom + ((group + 3) & ~3);                 // group*Lpad
  __hip_bfloat16* v_lds = reinterpret_cast<__hip_bfloat16*>(
      scores + (size_t)group * Lpad);                         // VTILE*D

  const int* bt = block_tables + (int64_t)seq * max_nb;
  const int64_t kv_row = (int64_t)KV * D;

  // ---- stage Q (group heads of this kv head) into LDS as fp32 ----
  for (int i = tid; i < group * D; i += blockDim.x) {
    const int h = kv * group + i / D;
    q_lds[i] = bf2f(q[((int64_t)seq * H + h) * D + i % D]);
  }
  __syncthreads();

  // ---- phase 1: scores ----
  const int gid = tid / TPG;            // 16 token-groups in flight
  const int lane = tid % TPG;
  constexpr int EPT = D / TPG;          // elems per thread (8 for D=128)
  for (int t = gid; t < L; t += blockDim.x / TPG) {
    const int64_t row = (int64_t)bt[t / block_size] * block_size
                        + t % block_size;
    const __hip_bfloat16* kp = kcache + row * kv_row + (int64_t)kv * D
                               + lane * EPT;
    float kf[EPT];
    if constexpr (EPT == 8) {
      bf16x8 kvec = *reinterpret_cast<const bf16x8*>(kp);
      #pragma unroll
      for (int j = 0; j < 8; ++j) kf[j] = bf2f(kvec.v[j]);
    } else {
      #pragma unroll
      for (int j = 0; j < EPT; ++j) kf[j] = bf2f(kp[j]);
    }
    for (int h = 0; h < group; ++h) {
      const float* qh = q_lds + h * D + lane * EPT;
      float p = 0.f;
      #pragma unroll
      for (int j = 0; j < EPT; ++j) p += qh[j] * kf[j];
      #pragma unroll
      for (int off = TPG / 2; off > 0; off >>= 1) p += __shfl_xor(p, off, WAVE);
      if (lane == 0) scores[(size_t)h * Lpad + t] = p * scale;
    }
  }
  __syncthreads();

  // ---- phase 2: per-head softmax over the LDS scores ----
  const int wid = tid / WAVE;
  const int wlane = tid % WAVE;
  const int nw = blockDim.x / WAVE;
  for (int h = wid; h < group; h += nw) {
    float* s = scores + (size_t)h * Lpad;
    float m = -INFINITY;
    for (int t = wlane; t < L; t += WAVE) m = fmaxf(m, s[t]);
    m = wave_max(m);
    float d = 0.f;
    for (int t = wlane; t < L; t += WAVE) {
      float e = __expf(s[t] - m);
      s[t] = e;
      d += e;
    }
    d = wave_sum(d);
    if (wlane == 0) denom[h] = d;
  }
  __syncthreads();

  // ---- phase 3: PV accumulation through LDS V tiles ----
  constexpr int DV = D / 8;             // bf16x8 units per row
  const int units = group * DV;         // <= 8*16 = 128
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const int u = tid;                    // one unit per thread (tid < units)
  const int uh = u / DV, ud = u % DV;

  for (int base = 0; base < L; base += VTILE) {
    const int tile = min(VTILE, L - base);
    // stage V rows [base, base+tile) for this kv head
    for (int i = tid; i < tile * DV; i += blockDim.x) {
      const int tt = base + i / DV;
      const int64_t row = (int64_t)bt[tt / block_size] * block_size
                          + tt % block_size;
      reinterpret_cast<bf16x8*>(v_lds)[i] =
          *reinterpret_cast<const bf16x8*>(
              vcache + row * kv_row + (int64_t)kv * D + (i % DV) * 8);
    }
    __syncthreads();
    if (u < units) {
      const float* ps = scores + (size_t)uh * Lpad + base;
      #pragma unroll 4
      for (int j = 0; j < tile; ++j) {
        const float p = ps[j];
        const bf16x8 vv = reinterpret_cast<const bf16x8*>(v_lds)[j * DV + ud];
        #pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += p * bf2f(vv.v[e]);
      }
    }
    __syncthreads();
  }

  if (u < units) {
    const float inv = 1.f / denom[uh];
    const int h = kv * group + uh;
    bf16x8 o;
    #pragma unroll
    for (int e = 0; e < 8; ++e) o.v[e] = f2bf(acc[e] * inv);
    *reinterpret_cast<bf16x8*>(out + ((int64_t)seq * H + h) * D + ud * 8) = o;
  }
}

}  // namespace

torch::Tensor paged_attention_decode(torch::Tensor q, torch::Tensor kcache,
                                     torch::Tensor vcache,
                                     torch::Tensor block_tables,
                                     torch::Tensor ctx_lens, double scale) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt &&
              ctx_lens.scalar_type() == at::kInt);
  const int N = q.size(0), H = q.size(1), D = q.size(2);
  const int KV = kcache.size(2);
  const int block_size = kcache.size(1);
  const int max_nb = block_tables.size(1);
  const int group = H / KV;
  TORCH_CHECK(H % KV == 0 && group <= 16, "GQA group too large");

  auto out = torch::empty_like(q);
  if (N == 0) return out;

  const int max_ctx = max_nb * block_size;
  const int Lpad = max_ctx + 4;  // +pad to stagger LDS banks across heads
  size_t smem = (size_t)group * D * 4 + ((group + 3) & ~3) * 4
                + (size_t)group * Lpad * 4 + (size_t)VTILE * D * 2;
  TORCH_CHECK(smem <= 160 * 1024,
              "context too long for single-pass decode kernel: ", max_ctx);

  dim3 grid(N, KV), block(256);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (D == 128) {
    hipLaunchKernelGGL(paged_decode_kernel<128>, grid, block, smem, stream,
                       reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(kcache.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(vcache.data_ptr()),
                       block_tables.data_ptr<int>(), ctx_lens.data_ptr<int>(),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       H, KV, max_nb, block_size, Lpad, (float)scale);
  } else if (D == 64) {
    hipLaunchKernelGGL(paged_decode_kernel<64>, grid, block, smem, stream,
                       reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(kcache.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(vcache.data_ptr()),
                       block_tables.data_ptr<int>(), ctx_lens.data_ptr<int>(),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       H, KV, max_nb, block_size, Lpad, (float)scale);
  } else {
    TORCH_CHECK(false, "paged decode: head_dim 64 or 128 only, got ", D);
  }
  HIP_CHECK_LAST();
  return out;
}
