// Paged-attention decode (GQA) over the block-paged KV pool — the decode
// hot loop of the generation engine (reference counterpart: vLLM
// paged_attention_v1/v2), written CDNA4-first:
//
//   grid = (num_seqs, n_kv_heads, context_slices); one workgroup owns one
//   (sequence, kv-head, context-slice) triple and the kv-head's whole
//   group of query heads (GQA 7:1 on Qwen2.5-7B). 4 waves / 256 threads.
//   The flash-decoding context split (chosen to fill the chip at decode
//   batch sizes) emits unnormalized fp32 partials + (m, l) per slice,
//   combined by paged_decode_combine_kernel.
//   Phase 1 (scores): each LANE owns one context token — it streams that
//   token's K row with bf16x8 loads and keeps one fp32 partial per query
//   head in registers (GROUP is a template param so the per-head array
//   stays in VGPRs); Q is staged in LDS and read as wave-uniform
//   broadcasts. No cross-lane reduction at all.
//   Phase 2: per-head softmax (wave-strided max/sum over the LDS scores).
//   Phase 3 (PV): V streamed through LDS in 32-token tiles; each thread
//   accumulates one (head, 8-dim) slice in registers; epilogue divides by
//   the softmax denominator and stores bf16x8.
//
// K and V are each read exactly once from HBM per step; the kernel is
// KV-bandwidth-bound as it should be.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

constexpr int VTILE = 64;        // V tokens staged per LDS tile

template <int D, int GROUP>
__global__ __launch_bounds__(256)
void paged_decode_kernel(const __hip_bfloat16* __restrict__ q,      // (N,H,D)
                         const __hip_bfloat16* __restrict__ kcache, // (nb,bs,KV,D)
                         const __hip_bfloat16* __restrict__ vcache,
                         const int* __restrict__ block_tables,      // (N,max_nb)
                         const int* __restrict__ ctx_lens,          // (N,)
                         __hip_bfloat16* __restrict__ out,          // (N,H,D)
                         float* __restrict__ o_part,  // (CS,N,H,D) | null
                         float* __restrict__ ml_part, // (CS,N,H,2) | null
                         int H, int KV, int max_nb, int block_size,
                         int Lpad, float scale, int64_t q_row_stride,
                         int slice_len) {
  const int seq = blockIdx.x;
  const int kv = blockIdx.y;
  const int zid = blockIdx.z;     // context slice (flash-decoding split)
  const int Lall = ctx_lens[seq];
  const int c0 = zid * slice_len;
  const int L = min(Lall, c0 + slice_len);  // this slice: [c0, L)
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* q_lds = reinterpret_cast<float*>(smem_raw);          // GROUP*D
  float* denom = q_lds + GROUP * D;                           // padded to 4
  float* gmax = denom + ((GROUP + 3) & ~3);                   // slice maxes
  float* scores = gmax + ((GROUP + 3) & ~3);                  // GROUP*Lpad
  __hip_bfloat16* v_lds = reinterpret_cast<__hip_bfloat16*>(
      scores + (size_t)GROUP * Lpad);                         // VTILE*D

  const int* bt = block_tables + (int64_t)seq * max_nb;
  const int64_t kv_row = (int64_t)KV * D;

  // ---- stage Q (group heads of this kv head) into LDS as fp32 ----
  for (int i = tid; i < GROUP * D; i += blockDim.x) {
    const int h = kv * GROUP + i / D;
    q_lds[i] = bf2f(q[(int64_t)seq * q_row_stride + h * D + i % D]) * scale;
  }
  __syncthreads();

  // ---- phase 1: one token per lane, per-head partials in VGPRs ----
  for (int t = c0 + tid; t < L; t += blockDim.x) {
    const int64_t row = (int64_t)bt[t / block_size] * block_size
                        + t % block_size;
    const bf16x8* kp = reinterpret_cast<const bf16x8*>(
        kcache + row * kv_row + (int64_t)kv * D);
    float part[GROUP];
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) part[h] = 0.f;
    #pragma unroll
    for (int c = 0; c < D / 8; ++c) {
      const bf16x8 kvec = kp[c];
      float kf[8];
      #pragma unroll
      for (int j = 0; j < 8; ++j) kf[j] = bf2f(kvec.v[j]);
      #pragma unroll
      for (int h = 0; h < GROUP; ++h) {
        const float* qh = q_lds + h * D + c * 8;   // wave-uniform broadcast
        #pragma unroll
        for (int j = 0; j < 8; ++j) part[h] += qh[j] * kf[j];
      }
    }
    #pragma unroll
    for (int h = 0; h < GROUP; ++h) scores[(size_t)h * Lpad + (t - c0)] = part[h];
  }
  __syncthreads();

  // ---- phase 2: per-head softmax over the LDS scores ----
  const int wid = tid / WAVE;
  const int wlane = tid % WAVE;
  const int nw = blockDim.x / WAVE;
  const int Ls = L - c0;  // tokens in this slice (may be <= 0)
  for (int h = wid; h < GROUP; h += nw) {
    float* s = scores + (size_t)h * Lpad;
    float m = -1e30f;
    for (int t = wlane; t < Ls; t += WAVE) m = fmaxf(m, s[t]);
    m = wave_max(m);
    float d = 0.f;
    for (int t = wlane; t < Ls; t += WAVE) {
      float e = __expf(s[t] - m);
      s[t] = e;
      d += e;
    }
    d = wave_sum(d);
    if (wlane == 0) {
      denom[h] = d;
      gmax[h] = m;
    }
  }
  __syncthreads();

  // ---- phase 3: PV accumulation through LDS V tiles ----
  constexpr int DV = D / 8;             // bf16x8 units per row
  constexpr int UNITS = GROUP * DV;     // <= 16*16 = 256
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const int u = tid;                    // one unit per thread (tid < UNITS)
  const int uh = u / DV, ud = u % DV;

  for (int base = 0; base < Ls; base += VTILE) {
    const int tile = min(VTILE, Ls - base);
    // stage V rows [base, base+tile) for this kv head
    for (int i = tid; i < tile * DV; i += blockDim.x) {
      const int tt = c0 + base + i / DV;
      const int64_t row = (int64_t)bt[tt / block_size] * block_size
                          + tt % block_size;
      reinterpret_cast<bf16x8*>(v_lds)[i] =
          *reinterpret_cast<const bf16x8*>(
              vcache + row * kv_row + (int64_t)kv * D + (i % DV) * 8);
    }
    __syncthreads();
    if (u < UNITS) {
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

  if (u < UNITS) {
    const int h = kv * GROUP + uh;
    if (o_part == nullptr) {
      const float inv = 1.f / denom[uh];
      bf16x8 o;
      #pragma unroll
      for (int e = 0; e < 8; ++e) o.v[e] = f2bf(acc[e] * inv);
      *reinterpret_cast<bf16x8*>(out + ((int64_t)seq * H + h) * D + ud * 8) = o;
    } else {
      // flash-decoding partial: unnormalized o + (m, l) per slice
      const int64_t oz = (((int64_t)zid * gridDim.x + seq) * H + h) * D + ud * 8;
      #pragma unroll
      for (int e = 0; e < 8; ++e) o_part[oz + e] = acc[e];
      if (ud == 0) {
        const int64_t mz = (((int64_t)zid * gridDim.x + seq) * H + h) * 2;
        ml_part[mz] = (Ls > 0) ? gmax[uh] : -1e30f;
        ml_part[mz + 1] = (Ls > 0) ? denom[uh] : 0.f;
      }
    }
  }
}

// flash-decoding combine: out[seq,h] = sum_z w_z*o_z / sum_z w_z*l_z,
// w_z = exp(m_z - max_z m_z). grid (N, H), D/64 elems per lane.
template <int D>
__global__ __launch_bounds__(64)
void paged_decode_combine_kernel(const float* __restrict__ o_part,
                                 const float* __restrict__ ml_part,
                                 __hip_bfloat16* __restrict__ out,
                                 int H, int cs) {
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int N = gridDim.x;
  const int lane = threadIdx.x;
  float M = -1e30f;
  for (int z = 0; z < cs; ++z)
    M = fmaxf(M, ml_part[(((int64_t)z * N + seq) * H + h) * 2]);
  float Lsum = 0.f;
  float o[D / 64];
  #pragma unroll
  for (int e = 0; e < D / 64; ++e) o[e] = 0.f;
  for (int z = 0; z < cs; ++z) {
    const int64_t mz = (((int64_t)z * N + seq) * H + h) * 2;
    const float w = __expf(ml_part[mz] - M);
    Lsum += w * ml_part[mz + 1];
    const int64_t oz = (((int64_t)z * N + seq) * H + h) * D;
    #pragma unroll
    for (int e = 0; e < D / 64; ++e)
      o[e] += w * o_part[oz + lane + e * 64];
  }
  const float inv = 1.f / Lsum;
  #pragma unroll
  for (int e = 0; e < D / 64; ++e)
    out[((int64_t)seq * H + h) * D + lane + e * 64] = f2bf(o[e] * inv);
}

template <int D, int GROUP>
void launch(const torch::Tensor& q, const torch::Tensor& kcache,
            const torch::Tensor& vcache, const torch::Tensor& bt,
            const torch::Tensor& ctx, torch::Tensor& out, int H, int KV,
            int max_nb, int block_size, int Lpad, float scale, size_t smem,
            int64_t q_row_stride, int cs, int slice_len) {
  const int N = q.size(0);
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 grid(N, KV, cs), block(256);
  float* o_p = nullptr;
  float* ml_p = nullptr;
  torch::Tensor o_part, ml_part;
  if (cs > 1) {
    auto opts = out.options().dtype(at::kFloat);
    o_part = torch::empty({cs, N, H, D}, opts);
    ml_part = torch::empty({cs, N, H, 2}, opts);
    o_p = o_part.data_ptr<float>();
    ml_p = ml_part.data_ptr<float>();
  }
  hipLaunchKernelGGL((paged_decode_kernel<D, GROUP>), grid, block, smem,
                     stream,
                     reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(kcache.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(vcache.data_ptr()),
                     bt.data_ptr<int>(), ctx.data_ptr<int>(),
                     reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                     o_p, ml_p,
                     H, KV, max_nb, block_size, Lpad, scale, q_row_stride,
                     slice_len);
  if (cs > 1) {
    hipLaunchKernelGGL((paged_decode_combine_kernel<D>), dim3(N, H), dim3(64),
                       0, stream, o_p, ml_p,
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       H, cs);
  }
}

}  // namespace

torch::Tensor paged_attention_decode_strided(
    torch::Tensor q, int64_t n_heads, int64_t head_dim, int64_t q_row_stride,
    torch::Tensor kcache, torch::Tensor vcache, torch::Tensor block_tables,
    torch::Tensor ctx_lens, double scale) {
  // q: (N, q_row_stride) packed rows (e.g. the fused qkv GEMM output);
  // query heads occupy the first n_heads*head_dim elements of each row.
  TORCH_CHECK(q.is_cuda() && q.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt &&
              ctx_lens.scalar_type() == at::kInt);
  const int N = q.size(0), H = (int)n_heads, D = (int)head_dim;
  const int KV = kcache.size(2);
  const int block_size = kcache.size(1);
  const int max_nb = block_tables.size(1);
  const int group = H / KV;
  TORCH_CHECK(H % KV == 0 && group <= 16, "GQA group too large");

  auto out = torch::empty({N, (int64_t)H, (int64_t)D}, q.options());
  if (N == 0) return out;

  const int max_ctx = max_nb * block_size;
  // flash-decoding context split: fill the chip (N*KV blocks alone leave
  // most CUs idle at decode batch sizes) and shorten the per-block chain
  // sweep-tuned: ~2048 blocks beats ~1024 at decode batch 160 (113 vs
  // 137 us at ctx 800 — smaller LDS score tiles double the blocks/CU);
  // cs=8+ loses to combine overhead (profiles/)
  int cs = 1;
  while (cs * 2 <= 8 && N * KV * cs < 2048
         && (max_ctx + cs * 2 - 1) / (cs * 2) >= 128)
    cs *= 2;
  if (const char* e = getenv("DISTRL_PAGED_CS")) {
    cs = std::max(1, std::min(atoi(e), 16));
    while (cs > 1 && (max_ctx + cs - 1) / cs < 32) cs /= 2;
  }
  const int slice_len = (max_ctx + cs - 1) / cs;
  const int Lpad = slice_len + 4;  // +pad to stagger LDS banks across heads
  size_t smem = (size_t)group * D * 4 + 2 * ((group + 3) & ~3) * 4
                + (size_t)group * Lpad * 4 + (size_t)VTILE * D * 2;
  TORCH_CHECK(smem <= 160 * 1024,
              "context too long for single-pass decode kernel: ", max_ctx);

  auto bt = block_tables;
  auto ctx = ctx_lens;
  const float sc = (float)scale;
  #define CASE(DD, GG) \
    if (D == DD && group == GG) { \
      launch<DD, GG>(q, kcache, vcache, bt, ctx, out, H, KV, max_nb, \
                     block_size, Lpad, sc, smem, q_row_stride, cs, \
                     slice_len); \
      HIP_CHECK_LAST(); return out; }
  CASE(128, 7) CASE(128, 5) CASE(128, 4) CASE(128, 8) CASE(128, 6)
  CASE(128, 2) CASE(128, 1) CASE(64, 7) CASE(64, 4) CASE(64, 2) CASE(64, 1)
  #undef CASE
  TORCH_CHECK(false, "paged decode: unsupported (head_dim=", D,
              ", gqa group=", group, ")");
}


torch::Tensor paged_attention_decode(torch::Tensor q, torch::Tensor kcache,
                                     torch::Tensor vcache,
                                     torch::Tensor block_tables,
                                     torch::Tensor ctx_lens, double scale) {
  TORCH_CHECK(q.dim() == 3);
  return paged_attention_decode_strided(
      q.view({q.size(0), q.size(1) * q.size(2)}), q.size(1), q.size(2),
      q.size(1) * q.size(2), kcache, vcache, block_tables, ctx_lens, scale);
}

// ---- varlen causal prefill attention (prompt phase) ----
// One workgroup per (q-row-tile of 16, q-head): same three-phase structure
// as the decode kernel with the GQA group slot re-used for the 16 q rows
// of the tile — lane-per-kv-token scoring against the LDS-staged q tile
// (with causal masking), per-row softmax, PV through LDS V tiles.
// K/V are the flat pre-RoPE'd prefill tensors (no paging needed here).
// Completes the "prefill + decode as HIP kernels" surface (BASELINE.json
// north star); replaces the torch-SDPA library call in the engine.
namespace {

// the legacy prefill kernel keeps the original 32-token V tile: the
// decode kernel's VTILE=64 retune doubled this kernel's LDS and cost it
// ~10x in occupancy (one reason the MFMA flash kernel is now the
// prefill default)
constexpr int PF_VTILE = 32;

template <int D>
__global__ __launch_bounds__(256)
void prefill_attn_kernel(const __hip_bfloat16* __restrict__ q,   // (T,H,D)
                         const __hip_bfloat16* __restrict__ k,   // (T,KV,D)
                         const __hip_bfloat16* __restrict__ v,   // (T,KV,D)
                         const int* __restrict__ tile_q0,    // global q row
                         const int* __restrict__ tile_rows,  // rows in tile
                         const int* __restrict__ tile_kv0,   // prompt start
                         __hip_bfloat16* __restrict__ out,       // (T,H,D)
                         int H, int KV, int Lpad, float scale) {
  constexpr int QT = 16;
  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int kvh = h / (H / KV);
  const int q0 = tile_q0[tile];
  const int rows = tile_rows[tile];
  const int kv0 = tile_kv0[tile];
  // causal: tile rows attend to kv positions [kv0, q0 + rows)
  const int Lmax = q0 - kv0 + rows;
  const int tid = threadIdx.x;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* q_lds = reinterpret_cast<float*>(smem_raw);          // QT*D
  float* denom = q_lds + QT * D;                              // QT
  float* scores = denom + QT;                                 // QT*Lpad
  __hip_bfloat16* v_lds = reinterpret_cast<__hip_bfloat16*>(
      scores + (size_t)QT * Lpad);                            // PF_VTILE*D

  for (int i = tid; i < QT * D; i += blockDim.x) {
    const int r = i / D;
    q_lds[i] = (r < rows)
        ? bf2f(q[((int64_t)(q0 + r) * H + h) * D + i % D]) * scale
        : 0.f;
  }
  __syncthreads();

  // phase 1: lane-per-kv-token scoring with causal mask
  for (int t = tid; t < Lmax; t += blockDim.x) {
    const __hip_bfloat16* kp = k + ((int64_t)(kv0 + t) * KV + kvh) * D;
    float part[QT];
    #pragma unroll
    for (int r = 0; r < QT; ++r) part[r] = 0.f;
    #pragma unroll
    for (int c = 0; c < D / 8; ++c) {
      const bf16x8 kvec = *reinterpret_cast<const bf16x8*>(kp + c * 8);
      float kf[8];
      #pragma unroll
      for (int j = 0; j < 8; ++j) kf[j] = bf2f(kvec.v[j]);
      #pragma unroll
      for (int r = 0; r < QT; ++r) {
        const float* qr = q_lds + r * D + c * 8;
        #pragma unroll
        for (int j = 0; j < 8; ++j) part[r] += qr[j] * kf[j];
      }
    }
    #pragma unroll
    for (int r = 0; r < QT; ++r) {
      // causal: kv position kv0+t visible to q row q0+r iff kv0+t <= q0+r
      const bool vis = (kv0 + t) <= (q0 + r);
      scores[(size_t)r * Lpad + t] = vis ? part[r] : -1e30f;
    }
  }
  __syncthreads();

  // phase 2: per-row softmax
  const int wid = tid / WAVE;
  const int wlane = tid % WAVE;
  const int nw = blockDim.x / WAVE;
  for (int r = wid; r < QT; r += nw) {
    float* s = scores + (size_t)r * Lpad;
    float m = -1e30f;
    for (int t = wlane; t < Lmax; t += WAVE) m = fmaxf(m, s[t]);
    m = wave_max(m);
    float d = 0.f;
    for (int t = wlane; t < Lmax; t += WAVE) {
      float e = __expf(s[t] - m);
      s[t] = e;
      d += e;
    }
    d = wave_sum(d);
    if (wlane == 0) denom[r] = d;
  }
  __syncthreads();

  // phase 3: PV through LDS V tiles
  constexpr int DV = D / 8;
  constexpr int UNITS = QT * DV;  // 128 (D=64) or 256 (D=128)
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  const int u = tid;
  const int ur = u / DV, ud = u % DV;
  for (int base = 0; base < Lmax; base += PF_VTILE) {
    const int tl = min(PF_VTILE, Lmax - base);
    for (int i = tid; i < tl * DV; i += blockDim.x) {
      reinterpret_cast<bf16x8*>(v_lds)[i] =
          *reinterpret_cast<const bf16x8*>(
              v + ((int64_t)(kv0 + base + i / DV) * KV + kvh) * D
              + (i % DV) * 8);
    }
    __syncthreads();
    if (u < UNITS) {
      const float* ps = scores + (size_t)ur * Lpad + base;
      #pragma unroll 4
      for (int j = 0; j < tl; ++j) {
        const float p = ps[j];
        const bf16x8 vv = reinterpret_cast<const bf16x8*>(v_lds)[j * DV + ud];
        #pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += p * bf2f(vv.v[e]);
      }
    }
    __syncthreads();
  }

  if (u < UNITS && ur < rows) {
    const float inv = 1.f / denom[ur];
    bf16x8 o;
    #pragma unroll
    for (int e = 0; e < 8; ++e) o.v[e] = f2bf(acc[e] * inv);
    *reinterpret_cast<bf16x8*>(
        out + ((int64_t)(q0 + ur) * H + h) * D + ud * 8) = o;
  }
}

}  // namespace

torch::Tensor prefill_attention(torch::Tensor q, torch::Tensor k,
                                torch::Tensor v, torch::Tensor tile_q0,
                                torch::Tensor tile_rows, torch::Tensor tile_kv0,
                                int64_t max_len, double scale) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous()
              && v.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(tile_q0.scalar_type() == at::kInt);
  const int H = q.size(1), D = q.size(2), KV = k.size(1);
  const int n_tiles = tile_q0.size(0);
  auto out = torch::empty_like(q);
  if (n_tiles == 0) return out;
  const int Lpad = (int)max_len + 4;
  size_t smem = 16 * D * 4 + 16 * 4 + (size_t)16 * Lpad * 4
                + (size_t)PF_VTILE * D * 2;
  TORCH_CHECK(smem <= 160 * 1024, "prompt too long for prefill kernel: ",
              max_len);
  dim3 grid(n_tiles, H), block(256);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (D == 128) {
    hipLaunchKernelGGL(prefill_attn_kernel<128>, grid, block, smem, stream,
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
        tile_q0.data_ptr<int>(), tile_rows.data_ptr<int>(),
        tile_kv0.data_ptr<int>(),
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
        H, KV, Lpad, (float)scale);
  } else if (D == 64) {
    hipLaunchKernelGGL(prefill_attn_kernel<64>, grid, block, smem, stream,
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
        tile_q0.data_ptr<int>(), tile_rows.data_ptr<int>(),
        tile_kv0.data_ptr<int>(),
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
        H, KV, Lpad, (float)scale);
  } else {
    TORCH_CHECK(false, "prefill attention: head_dim 64 or 128 only");
  }
  HIP_CHECK_LAST();
  return out;
}
