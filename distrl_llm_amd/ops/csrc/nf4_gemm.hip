// Fused nf4-dequant + MFMA GEMM — the 4-bit weight path of both the
// generation engine and the learner base projections (SURVEY.md §2.4:
// "hand-written fused nf4-dequant+MFMA GEMM" north star; replaces
// bitsandbytes kDequantizeBlockwise + cuBLAS in the reference's stack).
//
//   y[M,N] = x[M,K] @ dequant(W4)[N,K]^T (+ bias) (+ u[M,r] @ Bs[N,r]^T)
//
// Weights are PREPACKED into MFMA B-fragment order at load time (we own
// the format): the lane's 4 nibble-dwords for one 64-deep K chunk of a
// wave's 2 n-tiles are CONTIGUOUS (layout v3: [ngroup2][kchunk][lane][4]),
// so a whole chunk is ONE dwordx4 load per lane (v1 was 8 scalar dword
// loads) and dequant is 8 LUT-mul-cvt ops straight into the MFMA
// B fragment. absmax is fragment-ordered fp32 ([ngroup2][kchunk][16][2],
// one dwordx2 per lane per chunk). The LoRA correction rides
// the same accumulators: u = x@A^T (computed by the split-K lora_u kernel
// below) enters as ONE extra MFMA k-step per rank-32 block against the
// bf16-prepacked, scale-folded B matrix — the adapter stays exact bf16
// (never quantized), matching the reference's nf4-base + bf16-LoRA
// semantics (distributed_actor.py:58-66 + helper.py:25-46).
//
// Fragment mapping (mfma_f32_16x16x32_bf16, verified by mfma_probe test):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7
//   B: lane l holds B[col = l&15][k = (l>>4)*8 + j]   (B consumed as N x K)
//   D: lane l, reg r -> row = (l>>4)*4 + r, col = l&15
//
// Geometry: NW waves of 2 n-tiles each (NW=8 -> BN 256, 512 threads,
// 1 block/CU; NW=4 -> BN 128, 256 threads, 2 blocks/CU) x BM = 16*MT
// rows; K loop in 64-deep chunks with the x tile staged in LDS behind an
// XOR swizzle (byte ^= (row&15)<<4) so the ds_read_b128 A-fragment reads
// are conflict-free (guide G4/T2). The 2-tile wave keeps total VGPR+AGPR
// under 256 -> 2 waves/SIMD, which is what hides the ds_bpermute dequant
// latency (the 4-tile variant allocated 338 regs = 1 wave/SIMD and ran
// ~2x slower end-to-end).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16v8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

namespace {

// nf4 codebook. NOTE: indexed per-lane with a divergent nibble, so it must
// live in LDS at run time — a divergent index into __constant__/global
// memory scalarizes into a waterfall on CDNA (measured 8x whole-kernel
// slowdown). Each kernel copies it to LDS once.
__device__ const float NF4_LUT[16] = {
    -1.0f, -0.6961928009986877f, -0.5250730514526367f, -0.39491748809814453f,
    -0.28444138169288635f, -0.18477343022823334f, -0.09105003625154495f, 0.0f,
    0.07958029955625534f, 0.16093020141124725f, 0.24611230194568634f,
    0.33791524171829224f, 0.44070982933044434f, 0.5626170039176941f,
    0.7229568362236023f, 1.0f};

DEV_INLINE bf16v8 lds_read_frag(const char* base, int byte_off) {
  return *reinterpret_cast<const bf16v8*>(base + byte_off);
}

template <int MT, int NW, int DBG = 0>
// NW = waves per block (each wave owns TWO n-tiles; BN = NW*32);
// DBG: 1 = skip LUT dequant, 2 = skip x LDS
__global__ __launch_bounds__(NW * 64)
void nf4_gemm_kernel(const __hip_bfloat16* __restrict__ x,   // (M, K)
                     const uint32_t* __restrict__ w4f,       // frag v3
                     const float* __restrict__ amaxf,        // frag v3
                     const __hip_bfloat16* __restrict__ bias,  // (N) | null
                     const float* __restrict__ u,            // (M, r) | null
                     const uint32_t* __restrict__ bfrag,     // B frag | null
                     __hip_bfloat16* __restrict__ y,         // (M, N)
                     float* __restrict__ ws,                 // split-K | null
                     int M, int N, int K, int r, int u_stride,
                     int nsp_per) {
  constexpr int BM = 16 * MT;
  constexpr int THREADS = NW * 64;
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int wave = tid >> 6;
  const int lrow = l & 15;          // fragment row/col index
  const int lk = l >> 4;            // fragment k-group
  // grid: x = m-block (fast dim), y = n-block — adjacent linear block ids
  // share the same weight panel, so the XCD L2 absorbs re-reads when the
  // m dimension is tiled (T1 locality without an explicit remap)
  const int mbase = blockIdx.x * BM;
  const int g2 = blockIdx.y * NW + wave;          // n-group of 2 tiles
  const int ntile0 = g2 * 2;                      // this wave's 2 n-tiles
  const int zid = blockIdx.z;                     // split-K slice
  const int nkb = K / 64;                         // total 64-deep chunks

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // x super-panel columns: 512 when the LDS budget allows (8-wave blocks
  // run 1 block/CU so the whole 160 KB is ours) — halves the barrier
  // frequency of the staging loop
  constexpr int SK_ = (MT <= 2 || NW == 8) ? 512 : 256;
  char* x_lds = smem;                         // BM * SK * 2 bytes
  char* u_lds = x_lds + BM * SK_ * 2;         // BM * r * 2 bytes
  // per-lane register copy of the codebook for the shuffle-LUT dequant
  // (a pure-VALU cndmask-tree variant measured 1.5x SLOWER: ~9 VALU/value
  // beats ds_bpermute's latency only on paper — the compiler pipelines
  // the shuffles ~2-deep and the VALU tree just adds issue pressure).
  // The table is replicated every 16 lanes, which makes the raw-address
  // bpermute trick below legal (garbage bits [7:6] only pick the replica).
  const float lut_reg = NF4_LUT[tid & 15];
  const int lut_bits = __builtin_bit_cast(int, lut_reg);

  f32x4 acc[MT][2];
  #pragma unroll
  for (int mt = 0; mt < MT; ++mt)
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) acc[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};

  // stage u (fp32 -> bf16) once; it is tiny (BM x r)
  if (u != nullptr) {
    for (int i = tid; i < BM * (r / 8); i += THREADS) {
      const int row = i / (r / 8);
      const int unit = i % (r / 8);
      bf16x8 s;
      if (mbase + row < M) {
        const float* up = u + (int64_t)(mbase + row) * u_stride + unit * 8;
        #pragma unroll
        for (int j = 0; j < 8; ++j) s.v[j] = f2bf(up[j]);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j) s.v[j] = f2bf(0.f);
      }
      *reinterpret_cast<bf16x8*>(u_lds + (int64_t)row * r * 2 + unit * 16) = s;
    }
  }

  // ---- main K loop: 64-deep chunks streamed through a static 4-slot
  // register ring (3 chunks prefetched ahead). One chunk = 1 dwordx4 W
  // load + 1 dwordx2 absmax load per lane (layout v3) — hipcc's counted
  // vmcnt bookkeeping keeps up to 8 loads in flight, and the 2-tile wave
  // keeps total registers low enough for 2 waves/SIMD (the co-resident
  // wave is what hides the dequant-shuffle latency). Without prefetch the
  // compiler issues each weight dword right before its use behind
  // vmcnt(0) — every fragment pays full HBM latency serially (measured
  // 8-20x slowdown).
  uint32_t wb[4][4];
  f32x2 am[4];

  #define LOAD_WCHUNK(KB, RS)                                             \
    {                                                                     \
      const int64_t cb_ = (int64_t)g2 * nkb + (KB);                       \
      *reinterpret_cast<u32x4*>(&wb[RS][0]) =                             \
          reinterpret_cast<const u32x4*>(w4f)[cb_ * 64 + l];              \
      am[RS] = reinterpret_cast<const f32x2*>(amaxf + cb_ * 32)[lrow];    \
    }

  // this slice's super-panel range (split-K over whole panels)
  constexpr int SKC = SK_ / 64;                      // chunks per panel
  static_assert(SKC % 4 == 0, "ring phase must stay aligned per panel");
  const int npanels = K / SK_;
  const int sp0 = zid * nsp_per;
  const int sp1 = min(npanels, sp0 + nsp_per);
  const int kb0 = sp0 * SKC;
  const int kb_end = sp1 * SKC;
  if (kb0 < kb_end) LOAD_WCHUNK(kb0, 0);
  if (kb0 + 1 < kb_end) LOAD_WCHUNK(kb0 + 1, 1);
  if (kb0 + 2 < kb_end) LOAD_WCHUNK(kb0 + 2, 2);

  // double-buffered PRE-DEQUANTED B fragments: chunk kc's MFMAs consume
  // bfr[kc&1] while chunk kc+1 dequants into bfr[(kc+1)&1] — the
  // ds_bpermute latency hides under the matrix pipe instead of gating it
  bf16v8 bfr[2][4];
  #define DEQUANT_CHUNK(RS, PB)                                           \
    {                                                                     \
      if constexpr (DBG == 1) {                                           \
        _Pragma("unroll")                                                 \
        for (int q_ = 0; q_ < 4; ++q_) {                                  \
          const uint32_t wbits_ = wb[RS][q_];                             \
          _Pragma("unroll")                                               \
          for (int j_ = 0; j_ < 8; ++j_)                                  \
            bfr[PB][q_][j_] = (__bf16)(am[RS][q_ & 1]                     \
                                       + (float)(wbits_ & 1));            \
        }                                                                 \
      } else {                                                            \
        float cval_[4][8];                                                \
        _Pragma("unroll")                                                 \
        for (int q_ = 0; q_ < 4; ++q_) {                                  \
          const uint32_t wbits_ = wb[RS][q_];                             \
          _Pragma("unroll")                                               \
          for (int j_ = 0; j_ < 8; ++j_) {                                \
            const int addr_ = (j_ == 0) ? (int)(wbits_ << 2)              \
                                        : (int)(wbits_ >> (4 * j_ - 2));  \
            cval_[q_][j_] = __builtin_bit_cast(                           \
                float, __builtin_amdgcn_ds_bpermute(addr_, lut_bits));    \
          }                                                               \
        }                                                                 \
        _Pragma("unroll")                                                 \
        for (int q_ = 0; q_ < 4; ++q_) {                                  \
          const float amv_ = am[RS][q_ & 1];                              \
          _Pragma("unroll")                                               \
          for (int j_ = 0; j_ < 8; ++j_)                                  \
            bfr[PB][q_][j_] = (__bf16)(cval_[q_][j_] * amv_);             \
        }                                                                 \
      }                                                                   \
    }
  if (kb0 < kb_end) DEQUANT_CHUNK(0, 0);

  // x is staged in SUPER-panels of SK columns: one barrier pair per
  // SK/64 weight chunks, so the chunk loop in between runs barrier-free
  // and the one-chunk-ahead weight prefetch is never drained (hipcc puts
  // a vmcnt(0) on any in-loop ds_write/barrier path — guide §5 traps).
  constexpr int SK = SK_;                     // host asserts K % SK == 0
  constexpr int SKU = SK / 8;                 // bf16x8 units per row
  constexpr int XIT = (BM * SKU + THREADS - 1) / THREADS;  // staging iters

  // full-tile blocks (every row valid) issue the staging loads
  // unpredicated — a per-iteration row guard makes hipcc branch around
  // EACH load (one basic block per load, guide §5 trap c); only the
  // ragged last m-block pays the guarded path
  const bool full_rows = (mbase + BM <= M);

  for (int sp = sp0; sp < sp1; ++sp) {
    const int sk = sp * SK;
    // issue-early x loads for the whole super-panel
    bf16x8 xs[XIT];
    if (full_rows) {
      #pragma unroll
      for (int it = 0; it < XIT; ++it) {
        const int i = tid + it * THREADS;
        if (i < BM * SKU)
          xs[it] = *reinterpret_cast<const bf16x8*>(
              x + (int64_t)(mbase + i / SKU) * K + sk + (i % SKU) * 8);
      }
    } else {
      #pragma unroll
      for (int it = 0; it < XIT; ++it) {
        const int i = tid + it * THREADS;
        if (i < BM * SKU) {
          const int row = i / SKU;
          if (mbase + row < M) {
            xs[it] = *reinterpret_cast<const bf16x8*>(
                x + (int64_t)(mbase + row) * K + sk + (i % SKU) * 8);
          } else {
            #pragma unroll
            for (int j = 0; j < 8; ++j) xs[it].v[j] = f2bf(0.f);
          }
        }
      }
    }
    __syncthreads();  // previous panel fully consumed
    #pragma unroll
    for (int it = 0; it < XIT; ++it) {
      const int i = tid + it * THREADS;
      if (i < BM * SKU) {
        const int row = i / SKU;
        const int off = (row * (SK * 2) + (i % SKU) * 16)
                        ^ ((row & 15) << 4);
        *reinterpret_cast<bf16x8*>(x_lds + off) = xs[it];
      }
    }
    __syncthreads();

    #pragma unroll
    for (int kc = 0; kc < SKC; ++kc) {
      const int kb = sp * SKC + kc;
      // ring slot indices are compile-time: SKC % 4 == 0 keeps the
      // panel-relative phase aligned, so slot = kc & 3 in the unrolled
      // body (no register-shuffle rotation between chunks)
      if (kb + 3 < kb_end) LOAD_WCHUNK(kb + 3, (kc + 3) & 3);

      // ---- A) MFMAs of the CURRENT chunk from the PRE-DEQUANTED
      // fragments (bfr[kc&1], produced one chunk ago — the bpermute
      // latency of the dequant overlaps the previous chunk's MFMAs)
      #pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16v8 afrag[MT];
        #pragma unroll
        for (int mt = 0; mt < MT; ++mt) {
          if constexpr (DBG == 2) {
            #pragma unroll
            for (int j = 0; j < 8; ++j) afrag[mt][j] = (__bf16)1.0f;
          } else {
            const int row = mt * 16 + lrow;
            const int off = (row * (SK * 2)
                             + (kc * 8 + ks * 4 + lk) * 16)
                            ^ ((row & 15) << 4);
            afrag[mt] = lds_read_frag(x_lds, off);
          }
        }
        #pragma unroll
        for (int nt = 0; nt < 2; ++nt)
          #pragma unroll
          for (int mt = 0; mt < MT; ++mt)
            acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[mt], bfr[kc & 1][ks * 2 + nt], acc[mt][nt], 0, 0, 0);
      }

      // ---- B) dequant the NEXT chunk into the spare fragment buffer
      if (kb + 1 < kb_end) {
        DEQUANT_CHUNK((kc + 1) & 3, (kc + 1) & 1);
      }
    }
  }
  #undef LOAD_WCHUNK
  #undef DEQUANT_CHUNK

  // ---- LoRA epilogue: one extra MFMA k-step per rank-32 block ----
  // (slice 0 only under split-K; slices contribute additively)
  if (u != nullptr && zid == 0) {
    __syncthreads();
    for (int rs = 0; rs < r / 32; ++rs) {
      bf16v8 afrag[MT];
      #pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        const int row = mt * 16 + lrow;
        afrag[mt] = *reinterpret_cast<const bf16v8*>(
            u_lds + (int64_t)row * r * 2 + (rs * 32 + lk * 8) * 2);
      }
      #pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        const int ntg = ntile0 + nt;
        const uint4 bw = *reinterpret_cast<const uint4*>(
            bfrag + ((int64_t)ntg * (r / 32) + rs) * 64 * 4 + l * 4);
        const bf16v8 bfr = *reinterpret_cast<const bf16v8*>(&bw);
        #pragma unroll
        for (int mt = 0; mt < MT; ++mt)
          acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mt], bfr, acc[mt][nt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: bias + store (fp32 slab under split-K) ----
  #pragma unroll
  for (int nt = 0; nt < 2; ++nt) {
    const int n = (ntile0 + nt) * 16 + lrow;
    const float bv = (bias != nullptr && zid == 0) ? bf2f(bias[n]) : 0.f;
    #pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int m = mbase + mt * 16 + lk * 4 + rr;
        if (m < M) {
          if (ws != nullptr)
            ws[((int64_t)zid * M + m) * N + n] = acc[mt][nt][rr] + bv;
          else
            y[(int64_t)m * N + n] = f2bf(acc[mt][nt][rr] + bv);
        }
      }
    }
  }
}

// split-K combine: y = bf16(sum_z ws[z])
__global__ __launch_bounds__(256)
void splitk_reduce_kernel(const float* __restrict__ ws,
                          __hip_bfloat16* __restrict__ y,
                          int64_t mn, int ksplit) {
  const int64_t i0 = ((int64_t)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i0 + 3 >= mn) {
    for (int64_t i = i0; i < mn; ++i) {
      float a = 0.f;
      for (int z = 0; z < ksplit; ++z) a += ws[(int64_t)z * mn + i];
      y[i] = f2bf(a);
    }
    return;
  }
  float4 a = *reinterpret_cast<const float4*>(ws + i0);
  for (int z = 1; z < ksplit; ++z) {
    const float4 b = *reinterpret_cast<const float4*>(ws + (int64_t)z * mn + i0);
    a.x += b.x; a.y += b.y; a.z += b.z; a.w += b.w;
  }
  bf16x8* dummy;
  __hip_bfloat16 o[4] = {f2bf(a.x), f2bf(a.y), f2bf(a.z), f2bf(a.w)};
  *reinterpret_cast<uint2*>(y + i0) = *reinterpret_cast<uint2*>(o);
  (void)dummy;
}

// ---- split-K LoRA A kernel: u[M,r] = x[M,K] @ A^T, fp32 atomic combine.
// A is prepacked into B-fragment order (bf16, 16 B per (r-tile, k-step,
// lane)); x A-fragments are read straight from global (x is L2-resident
// at decode batch sizes). One wave per (m-tile, r-tile, k-split) block.
__global__ __launch_bounds__(64)
void lora_u_kernel(const __hip_bfloat16* __restrict__ x,   // (M, K)
                   const uint32_t* __restrict__ afrag,     // (r/16, K/32, 64, 4)
                   float* __restrict__ u,                  // (M, r) zeroed
                   int M, int K, int r, int ksplit, int u_stride) {
  const int l = threadIdx.x;
  const int lrow = l & 15;
  const int lk = l >> 4;
  const int mt = blockIdx.x;
  const int rt = blockIdx.y;
  const int ks_id = blockIdx.z;
  const int ksteps = K / 32;
  const int per = (ksteps + ksplit - 1) / ksplit;
  const int k0 = ks_id * per;
  const int k1 = min(ksteps, k0 + per);

  const int m = mt * 16 + lrow;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  // software-pipelined 2-deep: issue next k-steps' fragments while the
  // current MFMA runs (the bare loop serializes on load latency)
  auto load_af = [&](int kstep, bf16v8& af, uint4& bw) {
    if (m < M) {
      af = *reinterpret_cast<const bf16v8*>(
          x + (int64_t)m * K + kstep * 32 + lk * 8);
    } else {
      #pragma unroll
      for (int j = 0; j < 8; ++j) af[j] = (__bf16)0.f;
    }
    bw = *reinterpret_cast<const uint4*>(
        afrag + ((int64_t)rt * ksteps + kstep) * 64 * 4 + l * 4);
  };
  bf16v8 af_c, af_n;
  uint4 bw_c, bw_n;
  if (k0 < k1) load_af(k0, af_c, bw_c);
  if (k0 + 1 < k1) load_af(k0 + 1, af_n, bw_n);
  for (int kstep = k0; kstep < k1; ++kstep) {
    bf16v8 af_f;
    uint4 bw_f;
    if (kstep + 2 < k1) load_af(kstep + 2, af_f, bw_f);
    const bf16v8 bf = *reinterpret_cast<const bf16v8*>(&bw_c);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af_c, bf, acc, 0, 0, 0);
    af_c = af_n; bw_c = bw_n;
    af_n = af_f; bw_n = bw_f;
  }
  #pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int mrow = mt * 16 + lk * 4 + rr;
    if (mrow < M)
      atomicAdd(&u[(int64_t)mrow * u_stride + rt * 16 + lrow], acc[rr]);
  }
}

// ---- single-tile MFMA mapping probe (numerics test support) ----
__global__ void mfma_probe_kernel(const __hip_bfloat16* a,  // (16, 32)
                                  const __hip_bfloat16* b,  // (16, 32)
                                  float* c) {               // (16, 16)
  const int l = threadIdx.x;
  bf16v8 av, bv;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    av[j] = (__bf16)bf2f(a[(l & 15) * 32 + (l >> 4) * 8 + j]);
    bv[j] = (__bf16)bf2f(b[(l & 15) * 32 + (l >> 4) * 8 + j]);
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc, 0, 0, 0);
  #pragma unroll
  for (int rr = 0; rr < 4; ++rr)
    c[((l >> 4) * 4 + rr) * 16 + (l & 15)] = acc[rr];
}

}  // namespace

torch::Tensor nf4_gemm(torch::Tensor x, torch::Tensor w4f, torch::Tensor amaxf,
                       c10::optional<torch::Tensor> bias,
                       c10::optional<torch::Tensor> u,
                       c10::optional<torch::Tensor> bfrag,
                       int64_t N, int64_t K, int64_t r) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(w4f.scalar_type() == at::kInt || w4f.scalar_type() == at::kUInt32);
  const int M = x.size(0);
  TORCH_CHECK(x.size(1) == K);
  {
    // the prepack pads K to a multiple of 512 (zero absmax): mirror it
    const int64_t Kp = (K + 511) / 512 * 512;
    if (Kp != K) {
      x = torch::constant_pad_nd(x, {0, Kp - K}, 0);
      K = Kp;
    }
  }
  TORCH_CHECK(N % 128 == 0, "nf4_gemm: N must be a multiple of 128, got ", N);
  TORCH_CHECK(K % 64 == 0);
  auto y = torch::empty({(int64_t)M, N}, x.options());
  if (M == 0) return y;

  const bool has_lora = u.has_value() && r > 0;
  int u_stride = 0;
  if (has_lora) {
    TORCH_CHECK(r % 32 == 0 && bfrag.has_value());
    TORCH_CHECK(u->scalar_type() == at::kFloat && u->stride(1) == 1);
    u_stride = u->stride(0);
  }

  // m-tile size: MT5 + 2-tile waves fits 2 waves/SIMD
  int mtcap = 5;
  if (const char* e = getenv("DISTRL_NF4_MTCAP")) mtcap = atoi(e);
  int mt = std::min<int>((M + 15) / 16, std::max(1, mtcap));
  const int BM = 16 * mt;
  // waves per block: 4-wave (BN=128, 2 blocks/CU) measured >= the 8-wave
  // variant on every decode shape (co-resident independent blocks
  // decorrelate the dequant-latency stalls); the 8-wave template is kept
  // behind DISTRL_NF4_NW=8 for experiments
  int nw = 4;
  if (const char* e = getenv("DISTRL_NF4_NW")) nw = atoi(e);
  TORCH_CHECK(nw == 4 || nw == 8);
  TORCH_CHECK(N % (nw * 32) == 0);
  const int SK = (mt <= 2 || nw == 8) ? 512 : 256;  // mirror kernel SK_
  TORCH_CHECK(K % SK == 0, "nf4_gemm: K (", K, ") % ", SK, " != 0");
  const int nblk = (int)N / (nw * 32);
  // split-K over super-panels until the grid fills the chip (the skinny
  // decode shapes otherwise run at ~1 block-wave, fully latency-exposed)
  const int npanels = (int)K / SK;
  const int base_blocks = ((M + BM - 1) / BM) * nblk;
  int blk_target = (nw == 8) ? 384 : 640;  // sweep-tuned (profiles/)
  if (const char* e = getenv("DISTRL_NF4_BLKTGT")) blk_target = atoi(e);
  int ksplit = 1;
  while (ksplit * 2 <= npanels && base_blocks * ksplit < blk_target
         && ksplit < 16)
    ksplit *= 2;
  // deep-K shapes (down-proj): past ksplit 8 the fp32 slab traffic
  // outweighs the fill gain (sweep: down 60.8us@8 vs 64.1@16)
  if (npanels >= 32 && base_blocks * 8 >= 256) ksplit = std::min(ksplit, 8);
  if (const char* e = getenv("DISTRL_NF4_KSPLIT")) ksplit = atoi(e);
  ksplit = std::max(1, std::min(ksplit, npanels));
  const int nsp_per = (npanels + ksplit - 1) / ksplit;
  ksplit = (npanels + nsp_per - 1) / nsp_per;
  dim3 grid((M + BM - 1) / BM, nblk, ksplit), block(nw * 64);
  size_t smem = (size_t)BM * SK * 2
                + (has_lora ? (size_t)BM * r * 2 : 0);
  if (getenv("DISTRL_NF4_LDSPAD")) smem += 8192;  // debug: OOB guard
  torch::Tensor ws;
  float* ws_p = nullptr;
  if (ksplit > 1) {
    ws = torch::empty({ksplit, (int64_t)M, N}, x.options().dtype(at::kFloat));
    if (getenv("DISTRL_NF4_WSZERO")) ws.zero_();
    ws_p = ws.data_ptr<float>();
  }

  auto stream = at::cuda::getCurrentCUDAStream();
  const __hip_bfloat16* bias_p = nullptr;
  if (bias.has_value() && bias->defined() && bias->numel() > 0)
    bias_p = reinterpret_cast<const __hip_bfloat16*>(bias->data_ptr());
  const float* u_p = has_lora ? u->data_ptr<float>() : nullptr;
  const uint32_t* bf_p = has_lora
      ? reinterpret_cast<const uint32_t*>(bfrag->data_ptr()) : nullptr;

  int dbg = 0;
  if (const char* e = getenv("DISTRL_NF4_DBG")) dbg = atoi(e);
  #define LAUNCH_1(MTV, NWV, DBGV) \
    hipLaunchKernelGGL((nf4_gemm_kernel<MTV, NWV, DBGV>), grid, block, smem, \
        stream, \
        reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()), \
        reinterpret_cast<const uint32_t*>(w4f.data_ptr()), \
        amaxf.data_ptr<float>(), bias_p, u_p, bf_p, \
        reinterpret_cast<__hip_bfloat16*>(y.data_ptr()), ws_p, \
        M, (int)N, (int)K, (int)r, u_stride, nsp_per)
  #define LAUNCH(MTV) \
    do { if (nw == 8) LAUNCH_1(MTV, 8, 0); else LAUNCH_1(MTV, 4, 0); } \
    while (0)
  if (dbg == 1 && mt == 1 && nw == 4) {
    LAUNCH_1(1, 4, 1);
  } else if (dbg == 2 && mt == 1 && nw == 4) {
    LAUNCH_1(1, 4, 2);
  } else if (dbg == 1 && mt == 5) {        // ablation: no LUT dequant
    if (nw == 8) LAUNCH_1(5, 8, 1); else LAUNCH_1(5, 4, 1);
  } else if (dbg == 2 && mt == 5) {        // ablation: no x LDS reads
    if (nw == 8) LAUNCH_1(5, 8, 2); else LAUNCH_1(5, 4, 2);
  } else
  switch (mt) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 3: LAUNCH(3); break;
    case 4: LAUNCH(4); break;
    default: LAUNCH(5); break;
  }
  #undef LAUNCH
  #undef LAUNCH_1
  if (ksplit > 1 && getenv("DISTRL_NF4_RETWS")) return ws;  // debug
  if (ksplit > 1) {
    const int64_t mn = (int64_t)M * N;
    hipLaunchKernelGGL(splitk_reduce_kernel, dim3(CDIV(mn, 1024)), dim3(256),
                       0, stream, ws_p,
                       reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                       mn, ksplit);
  }
  HIP_CHECK_LAST();
  return y;
}

void lora_u(torch::Tensor x, torch::Tensor afrag, torch::Tensor u,
            int64_t r, int64_t ksplit) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(u.scalar_type() == at::kFloat && u.stride(1) == 1);
  const int M = x.size(0), K = x.size(1);
  TORCH_CHECK(r % 16 == 0 && K % 32 == 0);
  // caller zeroes u (batched across all sites/layers per step)
  dim3 grid((M + 15) / 16, r / 16, ksplit), block(64);
  hipLaunchKernelGGL(lora_u_kernel, grid, block, 0,
                     at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                     reinterpret_cast<const uint32_t*>(afrag.data_ptr()),
                     u.data_ptr<float>(), M, K, (int)r, (int)ksplit,
                     (int)u.stride(0));
  HIP_CHECK_LAST();
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  auto c = torch::empty({16, 16}, a.options().dtype(at::kFloat));
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0,
                     at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<const __hip_bfloat16*>(a.contiguous().data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(b.contiguous().data_ptr()),
                     c.data_ptr<float>());
  HIP_CHECK_LAST();
  return c;
}
