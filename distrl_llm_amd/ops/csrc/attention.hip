// First-party CDNA4 flash attention (forward + backward) for the learner's
// teacher-forced training pass and the engine's long-prompt prefill —
// replaces torch SDPA (aotriton, Triton-derived) on the hot path
// (reference distributed_actor.py:241-243 forward + loss.backward();
// SURVEY.md §2.4-B "causal attention fwd/bwd" row; north star: no Triton).
//
// Design (MFMA 16x16x32, online softmax, batched causal GQA):
//  * The S-matrix is never materialized: per 64-row Q block (4 waves x 16
//    rows), KV is streamed in 32-token tiles through LDS.
//  * Every MFMA operand is a ROW-MAJOR fragment: QK^T is computed in the
//    SWAPPED orientation ST[kv][q] = mfma(A=K rows, B=Q rows), so softmax
//    statistics reduce over the register rows + two cross-lane shuffles,
//    and P@V consumes a PRE-TRANSPOSED V (v_t = V^T, one cheap torch
//    transpose per call) so its B fragments are row-major reads too.
//  * The only layout conversion is D-layout (f32 accumulator) ->
//    A/B-fragment (bf16), done in-register with v_cvt_pk_bf16_f32 + 8
//    ds_bpermute per 16x32 fragment (guide T12's idea at 16x16 shape).
//  * Backward runs in two passes with NO atomics: pass Q (q-block outer,
//    recomputes ST from the saved row LSE, produces dQ) and pass KV
//    (kv-block outer in the UNSWAPPED orientation, produces dK^T/dV^T).
//    delta = rowsum(dO*O) is computed by the host in torch.
//
// Fragment mapping (mfma_f32_16x16x32_bf16, same as nf4_gemm.hip):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j]
//   B: lane l holds B[col = l&15][k = (l>>4)*8 + j]   (D = A @ B^T)
//   D: lane l, reg r -> row = (l>>4)*4 + r, col = l&15
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16v8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace {

constexpr int QB = 16;    // q rows per wave
constexpr int KVB = 32;   // kv tile
constexpr int NWAVE = 4;  // waves per block -> 64 q rows per block

DEV_INLINE float wave_xor_max(float v) {
  v = fmaxf(v, __shfl_xor(v, 16, 64));
  v = fmaxf(v, __shfl_xor(v, 32, 64));
  return v;
}

DEV_INLINE float wave_xor_sum(float v) {
  v += __shfl_xor(v, 16, 64);
  v += __shfl_xor(v, 32, 64);
  return v;
}

// Pack two f32 into one dword of bf16x2 (round-to-nearest-even).
DEV_INLINE uint32_t cvt_pk_bf16(float lo, float hi) {
  union { __hip_bfloat162 h; uint32_t u; } cv;
  cv.h = __float22bfloat162_rn(float2{lo, hi});
  return cv.u;
}

// D-layout -> A/B-fragment relayout.
// Sources: two 16x16 D-layout tiles T0 (k-rows 0-15) and T1 (k-rows
// 16-31) over the SAME preserved column index c = l&15; lane l holds
// T[4*(l>>4)+r][l&15]. Target: bf16v8 with lane l = F[c = l&15]
// [k = 8*(l>>4)+j]. Each lane pulls two packed dwords from source lanes
// s0/s1 of T0 AND T1, then selects by its own group half.
DEV_INLINE bf16v8 relayout_frag(const f32x4& t0, const f32x4& t1) {
  const int l = threadIdx.x & 63;
  const int g = l >> 4;
  const int c = l & 15;
  const uint32_t a0_0 = cvt_pk_bf16(t0[0], t0[1]);
  const uint32_t a1_0 = cvt_pk_bf16(t0[2], t0[3]);
  const uint32_t a0_1 = cvt_pk_bf16(t1[0], t1[1]);
  const uint32_t a1_1 = cvt_pk_bf16(t1[2], t1[3]);
  const int s0 = ((g & 1) << 5) + c;        // source lane: group 2*(g&1)
  const int s1 = s0 + 16;                   // group 2*(g&1)+1
  union { uint32_t w[4]; bf16v8 v; } out;
  const uint32_t p00 = __shfl((int)a0_0, s0, 64), p01 = __shfl((int)a1_0, s0, 64);
  const uint32_t p02 = __shfl((int)a0_0, s1, 64), p03 = __shfl((int)a1_0, s1, 64);
  const uint32_t p10 = __shfl((int)a0_1, s0, 64), p11 = __shfl((int)a1_1, s0, 64);
  const uint32_t p12 = __shfl((int)a0_1, s1, 64), p13 = __shfl((int)a1_1, s1, 64);
  const bool hi = g >= 2;                   // k 16..31 -> T1
  out.w[0] = hi ? p10 : p00;
  out.w[1] = hi ? p11 : p01;
  out.w[2] = hi ? p12 : p02;
  out.w[3] = hi ? p13 : p03;
  return out.v;
}

// LDS tile staging: [ROWS][COLS] bf16 row-major with an XOR swizzle on
// the 16-B column slot, MASKED to the row width (COLS/8 slots per row)
// so it never escapes the row — (row & 15) << 4 for 256-B rows,
// (row & 3) << 4 for 64-B rows. Readers use lds_frag with the same XOR.
// Guards rows >= limit with zero fill.
template <int COLS>
DEV_INLINE int tile_off(int row, int cu) {
  constexpr int SWZ_MASK = (COLS / 8) - 1;
  return (row * (COLS * 2) + cu * 16) ^ ((row & SWZ_MASK) << 4);
}

template <int ROWS, int COLS>
DEV_INLINE void stage_tile(char* lds, const __hip_bfloat16* src,
                           int64_t src_row_stride, int rows_valid, int tid,
                           int nthreads) {
  constexpr int UN = COLS / 8;  // bf16x8 units per row
  for (int i = tid; i < ROWS * UN; i += nthreads) {
    const int row = i / UN;
    const int cu = i % UN;
    bf16x8 v;
    if (row < rows_valid) {
      v = *reinterpret_cast<const bf16x8*>(src + (int64_t)row * src_row_stride
                                           + cu * 8);
    } else {
      #pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = f2bf(0.f);
    }
    *reinterpret_cast<bf16x8*>(lds + tile_off<COLS>(row, cu)) = v;
  }
}

template <int COLS>
DEV_INLINE bf16v8 lds_frag(const char* lds, int row, int col8) {
  return *reinterpret_cast<const bf16v8*>(lds + tile_off<COLS>(row, col8));
}

// ---------------------------------------------------------------- forward
// o[b,h,q,d], lse[b,h,q] from q/k/v. q,k: (B,H*/Hkv,T,D) row-major over
// (T,D); v_t: (B,Hkv,D,T) (V pre-transposed). Causal. T arbitrary.
template <int D>
__global__ __launch_bounds__(256)
void flash_fwd_kernel(const __hip_bfloat16* __restrict__ q,
                      const __hip_bfloat16* __restrict__ k,
                      const __hip_bfloat16* __restrict__ vt,
                      __hip_bfloat16* __restrict__ o,
                      float* __restrict__ lse,
                      int B, int Hq, int Hkv, int T, int Tp, float scale) {
  const int qblk = blockIdx.x;          // 64-row q block
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4;
  const int lc = l & 15;

  const int q0 = qblk * (QB * NWAVE) + wave * QB;  // this wave's q rows
  const int q_end = min(q0 + QB, T);  // waves past T still hit barriers

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                        // KVB x D bf16 (swizzled)
  char* v_lds = k_lds + KVB * D * 2;         // D x KVB bf16 (swizzled)
  float* o_lds = reinterpret_cast<float*>(v_lds + (size_t)D * KVB * 2);
  // o_lds: per-wave 16 x D f32 staging for the epilogue (NWAVE*16*D)

  const __hip_bfloat16* qp = q + (((int64_t)b * Hq + h) * T) * D;
  const __hip_bfloat16* kp = k + (((int64_t)b * Hkv + kvh) * T) * D;
  const __hip_bfloat16* vp = vt + (((int64_t)b * Hkv + kvh) * D) * Tp;

  // Q fragments: B[col = q-row = lc][kd = lg*8+j], one per 32-d step.
  bf16v8 qf[D / 32];
  {
    const int row = q0 + lc;
    #pragma unroll
    for (int ks = 0; ks < D / 32; ++ks) {
      if (row < T) {
        qf[ks] = *reinterpret_cast<const bf16v8*>(
            qp + (int64_t)row * D + ks * 32 + lg * 8);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j) qf[ks][j] = (__bf16)0.f;
      }
    }
  }

  float m_run = -1e30f;   // running max for this lane's q column
  float l_run = 0.f;      // running denominator
  f32x4 oacc[D / 16];     // O D-layout: col d = lc (+16*dt), row q = 4*lg+r
  #pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) oacc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  // causal: this BLOCK needs kv tiles up to the block's last q row
  const int blk_q_end = min(qblk * (QB * NWAVE) + QB * NWAVE, T);
  const int n_tiles = (blk_q_end + KVB - 1) / KVB;

  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * KVB;
    const int kv_valid = min(KVB, T - kv0);
    __syncthreads();
    stage_tile<KVB, D>(k_lds, kp + (int64_t)kv0 * D, D, kv_valid, tid, 256);
    stage_tile<D, KVB>(v_lds, vp + kv0, Tp, D, tid, 256);  // padded cols
    __syncthreads();

    if (q0 < T && kv0 <= q_end - 1) {
      // ST[kv][q] tiles: kv subtile s (rows s*16..), cols = q block
      f32x4 st[KVB / 16];
      #pragma unroll
      for (int s = 0; s < KVB / 16; ++s) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        #pragma unroll
        for (int ks = 0; ks < D / 32; ++ks) {
          const bf16v8 kf = lds_frag<D>(k_lds, s * 16 + lc, ks * 4 + lg);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qf[ks], acc,
                                                        0, 0, 0);
        }
        st[s] = acc;
      }
      // scale + causal mask + running softmax (per q column = lane lc)
      float tile_max = -1e30f;
      #pragma unroll
      for (int s = 0; s < KVB / 16; ++s) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kv = kv0 + s * 16 + 4 * lg + r;
          const int qq = q0 + lc;
          float v = st[s][r] * scale;
          if (kv > qq || qq >= T) v = -1e30f;
          st[s][r] = v;
          tile_max = fmaxf(tile_max, v);
        }
      }
      tile_max = wave_xor_max(tile_max);
      const float m_new = fmaxf(m_run, tile_max);
      const float alpha = __expf(m_run - m_new);
      float psum = 0.f;
      #pragma unroll
      for (int s = 0; s < KVB / 16; ++s) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float p = __expf(st[s][r] - m_new);
          st[s][r] = p;
          psum += p;
        }
      }
      psum = wave_xor_sum(psum);
      l_run = l_run * alpha + psum;
      m_run = m_new;
      // rescale O. The softmax stats live per LANE COLUMN (q = lc), but
      // the accumulator's rows are q = 4*lg + r — fetch each row's alpha
      // from the lane that owns that q column.
      float alpha_row[4];
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        alpha_row[r] = __shfl(alpha, 4 * lg + r, 64);
      #pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) oacc[dt][r] *= alpha_row[r];
      }
      // PV: A = PT^T fragments (relayout), B = v_t rows
      const bf16v8 pa = relayout_frag(st[0], st[1]);
      #pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const bf16v8 vf = lds_frag<KVB>(v_lds, dt * 16 + lc, lg);
        oacc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vf, oacc[dt],
                                                           0, 0, 0);
      }
    }
  }

  // epilogue: normalize, stage O tile through LDS (D-layout columns ->
  // row-major global), write lse. As with alpha, 1/l for accumulator row
  // q = 4*lg+r lives in lane lc = 4*lg+r.
  const float inv = (l_run > 0.f) ? 1.f / l_run : 0.f;
  float inv_row[4];
  #pragma unroll
  for (int r = 0; r < 4; ++r) inv_row[r] = __shfl(inv, 4 * lg + r, 64);
  float* ow = o_lds + (size_t)wave * QB * D;
  #pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qq = 4 * lg + r;            // row within the wave's tile
      ow[qq * D + dt * 16 + lc] = oacc[dt][r] * inv_row[r];
    }
  }
  if (q0 < T && lg == 0) {
    // lane lc owns q column q0+lc
    if (q0 + lc < T)
      lse[(((int64_t)b * Hq + h) * T) + q0 + lc] = m_run + __logf(l_run);
  }
  __syncthreads();
  // coalesced store: each lane writes bf16x8 rows
  for (int i = tid; i < NWAVE * QB * (D / 8); i += 256) {
    const int w2 = i / (QB * D / 8);
    const int rem = i % (QB * D / 8);
    const int row = rem / (D / 8);
    const int cu = rem % (D / 8);
    const int qq = qblk * (QB * NWAVE) + w2 * QB + row;
    if (qq < T) {
      const float* src = o_lds + ((size_t)w2 * QB + row) * D + cu * 8;
      bf16x8 vv;
      #pragma unroll
      for (int j = 0; j < 8; ++j) vv.v[j] = f2bf(src[j]);
      *reinterpret_cast<bf16x8*>(o + (((int64_t)b * Hq + h) * T + qq) * D
                                 + cu * 8) = vv;
    }
  }
}

// ------------------------------------------------------------- backward
// Pass Q (dQ): q-block outer, kv loop; swapped orientation (like fwd).
//   dPT[kv][q] = mfma(A=V rows, B=dO rows)
//   PT[kv][q]  = exp(ST*scale - lse[q])   (ST recomputed)
//   dST        = PT * (dPT - delta[q]) * scale
//   dQ[q][d]  += mfma(A = relayout(dST), B = k_t rows)
template <int D>
__global__ __launch_bounds__(256)
void flash_bwd_dq_kernel(const __hip_bfloat16* __restrict__ q,
                         const __hip_bfloat16* __restrict__ k,
                         const __hip_bfloat16* __restrict__ kt,
                         const __hip_bfloat16* __restrict__ v,
                         const __hip_bfloat16* __restrict__ dout,
                         const float* __restrict__ lse,
                         const float* __restrict__ delta,
                         __hip_bfloat16* __restrict__ dq,
                         int B, int Hq, int Hkv, int T, int Tp, float scale) {
  const int qblk = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int kvh = h / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4;
  const int lc = l & 15;
  const int q0 = qblk * (QB * NWAVE) + wave * QB;
  const int q_end = min(q0 + QB, T);

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                          // KVB x D
  char* kt_lds = k_lds + KVB * D * 2;          // D x KVB (K^T rows)
  char* v_lds = kt_lds + (size_t)D * KVB * 2;  // KVB x D (V rows)
  float* s_lds = reinterpret_cast<float*>(v_lds + (size_t)KVB * D * 2);

  const int64_t bh = (int64_t)b * Hq + h;
  const int64_t bkv = (int64_t)b * Hkv + kvh;
  const __hip_bfloat16* qp = q + bh * T * D;
  const __hip_bfloat16* kp = k + bkv * T * D;
  const __hip_bfloat16* ktp = kt + bkv * D * Tp;
  const __hip_bfloat16* vp = v + bkv * T * D;
  const __hip_bfloat16* dop = dout + bh * T * D;

  // Q and dO fragments (B-frags over d)
  bf16v8 qf[D / 32], dof[D / 32];
  {
    const int row = q0 + lc;
    #pragma unroll
    for (int ks = 0; ks < D / 32; ++ks) {
      if (row < T) {
        qf[ks] = *reinterpret_cast<const bf16v8*>(
            qp + (int64_t)row * D + ks * 32 + lg * 8);
        dof[ks] = *reinterpret_cast<const bf16v8*>(
            dop + (int64_t)row * D + ks * 32 + lg * 8);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j) { qf[ks][j] = (__bf16)0.f;
                                      dof[ks][j] = (__bf16)0.f; }
      }
    }
  }
  const float lse_q = (q0 + lc < T) ? lse[bh * T + q0 + lc] : 0.f;
  const float delta_q = (q0 + lc < T) ? delta[bh * T + q0 + lc] : 0.f;

  f32x4 dqacc[D / 16];
  #pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) dqacc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int blk_q_end = min(qblk * (QB * NWAVE) + QB * NWAVE, T);
  const int n_tiles = (blk_q_end + KVB - 1) / KVB;

  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * KVB;
    const int kv_valid = min(KVB, T - kv0);
    __syncthreads();
    stage_tile<KVB, D>(k_lds, kp + (int64_t)kv0 * D, D, kv_valid, tid, 256);
    stage_tile<D, KVB>(kt_lds, ktp + kv0, Tp, D, tid, 256);
    stage_tile<KVB, D>(v_lds, vp + (int64_t)kv0 * D, D, kv_valid, tid, 256);
    __syncthreads();

    if (q0 < T && kv0 <= q_end - 1) {
      f32x4 st[KVB / 16], dpt[KVB / 16];
      #pragma unroll
      for (int s = 0; s < KVB / 16; ++s) {
        f32x4 a1 = {0.f, 0.f, 0.f, 0.f}, a2 = {0.f, 0.f, 0.f, 0.f};
        #pragma unroll
        for (int ks = 0; ks < D / 32; ++ks) {
          const bf16v8 kf = lds_frag<D>(k_lds, s * 16 + lc, ks * 4 + lg);
          const bf16v8 vf = lds_frag<D>(v_lds, s * 16 + lc, ks * 4 + lg);
          a1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qf[ks], a1, 0, 0, 0);
          a2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf, dof[ks], a2, 0, 0, 0);
        }
        st[s] = a1;
        dpt[s] = a2;
      }
      // dST = PT * (dPT - delta) * scale with causal mask
      #pragma unroll
      for (int s = 0; s < KVB / 16; ++s) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int kv = kv0 + s * 16 + 4 * lg + r;
          const int qq = q0 + lc;
          float p = 0.f;
          if (kv <= qq && qq < T && kv < T)
            p = __expf(st[s][r] * scale - lse_q);
          st[s][r] = p * (dpt[s][r] - delta_q) * scale;
        }
      }
      const bf16v8 da = relayout_frag(st[0], st[1]);
      #pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const bf16v8 ktf = lds_frag<KVB>(kt_lds, dt * 16 + lc, lg);
        dqacc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(da, ktf,
                                                            dqacc[dt], 0, 0, 0);
      }
    }
  }

  // epilogue via LDS (same as fwd)
  float* ow = s_lds + (size_t)wave * QB * D;
  #pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) {
    #pragma unroll
    for (int r = 0; r < 4; ++r)
      ow[(4 * lg + r) * D + dt * 16 + lc] = dqacc[dt][r];
  }
  __syncthreads();
  for (int i = tid; i < NWAVE * QB * (D / 8); i += 256) {
    const int w2 = i / (QB * D / 8);
    const int rem = i % (QB * D / 8);
    const int row = rem / (D / 8);
    const int cu = rem % (D / 8);
    const int qq = qblk * (QB * NWAVE) + w2 * QB + row;
    if (qq < T) {
      const float* src = s_lds + ((size_t)w2 * QB + row) * D + cu * 8;
      bf16x8 vv;
      #pragma unroll
      for (int j = 0; j < 8; ++j) vv.v[j] = f2bf(src[j]);
      *reinterpret_cast<bf16x8*>(dq + (bh * T + qq) * D + cu * 8) = vv;
    }
  }
}

// Pass KV (dK^T, dV^T): kv-block outer, UNSWAPPED orientation.
//   S[q][kv]  = mfma(A=Q rows, B=K rows);  P = exp(S*scale - lse[q])
//   dP[q][kv] = mfma(A=dO rows, B=V rows)
//   dS        = P * (dP - delta[q]) * scale
//   dV^T[d][kv] += mfma(A = dO^T rows, B = relayout(P))
//   dK^T[d][kv] += mfma(A = q^T rows,  B = relayout(dS))
// One wave owns 16 kv columns; the block owns 64 kv rows (KVT tile).
template <int D>
__global__ __launch_bounds__(256)
void flash_bwd_dkv_kernel(const __hip_bfloat16* __restrict__ q,
                          const __hip_bfloat16* __restrict__ qt,
                          const __hip_bfloat16* __restrict__ k,
                          const __hip_bfloat16* __restrict__ v,
                          const __hip_bfloat16* __restrict__ dout,
                          const __hip_bfloat16* __restrict__ dot_t,
                          const float* __restrict__ lse,
                          const float* __restrict__ delta,
                          __hip_bfloat16* __restrict__ dkt,  // (B,Hq,D,T)
                          __hip_bfloat16* __restrict__ dvt,   // per Q-HEAD
                          int B, int Hq, int Hkv, int T, int Tp, float scale) {
  constexpr int KB = 16;     // kv cols per wave
  const int kvblk = blockIdx.x;          // 64 kv rows per block
  const int h = blockIdx.y;              // q head (accumulate into kv head)
  const int b = blockIdx.z;
  const int group = Hq / Hkv;
  const int kvh = h / group;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int l = tid & 63;
  const int lg = l >> 4;
  const int lc = l & 15;
  const int kv0 = kvblk * (KB * NWAVE) + wave * KB;  // wave's 16 kv cols

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* q_lds = smem;                            // QTile x D (Q rows)
  char* do_lds = q_lds + KVB * D * 2;            // QTile x D (dO rows)
  char* qt_lds = do_lds + KVB * D * 2;           // D x QTile (Q^T rows)
  char* dot_lds = qt_lds + (size_t)D * KVB * 2;  // D x QTile (dO^T rows)

  const int64_t bh = (int64_t)b * Hq + h;
  const int64_t bkv = (int64_t)b * Hkv + kvh;
  const __hip_bfloat16* qp = q + bh * T * D;
  const __hip_bfloat16* qtp = qt + bh * D * Tp;
  const __hip_bfloat16* kp = k + bkv * T * D;
  const __hip_bfloat16* vp = v + bkv * T * D;
  const __hip_bfloat16* dop = dout + bh * T * D;
  const __hip_bfloat16* dotp = dot_t + bh * D * Tp;

  // K and V fragments for this wave's 16 kv columns (B-frags over d)
  bf16v8 kf[D / 32], vf[D / 32];
  {
    const int row = kv0 + lc;
    #pragma unroll
    for (int ks = 0; ks < D / 32; ++ks) {
      if (row < T) {
        kf[ks] = *reinterpret_cast<const bf16v8*>(
            kp + (int64_t)row * D + ks * 32 + lg * 8);
        vf[ks] = *reinterpret_cast<const bf16v8*>(
            vp + (int64_t)row * D + ks * 32 + lg * 8);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j) { kf[ks][j] = (__bf16)0.f;
                                      vf[ks][j] = (__bf16)0.f; }
      }
    }
  }

  f32x4 dkacc[D / 16], dvacc[D / 16];
  #pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) {
    dkacc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
    dvacc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  // causal: q tiles from the block's FIRST kv row onward
  const int blk_kv0 = kvblk * (KB * NWAVE);
  const int t0 = blk_kv0 / KVB;
  const int n_tiles = (T + KVB - 1) / KVB;

  for (int t = t0; t < n_tiles; ++t) {
    const int tq0 = t * KVB;                 // 32 q rows per tile
    const int q_valid = min(KVB, T - tq0);
    __syncthreads();
    stage_tile<KVB, D>(q_lds, qp + (int64_t)tq0 * D, D, q_valid, tid, 256);
    stage_tile<KVB, D>(do_lds, dop + (int64_t)tq0 * D, D, q_valid, tid, 256);
    stage_tile<D, KVB>(qt_lds, qtp + tq0, Tp, D, tid, 256);
    stage_tile<D, KVB>(dot_lds, dotp + tq0, Tp, D, tid, 256);
    __syncthreads();

    if (kv0 < T && tq0 + q_valid - 1 >= kv0) {
      f32x4 s_t[KVB / 16], dp_t[KVB / 16];
      #pragma unroll
      for (int s = 0; s < KVB / 16; ++s) {
        f32x4 a1 = {0.f, 0.f, 0.f, 0.f}, a2 = {0.f, 0.f, 0.f, 0.f};
        #pragma unroll
        for (int ks = 0; ks < D / 32; ++ks) {
          // S[q][kv]: A = Q rows (LDS), B = K frags (regs)
          const bf16v8 qfr = lds_frag<D>(q_lds, s * 16 + lc, ks * 4 + lg);
          const bf16v8 dofr = lds_frag<D>(do_lds, s * 16 + lc, ks * 4 + lg);
          a1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfr, kf[ks], a1, 0, 0, 0);
          a2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dofr, vf[ks], a2, 0, 0, 0);
        }
        s_t[s] = a1;   // D-layout: col = kv = lc + kv0, row = q = tq0+s*16+4lg+r
        dp_t[s] = a2;
      }
      f32x4 p_t[KVB / 16];
      #pragma unroll
      for (int s = 0; s < KVB / 16; ++s) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qq = tq0 + s * 16 + 4 * lg + r;
          const int kv = kv0 + lc;
          float p = 0.f;
          if (kv <= qq && qq < T && kv < T)
            p = __expf(s_t[s][r] * scale
                       - lse[bh * T + min(qq, T - 1)]);
          const float dlt = (qq < T) ? delta[bh * T + qq] : 0.f;
          p_t[s][r] = p;
          s_t[s][r] = p * (dp_t[s][r] - dlt) * scale;
        }
      }
      const bf16v8 pb = relayout_frag(p_t[0], p_t[1]);    // B[kv][q-k]
      const bf16v8 dsb = relayout_frag(s_t[0], s_t[1]);
      #pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const bf16v8 dotf = lds_frag<KVB>(dot_lds, dt * 16 + lc, lg);
        const bf16v8 qtf = lds_frag<KVB>(qt_lds, dt * 16 + lc, lg);
        dvacc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dotf, pb,
                                                            dvacc[dt], 0, 0, 0);
        dkacc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qtf, dsb,
                                                            dkacc[dt], 0, 0, 0);
      }
    }
  }

  // per-Q-HEAD bf16 outputs (B,Hq,D,T): each (kvblk, h) block owns its
  // region exclusively — no atomics; the host sums the GQA group in fp32
  // (288 GB makes the extra head-resolved image free)
  if (kv0 < T) {
    #pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int d = dt * 16 + 4 * lg + r;   // D-layout row = d
        const int kv = kv0 + lc;              // col = kv
        if (kv < T) {
          const int64_t off = (bh * D + d) * T + kv;
          dkt[off] = f2bf(dkacc[dt][r]);
          dvt[off] = f2bf(dvacc[dt][r]);
        }
      }
    }
  }
  (void)group;
}

}  // namespace

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, double scale) {
  // q: (B, Hq, T, D); k, v: (B, Hkv, T, D) — causal, bf16
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(k.scalar_type() == at::kBFloat16 &&
              v.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4);
  const int B = q.size(0), Hq = q.size(1), T = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  TORCH_CHECK(k.size(0) == B && k.size(2) == T && k.size(3) == D &&
              v.sizes() == k.sizes(), "flash_attn_fwd: k/v shape mismatch");
  TORCH_CHECK(Hq % Hkv == 0);
  TORCH_CHECK(D == 64 || D == 128, "flash_attn_fwd: head_dim 64/128 only");
  auto qc = q.contiguous(), kc = k.contiguous();
  // transpose with columns padded to a multiple of 32 (KVB): keeps every
  // bf16x8 staging read 16B-aligned and in-bounds for arbitrary T
  const int Tp = (T + 31) / 32 * 32;
  auto vt = torch::constant_pad_nd(v.transpose(2, 3), {0, Tp - T}, 0)
                .contiguous();                 // (B, Hkv, D, Tp)
  auto o = torch::empty_like(qc);
  auto lse = torch::empty({B, Hq, T}, q.options().dtype(at::kFloat));
  dim3 grid((T + QB * NWAVE - 1) / (QB * NWAVE), Hq, B), block(256);
  const size_t smem = (size_t)KVB * D * 2 * 2 + (size_t)NWAVE * QB * D * 4;
  auto stream = at::cuda::getCurrentCUDAStream();
  #define LAUNCH_FWD(DV) \
    hipLaunchKernelGGL((flash_fwd_kernel<DV>), grid, block, smem, stream, \
        reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(vt.data_ptr()), \
        reinterpret_cast<__hip_bfloat16*>(o.data_ptr()), \
        lse.data_ptr<float>(), B, Hq, Hkv, T, Tp, (float)scale)
  if (D == 128) LAUNCH_FWD(128); else LAUNCH_FWD(64);
  #undef LAUNCH_FWD
  HIP_CHECK_LAST();
  return {o, lse};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor o, torch::Tensor lse,
                                          double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(dout.scalar_type() == at::kBFloat16 &&
              k.scalar_type() == at::kBFloat16 &&
              v.scalar_type() == at::kBFloat16 &&
              o.scalar_type() == at::kBFloat16 &&
              lse.scalar_type() == at::kFloat);
  TORCH_CHECK(dout.sizes() == q.sizes() && o.sizes() == q.sizes(),
              "flash_attn_bwd: dout/o shape mismatch");
  const int B = q.size(0), Hq = q.size(1), T = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  TORCH_CHECK(k.size(0) == B && k.size(2) == T && k.size(3) == D &&
              v.sizes() == k.sizes(), "flash_attn_bwd: k/v shape mismatch");
  auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
  auto doc = dout.contiguous();
  const int Tp = (T + 31) / 32 * 32;
  auto kt = torch::constant_pad_nd(kc.transpose(2, 3), {0, Tp - T}, 0)
                .contiguous();
  auto qt = torch::constant_pad_nd(qc.transpose(2, 3), {0, Tp - T}, 0)
                .contiguous();
  auto dot_t = torch::constant_pad_nd(doc.transpose(2, 3), {0, Tp - T}, 0)
                .contiguous();
  auto delta = (doc.to(at::kFloat) * o.to(at::kFloat)).sum(-1);  // (B,Hq,T)
  auto dq = torch::empty_like(qc);
  auto dkt = torch::zeros({B, Hq, D, T}, q.options());
  auto dvt = torch::zeros({B, Hq, D, T}, q.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 block(256);
  dim3 grid_q((T + QB * NWAVE - 1) / (QB * NWAVE), Hq, B);
  const size_t smem_q = (size_t)KVB * D * 2 * 3 + (size_t)NWAVE * QB * D * 4;
  #define LAUNCH_DQ(DV) \
    hipLaunchKernelGGL((flash_bwd_dq_kernel<DV>), grid_q, block, smem_q, \
        stream, \
        reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(kt.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(doc.data_ptr()), \
        lse.data_ptr<float>(), delta.data_ptr<float>(), \
        reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()), \
        B, Hq, Hkv, T, Tp, (float)scale)
  if (D == 128) LAUNCH_DQ(128); else LAUNCH_DQ(64);
  #undef LAUNCH_DQ
  dim3 grid_kv((T + 16 * NWAVE - 1) / (16 * NWAVE), Hq, B);
  const size_t smem_kv = (size_t)KVB * D * 2 * 4;
  #define LAUNCH_DKV(DV) \
    hipLaunchKernelGGL((flash_bwd_dkv_kernel<DV>), grid_kv, block, smem_kv, \
        stream, \
        reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(qt.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(doc.data_ptr()), \
        reinterpret_cast<const __hip_bfloat16*>(dot_t.data_ptr()), \
        lse.data_ptr<float>(), delta.data_ptr<float>(), \
        reinterpret_cast<__hip_bfloat16*>(dkt.data_ptr()), \
        reinterpret_cast<__hip_bfloat16*>(dvt.data_ptr()), \
        B, Hq, Hkv, T, Tp, (float)scale)
  if (D == 128) LAUNCH_DKV(128); else LAUNCH_DKV(64);
  #undef LAUNCH_DKV
  HIP_CHECK_LAST();
  const int group = Hq / Hkv;
  auto dk = dkt.view({B, Hkv, group, D, T}).to(at::kFloat).sum(2)
                .transpose(2, 3).contiguous().to(at::kBFloat16);
  auto dv = dvt.view({B, Hkv, group, D, T}).to(at::kFloat).sum(2)
                .transpose(2, 3).contiguous().to(at::kBFloat16);
  return {dq, dk, dv};
}
