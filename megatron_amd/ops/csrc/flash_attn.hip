// FlashAttention-2-equivalent fused attention for MI355X (gfx950, CDNA4).
//
// Replaces the reference's external flash-attn dependency
// (transformer.py:9,369,528-552): causal + sliding-window attention with
// native GQA/MQA head broadcast (no K/V expansion), forward + backward.
//
// Design (v1, correctness-first MFMA structure — tuned variants iterate on
// this skeleton):
//  - mfma_f32_16x16x32_bf16 tiles; 256-thread blocks = 4 waves;
//    BLOCK_M = BLOCK_N = 64; head dims 64 / 128.
//  - Operand layouts chosen so Q/K A- and B-fragments load as contiguous
//    16-byte chunks straight from HBM ([S][D] rows match the fragment's
//    8-element k-runs); only P (QK^T output -> PV input) round-trips
//    through a padded LDS strip, and V/dO/Q B-fragments that need the
//    d-major orientation are staged in LDS tiles.
//  - online softmax in fp32 registers; cross-lane row reductions are
//    __shfl_xor within 16-lane groups (wave64).
//  - backward is split k-parallel (dK/dV, GQA-group loop => register
//    accumulation, no atomics) and q-parallel (dQ) plus a small
//    delta = rowsum(dO*O) preprocess kernel, using the stored
//    logsumexp as in FlashAttention-2.
//
// Fragment maps (gfx950 mfma_f32_16x16x32_bf16, verified by the
// tests/gpu parity suite):
//   A[i][k]: lane l holds A[l%16][(l/16)*8 + j], j = 0..7
//   B[k][j]: lane l holds B[(l/16)*8 + j][l%16]
//   C[r][c]: lane l, reg r holds C[(l/16)*4 + r][l%16]

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <vector>

namespace {

using frag_b16 = __attribute__((ext_vector_type(8))) short;
using frag_f32 = __attribute__((ext_vector_type(4))) float;

constexpr int kBlockM = 64;
constexpr int kBlockN = 64;
constexpr int kThreads = 256;
constexpr int kStrip = 72;  // padded LDS strip stride (bf16 elems) per 64 cols

__device__ __forceinline__ frag_b16 load_frag_global(const __hip_bfloat16* p) {
  // 8 contiguous bf16 = 16 B
  uint4 u = *reinterpret_cast<const uint4*>(p);
  union {
    uint4 u;
    frag_b16 f;
  } cvt;
  cvt.u = u;
  return cvt.f;
}

__device__ __forceinline__ float group16_max(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) {
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  }
  return v;
}

__device__ __forceinline__ float group16_sum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) {
    v += __shfl_xor(v, off, WAVE_SIZE);
  }
  return v;
}

// ---------------------------------------------------------------------------
// forward

template <int D>
__global__ __launch_bounds__(kThreads) void fa_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse, int B, int Sq, int Sk, int Hq, int Hkv,
    float scale, int causal, int window) {
  constexpr int KFRAGS = D / 32;   // A/B fragment K-steps
  constexpr int DTILES = D / 16;   // output col tiles
  constexpr int VPAD = D + 8;

  __shared__ __hip_bfloat16 p_lds[4][16 * kStrip];
  __shared__ __hip_bfloat16 v_lds[kBlockN * VPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;    // A-frag row / C col
  const int kgroup = lane >> 4;         // 0..3

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long q_base = ((long)b * Sq * Hq + h) * D;
  const long k_base = ((long)b * Sk * Hkv + hkv) * D;
  const long v_base = k_base;
  const long o_base = q_base;
  const int rq = Hq * D;   // row stride in q/out
  const int rk = Hkv * D;  // row stride in k/v

  const int qrow0 = qb * kBlockM + wave * 16;  // wave's first q row
  const int skq = Sk - Sq;

  // Q A-fragments (row = qrow0 + lane%16, k-run = kgroup*8)
  frag_b16 qf[KFRAGS];
  {
    int qrow = qrow0 + row_in_tile;
    int qr = qrow < Sq ? qrow : Sq - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      qf[kk] = load_frag_global(q + q_base + (long)qr * rq + kk * 32 +
                                kgroup * 8);
    }
  }

  frag_f32 o_acc[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) o_acc[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }

  // key-block range for the whole block (waves mask individually)
  int kb_end = (Sk + kBlockN - 1) / kBlockN;
  if (causal) {
    int max_qrow = qb * kBlockM + kBlockM - 1;
    int max_key = max_qrow + skq;
    kb_end = min(kb_end, max_key / kBlockN + 1);
  }
  int kb_start = 0;
  if (window > 0) {
    int min_qrow = qb * kBlockM;
    int min_key = min_qrow + skq - window + 1;
    if (min_key > 0) kb_start = min_key / kBlockN;
  }

  for (int kb = kb_start; kb < kb_end; ++kb) {
    const int kstart = kb * kBlockN;

    // stage V tile cooperatively: 64 rows x D cols
    {
      const int vec_per_row = D / 8;  // uint4 count per row
      for (int idx = threadIdx.x; idx < kBlockN * vec_per_row;
           idx += kThreads) {
        int krow = idx / vec_per_row;
        int c8 = (idx % vec_per_row) * 8;
        int key = kstart + krow;
        uint4 val;
        if (key < Sk) {
          val = *reinterpret_cast<const uint4*>(v + v_base + (long)key * rk +
                                                c8);
        } else {
          val = make_uint4(0, 0, 0, 0);
        }
        *reinterpret_cast<uint4*>(&v_lds[krow * VPAD + c8]) = val;
      }
    }
    __syncthreads();

    // S = Q K^T for this wave's 16 rows x 64 cols
    frag_f32 st[4];
#pragma unroll
    for (int t = 0; t < 4; ++t) st[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      int key = kstart + t * 16 + row_in_tile;
      int kr = key < Sk ? key : Sk - 1;
#pragma unroll
      for (int kk = 0; kk < KFRAGS; ++kk) {
        frag_b16 bf = load_frag_global(k + k_base + (long)kr * rk + kk * 32 +
                                       kgroup * 8);
        st[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[kk], bf, st[t], 0,
                                                        0, 0);
      }
    }

    // mask + scale into s[t][r]; track masked-ness
    float s_val[4][4];
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int col = kstart + t * 16 + row_in_tile;
        int row = qrow0 + kgroup * 4 + r;
        bool masked = (col >= Sk) || (row >= Sq);
        if (causal && col > row + skq) masked = true;
        if (window > 0 && col < row + skq - window + 1) masked = true;
        s_val[t][r] = masked ? -1e30f : st[t][r] * scale;
      }
    }

    // NOTE: C layout row = kgroup*4 + r with col = row_in_tile; the row-wise
    // reduction is over col => over the 16 lanes of the row_in_tile group.
    float m_new[4], alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float pm = fmaxf(fmaxf(s_val[0][r], s_val[1][r]),
                       fmaxf(s_val[2][r], s_val[3][r]));
      pm = group16_max(pm);
      m_new[r] = fmaxf(m_run[r], pm);
      alpha[r] = __expf(m_run[r] - m_new[r]);
      m_run[r] = m_new[r];
    }

    float p_val[4][4];
    float rowsum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float acc = 0.f;
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        float pv = (s_val[t][r] < -1e29f)
                       ? 0.f
                       : __expf(s_val[t][r] - m_new[r]);
        p_val[t][r] = pv;
        acc += pv;
      }
      rowsum[r] = group16_sum(acc);
      l_run[r] = l_run[r] * alpha[r] + rowsum[r];
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        o_acc[t][r] *= alpha[r];
      }
    }

    // P -> LDS strip [row 16][col 64] stride kStrip
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = kgroup * 4 + r;
        int col = t * 16 + row_in_tile;
        p_lds[wave][row * kStrip + col] = __float2bfloat16(p_val[t][r]);
      }
    }
    __syncthreads();  // also covers v_lds reuse across iterations

    // O += P V
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      // A-frag: P rows (lane%16), k-run = kk2*32 + kgroup*8
      frag_b16 pf;
      {
        const __hip_bfloat16* src =
            &p_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8];
        uint4 u = *reinterpret_cast<const uint4*>(src);
        union {
          uint4 u;
          frag_b16 f;
        } cvt;
        cvt.u = u;
        pf = cvt.f;
      }
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        // B-frag: V[key = kk2*32 + kgroup*8 + j][d = t*16 + lane%16]
        frag_b16 vf;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int key = kk2 * 32 + kgroup * 8 + j;
          vf[j] = __bfloat16_as_short(
              v_lds[key * VPAD + t * 16 + row_in_tile]);
        }
        o_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, o_acc[t],
                                                           0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: normalize + store O (bf16) and lse (fp32)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = qrow0 + kgroup * 4 + r;
    float inv_l = l_run[r] > 0.f ? 1.0f / l_run[r] : 0.f;
    if (row < Sq) {
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        out[o_base + (long)row * rq + t * 16 + row_in_tile] =
            __float2bfloat16(o_acc[t][r] * inv_l);
      }
      if (row_in_tile == 0) {
        lse[((long)b * Hq + h) * Sq + row] =
            m_run[r] + logf(fmaxf(l_run[r], 1e-30f));
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward preprocess: delta[b,h,s] = sum_d dO * O  (fp32)

template <int D>
__global__ void fa_bwd_delta_kernel(const __hip_bfloat16* __restrict__ dout,
                                    const __hip_bfloat16* __restrict__ out,
                                    float* __restrict__ delta, int B, int Sq,
                                    int Hq) {
  const long row = blockIdx.x;  // b*Sq*Hq rows? use (b*Hq+h)*Sq + s layout
  const int s = row % Sq;
  const long bh = row / Sq;
  const int b = bh / Hq;
  const int h = bh % Hq;
  const long base = (((long)b * Sq + s) * Hq + h) * D;
  float acc = 0.f;
  for (int i = threadIdx.x; i < D; i += 64) {
    acc += __bfloat162float(dout[base + i]) * __bfloat162float(out[base + i]);
  }
  acc = wave_reduce_sum(acc);
  if (threadIdx.x == 0) delta[row] = acc;
}

// ---------------------------------------------------------------------------
// backward dK/dV (k-parallel; loops over q heads of the GQA group and q
// blocks; accumulates dK/dV in registers — no atomics)

template <int D>
__global__ __launch_bounds__(kThreads) void fa_bwd_dkdv_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const __hip_bfloat16* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, __hip_bfloat16* __restrict__ dk,
    __hip_bfloat16* __restrict__ dv, int B, int Sq, int Sk, int Hq, int Hkv,
    float scale, int causal, int window) {
  constexpr int KFRAGS = D / 32;
  constexpr int DTILES = D / 16;
  constexpr int TPAD = D + 8;

  __shared__ __hip_bfloat16 pt_lds[4][16 * kStrip];   // P^T strips
  __shared__ __hip_bfloat16 dst_lds[4][16 * kStrip];  // dS^T strips
  __shared__ __hip_bfloat16 do_lds[kBlockM * TPAD];   // dO tile
  __shared__ __hip_bfloat16 q_tile_lds[kBlockM * TPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;
  const int kgroup = lane >> 4;

  const int kb = blockIdx.x;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int gqa = Hq / Hkv;

  const long k_base = ((long)b * Sk * Hkv + hkv) * D;
  const int rk = Hkv * D;
  const int rq = Hq * D;
  const int skq = Sk - Sq;

  const int key0 = kb * kBlockN + wave * 16;  // wave's first key

  // K and V A-fragments for this wave's 16 keys
  frag_b16 ka[KFRAGS], va[KFRAGS];
  {
    int key = key0 + row_in_tile;
    int kr = key < Sk ? key : Sk - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      ka[kk] = load_frag_global(k + k_base + (long)kr * rk + kk * 32 +
                                kgroup * 8);
      va[kk] = load_frag_global(v + k_base + (long)kr * rk + kk * 32 +
                                kgroup * 8);
    }
  }

  frag_f32 dv_acc[DTILES], dk_acc[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) {
    dv_acc[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
    dk_acc[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
  }

  // q-block range for this k-block
  int qb_start = 0;
  if (causal) {
    int min_key = kb * kBlockN;
    int min_qrow = min_key - skq;  // first q row that can attend min_key
    if (min_qrow > 0) qb_start = min_qrow / kBlockM;
  }
  int qb_end = (Sq + kBlockM - 1) / kBlockM;
  if (window > 0) {
    // q rows beyond key + window - 1 - skq can't see this block
    int max_key = kb * kBlockN + kBlockN - 1;
    int max_qrow = max_key + window - 1 - skq;
    qb_end = min(qb_end, max_qrow / kBlockM + 1);
  }

  for (int hq = hkv * gqa; hq < (hkv + 1) * gqa; ++hq) {
    const long q_base = ((long)b * Sq * Hq + hq) * D;
    const float* lse_h = lse + ((long)b * Hq + hq) * Sq;
    const float* delta_h = delta + ((long)b * Hq + hq) * Sq;

    for (int qb = qb_start; qb < qb_end; ++qb) {
      const int qstart = qb * kBlockM;

      // stage dO and Q tiles cooperatively
      {
        const int vec_per_row = D / 8;
        for (int idx = threadIdx.x; idx < kBlockM * vec_per_row;
             idx += kThreads) {
          int qrow = idx / vec_per_row;
          int c8 = (idx % vec_per_row) * 8;
          int sq_idx = qstart + qrow;
          uint4 dval, qval;
          if (sq_idx < Sq) {
            dval = *reinterpret_cast<const uint4*>(dout + q_base +
                                                   (long)sq_idx * rq + c8);
            qval = *reinterpret_cast<const uint4*>(q + q_base +
                                                   (long)sq_idx * rq + c8);
          } else {
            dval = make_uint4(0, 0, 0, 0);
            qval = make_uint4(0, 0, 0, 0);
          }
          *reinterpret_cast<uint4*>(&do_lds[qrow * TPAD + c8]) = dval;
          *reinterpret_cast<uint4*>(&q_tile_lds[qrow * TPAD + c8]) = qval;
        }
      }
      __syncthreads();

      // S^T = K Q^T : rows = keys (this wave's 16), cols = 64 q rows
      frag_f32 stt[4], dpt[4];
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        stt[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
        dpt[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
      }
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        int qrow = qstart + t * 16 + row_in_tile;
        int qr = qrow < Sq ? qrow : Sq - 1;
#pragma unroll
        for (int kk = 0; kk < KFRAGS; ++kk) {
          // B-frag Q^T: Q[qrow l%16][d-run]
          frag_b16 qbf = load_frag_global(q + q_base + (long)qr * rq +
                                          kk * 32 + kgroup * 8);
          stt[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ka[kk], qbf,
                                                           stt[t], 0, 0, 0);
          // B-frag dO^T for dP^T = V dO^T
          frag_b16 dbf = load_frag_global(dout + q_base + (long)qr * rq +
                                          kk * 32 + kgroup * 8);
          dpt[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(va[kk], dbf,
                                                           dpt[t], 0, 0, 0);
        }
      }

      // P^T = exp(S^T*scale - lse), dS^T = P^T*(dP^T - delta)*scale
#pragma unroll
      for (int t = 0; t < 4; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int key = kb * kBlockN + wave * 16 + kgroup * 4 + r;
          int qrow = qstart + t * 16 + row_in_tile;
          bool masked = (key >= Sk) || (qrow >= Sq);
          if (causal && key > qrow + skq) masked = true;
          if (window > 0 && key < qrow + skq - window + 1) masked = true;
          float pt = 0.f, dst = 0.f;
          if (!masked) {
            float l = lse_h[qrow];
            pt = __expf(stt[t][r] * scale - l);
            dst = pt * (dpt[t][r] - delta_h[qrow]) * scale;
          }
          int lrow = kgroup * 4 + r;    // key within wave strip
          int lcol = t * 16 + row_in_tile;
          pt_lds[wave][lrow * kStrip + lcol] = __float2bfloat16(pt);
          dst_lds[wave][lrow * kStrip + lcol] = __float2bfloat16(dst);
        }
      }
      __syncthreads();

      // dV += P^T dO ; dK += dS^T Q
#pragma unroll
      for (int kk2 = 0; kk2 < 2; ++kk2) {
        frag_b16 ptf, dstf;
        {
          const __hip_bfloat16* src1 =
              &pt_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8];
          const __hip_bfloat16* src2 =
              &dst_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8];
          union {
            uint4 u;
            frag_b16 f;
          } c1, c2;
          c1.u = *reinterpret_cast<const uint4*>(src1);
          c2.u = *reinterpret_cast<const uint4*>(src2);
          ptf = c1.f;
          dstf = c2.f;
        }
#pragma unroll
        for (int t = 0; t < DTILES; ++t) {
          frag_b16 dof, qtf;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            int qrow = kk2 * 32 + kgroup * 8 + j;
            dof[j] = __bfloat16_as_short(
                do_lds[qrow * TPAD + t * 16 + row_in_tile]);
            qtf[j] = __bfloat16_as_short(
                q_tile_lds[qrow * TPAD + t * 16 + row_in_tile]);
          }
          dv_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ptf, dof,
                                                              dv_acc[t], 0,
                                                              0, 0);
          dk_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dstf, qtf,
                                                              dk_acc[t], 0,
                                                              0, 0);
        }
      }
      __syncthreads();
    }
  }

  // store dK/dV (C layout: row = key kgroup*4+r, col = d)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int key = kb * kBlockN + wave * 16 + kgroup * 4 + r;
    if (key < Sk) {
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        dk[k_base + (long)key * rk + t * 16 + row_in_tile] =
            __float2bfloat16(dk_acc[t][r]);
        dv[k_base + (long)key * rk + t * 16 + row_in_tile] =
            __float2bfloat16(dv_acc[t][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward dQ (q-parallel)

template <int D>
__global__ __launch_bounds__(kThreads) void fa_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const __hip_bfloat16* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, __hip_bfloat16* __restrict__ dq, int B,
    int Sq, int Sk, int Hq, int Hkv, float scale, int causal, int window) {
  constexpr int KFRAGS = D / 32;
  constexpr int DTILES = D / 16;
  constexpr int TPAD = D + 8;

  __shared__ __hip_bfloat16 ds_lds[4][16 * kStrip];
  __shared__ __hip_bfloat16 k_tile_lds[kBlockN * TPAD];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;
  const int kgroup = lane >> 4;

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long q_base = ((long)b * Sq * Hq + h) * D;
  const long k_base = ((long)b * Sk * Hkv + hkv) * D;
  const int rq = Hq * D;
  const int rk = Hkv * D;
  const int skq = Sk - Sq;
  const int qrow0 = qb * kBlockM + wave * 16;

  const float* lse_h = lse + ((long)b * Hq + h) * Sq;
  const float* delta_h = delta + ((long)b * Hq + h) * Sq;

  // Q and dO A-fragments for this wave's rows
  frag_b16 qf[KFRAGS], dof[KFRAGS];
  {
    int qrow = qrow0 + row_in_tile;
    int qr = qrow < Sq ? qrow : Sq - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      qf[kk] = load_frag_global(q + q_base + (long)qr * rq + kk * 32 +
                                kgroup * 8);
      dof[kk] = load_frag_global(dout + q_base + (long)qr * rq + kk * 32 +
                                 kgroup * 8);
    }
  }

  frag_f32 dq_acc[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) dq_acc[t] = frag_f32{0.f, 0.f, 0.f, 0.f};

  int kb_end = (Sk + kBlockN - 1) / kBlockN;
  if (causal) {
    int max_qrow = qb * kBlockM + kBlockM - 1;
    kb_end = min(kb_end, (max_qrow + skq) / kBlockN + 1);
  }
  int kb_start = 0;
  if (window > 0) {
    int min_key = qb * kBlockM + skq - window + 1;
    if (min_key > 0) kb_start = min_key / kBlockN;
  }

  float lse_r[4], delta_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = qrow0 + kgroup * 4 + r;
    lse_r[r] = row < Sq ? lse_h[row] : 0.f;
    delta_r[r] = row < Sq ? delta_h[row] : 0.f;
  }

  for (int kb = kb_start; kb < kb_end; ++kb) {
    const int kstart = kb * kBlockN;

    // stage K tile
    {
      const int vec_per_row = D / 8;
      for (int idx = threadIdx.x; idx < kBlockN * vec_per_row;
           idx += kThreads) {
        int krow = idx / vec_per_row;
        int c8 = (idx % vec_per_row) * 8;
        int key = kstart + krow;
        uint4 val;
        if (key < Sk) {
          val = *reinterpret_cast<const uint4*>(k + k_base + (long)key * rk +
                                                c8);
        } else {
          val = make_uint4(0, 0, 0, 0);
        }
        *reinterpret_cast<uint4*>(&k_tile_lds[krow * TPAD + c8]) = val;
      }
    }
    __syncthreads();

    // S and dP tiles
    frag_f32 st[4], dp[4];
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      st[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
      dp[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      int key = kstart + t * 16 + row_in_tile;
      int kr = key < Sk ? key : Sk - 1;
#pragma unroll
      for (int kk = 0; kk < KFRAGS; ++kk) {
        frag_b16 kbf = load_frag_global(k + k_base + (long)kr * rk + kk * 32 +
                                        kgroup * 8);
        st[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[kk], kbf, st[t], 0,
                                                        0, 0);
        frag_b16 vbf = load_frag_global(v + k_base + (long)kr * rk + kk * 32 +
                                        kgroup * 8);
        dp[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof[kk], vbf, dp[t],
                                                        0, 0, 0);
      }
    }

    // dS = P*(dP - delta)*scale -> LDS strip
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int col = kstart + t * 16 + row_in_tile;
        int row = qrow0 + kgroup * 4 + r;
        bool masked = (col >= Sk) || (row >= Sq);
        if (causal && col > row + skq) masked = true;
        if (window > 0 && col < row + skq - window + 1) masked = true;
        float ds = 0.f;
        if (!masked) {
          float p = __expf(st[t][r] * scale - lse_r[r]);
          ds = p * (dp[t][r] - delta_r[r]) * scale;
        }
        int lrow = kgroup * 4 + r;
        int lcol = t * 16 + row_in_tile;
        ds_lds[wave][lrow * kStrip + lcol] = __float2bfloat16(ds);
      }
    }
    __syncthreads();

    // dQ += dS K
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      frag_b16 dsf;
      {
        const __hip_bfloat16* src =
            &ds_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8];
        union {
          uint4 u;
          frag_b16 f;
        } cvt;
        cvt.u = *reinterpret_cast<const uint4*>(src);
        dsf = cvt.f;
      }
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        frag_b16 ktf;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int key = kk2 * 32 + kgroup * 8 + j;
          ktf[j] = __bfloat16_as_short(
              k_tile_lds[key * TPAD + t * 16 + row_in_tile]);
        }
        dq_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, ktf,
                                                            dq_acc[t], 0, 0,
                                                            0);
      }
    }
    __syncthreads();
  }

  // store dQ
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = qrow0 + kgroup * 4 + r;
    if (row < Sq) {
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        dq[q_base + (long)row * rq + t * 16 + row_in_tile] =
            __float2bfloat16(dq_acc[t][r]);
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, bool causal,
                                          double softmax_scale,
                                          int64_t window_size) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 4 && q.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16,
              "flash_attn: bf16 only (got ", q.scalar_type(), ")");
  int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  int Sk = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(D == 64 || D == 128, "flash_attn: head dim must be 64/128");
  TORCH_CHECK(Hq % Hkv == 0);

  auto out = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, Sq},
                          q.options().dtype(torch::kFloat32));
  dim3 grid((Sq + kBlockM - 1) / kBlockM, Hq, B);
  auto stream = c10::hip::getCurrentHIPStream();
  int win = window_size > 0 ? (int)window_size : 0;
  if (D == 128) {
    hipLaunchKernelGGL((fa_fwd_kernel<128>), grid, dim3(kThreads), 0, stream,
                       (const __hip_bfloat16*)q.data_ptr(),
                       (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(),
                       (__hip_bfloat16*)out.data_ptr(),
                       lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                       (float)softmax_scale, causal ? 1 : 0, win);
  } else {
    hipLaunchKernelGGL((fa_fwd_kernel<64>), grid, dim3(kThreads), 0, stream,
                       (const __hip_bfloat16*)q.data_ptr(),
                       (const __hip_bfloat16*)k.data_ptr(),
                       (const __hip_bfloat16*)v.data_ptr(),
                       (__hip_bfloat16*)out.data_ptr(),
                       lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                       (float)softmax_scale, causal ? 1 : 0, win);
  }
  return {out, lse};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor out, torch::Tensor lse,
                                          bool causal, double softmax_scale,
                                          int64_t window_size) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous());
  int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  int Sk = k.size(1), Hkv = k.size(2);
  int win = window_size > 0 ? (int)window_size : 0;

  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto delta = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();

  long rows = (long)B * Hq * Sq;
#define LAUNCH_BWD(DD)                                                        \
  do {                                                                        \
    hipLaunchKernelGGL((fa_bwd_delta_kernel<DD>), dim3(rows), dim3(64), 0,    \
                       stream, (const __hip_bfloat16*)dout.data_ptr(),        \
                       (const __hip_bfloat16*)out.data_ptr(),                 \
                       delta.data_ptr<float>(), B, Sq, Hq);                   \
    dim3 gridk((Sk + kBlockN - 1) / kBlockN, Hkv, B);                         \
    hipLaunchKernelGGL((fa_bwd_dkdv_kernel<DD>), gridk, dim3(kThreads), 0,    \
                       stream, (const __hip_bfloat16*)q.data_ptr(),           \
                       (const __hip_bfloat16*)k.data_ptr(),                   \
                       (const __hip_bfloat16*)v.data_ptr(),                   \
                       (const __hip_bfloat16*)dout.data_ptr(),                \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),        \
                       (__hip_bfloat16*)dk.data_ptr(),                        \
                       (__hip_bfloat16*)dv.data_ptr(), B, Sq, Sk, Hq, Hkv,    \
                       (float)softmax_scale, causal ? 1 : 0, win);            \
    dim3 gridq((Sq + kBlockM - 1) / kBlockM, Hq, B);                          \
    hipLaunchKernelGGL((fa_bwd_dq_kernel<DD>), gridq, dim3(kThreads), 0,      \
                       stream, (const __hip_bfloat16*)q.data_ptr(),           \
                       (const __hip_bfloat16*)k.data_ptr(),                   \
                       (const __hip_bfloat16*)v.data_ptr(),                   \
                       (const __hip_bfloat16*)dout.data_ptr(),                \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),        \
                       (__hip_bfloat16*)dq.data_ptr(), B, Sq, Sk, Hq, Hkv,    \
                       (float)softmax_scale, causal ? 1 : 0, win);            \
  } while (0)

  if (D == 128) {
    LAUNCH_BWD(128);
  } else {
    TORCH_CHECK(D == 64);
    LAUNCH_BWD(64);
  }
  return {dq, dk, dv};
}
