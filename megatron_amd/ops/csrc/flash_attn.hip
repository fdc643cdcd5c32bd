// FlashAttention-2-equivalent fused attention for MI355X (gfx950, CDNA4).
//
// Replaces the reference's external flash-attn dependency
// (transformer.py:9,369,528-552): causal + sliding-window attention with
// native GQA/MQA head broadcast (no K/V expansion), forward + backward.
//
// v4 design (measured on MI355X; see profiles/):
//  - mfma_f32_16x16x32_bf16 tiles; 256-thread blocks = 4 waves;
//    BLOCK_M = BLOCK_N = 64; head dims 64 / 128.
//  - every K/V/dO/Q tile is staged ONCE per block into a padded LDS
//    "tr image" ([row/4] tile-rows of (D/16) [4][16] segments, +8-element
//    row pad) that serves BOTH fragment orientations conflict-free:
//      * d-run A/B fragments (8 contiguous head-dims of one row) as single
//        16-byte ds_read_b128 — lanes' 4-dword footprints interleave across
//        the 64 banks thanks to the +8 pad;
//      * row-run B fragments (8 contiguous rows at one head-dim) via the
//        gfx950 hardware transpose read ds_read_b64_tr_b16 (lane mapping
//        verified empirically with ops/csrc/probe_tr.hip: per-lane address
//        base + (lane&15)*8B delivers lds16[base + (lane&15) + 16*j]).
//    The inner loop therefore touches HBM only through the cooperative
//    staging copies, which are double-buffered so the next tile's loads
//    overlap this tile's MFMAs with ONE barrier per iteration (v2/v3's
//    per-wave global B-fragment loads serialized a load->MFMA chain on
//    every k-step — ~70% SQ_WAIT_ANY in PMC, profiles/).
//  - online softmax in fp32 registers; row reductions are __shfl_xor within
//    16-lane groups (wave64); P/dS round-trip through wave-private padded
//    strips (no barrier).
//  - backward splits k-parallel dK/dV (flattened (gqa-head, q-block) loop =>
//    register accumulation, no atomics) and q-parallel dQ, plus a
//    delta = rowsum(dO*O) preprocess, using the stored logsumexp
//    (FlashAttention-2 scheme).
//
// Fragment maps (gfx950 mfma_f32_16x16x32_bf16, verified by the GPU parity
// suite):
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
using v4s = __attribute__((ext_vector_type(4))) short;

constexpr int kBlockM = 64;
constexpr int kBlockN = 64;
constexpr int kThreads = 256;
constexpr int kStrip = 72;  // padded strip stride (bf16), P/dS tiles

// per-tensor strides in elements (batch, seq, head); the head_dim axis must
// be dense. Lets q/k/v/out/grads be strided VIEWS — sbhd transposes and the
// per-head slices of a fused QKV projection — with no host-side copies.
struct TStr {
  long bs, ss, hs;
};

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

// Attention-dropout keep decision for element (row=bh_row%..., col). The
// (b,h,row,col) -> philox mapping is shared by forward and both backward
// kernels (and mirrored in tests/philox_ref.py + the CPU reference), so the
// backward regenerates the forward's mask exactly.
__device__ __forceinline__ bool attn_drop_keep(unsigned long long seed,
                                               unsigned long long offset,
                                               long long bh_row, long long Skp,
                                               int col, float p) {
  unsigned long long e = (unsigned long long)bh_row * Skp + col;
  uint4 r = philox10_ctr64(seed, offset, e >> 2);
  unsigned v = (&r.x)[e & 3];
  return uint_to_uniform(v) > p;
}

// --- padded tr-image -------------------------------------------------------

// Bank-conflict-free tile-row geometry (derived analytically from the r02
// PMC finding of +39% SQ_LDS_BANK_CONFLICT with the old D*4+8 stride, then
// re-measured):
//  - stride D*4+128 elems (== 0 mod 128) makes a d-run ds_read_b128's 16
//    hardware-group lanes (rows t*16+0..15 at one col) land on 16 DISTINCT
//    4-dword bank ranges = all 64 banks exactly once, via
//  - the alternating +64-elem shift per tile-row PAIR (((rt>>1)&1)*64),
//    which also puts a tr-read's partner tile-rows (ktile, ktile+2) exactly
//    32 banks apart, so the 2x32-lane ds_read_b64_tr_b16 groups split the
//    bank space instead of colliding.
template <int D>
constexpr int tr_stride() {
  return D * 4 + 128;
}

template <int D>
constexpr int tr_elems(int rows) {
  return (rows / 4) * tr_stride<D>();
}

__device__ __forceinline__ int tr_shift(int rowtile) {
  return ((rowtile >> 1) & 1) * 64;
}

template <int D>
__device__ __forceinline__ int tr_off(int row, int col) {
  return (row >> 2) * tr_stride<D>() + tr_shift(row >> 2) + (col >> 4) * 64 +
         (row & 3) * 16 + (col & 15);
}

__device__ __forceinline__ frag_b16 lds_read16(const __hip_bfloat16* p) {
  union {
    uint4 u;
    frag_b16 f;
  } cvt;
  cvt.u = *reinterpret_cast<const uint4*>(p);
  return cvt.f;
}

__device__ __forceinline__ frag_b16 global_read16(const __hip_bfloat16* p) {
  union {
    uint4 u;
    frag_b16 f;
  } cvt;
  cvt.u = *reinterpret_cast<const uint4*>(p);
  return cvt.f;
}

// d-run fragment: 8 contiguous head-dims of row `row`, starting at d0
// (d0 % 16 in {0, 8}).
template <int D>
__device__ __forceinline__ frag_b16 img_dfrag(const __hip_bfloat16* img,
                                              int row, int d0) {
  return lds_read16(&img[tr_off<D>(row, d0)]);
}

// row-run B fragment: B[row0 + (l>>4)*8 + jj][dtile*16 + (l&15)] via two
// hardware transpose reads.
template <int D>
__device__ __forceinline__ frag_b16 tr_bfrag(const __hip_bfloat16* img,
                                             int row0, int dtile, int lane) {
  const int c = lane & 15;
  const int ktile = (row0 >> 2) + (lane >> 4) * 2;
  const int b1 = ktile * tr_stride<D>() + tr_shift(ktile) + dtile * 64;
  const int b2 =
      (ktile + 1) * tr_stride<D>() + tr_shift(ktile + 1) + dtile * 64;
  const __attribute__((address_space(3))) v4s* p1 =
      (const __attribute__((address_space(3))) v4s*)&img[b1] + c;
  const __attribute__((address_space(3))) v4s* p2 =
      (const __attribute__((address_space(3))) v4s*)&img[b2] + c;
  v4s lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) v4s*)p1);
  v4s hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) v4s*)p2);
  frag_b16 f;
  f[0] = lo[0];
  f[1] = lo[1];
  f[2] = lo[2];
  f[3] = lo[3];
  f[4] = hi[0];
  f[5] = hi[1];
  f[6] = hi[2];
  f[7] = hi[3];
  return f;
}

// cooperative stage of a [rows x D] bf16 tile from global into a tr image
template <int D, int ROWS, int NT = kThreads>
__device__ __forceinline__ void stage_tr_image(
    __hip_bfloat16* img, const __hip_bfloat16* src_base, long row_stride,
    int first_row, int max_row) {
  const int vec_per_row = D / 8;
  for (int idx = threadIdx.x; idx < ROWS * vec_per_row; idx += NT) {
    int r = idx / vec_per_row;
    int c8 = (idx % vec_per_row) * 8;
    int row = first_row + r;
    uint4 val;
    if (row < max_row) {
      val = *reinterpret_cast<const uint4*>(src_base + (long)row * row_stride +
                                            c8);
    } else {
      val = make_uint4(0, 0, 0, 0);
    }
    *reinterpret_cast<uint4*>(&img[tr_off<D>(r, c8)]) = val;
  }
}

// ---------------------------------------------------------------------------
// forward

template <int D, int NW, bool DROP = false>
__global__ __launch_bounds__(NW * 64) void fa_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse, int B, int Sq, int Sk, int Hq, int Hkv,
    float scale, int causal, int window, TStr qs, TStr ks, TStr vs, TStr os,
    float drop_p = 0.f, unsigned long long drop_seed = 0,
    unsigned long long drop_offset = 0) {
  constexpr int KFRAGS = D / 32;
  constexpr int DTILES = D / 16;
  constexpr int IMG = tr_elems<D>(kBlockN);
  constexpr int BM = NW * 16;       // q rows per workgroup
  constexpr int NT = NW * 64;       // threads

  __shared__ __hip_bfloat16 p_lds[NW][16 * kStrip];
  __shared__ __hip_bfloat16 k_img[2][IMG];
  __shared__ __hip_bfloat16 v_img[2][IMG];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;
  const int kgroup = lane >> 4;

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long q_base = (long)b * qs.bs + (long)h * qs.hs;
  const long k_base = (long)b * ks.bs + (long)hkv * ks.hs;
  const long v_base = (long)b * vs.bs + (long)hkv * vs.hs;
  const long o_base = (long)b * os.bs + (long)h * os.hs;
  const long rq = qs.ss;
  const long rk = ks.ss;
  const long rv = vs.ss;

  const int qrow0 = qb * BM + wave * 16;
  const int skq = Sk - Sq;

  frag_b16 qf[KFRAGS];
  {
    int qrow = qrow0 + row_in_tile;
    int qr = qrow < Sq ? qrow : Sq - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      qf[kk] = global_read16(q + q_base + (long)qr * rq + kk * 32 +
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

  int kb_end = (Sk + kBlockN - 1) / kBlockN;
  if (causal) {
    int max_qrow = qb * BM + BM - 1;
    int max_key = max_qrow + skq;
    kb_end = min(kb_end, max_key / kBlockN + 1);
  }
  int kb_start = 0;
  if (window > 0) {
    int min_qrow = qb * BM;
    int min_key = min_qrow + skq - window + 1;
    if (min_key > 0) kb_start = min_key / kBlockN;
  }

  stage_tr_image<D, kBlockN, NT>(k_img[kb_start & 1], k + k_base, rk,
                                 kb_start * kBlockN, Sk);
  stage_tr_image<D, kBlockN, NT>(v_img[kb_start & 1], v + v_base, rv,
                                 kb_start * kBlockN, Sk);
  __syncthreads();

  for (int kb = kb_start; kb < kb_end; ++kb) {
    const int kstart = kb * kBlockN;
    const __hip_bfloat16* k_cur = k_img[kb & 1];
    const __hip_bfloat16* v_cur = v_img[kb & 1];
    // issue next tile's staging loads; they land behind this iteration's
    // compute and are published by the single end-of-iteration barrier
    if (kb + 1 < kb_end) {
      stage_tr_image<D, kBlockN, NT>(k_img[(kb + 1) & 1], k + k_base, rk,
                                     (kb + 1) * kBlockN, Sk);
      stage_tr_image<D, kBlockN, NT>(v_img[(kb + 1) & 1], v + v_base, rv,
                                     (kb + 1) * kBlockN, Sk);
    }

    // per-wave skip: with BM = NW*16 q rows per workgroup, kb_end covers the
    // LAST wave's diagonal — earlier waves' tail tiles are fully masked
    bool wave_live = true;
    if (causal && kstart > qrow0 + 15 + skq) wave_live = false;
    if (window > 0 && kstart + kBlockN - 1 < qrow0 + skq - window + 1) {
      wave_live = false;
    }
    if (!wave_live) {
      __syncthreads();
      continue;
    }

    // S = Q K^T : B-fragments = K d-runs from LDS
    frag_f32 st[4];
#pragma unroll
    for (int t = 0; t < 4; ++t) st[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        frag_b16 bf = img_dfrag<D>(k_cur, t * 16 + row_in_tile,
                                   kk * 32 + kgroup * 8);
        st[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[kk], bf, st[t], 0,
                                                        0, 0);
      }
    }

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

    // p overwrites s in place (overlapping lifetimes — keeps the register
    // count under the 4-waves/SIMD budget for the 16-wave variant)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float acc = 0.f;
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        float pv = (s_val[t][r] < -1e29f) ? 0.f
                                          : __expf(s_val[t][r] - m_new[r]);
        s_val[t][r] = pv;
        acc += pv;
      }
      float rowsum = group16_sum(acc);
      l_run[r] = l_run[r] * alpha[r] + rowsum;
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        o_acc[t][r] *= alpha[r];
      }
    }

#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = kgroup * 4 + r;
        int col = t * 16 + row_in_tile;
        float pw = s_val[t][r];
        if (DROP) {
          // dropout applies to P feeding O (the lse/l_run softmax state
          // stays dropout-independent)
          bool keep = attn_drop_keep(
              drop_seed, drop_offset,
              ((long long)b * Hq + h) * Sq + (qrow0 + row),
              ((long long)Sk + 3) & ~3LL, kstart + col, drop_p);
          pw = keep ? pw * (1.0f / (1.0f - drop_p)) : 0.f;
        }
        p_lds[wave][row * kStrip + col] = __float2bfloat16(pw);
      }
    }
    // strips are wave-private: lgkmcnt ordering suffices, no barrier

    // O += P V : A = P strip, B = V row-runs (transpose reads)
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      frag_b16 pf = lds_read16(
          &p_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8]);
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        frag_b16 vf = tr_bfrag<D>(v_cur, kk2 * 32, t, lane);
        o_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, o_acc[t],
                                                           0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: pack the 16 x D wave tile through the (now idle) P strip so
  // the global stores are 16-byte dwordx4 — the scalar bf16 tail is
  // store-issue-bound (32 instructions/lane)
  float inv_l[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = qrow0 + kgroup * 4 + r;
    inv_l[r] = l_run[r] > 0.f ? 1.0f / l_run[r] : 0.f;
    if (row < Sq && row_in_tile == 0) {
      lse[((long)b * Hq + h) * Sq + row] =
          m_run[r] + logf(fmaxf(l_run[r], 1e-30f));
    }
  }
#pragma unroll
  for (int half = 0; half < D / 64; ++half) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p_lds[wave][(kgroup * 4 + r) * kStrip + t * 16 + row_in_tile] =
            __float2bfloat16(o_acc[half * 4 + t][r] * inv_l[r]);
      }
    }
    // wave-private strip: in-order LDS, no barrier
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int e = j * 64 + lane;
      int srow = e / 8;
      int scol = (e % 8) * 8;
      int row = qrow0 + srow;
      if (row < Sq) {
        *reinterpret_cast<uint4*>(out + o_base + (long)row * os.ss +
                                  half * 64 + scol) =
            *reinterpret_cast<const uint4*>(
                &p_lds[wave][srow * kStrip + scol]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// forward mb2: 8 waves x TWO 16-row bands per wave (BM = 256). Every K
// d-run fragment and V tr-read pair feeds TWO MFMAs (one per band), halving
// LDS fragment traffic per FLOP — the PMC-identified bottleneck (LDS array
// ~48% busy vs MFMA pipe 14.5% at one band). 2 waves/SIMD (VGPR ~220).

template <int D>
__global__ __launch_bounds__(512, 2) void fa_fwd_mb2_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse, int B, int Sq, int Sk, int Hq, int Hkv,
    float scale, int causal, int window, TStr qs, TStr ks, TStr vs, TStr os) {
  constexpr int NW = 8;
  constexpr int MB = 2;
  constexpr int KFRAGS = D / 32;
  constexpr int DTILES = D / 16;
  constexpr int IMG = tr_elems<D>(kBlockN);
  constexpr int BM = NW * 16 * MB;  // 256 q rows per workgroup
  constexpr int NT = NW * 64;

  __shared__ __hip_bfloat16 p_lds[NW][MB * 16 * kStrip];
  __shared__ __hip_bfloat16 k_img[2][IMG];
  __shared__ __hip_bfloat16 v_img[2][IMG];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;
  const int kgroup = lane >> 4;

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long q_base = (long)b * qs.bs + (long)h * qs.hs;
  const long k_base = (long)b * ks.bs + (long)hkv * ks.hs;
  const long v_base = (long)b * vs.bs + (long)hkv * vs.hs;
  const long o_base = (long)b * os.bs + (long)h * os.hs;
  const long rq = qs.ss;
  const long rk = ks.ss;
  const long rv = vs.ss;

  // band mb covers rows [qrow0(mb), qrow0(mb)+16)
  const int wrow0 = qb * BM + wave * (16 * MB);
  const int skq = Sk - Sq;

  frag_b16 qf[MB][KFRAGS];
#pragma unroll
  for (int mb = 0; mb < MB; ++mb) {
    int qrow = wrow0 + mb * 16 + row_in_tile;
    int qr = qrow < Sq ? qrow : Sq - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      qf[mb][kk] = global_read16(q + q_base + (long)qr * rq + kk * 32 +
                                 kgroup * 8);
    }
  }

  frag_f32 o_acc[MB][DTILES];
  float m_run[MB][4], l_run[MB][4];
#pragma unroll
  for (int mb = 0; mb < MB; ++mb) {
#pragma unroll
    for (int t = 0; t < DTILES; ++t) {
      o_acc[mb][t] = frag_f32{0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[mb][r] = -1e30f;
      l_run[mb][r] = 0.f;
    }
  }

  int kb_end = (Sk + kBlockN - 1) / kBlockN;
  if (causal) {
    int max_key = qb * BM + BM - 1 + skq;
    kb_end = min(kb_end, max_key / kBlockN + 1);
  }
  int kb_start = 0;
  if (window > 0) {
    int min_key = qb * BM + skq - window + 1;
    if (min_key > 0) kb_start = min_key / kBlockN;
  }

  stage_tr_image<D, kBlockN, NT>(k_img[kb_start & 1], k + k_base, rk,
                                 kb_start * kBlockN, Sk);
  stage_tr_image<D, kBlockN, NT>(v_img[kb_start & 1], v + v_base, rv,
                                 kb_start * kBlockN, Sk);
  __syncthreads();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= NT / 2) {
    __builtin_amdgcn_s_setprio(1);
  }

  for (int kb = kb_start; kb < kb_end; ++kb) {
    const int kstart = kb * kBlockN;
    const __hip_bfloat16* k_cur = k_img[kb & 1];
    const __hip_bfloat16* v_cur = v_img[kb & 1];
    if (kb + 1 < kb_end) {
      stage_tr_image<D, kBlockN, NT>(k_img[(kb + 1) & 1], k + k_base, rk,
                                     (kb + 1) * kBlockN, Sk);
      stage_tr_image<D, kBlockN, NT>(v_img[(kb + 1) & 1], v + v_base, rv,
                                     (kb + 1) * kBlockN, Sk);
    }

    bool wave_live = true;
    if (causal && kstart > wrow0 + 16 * MB - 1 + skq) wave_live = false;
    if (window > 0 &&
        kstart + kBlockN - 1 < wrow0 + skq - window + 1) {
      wave_live = false;
    }
    if (!wave_live) {
      __syncthreads();
      continue;
    }

    // S = Q K^T, both bands per B-fragment read
    frag_f32 st[MB][4];
#pragma unroll
    for (int mb = 0; mb < MB; ++mb) {
#pragma unroll
      for (int t = 0; t < 4; ++t) st[mb][t] = frag_f32{0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        frag_b16 bf = img_dfrag<D>(k_cur, t * 16 + row_in_tile,
                                   kk * 32 + kgroup * 8);
#pragma unroll
        for (int mb = 0; mb < MB; ++mb) {
          st[mb][t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[mb][kk], bf, st[mb][t], 0, 0, 0);
        }
      }
    }

#pragma unroll
    for (int mb = 0; mb < MB; ++mb) {
      const int qrow0 = wrow0 + mb * 16;
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
          s_val[t][r] = masked ? -1e30f : st[mb][t][r] * scale;
        }
      }
      float m_new[4], alpha[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pm = fmaxf(fmaxf(s_val[0][r], s_val[1][r]),
                         fmaxf(s_val[2][r], s_val[3][r]));
        pm = group16_max(pm);
        m_new[r] = fmaxf(m_run[mb][r], pm);
        alpha[r] = __expf(m_run[mb][r] - m_new[r]);
        m_run[mb][r] = m_new[r];
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float acc = 0.f;
#pragma unroll
        for (int t = 0; t < 4; ++t) {
          float pv = (s_val[t][r] < -1e29f)
                         ? 0.f
                         : __expf(s_val[t][r] - m_new[r]);
          s_val[t][r] = pv;
          acc += pv;
        }
        float rowsum = group16_sum(acc);
        l_run[mb][r] = l_run[mb][r] * alpha[r] + rowsum;
#pragma unroll
        for (int t = 0; t < DTILES; ++t) {
          o_acc[mb][t][r] *= alpha[r];
        }
      }
#pragma unroll
      for (int t = 0; t < 4; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int srow = mb * 16 + kgroup * 4 + r;
          p_lds[wave][srow * kStrip + t * 16 + row_in_tile] =
              __float2bfloat16(s_val[t][r]);
        }
      }
    }
    // strips wave-private: no barrier

    // O += P V, both bands per V tr-read pair
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      frag_b16 pf[MB];
#pragma unroll
      for (int mb = 0; mb < MB; ++mb) {
        pf[mb] = lds_read16(&p_lds[wave][(mb * 16 + row_in_tile) * kStrip +
                                         kk2 * 32 + kgroup * 8]);
      }
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        frag_b16 vf = tr_bfrag<D>(v_cur, kk2 * 32, t, lane);
#pragma unroll
        for (int mb = 0; mb < MB; ++mb) {
          o_acc[mb][t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pf[mb], vf, o_acc[mb][t], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int mb = 0; mb < MB; ++mb) {
    const int qrow0 = wrow0 + mb * 16;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = qrow0 + kgroup * 4 + r;
      float inv_l = l_run[mb][r] > 0.f ? 1.0f / l_run[mb][r] : 0.f;
      if (row < Sq) {
#pragma unroll
        for (int t = 0; t < DTILES; ++t) {
          out[o_base + (long)row * os.ss + t * 16 + row_in_tile] =
              __float2bfloat16(o_acc[mb][t][r] * inv_l);
        }
        if (row_in_tile == 0) {
          lse[((long)b * Hq + h) * Sq + row] =
              m_run[mb][r] + logf(fmaxf(l_run[mb][r], 1e-30f));
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// forward v8: 8-wave workgroup (2 waves/SIMD — the tuned CDNA4 attention
// regime) with the T14 async-stage split: K/V tile loads are ISSUED one full
// compute phase before their vmcnt wait + LDS write, so the HBM latency
// hides under the previous tile's MFMAs instead of stalling at the top of
// the iteration (the v4 stage_tr_image load+wait+write ran inline: PMC
// showed 44% SQ_WAIT_ANY). Additions: static s_setprio(1) for the
// younger dispatch half, per-wave skip of fully-masked causal tiles, and an
// LDS-packed dwordx4 epilogue (v4 stored 32 scalar bf16 per lane —
// issue-bound store tail).

template <int D>
__global__ __launch_bounds__(512, 2) void fa_fwd_v8_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ out,
    float* __restrict__ lse, int B, int Sq, int Sk, int Hq, int Hkv,
    float scale, int causal, int window, TStr qs, TStr ks, TStr vs, TStr os) {
  constexpr int NW = 8;
  constexpr int KFRAGS = D / 32;
  constexpr int DTILES = D / 16;
  constexpr int IMG = tr_elems<D>(kBlockN);
  constexpr int BM = NW * 16;  // 128 q rows per workgroup
  constexpr int NT = NW * 64;  // 512 threads
  // staging: one K image + one V image = 2 * 64 rows * (D/8) uint4 chunks
  constexpr int CHUNKS = kBlockN * (D / 8);   // per image
  constexpr int PER_THREAD = CHUNKS / NT;     // 2 at D=128, 1 at D=64
  static_assert(CHUNKS % NT == 0);

  __shared__ __hip_bfloat16 p_lds[NW][16 * kStrip];
  __shared__ __hip_bfloat16 k_img[2][IMG];
  __shared__ __hip_bfloat16 v_img[2][IMG];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;
  const int kgroup = lane >> 4;

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long q_base = (long)b * qs.bs + (long)h * qs.hs;
  const long k_base = (long)b * ks.bs + (long)hkv * ks.hs;
  const long v_base = (long)b * vs.bs + (long)hkv * vs.hs;
  const long o_base = (long)b * os.bs + (long)h * os.hs;
  const long rq = qs.ss;
  const long rk = ks.ss;
  const long rv = vs.ss;

  const int qrow0 = qb * BM + wave * 16;
  const int skq = Sk - Sq;

  frag_b16 qf[KFRAGS];
  {
    int qrow = qrow0 + row_in_tile;
    int qr = qrow < Sq ? qrow : Sq - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      qf[kk] = global_read16(q + q_base + (long)qr * rq + kk * 32 +
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

  int kb_end = (Sk + kBlockN - 1) / kBlockN;
  if (causal) {
    int max_qrow = qb * BM + BM - 1;
    int max_key = max_qrow + skq;
    kb_end = min(kb_end, max_key / kBlockN + 1);
  }
  int kb_start = 0;
  if (window > 0) {
    int min_qrow = qb * BM;
    int min_key = min_qrow + skq - window + 1;
    if (min_key > 0) kb_start = min_key / kBlockN;
  }

  // per-thread staging registers: tile loads issued one iteration early
  uint4 kreg[PER_THREAD], vreg[PER_THREAD];
  // chunk idx -> (row, col8) of the 64 x D tile; write to tr_off(row, col8)
  auto stage_load = [&](int kb) {
#pragma unroll
    for (int j = 0; j < PER_THREAD; ++j) {
      int idx = threadIdx.x + j * NT;
      int r = idx / (D / 8);
      int c8 = (idx % (D / 8)) * 8;
      int row = kb * kBlockN + r;
      int rr = row < Sk ? row : Sk - 1;
      uint4 kv = *reinterpret_cast<const uint4*>(k + k_base + (long)rr * rk +
                                                 c8);
      uint4 vv = *reinterpret_cast<const uint4*>(v + v_base + (long)rr * rv +
                                                 c8);
      if (row >= Sk) {
        kv = make_uint4(0, 0, 0, 0);
        vv = make_uint4(0, 0, 0, 0);
      }
      kreg[j] = kv;
      vreg[j] = vv;
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < PER_THREAD; ++j) {
      int idx = threadIdx.x + j * NT;
      int r = idx / (D / 8);
      int c8 = (idx % (D / 8)) * 8;
      *reinterpret_cast<uint4*>(&k_img[buf][tr_off<D>(r, c8)]) = kreg[j];
      *reinterpret_cast<uint4*>(&v_img[buf][tr_off<D>(r, c8)]) = vreg[j];
    }
  };

  stage_load(kb_start);
  stage_write(kb_start & 1);
  if (kb_start + 1 < kb_end) stage_load(kb_start + 1);
  __syncthreads();

  // static priority for the younger dispatch half (T5 static form): at 2
  // waves/SIMD the second half loses VALU arbitration on every segment
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= NT / 2) {
    __builtin_amdgcn_s_setprio(1);
  }

  for (int kb = kb_start; kb < kb_end; ++kb) {
    const int kstart = kb * kBlockN;
    const __hip_bfloat16* k_cur = k_img[kb & 1];
    const __hip_bfloat16* v_cur = v_img[kb & 1];
    // T14 write-late: publish tile kb+1 (loads issued LAST iteration, a
    // full compute phase of latency slack), then issue tile kb+2's loads
    if (kb + 1 < kb_end) stage_write((kb + 1) & 1);
    if (kb + 2 < kb_end) stage_load(kb + 2);

    // per-wave skip: every row of this wave's 16-row band masked out
    bool wave_live = true;
    if (causal && kstart > qrow0 + 15 + skq) wave_live = false;
    if (window > 0 && kstart + kBlockN - 1 < qrow0 + skq - window + 1) {
      wave_live = false;
    }
    if (wave_live) {
      // S = Q K^T : B-fragments = K d-runs from LDS
      frag_f32 st[4];
#pragma unroll
      for (int t = 0; t < 4; ++t) st[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < KFRAGS; ++kk) {
#pragma unroll
        for (int t = 0; t < 4; ++t) {
          frag_b16 bf = img_dfrag<D>(k_cur, t * 16 + row_in_tile,
                                     kk * 32 + kgroup * 8);
          st[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[kk], bf, st[t],
                                                          0, 0, 0);
        }
      }

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
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float acc = 0.f;
#pragma unroll
        for (int t = 0; t < 4; ++t) {
          float pv = (s_val[t][r] < -1e29f) ? 0.f
                                            : __expf(s_val[t][r] - m_new[r]);
          p_val[t][r] = pv;
          acc += pv;
        }
        float rowsum = group16_sum(acc);
        l_run[r] = l_run[r] * alpha[r] + rowsum;
#pragma unroll
        for (int t = 0; t < DTILES; ++t) {
          o_acc[t][r] *= alpha[r];
        }
      }

#pragma unroll
      for (int t = 0; t < 4; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = kgroup * 4 + r;
          int col = t * 16 + row_in_tile;
          p_lds[wave][row * kStrip + col] = __float2bfloat16(p_val[t][r]);
        }
      }
      // strips are wave-private: lgkmcnt ordering suffices, no barrier

      // O += P V : A = P strip, B = V row-runs (transpose reads)
#pragma unroll
      for (int kk2 = 0; kk2 < 2; ++kk2) {
        frag_b16 pf = lds_read16(
            &p_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8]);
#pragma unroll
        for (int t = 0; t < DTILES; ++t) {
          frag_b16 vf = tr_bfrag<D>(v_cur, kk2 * 32, t, lane);
          o_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, o_acc[t],
                                                             0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // epilogue: round-trip the 16 x D wave tile through the P strip so the
  // global stores are 16-byte dwordx4 instead of 32 scalar bf16 per lane
  float inv_l[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    inv_l[r] = l_run[r] > 0.f ? 1.0f / l_run[r] : 0.f;
    int row = qrow0 + kgroup * 4 + r;
    if (row < Sq && row_in_tile == 0) {
      lse[((long)b * Hq + h) * Sq + row] =
          m_run[r] + logf(fmaxf(l_run[r], 1e-30f));
    }
  }
#pragma unroll
  for (int half = 0; half < D / 64; ++half) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int srow = kgroup * 4 + r;
        int scol = t * 16 + row_in_tile;
        p_lds[wave][srow * kStrip + scol] =
            __float2bfloat16(o_acc[half * 4 + t][r] * inv_l[r]);
      }
    }
    // wave-private strip: no barrier
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int e = j * 64 + lane;       // 8-elem chunk index in the 16x64 strip
      int srow = e / 8;
      int scol = (e % 8) * 8;
      int row = qrow0 + srow;
      if (row < Sq) {
        uint4 val = *reinterpret_cast<const uint4*>(
            &p_lds[wave][srow * kStrip + scol]);
        *reinterpret_cast<uint4*>(
            out + o_base + (long)row * os.ss + half * 64 + scol) = val;
      }
    }
    // all lanes re-write the strip next half: wave-private, in-order LDS
  }
}

// ---------------------------------------------------------------------------
// backward preprocess: delta[b,h,s] = sum_d dO * O (fp32)

template <int D>
__global__ void fa_bwd_delta_kernel(const __hip_bfloat16* __restrict__ dout,
                                    const __hip_bfloat16* __restrict__ out,
                                    float* __restrict__ delta, int B, int Sq,
                                    int Hq, TStr ds, TStr os) {
  const long row = blockIdx.x;
  const int s = row % Sq;
  const long bh = row / Sq;
  const int b = bh / Hq;
  const int h = bh % Hq;
  const long do_base = (long)b * ds.bs + (long)s * ds.ss + (long)h * ds.hs;
  const long o_base = (long)b * os.bs + (long)s * os.ss + (long)h * os.hs;
  float acc = 0.f;
  for (int i = threadIdx.x; i < D; i += 64) {
    acc += __bfloat162float(dout[do_base + i]) *
           __bfloat162float(out[o_base + i]);
  }
  acc = wave_reduce_sum(acc);
  if (threadIdx.x == 0) delta[row] = acc;
}

// ---------------------------------------------------------------------------
// backward dK/dV (k-parallel; flattened (gqa head, q-block) loop with
// double-buffered dO/Q images)

template <int D, int NW, bool DROP = false>
__global__ __launch_bounds__(NW * 64) void fa_bwd_dkdv_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const __hip_bfloat16* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, __hip_bfloat16* __restrict__ dk,
    __hip_bfloat16* __restrict__ dv, int B, int Sq, int Sk, int Hq, int Hkv,
    float scale, int causal, int window, TStr qs, TStr ks, TStr vs, TStr ds,
    TStr dks, TStr dvs, float drop_p = 0.f, unsigned long long drop_seed = 0,
    unsigned long long drop_offset = 0) {
  constexpr int KFRAGS = D / 32;
  constexpr int DTILES = D / 16;
  constexpr int IMG = tr_elems<D>(kBlockM);
  constexpr int BN = NW * 16;  // keys per workgroup
  constexpr int NT = NW * 64;

  __shared__ __hip_bfloat16 pt_lds[NW][16 * kStrip];
  __shared__ __hip_bfloat16 dst_lds[NW][16 * kStrip];
  __shared__ __hip_bfloat16 do_img[2][IMG];
  __shared__ __hip_bfloat16 q_img[2][IMG];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;
  const int kgroup = lane >> 4;

  const int kb = blockIdx.x;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int gqa = Hq / Hkv;

  const long k_base = (long)b * ks.bs + (long)hkv * ks.hs;
  const long v_base = (long)b * vs.bs + (long)hkv * vs.hs;
  const long rk = ks.ss;
  const long rv = vs.ss;
  const int skq = Sk - Sq;

  const int key0 = kb * BN + wave * 16;

  frag_b16 ka[KFRAGS], va[KFRAGS];
  {
    int key = key0 + row_in_tile;
    int kr = key < Sk ? key : Sk - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      ka[kk] = global_read16(k + k_base + (long)kr * rk + kk * 32 +
                             kgroup * 8);
      va[kk] = global_read16(v + v_base + (long)kr * rv + kk * 32 +
                             kgroup * 8);
    }
  }

  frag_f32 dv_acc[DTILES], dk_acc[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) {
    dv_acc[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
    dk_acc[t] = frag_f32{0.f, 0.f, 0.f, 0.f};
  }

  int qb_start = 0;
  if (causal) {
    int min_key = kb * BN;
    int min_qrow = min_key - skq;
    if (min_qrow > 0) qb_start = min_qrow / kBlockM;
  }
  int qb_end = (Sq + kBlockM - 1) / kBlockM;
  if (window > 0) {
    int max_key = kb * BN + BN - 1;
    int max_qrow = max_key + window - 1 - skq;
    qb_end = min(qb_end, max_qrow / kBlockM + 1);
  }

  const int nqb = qb_end - qb_start;
  const int iters = gqa * (nqb > 0 ? nqb : 0);

  auto stage_iter = [&](int it, int buf) {
    int hq = hkv * gqa + it / nqb;
    int qbx = qb_start + it % nqb;
    stage_tr_image<D, kBlockM, NT>(
        do_img[buf], dout + (long)b * ds.bs + (long)hq * ds.hs, ds.ss,
        qbx * kBlockM, Sq);
    stage_tr_image<D, kBlockM, NT>(
        q_img[buf], q + (long)b * qs.bs + (long)hq * qs.hs, qs.ss,
        qbx * kBlockM, Sq);
  };

  if (iters > 0) {
    stage_iter(0, 0);
    __syncthreads();
  }

  for (int it = 0; it < iters; ++it) {
    const int hq = hkv * gqa + it / nqb;
    const int qbx = qb_start + it % nqb;
    const int qstart = qbx * kBlockM;
    const __hip_bfloat16* do_cur = do_img[it & 1];
    const __hip_bfloat16* q_cur = q_img[it & 1];
    const float* lse_h = lse + ((long)b * Hq + hq) * Sq;
    const float* delta_h = delta + ((long)b * Hq + hq) * Sq;

    if (it + 1 < iters) {
      stage_iter(it + 1, (it + 1) & 1);
    }

    // per-wave skip: this wave's 16 keys all causally after (or all
    // SWA-expired before) every q row of this tile
    bool wave_live = true;
    if (causal && key0 > qstart + kBlockM - 1 + skq) wave_live = false;
    if (window > 0 && key0 + 15 < qstart + skq - window + 1) wave_live = false;
    if (!wave_live) {
      __syncthreads();
      continue;
    }

    // S^T = K Q^T ; dP^T = V dO^T : B-fragments are Q/dO d-runs from LDS.
    // One tile's fragment pair live at a time (VGPR pressure), softmax +
    // strip write immediately after each tile's K-reduction.
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      frag_f32 stt = frag_f32{0.f, 0.f, 0.f, 0.f};
      frag_f32 dpt = frag_f32{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < KFRAGS; ++kk) {
        frag_b16 qbf = img_dfrag<D>(q_cur, t * 16 + row_in_tile,
                                    kk * 32 + kgroup * 8);
        stt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ka[kk], qbf, stt, 0, 0,
                                                      0);
        frag_b16 dbf = img_dfrag<D>(do_cur, t * 16 + row_in_tile,
                                    kk * 32 + kgroup * 8);
        dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(va[kk], dbf, dpt, 0, 0,
                                                      0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int key = kb * BN + wave * 16 + kgroup * 4 + r;
        int qrow = qstart + t * 16 + row_in_tile;
        bool masked = (key >= Sk) || (qrow >= Sq);
        if (causal && key > qrow + skq) masked = true;
        if (window > 0 && key < qrow + skq - window + 1) masked = true;
        float pt = 0.f, dst = 0.f;
        if (!masked) {
          float l = lse_h[qrow];
          pt = __expf(stt[r] * scale - l);
          float dp_eff = dpt[r];
          if (DROP) {
            bool keep = attn_drop_keep(
                drop_seed, drop_offset,
                ((long long)b * Hq + hq) * Sq + qrow,
                ((long long)Sk + 3) & ~3LL, key, drop_p);
            float rs = 1.0f / (1.0f - drop_p);
            dp_eff = keep ? dp_eff * rs : 0.f;
            dst = pt * (dp_eff - delta_h[qrow]) * scale;
            pt = keep ? pt * rs : 0.f;  // dV uses the dropped P
          } else {
            dst = pt * (dp_eff - delta_h[qrow]) * scale;
          }
        }
        int lrow = kgroup * 4 + r;
        int lcol = t * 16 + row_in_tile;
        pt_lds[wave][lrow * kStrip + lcol] = __float2bfloat16(pt);
        dst_lds[wave][lrow * kStrip + lcol] = __float2bfloat16(dst);
      }
    }
    // strips are wave-private: no barrier

    // dV += P^T dO ; dK += dS^T Q  (B = dO/Q row-runs: transpose reads)
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      frag_b16 ptf = lds_read16(
          &pt_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8]);
      frag_b16 dstf = lds_read16(
          &dst_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8]);
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        frag_b16 dof = tr_bfrag<D>(do_cur, kk2 * 32, t, lane);
        dv_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ptf, dof,
                                                            dv_acc[t], 0, 0,
                                                            0);
        frag_b16 qtf = tr_bfrag<D>(q_cur, kk2 * 32, t, lane);
        dk_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dstf, qtf,
                                                            dk_acc[t], 0, 0,
                                                            0);
      }
    }
    __syncthreads();
  }

  // epilogue: pack dk through pt_lds and dv through dst_lds (both idle by
  // now) for dwordx4 stores instead of 2x32 scalar bf16 per lane
  {
    const long dk_base = (long)b * dks.bs + (long)hkv * dks.hs;
    const long dv_base = (long)b * dvs.bs + (long)hkv * dvs.hs;
    const int key0w = kb * BN + wave * 16;
#pragma unroll
    for (int half = 0; half < D / 64; ++half) {
#pragma unroll
      for (int t = 0; t < 4; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int srow = kgroup * 4 + r;
          int scol = t * 16 + row_in_tile;
          pt_lds[wave][srow * kStrip + scol] =
              __float2bfloat16(dk_acc[half * 4 + t][r]);
          dst_lds[wave][srow * kStrip + scol] =
              __float2bfloat16(dv_acc[half * 4 + t][r]);
        }
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int e = j * 64 + lane;
        int srow = e / 8;
        int scol = (e % 8) * 8;
        int key = key0w + srow;
        if (key < Sk) {
          *reinterpret_cast<uint4*>(dk + dk_base + (long)key * dks.ss +
                                    half * 64 + scol) =
              *reinterpret_cast<const uint4*>(
                  &pt_lds[wave][srow * kStrip + scol]);
          *reinterpret_cast<uint4*>(dv + dv_base + (long)key * dvs.ss +
                                    half * 64 + scol) =
              *reinterpret_cast<const uint4*>(
                  &dst_lds[wave][srow * kStrip + scol]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward dQ (q-parallel)

template <int D, int NW, bool DROP = false>
__global__ __launch_bounds__(NW * 64) void fa_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v,
    const __hip_bfloat16* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ delta, __hip_bfloat16* __restrict__ dq, int B,
    int Sq, int Sk, int Hq, int Hkv, float scale, int causal, int window,
    TStr qs, TStr ks, TStr vs, TStr ds, TStr dqs, float drop_p = 0.f,
    unsigned long long drop_seed = 0, unsigned long long drop_offset = 0) {
  constexpr int KFRAGS = D / 32;
  constexpr int DTILES = D / 16;
  constexpr int IMG = tr_elems<D>(kBlockN);
  constexpr int BM = NW * 16;
  constexpr int NT = NW * 64;

  __shared__ __hip_bfloat16 ds_lds[NW][16 * kStrip];
  __shared__ __hip_bfloat16 k_img[2][IMG];
  __shared__ __hip_bfloat16 v_img[2][IMG];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row_in_tile = lane & 15;
  const int kgroup = lane >> 4;

  const int qb = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long q_base = (long)b * qs.bs + (long)h * qs.hs;
  const long do_base = (long)b * ds.bs + (long)h * ds.hs;
  const long k_base = (long)b * ks.bs + (long)hkv * ks.hs;
  const long v_base = (long)b * vs.bs + (long)hkv * vs.hs;
  const long rq = qs.ss;
  const long rk = ks.ss;
  const long rv = vs.ss;
  const int skq = Sk - Sq;
  const int qrow0 = qb * BM + wave * 16;

  const float* lse_h = lse + ((long)b * Hq + h) * Sq;
  const float* delta_h = delta + ((long)b * Hq + h) * Sq;

  frag_b16 qf[KFRAGS], dof[KFRAGS];
  {
    int qrow = qrow0 + row_in_tile;
    int qr = qrow < Sq ? qrow : Sq - 1;
#pragma unroll
    for (int kk = 0; kk < KFRAGS; ++kk) {
      qf[kk] = global_read16(q + q_base + (long)qr * rq + kk * 32 +
                             kgroup * 8);
      dof[kk] = global_read16(dout + do_base + (long)qr * ds.ss + kk * 32 +
                              kgroup * 8);
    }
  }

  frag_f32 dq_acc[DTILES];
#pragma unroll
  for (int t = 0; t < DTILES; ++t) dq_acc[t] = frag_f32{0.f, 0.f, 0.f, 0.f};

  int kb_end = (Sk + kBlockN - 1) / kBlockN;
  if (causal) {
    int max_qrow = qb * BM + BM - 1;
    kb_end = min(kb_end, (max_qrow + skq) / kBlockN + 1);
  }
  int kb_start = 0;
  if (window > 0) {
    int min_key = qb * BM + skq - window + 1;
    if (min_key > 0) kb_start = min_key / kBlockN;
  }

  float lse_r[4], delta_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = qrow0 + kgroup * 4 + r;
    lse_r[r] = row < Sq ? lse_h[row] : 0.f;
    delta_r[r] = row < Sq ? delta_h[row] : 0.f;
  }

  stage_tr_image<D, kBlockN, NT>(k_img[kb_start & 1], k + k_base, rk,
                                 kb_start * kBlockN, Sk);
  stage_tr_image<D, kBlockN, NT>(v_img[kb_start & 1], v + v_base, rv,
                                 kb_start * kBlockN, Sk);
  __syncthreads();

  for (int kb = kb_start; kb < kb_end; ++kb) {
    const int kstart = kb * kBlockN;
    const __hip_bfloat16* k_cur = k_img[kb & 1];
    const __hip_bfloat16* v_cur = v_img[kb & 1];
    if (kb + 1 < kb_end) {
      stage_tr_image<D, kBlockN, NT>(k_img[(kb + 1) & 1], k + k_base, rk,
                                     (kb + 1) * kBlockN, Sk);
      stage_tr_image<D, kBlockN, NT>(v_img[(kb + 1) & 1], v + v_base, rv,
                                     (kb + 1) * kBlockN, Sk);
    }

    // per-wave skip (mirror of the forward: kb_end covers the last wave's
    // diagonal, earlier waves' tail tiles are fully masked)
    bool wave_live = true;
    if (causal && kstart > qrow0 + 15 + skq) wave_live = false;
    if (window > 0 && kstart + kBlockN - 1 < qrow0 + skq - window + 1) {
      wave_live = false;
    }
    if (!wave_live) {
      __syncthreads();
      continue;
    }

    // S = Q K^T ; dP = dO V^T : B-fragments are K/V d-runs from LDS.
    // Tile-at-a-time (one st/dp fragment pair live) for VGPR pressure —
    // dS needs only the stored lse/delta, no cross-tile softmax state.
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      frag_f32 st = frag_f32{0.f, 0.f, 0.f, 0.f};
      frag_f32 dp = frag_f32{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < KFRAGS; ++kk) {
        frag_b16 kbf = img_dfrag<D>(k_cur, t * 16 + row_in_tile,
                                    kk * 32 + kgroup * 8);
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[kk], kbf, st, 0, 0,
                                                     0);
        frag_b16 vbf = img_dfrag<D>(v_cur, t * 16 + row_in_tile,
                                    kk * 32 + kgroup * 8);
        dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof[kk], vbf, dp, 0, 0,
                                                     0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int col = kstart + t * 16 + row_in_tile;
        int row = qrow0 + kgroup * 4 + r;
        bool masked = (col >= Sk) || (row >= Sq);
        if (causal && col > row + skq) masked = true;
        if (window > 0 && col < row + skq - window + 1) masked = true;
        float ds = 0.f;
        if (!masked) {
          float p = __expf(st[r] * scale - lse_r[r]);
          float dp_eff = dp[r];
          if (DROP) {
            bool keep = attn_drop_keep(
                drop_seed, drop_offset,
                ((long long)b * Hq + h) * Sq + row,
                ((long long)Sk + 3) & ~3LL, col, drop_p);
            dp_eff = keep ? dp_eff * (1.0f / (1.0f - drop_p)) : 0.f;
          }
          ds = p * (dp_eff - delta_r[r]) * scale;
        }
        int lrow = kgroup * 4 + r;
        int lcol = t * 16 + row_in_tile;
        ds_lds[wave][lrow * kStrip + lcol] = __float2bfloat16(ds);
      }
    }
    // wave-private strip: no barrier needed

    // dQ += dS K  (B = K row-runs: transpose reads)
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      frag_b16 dsf = lds_read16(
          &ds_lds[wave][row_in_tile * kStrip + kk2 * 32 + kgroup * 8]);
#pragma unroll
      for (int t = 0; t < DTILES; ++t) {
        frag_b16 ktf = tr_bfrag<D>(k_cur, kk2 * 32, t, lane);
        dq_acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf, ktf,
                                                            dq_acc[t], 0, 0,
                                                            0);
      }
    }
    __syncthreads();
  }

  // epilogue: pack dq through the idle dS strip for dwordx4 stores
  {
    const long dq_base = (long)b * dqs.bs + (long)h * dqs.hs;
#pragma unroll
    for (int half = 0; half < D / 64; ++half) {
#pragma unroll
      for (int t = 0; t < 4; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          ds_lds[wave][(kgroup * 4 + r) * kStrip + t * 16 + row_in_tile] =
              __float2bfloat16(dq_acc[half * 4 + t][r]);
        }
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int e = j * 64 + lane;
        int srow = e / 8;
        int scol = (e % 8) * 8;
        int row = qrow0 + srow;
        if (row < Sq) {
          *reinterpret_cast<uint4*>(dq + dq_base + (long)row * dqs.ss +
                                    half * 64 + scol) =
              *reinterpret_cast<const uint4*>(
                  &ds_lds[wave][srow * kStrip + scol]);
        }
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------

// [b,s,n,h] tensors need only a dense head_dim axis; batch/seq/head strides
// are free, so sbhd transposes AND the per-head q/k/v slices of a fused QKV
// projection pass with no copies. 16-byte-aligned rows required (head_dim is
// 64/128 and torch allocations are 256B-aligned, so any element-strided view
// of a bf16 buffer qualifies as long as the strides are multiples of 8).
static void check_bshd(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.dim() == 4, name, ": need 4-D CUDA tensor");
  TORCH_CHECK(t.stride(3) == 1, name, ": head_dim must be dense");
  TORCH_CHECK(t.stride(0) % 8 == 0 && t.stride(1) % 8 == 0 &&
                  t.stride(2) % 8 == 0,
              name, ": strides must keep rows 16-byte aligned");
}

static TStr tstr(const torch::Tensor& t) {
  return TStr{t.stride(0), t.stride(1), t.stride(2)};
}

// allocate like ref, preserving strides only when they tile storage densely
// (keeps sbhd buffers sbhd so downstream [s,b,...] reshapes stay views); a
// sparse view (e.g. a QKV slice) gets a fresh contiguous buffer instead
static torch::Tensor empty_like_strided(const torch::Tensor& ref) {
  long extent = 1;
  for (int i = 0; i < ref.dim(); ++i) {
    extent += (ref.size(i) - 1) * ref.stride(i);
  }
  if (extent != ref.numel()) {
    return torch::empty(ref.sizes(), ref.options());
  }
  return torch::empty_strided(ref.sizes(), ref.strides(), ref.options());
}

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, bool causal,
                                          double softmax_scale,
                                          int64_t window_size,
                                          double dropout_p, int64_t drop_seed,
                                          int64_t drop_offset) {
  check_bshd(q, "q");
  check_bshd(k, "k");
  check_bshd(v, "v");
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16,
              "flash_attn: bf16 only (got ", q.scalar_type(), ")");
  int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  int Sk = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(D == 64 || D == 128, "flash_attn: head dim must be 64/128");
  TORCH_CHECK(Hq % Hkv == 0);

  auto out = empty_like_strided(q);
  const TStr qs = tstr(q), ks = tstr(k), vs = tstr(v), os = tstr(out);
  auto lse = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();
  int win = window_size > 0 ? (int)window_size : 0;
  static const int nw_env = []() {
    const char* e = getenv("MEGATRON_AMD_FA_FWD_WAVES");
    // 12 = the measured-best 12-wave kernel; 8 selects the T14-split
    // experiment (measured SLOWER: 172 vs 277 TF — kept for A/B)
    return e ? atoi(e) : 12;
  }();
  if (dropout_p > 0.0) {
    // attention dropout: 12-wave DROP instantiation only (the experimental
    // variants fall through to it)
    dim3 grid((Sq + 12 * 16 - 1) / (12 * 16), Hq, B);
    const float dp = (float)dropout_p;
    const unsigned long long dseed = (unsigned long long)drop_seed;
    const unsigned long long doff = (unsigned long long)drop_offset;
    if (D == 128) {
      hipLaunchKernelGGL((fa_fwd_kernel<128, 12, true>), grid, dim3(12 * 64),
                         0, stream, (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(),
                         (__hip_bfloat16*)out.data_ptr(),
                         lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                         (float)softmax_scale, causal ? 1 : 0, win, qs, ks,
                         vs, os, dp, dseed, doff);
    } else {
      hipLaunchKernelGGL((fa_fwd_kernel<64, 12, true>), grid, dim3(12 * 64),
                         0, stream, (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(),
                         (__hip_bfloat16*)out.data_ptr(),
                         lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                         (float)softmax_scale, causal ? 1 : 0, win, qs, ks,
                         vs, os, dp, dseed, doff);
    }
    return {out, lse};
  }
  if (nw_env == 82) {
    // 8 waves x 2 bands (BM = 256)
    dim3 grid((Sq + 255) / 256, Hq, B);
    if (D == 128) {
      hipLaunchKernelGGL((fa_fwd_mb2_kernel<128>), grid, dim3(512), 0, stream,
                         (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(),
                         (__hip_bfloat16*)out.data_ptr(),
                         lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                         (float)softmax_scale, causal ? 1 : 0, win, qs, ks,
                         vs, os);
    } else {
      hipLaunchKernelGGL((fa_fwd_mb2_kernel<64>), grid, dim3(512), 0, stream,
                         (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(),
                         (__hip_bfloat16*)out.data_ptr(),
                         lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                         (float)softmax_scale, causal ? 1 : 0, win, qs, ks,
                         vs, os);
    }
    return {out, lse};
  }
  if (nw_env == 8) {
    dim3 grid((Sq + 127) / 128, Hq, B);
    if (D == 128) {
      hipLaunchKernelGGL((fa_fwd_v8_kernel<128>), grid, dim3(512), 0, stream,
                         (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(),
                         (__hip_bfloat16*)out.data_ptr(),
                         lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                         (float)softmax_scale, causal ? 1 : 0, win, qs, ks,
                         vs, os);
    } else {
      hipLaunchKernelGGL((fa_fwd_v8_kernel<64>), grid, dim3(512), 0, stream,
                         (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k.data_ptr(),
                         (const __hip_bfloat16*)v.data_ptr(),
                         (__hip_bfloat16*)out.data_ptr(),
                         lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,
                         (float)softmax_scale, causal ? 1 : 0, win, qs, ks,
                         vs, os);
    }
    return {out, lse};
  }
#define LAUNCH_FWD(DD, NW)                                                      do {                                                                            dim3 grid((Sq + NW * 16 - 1) / (NW * 16), Hq, B);                             hipLaunchKernelGGL((fa_fwd_kernel<DD, NW>), grid, dim3(NW * 64), 0,                              stream, (const __hip_bfloat16*)q.data_ptr(),                                  (const __hip_bfloat16*)k.data_ptr(),                                          (const __hip_bfloat16*)v.data_ptr(),                                          (__hip_bfloat16*)out.data_ptr(),                                              lse.data_ptr<float>(), B, Sq, Sk, Hq, Hkv,                                    (float)softmax_scale, causal ? 1 : 0, win,                                    qs, ks, vs, os);              } while (0)
  if (D == 128) {
    if (nw_env == 16) LAUNCH_FWD(128, 16);
    else if (nw_env >= 12) LAUNCH_FWD(128, 12);
    else LAUNCH_FWD(128, 4);
  } else {
    if (nw_env == 16) LAUNCH_FWD(64, 16);
    else if (nw_env >= 12) LAUNCH_FWD(64, 12);
    else LAUNCH_FWD(64, 4);
  }
#undef LAUNCH_FWD
  return {out, lse};
}

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q,
                                          torch::Tensor k, torch::Tensor v,
                                          torch::Tensor out, torch::Tensor lse,
                                          bool causal, double softmax_scale,
                                          int64_t window_size,
                                          double dropout_p, int64_t drop_seed,
                                          int64_t drop_offset) {
  check_bshd(dout, "dout");
  check_bshd(q, "q");
  check_bshd(k, "k");
  check_bshd(v, "v");
  check_bshd(out, "out");
  int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  int Sk = k.size(1), Hkv = k.size(2);
  int win = window_size > 0 ? (int)window_size : 0;

  auto dq = empty_like_strided(q);
  auto dk = empty_like_strided(k);
  auto dv = empty_like_strided(v);
  auto delta = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();
  const TStr qs = tstr(q), ks = tstr(k), vs = tstr(v), ds = tstr(dout),
             os = tstr(out), dqs = tstr(dq), dks = tstr(dk), dvs = tstr(dv);

  long rows = (long)B * Hq * Sq;
  const float dp = (float)dropout_p;
  const unsigned long long dseed = (unsigned long long)drop_seed;
  const unsigned long long doff = (unsigned long long)drop_offset;
#define LAUNCH_BWD(DD, DROP)                                                  \
  do {                                                                        \
    hipLaunchKernelGGL((fa_bwd_delta_kernel<DD>), dim3(rows), dim3(64), 0,    \
                       stream, (const __hip_bfloat16*)dout.data_ptr(),        \
                       (const __hip_bfloat16*)out.data_ptr(),                 \
                       delta.data_ptr<float>(), B, Sq, Hq, ds, os);           \
    dim3 gridk((Sk + 12 * 16 - 1) / (12 * 16), Hkv, B);                       \
    hipLaunchKernelGGL((fa_bwd_dkdv_kernel<DD, 12, DROP>), gridk,             \
                       dim3(12 * 64), 0,                                      \
                       stream, (const __hip_bfloat16*)q.data_ptr(),           \
                       (const __hip_bfloat16*)k.data_ptr(),                   \
                       (const __hip_bfloat16*)v.data_ptr(),                   \
                       (const __hip_bfloat16*)dout.data_ptr(),                \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),        \
                       (__hip_bfloat16*)dk.data_ptr(),                        \
                       (__hip_bfloat16*)dv.data_ptr(), B, Sq, Sk, Hq, Hkv,    \
                       (float)softmax_scale, causal ? 1 : 0, win, qs, ks, vs, \
                       ds, dks, dvs, dp, dseed, doff);                        \
    dim3 gridq((Sq + 12 * 16 - 1) / (12 * 16), Hq, B);                        \
    hipLaunchKernelGGL((fa_bwd_dq_kernel<DD, 12, DROP>), gridq,               \
                       dim3(12 * 64), 0,                                      \
                       stream, (const __hip_bfloat16*)q.data_ptr(),           \
                       (const __hip_bfloat16*)k.data_ptr(),                   \
                       (const __hip_bfloat16*)v.data_ptr(),                   \
                       (const __hip_bfloat16*)dout.data_ptr(),                \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),        \
                       (__hip_bfloat16*)dq.data_ptr(), B, Sq, Sk, Hq, Hkv,    \
                       (float)softmax_scale, causal ? 1 : 0, win, qs, ks, vs, \
                       ds, dqs, dp, dseed, doff);                             \
  } while (0)

  const bool drop = dropout_p > 0.0;
  if (D == 128) {
    if (drop) LAUNCH_BWD(128, true);
    else LAUNCH_BWD(128, false);
  } else {
    TORCH_CHECK(D == 64);
    if (drop) LAUNCH_BWD(64, true);
    else LAUNCH_BWD(64, false);
  }
  return {dq, dk, dv};
}
