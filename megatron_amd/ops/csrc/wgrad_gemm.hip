// Hand-written CDNA4 weight-gradient GEMM: C_f32[M,N] += A^T B with
// A = grad_output [K, M] bf16 and B = input [K, N] bf16, both K-major
// (row-major with M/N contiguous), K large (32768 at mbs8 seq4096).
//
// hipBLASLt's f32-accumulate kernel pool tops out at 1.03-1.14 PF/s on the
// Llama-7B wgrad shapes (profiles/r02_progress.md) while its bf16-out
// kernels of the same shapes reach 1.4-2.1 PF — this kernel exists to close
// that gap with an MI355X-native structure:
//  - 256x256 C tile per 512-thread workgroup (8 waves as 2M x 4N), BK=64,
//    mfma_f32_16x16x32_bf16, 64 MFMAs per wave per K-step;
//  - both operands staged K-major into LDS "tr images" (the FA kernels'
//    conflict-free tile-row geometry: stride ≡ 0 mod 128 elems + alternating
//    64-elem shift) and read as MFMA fragments with the gfx950 hardware
//    transpose ds_read_b64_tr_b16 — A and B fragments are both k-runs, so
//    NO transposing copies anywhere;
//  - staging via global_load_lds dwordx4 (lane-linear LDS dest, per-lane
//    SOURCE permutation produces the tr-image layout directly);
//  - grouped blockIdx->tile mapping so concurrently resident workgroups
//    share operand panels through L2/L3 (K-major streaming would otherwise
//    be HBM-bound at 24+ GB per GEMM);
//  - double-buffered K-steps; counted-vmcnt phase pipeline layered on in
//    the PIPE variant.

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace {

using frag_b16 = __attribute__((ext_vector_type(8))) short;
using frag_f32 = __attribute__((ext_vector_type(4))) float;
using v4s = __attribute__((ext_vector_type(4))) short;

constexpr int kBM = 256;
constexpr int kBN = 256;
constexpr int kBK = 64;
constexpr int kThreads = 512;

// tile-row geometry for a [kBK x 256] K-major tile (see flash_attn.hip):
// 16 tile-rows of 4 k x 256 cols; stride 256*4+128 = 1152 elems (2304 B
// ≡ 0 mod 256 B so d-run banks cycle cleanly) + ((rt>>1)&1)*64-elem shift
// so ds_read_b64_tr_b16 partner tile-rows sit 32 banks apart.
constexpr int kTRS = 256 * 4 + 128;         // 1152 elems per tile-row
constexpr int kImgElems = (kBK / 4) * kTRS;  // 18432 elems = 36864 B

__device__ __forceinline__ int tr_shift(int rt) { return ((rt >> 1) & 1) * 64; }

__device__ __forceinline__ int img_off(int k, int col) {
  int rt = k >> 2;
  return rt * kTRS + tr_shift(rt) + (col >> 4) * 64 + (k & 3) * 16 +
         (col & 15);
}

// B-fragment (k-run) via two hardware transpose reads: delivers
// img[k0 + (l>>4)*8 + jj][ct*16 + (l&15)] for jj = 0..7.
__device__ __forceinline__ frag_b16 tr_frag(const __hip_bfloat16* img, int k0,
                                            int ct, int lane) {
  const int c = lane & 15;
  const int kt = (k0 >> 2) + (lane >> 4) * 2;
  const int b1 = kt * kTRS + tr_shift(kt) + ct * 64;
  const int b2 = (kt + 1) * kTRS + tr_shift(kt + 1) + ct * 64;
  const __attribute__((address_space(3))) v4s* p1 =
      (const __attribute__((address_space(3))) v4s*)&img[b1] + c;
  const __attribute__((address_space(3))) v4s* p2 =
      (const __attribute__((address_space(3))) v4s*)&img[b2] + c;
  v4s lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) v4s*)p1);
  v4s hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) v4s*)p2);
  frag_b16 f;
  f[0] = lo[0]; f[1] = lo[1]; f[2] = lo[2]; f[3] = lo[3];
  f[4] = hi[0]; f[5] = hi[1]; f[6] = hi[2]; f[7] = hi[3];
  return f;
}

// Stage one [kBK x 256] K-major tile into its tr image with LDS-DMA.
// Each 64-lane wave instruction writes 1024 contiguous LDS bytes (= half a
// tile-row's data); the per-lane SOURCE address realizes the layout:
// dest elem 8*l  <=>  (k = rt*4 + ((l>>1)&3), col = half*128 + (l>>3)*16 +
// (l&1)*8). 32 instructions per tile, spread over the workgroup's 8 waves.
// `piece` selects which of the 4 (tile-row-pair, half) units this wave
// stages this call (4 calls stage the whole tile).
__device__ __forceinline__ void stage_piece(const __hip_bfloat16* src,
                                            long row_stride,
                                            __hip_bfloat16* img, int k_base,
                                            int wave, int lane, int piece) {
  // 32 wave-instructions total: unit u = wave*4 + piece in [0,32):
  // rt = u >> 1, half = u & 1
  int u = wave * 4 + piece;
  int rt = u >> 1;
  int half = u & 1;
  int k = rt * 4 + ((lane >> 1) & 3);
  int col = half * 128 + (lane >> 3) * 16 + (lane & 1) * 8;
  const __hip_bfloat16* gsrc =
      src + (long)(k_base + k) * row_stride + col;
  __hip_bfloat16* dst = img + rt * kTRS + tr_shift(rt) + half * 512;
  // lane-linear LDS-DMA: dest = base + lane*16 B
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
}

// ---------------------------------------------------------------------------
// main kernel. Grid: one block per 256x256 C tile, grouped mapping applied
// host-side via a (tile_m, tile_n) lookup-free swizzle.

__global__ __launch_bounds__(kThreads, 2) void wgrad_gemm_kernel(
    const __hip_bfloat16* __restrict__ a,  // [K, M] grad_output
    const __hip_bfloat16* __restrict__ b,  // [K, N] input
    float* __restrict__ c,                 // [M, N] main_grad (+=)
    int M, int N, int K, int group_m) {
  __shared__ __hip_bfloat16 a_img[2][kImgElems];
  __shared__ __hip_bfloat16 b_img[2][kImgElems];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;       // 0..1: 128-row band
  const int wn = wave & 3;        // 0..3: 64-col band

  // grouped tile mapping: consecutive blocks walk a group_m-tall column of
  // M-tiles before moving to the next N-tile column, so resident blocks
  // share A panels (and successive columns share B via L3)
  const int tiles_m = M / kBM;
  int bid = blockIdx.x;
  int group_sz = group_m * (N / kBN);
  int g = bid / group_sz;
  int r = bid % group_sz;
  int gm = min(group_m, tiles_m - g * group_m);
  int tm = g * group_m + r % gm;
  int tn = r / gm;

  const int m0 = tm * kBM;
  const int n0 = tn * kBN;

  // 32 accumulator fragments: 8 m-frags x 4 n-frags of 16x16
  frag_f32 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = frag_f32{0.f, 0.f, 0.f, 0.f};
  }

  const int ksteps = K / kBK;
  // prologue: stage tile 0 (all 4 pieces per wave), start tile 1
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    stage_piece(a + m0, M, a_img[0], 0, wave, lane, p);
    stage_piece(b + n0, N, b_img[0], 0, wave, lane, p);
  }
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < ksteps; ++kt) {
    const __hip_bfloat16* ai = a_img[kt & 1];
    const __hip_bfloat16* bi = b_img[kt & 1];
    // issue next tile's staging (8 glds per wave) — lands behind this
    // step's MFMAs, drained by the counted wait before the flip barrier
    if (kt + 1 < ksteps) {
#pragma unroll
      for (int p = 0; p < 4; ++p) {
        stage_piece(a + m0, M, a_img[(kt + 1) & 1], (kt + 1) * kBK, wave,
                    lane, p);
        stage_piece(b + n0, N, b_img[(kt + 1) & 1], (kt + 1) * kBK, wave,
                    lane, p);
      }
    }

    // 64 MFMAs: 2 k-chunks x 8 m-frags x 4 n-frags, B-frags hoisted per
    // chunk (reused by all 8 m-frags)
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      const int k0 = kc * 32;
      frag_b16 bf[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        bf[j] = tr_frag(bi, k0, (wn * 64 + j * 16) >> 4, lane);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        frag_b16 af = tr_frag(ai, k0, (wm * 128 + i * 16) >> 4, lane);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf[j],
                                                              acc[i][j], 0,
                                                              0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    // drain this iteration's glds before any wave flips buffers
    asm volatile("s_waitcnt vmcnt(0)");
    asm volatile("s_waitcnt lgkmcnt(0)");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: C += acc. Lane l's fragment element (r, c): row = (l>>4)*4+r,
  // col = l%16 — 16 lanes per row are contiguous f32 (64 B stores).
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int row_base = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      long row = row_base + r;
      float* crow = c + row * (long)N + n0 + wn * 64 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        crow[j * 16] += acc[i][j][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// v2: counted-vmcnt half-tile pipeline (the T3+T4 structure). Two phases per
// K-step, each {issue 4 glds for tile t+1's half h; 32 MFMAs on tile t's
// half h; s_waitcnt vmcnt(4) — the NEXT phase's half has landed, the one
// just issued stays in flight ACROSS the barrier; raw s_barrier}. Never
// vmcnt(0) in the main loop (the 2-phase v1 above drains the whole glds
// queue at every barrier — its measured ~900 TF ceiling).

__global__ __launch_bounds__(kThreads, 2) void wgrad_gemm_pipe_kernel(
    const __hip_bfloat16* __restrict__ a,  // [K, M]
    const __hip_bfloat16* __restrict__ b,  // [K, N]
    float* __restrict__ c, int M, int N, int K, int group_m) {
  __shared__ __hip_bfloat16 a_img[2][kImgElems];
  __shared__ __hip_bfloat16 b_img[2][kImgElems];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  const int tiles_m = M / kBM;
  int bid = blockIdx.x;
  int group_sz = group_m * (N / kBN);
  int g = bid / group_sz;
  int r = bid % group_sz;
  int gm = min(group_m, tiles_m - g * group_m);
  int tm = g * group_m + r % gm;
  int tn = r / gm;
  const int m0 = tm * kBM;
  const int n0 = tn * kBN;

  frag_f32 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = frag_f32{0.f, 0.f, 0.f, 0.f};
  }

  // stage_half: half h of tile kt = tile-rows [h*8, h*8+8) = k in
  // [h*32, h*32+32); 16 glds instructions over 8 waves = 2 per wave
  auto stage_half = [&](const __hip_bfloat16* src, long ld,
                        __hip_bfloat16* img, int k_base, int h) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      int u = h * 16 + wave * 2 + p;  // unit in [h*16, h*16+16)
      int rt = u >> 1;
      int half = u & 1;
      int k = rt * 4 + ((lane >> 1) & 3);
      int col = half * 128 + (lane >> 3) * 16 + (lane & 1) * 8;
      const __hip_bfloat16* gsrc = src + (long)(k_base + k) * ld + col;
      __hip_bfloat16* dst = img + rt * kTRS + tr_shift(rt) + half * 512;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gsrc,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  };
  // one compute phase: 32 MFMAs on k-chunk kc of the given images
  auto compute_half = [&](const __hip_bfloat16* ai, const __hip_bfloat16* bi,
                          int kc) {
    const int k0 = kc * 32;
    frag_b16 bf[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf[j] = tr_frag(bi, k0, (wn * 64 + j * 16) >> 4, lane);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      frag_b16 af = tr_frag(ai, k0, (wm * 128 + i * 16) >> 4, lane);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf[j],
                                                            acc[i][j], 0, 0,
                                                            0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  };

  const int ksteps = K / kBK;
  // prologue: tile 0 fully, tile 1 half 0; drain everything once
  stage_half(a + m0, M, a_img[0], 0, 0);
  stage_half(b + n0, N, b_img[0], 0, 0);
  stage_half(a + m0, M, a_img[0], 0, 1);
  stage_half(b + n0, N, b_img[0], 0, 1);
  if (ksteps > 1) {
    stage_half(a + m0, M, a_img[1], kBK, 0);
    stage_half(b + n0, N, b_img[1], kBK, 0);
  }
  if (ksteps > 1) {
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // tile 0 landed
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < ksteps; ++kt) {
    const int cur = kt & 1;
    const int nxt = cur ^ 1;
    const bool more = kt + 1 < ksteps;
    // phase A: issue t+1 half1, compute t half0
    if (more) {
      stage_half(a + m0, M, a_img[nxt], (kt + 1) * kBK, 1);
      stage_half(b + n0, N, b_img[nxt], (kt + 1) * kBK, 1);
    }
    compute_half(a_img[cur], b_img[cur], 0);
    // t's halves landed long ago; t+1.h0 must land before NEXT tile's
    // phase A, but phase B only needs t.h1 (resident) — keep everything
    // in flight here, no wait, LDS-safety barrier only
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    // phase B: issue t+2 half0, compute t half1
    if (kt + 2 < ksteps) {
      stage_half(a + m0, M, a_img[cur], (kt + 2) * kBK, 0);
      stage_half(b + n0, N, b_img[cur], (kt + 2) * kBK, 0);
    }
    compute_half(a_img[cur], b_img[cur], 1);
    // before the buffer flip: t+1's halves (issued 2 and 1 phases ago)
    // must be LDS-visible; t+2.h0 (4 glds/wave) may stay in flight
    if (more) {
      if (kt + 2 < ksteps) {
        asm volatile("s_waitcnt vmcnt(4) lgkmcnt(0)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
      }
    } else {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int row_base = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int r2 = 0; r2 < 4; ++r2) {
      long row = row_base + r2;
      float* crow = c + row * (long)N + n0 + wn * 64 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        crow[j * 16] += acc[i][j][r2];
      }
    }
  }
}

}  // namespace

// Host entry: returns false when the shape does not fit the fast path
// (caller falls back to hipBLASLt).
bool wgrad_gemm_hand(torch::Tensor input, torch::Tensor grad_output,
                     torch::Tensor main_grad) {
  long K = input.size(0);
  long N = input.size(1);    // in_dim
  long M = grad_output.size(1);  // out_dim
  if (input.scalar_type() != torch::kBFloat16 ||
      grad_output.scalar_type() != torch::kBFloat16) {
    return false;
  }
  if (M % kBM || N % kBN || K % kBK) return false;

  static const int group_m = []() {
    const char* e = getenv("MEGATRON_AMD_WGRAD_GROUP_M");
    return e ? atoi(e) : 8;
  }();
  static const int variant = []() {
    const char* e = getenv("MEGATRON_AMD_WGRAD_V");
    return e ? atoi(e) : 2;  // 2 = counted-vmcnt pipeline, 1 = simple
  }();
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid((M / kBM) * (N / kBN));
  if (variant == 2) {
    hipLaunchKernelGGL(wgrad_gemm_pipe_kernel, grid, dim3(kThreads), 0,
                       stream,
                       (const __hip_bfloat16*)grad_output.data_ptr(),
                       (const __hip_bfloat16*)input.data_ptr(),
                       main_grad.data_ptr<float>(), (int)M, (int)N, (int)K,
                       group_m);
  } else {
    hipLaunchKernelGGL(wgrad_gemm_kernel, grid, dim3(kThreads), 0, stream,
                       (const __hip_bfloat16*)grad_output.data_ptr(),
                       (const __hip_bfloat16*)input.data_ptr(),
                       main_grad.data_ptr<float>(), (int)M, (int)N, (int)K,
                       group_m);
  }
  return true;
}
