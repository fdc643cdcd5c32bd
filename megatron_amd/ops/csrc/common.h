// Shared helpers for the MI355X (gfx950, CDNA4) ops extension.
//
// Conventions:
//  - wavefront is 64 lanes (CDNA); all cross-lane reductions use
//    __shfl_xor over 64 lanes;
//  - blocks are multiples of 64 threads (256 default);
//  - bf16/fp16 global traffic is vectorized 8-elements-per-lane (16 B)
//    wherever the layout allows (HBM3E-bound kernels want 16 B/lane).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e = (cmd);                                                     \
    if (e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(e));                         \
    }                                                                         \
  } while (0)

// ---- dtype traits ---------------------------------------------------------

template <typename T>
struct DTypeTraits;

template <>
struct DTypeTraits<float> {
  using Vec = float4;                      // 4 floats = 16 B
  static constexpr int kVecLen = 4;
  static __device__ __forceinline__ float to_float(float v) { return v; }
  static __device__ __forceinline__ float from_float(float v) { return v; }
};

template <>
struct DTypeTraits<__hip_bfloat16> {
  static constexpr int kVecLen = 8;        // 8 bf16 = 16 B
  static __device__ __forceinline__ float to_float(__hip_bfloat16 v) {
    return __bfloat162float(v);
  }
  static __device__ __forceinline__ __hip_bfloat16 from_float(float v) {
    return __float2bfloat16(v);
  }
};

template <>
struct DTypeTraits<__half> {
  static constexpr int kVecLen = 8;
  static __device__ __forceinline__ float to_float(__half v) {
    return __half2float(v);
  }
  static __device__ __forceinline__ __half from_float(float v) {
    return __float2half(v);
  }
};

// ---- wave + block reductions ---------------------------------------------

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int offset = WAVE_SIZE / 2; offset > 0; offset >>= 1) {
    v += __shfl_xor(v, offset, WAVE_SIZE);
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int offset = WAVE_SIZE / 2; offset > 0; offset >>= 1) {
    v = fmaxf(v, __shfl_xor(v, offset, WAVE_SIZE));
  }
  return v;
}

// Block reduction across up to 16 waves (1024 threads) via LDS.
template <int BLOCK>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds) {
  constexpr int NWAVES = BLOCK / WAVE_SIZE;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  if (wave == 0) {
    float w = (lane < NWAVES) ? lds[lane] : 0.f;
    w = wave_reduce_sum(w);
    if (lane == 0) lds[0] = w;
  }
  __syncthreads();
  return lds[0];
}

template <int BLOCK>
__device__ __forceinline__ float block_reduce_max(float v, float* lds) {
  constexpr int NWAVES = BLOCK / WAVE_SIZE;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  if (wave == 0) {
    float w = (lane < NWAVES) ? lds[lane] : -INFINITY;
    w = wave_reduce_max(w);
    if (lane == 0) lds[0] = w;
  }
  __syncthreads();
  return lds[0];
}

// ---- vectorized 8x bf16/half load/store as uint4 --------------------------

union Bf16x8 {
  uint4 u;
  __hip_bfloat16 h[8];
};

union Half8 {
  uint4 u;
  __half h[8];
};

union Float4 {
  float4 f4;
  float f[4];
};

// ---- philox4x32-10 counter-based RNG --------------------------------------

__device__ __forceinline__ void philox_round(uint4& ctr, uint2& key) {
  const unsigned PHILOX_M0 = 0xD2511F53u;
  const unsigned PHILOX_M1 = 0xCD9E8D57u;
  unsigned hi0 = __umulhi(PHILOX_M0, ctr.x);
  unsigned lo0 = PHILOX_M0 * ctr.x;
  unsigned hi1 = __umulhi(PHILOX_M1, ctr.z);
  unsigned lo1 = PHILOX_M1 * ctr.z;
  ctr = make_uint4(hi1 ^ ctr.y ^ key.x, lo1, hi0 ^ ctr.w ^ key.y, lo0);
  key.x += 0x9E3779B9u;
  key.y += 0xBB67AE85u;
}

__device__ __forceinline__ uint4 philox10(unsigned long long seed,
                                          unsigned long long offset,
                                          unsigned idx) {
  uint2 key = make_uint2((unsigned)seed, (unsigned)(seed >> 32));
  uint4 ctr = make_uint4(idx, (unsigned)offset, (unsigned)(offset >> 32), 0);
#pragma unroll
  for (int i = 0; i < 10; ++i) philox_round(ctr, key);
  return ctr;
}

__device__ __forceinline__ float uint_to_uniform(unsigned x) {
  // (0,1] uniform
  return (x >> 8) * (1.0f / 16777216.0f);
}

// 64-bit-counter variant for index spaces past 2^32 (FA attention dropout:
// counter n = element/4, ctr = (n_lo, n_hi, offset_lo, offset_hi), key =
// seed). Mirrored bit-exactly by tests/philox_ref.py.
__device__ __forceinline__ uint4 philox10_ctr64(unsigned long long seed,
                                                unsigned long long offset,
                                                unsigned long long n) {
  uint2 key = make_uint2((unsigned)seed, (unsigned)(seed >> 32));
  uint4 ctr = make_uint4((unsigned)n, (unsigned)(n >> 32), (unsigned)offset,
                         (unsigned)(offset >> 32));
#pragma unroll
  for (int i = 0; i < 10; ++i) philox_round(ctr, key);
  return ctr;
}
