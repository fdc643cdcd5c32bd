// Fused single-token decode attention over the static KV cache.
//
// Replaces the ~10-kernel eager chain in the graph-captured decode step
// (fp32 casts of the whole cache + bmm + masked_fill + softmax + bmm,
// models/transformer.py _static_decode): one kernel per layer reads the
// cache once in bf16, does the online softmax in registers, and emits the
// context row. Length comes from a DEVICE tensor so the kernel is
// hipGraph-capturable with a growing cache.
//
//   q:       [b, n, h]     (one token per sequence)
//   k,v:     [L, b, nkv, h] (the static cache)
//   pos:     int32/int64 device scalar — attend to l in [0, pos]
//   out:     [b, n*h]
//
// Geometry: one 256-thread workgroup per (b, head); 4 waves stride the
// cache rows (one row per wave per round, 64 lanes x 4 B = the full 256 B
// row, coalesced); per-wave online softmax state; cross-wave combine
// through LDS at the end. h = 64 or 128.

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace {

template <int H>
__global__ __launch_bounds__(256) void decode_attn_kernel(
    const __hip_bfloat16* __restrict__ q,
    const __hip_bfloat16* __restrict__ k_cache,
    const __hip_bfloat16* __restrict__ v_cache,
    const long* __restrict__ pos_ptr, __hip_bfloat16* __restrict__ out,
    int B, int N, int NKV, int L, float scale, int window) {
  constexpr int kPerLane = H / 64;  // q/k/v elements per lane (1 or 2)
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const int hkv = h / (N / NKV);
  const long pos = *pos_ptr;  // last valid cache row

  // q row -> registers (each lane holds its kPerLane elems)
  float qr[kPerLane];
  const __hip_bfloat16* qrow = q + ((long)b * N + h) * H;
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) {
    qr[e] = __bfloat162float(qrow[lane * kPerLane + e]) * scale;
  }

  const long row_stride = (long)B * NKV * H;
  const __hip_bfloat16* kb = k_cache + ((long)b * NKV + hkv) * H;
  const __hip_bfloat16* vb = v_cache + ((long)b * NKV + hkv) * H;

  float m = -1e30f, s = 0.f;
  float o[kPerLane];
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) o[e] = 0.f;

  // sliding window: keep `window` keys INCLUDING the current position
  // (HF/Mistral convention, see ops/functional.py _attn_mask)
  long lo = 0;
  if (window > 0 && pos - window + 1 > 0) lo = pos - window + 1;
  for (long l = lo + wave; l <= pos; l += 4) {
    const __hip_bfloat16* krow = kb + l * row_stride;
    float dot = 0.f;
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) {
      dot += qr[e] * __bfloat162float(krow[lane * kPerLane + e]);
    }
    dot = wave_reduce_sum(dot);  // full row dot on every lane
    float m_new = fmaxf(m, dot);
    float alpha = __expf(m - m_new);
    float p = __expf(dot - m_new);
    s = s * alpha + p;
    const __hip_bfloat16* vrow = vb + l * row_stride;
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) {
      o[e] = o[e] * alpha + p * __bfloat162float(vrow[lane * kPerLane + e]);
    }
    m = m_new;
  }

  // cross-wave combine: waves publish (m, s, o) and wave 0 folds
  __shared__ float lds_m[4], lds_s[4];
  __shared__ float lds_o[4][H];
  lds_m[wave] = m;
  lds_s[wave] = s;
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) {
    lds_o[wave][lane * kPerLane + e] = o[e];
  }
  __syncthreads();
  if (wave == 0) {
    float gm = fmaxf(fmaxf(lds_m[0], lds_m[1]), fmaxf(lds_m[2], lds_m[3]));
    float gs = 0.f;
    float acc[kPerLane];
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) acc[e] = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      float a = __expf(lds_m[w] - gm);
      gs += lds_s[w] * a;
#pragma unroll
      for (int e = 0; e < kPerLane; ++e) {
        acc[e] += lds_o[w][lane * kPerLane + e] * a;
      }
    }
    float inv = gs > 0.f ? 1.0f / gs : 0.f;
    __hip_bfloat16* orow = out + ((long)b * N + h) * H;
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) {
      orow[lane * kPerLane + e] = __float2bfloat16(acc[e] * inv);
    }
  }
}

// ---------------------------------------------------------------------------
// split-L ("flash-decoding") path for long caches: grid (n, b, C) computes
// per-chunk online-softmax partials (m, s, o) over an equal slice of
// [lo, pos]; a small combine kernel folds the C partials. At bs1 the base
// kernel has only b*n workgroups — at 32k cache rows that serializes ~8 MB
// of reads per workgroup; the split spreads it across C*b*n workgroups.

template <int H>
__global__ __launch_bounds__(256) void decode_attn_part_kernel(
    const __hip_bfloat16* __restrict__ q,
    const __hip_bfloat16* __restrict__ k_cache,
    const __hip_bfloat16* __restrict__ v_cache,
    const long* __restrict__ pos_ptr, float* __restrict__ part,
    int B, int N, int NKV, int L, float scale, int window, int chunks) {
  constexpr int kPerLane = H / 64;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const int c = blockIdx.z;
  const int hkv = h / (N / NKV);
  const long pos = *pos_ptr;

  long lo = 0;
  if (window > 0 && pos - window + 1 > 0) lo = pos - window + 1;
  const long span = pos - lo + 1;
  const long per = (span + chunks - 1) / chunks;
  const long c_lo = lo + (long)c * per;
  const long c_hi = min(pos, c_lo + per - 1);

  float qr[kPerLane];
  const __hip_bfloat16* qrow = q + ((long)b * N + h) * H;
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) {
    qr[e] = __bfloat162float(qrow[lane * kPerLane + e]) * scale;
  }
  const long row_stride = (long)B * NKV * H;
  const __hip_bfloat16* kb = k_cache + ((long)b * NKV + hkv) * H;
  const __hip_bfloat16* vb = v_cache + ((long)b * NKV + hkv) * H;

  float m = -1e30f, sacc = 0.f;
  float o[kPerLane];
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) o[e] = 0.f;
  for (long l = c_lo + wave; l <= c_hi; l += 4) {
    const __hip_bfloat16* krow = kb + l * row_stride;
    float dot = 0.f;
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) {
      dot += qr[e] * __bfloat162float(krow[lane * kPerLane + e]);
    }
    dot = wave_reduce_sum(dot);
    float m_new = fmaxf(m, dot);
    float alpha = __expf(m - m_new);
    float pv = __expf(dot - m_new);
    sacc = sacc * alpha + pv;
    const __hip_bfloat16* vrow = vb + l * row_stride;
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) {
      o[e] = o[e] * alpha + pv * __bfloat162float(vrow[lane * kPerLane + e]);
    }
    m = m_new;
  }

  __shared__ float lds_m[4], lds_s[4];
  __shared__ float lds_o[4][H];
  lds_m[wave] = m;
  lds_s[wave] = sacc;
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) {
    lds_o[wave][lane * kPerLane + e] = o[e];
  }
  __syncthreads();
  if (wave == 0) {
    float gm = fmaxf(fmaxf(lds_m[0], lds_m[1]), fmaxf(lds_m[2], lds_m[3]));
    float gs = 0.f;
    float acc[kPerLane];
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) acc[e] = 0.f;
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      float a = (gm > -1e29f) ? __expf(lds_m[w] - gm) : 0.f;
      gs += lds_s[w] * a;
#pragma unroll
      for (int e = 0; e < kPerLane; ++e) {
        acc[e] += lds_o[w][lane * kPerLane + e] * a;
      }
    }
    // partial record: [b, n, c, H + 2] = {o[H], m, s}
    float* rec = part + ((((long)b * N + h) * gridDim.z) + c) * (H + 2);
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) {
      rec[lane * kPerLane + e] = acc[e];
    }
    if (lane == 0) {
      rec[H] = gm;
      rec[H + 1] = gs;
    }
  }
}

template <int H>
__global__ __launch_bounds__(64) void decode_attn_combine_kernel(
    const float* __restrict__ part, __hip_bfloat16* __restrict__ out, int B,
    int N, int chunks) {
  constexpr int kPerLane = H / 64;
  const int lane = threadIdx.x;
  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const float* base = part + (((long)b * N + h) * chunks) * (H + 2);
  float gm = -1e30f;
  for (int c = 0; c < chunks; ++c) gm = fmaxf(gm, base[c * (H + 2) + H]);
  float gs = 0.f;
  float acc[kPerLane];
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) acc[e] = 0.f;
  for (int c = 0; c < chunks; ++c) {
    const float* rec = base + c * (H + 2);
    float a = (rec[H] > -1e29f) ? __expf(rec[H] - gm) : 0.f;
    gs += rec[H + 1] * a;
#pragma unroll
    for (int e = 0; e < kPerLane; ++e) {
      acc[e] += rec[lane * kPerLane + e] * a;
    }
  }
  float inv = gs > 0.f ? 1.0f / gs : 0.f;
  __hip_bfloat16* orow = out + ((long)b * N + h) * H;
#pragma unroll
  for (int e = 0; e < kPerLane; ++e) {
    orow[lane * kPerLane + e] = __float2bfloat16(acc[e] * inv);
  }
}

}  // namespace

torch::Tensor decode_attn(torch::Tensor q, torch::Tensor k_cache,
                          torch::Tensor v_cache, torch::Tensor pos,
                          double scale, int64_t window) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3 && q.is_contiguous());
  TORCH_CHECK(k_cache.dim() == 4 && k_cache.is_contiguous());
  TORCH_CHECK(v_cache.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(pos.is_cuda() && pos.scalar_type() == torch::kLong &&
              pos.numel() == 1);
  int B = q.size(0), N = q.size(1), H = q.size(2);
  int L = k_cache.size(0), NKV = k_cache.size(2);
  TORCH_CHECK(k_cache.size(1) == B && k_cache.size(3) == H);
  TORCH_CHECK(H == 64 || H == 128);
  TORCH_CHECK(N % NKV == 0);

  auto out = torch::empty({B, (long)N * H}, q.options());
  auto stream = c10::hip::getCurrentHIPStream();

  // flash-decoding split for long caches: spread the cache sweep across
  // fixed C chunks (device-side pos keeps it graph-capturable; empty
  // chunks cost one kernel round)
  if (L >= 1024) {
    const int C = 8;
    auto part = torch::empty({(long)B * N * C * (H + 2)},
                             q.options().dtype(torch::kFloat32));
    dim3 gridp(N, B, C);
    dim3 gridc(N, B);
    if (H == 128) {
      hipLaunchKernelGGL((decode_attn_part_kernel<128>), gridp, dim3(256), 0,
                         stream, (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k_cache.data_ptr(),
                         (const __hip_bfloat16*)v_cache.data_ptr(),
                         pos.data_ptr<long>(), part.data_ptr<float>(), B, N,
                         NKV, L, (float)scale, (int)window, C);
      hipLaunchKernelGGL((decode_attn_combine_kernel<128>), gridc, dim3(64),
                         0, stream, part.data_ptr<float>(),
                         (__hip_bfloat16*)out.data_ptr(), B, N, C);
    } else {
      hipLaunchKernelGGL((decode_attn_part_kernel<64>), gridp, dim3(256), 0,
                         stream, (const __hip_bfloat16*)q.data_ptr(),
                         (const __hip_bfloat16*)k_cache.data_ptr(),
                         (const __hip_bfloat16*)v_cache.data_ptr(),
                         pos.data_ptr<long>(), part.data_ptr<float>(), B, N,
                         NKV, L, (float)scale, (int)window, C);
      hipLaunchKernelGGL((decode_attn_combine_kernel<64>), gridc, dim3(64),
                         0, stream, part.data_ptr<float>(),
                         (__hip_bfloat16*)out.data_ptr(), B, N, C);
    }
    return out;
  }
  dim3 grid(N, B);
  if (H == 128) {
    hipLaunchKernelGGL((decode_attn_kernel<128>), grid, dim3(256), 0, stream,
                       (const __hip_bfloat16*)q.data_ptr(),
                       (const __hip_bfloat16*)k_cache.data_ptr(),
                       (const __hip_bfloat16*)v_cache.data_ptr(),
                       pos.data_ptr<long>(), (__hip_bfloat16*)out.data_ptr(),
                       B, N, NKV, L, (float)scale, (int)window);
  } else {
    hipLaunchKernelGGL((decode_attn_kernel<64>), grid, dim3(256), 0, stream,
                       (const __hip_bfloat16*)q.data_ptr(),
                       (const __hip_bfloat16*)k_cache.data_ptr(),
                       (const __hip_bfloat16*)v_cache.data_ptr(),
                       pos.data_ptr<long>(), (__hip_bfloat16*)out.data_ptr(),
                       B, N, NKV, L, (float)scale, (int)window);
  }
  return out;
}
