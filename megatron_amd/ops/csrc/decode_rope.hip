// Fused decode-step RoPE + KV-cache append.
//
// One kernel replaces the per-layer decode chain {2x .contiguous() slice
// copies, rope(q), rope(k), 2x index_copy into the cache} — with the
// position read from a DEVICE tensor so the op is hipGraph-capturable.
//
//   q,k,v:   [b, n|nkv, h] strided views of the QKV projection (sq = 1)
//   cos,sin: [S, h/2] fp32 rope tables (interleaved-pair convention,
//            models/rope.py)
//   pos:     int64 device scalar — the cache row to write / rope position
//   k_cache, v_cache: [L, b, nkv, h] contiguous
//   returns: q_rot [b, n, h] contiguous (bf16)

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace {

__global__ __launch_bounds__(64) void decode_rope_append_kernel(
    const __hip_bfloat16* __restrict__ q, long qb, long qh,
    const __hip_bfloat16* __restrict__ k, long kb, long kh,
    const __hip_bfloat16* __restrict__ v, long vb, long vh,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    const long* __restrict__ pos_ptr, __hip_bfloat16* __restrict__ k_cache,
    __hip_bfloat16* __restrict__ v_cache, __hip_bfloat16* __restrict__ q_out,
    int B, int N, int NKV, int H) {
  const int lane = threadIdx.x;
  const int head = blockIdx.x;  // [0,N) q | [N,N+NKV) k | [N+NKV,N+2NKV) v
  const int b = blockIdx.y;
  const long pos = *pos_ptr;
  const float* crow = cos_t + pos * (H / 2);
  const float* srow = sin_t + pos * (H / 2);
  const long cache_row = pos * (long)B * NKV * H;

  if (head < N) {
    const __hip_bfloat16* src = q + b * qb + head * qh;
    __hip_bfloat16* dst = q_out + ((long)b * N + head) * H;
    for (int i = lane; i < H / 2; i += 64) {
      float x1 = __bfloat162float(src[2 * i]);
      float x2 = __bfloat162float(src[2 * i + 1]);
      float c = crow[i], s = srow[i];
      dst[2 * i] = __float2bfloat16(x1 * c - x2 * s);
      dst[2 * i + 1] = __float2bfloat16(x2 * c + x1 * s);
    }
  } else if (head < N + NKV) {
    const int hk = head - N;
    const __hip_bfloat16* src = k + b * kb + hk * kh;
    __hip_bfloat16* dst = k_cache + cache_row + ((long)b * NKV + hk) * H;
    for (int i = lane; i < H / 2; i += 64) {
      float x1 = __bfloat162float(src[2 * i]);
      float x2 = __bfloat162float(src[2 * i + 1]);
      float c = crow[i], s = srow[i];
      dst[2 * i] = __float2bfloat16(x1 * c - x2 * s);
      dst[2 * i + 1] = __float2bfloat16(x2 * c + x1 * s);
    }
  } else {
    const int hv = head - N - NKV;
    const __hip_bfloat16* src = v + b * vb + hv * vh;
    __hip_bfloat16* dst = v_cache + cache_row + ((long)b * NKV + hv) * H;
    for (int i = lane; i < H / 2; i += 64) {
      dst[2 * i] = src[2 * i];
      dst[2 * i + 1] = src[2 * i + 1];
    }
  }
}

}  // namespace

torch::Tensor decode_rope_append(torch::Tensor q, torch::Tensor k,
                                 torch::Tensor v, torch::Tensor cos,
                                 torch::Tensor sin, torch::Tensor pos,
                                 torch::Tensor k_cache,
                                 torch::Tensor v_cache) {
  TORCH_CHECK(q.is_cuda() && q.dim() == 3 && q.stride(2) == 1);
  TORCH_CHECK(k.dim() == 3 && k.stride(2) == 1);
  TORCH_CHECK(v.dim() == 3 && v.stride(2) == 1);
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(cos.scalar_type() == torch::kFloat32 && cos.is_contiguous());
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(pos.scalar_type() == torch::kLong && pos.numel() == 1);
  int B = q.size(0), N = q.size(1), H = q.size(2);
  int NKV = k.size(1);
  TORCH_CHECK(k_cache.size(1) == B && k_cache.size(2) == NKV &&
              k_cache.size(3) == H);
  TORCH_CHECK(H % 2 == 0);

  auto q_out = torch::empty({B, N, H}, q.options());
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid(N + 2 * NKV, B);
  hipLaunchKernelGGL(decode_rope_append_kernel, grid, dim3(64), 0, stream,
                     (const __hip_bfloat16*)q.data_ptr(), q.stride(0),
                     q.stride(1),
                     (const __hip_bfloat16*)k.data_ptr(), k.stride(0),
                     k.stride(1),
                     (const __hip_bfloat16*)v.data_ptr(), v.stride(0),
                     v.stride(1), cos.data_ptr<float>(),
                     sin.data_ptr<float>(), pos.data_ptr<long>(),
                     (__hip_bfloat16*)k_cache.data_ptr(),
                     (__hip_bfloat16*)v_cache.data_ptr(),
                     (__hip_bfloat16*)q_out.data_ptr(), B, N, NKV, H);
  return q_out;
}
