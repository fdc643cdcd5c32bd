// Decode-time GEMV: y[T, M] = x[T, K] @ W[M, K]^T for T <= 8 tokens, bf16.
//
// hipBLASLt's M=1 GEMM selection leaves the single-stream decode far off
// the weight-streaming roofline (the 7B bf16 decode step reads ~13.5 GB of
// weights; 8 TB/s HBM3E bounds ~470 tok/s while the library path measured
// ~111). This kernel is a plain weight-stream: one wave per output row,
// dwordx4 nontemporal loads of the row (each CU reads each weight byte
// exactly once per token — the guide's nt-weights case), x served from
// L2, fp32 accumulation, wave-reduced. No LDS round trip (GEMV operands
// are not shared across waves).
//
// Replaces the reference's reliance on cuBLAS for generation-time matmuls
// (megatron/text_generation/forward_step.py) on the hot decode path.

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace {

template <int T>
__global__ __launch_bounds__(256) void gemv_kernel(
    const __hip_bfloat16* __restrict__ w,  // [M, K]
    const __hip_bfloat16* __restrict__ x,  // [T, K]
    __hip_bfloat16* __restrict__ y,        // [T, M]
    int M, int K) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row = blockIdx.x * 4 + wave;
  if (row >= M) return;

  using v4u = __attribute__((ext_vector_type(4))) unsigned int;
  const v4u* wrow = reinterpret_cast<const v4u*>(w + (long)row * K);
  float acc[T];
#pragma unroll
  for (int t = 0; t < T; ++t) acc[t] = 0.f;

  const int kv = K / 8;  // dwordx4 chunks per row
  for (int i = lane; i < kv; i += 64) {
    v4u wraw = __builtin_nontemporal_load(&wrow[i]);
    Bf16x8 wv;
    wv.u = make_uint4(wraw.x, wraw.y, wraw.z, wraw.w);
#pragma unroll
    for (int t = 0; t < T; ++t) {
      Bf16x8 xv;
      xv.u = *reinterpret_cast<const uint4*>(x + (long)t * K + i * 8);
      float a = acc[t];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        a = fmaf(__bfloat162float(wv.h[e]), __bfloat162float(xv.h[e]), a);
      }
      acc[t] = a;
    }
  }
#pragma unroll
  for (int t = 0; t < T; ++t) {
    float r = wave_reduce_sum(acc[t]);
    if (lane == 0) {
      y[(long)t * M + row] = __float2bfloat16(r);
    }
  }
}

}  // namespace

torch::Tensor gemv_bf16(torch::Tensor weight, torch::Tensor x) {
  TORCH_CHECK(weight.is_cuda() && weight.dim() == 2 && weight.is_contiguous());
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(weight.scalar_type() == torch::kBFloat16 &&
              x.scalar_type() == torch::kBFloat16);
  long M = weight.size(0), K = weight.size(1);
  long T = x.size(0);
  TORCH_CHECK(x.size(1) == K && K % 8 == 0 && T >= 1 && T <= 4);

  auto y = torch::empty({T, M}, x.options());
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid((M + 3) / 4);
#define LAUNCH_GEMV(TT)                                                   \
  hipLaunchKernelGGL((gemv_kernel<TT>), grid, dim3(256), 0, stream,       \
                     (const __hip_bfloat16*)weight.data_ptr(),            \
                     (const __hip_bfloat16*)x.data_ptr(),                 \
                     (__hip_bfloat16*)y.data_ptr(), (int)M, (int)K)
  switch (T) {
    case 1: LAUNCH_GEMV(1); break;
    case 2: LAUNCH_GEMV(2); break;
    case 3: LAUNCH_GEMV(3); break;
    default: LAUNCH_GEMV(4); break;
  }
#undef LAUNCH_GEMV
  return y;
}
