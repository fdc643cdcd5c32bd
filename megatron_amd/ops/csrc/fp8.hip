// fp8 (e4m3) quantization for the fp8 GEMM recipe (megatron_amd/fp8.py).
//
// ONE pass over the input: read bf16, write the saturated e4m3 value using
// the scale from the PREVIOUS step (delayed scaling a la TransformerEngine)
// and reduce this tensor's amax on the way through — the eager-torch
// version costs three passes (abs().amax(), mul, cast). amax reduction is
// per-block into registers/LDS, then one fp32 atomicMax per block (positive
// floats compare correctly as uint32 bit patterns).

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <c10/util/Float8_e4m3fn.h>

#include <vector>

namespace {

constexpr int kBlock = 256;
constexpr float kE4M3Max = 448.0f;

__device__ __forceinline__ void atomic_max_pos_float(float* addr, float val) {
  // positive IEEE floats order like their uint bit patterns
  atomicMax(reinterpret_cast<unsigned int*>(addr), __float_as_uint(val));
}

__global__ void fp8_quantize_kernel(const __hip_bfloat16* __restrict__ x,
                                    const float* __restrict__ scale_inv_src,
                                    uint8_t* __restrict__ q,
                                    float* __restrict__ amax, long n) {
  // scale_inv_src holds the DEQUANT scale s (x ~= q * s); quantize by 1/s
  const float qscale = 1.0f / scale_inv_src[0];
  float local_amax = 0.f;
  for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n;
       i += (long)gridDim.x * kBlock) {
    float v = __bfloat162float(x[i]);
    local_amax = fmaxf(local_amax, fabsf(v));
    float scaled = v * qscale;
    scaled = fminf(fmaxf(scaled, -kE4M3Max), kE4M3Max);
    q[i] = c10::Float8_e4m3fn(scaled).x;
  }
  __shared__ float lds[kBlock / WAVE_SIZE];
  local_amax = wave_reduce_max(local_amax);
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  if (lane == 0) lds[wave] = local_amax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = 0.f;
    for (int w = 0; w < kBlock / WAVE_SIZE; ++w) m = fmaxf(m, lds[w]);
    atomic_max_pos_float(amax, m);
  }
}

inline int grid_for(long total) {
  long g = (total + kBlock - 1) / kBlock;
  return (int)std::min<long>(g, 8192);
}

}  // namespace

std::vector<torch::Tensor> fp8_quantize(torch::Tensor x,
                                        torch::Tensor scale) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous()
              && x.scalar_type() == torch::kBFloat16,
              "fp8_quantize: contiguous bf16 input required");
  TORCH_CHECK(scale.is_cuda() && scale.numel() == 1
              && scale.scalar_type() == torch::kFloat32);
  long n = x.numel();
  auto q = torch::empty_like(x, x.options().dtype(torch::kFloat8_e4m3fn));
  auto amax = torch::zeros({1}, x.options().dtype(torch::kFloat32));
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fp8_quantize_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                     stream, (const __hip_bfloat16*)x.data_ptr(),
                     scale.data_ptr<float>(), (uint8_t*)q.data_ptr(),
                     amax.data_ptr<float>(), n);
  return {q, amax};
}
