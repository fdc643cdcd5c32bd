// Fused scale + (causal / boolean-masked) softmax forward + backward.
//
// Replaces the reference's scaled_upper_triang_masked_softmax /
// scaled_masked_softmax / scaled_softmax CUDA extensions
// (megatron/fused_kernels/*.cu, fused_softmax.py:9-99).
//
// Input viewed as (b, np, sq, sk): one 256-thread block per (b, np, sq) row,
// fp32 math, output stored in the input dtype. Causal masking is computed
// from indices (no mask tensor); the boolean mask path takes a
// (b, 1, sq, sk)-broadcastable uint8 mask (1 = masked out).
// Wave64 reductions; sk is unbounded (thread-strided row loop).

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace {

constexpr int kBlock = 256;

template <typename T, bool CAUSAL, bool HAS_MASK>
__global__ void softmax_fwd_kernel(const T* __restrict__ x,
                                   const uint8_t* __restrict__ mask,
                                   T* __restrict__ y, float scale, int b,
                                   int np, int sq, int sk) {
  __shared__ float lds[kBlock / WAVE_SIZE];
  const long row = blockIdx.x;          // row in (b*np*sq)
  const int q_idx = row % sq;
  const long bn = row / sq;             // b*np index
  const int b_idx = bn / np;

  const T* xr = x + row * (long)sk;
  T* yr = y + row * (long)sk;
  const uint8_t* mr = HAS_MASK ? mask + ((long)b_idx * sq + q_idx) * sk
                               : nullptr;

  // allowed key range for causal: k <= q + (sk - sq)
  const int limit = CAUSAL ? (q_idx + (sk - sq) + 1) : sk;

  float maxv = -INFINITY;
  for (int i = threadIdx.x; i < sk; i += kBlock) {
    bool masked = (CAUSAL && i >= limit) || (HAS_MASK && mr[i]);
    float v = masked ? -INFINITY
                     : DTypeTraits<T>::to_float(xr[i]) * scale;
    maxv = fmaxf(maxv, v);
  }
  maxv = block_reduce_max<kBlock>(maxv, lds);
  __syncthreads();

  float sum = 0.f;
  for (int i = threadIdx.x; i < sk; i += kBlock) {
    bool masked = (CAUSAL && i >= limit) || (HAS_MASK && mr[i]);
    float v = masked ? 0.f
                     : __expf(DTypeTraits<T>::to_float(xr[i]) * scale - maxv);
    sum += v;
    // stash exp in output to avoid recomputing
    yr[i] = DTypeTraits<T>::from_float(v);
  }
  sum = block_reduce_sum<kBlock>(sum, lds);
  float rsum = (sum > 0.f) ? 1.0f / sum : 0.f;
  for (int i = threadIdx.x; i < sk; i += kBlock) {
    float v = DTypeTraits<T>::to_float(yr[i]);
    yr[i] = DTypeTraits<T>::from_float(v * rsum);
  }
}

template <typename T>
__global__ void softmax_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ y,
                                   T* __restrict__ dx, float scale, int sk) {
  __shared__ float lds[kBlock / WAVE_SIZE];
  const long row = blockIdx.x;
  const T* dyr = dy + row * (long)sk;
  const T* yr = y + row * (long)sk;
  T* dxr = dx + row * (long)sk;

  float dot = 0.f;
  for (int i = threadIdx.x; i < sk; i += kBlock) {
    dot += DTypeTraits<T>::to_float(dyr[i]) * DTypeTraits<T>::to_float(yr[i]);
  }
  dot = block_reduce_sum<kBlock>(dot, lds);
  for (int i = threadIdx.x; i < sk; i += kBlock) {
    float g = DTypeTraits<T>::to_float(dyr[i]);
    float p = DTypeTraits<T>::to_float(yr[i]);
    dxr[i] = DTypeTraits<T>::from_float((g - dot) * p * scale);
  }
}

template <typename T>
void softmax_fwd_dispatch(const torch::Tensor& x, const torch::Tensor* mask,
                          torch::Tensor& y, float scale, bool causal) {
  int b = x.size(0), np = x.size(1), sq = x.size(2), sk = x.size(3);
  long rows = (long)b * np * sq;
  auto stream = c10::hip::getCurrentHIPStream();
  const uint8_t* mptr =
      mask ? (const uint8_t*)mask->data_ptr<bool>() : nullptr;
  if (causal && mask) {
    hipLaunchKernelGGL((softmax_fwd_kernel<T, true, true>), dim3(rows),
                       dim3(kBlock), 0, stream, (const T*)x.data_ptr(), mptr,
                       (T*)y.data_ptr(), scale, b, np, sq, sk);
  } else if (causal) {
    hipLaunchKernelGGL((softmax_fwd_kernel<T, true, false>), dim3(rows),
                       dim3(kBlock), 0, stream, (const T*)x.data_ptr(),
                       nullptr, (T*)y.data_ptr(), scale, b, np, sq, sk);
  } else if (mask) {
    hipLaunchKernelGGL((softmax_fwd_kernel<T, false, true>), dim3(rows),
                       dim3(kBlock), 0, stream, (const T*)x.data_ptr(), mptr,
                       (T*)y.data_ptr(), scale, b, np, sq, sk);
  } else {
    hipLaunchKernelGGL((softmax_fwd_kernel<T, false, false>), dim3(rows),
                       dim3(kBlock), 0, stream, (const T*)x.data_ptr(),
                       nullptr, (T*)y.data_ptr(), scale, b, np, sq, sk);
  }
}

}  // namespace

torch::Tensor scaled_masked_softmax_fwd(torch::Tensor x,
                                        c10::optional<torch::Tensor> mask,
                                        double scale, bool causal) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous());
  auto y = torch::empty_like(x);
  const torch::Tensor* mptr = nullptr;
  torch::Tensor m;
  if (mask.has_value()) {
    m = mask.value();
    TORCH_CHECK(m.scalar_type() == torch::kBool);
    // broadcast (b,1,sq,sk) -> (b,sq,sk) contiguous
    m = m.expand({x.size(0), 1, x.size(2), x.size(3)}).contiguous();
    mptr = &m;
  }
  if (x.scalar_type() == torch::kBFloat16) {
    softmax_fwd_dispatch<__hip_bfloat16>(x, mptr, y, (float)scale, causal);
  } else if (x.scalar_type() == torch::kFloat16) {
    softmax_fwd_dispatch<__half>(x, mptr, y, (float)scale, causal);
  } else {
    softmax_fwd_dispatch<float>(x, mptr, y, (float)scale, causal);
  }
  return y;
}

torch::Tensor scaled_masked_softmax_bwd(torch::Tensor dy, torch::Tensor y,
                                        double scale) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4 && dy.is_contiguous());
  auto dx = torch::empty_like(dy);
  long rows = (long)dy.size(0) * dy.size(1) * dy.size(2);
  int sk = dy.size(3);
  auto stream = c10::hip::getCurrentHIPStream();
  if (dy.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((softmax_bwd_kernel<__hip_bfloat16>), dim3(rows),
                       dim3(kBlock), 0, stream,
                       (const __hip_bfloat16*)dy.data_ptr(),
                       (const __hip_bfloat16*)y.data_ptr(),
                       (__hip_bfloat16*)dx.data_ptr(), (float)scale, sk);
  } else if (dy.scalar_type() == torch::kFloat16) {
    hipLaunchKernelGGL((softmax_bwd_kernel<__half>), dim3(rows), dim3(kBlock),
                       0, stream, (const __half*)dy.data_ptr(),
                       (const __half*)y.data_ptr(), (__half*)dx.data_ptr(),
                       (float)scale, sk);
  } else {
    hipLaunchKernelGGL((softmax_bwd_kernel<float>), dim3(rows), dim3(kBlock),
                       0, stream, (const float*)dy.data_ptr(),
                       (const float*)y.data_ptr(), (float*)dx.data_ptr(),
                       (float)scale, sk);
  }
  return dx;
}
