// Fused elementwise kernels: GLU family (SwiGLU/GeGLU/ReGLU/LiGLU),
// rotary embedding, bias+dropout+residual-add.
//
// Replaces the reference's jit-scripted fusions (fused_bias_gelu.py:14-43,
// transformer.py:596-609 bias_dropout_add, glu_activations.py:8-49) and the
// fp32 complex-multiply RoPE (positional_embeddings.py:27-51). All are
// HBM-bound: one pass, vectorization where layout allows, fp32 math.

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <vector>

namespace {

constexpr int kBlock = 256;

__device__ __forceinline__ float act_eval(int mode, float v) {
  switch (mode) {
    case 0: return v;                                   // identity
    case 1: {                                           // exact gelu (erf)
      return 0.5f * v * (1.0f + erff(v * 0.70710678118654752440f));
    }
    case 2: return v > 0.f ? v : 0.f;                   // relu
    default: {                                          // silu
      return v / (1.0f + __expf(-v));
    }
  }
}

__device__ __forceinline__ float act_grad(int mode, float v) {
  switch (mode) {
    case 0: return 1.f;
    case 1: {
      // d/dv gelu = Phi(v) + v*phi(v)
      const float kInvSqrt2 = 0.70710678118654752440f;
      const float kInvSqrt2Pi = 0.39894228040143267794f;
      float cdf = 0.5f * (1.0f + erff(v * kInvSqrt2));
      float pdf = kInvSqrt2Pi * __expf(-0.5f * v * v);
      return cdf + v * pdf;
    }
    case 2: return v > 0.f ? 1.f : 0.f;
    default: {
      float s = 1.0f / (1.0f + __expf(-v));
      return s * (1.0f + v * (1.0f - s));
    }
  }
}

// vectorized bf16 GLU (F % 8 == 0): 16-B loads of both halves
__global__ void glu_fwd_kernel_bf16v(const __hip_bfloat16* __restrict__ x,
                                     __hip_bfloat16* __restrict__ y,
                                     long rows, int F, int mode) {
  const int FV = F / 8;
  const long total = rows * (long)FV;
  for (long idx = (long)blockIdx.x * kBlock + threadIdx.x; idx < total;
       idx += (long)gridDim.x * kBlock) {
    const long r = idx / FV;
    const int j = idx % FV;
    const uint4* xr = reinterpret_cast<const uint4*>(x + r * (long)(2 * F));
    Bf16x8 v1, v2, vo;
    v1.u = xr[j];
    v2.u = xr[FV + j];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      vo.h[k] = __float2bfloat16(
          __bfloat162float(v1.h[k]) *
          act_eval(mode, __bfloat162float(v2.h[k])));
    }
    reinterpret_cast<uint4*>(y + r * (long)F)[j] = vo.u;
  }
}

__global__ void glu_bwd_kernel_bf16v(const __hip_bfloat16* __restrict__ dy,
                                     const __hip_bfloat16* __restrict__ x,
                                     __hip_bfloat16* __restrict__ dx,
                                     long rows, int F, int mode) {
  const int FV = F / 8;
  const long total = rows * (long)FV;
  for (long idx = (long)blockIdx.x * kBlock + threadIdx.x; idx < total;
       idx += (long)gridDim.x * kBlock) {
    const long r = idx / FV;
    const int j = idx % FV;
    const uint4* xr = reinterpret_cast<const uint4*>(x + r * (long)(2 * F));
    uint4* dxr = reinterpret_cast<uint4*>(dx + r * (long)(2 * F));
    Bf16x8 vg, v1, v2, d1, d2;
    vg.u = reinterpret_cast<const uint4*>(dy + r * (long)F)[j];
    v1.u = xr[j];
    v2.u = xr[FV + j];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float g = __bfloat162float(vg.h[k]);
      float x1 = __bfloat162float(v1.h[k]);
      float x2 = __bfloat162float(v2.h[k]);
      d1.h[k] = __float2bfloat16(g * act_eval(mode, x2));
      d2.h[k] = __float2bfloat16(g * x1 * act_grad(mode, x2));
    }
    dxr[j] = d1.u;
    dxr[FV + j] = d2.u;
  }
}

// y[i, j] = x1[i, j] * act(x2[i, j]),  x = [x1 | x2] along last dim
template <typename T>
__global__ void glu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               long rows, int F, int mode) {
  const long total = rows * (long)F;
  for (long idx = (long)blockIdx.x * kBlock + threadIdx.x; idx < total;
       idx += (long)gridDim.x * kBlock) {
    const long r = idx / F;
    const int j = idx % F;
    const T* xr = x + r * (long)(2 * F);
    float x1 = DTypeTraits<T>::to_float(xr[j]);
    float x2 = DTypeTraits<T>::to_float(xr[F + j]);
    y[idx] = DTypeTraits<T>::from_float(x1 * act_eval(mode, x2));
  }
}

template <typename T>
__global__ void glu_bwd_kernel(const T* __restrict__ dy,
                               const T* __restrict__ x, T* __restrict__ dx,
                               long rows, int F, int mode) {
  const long total = rows * (long)F;
  for (long idx = (long)blockIdx.x * kBlock + threadIdx.x; idx < total;
       idx += (long)gridDim.x * kBlock) {
    const long r = idx / F;
    const int j = idx % F;
    const T* xr = x + r * (long)(2 * F);
    T* dxr = dx + r * (long)(2 * F);
    float g = DTypeTraits<T>::to_float(dy[idx]);
    float x1 = DTypeTraits<T>::to_float(xr[j]);
    float x2 = DTypeTraits<T>::to_float(xr[F + j]);
    float a = act_eval(mode, x2);
    dxr[j] = DTypeTraits<T>::from_float(g * a);
    dxr[F + j] = DTypeTraits<T>::from_float(g * x1 * act_grad(mode, x2));
  }
}

// RoPE on interleaved pairs: x (s, b, n, h), cos/sin (s, h/2) fp32.
// One thread per pair. DIR=+1 fwd, -1 bwd (rotation by -theta).
template <typename T, int DIR>
__global__ void rope_kernel(const T* __restrict__ x,
                            const float* __restrict__ cosT,
                            const float* __restrict__ sinT,
                            T* __restrict__ y, long sbn, int b, int n,
                            int h_half, long x_ss, long x_bs, long x_ns,
                            long y_ss, long y_bs, long y_ns) {
  // rows = s*b*n, each row has h = 2*h_half elements. x and y may be strided
  // VIEWS (e.g. the q/k head slices of a fused QKV projection or of its
  // gradient buffer); head_dim is dense.
  for (long idx = (long)blockIdx.x * kBlock + threadIdx.x;
       idx < sbn * (long)h_half; idx += (long)gridDim.x * kBlock) {
    const long row = idx / h_half;
    const int j = idx % h_half;
    const int n_idx = row % n;
    const long sb = row / n;
    const int b_idx = sb % b;
    const int s_idx = sb / b;
    const long x_base =
        (long)s_idx * x_ss + (long)b_idx * x_bs + (long)n_idx * x_ns + 2 * j;
    const long y_base =
        (long)s_idx * y_ss + (long)b_idx * y_bs + (long)n_idx * y_ns + 2 * j;
    float c = cosT[(long)s_idx * h_half + j];
    float sn = sinT[(long)s_idx * h_half + j] * DIR;
    float x1 = DTypeTraits<T>::to_float(x[x_base]);
    float x2 = DTypeTraits<T>::to_float(x[x_base + 1]);
    y[y_base] = DTypeTraits<T>::from_float(x1 * c - x2 * sn);
    y[y_base + 1] = DTypeTraits<T>::from_float(x2 * c + x1 * sn);
  }
}

// out = residual + dropout(x [+ bias]); mask stored as uint8
template <typename T, bool HAS_BIAS>
__global__ void bias_dropout_add_kernel(const T* __restrict__ x,
                                        const T* __restrict__ bias,
                                        const T* __restrict__ residual,
                                        T* __restrict__ y,
                                        uint8_t* __restrict__ mask_out,
                                        long total, int H, float p,
                                        float rescale,
                                        unsigned long long seed,
                                        unsigned long long offset) {
  for (long idx = ((long)blockIdx.x * kBlock + threadIdx.x) * 4; idx < total;
       idx += (long)gridDim.x * kBlock * 4) {
    uint4 r = philox10(seed, offset, (unsigned)(idx / 4));
    unsigned rv[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      long i = idx + k;
      if (i >= total) break;
      bool keep = uint_to_uniform(rv[k]) > p;
      float v = DTypeTraits<T>::to_float(x[i]);
      if (HAS_BIAS) v += DTypeTraits<T>::to_float(bias[i % H]);
      float res = DTypeTraits<T>::to_float(residual[i]);
      y[i] = DTypeTraits<T>::from_float(
          (keep ? v * rescale : 0.f) + res);
      mask_out[i] = keep ? 1 : 0;
    }
  }
}

template <typename T>
__global__ void dropout_bwd_kernel(const T* __restrict__ dy,
                                   const uint8_t* __restrict__ mask,
                                   T* __restrict__ dx, long total,
                                   float rescale) {
  for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < total;
       i += (long)gridDim.x * kBlock) {
    float g = DTypeTraits<T>::to_float(dy[i]);
    dx[i] = DTypeTraits<T>::from_float(mask[i] ? g * rescale : 0.f);
  }
}

inline int grid_for(long total) {
  long g = (total + kBlock - 1) / kBlock;
  // ≫256 workgroups to fill 8 XCDs x 32 CUs
  return (int)std::min<long>(g, 32768);
}

#define DISPATCH_DTYPE(TENSOR, FN)                                           \
  do {                                                                       \
    if ((TENSOR).scalar_type() == torch::kBFloat16) {                        \
      FN(__hip_bfloat16);                                                    \
    } else if ((TENSOR).scalar_type() == torch::kFloat16) {                  \
      FN(__half);                                                            \
    } else {                                                                 \
      FN(float);                                                             \
    }                                                                        \
  } while (0)

}  // namespace

torch::Tensor glu_fwd(torch::Tensor x, int64_t mode) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  long rows = x.size(0);
  int F = x.size(1) / 2;
  auto y = torch::empty({rows, F}, x.options());
  auto stream = c10::hip::getCurrentHIPStream();
  long total = rows * (long)F;
  if (x.scalar_type() == torch::kBFloat16 && F % 8 == 0) {
    hipLaunchKernelGGL(glu_fwd_kernel_bf16v, dim3(grid_for(total / 8)),
                       dim3(kBlock), 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(),
                       (__hip_bfloat16*)y.data_ptr(), rows, F, (int)mode);
    return y;
  }
#define LAUNCH_GLU_F(T)                                                      \
  hipLaunchKernelGGL((glu_fwd_kernel<T>), dim3(grid_for(total)),             \
                     dim3(kBlock), 0, stream, (const T*)x.data_ptr(),        \
                     (T*)y.data_ptr(), rows, F, (int)mode)
  DISPATCH_DTYPE(x, LAUNCH_GLU_F);
  return y;
}

torch::Tensor glu_bwd(torch::Tensor dy, torch::Tensor x, int64_t mode) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  long rows = x.size(0);
  int F = x.size(1) / 2;
  auto dx = torch::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  long total = rows * (long)F;
  if (x.scalar_type() == torch::kBFloat16 && F % 8 == 0) {
    hipLaunchKernelGGL(glu_bwd_kernel_bf16v, dim3(grid_for(total / 8)),
                       dim3(kBlock), 0, stream,
                       (const __hip_bfloat16*)dy.data_ptr(),
                       (const __hip_bfloat16*)x.data_ptr(),
                       (__hip_bfloat16*)dx.data_ptr(), rows, F, (int)mode);
    return dx;
  }
#define LAUNCH_GLU_B(T)                                                      \
  hipLaunchKernelGGL((glu_bwd_kernel<T>), dim3(grid_for(total)),             \
                     dim3(kBlock), 0, stream, (const T*)dy.data_ptr(),       \
                     (const T*)x.data_ptr(), (T*)dx.data_ptr(), rows, F,     \
                     (int)mode)
  DISPATCH_DTYPE(x, LAUNCH_GLU_B);
  return dx;
}

static torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cosT,
                                torch::Tensor sinT, int dir,
                                c10::optional<torch::Tensor> out = {}) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.stride(3) == 1,
              "rope: need 4-D x with dense head_dim");
  TORCH_CHECK(cosT.scalar_type() == torch::kFloat32);
  int s = x.size(0), b = x.size(1), n = x.size(2), h = x.size(3);
  TORCH_CHECK(cosT.size(0) >= s && cosT.size(1) == h / 2,
              "rope table too small");
  torch::Tensor y;
  if (out.has_value()) {
    y = *out;
    TORCH_CHECK(y.sizes() == x.sizes() && y.stride(3) == 1 &&
                    y.scalar_type() == x.scalar_type(),
                "rope: bad out tensor");
  } else {
    y = torch::empty({s, b, n, h}, x.options());
  }
  long sbn = (long)s * b * n;
  long total = sbn * (h / 2);
  auto stream = c10::hip::getCurrentHIPStream();
  auto cosc = cosT.contiguous();
  auto sinc = sinT.contiguous();
#define LAUNCH_ROPE_F(T)                                                     \
  hipLaunchKernelGGL((rope_kernel<T, 1>), dim3(grid_for(total)),             \
                     dim3(kBlock), 0, stream, (const T*)x.data_ptr(),        \
                     cosc.data_ptr<float>(), sinc.data_ptr<float>(),         \
                     (T*)y.data_ptr(), sbn, b, n, h / 2, x.stride(0),        \
                     x.stride(1), x.stride(2), y.stride(0), y.stride(1),     \
                     y.stride(2))
#define LAUNCH_ROPE_B(T)                                                     \
  hipLaunchKernelGGL((rope_kernel<T, -1>), dim3(grid_for(total)),            \
                     dim3(kBlock), 0, stream, (const T*)x.data_ptr(),        \
                     cosc.data_ptr<float>(), sinc.data_ptr<float>(),         \
                     (T*)y.data_ptr(), sbn, b, n, h / 2, x.stride(0),        \
                     x.stride(1), x.stride(2), y.stride(0), y.stride(1),     \
                     y.stride(2))
  if (dir > 0) {
    DISPATCH_DTYPE(x, LAUNCH_ROPE_F);
  } else {
    DISPATCH_DTYPE(x, LAUNCH_ROPE_B);
  }
  return y;
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cosT,
                       torch::Tensor sinT) {
  return rope_apply(x, cosT, sinT, +1);
}

torch::Tensor rope_bwd(torch::Tensor dy, torch::Tensor cosT,
                       torch::Tensor sinT) {
  return rope_apply(dy, cosT, sinT, -1);
}

// rotate dy and write the result into `out` (typically a strided slice of a
// fused-QKV gradient buffer) — lets the backward of the QKV split assemble
// d_mixed in one pass with no zeros+copy+add chain.
torch::Tensor rope_bwd_into(torch::Tensor dy, torch::Tensor cosT,
                            torch::Tensor sinT, torch::Tensor out) {
  return rope_apply(dy, cosT, sinT, -1, out);
}

std::vector<torch::Tensor> bias_dropout_add_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> bias,
    torch::Tensor residual, double p, int64_t seed, int64_t offset) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && residual.is_contiguous());
  auto y = torch::empty_like(x);
  auto mask = torch::empty(x.sizes(), x.options().dtype(torch::kUInt8));
  long total = x.numel();
  int H = x.size(-1);
  float rescale = 1.0f / (1.0f - (float)p);
  auto stream = c10::hip::getCurrentHIPStream();
  long vec_total = (total + 3) / 4;
#define LAUNCH_BDA(T)                                                        \
  do {                                                                       \
    if (bias.has_value()) {                                                  \
      hipLaunchKernelGGL((bias_dropout_add_kernel<T, true>),                 \
                         dim3(grid_for(vec_total)), dim3(kBlock), 0, stream, \
                         (const T*)x.data_ptr(),                             \
                         (const T*)bias->data_ptr(),                         \
                         (const T*)residual.data_ptr(), (T*)y.data_ptr(),    \
                         mask.data_ptr<uint8_t>(), total, H, (float)p,       \
                         rescale, (unsigned long long)seed,                  \
                         (unsigned long long)offset);                        \
    } else {                                                                 \
      hipLaunchKernelGGL((bias_dropout_add_kernel<T, false>),                \
                         dim3(grid_for(vec_total)), dim3(kBlock), 0, stream, \
                         (const T*)x.data_ptr(), nullptr,                    \
                         (const T*)residual.data_ptr(), (T*)y.data_ptr(),    \
                         mask.data_ptr<uint8_t>(), total, H, (float)p,       \
                         rescale, (unsigned long long)seed,                  \
                         (unsigned long long)offset);                        \
    }                                                                        \
  } while (0)
  DISPATCH_DTYPE(x, LAUNCH_BDA);
  return {y, mask};
}

torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  auto dx = torch::empty_like(dy);
  long total = dy.numel();
  float rescale = 1.0f / (1.0f - (float)p);
  auto stream = c10::hip::getCurrentHIPStream();
#define LAUNCH_DB(T)                                                         \
  hipLaunchKernelGGL((dropout_bwd_kernel<T>), dim3(grid_for(total)),         \
                     dim3(kBlock), 0, stream, (const T*)dy.data_ptr(),       \
                     mask.data_ptr<uint8_t>(), (T*)dx.data_ptr(), total,     \
                     rescale)
  DISPATCH_DTYPE(dy, LAUNCH_DB);
  return dx;
}
