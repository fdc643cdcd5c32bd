// Fused RMSNorm / LayerNorm forward + backward for MI355X (gfx950).
//
// Replaces the reference's CUDA layer_norm_cuda_kernel.cu (818 LoC) and its
// UNFUSED Python RMSNorm (fused_layer_norm.py:125-139 — on Llama this sits on
// the critical path twice per layer).
//
// Design (HBM-bound op, target ≈6.3 TB/s streaming):
//  - one 256-thread block per row (H = hidden size, fp32 statistics);
//  - bf16/half traffic vectorized 16 B/lane (8 elems) when H % 8 == 0;
//  - backward: grid-strided blocks accumulate dweight/dbias in registers
//    (H/256 fp32 accumulators per thread) and atomically fold into the
//    global fp32 dweight once per block — O(grid) atomics instead of
//    O(rows).

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <vector>

namespace {

template <typename T, bool RMS, int BLOCK>
__global__ void norm_fwd_kernel(const T* __restrict__ x,
                                const T* __restrict__ weight,
                                const T* __restrict__ bias,
                                T* __restrict__ y,
                                float* __restrict__ mean_out,
                                float* __restrict__ inv_out, int H,
                                float eps) {
  __shared__ float lds[BLOCK / WAVE_SIZE];
  const long row = blockIdx.x;
  const T* xrow = x + row * (long)H;
  T* yrow = y + row * (long)H;

  float sum = 0.f, sumsq = 0.f;
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float v = DTypeTraits<T>::to_float(xrow[i]);
    sum += v;
    sumsq += v * v;
  }
  float mean = 0.f;
  if (!RMS) {
    mean = block_reduce_sum<BLOCK>(sum, lds) / H;
    __syncthreads();
  }
  float var = block_reduce_sum<BLOCK>(sumsq, lds) / H;
  if (!RMS) var -= mean * mean;
  float inv = rsqrtf(var + eps);

  if (threadIdx.x == 0) {
    if (!RMS) mean_out[row] = mean;
    inv_out[row] = inv;
  }
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float v = DTypeTraits<T>::to_float(xrow[i]);
    float w = DTypeTraits<T>::to_float(weight[i]);
    float o = (v - mean) * inv * w;
    if (!RMS) o += DTypeTraits<T>::to_float(bias[i]);
    yrow[i] = DTypeTraits<T>::from_float(o);
  }
}

// Vectorized bf16 x8 variant (H % 8 == 0).
template <bool RMS, int BLOCK>
__global__ void norm_fwd_kernel_bf16v(const __hip_bfloat16* __restrict__ x,
                                      const __hip_bfloat16* __restrict__ weight,
                                      const __hip_bfloat16* __restrict__ bias,
                                      __hip_bfloat16* __restrict__ y,
                                      float* __restrict__ mean_out,
                                      float* __restrict__ inv_out, int H,
                                      float eps) {
  __shared__ float lds[BLOCK / WAVE_SIZE];
  const long row = blockIdx.x;
  const uint4* xrow = reinterpret_cast<const uint4*>(x + row * (long)H);
  uint4* yrow = reinterpret_cast<uint4*>(y + row * (long)H);
  const uint4* wv = reinterpret_cast<const uint4*>(weight);
  const uint4* bv = reinterpret_cast<const uint4*>(bias);
  const int HV = H / 8;

  float sum = 0.f, sumsq = 0.f;
  for (int i = threadIdx.x; i < HV; i += BLOCK) {
    Bf16x8 vx; vx.u = xrow[i];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float v = __bfloat162float(vx.h[k]);
      sum += v;
      sumsq += v * v;
    }
  }
  float mean = 0.f;
  if (!RMS) {
    mean = block_reduce_sum<BLOCK>(sum, lds) / H;
    __syncthreads();
  }
  float var = block_reduce_sum<BLOCK>(sumsq, lds) / H;
  if (!RMS) var -= mean * mean;
  float inv = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    if (!RMS) mean_out[row] = mean;
    inv_out[row] = inv;
  }
  for (int i = threadIdx.x; i < HV; i += BLOCK) {
    Bf16x8 vx; vx.u = xrow[i];
    Bf16x8 vw; vw.u = wv[i];
    Bf16x8 vo;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float v = __bfloat162float(vx.h[k]);
      float o = (v - mean) * inv * __bfloat162float(vw.h[k]);
      if (!RMS) o += 0.f;  // bias handled below for LN
      vo.h[k] = __float2bfloat16(o);
    }
    if (!RMS) {
      Bf16x8 vb; vb.u = bv[i];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        vo.h[k] = __float2bfloat16(__bfloat162float(vo.h[k]) +
                                   __bfloat162float(vb.h[k]));
      }
    }
    yrow[i] = vo.u;
  }
}

// Vectorized bf16 RMSNorm backward (H % 8 == 0, H <= BLOCK*8*MAX_CV):
// 16-B loads/stores, dweight accumulated in registers per block and folded
// with one atomicAdd per element at the end.
template <bool RMS, int BLOCK, int MAX_CV>
__global__ void norm_bwd_kernel_bf16v(const __hip_bfloat16* __restrict__ dy,
                                      const __hip_bfloat16* __restrict__ x,
                                      const __hip_bfloat16* __restrict__ weight,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ inv,
                                      const __hip_bfloat16* __restrict__ dres,
                                      __hip_bfloat16* __restrict__ dx,
                                      float* __restrict__ dweight,
                                      float* __restrict__ dbias, long rows,
                                      int H) {
  __shared__ float lds[BLOCK / WAVE_SIZE];
  const int HV = H / 8;
  const int CV = (HV + BLOCK - 1) / BLOCK;  // <= MAX_CV vec-chunks per thread

  float acc_dw[MAX_CV][8];
  float acc_db[MAX_CV][8];
#pragma unroll
  for (int c = 0; c < MAX_CV; ++c) {
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      acc_dw[c][k] = 0.f;
      acc_db[c][k] = 0.f;
    }
  }

  const uint4* wv = reinterpret_cast<const uint4*>(weight);

  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const uint4* dyr = reinterpret_cast<const uint4*>(dy + row * (long)H);
    const uint4* xr = reinterpret_cast<const uint4*>(x + row * (long)H);
    uint4* dxr = reinterpret_cast<uint4*>(dx + row * (long)H);
    const float m = RMS ? 0.f : mean[row];
    const float r = inv[row];

    float dot = 0.f, gsum = 0.f;
    for (int i = threadIdx.x; i < HV; i += BLOCK) {
      Bf16x8 vg, vx, vw;
      vg.u = dyr[i];
      vx.u = xr[i];
      vw.u = wv[i];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = __bfloat162float(vg.h[k]);
        float xv = __bfloat162float(vx.h[k]);
        float w = __bfloat162float(vw.h[k]);
        float xhat = (xv - m) * r;
        float gw = g * w;
        dot += gw * xhat;
        if (!RMS) gsum += gw;
      }
    }
    dot = block_reduce_sum<BLOCK>(dot, lds);
    if (!RMS) {
      __syncthreads();
      gsum = block_reduce_sum<BLOCK>(gsum, lds);
    }
    const float inv_H = 1.0f / H;

    const uint4* drr =
        dres ? reinterpret_cast<const uint4*>(dres + row * (long)H) : nullptr;
    for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
      Bf16x8 vg, vx, vw, vo, vr;
      vg.u = dyr[i];
      vx.u = xr[i];
      vw.u = wv[i];
      if (drr) vr.u = drr[i];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = __bfloat162float(vg.h[k]);
        float xv = __bfloat162float(vx.h[k]);
        float w = __bfloat162float(vw.h[k]);
        float xhat = (xv - m) * r;
        float gw = g * w;
        float d;
        if (RMS) {
          d = (gw - xhat * dot * inv_H) * r;
        } else {
          d = (gw - gsum * inv_H - xhat * dot * inv_H) * r;
        }
        // fused residual-grad accumulate: dx_total = norm_dx + dres
        if (drr) d += __bfloat162float(vr.h[k]);
        vo.h[k] = __float2bfloat16(d);
        acc_dw[c][k] += g * xhat;
        if (!RMS) acc_db[c][k] += g;
      }
      dxr[i] = vo.u;
    }
    __syncthreads();
  }

  // per-block partial rows (no atomics: fp32 atomic contention on H hot
  // addresses dominated the kernel) — reduced by norm_bwd_reduce_kernel
  for (int i = threadIdx.x, c = 0; i < HV; i += BLOCK, ++c) {
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      dweight[(long)blockIdx.x * H + i * 8 + k] = acc_dw[c][k];
      if (!RMS) dbias[(long)blockIdx.x * H + i * 8 + k] = acc_db[c][k];
    }
  }
}

// column-sum of the [grid][H] partial buffers. blockIdx.y slices the row
// range so the read is parallel across CUs; a second launch with nblocks =
// n_slices folds the slice partials — a fixed-order (deterministic) tree.
__global__ void norm_bwd_reduce_kernel(const float* __restrict__ partial,
                                       float* __restrict__ out, int nblocks,
                                       int H, int rows_per_slice) {
  int i = blockIdx.x * 256 + threadIdx.x;
  if (i >= H) return;
  int b0 = blockIdx.y * rows_per_slice;
  int b1 = min(nblocks, b0 + rows_per_slice);
  float acc = 0.f;
  for (int b = b0; b < b1; ++b) acc += partial[(long)b * H + i];
  out[(long)blockIdx.y * H + i] = acc;
}

// helper: deterministic two-level column reduce of [nblocks][H] -> out[H]
static void reduce_partials(const float* part, float* out, int nblocks, int H,
                            torch::Tensor& tmp, hipStream_t stream) {
  const int kSlices = 32;
  if (nblocks <= 64) {
    hipLaunchKernelGGL(norm_bwd_reduce_kernel, dim3((H + 255) / 256, 1),
                       dim3(256), 0, stream, part, out, nblocks, H, nblocks);
    return;
  }
  int rps = (nblocks + kSlices - 1) / kSlices;
  int slices = (nblocks + rps - 1) / rps;
  hipLaunchKernelGGL(norm_bwd_reduce_kernel, dim3((H + 255) / 256, slices),
                     dim3(256), 0, stream, part, tmp.data_ptr<float>(),
                     nblocks, H, rps);
  hipLaunchKernelGGL(norm_bwd_reduce_kernel, dim3((H + 255) / 256, 1),
                     dim3(256), 0, stream, tmp.data_ptr<float>(), out, slices,
                     H, slices);
}

// Backward. MAX_ACC register accumulators per thread for dweight/dbias.
template <typename T, bool RMS, int BLOCK, int MAX_ACC>
__global__ void norm_bwd_kernel(const T* __restrict__ dy,
                                const T* __restrict__ x,
                                const T* __restrict__ weight,
                                const float* __restrict__ mean,
                                const float* __restrict__ inv,
                                const T* __restrict__ dres,
                                T* __restrict__ dx,
                                float* __restrict__ dweight,
                                float* __restrict__ dbias, long rows, int H) {
  __shared__ float lds[BLOCK / WAVE_SIZE];
  float acc_dw[MAX_ACC];
  float acc_db[MAX_ACC];
  const int per_thread = (H + BLOCK - 1) / BLOCK;
#pragma unroll
  for (int k = 0; k < MAX_ACC; ++k) { acc_dw[k] = 0.f; acc_db[k] = 0.f; }

  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * (long)H;
    const T* xr = x + row * (long)H;
    T* dxr = dx + row * (long)H;
    const float m = RMS ? 0.f : mean[row];
    const float r = inv[row];

    // reductions: sum(g*xhat) and (LN only) sum(g)
    float dot = 0.f, gsum = 0.f;
    for (int i = threadIdx.x, k = 0; i < H; i += BLOCK, ++k) {
      float g = DTypeTraits<T>::to_float(dyr[i]);
      float w = DTypeTraits<T>::to_float(weight[i]);
      float xv = DTypeTraits<T>::to_float(xr[i]);
      float xhat = (xv - m) * r;
      float gw = g * w;
      dot += gw * xhat;
      if (!RMS) gsum += gw;
      if (k < MAX_ACC) {
        acc_dw[k] += g * xhat;
        if (!RMS) acc_db[k] += g;
      }
    }
    dot = block_reduce_sum<BLOCK>(dot, lds);
    if (!RMS) {
      __syncthreads();
      gsum = block_reduce_sum<BLOCK>(gsum, lds);
    }
    const float inv_H = 1.0f / H;

    for (int i = threadIdx.x; i < H; i += BLOCK) {
      float g = DTypeTraits<T>::to_float(dyr[i]);
      float w = DTypeTraits<T>::to_float(weight[i]);
      float xv = DTypeTraits<T>::to_float(xr[i]);
      float xhat = (xv - m) * r;
      float gw = g * w;
      float d;
      if (RMS) {
        d = (gw - xhat * dot * inv_H) * r;
      } else {
        d = (gw - gsum * inv_H - xhat * dot * inv_H) * r;
      }
      if (dres) d += DTypeTraits<T>::to_float(dres[row * (long)H + i]);
      dxr[i] = DTypeTraits<T>::from_float(d);
    }
    __syncthreads();
  }

  // fold register accumulators into global dweight/dbias
  for (int i = threadIdx.x, k = 0; i < H && k < MAX_ACC; i += BLOCK, ++k) {
    atomicAdd(&dweight[i], acc_dw[k]);
    if (!RMS) atomicAdd(&dbias[i], acc_db[k]);
  }
}

template <typename T>
struct TorchDtype;
template <>
struct TorchDtype<float> {
  static constexpr auto value = torch::kFloat32;
};
template <>
struct TorchDtype<__hip_bfloat16> {
  static constexpr auto value = torch::kBFloat16;
};
template <>
struct TorchDtype<__half> {
  static constexpr auto value = torch::kFloat16;
};

constexpr int kBlock = 256;
constexpr int kMaxAcc = 32;  // supports H up to 8192 in register accumulation

template <typename T, bool RMS>
void norm_fwd_launch(const torch::Tensor& x, const torch::Tensor& w,
                     const torch::Tensor* b, torch::Tensor& y,
                     torch::Tensor& mean, torch::Tensor& inv, double eps) {
  long rows = x.size(0);
  int H = x.size(1);
  auto stream = c10::hip::getCurrentHIPStream();
  if constexpr (std::is_same_v<T, __hip_bfloat16>) {
    if (H % 8 == 0) {
      hipLaunchKernelGGL((norm_fwd_kernel_bf16v<RMS, kBlock>), dim3(rows),
                         dim3(kBlock), 0, stream,
                         (const __hip_bfloat16*)x.data_ptr(),
                         (const __hip_bfloat16*)w.data_ptr(),
                         RMS ? nullptr : (const __hip_bfloat16*)b->data_ptr(),
                         (__hip_bfloat16*)y.data_ptr(),
                         RMS ? nullptr : mean.data_ptr<float>(),
                         inv.data_ptr<float>(), H, (float)eps);
      return;
    }
  }
  hipLaunchKernelGGL((norm_fwd_kernel<T, RMS, kBlock>), dim3(rows),
                     dim3(kBlock), 0, stream, (const T*)x.data_ptr(),
                     (const T*)w.data_ptr(),
                     RMS ? nullptr : (const T*)b->data_ptr(),
                     (T*)y.data_ptr(),
                     RMS ? nullptr : mean.data_ptr<float>(),
                     inv.data_ptr<float>(), H, (float)eps);
}

template <typename T, bool RMS>
void norm_bwd_launch(const torch::Tensor& dy, const torch::Tensor& x,
                     const torch::Tensor& w, const torch::Tensor* mean,
                     const torch::Tensor& inv, const torch::Tensor* dres,
                     torch::Tensor& dx, torch::Tensor& dw,
                     torch::Tensor* db) {
  long rows = x.size(0);
  int H = x.size(1);
  TORCH_CHECK(H <= kBlock * kMaxAcc,
              "hidden size too large for norm backward (max ",
              kBlock * kMaxAcc, ")");
  int grid = (int)std::min<long>(rows, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  if constexpr (std::is_same_v<T, __hip_bfloat16>) {
    if (H % 8 == 0 && H <= kBlock * 8 * 4) {
      static const int grid_env = []() {
        const char* e = getenv("MEGATRON_AMD_NORM_BWD_GRID");
        return e ? atoi(e) : 608;
      }();
      grid = (int)std::min<long>(rows, grid_env);
      auto opts = dy.options().dtype(torch::kFloat32);
      auto dw_part = torch::empty({grid, H}, opts);
      auto red_tmp = torch::empty({32, H}, opts);
      torch::Tensor db_part;
      if (!RMS) db_part = torch::empty({grid, H}, opts);
      hipLaunchKernelGGL((norm_bwd_kernel_bf16v<RMS, kBlock, 4>), dim3(grid),
                         dim3(kBlock), 0, stream,
                         (const __hip_bfloat16*)dy.data_ptr(),
                         (const __hip_bfloat16*)x.data_ptr(),
                         (const __hip_bfloat16*)w.data_ptr(),
                         RMS ? nullptr : mean->data_ptr<float>(),
                         inv.data_ptr<float>(),
                         dres ? (const __hip_bfloat16*)dres->data_ptr()
                              : nullptr,
                         (__hip_bfloat16*)dx.data_ptr(),
                         dw_part.data_ptr<float>(),
                         RMS ? nullptr : db_part.data_ptr<float>(), rows, H);
      reduce_partials(dw_part.data_ptr<float>(), dw.data_ptr<float>(), grid,
                      H, red_tmp, stream);
      if (!RMS) {
        reduce_partials(db_part.data_ptr<float>(), db->data_ptr<float>(),
                        grid, H, red_tmp, stream);
      }
      return;
    }
  }
  hipLaunchKernelGGL((norm_bwd_kernel<T, RMS, kBlock, kMaxAcc>), dim3(grid),
                     dim3(kBlock), 0, stream, (const T*)dy.data_ptr(),
                     (const T*)x.data_ptr(), (const T*)w.data_ptr(),
                     RMS ? nullptr : mean->data_ptr<float>(),
                     inv.data_ptr<float>(),
                     dres ? (const T*)dres->data_ptr() : nullptr,
                     (T*)dx.data_ptr(),
                     dw.data_ptr<float>(),
                     RMS ? nullptr : db->data_ptr<float>(), rows, H);
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor weight,
                                       double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  long rows = x.size(0);
  auto y = torch::empty_like(x);
  auto inv = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto mean = torch::Tensor();
  if (x.scalar_type() == torch::kBFloat16) {
    norm_fwd_launch<__hip_bfloat16, true>(x, weight, nullptr, y, mean, inv, eps);
  } else if (x.scalar_type() == torch::kFloat16) {
    norm_fwd_launch<__half, true>(x, weight, nullptr, y, mean, inv, eps);
  } else {
    norm_fwd_launch<float, true>(x, weight, nullptr, y, mean, inv, eps);
  }
  return {y, inv};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor weight, torch::Tensor inv,
                                       c10::optional<torch::Tensor> dres) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const torch::Tensor* dr = nullptr;
  torch::Tensor dres_c;
  if (dres.has_value()) {
    dres_c = dres->contiguous();
    TORCH_CHECK(dres_c.scalar_type() == x.scalar_type() &&
                dres_c.numel() == x.numel());
    dr = &dres_c;
  }
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({x.size(1)}, x.options().dtype(torch::kFloat32));
  if (x.scalar_type() == torch::kBFloat16) {
    norm_bwd_launch<__hip_bfloat16, true>(dy, x, weight, nullptr, inv, dr, dx,
                                          dw32, nullptr);
  } else if (x.scalar_type() == torch::kFloat16) {
    norm_bwd_launch<__half, true>(dy, x, weight, nullptr, inv, dr, dx, dw32,
                                  nullptr);
  } else {
    norm_bwd_launch<float, true>(dy, x, weight, nullptr, inv, dr, dx, dw32,
                                 nullptr);
  }
  return {dx, dw32.to(weight.scalar_type())};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor weight,
                                         torch::Tensor bias, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  long rows = x.size(0);
  auto y = torch::empty_like(x);
  auto inv = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  if (x.scalar_type() == torch::kBFloat16) {
    norm_fwd_launch<__hip_bfloat16, false>(x, weight, &bias, y, mean, inv, eps);
  } else if (x.scalar_type() == torch::kFloat16) {
    norm_fwd_launch<__half, false>(x, weight, &bias, y, mean, inv, eps);
  } else {
    norm_fwd_launch<float, false>(x, weight, &bias, y, mean, inv, eps);
  }
  return {y, mean, inv};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor weight,
                                         torch::Tensor mean, torch::Tensor inv,
                                         c10::optional<torch::Tensor> dres) {
  const torch::Tensor* dr = nullptr;
  torch::Tensor dres_c;
  if (dres.has_value()) {
    dres_c = dres->contiguous();
    TORCH_CHECK(dres_c.scalar_type() == x.scalar_type() &&
                dres_c.numel() == x.numel());
    dr = &dres_c;
  }
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({x.size(1)}, x.options().dtype(torch::kFloat32));
  auto db32 = torch::zeros({x.size(1)}, x.options().dtype(torch::kFloat32));
  if (x.scalar_type() == torch::kBFloat16) {
    norm_bwd_launch<__hip_bfloat16, false>(dy, x, weight, &mean, inv, dr, dx,
                                           dw32, &db32);
  } else if (x.scalar_type() == torch::kFloat16) {
    norm_bwd_launch<__half, false>(dy, x, weight, &mean, inv, dr, dx, dw32,
                                   &db32);
  } else {
    norm_bwd_launch<float, false>(dy, x, weight, &mean, inv, dr, dx, dw32,
                                  &db32);
  }
  return {dx, dw32.to(weight.scalar_type()), db32.to(weight.scalar_type())};
}
