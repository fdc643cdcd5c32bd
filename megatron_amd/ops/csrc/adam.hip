// Fused Adam(W) + grad-utility kernels (replaces apex FusedAdam / amp_C).
//
// The optimizer state lives in fp32 (master params, m, v). One elementwise
// kernel per tensor, launched back-to-back on the current stream (launches
// are ~3 µs and overlap; the op itself is HBM-bound). An optional model-param
// output writes the bf16/fp16 copy in the same pass, fusing the
// master->model copy of reference optimizer.py:435.

#include "common.h"

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <vector>

namespace {

constexpr int kBlock = 256;

template <typename OutT>
__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            OutT* __restrict__ model_out, long n, float lr,
                            float beta1, float beta2, float eps, float wd,
                            float bc1, float bc2_sqrt, int adam_w,
                            float grad_scale) {
  // grad_scale folds the grad-clip coefficient (and any unscale factor)
  // into this pass, avoiding a separate whole-buffer multiply
  for (long i = (long)blockIdx.x * kBlock + threadIdx.x; i < n;
       i += (long)gridDim.x * kBlock) {
    float grad = g[i] * grad_scale;
    float param = p[i];
    if (!adam_w && wd != 0.f) grad += wd * param;
    float m_new = beta1 * m[i] + (1.f - beta1) * grad;
    float v_new = beta2 * v[i] + (1.f - beta2) * grad * grad;
    m[i] = m_new;
    v[i] = v_new;
    float update = (m_new / bc1) / (sqrtf(v_new) / bc2_sqrt + eps);
    if (adam_w && wd != 0.f) param -= lr * wd * param;
    param -= lr * update;
    p[i] = param;
    if (model_out != nullptr) {
      model_out[i] = DTypeTraits<OutT>::from_float(param);
    }
  }
}

inline int grid_for(long total) {
  long g = (total + kBlock - 1) / kBlock;
  return (int)std::min<long>(g, 16384);
}

}  // namespace

void fused_adam(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> exp_avgs,
                std::vector<torch::Tensor> exp_avg_sqs, double lr,
                double beta1, double beta2, double eps, double wd,
                int64_t step, int64_t adam_w_mode, double grad_scale) {
  auto stream = c10::hip::getCurrentHIPStream();
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2_sqrt = sqrtf(1.f - powf((float)beta2, (float)step));
  for (size_t i = 0; i < params.size(); ++i) {
    TORCH_CHECK(params[i].scalar_type() == torch::kFloat32,
                "fused_adam expects fp32 master params");
    long n = params[i].numel();
    hipLaunchKernelGGL(adam_kernel<float>, dim3(grid_for(n)), dim3(kBlock), 0,
                       stream, params[i].data_ptr<float>(),
                       grads[i].data_ptr<float>(),
                       exp_avgs[i].data_ptr<float>(),
                       exp_avg_sqs[i].data_ptr<float>(), nullptr, n,
                       (float)lr, (float)beta1, (float)beta2, (float)eps,
                       (float)wd, bc1, bc2_sqrt, (int)adam_w_mode,
                       (float)grad_scale);
  }
}

// Variant fusing the fp32-master -> bf16/fp16 model-param copy
// (reference optimizer.py:435 _copy_main_params_to_model_params) into the
// same memory pass.
void fused_adam_with_model_copy(std::vector<torch::Tensor> params,
                                std::vector<torch::Tensor> grads,
                                std::vector<torch::Tensor> exp_avgs,
                                std::vector<torch::Tensor> exp_avg_sqs,
                                std::vector<torch::Tensor> model_params,
                                double lr, double beta1, double beta2,
                                double eps, double wd, int64_t step,
                                int64_t adam_w_mode, double grad_scale) {
  auto stream = c10::hip::getCurrentHIPStream();
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2_sqrt = sqrtf(1.f - powf((float)beta2, (float)step));
  for (size_t i = 0; i < params.size(); ++i) {
    TORCH_CHECK(params[i].scalar_type() == torch::kFloat32);
    long n = params[i].numel();
    TORCH_CHECK(model_params[i].numel() == n);
    if (model_params[i].scalar_type() == torch::kBFloat16) {
      hipLaunchKernelGGL(adam_kernel<__hip_bfloat16>, dim3(grid_for(n)),
                         dim3(kBlock), 0, stream,
                         params[i].data_ptr<float>(),
                         grads[i].data_ptr<float>(),
                         exp_avgs[i].data_ptr<float>(),
                         exp_avg_sqs[i].data_ptr<float>(),
                         (__hip_bfloat16*)model_params[i].data_ptr(), n,
                         (float)lr, (float)beta1, (float)beta2, (float)eps,
                         (float)wd, bc1, bc2_sqrt, (int)adam_w_mode,
                         (float)grad_scale);
    } else if (model_params[i].scalar_type() == torch::kFloat16) {
      hipLaunchKernelGGL(adam_kernel<__half>, dim3(grid_for(n)),
                         dim3(kBlock), 0, stream,
                         params[i].data_ptr<float>(),
                         grads[i].data_ptr<float>(),
                         exp_avgs[i].data_ptr<float>(),
                         exp_avg_sqs[i].data_ptr<float>(),
                         (__half*)model_params[i].data_ptr(), n, (float)lr,
                         (float)beta1, (float)beta2, (float)eps, (float)wd,
                         bc1, bc2_sqrt, (int)adam_w_mode,
                         (float)grad_scale);
    } else {
      TORCH_CHECK(false, "model params must be bf16/fp16");
    }
  }
}
