// Weight-gradient GEMM with fp32 accumulation directly into main_grad.
//
// Replaces apex fused_weight_gradient_mlp_cuda (reference
// fused_weight_gradient_dense.cu:128-151, consumed at layers.py:298-304):
//   main_grad(fp32)[out, in] += grad_output[K, out]^T @ input[K, in]
// One hipBLASLt/rocBLAS GEMM (HIPBLAS_COMPUTE_32F, beta = 1).
//
// Row-major [out, in] viewed column-major is [in, out], so in col-major:
//   C[in, out] += A(=input^T viewed cm: [in, K], op N) * B(=grad_out viewed
//   cm: [out, K], op T)

#include <hip/hip_runtime.h>
#include <hipblas/hipblas.h>

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace {

hipblasHandle_t get_handle() {
  static hipblasHandle_t handle = nullptr;
  if (handle == nullptr) {
    hipblasCreate(&handle);
  }
  return handle;
}

}  // namespace

void wgrad_gemm_accum_fp32(torch::Tensor input, torch::Tensor grad_output,
                           torch::Tensor main_grad) {
  TORCH_CHECK(input.is_cuda() && input.dim() == 2 && input.is_contiguous());
  TORCH_CHECK(grad_output.dim() == 2 && grad_output.is_contiguous());
  TORCH_CHECK(main_grad.scalar_type() == torch::kFloat32 &&
              main_grad.is_contiguous());
  long K = input.size(0);
  long in_dim = input.size(1);
  long out_dim = grad_output.size(1);
  TORCH_CHECK(grad_output.size(0) == K);
  TORCH_CHECK(main_grad.size(0) == out_dim && main_grad.size(1) == in_dim);

  hipDataType ab_type;
  if (input.scalar_type() == torch::kBFloat16) {
    ab_type = HIP_R_16BF;
  } else if (input.scalar_type() == torch::kFloat16) {
    ab_type = HIP_R_16F;
  } else {
    ab_type = HIP_R_32F;
  }
  TORCH_CHECK(grad_output.scalar_type() == input.scalar_type());

  float alpha = 1.0f, beta = 1.0f;
  auto handle = get_handle();
  hipblasSetStream(handle, c10::hip::getCurrentHIPStream());
  auto status = hipblasGemmEx(
      handle, HIPBLAS_OP_N, HIPBLAS_OP_T,
      (int)in_dim, (int)out_dim, (int)K, &alpha,
      input.data_ptr(), ab_type, (int)in_dim,
      grad_output.data_ptr(), ab_type, (int)out_dim, &beta,
      main_grad.data_ptr(), HIP_R_32F, (int)in_dim,
      HIPBLAS_COMPUTE_32F, HIPBLAS_GEMM_DEFAULT);
  TORCH_CHECK(status == HIPBLAS_STATUS_SUCCESS,
              "hipblasGemmEx failed with status ", (int)status);
}
