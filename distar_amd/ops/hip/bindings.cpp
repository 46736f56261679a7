// Python bindings for the distar_amd HIP/CDNA4 kernels (_hip_ops).
#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

extern "C" __global__ void lambda_return_kernel(
    const float*, const float*, const float*, const float*, float*, int, int);
extern "C" __global__ void vtrace_kernel(
    const float*, const float*, const float*, const float*, const float*,
    const float*, float*, int, int);

namespace {

inline void check_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on device");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

torch::Tensor lambda_return_scan(torch::Tensor rewards, torch::Tensor gammas,
                                 torch::Tensor values_tp1, torch::Tensor lambdas) {
  check_2d(rewards, "rewards");
  check_2d(gammas, "gammas");
  check_2d(values_tp1, "values_tp1");
  check_2d(lambdas, "lambdas");
  int64_t T = rewards.size(0), B = rewards.numel() / T;
  auto out = torch::empty_like(rewards);
  int threads = 256;
  int blocks = std::min<int64_t>((B + threads - 1) / threads, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(lambda_return_kernel, dim3(blocks), dim3(threads), 0,
                     stream.stream(),
                     rewards.data_ptr<float>(), gammas.data_ptr<float>(),
                     values_tp1.data_ptr<float>(), lambdas.data_ptr<float>(),
                     out.data_ptr<float>(), (int)T, (int)B);
  return out;
}

torch::Tensor vtrace_scan(torch::Tensor clipped_rhos, torch::Tensor clipped_cs,
                          torch::Tensor rewards, torch::Tensor values,
                          torch::Tensor gammas, torch::Tensor lambdas) {
  check_2d(clipped_rhos, "clipped_rhos");
  check_2d(rewards, "rewards");
  check_2d(values, "values");
  int64_t T = rewards.size(0), B = rewards.numel() / T;
  TORCH_CHECK(values.size(0) == T + 1, "values must be (T+1, B)");
  auto out = torch::empty_like(values);
  int threads = 256;
  int blocks = std::min<int64_t>((B + threads - 1) / threads, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(vtrace_kernel, dim3(blocks), dim3(threads), 0,
                     stream.stream(),
                     clipped_rhos.data_ptr<float>(), clipped_cs.data_ptr<float>(),
                     rewards.data_ptr<float>(), values.data_ptr<float>(),
                     gammas.data_ptr<float>(), lambdas.data_ptr<float>(),
                     out.data_ptr<float>(), (int)T, (int)B);
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("lambda_return_scan", &lambda_return_scan,
        "generalized lambda-return reverse scan (T,B)");
  m.def("vtrace_scan", &vtrace_scan, "v-trace corrected-value reverse scan");
}
