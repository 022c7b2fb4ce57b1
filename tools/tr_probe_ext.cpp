#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
extern "C" __global__ void tr_probe(float*, float*);
std::vector<torch::Tensor> run() {
  auto opts = torch::TensorOptions().dtype(torch::kFloat32).device(torch::kCUDA);
  auto u = torch::zeros({256}, opts);
  auto l = torch::zeros({256}, opts);
  hipLaunchKernelGGL(tr_probe, dim3(1), dim3(64), 0,
                     at::hip::getCurrentHIPStream().stream(),
                     u.data_ptr<float>(), l.data_ptr<float>());
  (void)hipDeviceSynchronize();
  return {u, l};
}
PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) { m.def("run", &run); }
