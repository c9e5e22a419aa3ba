// Toolchain probe: minimal torch extension compiled natively with hipcc for gfx950.
#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

__global__ void add_one_kernel(float* x, int n) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) x[i] += 1.0f;
}

torch::Tensor add_one(torch::Tensor t) {
    if (t.is_cuda()) {
        auto stream = c10::hip::getCurrentHIPStream();
        int n = t.numel();
        hipLaunchKernelGGL(add_one_kernel, dim3((n + 255) / 256), dim3(256), 0,
                           stream.stream(), t.data_ptr<float>(), n);
        return t;
    }
    return t + 1;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) { m.def("add_one", &add_one); }
