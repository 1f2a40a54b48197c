#include "hip/hip_runtime.h"
// apex_amd._index_mul_2d — fused out[i,:] = in1[idx[i],:] * in2[i,:].
// Reference surface: apex/contrib/index_mul_2d/index_mul_2d.py (float/half
// forward + fused backward; grad_in1 is a scatter-add over repeated indices).
#include "common.h"

#include <vector>

namespace {

constexpr int IM_BLOCK = 256;

template <typename T>
__global__ void __launch_bounds__(IM_BLOCK) index_mul_fwd_kernel(
    T* __restrict__ out, const T* __restrict__ in1, const T* __restrict__ in2,
    const long* __restrict__ idx, long n, long d) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n * d;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / d, c = i % d;
    out[i] = from_float<T>(to_float(in1[idx[r] * d + c]) * to_float(in2[i]));
  }
}

// grad_in1 = scatter_add(idx, grad_out * in2), accumulated in fp32 atomics
// (indices may repeat; device-scope atomicAdd per G12)
template <typename T>
__global__ void __launch_bounds__(IM_BLOCK) index_mul_bwd_in1_kernel(
    float* __restrict__ grad_in1_f32, const T* __restrict__ grad_out, const T* __restrict__ in2,
    const long* __restrict__ idx, long n, long d) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n * d;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / d, c = i % d;
    atomicAdd(&grad_in1_f32[idx[r] * d + c], to_float(grad_out[i]) * to_float(in2[i]));
  }
}

}  // namespace

void index_mul_2d_forward(at::Tensor out, at::Tensor in1, at::Tensor in2, at::Tensor idx) {
  const long n = in2.size(0), d = in2.size(1);
  const int grid = (int)std::min<long>((n * d + IM_BLOCK - 1) / IM_BLOCK, 8192);
  auto idxl = idx.to(at::kLong);
  APEX_DISPATCH_FLOAT_HALF_BF(in1.scalar_type(), "index_mul_2d_forward", ([&] {
    hipLaunchKernelGGL((index_mul_fwd_kernel<scalar_t>), dim3(grid), dim3(IM_BLOCK), 0,
                       current_stream(), (scalar_t*)out.data_ptr(),
                       (const scalar_t*)in1.data_ptr(), (const scalar_t*)in2.data_ptr(),
                       idxl.data_ptr<long>(), n, d);
  }()));
  HIP_CHECK(hipGetLastError());
}

std::vector<at::Tensor> index_mul_2d_backward(at::Tensor in1, at::Tensor in2, at::Tensor idx,
                                              at::Tensor grad_out) {
  const long n = in2.size(0), d = in2.size(1);
  auto go = grad_out.contiguous();
  auto idxl = idx.to(at::kLong);
  auto grad_in2 = at::empty_like(in2);
  auto grad_in1_f32 = at::zeros({in1.size(0), in1.size(1)},
                                in1.options().dtype(at::kFloat));
  const int grid = (int)std::min<long>((n * d + IM_BLOCK - 1) / IM_BLOCK, 8192);

  APEX_DISPATCH_FLOAT_HALF_BF(in1.scalar_type(), "index_mul_2d_backward", ([&] {
    // grad_in2 elementwise
    hipLaunchKernelGGL((index_mul_fwd_kernel<scalar_t>), dim3(grid), dim3(IM_BLOCK), 0,
                       current_stream(), (scalar_t*)grad_in2.data_ptr(),
                       (const scalar_t*)in1.data_ptr(), (const scalar_t*)go.data_ptr(),
                       idxl.data_ptr<long>(), n, d);
    // grad_in1 scatter-add in fp32
    hipLaunchKernelGGL((index_mul_bwd_in1_kernel<scalar_t>), dim3(grid), dim3(IM_BLOCK), 0,
                       current_stream(), grad_in1_f32.data_ptr<float>(),
                       (const scalar_t*)go.data_ptr(), (const scalar_t*)in2.data_ptr(),
                       idxl.data_ptr<long>(), n, d);
  }()));
  HIP_CHECK(hipGetLastError());
  return {grad_in1_f32.to(in1.scalar_type()), grad_in2};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("forward", &index_mul_2d_forward);
  m.def("backward", &index_mul_2d_backward);
}
