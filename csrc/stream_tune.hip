// Streaming-config sweep for the Adam-class multi-tensor kernels: pure
// 4-stream read / 3-stream write Adam math over a flat buffer, templated on
// (block size, ILP) to find the HBM-saturating launch shape on gfx950.
#include "common.h"
#include "multi_tensor_apply.h"

#include <vector>

namespace {

template <int BLOCK, int ILP>
__global__ void __launch_bounds__(BLOCK) adam_stream_kernel(
    const float* __restrict__ g, float* __restrict__ p, float* __restrict__ m,
    float* __restrict__ v, long n, float lr, float beta1, float beta2, float eps) {
  static_assert(ILP % 4 == 0);
  constexpr int NV = ILP / 4;
  const long base = ((long)blockIdx.x * BLOCK + threadIdx.x) * ILP;
  const long stride = (long)gridDim.x * BLOCK * ILP;
  for (long i = base; i + ILP <= n; i += stride) {
#pragma unroll
    for (int q = 0; q < NV; ++q) {
      Vec4<float> vg, vp, vm, vv;
      load_vec4(vg, g + i + q * 4);
      load_vec4(vp, p + i + q * 4);
      load_vec4(vm, m + i + q * 4);
      load_vec4(vv, v + i + q * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float gf = vg.a[j];
        float mf = fmaf(beta1, vm.a[j], (1.f - beta1) * gf);
        float vf = fmaf(beta2, vv.a[j], (1.f - beta2) * gf * gf);
        vp.a[j] -= lr * mf / (sqrtf(vf) + eps);
        vm.a[j] = mf;
        vv.a[j] = vf;
      }
      store_vec4(p + i + q * 4, vp);
      store_vec4(m + i + q * 4, vm);
      store_vec4(v + i + q * 4, vv);
    }
  }
}

}  // namespace

void adam_stream_probe(at::Tensor g, at::Tensor p, at::Tensor m, at::Tensor v, long variant,
                       long grid_blocks) {
  const long n = g.numel();
  auto stream = current_stream();
#define CASE(ID, BLOCK, ILP)                                                          \
  if (variant == ID) {                                                                \
    const int grid = grid_blocks > 0 ? (int)grid_blocks                               \
                                     : (int)std::min<long>((n + BLOCK * ILP - 1) /    \
                                                           ((long)BLOCK * ILP), 65535); \
    hipLaunchKernelGGL((adam_stream_kernel<BLOCK, ILP>), dim3(grid), dim3(BLOCK), 0,  \
                       stream, g.data_ptr<float>(), p.data_ptr<float>(),              \
                       m.data_ptr<float>(), v.data_ptr<float>(), n, 1e-3f, 0.9f,      \
                       0.999f, 1e-8f);                                                \
  }
  CASE(0, 256, 4)
  CASE(1, 256, 8)
  CASE(2, 512, 4)
  CASE(3, 512, 8)
  CASE(4, 1024, 4)
  CASE(5, 128, 8)
  CASE(6, 256, 16)
  CASE(7, 512, 16)
#undef CASE
  HIP_CHECK(hipGetLastError());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m_) {
  m_.def("adam_stream_probe", &adam_stream_probe);
}
