// Verification probe for the gfx950 mfma_f32_16x16x32_bf16 fragment
// layouts used by the hand-written MFMA kernels (guide §3 + G9: verify with
// random asymmetric inputs against a host reference before building on it).
//
// Assumed layouts (standard CDNA4 per-wave mapping, C/D measured in-guide):
//   A[r][k]: lane l holds r = l%16, k = (l/16)*8 + j   (j = 0..7)
//   B[k][c]: lane l holds c = l%16, k = (l/16)*8 + j
//   D[r][c]: lane l reg q holds row = (l>>4)*4 + q, col = l&15
#include "common.h"

#include <vector>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void mfma_tile_kernel(const short* __restrict__ A /*16x32 row-major*/,
                                 const short* __restrict__ B /*32x16 row-major*/,
                                 float* __restrict__ D /*16x16 row-major*/) {
  const int lane = threadIdx.x;
  bf16x8 a, b;
  const int r = lane % 16;
  const int c = lane % 16;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int k = (lane / 16) * 8 + j;
    a[j] = A[r * 32 + k];
    b[j] = B[k * 16 + c];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    D[((lane >> 4) * 4 + q) * 16 + (lane & 15)] = acc[q];
  }
}

}  // namespace

at::Tensor mfma_tile_probe(at::Tensor A_bf16, at::Tensor B_bf16) {
  TORCH_CHECK(A_bf16.sizes() == at::IntArrayRef({16, 32}) &&
              B_bf16.sizes() == at::IntArrayRef({32, 16}));
  auto A = A_bf16.contiguous();
  auto B = B_bf16.contiguous();
  auto D = at::zeros({16, 16}, A.options().dtype(at::kFloat));
  hipLaunchKernelGGL(mfma_tile_kernel, dim3(1), dim3(64), 0, current_stream(),
                     (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                     D.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return D;
}
