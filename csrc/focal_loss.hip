// apex_amd._focal_loss — fused sigmoid focal loss (loss + partial grad in
// one pass). Reference surface: apex/contrib/focal_loss/focal_loss.py:6-60
// (forward -> (loss, partial_grad); backward scales partial_grad in place by
// grad_loss / num_positives_sum).
//
// Semantics (per reference kernel focal_loss_cuda_kernel.cu): targets y per
// anchor; y == -2 ignores the anchor; y >= 0 marks class y positive, all
// other (real) classes negative; pad classes (>= num_real_classes) are
// skipped; optional label smoothing splits targets to 1-s/2 and s/2.
// Loss is summed then normalized by num_positives_sum (device scalar).
//
// MI355X design: grid-stride elementwise with per-block partial sums reduced
// by a fixed-order second kernel (deterministic — no fp32 atomics).
#include "common.h"

#include <vector>

namespace {

constexpr int FL_BLOCK = 256;
constexpr int FL_GRID = 1024;

template <typename T, bool SMOOTH>
__global__ void __launch_bounds__(FL_BLOCK) focal_fwd_kernel(
    const T* __restrict__ logits, const long* __restrict__ targets, T* __restrict__ partial_grad,
    float* __restrict__ block_sums, long num_examples, long C, long C_real, float alpha,
    float gamma, float smoothing) {
  const float half_s = smoothing * 0.5f;
  float acc = 0.f;
  const long total = num_examples * C;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long a = i / C;
    const long j = i % C;
    const long y = targets[a];
    if (y == -2 || j >= C_real) {
      partial_grad[i] = from_float<T>(0.f);
      continue;
    }
    const float p = to_float(logits[i]);
    const float sigma = 1.f / (1.f + __expf(-p));
    // stable softplus(-p) = -log(sigma)
    const float softplus_neg = (p >= 0.f ? 0.f : -p) + __logf(1.f + __expf(-fabsf(p)));

    const bool pos = (y >= 0) && (j == y);
    // target prob under smoothing
    const float t = pos ? (SMOOTH ? 1.f - half_s : 1.f) : (SMOOTH ? half_s : 0.f);
    // BCE-with-logits: ce = -t*log(sigma) - (1-t)*log(1-sigma)
    //   log(sigma) = -softplus(-p) ; log(1-sigma) = -p - softplus(-p)
    const float ce = t * softplus_neg + (1.f - t) * (p + softplus_neg);
    const float pt = pos ? sigma : 1.f - sigma;          // prob of the hard target
    const float a_t = pos ? alpha : 1.f - alpha;
    const float mod = __powf(1.f - pt, gamma);           // focal modulator
    const float loss = a_t * mod * ce;

    // d/dp [ a_t * (1-pt)^g * ce ]
    //   dce/dp = sigma - t
    //   dpt/dp = pos ?  sigma(1-sigma) : -sigma(1-sigma)
    const float dce = sigma - t;
    const float dpt = (pos ? 1.f : -1.f) * sigma * (1.f - sigma);
    float dmod = 0.f;
    if (gamma != 0.f) dmod = -gamma * __powf(1.f - pt, gamma - 1.f) * dpt;
    const float grad = a_t * (dmod * ce + mod * dce);

    acc += loss;
    partial_grad[i] = from_float<T>(grad);
  }
  __shared__ float smem[FL_BLOCK / WAVE_SIZE];
  float r = block_reduce_sum(acc, smem);
  if (threadIdx.x == 0) block_sums[blockIdx.x] = r;
}

__global__ void focal_finish_kernel(const float* __restrict__ block_sums, int nblocks,
                                    const float* __restrict__ num_positives_sum,
                                    float* __restrict__ loss_out) {
  __shared__ float smem[FL_BLOCK / WAVE_SIZE];
  float acc = 0.f;
  for (int i = threadIdx.x; i < nblocks; i += blockDim.x) acc += block_sums[i];
  float r = block_reduce_sum(acc, smem);
  if (threadIdx.x == 0) loss_out[0] = r / num_positives_sum[0];
}

template <typename T, typename G>
__global__ void __launch_bounds__(FL_BLOCK) focal_bwd_kernel(
    const G* __restrict__ grad_loss, T* __restrict__ partial_grad,
    const float* __restrict__ num_positives_sum, long total) {
  const float scale = to_float(grad_loss[0]) / num_positives_sum[0];
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    partial_grad[i] = from_float<T>(to_float(partial_grad[i]) * scale);
  }
}

}  // namespace

std::vector<at::Tensor> focal_loss_forward(at::Tensor cls_output, at::Tensor cls_targets,
                                           at::Tensor num_positives_sum, long num_real_classes,
                                           double alpha, double gamma, double label_smoothing) {
  auto x = cls_output.contiguous();
  auto t = cls_targets.contiguous().to(at::kLong);
  auto nps = num_positives_sum.contiguous().to(at::kFloat);
  const long C = x.size(-1);
  const long num_examples = x.numel() / C;
  auto partial_grad = at::empty_like(x);
  auto fopts = x.options().dtype(at::kFloat);
  auto block_sums = at::empty({FL_GRID}, fopts);
  auto loss = at::empty({}, fopts);

  APEX_DISPATCH_FLOAT_HALF_BF(x.scalar_type(), "focal_loss_forward", ([&] {
    if (label_smoothing != 0.0) {
      hipLaunchKernelGGL((focal_fwd_kernel<scalar_t, true>), dim3(FL_GRID), dim3(FL_BLOCK), 0,
                         current_stream(), (const scalar_t*)x.data_ptr(), t.data_ptr<long>(),
                         (scalar_t*)partial_grad.data_ptr(), block_sums.data_ptr<float>(),
                         num_examples, C, num_real_classes, (float)alpha, (float)gamma,
                         (float)label_smoothing);
    } else {
      hipLaunchKernelGGL((focal_fwd_kernel<scalar_t, false>), dim3(FL_GRID), dim3(FL_BLOCK), 0,
                         current_stream(), (const scalar_t*)x.data_ptr(), t.data_ptr<long>(),
                         (scalar_t*)partial_grad.data_ptr(), block_sums.data_ptr<float>(),
                         num_examples, C, num_real_classes, (float)alpha, (float)gamma, 0.f);
    }
  }()));
  HIP_CHECK(hipGetLastError());
  hipLaunchKernelGGL(focal_finish_kernel, dim3(1), dim3(FL_BLOCK), 0, current_stream(),
                     block_sums.data_ptr<float>(), FL_GRID, nps.data_ptr<float>(),
                     loss.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return {loss, partial_grad};
}

at::Tensor focal_loss_backward(at::Tensor grad_loss, at::Tensor partial_grad,
                               at::Tensor num_positives_sum) {
  auto nps = num_positives_sum.contiguous().to(at::kFloat);
  const long total = partial_grad.numel();
  const int grid = (int)std::min<long>((total + FL_BLOCK - 1) / FL_BLOCK, 8192);
  auto gl = grad_loss.contiguous();
  APEX_DISPATCH_FLOAT_HALF_BF(partial_grad.scalar_type(), "focal_loss_backward", ([&] {
    using T = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(gl.scalar_type(), "focal_loss_backward", ([&] {
      using G = scalar_t;
      hipLaunchKernelGGL((focal_bwd_kernel<T, G>), dim3(grid), dim3(FL_BLOCK), 0,
                         current_stream(), (const G*)gl.data_ptr(), (T*)partial_grad.data_ptr(),
                         nps.data_ptr<float>(), total);
    }()));
  }()));
  HIP_CHECK(hipGetLastError());
  return partial_grad;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("forward", &focal_loss_forward, "fused sigmoid focal loss -> (loss, partial_grad)");
  m.def("backward", &focal_loss_backward, "scale partial_grad in place");
}
