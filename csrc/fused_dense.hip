// apex_amd._fused_dense — GEMM+bias(+GELU) with hipBLASLt epilogue fusion,
// plus the fused weight-gradient accumulation GEMM.
// Reference surface: csrc/fused_dense.cpp:161-167 (linear_bias_forward/
// backward, linear_gelu_linear_forward/backward) and
// csrc/megatron/fused_weight_gradient_dense.cpp:11-13 (wgrad_gemm_accum_*).
#include "lt_gemm.h"
#include "multi_tensor_apply.h"  // VecPack

#include <vector>

namespace {

at::Tensor flat2d(const at::Tensor& t) {
  return t.contiguous().reshape({-1, t.size(-1)});
}

// tanh-GELU elementwise fwd/bwd — fallback for dtypes where hipBLASLt has no
// GELU_AUX_BIAS / DGELU_BGRAD algorithm (bf16 as of hipBLASLt 1.2). The
// tanh approximation matches hipBLASLt's fused GELU epilogue bit-for-bit in
// spirit (probed: |epilogue - tanh-gelu| ~ 3e-7 vs ~5e-4 for erf), so fused
// and fallback paths stay numerically consistent.
constexpr float kGeluC = 0.7978845608028654f;   // sqrt(2/pi)
constexpr float kGeluA = 0.044715f;

__device__ __forceinline__ float gelu_tanh(float v) {
  const float t = tanhf(kGeluC * (v + kGeluA * v * v * v));
  return 0.5f * v * (1.f + t);
}

__device__ __forceinline__ float dgelu_tanh(float v) {
  const float t = tanhf(kGeluC * (v + kGeluA * v * v * v));
  const float dt = (1.f - t * t) * kGeluC * (1.f + 3.f * kGeluA * v * v);
  return 0.5f * (1.f + t) + 0.5f * v * dt;
}

template <typename T>
__global__ void __launch_bounds__(256) gelu_fwd_kernel(const T* __restrict__ z,
                                                       T* __restrict__ y, long n) {
  constexpr int W = VecPack<T>::width;
  if ((n % W) == 0) {
    for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * W; i < n;
         i += (long)gridDim.x * blockDim.x * W) {
      VecPack<T> v, r;
      load_pack(v, z + i);
#pragma unroll
      for (int j = 0; j < W; ++j) r.a[j] = from_float<T>(gelu_tanh(to_float(v.a[j])));
      store_pack(y + i, r);
    }
  } else {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
      y[i] = from_float<T>(gelu_tanh(to_float(z[i])));
    }
  }
}

template <typename T>
__global__ void __launch_bounds__(256) dgelu_kernel(T* __restrict__ d, const T* __restrict__ z,
                                                    long n) {
  constexpr int W = VecPack<T>::width;
  if ((n % W) == 0) {
    for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * W; i < n;
         i += (long)gridDim.x * blockDim.x * W) {
      VecPack<T> vd, vz;
      load_pack(vd, d + i);
      load_pack(vz, z + i);
#pragma unroll
      for (int j = 0; j < W; ++j)
        vd.a[j] = from_float<T>(to_float(vd.a[j]) * dgelu_tanh(to_float(vz.a[j])));
      store_pack(d + i, vd);
    }
  } else {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
      d[i] = from_float<T>(to_float(d[i]) * dgelu_tanh(to_float(z[i])));
    }
  }
}

void gelu_fwd_inplace(const at::Tensor& z, at::Tensor& y) {
  const long n = z.numel();
  const int grid = (int)std::min<long>((n + 255) / 256, 8192);
  APEX_DISPATCH_FLOAT_HALF_BF(z.scalar_type(), "gelu_fwd", ([&] {
    hipLaunchKernelGGL((gelu_fwd_kernel<scalar_t>), dim3(grid), dim3(256), 0, current_stream(),
                       (const scalar_t*)z.data_ptr(), (scalar_t*)y.data_ptr(), n);
  }()));
  HIP_CHECK(hipGetLastError());
}

void dgelu_inplace(at::Tensor& d, const at::Tensor& z) {
  const long n = z.numel();
  const int grid = (int)std::min<long>((n + 255) / 256, 8192);
  APEX_DISPATCH_FLOAT_HALF_BF(z.scalar_type(), "dgelu", ([&] {
    hipLaunchKernelGGL((dgelu_kernel<scalar_t>), dim3(grid), dim3(256), 0, current_stream(),
                       (scalar_t*)d.data_ptr(), (const scalar_t*)z.data_ptr(), n);
  }()));
  HIP_CHECK(hipGetLastError());
}

}  // namespace

at::Tensor linear_bias_forward(at::Tensor input, at::Tensor weight, at::Tensor bias) {
  auto x = flat2d(input);
  auto w = weight.contiguous();
  auto b = bias.contiguous();
  auto out = at::empty({x.size(0), w.size(0)}, x.options());
  lt_linear(x, w, out, &b, HIPBLASLT_EPILOGUE_BIAS, nullptr);
  auto sizes = input.sizes().vec();
  sizes.back() = w.size(0);
  return out.reshape(sizes);
}

// out = relu(X @ W^T + b) in ONE hipBLASLt launch (RELU_BIAS epilogue) —
// the fused 1x1-conv/bias/ReLU building block (reference op surface:
// apex/contrib/csrc/conv_bias_relu/conv_bias_relu.cpp — cuDNN runtime
// fusion there; the CDNA4-native answer is a GEMM epilogue since a 1x1
// conv over NHWC IS a GEMM).
at::Tensor linear_bias_relu_forward(at::Tensor input, at::Tensor weight, at::Tensor bias) {
  auto x = flat2d(input);
  auto w = weight.contiguous();
  auto b = bias.contiguous();
  auto out = at::empty({x.size(0), w.size(0)}, x.options());
  if (!lt_linear(x, w, out, &b, HIPBLASLT_EPILOGUE_RELU_BIAS, nullptr, /*allow_fail=*/true)) {
    lt_linear(x, w, out, &b, HIPBLASLT_EPILOGUE_BIAS, nullptr);
    out.relu_();
  }
  auto sizes = input.sizes().vec();
  sizes.back() = w.size(0);
  return out.reshape(sizes);
}

at::Tensor linear_forward(at::Tensor input, at::Tensor weight) {
  auto x = flat2d(input);
  auto w = weight.contiguous();
  auto out = at::empty({x.size(0), w.size(0)}, x.options());
  lt_linear(x, w, out, nullptr, HIPBLASLT_EPILOGUE_DEFAULT, nullptr);
  auto sizes = input.sizes().vec();
  sizes.back() = w.size(0);
  return out.reshape(sizes);
}

std::vector<at::Tensor> linear_bias_backward(at::Tensor input, at::Tensor weight,
                                             at::Tensor grad_output) {
  auto x = flat2d(input);
  auto w = weight.contiguous();
  auto dy = flat2d(grad_output);
  auto dx = at::empty_like(x);
  auto dw = at::empty_like(w);
  auto db = at::empty({w.size(0)}, w.options());
  // dgrad: dX = dY @ W
  lt_linear_dgrad(dy, w, dx, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, nullptr);
  // wgrad + fused bias grad: dW = dY^T @ X, db = colsum(dY)
  lt_linear_wgrad(x, dy, dw, HIPBLASLT_EPILOGUE_BGRADB, &db, 0.f);
  return {dx.reshape(input.sizes()), dw, db};
}

std::vector<at::Tensor> linear_backward(at::Tensor input, at::Tensor weight,
                                        at::Tensor grad_output) {
  auto x = flat2d(input);
  auto w = weight.contiguous();
  auto dy = flat2d(grad_output);
  auto dx = at::empty_like(x);
  auto dw = at::empty_like(w);
  lt_linear_dgrad(dy, w, dx, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, nullptr);
  lt_linear_wgrad(x, dy, dw, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, 0.f);
  return {dx.reshape(input.sizes()), dw};
}

std::vector<at::Tensor> linear_gelu_linear_forward(at::Tensor input, at::Tensor weight1,
                                                   at::Tensor bias1, at::Tensor weight2,
                                                   at::Tensor bias2) {
  auto x = flat2d(input);
  auto w1 = weight1.contiguous();
  auto b1 = bias1.contiguous();
  auto w2 = weight2.contiguous();
  auto b2 = bias2.contiguous();
  const long m = x.size(0), n1 = w1.size(0), n2 = w2.size(0);
  auto output1 = at::empty({m, n1}, x.options());   // GELU(X@W1^T + b1)
  auto gelu_in = at::empty({m, n1}, x.options());   // pre-GELU (post-bias) aux
  auto output2 = at::empty({m, n2}, x.options());
  // GELU_AUX_BIAS trusted for fp16 only (see the DGELU note in backward).
  const bool fused_gelu = x.scalar_type() == at::ScalarType::Half &&
                          lt_linear(x, w1, output1, &b1, HIPBLASLT_EPILOGUE_GELU_AUX_BIAS,
                                    &gelu_in, /*allow_fail=*/true);
  if (!fused_gelu) {
    // split epilogue: BIAS GEMM into gelu_in, then elementwise tanh-GELU
    lt_linear(x, w1, gelu_in, &b1, HIPBLASLT_EPILOGUE_BIAS, nullptr);
    gelu_fwd_inplace(gelu_in, output1);
  }
  lt_linear(output1, w2, output2, &b2, HIPBLASLT_EPILOGUE_BIAS, nullptr);
  auto sizes = input.sizes().vec();
  sizes.back() = n2;
  return {output1, output2.reshape(sizes), gelu_in};
}

std::vector<at::Tensor> linear_gelu_linear_backward(at::Tensor input, at::Tensor gelu_in,
                                                    at::Tensor output1, at::Tensor weight1,
                                                    at::Tensor weight2, at::Tensor grad_output) {
  auto x = flat2d(input);
  auto o1 = flat2d(output1);
  auto gi = flat2d(gelu_in);
  auto w1 = weight1.contiguous();
  auto w2 = weight2.contiguous();
  auto dy2 = flat2d(grad_output);
  const long m = x.size(0), n1 = w1.size(0);

  auto dw2 = at::empty_like(w2);
  auto db2 = at::empty({w2.size(0)}, w2.options());
  lt_linear_wgrad(o1, dy2, dw2, HIPBLASLT_EPILOGUE_BGRADB, &db2, 0.f);

  // d_gelu = dGELU(dY2 @ W2, gelu_in), db1 fused.
  // The DGELU_BGRAD epilogue is only trusted for fp16: for bf16 the
  // heuristic reports algorithms but the aux (gelu_in) is misinterpreted and
  // the output is wrong (probed on MI355X: dx rel-err 0.8); fp32 has no
  // algorithms. Other dtypes take the split path (plain dgrad GEMM +
  // vectorized tanh-dGELU pass + bias-grad reduction).
  auto d_gelu = at::empty({m, n1}, x.options());
  auto db1 = at::empty({n1}, w1.options());
  const bool fused_dgelu = x.scalar_type() == at::ScalarType::Half &&
                           lt_linear_dgrad(dy2, w2, d_gelu, HIPBLASLT_EPILOGUE_DGELU_BGRAD, &gi,
                                           &db1, /*allow_fail=*/true);
  if (!fused_dgelu) {
    lt_linear_dgrad(dy2, w2, d_gelu, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, nullptr);
    dgelu_inplace(d_gelu, gi);
    db1.copy_(d_gelu.sum(0).to(db1.scalar_type()));
  }

  auto dw1 = at::empty_like(w1);
  lt_linear_wgrad(x, d_gelu, dw1, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, 0.f);
  auto dx = at::empty_like(x);
  lt_linear_dgrad(d_gelu, w1, dx, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, nullptr);
  return {dx.reshape(input.sizes()), dw1, db1, dw2, db2};
}

// main_grad[n,k] += grad_output[m,n]^T @ input[m,k] — fp32 accumulate,
// beta=1 so the result lands in the persistent main-grad buffer.
void wgrad_gemm_accum_fp32(at::Tensor input, at::Tensor grad_output, at::Tensor main_grad) {
  auto x = flat2d(input);
  auto dy = flat2d(grad_output);
  TORCH_CHECK(main_grad.scalar_type() == at::ScalarType::Float, "main_grad must be fp32");
  lt_linear_wgrad(x, dy, main_grad, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, 1.f);
}

void wgrad_gemm_accum_fp16(at::Tensor input, at::Tensor grad_output, at::Tensor main_grad) {
  auto x = flat2d(input);
  auto dy = flat2d(grad_output);
  TORCH_CHECK(main_grad.scalar_type() == at::ScalarType::Half ||
                  main_grad.scalar_type() == at::ScalarType::BFloat16,
              "main_grad must be fp16/bf16");
  lt_linear_wgrad(x, dy, main_grad, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, 1.f);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("linear_bias_relu_forward", &linear_bias_relu_forward);
  m.def("linear_bias_forward", &linear_bias_forward);
  m.def("linear_forward", &linear_forward);
  m.def("linear_bias_backward", &linear_bias_backward);
  m.def("linear_backward", &linear_backward);
  m.def("linear_gelu_linear_forward", &linear_gelu_linear_forward);
  m.def("linear_gelu_linear_backward", &linear_gelu_linear_backward);
  m.def("wgrad_gemm_accum_fp32", &wgrad_gemm_accum_fp32);
  m.def("wgrad_gemm_accum_fp16", &wgrad_gemm_accum_fp16);
}
