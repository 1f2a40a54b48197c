// apex_amd._mlp — N-layer perceptron in one fused call.
// Reference surface: csrc/mlp.cpp (mlp_forward returns {out, reserved_space}
// holding every intermediate activation; mlp_backward returns all grads).
// MI355X path: hipBLASLt GEMMs with RELU_BIAS/BIAS epilogues writing layer
// outputs directly into the flat reserved buffer; backward fuses the bias
// gradient into the wgrad GEMM (BGRADB epilogue).
#include "lt_gemm.h"

#include <vector>

namespace {

int num_layers(bool bias, size_t nargs) { return bias ? (int)(nargs - 1) / 2 : (int)nargs - 1; }

}  // namespace

std::vector<at::Tensor> mlp_forward(int bias, int activation, std::vector<at::Tensor> args) {
  auto input = args[0].contiguous();
  TORCH_CHECK(input.dim() == 2, "mlp expects 2D input [batch, features]");
  const int nl = num_layers(bias, args.size());
  const long batch = input.size(0);

  long total = 0;
  for (int i = 0; i < nl; ++i) total += batch * args[1 + i].size(0);
  auto reserved = at::empty({total}, input.options());

  at::Tensor x = input;
  long off = 0;
  for (int i = 0; i < nl; ++i) {
    auto w = args[1 + i].contiguous();
    const long n = w.size(0);
    auto y = reserved.narrow(0, off, batch * n).view({batch, n});
    off += batch * n;
    if (bias) {
      auto b = args[1 + nl + i].contiguous();
      hipblasLtEpilogue_t epi = activation == 1 ? HIPBLASLT_EPILOGUE_RELU_BIAS
                                                : HIPBLASLT_EPILOGUE_BIAS;
      lt_linear(x, w, y, &b, epi, nullptr);
    } else {
      hipblasLtEpilogue_t epi = activation == 1 ? HIPBLASLT_EPILOGUE_RELU
                                                : HIPBLASLT_EPILOGUE_DEFAULT;
      lt_linear(x, w, y, nullptr, epi, nullptr);
    }
    if (activation == 2) y.sigmoid_();
    x = y;
  }
  return {x.clone(), reserved};
}

std::vector<at::Tensor> mlp_backward(int bias, int activation, at::Tensor grad_o,
                                     at::Tensor reserved, std::vector<at::Tensor> args) {
  auto input = args[0].contiguous();
  const int nl = num_layers(bias, args.size());
  const long batch = input.size(0);

  // rebuild activation views
  std::vector<at::Tensor> acts(nl + 1);
  acts[0] = input;
  long off = 0;
  for (int i = 0; i < nl; ++i) {
    const long n = args[1 + i].size(0);
    acts[i + 1] = reserved.narrow(0, off, batch * n).view({batch, n});
    off += batch * n;
  }

  std::vector<at::Tensor> wgrads(nl), bgrads(nl);
  at::Tensor dy = grad_o.contiguous();
  for (int i = nl - 1; i >= 0; --i) {
    auto w = args[1 + i].contiguous();
    auto y = acts[i + 1];
    if (activation == 1) {
      dy = dy * (y > 0).to(dy.scalar_type());
    } else if (activation == 2) {
      dy = dy * y * (1.0 - y);
    }
    wgrads[i] = at::empty_like(w);
    if (bias) {
      bgrads[i] = at::empty({w.size(0)}, w.options());
      lt_linear_wgrad(acts[i], dy, wgrads[i], HIPBLASLT_EPILOGUE_BGRADB, &bgrads[i], 0.f);
    } else {
      lt_linear_wgrad(acts[i], dy, wgrads[i], HIPBLASLT_EPILOGUE_DEFAULT, nullptr, 0.f);
    }
    auto dx = at::empty({batch, w.size(1)}, dy.options());
    lt_linear_dgrad(dy, w, dx, HIPBLASLT_EPILOGUE_DEFAULT, nullptr, nullptr);
    dy = dx;
  }

  std::vector<at::Tensor> out;
  out.push_back(dy);  // grad_input
  for (auto& t : wgrads) out.push_back(t);
  if (bias)
    for (auto& t : bgrads) out.push_back(t);
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("forward", &mlp_forward);
  m.def("backward", &mlp_backward);
}
