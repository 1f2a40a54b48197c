// multi_tensor_sgd — fused SGD with momentum/nesterov/dampening and
// optional low-precision param copy-out (the amp-O2 FusedSGD contract).
// Reference behavior: csrc/multi_tensor_sgd_kernel.cu:29-180. The reference
// hardcodes 4 dtype cases; here the functor is generically templated on
// (grad, param, copy-out) dtypes and N=3/4.
#include "amp_C.h"
#include "multi_tensor_apply.h"

namespace {

template <typename grad_t, typename param_t, typename out_t, int N>
struct SGDFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<N>& meta, int t, long chunk, float wd,
                             float momentum, float dampening, float lr, int nesterov,
                             int first_run, int wd_after_momentum, float scale) const {
    const long base = chunk * chunk_size;
    const grad_t* g = reinterpret_cast<const grad_t*>(meta.addrs[0][t]) + base;
    param_t* p = reinterpret_cast<param_t*>(meta.addrs[1][t]) + base;
    param_t* m = reinterpret_cast<param_t*>(meta.addrs[2][t]) + base;
    out_t* out = N == 4 ? reinterpret_cast<out_t*>(meta.addrs[N - 1][t]) + base : nullptr;
    const long n = min(meta.sizes[t] - base, chunk_size);

    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gf = to_float(g[i]) * scale;
      float pf = to_float(p[i]);
      if (wd != 0.f && !wd_after_momentum) gf = fmaf(wd, pf, gf);
      if (momentum != 0.f) {
        float mf;
        if (first_run) {
          mf = gf;
        } else {
          mf = fmaf(momentum, to_float(m[i]), (1.f - dampening) * gf);
        }
        m[i] = from_float<param_t>(mf);
        gf = nesterov ? fmaf(momentum, mf, gf) : mf;
      }
      if (wd != 0.f && wd_after_momentum) gf = fmaf(wd, pf, gf);
      pf = pf - lr * gf;
      p[i] = from_float<param_t>(pf);
      if (N == 4) out[i] = from_float<out_t>(pf);
    }
  }
};

}  // namespace

void multi_tensor_sgd_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                           double wd, double momentum, double dampening, double lr,
                           bool nesterov, bool first_run, bool wd_after_momentum, double scale) {
  const int N = (int)tensor_lists.size();
  TORCH_CHECK(N == 3 || N == 4, "multi_tensor_sgd: expected 3 or 4 tensor lists");
  const auto g_t = tensor_lists[0][0].scalar_type();
  const auto p_t = tensor_lists[1][0].scalar_type();
  const auto o_t = N == 4 ? tensor_lists[3][0].scalar_type() : p_t;
  TORCH_CHECK(N == 3 || o_t == at::ScalarType::Half || o_t == at::ScalarType::BFloat16,
              "multi_tensor_sgd: the 4th list (param copy) must be fp16/bf16");

  APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_sgd", ([&] {
    using grad_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_sgd", ([&] {
      using param_scalar = scalar_t;
      APEX_DISPATCH_FLOAT_HALF_BF(o_t, "multi_tensor_sgd", ([&] {
        using out_scalar = scalar_t;
        if (N == 3) {
          multi_tensor_apply<3>(chunk_size, noop_flag, tensor_lists,
                                SGDFunctor<grad_scalar, param_scalar, out_scalar, 3>(), (float)wd,
                                (float)momentum, (float)dampening, (float)lr, (int)nesterov,
                                (int)first_run, (int)wd_after_momentum, (float)scale);
        } else {
          multi_tensor_apply<4>(chunk_size, noop_flag, tensor_lists,
                                SGDFunctor<grad_scalar, param_scalar, out_scalar, 4>(), (float)wd,
                                (float)momentum, (float)dampening, (float)lr, (int)nesterov,
                                (int)first_run, (int)wd_after_momentum, (float)scale);
        }
      }()));
    }()));
  }()));
}
