// multi_tensor_adagrad + multi_tensor_novograd.
// Reference behavior: csrc/multi_tensor_adagrad.cu:25-82 (mode 0 = L2 decay
// before accumulation, mode 1 = decoupled) and
// csrc/multi_tensor_novograd.cu:27-100 (per-TENSOR second moment passed as a
// flat norm vector; moment mode 0 = wd outside, 1 = wd inside).
#include "amp_C.h"
#include "multi_tensor_apply_hip.h"

namespace {

template <typename param_t, typename grad_t>
struct AdagradFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<3>& meta, int t, long chunk, float lr,
                             float eps, int mode, float decay) const {
    const long base = chunk * chunk_size;
    const grad_t* g = reinterpret_cast<const grad_t*>(meta.addrs[0][t]) + base;
    param_t* p = reinterpret_cast<param_t*>(meta.addrs[1][t]) + base;
    float* h = reinterpret_cast<float*>(meta.addrs[2][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);

    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gf = to_float(g[i]);
      float pf = to_float(p[i]);
      if (mode == 0 && decay != 0.f) gf = fmaf(decay, pf, gf);
      float hf = fmaf(gf, gf, h[i]);
      float update = gf / (sqrtf(hf) + eps);
      if (mode == 1 && decay != 0.f) update = fmaf(decay, pf, update);
      p[i] = from_float<param_t>(pf - lr * update);
      h[i] = hf;
    }
  }
};

// NovoGrad: v is per-tensor (indexed by the GLOBAL tensor index); the Python
// layer updates v from per-tensor grad norms before this launch.
template <typename param_t, typename grad_t>
struct NovoGradFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<3>& meta, int t, long chunk,
                             const float* v_vec, float lr, float beta1, float beta3,
                             float bc1_recip, float bc2_recip, float eps, int moment_mode,
                             float decay) const {
    const long base = chunk * chunk_size;
    const grad_t* g = reinterpret_cast<const grad_t*>(meta.addrs[0][t]) + base;
    param_t* p = reinterpret_cast<param_t*>(meta.addrs[1][t]) + base;
    float* m = reinterpret_cast<float*>(meta.addrs[2][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);

    const float v = v_vec[meta.tensor_offset + t];
    const float denom = sqrtf(v * bc2_recip) + eps;

    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gf = to_float(g[i]) / denom;
      float pf = to_float(p[i]);
      if (moment_mode == 1 && decay != 0.f) gf = fmaf(decay, pf, gf);
      float mf = fmaf(beta1, m[i], beta3 * gf);
      float update = mf * bc1_recip;
      if (moment_mode == 0 && decay != 0.f) update = fmaf(decay, pf, update);
      p[i] = from_float<param_t>(pf - lr * update);
      m[i] = mf;
    }
  }
};

}  // namespace

void multi_tensor_adagrad_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                               double lr, double eps, long mode, double weight_decay) {
  const auto g_t = tensor_lists[0][0].scalar_type();
  const auto p_t = tensor_lists[1][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_adagrad", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_adagrad", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<3>(chunk_size, noop_flag, tensor_lists,
                            AdagradFunctor<param_scalar, grad_scalar>(), (float)lr, (float)eps,
                            (int)mode, (float)weight_decay);
    }()));
  }()));
}

void multi_tensor_novograd_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                                at::Tensor per_tensor_v, double lr, double beta1, double beta2,
                                double eps, long step, long bias_correction, double weight_decay,
                                long grad_averaging, long moment_mode, long norm_type) {
  float bc1_recip = 1.f, bc2_recip = 1.f;
  if (bias_correction == 1) {
    bc1_recip = (float)(1.0 / (1.0 - std::pow(beta1, (double)step)));
    bc2_recip = (float)(1.0 / (1.0 - std::pow(beta2, (double)step)));
  }
  const float beta3 = grad_averaging ? (float)(1.0 - beta1) : 1.0f;
  const auto g_t = tensor_lists[0][0].scalar_type();
  const auto p_t = tensor_lists[1][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_novograd", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_novograd", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<3>(chunk_size, noop_flag, tensor_lists,
                            NovoGradFunctor<param_scalar, grad_scalar>(),
                            per_tensor_v.data_ptr<float>(), (float)lr, (float)beta1, beta3,
                            bc1_recip, bc2_recip, (float)eps, (int)moment_mode,
                            (float)weight_decay);
    }()));
  }()));
}
