// multi_tensor_adam — fused Adam/AdamW (+ hipGraph-capturable variants).
// Reference behavior: csrc/multi_tensor_adam.cu (AdamFunctor:24,
// AdamCapturableFunctor:111, AdamCapturableMasterFunctor:203; mode 0 = L2,
// mode 1 = decoupled AdamW; bias corrections precomputed host-side for the
// non-capturable path, in-kernel for capturable).
//
// MI355X roofline note: fp32 Adam moves 28 B/element (r g,p,m,v; w p,m,v) →
// a 350M-param step is ~9.8 GB. The r2 streaming sweep
// (profiles/probe_adam_tune2.log) measured 5.49 TB/s as the best ANY
// block/ILP config reaches for this 4-stream read + 3-stream write mix
// (1024/4 won; grid caps hurt), so ~1.8 ms is the practical floor — the
// measured 1.85-2.0 ms step is at ~95% of that. The kernel is pure
// streaming: 8/16-byte vector accesses, fp32 math in registers, one
// workgroup per 64K chunk.
#include "amp_C.h"
#include "multi_tensor_apply.h"

#define ADAM_MODE_L2 0
#define ADAM_MODE_ADAMW 1

namespace {

template <typename param_t, typename grad_t>
struct AdamFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<4>& meta, int t, long chunk, float lr,
                             float beta1, float beta2, float eps, float bc1_recip,
                             float bc2_recip, int mode, float decay) const {
    const long base = chunk * chunk_size;
    const grad_t* g = reinterpret_cast<const grad_t*>(meta.addrs[0][t]) + base;
    param_t* p = reinterpret_cast<param_t*>(meta.addrs[1][t]) + base;
    float* m = reinterpret_cast<float*>(meta.addrs[2][t]) + base;
    float* v = reinterpret_cast<float*>(meta.addrs[3][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);

    const bool vec_ok = is_vec4_aligned<grad_t>(g) && is_vec4_aligned<param_t>(p) &&
                        is_vec4_aligned<float>(m) && is_vec4_aligned<float>(v) &&
                        (n & (MTA_ILP - 1)) == 0;
    if (vec_ok) {
      for (long i = (long)threadIdx.x * MTA_ILP; i < n; i += (long)blockDim.x * MTA_ILP) {
        Vec4<grad_t> vg;
        Vec4<param_t> vp;
        Vec4<float> vm, vv;
        load_vec4(vg, g + i);
        load_vec4(vp, p + i);
        load_vec4(vm, m + i);
        load_vec4(vv, v + i);
#pragma unroll
        for (int j = 0; j < MTA_ILP; ++j) {
          float gf = to_float(vg.a[j]);
          float pf = to_float(vp.a[j]);
          if (mode == ADAM_MODE_L2) gf = fmaf(decay, pf, gf);
          float mf = fmaf(beta1, vm.a[j], (1.f - beta1) * gf);
          float vf = fmaf(beta2, vv.a[j], (1.f - beta2) * gf * gf);
          float update = (mf * bc1_recip) / (sqrtf(vf * bc2_recip) + eps);
          if (mode == ADAM_MODE_ADAMW) update = fmaf(decay, pf, update);
          vp.a[j] = from_float<param_t>(pf - lr * update);
          vm.a[j] = mf;
          vv.a[j] = vf;
        }
        store_vec4(p + i, vp);
        store_vec4(m + i, vm);
        store_vec4(v + i, vv);
      }
    } else {
      for (long i = threadIdx.x; i < n; i += blockDim.x) {
        float gf = to_float(g[i]);
        float pf = to_float(p[i]);
        if (mode == ADAM_MODE_L2) gf = fmaf(decay, pf, gf);
        float mf = fmaf(beta1, m[i], (1.f - beta1) * gf);
        float vf = fmaf(beta2, v[i], (1.f - beta2) * gf * gf);
        float update = (mf * bc1_recip) / (sqrtf(vf * bc2_recip) + eps);
        if (mode == ADAM_MODE_ADAMW) update = fmaf(decay, pf, update);
        p[i] = from_float<param_t>(pf - lr * update);
        m[i] = mf;
        v[i] = vf;
      }
    }
  }
};

// Capturable: lr / step / inv_scale live in device memory so the launch is
// hipGraph-replayable with updated values; grads are unscaled in-place; the
// whole block early-exits when found_inf (noop) is set.
template <typename param_t, bool kMaster>
struct AdamCapturableFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<kMaster ? 5 : 4>& meta, int t, long chunk,
                             const float* lr_ptr, float beta1, float beta2, float eps,
                             const int* step_ptr, int mode, int bias_correction, float decay,
                             const float* inv_scale_ptr) const {
    if (*noop == 1) return;
    const long base = chunk * chunk_size;
    param_t* g = reinterpret_cast<param_t*>(meta.addrs[0][t]) + base;
    param_t* p = reinterpret_cast<param_t*>(meta.addrs[1][t]) + base;
    float* m = reinterpret_cast<float*>(meta.addrs[2][t]) + base;
    float* v = reinterpret_cast<float*>(meta.addrs[3][t]) + base;
    float* master = kMaster ? reinterpret_cast<float*>(meta.addrs[4][t]) + base : nullptr;
    const long n = min(meta.sizes[t] - base, chunk_size);

    const float lr = *lr_ptr;
    const float inv_scale = *inv_scale_ptr;
    const int step = *step_ptr;
    float bc1_recip = 1.f, bc2_recip = 1.f;
    if (bias_correction) {
      bc1_recip = 1.f / (1.f - powf(beta1, (float)step));
      bc2_recip = 1.f / (1.f - powf(beta2, (float)step));
    }

    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gf = to_float(g[i]) * inv_scale;
      float pf = kMaster ? master[i] : to_float(p[i]);
      if (mode == ADAM_MODE_L2) gf = fmaf(decay, pf, gf);
      float mf = fmaf(beta1, m[i], (1.f - beta1) * gf);
      float vf = fmaf(beta2, v[i], (1.f - beta2) * gf * gf);
      float update = (mf * bc1_recip) / (sqrtf(vf * bc2_recip) + eps);
      if (mode == ADAM_MODE_ADAMW) update = fmaf(decay, pf, update);
      pf = pf - lr * update;
      p[i] = from_float<param_t>(pf);
      if (kMaster) master[i] = pf;
      g[i] = from_float<param_t>(gf);  // unscaled grad written back (reference :156)
      m[i] = mf;
      v[i] = vf;
    }
  }
};

}  // namespace

void multi_tensor_adam_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                            double lr, double beta1, double beta2, double eps, long step,
                            long mode, long bias_correction, double weight_decay) {
  float bc1_recip = 1.f, bc2_recip = 1.f;
  if (bias_correction == 1) {
    bc1_recip = (float)(1.0 / (1.0 - std::pow(beta1, (double)step)));
    bc2_recip = (float)(1.0 / (1.0 - std::pow(beta2, (double)step)));
  }
  const auto g_t = tensor_lists[0][0].scalar_type();
  const auto p_t = tensor_lists[1][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_adam", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_adam", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<4>(chunk_size, noop_flag, tensor_lists,
                            AdamFunctor<param_scalar, grad_scalar>(), (float)lr, (float)beta1,
                            (float)beta2, (float)eps, bc1_recip, bc2_recip, (int)mode,
                            (float)weight_decay);
    }()));
  }()));
}

void multi_tensor_adam_capturable_cuda(long chunk_size, at::Tensor noop_flag,
                                       TensorLists tensor_lists, at::Tensor lr, double beta1,
                                       double beta2, double eps, at::Tensor step, long mode,
                                       long bias_correction, double weight_decay,
                                       at::Tensor inv_scale) {
  APEX_DISPATCH_FLOAT_HALF_BF(tensor_lists[1][0].scalar_type(), "multi_tensor_adam_capturable", ([&] {
    multi_tensor_apply<4>(chunk_size, noop_flag, tensor_lists,
                          AdamCapturableFunctor<scalar_t, false>(), lr.data_ptr<float>(),
                          (float)beta1, (float)beta2, (float)eps, step.data_ptr<int>(),
                          (int)mode, (int)bias_correction, (float)weight_decay,
                          inv_scale.data_ptr<float>());
  }()));
}

void multi_tensor_adam_capturable_master_cuda(long chunk_size, at::Tensor noop_flag,
                                              TensorLists tensor_lists, at::Tensor lr,
                                              double beta1, double beta2, double eps,
                                              at::Tensor step, long mode, long bias_correction,
                                              double weight_decay, at::Tensor inv_scale) {
  APEX_DISPATCH_FLOAT_HALF_BF(tensor_lists[1][0].scalar_type(), "multi_tensor_adam_capturable_master", ([&] {
    multi_tensor_apply<5>(chunk_size, noop_flag, tensor_lists,
                          AdamCapturableFunctor<scalar_t, true>(), lr.data_ptr<float>(),
                          (float)beta1, (float)beta2, (float)eps, step.data_ptr<int>(),
                          (int)mode, (int)bias_correction, (float)weight_decay,
                          inv_scale.data_ptr<float>());
  }()));
}
