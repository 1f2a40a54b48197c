// multi_tensor_lamb (+_mp) — fused LAMB.
// Reference behavior: csrc/multi_tensor_lamb.cu (stage 1 Adam-style update
// with global-grad-norm clip written into g; stage 2 per-tensor trust-ratio
// apply) and csrc/multi_tensor_lamb_mp.cu (device-tensor lr/step/found_inf/
// inv_scale + fp32 master path).
//
// Host orchestration: stage1 launch → per-tensor L2 norms of params and
// updates (deterministic two-stage reduction, see multi_tensor_l2norm.hip)
// → stage2 launch. All launches cover every chunk in one grid.
#include "amp_C.h"
#include "multi_tensor_apply.h"

namespace {

// ---- stage 1: update_t = adam_update(g/clip) (+wd*p) written into g ----
template <typename param_t, typename grad_t, bool kMp>
struct LambStage1 {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<kMp ? 5 : 4>& meta, int t, long chunk,
                             float beta1, float beta2, float beta3, float bc1_recip,
                             float bc2_recip, float eps, int mode, float decay,
                             const float* global_grad_norm, float max_grad_norm,
                             const float* inv_scale_ptr, const float* found_inf) const {
    if (kMp && found_inf && *found_inf != 0.f) return;
    const long base = chunk * chunk_size;
    grad_t* g = reinterpret_cast<grad_t*>(meta.addrs[0][t]) + base;
    const param_t* p = reinterpret_cast<const param_t*>(meta.addrs[1][t]) + base;
    float* m = reinterpret_cast<float*>(meta.addrs[2][t]) + base;
    float* v = reinterpret_cast<float*>(meta.addrs[3][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);

    const float inv_scale = inv_scale_ptr ? *inv_scale_ptr : 1.0f;
    float gnorm = *global_grad_norm * inv_scale;
    const float clip = (max_grad_norm > 0.f && gnorm > max_grad_norm) ? gnorm / max_grad_norm : 1.0f;
    const float combined_scale = inv_scale / clip;

    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gf = to_float(g[i]) * combined_scale;
      float pf = to_float(p[i]);
      if (mode == 0 && decay != 0.f) gf = fmaf(decay, pf, gf);
      float mf = fmaf(beta1, m[i], beta3 * gf);
      float vf = fmaf(beta2, v[i], (1.f - beta2) * gf * gf);
      float update = (mf * bc1_recip) / (sqrtf(vf * bc2_recip) + eps);
      if (mode == 1 && decay != 0.f) update = fmaf(decay, pf, update);
      g[i] = from_float<grad_t>(update);
      m[i] = mf;
      v[i] = vf;
    }
  }
};

// capturable stage 1: bias corrections from a DEVICE step pointer (powf
// in-kernel, like AdamCapturableFunctor) so the launch replays correctly
// inside a hipGraph with an advancing step.
template <typename param_t, typename grad_t>
struct LambStage1Capturable {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<4>& meta, int t, long chunk, float beta1,
                             float beta2, float beta3, const int* step_ptr, int bias_correction,
                             float eps, int mode, float decay,
                             const float* global_grad_norm, float max_grad_norm) const {
    const long base = chunk * chunk_size;
    grad_t* g = reinterpret_cast<grad_t*>(meta.addrs[0][t]) + base;
    const param_t* p = reinterpret_cast<const param_t*>(meta.addrs[1][t]) + base;
    float* m = reinterpret_cast<float*>(meta.addrs[2][t]) + base;
    float* v = reinterpret_cast<float*>(meta.addrs[3][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);

    const int step = *step_ptr;
    float bc1_recip = 1.f, bc2_recip = 1.f;
    if (bias_correction) {
      bc1_recip = 1.f / (1.f - powf(beta1, (float)step));
      bc2_recip = 1.f / (1.f - powf(beta2, (float)step));
    }
    const float gnorm = *global_grad_norm;
    const float clip = (max_grad_norm > 0.f && gnorm > max_grad_norm) ? gnorm / max_grad_norm : 1.0f;
    const float combined_scale = 1.0f / clip;

    for (long i = threadIdx.x; i < n; i += blockDim.x) {
      float gf = to_float(g[i]) * combined_scale;
      float pf = to_float(p[i]);
      if (mode == 0 && decay != 0.f) gf = fmaf(decay, pf, gf);
      float mf = fmaf(beta1, m[i], beta3 * gf);
      float vf = fmaf(beta2, v[i], (1.f - beta2) * gf * gf);
      float update = (mf * bc1_recip) / (sqrtf(vf * bc2_recip) + eps);
      if (mode == 1 && decay != 0.f) update = fmaf(decay, pf, update);
      g[i] = from_float<grad_t>(update);
      m[i] = mf;
      v[i] = vf;
    }
  }
};

// ---- stage 2: p -= ratio * update, ratio = lr * pnorm/unorm (trust) ----
template <typename param_t, typename grad_t, bool kMp>
struct LambStage2 {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<kMp ? 3 : 2>& meta, int t, long chunk,
                             const float* param_norms, const float* update_norms,
                             const float* lr_ptr, float lr_const, float decay, int use_nvlamb,
                             const float* found_inf) const {
    if (kMp && found_inf && *found_inf != 0.f) return;
    const long base = chunk * chunk_size;
    param_t* p = reinterpret_cast<param_t*>(meta.addrs[0][t]) + base;
    const grad_t* u = reinterpret_cast<const grad_t*>(meta.addrs[1][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);

    const float lr = lr_ptr ? *lr_ptr : lr_const;
    const float pn = param_norms[meta.tensor_offset + t];
    const float un = update_norms[meta.tensor_offset + t];
    float ratio = lr;
    if ((use_nvlamb || decay != 0.f) && pn != 0.f && un != 0.f) ratio = lr * (pn / un);

    if (!kMp) {
      for (long i = threadIdx.x; i < n; i += blockDim.x) {
        p[i] = from_float<param_t>(to_float(p[i]) - ratio * to_float(u[i]));
      }
    } else {
      // kMp: list 2 = low-precision model copy written alongside fp32 master
      using copy_t = grad_t;  // model dtype == update (grad) dtype in mp path
      copy_t* pc = reinterpret_cast<copy_t*>(meta.addrs[2][t]) + base;
      for (long i = threadIdx.x; i < n; i += blockDim.x) {
        float pf = to_float(p[i]) - ratio * to_float(u[i]);
        p[i] = from_float<param_t>(pf);
        pc[i] = from_float<copy_t>(pf);
      }
    }
  }
};

}  // namespace

void multi_tensor_lamb_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                            double lr, double beta1, double beta2, double eps, long step,
                            long bias_correction, double weight_decay, long grad_averaging,
                            long mode, at::Tensor global_grad_norm, double max_grad_norm,
                            bool use_nvlamb) {
  float bc1_recip = 1.f, bc2_recip = 1.f;
  if (bias_correction == 1) {
    bc1_recip = (float)(1.0 / (1.0 - std::pow(beta1, (double)step)));
    bc2_recip = (float)(1.0 / (1.0 - std::pow(beta2, (double)step)));
  }
  const float beta3 = grad_averaging ? (float)(1.0 - beta1) : 1.0f;
  auto gnorm = global_grad_norm.to(at::kFloat);

  const auto g_t = tensor_lists[0][0].scalar_type();
  const auto p_t = tensor_lists[1][0].scalar_type();

  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_lamb", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_lamb", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<4>(chunk_size, noop_flag, tensor_lists,
                            LambStage1<param_scalar, grad_scalar, false>(), (float)beta1,
                            (float)beta2, beta3, bc1_recip, bc2_recip, (float)eps, (int)mode,
                            (float)weight_decay, gnorm.data_ptr<float>(), (float)max_grad_norm,
                            (const float*)nullptr, (const float*)nullptr);

      // per-tensor norms of params and updates (updates now live in g)
      auto pn = multi_tensor_l2norm_cuda(chunk_size, noop_flag, {tensor_lists[1]}, true)[1];
      auto un = multi_tensor_l2norm_cuda(chunk_size, noop_flag, {tensor_lists[0]}, true)[1];

      TensorLists stage2_lists = {tensor_lists[1], tensor_lists[0]};
      multi_tensor_apply<2>(chunk_size, noop_flag, stage2_lists,
                            LambStage2<param_scalar, grad_scalar, false>(),
                            pn.data_ptr<float>(), un.data_ptr<float>(), (const float*)nullptr,
                            (float)lr, (float)weight_decay, (int)use_nvlamb,
                            (const float*)nullptr);
    }()));
  }()));
}

void multi_tensor_lamb_mp_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                               at::Tensor lr, double beta1, double beta2, double eps,
                               at::Tensor step, long bias_correction, double weight_decay,
                               long grad_averaging, long mode, at::Tensor global_grad_norm,
                               double max_grad_norm, bool use_nvlamb, at::Tensor found_inf,
                               at::Tensor inv_scale) {
  // bias corrections need the host step only when bias_correction=1; the mp
  // path keeps step on device — read it back once (cheap, optimizer-rate).
  long step_host = step.to(at::kCPU).item<long>();
  float bc1_recip = 1.f, bc2_recip = 1.f;
  if (bias_correction == 1) {
    bc1_recip = (float)(1.0 / (1.0 - std::pow(beta1, (double)step_host)));
    bc2_recip = (float)(1.0 / (1.0 - std::pow(beta2, (double)step_host)));
  }
  const float beta3 = grad_averaging ? (float)(1.0 - beta1) : 1.0f;
  auto gnorm = global_grad_norm.to(at::kFloat);

  const auto g_t = tensor_lists[0][0].scalar_type();  // grads (model dtype)
  const auto p_t = tensor_lists[1][0].scalar_type();  // fp32 masters

  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_lamb_mp", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_lamb_mp", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<5>(chunk_size, noop_flag, tensor_lists,
                            LambStage1<param_scalar, grad_scalar, true>(), (float)beta1,
                            (float)beta2, beta3, bc1_recip, bc2_recip, (float)eps, (int)mode,
                            (float)weight_decay, gnorm.data_ptr<float>(), (float)max_grad_norm,
                            inv_scale.data_ptr<float>(), found_inf.data_ptr<float>());

      auto pn = multi_tensor_l2norm_cuda(chunk_size, noop_flag, {tensor_lists[1]}, true)[1];
      auto un = multi_tensor_l2norm_cuda(chunk_size, noop_flag, {tensor_lists[0]}, true)[1];

      TensorLists stage2_lists = {tensor_lists[1], tensor_lists[0], tensor_lists[4]};
      multi_tensor_apply<3>(chunk_size, noop_flag, stage2_lists,
                            LambStage2<param_scalar, grad_scalar, true>(),
                            pn.data_ptr<float>(), un.data_ptr<float>(), lr.data_ptr<float>(),
                            0.f, (float)weight_decay, (int)use_nvlamb,
                            found_inf.data_ptr<float>());
    }()));
  }()));
}

// ---- sharded (ZeRO) entry points for DistributedFusedLAMB ----
// Reference analogue: the legacy two-phase csrc/multi_tensor_lamb_stage_1.cu
// / stage_2.cu bindings the MLPerf-BERT DistributedFusedLAMB pipeline uses
// (distributed_fused_lamb.py:1105-1133) — here the phases are split so the
// caller can all-reduce the per-SEGMENT param/update norms across ranks
// between them (each rank holds only a shard of every tensor).

// stage 1 only: update written into g. global_grad_norm is a DEVICE scalar
// (no host sync on the norm path).
void multi_tensor_lamb_stage1_cuda(long chunk_size, at::Tensor noop_flag,
                                   TensorLists tensor_lists, double beta1, double beta2,
                                   double eps, long step, long bias_correction,
                                   double weight_decay, long grad_averaging, long mode,
                                   at::Tensor global_grad_norm, double max_grad_norm) {
  float bc1_recip = 1.f, bc2_recip = 1.f;
  if (bias_correction == 1) {
    bc1_recip = (float)(1.0 / (1.0 - std::pow(beta1, (double)step)));
    bc2_recip = (float)(1.0 / (1.0 - std::pow(beta2, (double)step)));
  }
  const float beta3 = grad_averaging ? (float)(1.0 - beta1) : 1.0f;
  auto gnorm = global_grad_norm.to(at::kFloat);
  const auto g_t = tensor_lists[0][0].scalar_type();
  const auto p_t = tensor_lists[1][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_lamb_stage1", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_lamb_stage1", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<4>(chunk_size, noop_flag, tensor_lists,
                            LambStage1<param_scalar, grad_scalar, false>(), (float)beta1,
                            (float)beta2, beta3, bc1_recip, bc2_recip, (float)eps, (int)mode,
                            (float)weight_decay, gnorm.data_ptr<float>(), (float)max_grad_norm,
                            (const float*)nullptr, (const float*)nullptr);
    }()));
  }()));
}

// stage 2 only: p -= trust_ratio * update with caller-supplied per-listed-
// tensor norms (DEVICE fp32 [n_tensors], already reduced across ranks).
void multi_tensor_lamb_stage2_cuda(long chunk_size, at::Tensor noop_flag,
                                   TensorLists tensor_lists, at::Tensor param_norms,
                                   at::Tensor update_norms, double lr, double weight_decay,
                                   bool use_nvlamb) {
  auto pn = param_norms.to(at::kFloat).contiguous();
  auto un = update_norms.to(at::kFloat).contiguous();
  const auto p_t = tensor_lists[0][0].scalar_type();
  const auto g_t = tensor_lists[1][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_lamb_stage2", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_lamb_stage2", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<2>(chunk_size, noop_flag, tensor_lists,
                            LambStage2<param_scalar, grad_scalar, false>(),
                            pn.data_ptr<float>(), un.data_ptr<float>(), (const float*)nullptr,
                            (float)lr, (float)weight_decay, (int)use_nvlamb,
                            (const float*)nullptr);
    }()));
  }()));
}

// hipGraph-capturable LAMB: device lr/step, in-kernel bias corrections,
// device global-grad-norm end to end.
void multi_tensor_lamb_capturable_cuda(long chunk_size, at::Tensor noop_flag,
                                       TensorLists tensor_lists, at::Tensor lr, double beta1,
                                       double beta2, double eps, at::Tensor step,
                                       long bias_correction, double weight_decay,
                                       long grad_averaging, long mode,
                                       at::Tensor global_grad_norm, double max_grad_norm,
                                       bool use_nvlamb) {
  const float beta3 = grad_averaging ? (float)(1.0 - beta1) : 1.0f;
  auto gnorm = global_grad_norm.to(at::kFloat);
  const auto g_t = tensor_lists[0][0].scalar_type();
  const auto p_t = tensor_lists[1][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(p_t, "multi_tensor_lamb_capturable", ([&] {
    using param_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(g_t, "multi_tensor_lamb_capturable", ([&] {
      using grad_scalar = scalar_t;
      multi_tensor_apply<4>(chunk_size, noop_flag, tensor_lists,
                            LambStage1Capturable<param_scalar, grad_scalar>(), (float)beta1,
                            (float)beta2, beta3, step.data_ptr<int>(), (int)bias_correction,
                            (float)eps, (int)mode, (float)weight_decay,
                            gnorm.data_ptr<float>(), (float)max_grad_norm);

      auto pn = multi_tensor_l2norm_cuda(chunk_size, noop_flag, {tensor_lists[1]}, true)[1];
      auto un = multi_tensor_l2norm_cuda(chunk_size, noop_flag, {tensor_lists[0]}, true)[1];

      TensorLists stage2_lists = {tensor_lists[1], tensor_lists[0]};
      multi_tensor_apply<2>(chunk_size, noop_flag, stage2_lists,
                            LambStage2<param_scalar, grad_scalar, false>(),
                            pn.data_ptr<float>(), un.data_ptr<float>(), lr.data_ptr<float>(),
                            0.f, (float)weight_decay, (int)use_nvlamb,
                            (const float*)nullptr);
    }()));
  }()));
}
