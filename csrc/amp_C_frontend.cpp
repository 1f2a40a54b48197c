// pybind frontend for apex_amd._amp_C (reference surface:
// csrc/amp_C_frontend.cpp:83-123).
#include "amp_C.h"

#include <torch/extension.h>

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("multi_tensor_scale", &multi_tensor_scale_cuda,
        "Fused out = in * scale with isfinite check (sets noop flag)");
  m.def("multi_tensor_axpby", &multi_tensor_axpby_cuda,
        "Fused out = a*x + b*y with selectable isfinite check");
  m.def("multi_tensor_l2norm", &multi_tensor_l2norm_cuda,
        "L2 norm over a tensor list (+ optional per-tensor norms)");
  m.def("multi_tensor_l2norm_mp", &multi_tensor_l2norm_cuda,
        "L2 norm, fp32 accumulation (alias: accumulation is always fp32 here)");
  m.def("multi_tensor_unscale_l2norm", &multi_tensor_unscale_l2norm_cuda,
        "L2 norm of in*inv_scale (device scalar) without writing in");
  m.def("multi_tensor_l2norm_scale", &multi_tensor_l2norm_scale_cuda,
        "Fused out = in*scale plus L2 norm of the scaled values");
  m.def("multi_tensor_maxnorm", &multi_tensor_maxnorm_cuda,
        "Per-tensor max-abs norms over a tensor list");
  m.def("multi_tensor_sgd", &multi_tensor_sgd_cuda,
        "Fused SGD (momentum/nesterov/dampening, optional fp16/bf16 copy-out)");
  m.def("multi_tensor_adam", &multi_tensor_adam_cuda, "Fused Adam/AdamW");
  m.def("multi_tensor_adam_capturable", &multi_tensor_adam_capturable_cuda,
        "hipGraph-capturable Adam (device lr/step/inv_scale)");
  m.def("multi_tensor_adam_capturable_master", &multi_tensor_adam_capturable_master_cuda,
        "hipGraph-capturable Adam with fp32 master params");
  m.def("multi_tensor_adagrad", &multi_tensor_adagrad_cuda, "Fused Adagrad");
  m.def("multi_tensor_novograd", &multi_tensor_novograd_cuda,
        "Fused NovoGrad (per-tensor second moment vector)");
  m.def("multi_tensor_lamb", &multi_tensor_lamb_cuda,
        "Fused LAMB (global-norm clip + per-tensor trust ratio)");
  m.def("multi_tensor_lamb_capturable", &multi_tensor_lamb_capturable_cuda,
        "hipGraph-capturable LAMB (device lr/step/global-norm)");
  m.def("multi_tensor_lamb_stage1", &multi_tensor_lamb_stage1_cuda,
        "LAMB stage 1 only (sharded: update into g; device global-grad-norm)");
  m.def("multi_tensor_lamb_stage2", &multi_tensor_lamb_stage2_cuda,
        "LAMB stage 2 only (sharded: caller-reduced per-tensor norms)");
  m.def("multi_tensor_lamb_mp", &multi_tensor_lamb_mp_cuda,
        "Graph-safe LAMB with device-tensor state and fp32 masters");
  // reference-name aliases (amp_C_frontend.cpp exports these with the
  // _cuda suffix; downstream callers use either)
  m.def("multi_tensor_lamb_stage1_cuda", &multi_tensor_lamb_stage1_cuda,
        "alias of multi_tensor_lamb_stage1");
  m.def("multi_tensor_lamb_stage2_cuda", &multi_tensor_lamb_stage2_cuda,
        "alias of multi_tensor_lamb_stage2");
  m.def("update_scale_hysteresis", &update_scale_hysteresis_cuda,
        "On-device dynamic loss-scale update with hysteresis");
}
