// Host entry points for the apex_amd._amp_C extension (multi-tensor fused
// kernels for gfx950). Python call signatures mirror the reference amp_C
// module (csrc/amp_C_frontend.cpp:83-123) so the optimizer layer stays
// familiar; the launcher and kernels are MI355X-native (see
// multi_tensor_apply.h).
#pragma once

#include <torch/extension.h>

#include <vector>

using TensorLists = std::vector<std::vector<at::Tensor>>;

void multi_tensor_scale_cuda(long chunk_size, at::Tensor noop_flag,
                             TensorLists tensor_lists, double scale);

void multi_tensor_axpby_cuda(long chunk_size, at::Tensor noop_flag,
                             TensorLists tensor_lists, double a, double b,
                             long arg_to_check);

std::vector<at::Tensor> multi_tensor_l2norm_cuda(long chunk_size, at::Tensor noop_flag,
                                                 TensorLists tensor_lists, bool per_tensor);

std::vector<at::Tensor> multi_tensor_unscale_l2norm_cuda(long chunk_size, at::Tensor noop_flag,
                                                         TensorLists tensor_lists,
                                                         at::Tensor inv_scale, bool per_tensor);

std::vector<at::Tensor> multi_tensor_l2norm_scale_cuda(long chunk_size, at::Tensor noop_flag,
                                                       TensorLists tensor_lists, double scale,
                                                       bool per_tensor);

at::Tensor multi_tensor_maxnorm_cuda(long chunk_size, at::Tensor noop_flag,
                                     TensorLists tensor_lists);

void multi_tensor_sgd_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                           double wd, double momentum, double dampening, double lr,
                           bool nesterov, bool first_run, bool wd_after_momentum, double scale);

void multi_tensor_adam_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                            double lr, double beta1, double beta2, double eps, long step,
                            long mode, long bias_correction, double weight_decay);

void multi_tensor_adam_capturable_cuda(long chunk_size, at::Tensor noop_flag,
                                       TensorLists tensor_lists, at::Tensor lr, double beta1,
                                       double beta2, double eps, at::Tensor step, long mode,
                                       long bias_correction, double weight_decay,
                                       at::Tensor inv_scale);

void multi_tensor_adam_capturable_master_cuda(long chunk_size, at::Tensor noop_flag,
                                              TensorLists tensor_lists, at::Tensor lr,
                                              double beta1, double beta2, double eps,
                                              at::Tensor step, long mode, long bias_correction,
                                              double weight_decay, at::Tensor inv_scale);

void multi_tensor_adagrad_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                               double lr, double eps, long mode, double weight_decay);

void multi_tensor_novograd_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                                at::Tensor per_tensor_v, double lr, double beta1, double beta2,
                                double eps, long step, long bias_correction, double weight_decay,
                                long grad_averaging, long moment_mode, long norm_type);

void multi_tensor_lamb_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                            double lr, double beta1, double beta2, double eps, long step,
                            long bias_correction, double weight_decay, long grad_averaging,
                            long mode, at::Tensor global_grad_norm, double max_grad_norm,
                            bool use_nvlamb);

void multi_tensor_lamb_stage1_cuda(long chunk_size, at::Tensor noop_flag,
                                   TensorLists tensor_lists, double beta1, double beta2,
                                   double eps, long step, long bias_correction,
                                   double weight_decay, long grad_averaging, long mode,
                                   at::Tensor global_grad_norm, double max_grad_norm);

void multi_tensor_lamb_capturable_cuda(long chunk_size, at::Tensor noop_flag,
                                       TensorLists tensor_lists, at::Tensor lr, double beta1,
                                       double beta2, double eps, at::Tensor step,
                                       long bias_correction, double weight_decay,
                                       long grad_averaging, long mode,
                                       at::Tensor global_grad_norm, double max_grad_norm,
                                       bool use_nvlamb);

void multi_tensor_lamb_stage2_cuda(long chunk_size, at::Tensor noop_flag,
                                   TensorLists tensor_lists, at::Tensor param_norms,
                                   at::Tensor update_norms, double lr, double weight_decay,
                                   bool use_nvlamb);

void multi_tensor_lamb_mp_cuda(long chunk_size, at::Tensor noop_flag, TensorLists tensor_lists,
                               at::Tensor lr, double beta1, double beta2, double eps,
                               at::Tensor step, long bias_correction, double weight_decay,
                               long grad_averaging, long mode, at::Tensor global_grad_norm,
                               double max_grad_norm, bool use_nvlamb, at::Tensor found_inf,
                               at::Tensor inv_scale);

void update_scale_hysteresis_cuda(at::Tensor current_scale, at::Tensor growth_tracker,
                                  at::Tensor hysteresis_tracker, at::Tensor found_inf,
                                  double growth_factor, double backoff_factor,
                                  long growth_interval, long hysteresis);
