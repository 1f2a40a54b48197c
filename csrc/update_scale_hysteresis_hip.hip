#include "hip/hip_runtime.h"
// On-device dynamic loss-scale update with hysteresis.
// Reference behavior: csrc/update_scale_hysteresis.cu:5-47 — a single-
// workitem kernel so the scale state never round-trips to the host: backoff
// only after `hysteresis` consecutive overflow steps, growth every
// `growth_interval` clean steps, never grow into inf.
#include "amp_C.h"
#include "common.h"

namespace {

__global__ void update_scale_hysteresis_kernel(float* current_scale, int* growth_tracker,
                                               int* hysteresis_tracker, const int* found_inf,
                                               float growth_factor, float backoff_factor,
                                               int growth_interval, int hysteresis) {
  if (*found_inf) {
    *growth_tracker = 0;
    (*hysteresis_tracker)--;
    if (*hysteresis_tracker <= 0) {
      *current_scale = (*current_scale) * backoff_factor;
      *hysteresis_tracker = hysteresis;
    }
  } else {
    int g = (*growth_tracker) + 1;
    if (g == growth_interval) {
      float new_scale = (*current_scale) * growth_factor;
      if (isfinite(new_scale)) *current_scale = new_scale;
      g = 0;
    }
    *growth_tracker = g;
    *hysteresis_tracker = hysteresis;
  }
}

}  // namespace

void update_scale_hysteresis_cuda(at::Tensor current_scale, at::Tensor growth_tracker,
                                  at::Tensor hysteresis_tracker, at::Tensor found_inf,
                                  double growth_factor, double backoff_factor,
                                  long growth_interval, long hysteresis) {
  hipLaunchKernelGGL(update_scale_hysteresis_kernel, dim3(1), dim3(1), 0, current_stream(),
                     current_scale.data_ptr<float>(), growth_tracker.data_ptr<int>(),
                     hysteresis_tracker.data_ptr<int>(), found_inf.data_ptr<int>(),
                     (float)growth_factor, (float)backoff_factor, (int)growth_interval,
                     (int)hysteresis);
  HIP_CHECK(hipGetLastError());
}
