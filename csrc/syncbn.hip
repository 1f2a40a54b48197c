// apex_amd._syncbn — Welford BatchNorm primitives for SyncBatchNorm.
// Reference surface: csrc/syncbn.cpp:72-89 + csrc/welford.cu (welford_mean_var,
// welford_parallel, batchnorm_forward, reduce_bn, batchnorm_backward, and
// NHWC *_c_last variants).
//
// MI355X design: NCHW stats use one workgroup per (channel, slice) with
// lane-local streaming Welford + wave64 Chan merges; NHWC stats give each
// lane ONE channel column (fully coalesced across C) and merge slices in a
// second tiny kernel. Elementwise apply/backward kernels are grid-stride
// with vectorized access where layout permits.
#include "common.h"

#include <vector>

namespace {

constexpr int BN_BLOCK = 256;

struct WelfordData {
  float mean, m2, count;
};

__device__ __forceinline__ void welford_add(float x, float& mean, float& m2, float& count) {
  count += 1.f;
  float delta = x - mean;
  mean += delta / count;
  m2 = fmaf(delta, x - mean, m2);
}

__device__ __forceinline__ void welford_combine(float& mean, float& m2, float& count, float mb,
                                                float m2b, float nb) {
  if (nb == 0.f) return;
  float n = count + nb;
  float delta = mb - mean;
  mean += delta * nb / n;
  m2 += m2b + delta * delta * count * nb / n;
  count = n;
}

// ---- NCHW stats: grid (C, S) ; block reduces its slice of N*HW ----
template <typename T>
__global__ void __launch_bounds__(BN_BLOCK) welford_nchw_kernel(
    const T* __restrict__ x, WelfordData* __restrict__ part, long N, long C, long HW, int S) {
  const int c = blockIdx.x;
  const int s = blockIdx.y;
  const long total = N * HW;
  const long per = (total + S - 1) / S;
  const long lo = (long)s * per;
  const long hi = min(lo + per, total);

  float mean = 0.f, m2 = 0.f, count = 0.f;
  for (long i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    const long n = i / HW, sp = i % HW;
    welford_add(to_float(x[(n * C + c) * HW + sp]), mean, m2, count);
  }
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    float mb = __shfl_xor(mean, off), m2b = __shfl_xor(m2, off), nb = __shfl_xor(count, off);
    welford_combine(mean, m2, count, mb, m2b, nb);
  }
  __shared__ float smem[3 * (BN_BLOCK / WAVE_SIZE)];
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  if (lane == 0) {
    smem[3 * wid] = mean;
    smem[3 * wid + 1] = m2;
    smem[3 * wid + 2] = count;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float tm = 0.f, t2 = 0.f, tc = 0.f;
    for (int i = 0; i < BN_BLOCK / WAVE_SIZE; ++i)
      welford_combine(tm, t2, tc, smem[3 * i], smem[3 * i + 1], smem[3 * i + 2]);
    part[c * S + s] = {tm, t2, tc};
  }
}

// ---- NHWC stats: lane owns one channel column over a slice of rows ----
template <typename T>
__global__ void __launch_bounds__(BN_BLOCK) welford_nhwc_kernel(
    const T* __restrict__ x, WelfordData* __restrict__ part, long rows, long C, int S) {
  const int s = blockIdx.y;
  const long per = (rows + S - 1) / S;
  const long lo = (long)s * per;
  const long hi = min(lo + per, rows);
  if (C < BN_BLOCK && (BN_BLOCK % C) == 0 && gridDim.x == 1) {
    // small-C cooperative form (round-2 fix: one-thread-per-channel left
    // 97% of the machine idle at ResNet's C=64 stages): R = BN_BLOCK/C row
    // lanes share each channel — full coalescing (consecutive threads =
    // consecutive channels within a row) and BN_BLOCK active lanes.
    const int R = BN_BLOCK / (int)C;
    const int c = threadIdx.x % (int)C;
    const int k = threadIdx.x / (int)C;
    float mean = 0.f, m2 = 0.f, count = 0.f;
    for (long r = lo + k; r < hi; r += R)
      welford_add(to_float(x[r * C + c]), mean, m2, count);
    __shared__ WelfordData lanes[BN_BLOCK];
    lanes[threadIdx.x] = {mean, m2, count};
    __syncthreads();
    if (k == 0) {
      for (int kk = 1; kk < R; ++kk) {
        const WelfordData w = lanes[kk * (int)C + c];
        welford_combine(mean, m2, count, w.mean, w.m2, w.count);
      }
      part[(long)c * S + s] = {mean, m2, count};
    }
    return;
  }
  const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = 0.f, m2 = 0.f, count = 0.f;
  for (long r = lo; r < hi; ++r) welford_add(to_float(x[r * C + c]), mean, m2, count);
  part[c * S + s] = {mean, m2, count};
}

// merge S partials per channel -> mean[C], var_biased[C]
__global__ void welford_merge_kernel(const WelfordData* __restrict__ part, float* __restrict__ mean_out,
                                     float* __restrict__ var_out, long C, int S) {
  const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = 0.f, m2 = 0.f, count = 0.f;
  for (int s = 0; s < S; ++s) {
    WelfordData w = part[c * S + s];
    welford_combine(mean, m2, count, w.mean, w.m2, w.count);
  }
  mean_out[c] = mean;
  var_out[c] = count > 0.f ? m2 / count : 0.f;
}

// stats epilogue: invstd = rsqrt(var + eps) and (optionally) the running
// mean/var EMA update — fused so the Python layer launches ONE kernel
// instead of the ~6 eager ops torch would emit per BatchNorm layer (round-2
// fix: the NHWC ResNet step was host-launch-bound, ~16 eager launches x 53
// BN layers).
__global__ void bn_stats_epilogue_kernel(const float* __restrict__ mean,
                                         const float* __restrict__ var,
                                         float* __restrict__ invstd, float eps, long C,
                                         float* __restrict__ running_mean,
                                         float* __restrict__ running_var, float momentum,
                                         float unbiased_factor) {
  const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float v = var[c];
  invstd[c] = rsqrtf(v + eps);
  if (running_mean) {
    running_mean[c] = running_mean[c] * (1.f - momentum) + mean[c] * momentum;
    running_var[c] = running_var[c] * (1.f - momentum) + v * unbiased_factor * momentum;
  }
}

// merge per-process (mean, var_biased, count) rows (the cross-GPU stat merge)
__global__ void welford_parallel_kernel(const float* __restrict__ mean_all,
                                        const float* __restrict__ var_all,
                                        const int* __restrict__ counts, float* __restrict__ mean_out,
                                        float* __restrict__ var_out, long C, int W) {
  const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = 0.f, m2 = 0.f, count = 0.f;
  for (int w = 0; w < W; ++w) {
    float nb = (float)counts[w];
    welford_combine(mean, m2, count, mean_all[w * C + c], var_all[w * C + c] * nb, nb);
  }
  mean_out[c] = mean;
  var_out[c] = count > 0.f ? m2 / count : 0.f;
}

// ---- elementwise forward ----
template <typename T, bool NHWC, bool AFFINE, bool RELU>
__global__ void __launch_bounds__(BN_BLOCK) bn_fwd_kernel(
    const T* __restrict__ x, T* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ w, const float* __restrict__ b,
    long total, long C, long HW) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long c = NHWC ? (i % C) : ((i / HW) % C);
    float r = (to_float(x[i]) - mean[c]) * invstd[c];
    if (AFFINE) r = fmaf(r, w[c], b[c]);
    if (RELU) r = fmaxf(r, 0.f);
    y[i] = from_float<T>(r);
  }
}

// ---- backward reductions: sum_dy, sum_dy_xmu per channel ----
template <typename T, bool NHWC>
__global__ void __launch_bounds__(BN_BLOCK) reduce_bn_partials_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, const float* __restrict__ mean,
    float* __restrict__ part, long N, long C, long HW, int S) {
  // part layout: [C, S, 2]
  if (!NHWC) {
    const int c = blockIdx.x;
    const int s = blockIdx.y;
    const long total = N * HW;
    const long per = (total + S - 1) / S;
    const long lo = (long)s * per, hi = min(lo + per, total);
    float s1 = 0.f, s2 = 0.f;
    const float mu = mean[c];
    for (long i = lo + threadIdx.x; i < hi; i += blockDim.x) {
      const long n = i / HW, sp = i % HW;
      const long idx = (n * C + c) * HW + sp;
      float d = to_float(dy[idx]);
      s1 += d;
      s2 = fmaf(d, to_float(x[idx]) - mu, s2);
    }
    __shared__ float smem[BN_BLOCK / WAVE_SIZE];
    s1 = block_reduce_sum(s1, smem);
    s2 = block_reduce_sum(s2, smem);
    if (threadIdx.x == 0) {
      part[(c * S + s) * 2] = s1;
      part[(c * S + s) * 2 + 1] = s2;
    }
  } else {
    const int s = blockIdx.y;
    const long rows = N * HW;
    const long per = (rows + S - 1) / S;
    const long lo = (long)s * per, hi = min(lo + per, rows);
    if (C < BN_BLOCK && (BN_BLOCK % C) == 0 && gridDim.x == 1) {
      // small-C cooperative form (see welford_nhwc_kernel)
      const int R = BN_BLOCK / (int)C;
      const int c = threadIdx.x % (int)C;
      const int k = threadIdx.x / (int)C;
      const float mu = mean[c];
      float s1 = 0.f, s2 = 0.f;
      for (long r = lo + k; r < hi; r += R) {
        float d = to_float(dy[r * C + c]);
        s1 += d;
        s2 = fmaf(d, to_float(x[r * C + c]) - mu, s2);
      }
      __shared__ float lanes[BN_BLOCK][2];
      lanes[threadIdx.x][0] = s1;
      lanes[threadIdx.x][1] = s2;
      __syncthreads();
      if (k == 0) {
        for (int kk = 1; kk < R; ++kk) {
          s1 += lanes[kk * (int)C + c][0];
          s2 += lanes[kk * (int)C + c][1];
        }
        part[((long)c * S + s) * 2] = s1;
        part[((long)c * S + s) * 2 + 1] = s2;
      }
      return;
    }
    const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float s1 = 0.f, s2 = 0.f;
    const float mu = mean[c];
    for (long r = lo; r < hi; ++r) {
      float d = to_float(dy[r * C + c]);
      s1 += d;
      s2 = fmaf(d, to_float(x[r * C + c]) - mu, s2);
    }
    part[(c * S + s) * 2] = s1;
    part[(c * S + s) * 2 + 1] = s2;
  }
}

__global__ void reduce_bn_merge_kernel(const float* __restrict__ part,
                                       const float* __restrict__ invstd, float* __restrict__ sum_dy,
                                       float* __restrict__ sum_dy_xmu, float* __restrict__ gw,
                                       float* __restrict__ gb, long C, int S) {
  const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s1 = 0.f, s2 = 0.f;
  for (int s = 0; s < S; ++s) {
    s1 += part[(c * S + s) * 2];
    s2 += part[(c * S + s) * 2 + 1];
  }
  sum_dy[c] = s1;
  sum_dy_xmu[c] = s2;
  if (gw) gw[c] = s2 * invstd[c];
  if (gb) gb[c] = s1;
}

// ---- elementwise backward ----
template <typename T, bool NHWC, bool AFFINE>
__global__ void __launch_bounds__(BN_BLOCK) bn_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ w, const float* __restrict__ sum_dy,
    const float* __restrict__ sum_dy_xmu, float inv_count, long total, long C, long HW) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long c = NHWC ? (i % C) : ((i / HW) % C);
    const float iv = invstd[c];
    float d = to_float(dy[i]);
    float r = d - sum_dy[c] * inv_count -
              (to_float(x[i]) - mean[c]) * iv * iv * sum_dy_xmu[c] * inv_count;
    r *= iv * (AFFINE ? w[c] : 1.f);
    dx[i] = from_float<T>(r);
  }
}

// ---------------- host helpers ----------------

void split_dims(const at::Tensor& x, bool nhwc, long& N, long& C, long& HW) {
  if (nhwc) {
    C = x.size(-1);
    N = x.size(0);
    HW = x.numel() / (N * C);
  } else {
    N = x.size(0);
    C = x.size(1);
    HW = x.numel() / (N * C);
  }
}

int pick_splits(long work_per_channel, long C) {
  // enough blocks to fill 256 CUs without over-splitting tiny slices
  long target_blocks = 2048;
  long s = target_blocks / std::max<long>(C / BN_BLOCK + 1, 1);
  s = std::min<long>(s, (work_per_channel + 4095) / 4096);
  return (int)std::max<long>(1, std::min<long>(s, 64));
}

// NHWC stats grids: gridDim.x = ceil(C/BN_BLOCK) blocks of channels, S row
// splits. The small-C cooperative kernels keep all BN_BLOCK lanes busy, so
// split rows until ~2048 workgroups (8 XCDs x 256 CUs need >> 256).
int pick_splits_nhwc(long rows, long C) {
  const long bx = std::max<long>((C + BN_BLOCK - 1) / BN_BLOCK, 1);
  long s = 2048 / bx;
  s = std::min<long>(s, std::max<long>(rows / 256, 1));
  return (int)std::max<long>(1, std::min<long>(s, 2048));
}

std::vector<at::Tensor> welford_impl(const at::Tensor& input, bool nhwc) {
  auto x = input.contiguous();
  long N, C, HW;
  split_dims(x, nhwc, N, C, HW);
  auto fopts = at::TensorOptions().dtype(at::kFloat).device(x.device());
  auto mean = at::empty({C}, fopts);
  auto var = at::empty({C}, fopts);
  auto stream = current_stream();

  if (!nhwc) {
    const int S = pick_splits(N * HW, C);
    auto part = at::empty({C, S, 3}, fopts);
    APEX_DISPATCH_FLOAT_HALF_BF(x.scalar_type(), "welford_mean_var", ([&] {
      hipLaunchKernelGGL((welford_nchw_kernel<scalar_t>), dim3((uint32_t)C, S), dim3(BN_BLOCK), 0,
                         stream, (const scalar_t*)x.data_ptr(), (WelfordData*)part.data_ptr(), N,
                         C, HW, S);
    }()));
    HIP_CHECK(hipGetLastError());
    hipLaunchKernelGGL(welford_merge_kernel, dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK)),
                       dim3(BN_BLOCK), 0, stream, (const WelfordData*)part.data_ptr(),
                       mean.data_ptr<float>(), var.data_ptr<float>(), C, S);
    HIP_CHECK(hipGetLastError());
  } else {
    const long rows = N * HW;
    const int S = pick_splits_nhwc(rows, C);
    auto part = at::empty({C, S, 3}, fopts);
    APEX_DISPATCH_FLOAT_HALF_BF(x.scalar_type(), "welford_mean_var_c_last", ([&] {
      hipLaunchKernelGGL((welford_nhwc_kernel<scalar_t>),
                         dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK), S), dim3(BN_BLOCK), 0,
                         stream, (const scalar_t*)x.data_ptr(), (WelfordData*)part.data_ptr(),
                         rows, C, S);
    }()));
    HIP_CHECK(hipGetLastError());
    hipLaunchKernelGGL(welford_merge_kernel, dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK)),
                       dim3(BN_BLOCK), 0, stream, (const WelfordData*)part.data_ptr(),
                       mean.data_ptr<float>(), var.data_ptr<float>(), C, S);
    HIP_CHECK(hipGetLastError());
  }
  return {mean, var};
}

at::Tensor bn_fwd_impl(const at::Tensor& input, const at::Tensor& mean, const at::Tensor& invstd,
                       const c10::optional<at::Tensor>& weight,
                       const c10::optional<at::Tensor>& bias, bool nhwc, bool relu) {
  auto x = input.contiguous();
  long N, C, HW;
  split_dims(x, nhwc, N, C, HW);
  auto y = at::empty_like(x);
  const long total = x.numel();
  const int grid = (int)std::min<long>((total + BN_BLOCK - 1) / BN_BLOCK, 8192);
  const bool affine = weight.has_value() && weight->defined();
  auto w32 = affine ? weight->to(at::kFloat) : at::Tensor();
  auto b32 = affine ? bias->to(at::kFloat) : at::Tensor();
  auto stream = current_stream();

  APEX_DISPATCH_FLOAT_HALF_BF(x.scalar_type(), "batchnorm_forward", ([&] {
    auto launch = [&](auto nhwc_c, auto aff_c, auto relu_c) {
      hipLaunchKernelGGL((bn_fwd_kernel<scalar_t, decltype(nhwc_c)::value, decltype(aff_c)::value,
                                        decltype(relu_c)::value>),
                         dim3(grid), dim3(BN_BLOCK), 0, stream, (const scalar_t*)x.data_ptr(),
                         (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), affine ? w32.data_ptr<float>() : nullptr,
                         affine ? b32.data_ptr<float>() : nullptr, total, C, HW);
    };
    using T = std::true_type;
    using F = std::false_type;
    if (nhwc && affine && relu) launch(T{}, T{}, T{});
    else if (nhwc && affine) launch(T{}, T{}, F{});
    else if (nhwc && relu) launch(T{}, F{}, T{});
    else if (nhwc) launch(T{}, F{}, F{});
    else if (affine && relu) launch(F{}, T{}, T{});
    else if (affine) launch(F{}, T{}, F{});
    else if (relu) launch(F{}, F{}, T{});
    else launch(F{}, F{}, F{});
  }()));
  HIP_CHECK(hipGetLastError());
  return y;
}

std::vector<at::Tensor> reduce_bn_impl(const at::Tensor& grad_out, const at::Tensor& input,
                                       const at::Tensor& mean, const at::Tensor& invstd,
                                       const c10::optional<at::Tensor>& weight, bool nhwc) {
  auto dy = grad_out.contiguous();
  auto x = input.contiguous();
  long N, C, HW;
  split_dims(x, nhwc, N, C, HW);
  auto fopts = at::TensorOptions().dtype(at::kFloat).device(x.device());
  auto sum_dy = at::empty({C}, fopts);
  auto sum_dy_xmu = at::empty({C}, fopts);
  const bool affine = weight.has_value() && weight->defined();
  auto gw = affine ? at::empty({C}, fopts) : at::Tensor();
  auto gb = affine ? at::empty({C}, fopts) : at::Tensor();
  auto stream = current_stream();

  const int S = nhwc ? pick_splits_nhwc(N * HW, C) : pick_splits(N * HW, C);
  auto part = at::empty({C, S, 2}, fopts);

  APEX_DISPATCH_FLOAT_HALF_BF(x.scalar_type(), "reduce_bn", ([&] {
    if (nhwc) {
      hipLaunchKernelGGL((reduce_bn_partials_kernel<scalar_t, true>),
                         dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK), S), dim3(BN_BLOCK), 0,
                         stream, (const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
                         mean.data_ptr<float>(), part.data_ptr<float>(), N, C, HW, S);
    } else {
      hipLaunchKernelGGL((reduce_bn_partials_kernel<scalar_t, false>), dim3((uint32_t)C, S),
                         dim3(BN_BLOCK), 0, stream, (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)x.data_ptr(), mean.data_ptr<float>(),
                         part.data_ptr<float>(), N, C, HW, S);
    }
  }()));
  HIP_CHECK(hipGetLastError());
  hipLaunchKernelGGL(reduce_bn_merge_kernel, dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK)),
                     dim3(BN_BLOCK), 0, stream, part.data_ptr<float>(), invstd.data_ptr<float>(),
                     sum_dy.data_ptr<float>(), sum_dy_xmu.data_ptr<float>(),
                     affine ? gw.data_ptr<float>() : nullptr,
                     affine ? gb.data_ptr<float>() : nullptr, C, S);
  HIP_CHECK(hipGetLastError());
  return {sum_dy, sum_dy_xmu, gw, gb};
}

at::Tensor bn_bwd_impl(const at::Tensor& grad_out, const at::Tensor& input,
                       const at::Tensor& mean, const at::Tensor& invstd,
                       const c10::optional<at::Tensor>& weight, const at::Tensor& sum_dy,
                       const at::Tensor& sum_dy_xmu, long count, bool nhwc) {
  auto dy = grad_out.contiguous();
  auto x = input.contiguous();
  long N, C, HW;
  split_dims(x, nhwc, N, C, HW);
  auto dx = at::empty_like(x);
  const long total = x.numel();
  const int grid = (int)std::min<long>((total + BN_BLOCK - 1) / BN_BLOCK, 8192);
  const bool affine = weight.has_value() && weight->defined();
  auto w32 = affine ? weight->to(at::kFloat) : at::Tensor();
  auto sdy = sum_dy.to(at::kFloat);
  auto sdyx = sum_dy_xmu.to(at::kFloat);
  auto stream = current_stream();

  APEX_DISPATCH_FLOAT_HALF_BF(x.scalar_type(), "batchnorm_backward", ([&] {
    auto launch = [&](auto nhwc_c, auto aff_c) {
      hipLaunchKernelGGL((bn_bwd_kernel<scalar_t, decltype(nhwc_c)::value, decltype(aff_c)::value>),
                         dim3(grid), dim3(BN_BLOCK), 0, stream, (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)x.data_ptr(), (scalar_t*)dx.data_ptr(),
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         affine ? w32.data_ptr<float>() : nullptr, sdy.data_ptr<float>(),
                         sdyx.data_ptr<float>(), 1.0f / (float)count, total, C, HW);
    };
    using T = std::true_type;
    using F = std::false_type;
    if (nhwc && affine) launch(T{}, T{});
    else if (nhwc) launch(T{}, F{});
    else if (affine) launch(F{}, T{});
    else launch(F{}, F{});
  }()));
  HIP_CHECK(hipGetLastError());
  return dx;
}

}  // namespace

// ---------------- bindings (match apex_amd.parallel.sync_batchnorm) ----------------

std::vector<at::Tensor> welford_mean_var(at::Tensor input) { return welford_impl(input, false); }
std::vector<at::Tensor> welford_mean_var_c_last(at::Tensor input) { return welford_impl(input, true); }

// fused stats: welford + invstd + running-stat EMA in-kernel.
// Returns (mean, var_biased, invstd); running stats updated in place when
// given (count = elements per channel for the unbiased correction).
std::vector<at::Tensor> bn_stats(at::Tensor input, bool nhwc, double eps,
                                 c10::optional<at::Tensor> running_mean,
                                 c10::optional<at::Tensor> running_var, double momentum,
                                 long total_count) {
  auto mv = welford_impl(input, nhwc);
  auto mean = mv[0];
  auto var = mv[1];
  const long C = mean.numel();
  auto invstd = at::empty_like(mean);
  const bool track = running_mean.has_value() && running_mean->defined();
  TORCH_CHECK(!track || (running_mean->scalar_type() == at::ScalarType::Float &&
                         running_var->scalar_type() == at::ScalarType::Float),
              "bn_stats: running stats must be fp32");
  const float unbiased = total_count > 1 ? (float)total_count / (float)(total_count - 1) : 1.f;
  hipLaunchKernelGGL(bn_stats_epilogue_kernel, dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK)),
                     dim3(BN_BLOCK), 0, current_stream(), mean.data_ptr<float>(),
                     var.data_ptr<float>(), invstd.data_ptr<float>(), (float)eps, C,
                     track ? running_mean->data_ptr<float>() : nullptr,
                     track ? running_var->data_ptr<float>() : nullptr, (float)momentum,
                     unbiased);
  HIP_CHECK(hipGetLastError());
  return {mean, var, invstd};
}

std::vector<at::Tensor> welford_parallel(at::Tensor mean_all, at::Tensor var_all,
                                         at::Tensor counts);

// fused cross-process variant: welford_parallel merge + invstd + running EMA
std::vector<at::Tensor> bn_stats_parallel(at::Tensor mean_all, at::Tensor var_all,
                                          at::Tensor counts, double eps,
                                          c10::optional<at::Tensor> running_mean,
                                          c10::optional<at::Tensor> running_var,
                                          double momentum, long total_count) {
  auto mv = welford_parallel(mean_all, var_all, counts);
  auto mean = mv[0];
  auto var = mv[1];
  const long C = mean.numel();
  auto invstd = at::empty_like(mean);
  const bool track = running_mean.has_value() && running_mean->defined();
  const float unbiased = total_count > 1 ? (float)total_count / (float)(total_count - 1) : 1.f;
  hipLaunchKernelGGL(bn_stats_epilogue_kernel, dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK)),
                     dim3(BN_BLOCK), 0, current_stream(), mean.data_ptr<float>(),
                     var.data_ptr<float>(), invstd.data_ptr<float>(), (float)eps, C,
                     track ? running_mean->data_ptr<float>() : nullptr,
                     track ? running_var->data_ptr<float>() : nullptr, (float)momentum,
                     unbiased);
  HIP_CHECK(hipGetLastError());
  return {mean, var, invstd};
}

std::vector<at::Tensor> welford_parallel(at::Tensor mean_all, at::Tensor var_all,
                                         at::Tensor counts) {
  TORCH_CHECK(mean_all.dim() == 2, "welford_parallel expects [world, C]");
  const long W = mean_all.size(0), C = mean_all.size(1);
  auto fopts = at::TensorOptions().dtype(at::kFloat).device(mean_all.device());
  auto mean = at::empty({C}, fopts);
  auto var = at::empty({C}, fopts);
  auto m = mean_all.contiguous().to(at::kFloat);
  auto v = var_all.contiguous().to(at::kFloat);
  auto c = counts.contiguous().to(at::kInt);
  hipLaunchKernelGGL(welford_parallel_kernel, dim3((uint32_t)((C + BN_BLOCK - 1) / BN_BLOCK)),
                     dim3(BN_BLOCK), 0, current_stream(), m.data_ptr<float>(), v.data_ptr<float>(),
                     c.data_ptr<int>(), mean.data_ptr<float>(), var.data_ptr<float>(), C, (int)W);
  HIP_CHECK(hipGetLastError());
  return {mean, var};
}

at::Tensor batchnorm_forward(at::Tensor input, at::Tensor mean, at::Tensor invstd,
                             c10::optional<at::Tensor> weight, c10::optional<at::Tensor> bias) {
  return bn_fwd_impl(input, mean, invstd, weight, bias, false, false);
}

at::Tensor batchnorm_forward_c_last(at::Tensor input, at::Tensor mean, at::Tensor invstd,
                                    c10::optional<at::Tensor> weight,
                                    c10::optional<at::Tensor> bias, bool fuse_relu) {
  return bn_fwd_impl(input, mean, invstd, weight, bias, true, fuse_relu);
}

std::vector<at::Tensor> reduce_bn(at::Tensor grad_out, at::Tensor input, at::Tensor mean,
                                  at::Tensor invstd, c10::optional<at::Tensor> weight) {
  return reduce_bn_impl(grad_out, input, mean, invstd, weight, false);
}

std::vector<at::Tensor> reduce_bn_c_last(at::Tensor grad_out, at::Tensor input, at::Tensor mean,
                                         at::Tensor invstd, c10::optional<at::Tensor> weight) {
  return reduce_bn_impl(grad_out, input, mean, invstd, weight, true);
}

at::Tensor batchnorm_backward(at::Tensor grad_out, at::Tensor input, at::Tensor mean,
                              at::Tensor invstd, c10::optional<at::Tensor> weight,
                              at::Tensor sum_dy, at::Tensor sum_dy_xmu, long count) {
  return bn_bwd_impl(grad_out, input, mean, invstd, weight, sum_dy, sum_dy_xmu, count, false);
}

at::Tensor batchnorm_backward_c_last(at::Tensor grad_out, at::Tensor input, at::Tensor mean,
                                     at::Tensor invstd, c10::optional<at::Tensor> weight,
                                     at::Tensor sum_dy, at::Tensor sum_dy_xmu, long count) {
  return bn_bwd_impl(grad_out, input, mean, invstd, weight, sum_dy, sum_dy_xmu, count, true);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("welford_mean_var", &welford_mean_var, "per-channel Welford mean/biased-var (NCHW)");
  m.def("welford_mean_var_c_last", &welford_mean_var_c_last, "NHWC variant");
  m.def("welford_parallel", &welford_parallel, "merge per-process (mean,var,count) rows");
  m.def("bn_stats", &bn_stats,
        "fused welford + invstd + running-EMA -> (mean, var_biased, invstd)");
  m.def("bn_stats_parallel", &bn_stats_parallel,
        "fused cross-process merge + invstd + running-EMA");
  m.def("batchnorm_forward", &batchnorm_forward, "BN apply (NCHW)");
  m.def("batchnorm_forward_c_last", &batchnorm_forward_c_last, "BN apply (NHWC, optional ReLU)");
  m.def("reduce_bn", &reduce_bn, "per-channel sum_dy/sum_dy_xmu + weight/bias grads (NCHW)");
  m.def("reduce_bn_c_last", &reduce_bn_c_last, "NHWC variant");
  m.def("batchnorm_backward", &batchnorm_backward, "BN grad_input (NCHW)");
  m.def("batchnorm_backward_c_last", &batchnorm_backward_c_last, "NHWC variant");
}
