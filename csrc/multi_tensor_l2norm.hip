// multi_tensor_l2norm family — fused norms over tensor lists.
// Reference behavior: csrc/multi_tensor_l2norm_kernel.cu (L2NormFunctor,
// UnscaleL2NormFunctor, MaxNormFunctor, cleanup kernels) and
// csrc/multi_tensor_l2norm_scale_kernel.cu.
//
// MI355X design: every chunk gets a workgroup; each workgroup writes one
// fp32 partial (wave64 shuffle + LDS tree reduction) to
// partials[global_chunk]. A deterministic cleanup kernel then reduces the
// per-tensor ranges and the full buffer in a FIXED tree order (bitwise
// reproducible across runs — required by the L1 determinism harness; fp32
// atomics would not be).
#include "amp_C.h"

#include <mutex>
#include <string>
#include <unordered_map>
#include "multi_tensor_apply.h"

namespace {

template <typename in_t, bool kMax>
struct NormFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<1>& meta, int t, long chunk,
                             float* partials, const float* inv_scale) const {
    const long base = chunk * chunk_size;
    const in_t* in = reinterpret_cast<const in_t*>(meta.addrs[0][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);
    const float s = inv_scale ? *inv_scale : 1.0f;

    float acc = kMax ? 0.f : 0.f;
    if (is_vec4_aligned<in_t>(in) && (n & (MTA_ILP - 1)) == 0) {
      for (long i = (long)threadIdx.x * MTA_ILP; i < n; i += (long)blockDim.x * MTA_ILP) {
        Vec4<in_t> v;
        load_vec4(v, in + i);
#pragma unroll
        for (int j = 0; j < MTA_ILP; ++j) {
          float f = to_float(v.a[j]) * s;
          acc = kMax ? fmaxf(acc, fabsf(f)) : fmaf(f, f, acc);
        }
      }
    } else {
      for (long i = threadIdx.x; i < n; i += blockDim.x) {
        float f = to_float(in[i]) * s;
        acc = kMax ? fmaxf(acc, fabsf(f)) : fmaf(f, f, acc);
      }
    }
    __shared__ float smem[MTA_BLOCK / WAVE_SIZE];
    float r = kMax ? block_reduce_max(acc, smem) : block_reduce_sum(acc, smem);
    if (threadIdx.x == 0) partials[meta.chunk_offset + blockIdx.x] = r;
  }
};

// Fused scale+norm: out = in*scale, partials accumulate (in*scale)^2.
template <typename in_t, typename out_t>
struct L2NormScaleFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<2>& meta, int t, long chunk,
                             float* partials, float scale) const {
    const long base = chunk * chunk_size;
    const in_t* in = reinterpret_cast<const in_t*>(meta.addrs[0][t]) + base;
    out_t* out = reinterpret_cast<out_t*>(meta.addrs[1][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);

    float acc = 0.f;
    bool finite = true;
    if (is_vec4_aligned<in_t>(in) && is_vec4_aligned<out_t>(out) && (n & (MTA_ILP - 1)) == 0) {
      for (long i = (long)threadIdx.x * MTA_ILP; i < n; i += (long)blockDim.x * MTA_ILP) {
        Vec4<in_t> v;
        load_vec4(v, in + i);
        Vec4<out_t> vo;
#pragma unroll
        for (int j = 0; j < MTA_ILP; ++j) {
          float f = to_float(v.a[j]) * scale;
          finite &= isfinite(f);
          acc = fmaf(f, f, acc);
          vo.a[j] = from_float<out_t>(f);
        }
        store_vec4(out + i, vo);
      }
    } else {
      for (long i = threadIdx.x; i < n; i += blockDim.x) {
        float f = to_float(in[i]) * scale;
        finite &= isfinite(f);
        acc = fmaf(f, f, acc);
        out[i] = from_float<out_t>(f);
      }
    }
    if (!finite) *noop = 1;
    __shared__ float smem[MTA_BLOCK / WAVE_SIZE];
    float r = block_reduce_sum(acc, smem);
    if (threadIdx.x == 0) partials[meta.chunk_offset + blockIdx.x] = r;
  }
};

// Deterministic cleanup: block i < ntensors reduces partials[pfx[i]..pfx[i+1])
// -> per_tensor_out[i] = sqrt(sum) (or max). Block ntensors reduces the whole
// buffer -> out[0]. Fixed tree order => bitwise-stable.
template <bool kMax>
__global__ void __launch_bounds__(MTA_BLOCK) norm_cleanup_kernel(
    const float* partials, const int* prefix, int ntensors, long total_chunks,
    float* global_out, float* per_tensor_out) {
  __shared__ float smem[MTA_BLOCK / WAVE_SIZE];
  const int b = blockIdx.x;
  long lo, hi;
  float* dst;
  if (b < ntensors) {
    if (per_tensor_out == nullptr) return;
    lo = prefix[b];
    hi = prefix[b + 1];
    dst = per_tensor_out + b;
  } else {
    lo = 0;
    hi = total_chunks;
    dst = global_out;
  }
  float acc = kMax ? 0.f : 0.f;
  for (long i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    acc = kMax ? fmaxf(acc, partials[i]) : acc + partials[i];
  }
  float r = kMax ? block_reduce_max(acc, smem) : block_reduce_sum(acc, smem);
  if (threadIdx.x == 0) *dst = kMax ? r : sqrtf(r);
}

// host helper: global chunk prefix per tensor -> device int tensor.
// CACHED by (device, chunk_size, sizes): the prefix depends only on tensor
// sizes, and the H2D upload must NOT happen inside a hipGraph capture — a
// captured pageable-host copy replays against freed host memory (this was a
// GPU memory fault in the captured FusedLAMB step). The first (eager,
// warmup) call uploads; captured calls reuse the resident device tensor.
at::Tensor make_prefix(const std::vector<at::Tensor>& ts, long chunk_size, at::Device dev) {
  static std::mutex mu;
  static std::unordered_map<std::string, at::Tensor> cache;
  std::string key = std::to_string((int)dev.index()) + ":" + std::to_string(chunk_size);
  for (auto& t : ts) {
    key += ',';
    key += std::to_string(t.numel());
  }
  {
    std::lock_guard<std::mutex> g(mu);
    auto it = cache.find(key);
    if (it != cache.end()) return it->second;
  }
  std::vector<int> pfx(ts.size() + 1, 0);
  long c = 0;
  for (size_t i = 0; i < ts.size(); ++i) {
    c += (ts[i].numel() + chunk_size - 1) / chunk_size;
    pfx[i + 1] = (int)c;
  }
  auto cpu = at::from_blob(pfx.data(), {(long)pfx.size()}, at::kInt).clone();
  auto devt = cpu.to(dev);
  std::lock_guard<std::mutex> g(mu);
  cache.emplace(key, devt);
  return devt;
}

std::vector<at::Tensor> l2norm_impl(long chunk_size, at::Tensor noop_flag,
                                    TensorLists tensor_lists, bool per_tensor,
                                    const float* inv_scale_ptr, at::Tensor* inv_scale_keepalive) {
  auto& ts = tensor_lists[0];
  const auto dev = ts[0].device();
  const long total_chunks = mta_total_chunks(ts, chunk_size);
  auto opts = at::TensorOptions().dtype(at::kFloat).device(dev);
  auto partials = at::empty({total_chunks}, opts);
  auto global_out = at::zeros({1}, opts);
  auto per_tensor_out = per_tensor ? at::empty({(long)ts.size()}, opts) : at::empty({0}, opts);

  APEX_DISPATCH_FLOAT_HALF_BF(ts[0].scalar_type(), "multi_tensor_l2norm", ([&] {
    multi_tensor_apply<1>(chunk_size, noop_flag, tensor_lists, NormFunctor<scalar_t, false>(),
                          partials.data_ptr<float>(), inv_scale_ptr);
  }()));

  at::Tensor prefix = make_prefix(ts, chunk_size, dev);
  const int nblocks = (int)ts.size() + 1;
  hipLaunchKernelGGL((norm_cleanup_kernel<false>), dim3(nblocks), dim3(MTA_BLOCK), 0,
                     current_stream(), partials.data_ptr<float>(), prefix.data_ptr<int>(),
                     (int)ts.size(), total_chunks, global_out.data_ptr<float>(),
                     per_tensor ? per_tensor_out.data_ptr<float>() : nullptr);
  HIP_CHECK(hipGetLastError());
  return {global_out, per_tensor_out};
}

}  // namespace

std::vector<at::Tensor> multi_tensor_l2norm_cuda(long chunk_size, at::Tensor noop_flag,
                                                 TensorLists tensor_lists, bool per_tensor) {
  return l2norm_impl(chunk_size, noop_flag, tensor_lists, per_tensor, nullptr, nullptr);
}

std::vector<at::Tensor> multi_tensor_unscale_l2norm_cuda(long chunk_size, at::Tensor noop_flag,
                                                         TensorLists tensor_lists,
                                                         at::Tensor inv_scale, bool per_tensor) {
  return l2norm_impl(chunk_size, noop_flag, tensor_lists, per_tensor,
                     inv_scale.data_ptr<float>(), &inv_scale);
}

std::vector<at::Tensor> multi_tensor_l2norm_scale_cuda(long chunk_size, at::Tensor noop_flag,
                                                       TensorLists tensor_lists, double scale,
                                                       bool per_tensor) {
  auto& ts = tensor_lists[0];
  const auto dev = ts[0].device();
  const long total_chunks = mta_total_chunks(ts, chunk_size);
  auto opts = at::TensorOptions().dtype(at::kFloat).device(dev);
  auto partials = at::empty({total_chunks}, opts);
  auto global_out = at::zeros({1}, opts);
  auto per_tensor_out = per_tensor ? at::empty({(long)ts.size()}, opts) : at::empty({0}, opts);

  APEX_DISPATCH_FLOAT_HALF_BF(ts[0].scalar_type(), "multi_tensor_l2norm_scale", ([&] {
    using in_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(tensor_lists[1][0].scalar_type(), "multi_tensor_l2norm_scale", ([&] {
      using out_scalar = scalar_t;
      multi_tensor_apply<2>(chunk_size, noop_flag, tensor_lists,
                            L2NormScaleFunctor<in_scalar, out_scalar>(),
                            partials.data_ptr<float>(), (float)scale);
    }()));
  }()));

  at::Tensor prefix = make_prefix(ts, chunk_size, dev);
  hipLaunchKernelGGL((norm_cleanup_kernel<false>), dim3((int)ts.size() + 1), dim3(MTA_BLOCK), 0,
                     current_stream(), partials.data_ptr<float>(), prefix.data_ptr<int>(),
                     (int)ts.size(), total_chunks, global_out.data_ptr<float>(),
                     per_tensor ? per_tensor_out.data_ptr<float>() : nullptr);
  HIP_CHECK(hipGetLastError());
  return {global_out, per_tensor_out};
}

at::Tensor multi_tensor_maxnorm_cuda(long chunk_size, at::Tensor noop_flag,
                                     TensorLists tensor_lists) {
  auto& ts = tensor_lists[0];
  const auto dev = ts[0].device();
  const long total_chunks = mta_total_chunks(ts, chunk_size);
  auto opts = at::TensorOptions().dtype(at::kFloat).device(dev);
  auto partials = at::empty({total_chunks}, opts);
  auto global_out = at::zeros({1}, opts);
  auto per_tensor_out = at::empty({(long)ts.size()}, opts);

  APEX_DISPATCH_FLOAT_HALF_BF(ts[0].scalar_type(), "multi_tensor_maxnorm", ([&] {
    multi_tensor_apply<1>(chunk_size, noop_flag, tensor_lists, NormFunctor<scalar_t, true>(),
                          partials.data_ptr<float>(), (const float*)nullptr);
  }()));

  at::Tensor prefix = make_prefix(ts, chunk_size, dev);
  hipLaunchKernelGGL((norm_cleanup_kernel<true>), dim3((int)ts.size() + 1), dim3(MTA_BLOCK), 0,
                     current_stream(), partials.data_ptr<float>(), prefix.data_ptr<int>(),
                     (int)ts.size(), total_chunks, global_out.data_ptr<float>(),
                     per_tensor_out.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return per_tensor_out;
}
