// apex_amd._rope — fused rotary positional embedding for gfx950.
// Reference surface: csrc/megatron/fused_rotary_positional_embedding.cpp:
// 176-193 (forward/backward for sbhd, cached cos/sin, packed-varlen thd, 2d).
//
// MI355X design: pure elementwise rotate-half passes. cos/sin always come
// from precomputed tables (freqs / cached) — on-device trig would turn this
// memory-bound op VALU-bound on CDNA4 (guide Appendix B). Backward = forward
// with negated sin (the rotation is orthogonal).
#include "common.h"

namespace {

constexpr int RP_BLOCK = 256;

// t: [s, b, h, d]; freqs: [s, 1, 1, d2] fp32 — out = t*cos + rotate_half(t)*sin
template <typename T, bool BWD>
__global__ void __launch_bounds__(RP_BLOCK) rope_sbhd_kernel(
    const T* __restrict__ t, T* __restrict__ out, const float* __restrict__ freqs, long total,
    long bhd, long d, long d2) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long did = i % d;
    if (did >= d2) {
      out[i] = t[i];
      continue;
    }
    const long sid = i / bhd;
    const float f = freqs[sid * d2 + did];
    const float c = __cosf(f);
    const float s = BWD ? -__sinf(f) : __sinf(f);
    const long half = d2 / 2;
    float x = to_float(t[i]);
    float partner = to_float(t[i + (did < half ? half : -half)]);
    float rot = did < half ? -partner : partner;
    out[i] = from_float<T>(fmaf(x, c, rot * s));
  }
}

// cached cos/sin tables [s, 1, 1, d2] in table dtype WT
template <typename T, typename WT, bool BWD>
__global__ void __launch_bounds__(RP_BLOCK) rope_cached_kernel(
    const T* __restrict__ t, T* __restrict__ out, const WT* __restrict__ cos_t,
    const WT* __restrict__ sin_t, long total, long bhd, long d, long d2) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long did = i % d;
    if (did >= d2) {
      out[i] = t[i];
      continue;
    }
    const long sid = i / bhd;
    const float c = to_float(cos_t[sid * d2 + did]);
    float s = to_float(sin_t[sid * d2 + did]);
    if (BWD) s = -s;
    const long half = d2 / 2;
    float x = to_float(t[i]);
    float partner = to_float(t[i + (did < half ? half : -half)]);
    float rot = did < half ? -partner : partner;
    out[i] = from_float<T>(fmaf(x, c, rot * s));
  }
}

// packed varlen: t [total_tokens, h, d]; cu_seqlens [batch+1]; freqs indexed
// by position within each sequence. Each token's seq-position found by
// binary search over cu_seqlens.
template <typename T, bool BWD>
__global__ void __launch_bounds__(RP_BLOCK) rope_thd_kernel(
    const T* __restrict__ t, T* __restrict__ out, const int* __restrict__ cu_seqlens,
    int nseq, const float* __restrict__ freqs, long total, long hd, long d, long d2) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long did = i % d;
    if (did >= d2) {
      out[i] = t[i];
      continue;
    }
    const long tok = i / hd;
    // binary search: largest seq with cu_seqlens[seq] <= tok
    int lo = 0, hi = nseq - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (cu_seqlens[mid] <= tok) lo = mid; else hi = mid - 1;
    }
    const long pos = tok - cu_seqlens[lo];
    const float f = freqs[pos * d2 + did];
    const float c = __cosf(f);
    const float s = BWD ? -__sinf(f) : __sinf(f);
    const long half = d2 / 2;
    float x = to_float(t[i]);
    float partner = to_float(t[i + (did < half ? half : -half)]);
    float rot = did < half ? -partner : partner;
    out[i] = from_float<T>(fmaf(x, c, rot * s));
  }
}

// 2d image grid: t [b, H, W, h, d]; first d/2 rotated by h-freqs, second d/2
// by w-freqs. cos_h/sin_h: [1, maxH, 1, d/2]; cos_w/sin_w: [1, maxW, 1, d/2].
template <typename T, typename WT, bool BWD>
__global__ void __launch_bounds__(RP_BLOCK) rope_2d_kernel(
    const T* __restrict__ t, T* __restrict__ out, const WT* __restrict__ cos_h,
    const WT* __restrict__ sin_h, const WT* __restrict__ cos_w, const WT* __restrict__ sin_w,
    long total, long H, long W, long hd, long d) {
  const long dh = d / 2;  // h-rotated block (itself rotate-halved by dh/2)
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long did = i % d;
    const long cell = i / hd;  // b*H*W + h*W + w position
    const long wpos = cell % W;
    const long hpos = (cell / W) % H;
    float c, s;
    long half;
    long base;  // did offset within its block
    if (did < dh) {
      base = did;
      half = dh / 2;
      c = to_float(cos_h[hpos * dh + base]);
      s = to_float(sin_h[hpos * dh + base]);
    } else {
      base = did - dh;
      half = dh / 2;
      c = to_float(cos_w[wpos * dh + base]);
      s = to_float(sin_w[wpos * dh + base]);
    }
    if (BWD) s = -s;
    float x = to_float(t[i]);
    float partner = to_float(t[i + (base < half ? half : -half)]);
    float rot = base < half ? -partner : partner;
    out[i] = from_float<T>(fmaf(x, c, rot * s));
  }
}

inline int rp_grid(long total) {
  return (int)std::min<long>((total + RP_BLOCK - 1) / RP_BLOCK, 8192);
}

template <bool BWD>
at::Tensor rope_sbhd_impl(const at::Tensor& t_in, const at::Tensor& freqs_in) {
  auto t = t_in.contiguous();
  auto freqs = freqs_in.contiguous().to(at::kFloat);
  auto out = at::empty_like(t);
  const long d = t.size(-1);
  const long d2 = freqs.size(-1);
  const long bhd = t.numel() / t.size(0);
  APEX_DISPATCH_FLOAT_HALF_BF(t.scalar_type(), "fused_rope", ([&] {
    hipLaunchKernelGGL((rope_sbhd_kernel<scalar_t, BWD>), dim3(rp_grid(t.numel())),
                       dim3(RP_BLOCK), 0, current_stream(), (const scalar_t*)t.data_ptr(),
                       (scalar_t*)out.data_ptr(), freqs.data_ptr<float>(), t.numel(), bhd, d, d2);
  }()));
  HIP_CHECK(hipGetLastError());
  return out;
}

template <bool BWD>
at::Tensor rope_cached_impl(const at::Tensor& t_in, const at::Tensor& cos_in,
                            const at::Tensor& sin_in) {
  auto t = t_in.contiguous();
  auto cos_ = cos_in.contiguous();
  auto sin_ = sin_in.contiguous();
  auto out = at::empty_like(t);
  const long d = t.size(-1);
  const long d2 = cos_.size(-1);
  const long bhd = t.numel() / t.size(0);
  APEX_DISPATCH_FLOAT_HALF_BF(t.scalar_type(), "fused_rope_cached", ([&] {
    using in_t = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(cos_.scalar_type(), "fused_rope_cached", ([&] {
      using w_t = scalar_t;
      hipLaunchKernelGGL((rope_cached_kernel<in_t, w_t, BWD>), dim3(rp_grid(t.numel())),
                         dim3(RP_BLOCK), 0, current_stream(), (const in_t*)t.data_ptr(),
                         (in_t*)out.data_ptr(), (const w_t*)cos_.data_ptr(),
                         (const w_t*)sin_.data_ptr(), t.numel(), bhd, d, d2);
    }()));
  }()));
  HIP_CHECK(hipGetLastError());
  return out;
}

template <bool BWD>
at::Tensor rope_thd_impl(const at::Tensor& t_in, const at::Tensor& cu_seqlens,
                         const at::Tensor& freqs_in) {
  auto t = t_in.contiguous();
  auto cu = cu_seqlens.contiguous().to(at::kInt);
  auto freqs = freqs_in.contiguous().to(at::kFloat);
  auto out = at::empty_like(t);
  const long d = t.size(-1);
  const long d2 = freqs.size(-1);
  const long hd = t.numel() / t.size(0);
  APEX_DISPATCH_FLOAT_HALF_BF(t.scalar_type(), "fused_rope_thd", ([&] {
    hipLaunchKernelGGL((rope_thd_kernel<scalar_t, BWD>), dim3(rp_grid(t.numel())),
                       dim3(RP_BLOCK), 0, current_stream(), (const scalar_t*)t.data_ptr(),
                       (scalar_t*)out.data_ptr(), cu.data_ptr<int>(), (int)(cu.numel() - 1),
                       freqs.data_ptr<float>(), t.numel(), hd, d, d2);
  }()));
  HIP_CHECK(hipGetLastError());
  return out;
}

template <bool BWD>
at::Tensor rope_2d_impl(const at::Tensor& t_in, const at::Tensor& cos_h, const at::Tensor& sin_h,
                        const at::Tensor& cos_w, const at::Tensor& sin_w) {
  auto t = t_in.contiguous();  // [b, H, W, h, d]
  TORCH_CHECK(t.dim() == 5, "rope_2d expects [b, H, W, heads, d]");
  auto ch = cos_h.contiguous();
  auto sh = sin_h.contiguous();
  auto cw = cos_w.contiguous();
  auto sw = sin_w.contiguous();
  auto out = at::empty_like(t);
  const long H = t.size(1), W = t.size(2), d = t.size(4);
  const long hd = t.size(3) * d;
  APEX_DISPATCH_FLOAT_HALF_BF(t.scalar_type(), "fused_rope_2d", ([&] {
    using in_t = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(ch.scalar_type(), "fused_rope_2d", ([&] {
      using w_t = scalar_t;
      hipLaunchKernelGGL((rope_2d_kernel<in_t, w_t, BWD>), dim3(rp_grid(t.numel())),
                         dim3(RP_BLOCK), 0, current_stream(), (const in_t*)t.data_ptr(),
                         (in_t*)out.data_ptr(), (const w_t*)ch.data_ptr(),
                         (const w_t*)sh.data_ptr(), (const w_t*)cw.data_ptr(),
                         (const w_t*)sw.data_ptr(), t.numel(), H, W, hd, d);
    }()));
  }()));
  HIP_CHECK(hipGetLastError());
  return out;
}

}  // namespace

at::Tensor rope_forward(at::Tensor t, at::Tensor freqs, bool transpose_output_memory) {
  return rope_sbhd_impl<false>(t, freqs);
}
at::Tensor rope_backward(at::Tensor g, at::Tensor freqs, bool transpose_output_memory) {
  return rope_sbhd_impl<true>(g, freqs);
}
at::Tensor rope_forward_cached(at::Tensor t, at::Tensor cos_, at::Tensor sin_) {
  return rope_cached_impl<false>(t, cos_, sin_);
}
at::Tensor rope_backward_cached(at::Tensor g, at::Tensor cos_, at::Tensor sin_) {
  return rope_cached_impl<true>(g, cos_, sin_);
}
at::Tensor rope_forward_thd(at::Tensor t, at::Tensor cu_seqlens, at::Tensor freqs) {
  return rope_thd_impl<false>(t, cu_seqlens, freqs);
}
at::Tensor rope_backward_thd(at::Tensor g, at::Tensor cu_seqlens, at::Tensor freqs) {
  return rope_thd_impl<true>(g, cu_seqlens, freqs);
}
at::Tensor rope_forward_2d(at::Tensor t, at::Tensor cos_h, at::Tensor sin_h, at::Tensor cos_w,
                           at::Tensor sin_w) {
  return rope_2d_impl<false>(t, cos_h, sin_h, cos_w, sin_w);
}
at::Tensor rope_backward_2d(at::Tensor g, at::Tensor cos_h, at::Tensor sin_h, at::Tensor cos_w,
                            at::Tensor sin_w) {
  return rope_2d_impl<true>(g, cos_h, sin_h, cos_w, sin_w);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("forward", &rope_forward);
  m.def("backward", &rope_backward);
  m.def("forward_cached", &rope_forward_cached);
  m.def("backward_cached", &rope_backward_cached);
  m.def("forward_thd", &rope_forward_thd);
  m.def("backward_thd", &rope_backward_thd);
  m.def("forward_2d", &rope_forward_2d);
  m.def("backward_2d", &rope_backward_2d);
}
