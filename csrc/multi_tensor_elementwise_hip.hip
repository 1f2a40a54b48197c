// multi_tensor_scale / multi_tensor_axpby — fused elementwise passes with
// isfinite tracking (the amp unscale path).
// Reference behavior: csrc/multi_tensor_scale_kernel.cu:29-94,
// csrc/multi_tensor_axpby_kernel.cu:28-100. MI355X implementation: wave64
// blocks, 8/16-byte vector loads (G13), one launch for all chunks.
#include "amp_C.h"
#include "multi_tensor_apply_hip.h"

namespace {

template <typename in_t, typename out_t>
struct ScaleFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<2>& meta, int t, long chunk,
                             float scale) const {
    const long base = chunk * chunk_size;
    const in_t* in = reinterpret_cast<const in_t*>(meta.addrs[0][t]) + base;
    out_t* out = reinterpret_cast<out_t*>(meta.addrs[1][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);
    bool finite = true;

    if (is_vec4_aligned<in_t>(in) && is_vec4_aligned<out_t>(out) && (n & (MTA_ILP - 1)) == 0) {
      for (long i = (long)threadIdx.x * MTA_ILP; i < n; i += (long)blockDim.x * MTA_ILP) {
        Vec4<in_t> vi;
        load_vec4(vi, in + i);
        Vec4<out_t> vo;
#pragma unroll
        for (int j = 0; j < MTA_ILP; ++j) {
          float f = to_float(vi.a[j]) * scale;
          finite &= isfinite(f);
          vo.a[j] = from_float<out_t>(f);
        }
        store_vec4(out + i, vo);
      }
    } else {
      for (long i = threadIdx.x; i < n; i += blockDim.x) {
        float f = to_float(in[i]) * scale;
        finite &= isfinite(f);
        out[i] = from_float<out_t>(f);
      }
    }
    if (!finite) *noop = 1;  // racy write, by design (reference :92)
  }
};

template <typename x_t, typename y_t, typename out_t>
struct AxpbyFunctor {
  __device__ void operator()(long chunk_size, volatile int* noop,
                             const TensorListMeta<3>& meta, int t, long chunk,
                             float a, float b, int arg_to_check) const {
    const long base = chunk * chunk_size;
    const x_t* x = reinterpret_cast<const x_t*>(meta.addrs[0][t]) + base;
    const y_t* y = reinterpret_cast<const y_t*>(meta.addrs[1][t]) + base;
    out_t* out = reinterpret_cast<out_t*>(meta.addrs[2][t]) + base;
    const long n = min(meta.sizes[t] - base, chunk_size);
    bool finite = true;

    if (is_vec4_aligned<x_t>(x) && is_vec4_aligned<y_t>(y) && is_vec4_aligned<out_t>(out) &&
        (n & (MTA_ILP - 1)) == 0) {
      for (long i = (long)threadIdx.x * MTA_ILP; i < n; i += (long)blockDim.x * MTA_ILP) {
        Vec4<x_t> vx;
        Vec4<y_t> vy;
        load_vec4(vx, x + i);
        load_vec4(vy, y + i);
        Vec4<out_t> vo;
#pragma unroll
        for (int j = 0; j < MTA_ILP; ++j) {
          float fx = to_float(vx.a[j]), fy = to_float(vy.a[j]);
          if (arg_to_check == -1) finite &= (isfinite(fx) && isfinite(fy));
          if (arg_to_check == 0) finite &= isfinite(fx);
          if (arg_to_check == 1) finite &= isfinite(fy);
          vo.a[j] = from_float<out_t>(a * fx + b * fy);
        }
        store_vec4(out + i, vo);
      }
    } else {
      for (long i = threadIdx.x; i < n; i += blockDim.x) {
        float fx = to_float(x[i]), fy = to_float(y[i]);
        if (arg_to_check == -1) finite &= (isfinite(fx) && isfinite(fy));
        if (arg_to_check == 0) finite &= isfinite(fx);
        if (arg_to_check == 1) finite &= isfinite(fy);
        out[i] = from_float<out_t>(a * fx + b * fy);
      }
    }
    if (!finite) *noop = 1;
  }
};

}  // namespace

void multi_tensor_scale_cuda(long chunk_size, at::Tensor noop_flag,
                             TensorLists tensor_lists, double scale) {
  const auto in_t = tensor_lists[0][0].scalar_type();
  const auto out_t = tensor_lists[1][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(in_t, "multi_tensor_scale", ([&] {
    using in_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(out_t, "multi_tensor_scale", ([&] {
      using out_scalar = scalar_t;
      multi_tensor_apply<2>(chunk_size, noop_flag, tensor_lists,
                            ScaleFunctor<in_scalar, out_scalar>(), (float)scale);
    }()));
  }()));
}

void multi_tensor_axpby_cuda(long chunk_size, at::Tensor noop_flag,
                             TensorLists tensor_lists, double a, double b,
                             long arg_to_check) {
  const auto x_t = tensor_lists[0][0].scalar_type();
  const auto y_t = tensor_lists[1][0].scalar_type();
  const auto o_t = tensor_lists[2][0].scalar_type();
  APEX_DISPATCH_FLOAT_HALF_BF(x_t, "multi_tensor_axpby", ([&] {
    using x_scalar = scalar_t;
    APEX_DISPATCH_FLOAT_HALF_BF(y_t, "multi_tensor_axpby", ([&] {
      using y_scalar = scalar_t;
      APEX_DISPATCH_FLOAT_HALF_BF(o_t, "multi_tensor_axpby", ([&] {
        using o_scalar = scalar_t;
        multi_tensor_apply<3>(chunk_size, noop_flag, tensor_lists,
                              AxpbyFunctor<x_scalar, y_scalar, o_scalar>(), (float)a, (float)b,
                              (int)arg_to_check);
      }()));
    }()));
  }()));
}
