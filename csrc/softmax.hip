// apex_amd._softmax — fused scaled softmax family for gfx950.
// Reference surface: csrc/megatron/{scaled_softmax, scaled_masked_softmax,
// scaled_upper_triang_masked_softmax, generic_scaled_masked_softmax}_cuda.
//
// MI355X design: one 256-thread workgroup per row (4 wave64), online
// max+sum accumulation in fp32 (single pass over the row, lane-local online
// rescaling merged by wave64 shuffles + LDS), then a write pass that re-reads
// the row from cache. No 16K row-length ceiling (the reference's warp kernel
// is bounded at sk<=16384; the block-streaming form handles any sk, so
// "generic" binds to the same kernels). Masked positions write exact 0.
#include "common.h"
#include "multi_tensor_apply.h"  // VecPack 16B loads

namespace {

constexpr int SM_BLOCK = 256;

struct OnlineSM {
  float m = -INFINITY, s = 0.f;
  __device__ void add(float x) {
    if (x > m) {
      s = s * __expf(m - x) + 1.f;
      m = x;
    } else {
      s += __expf(x - m);
    }
  }
  __device__ void combine(float mb, float sb) {
    if (sb == 0.f) return;
    if (mb > m) {
      s = s * __expf(m - mb) + sb;
      m = mb;
    } else {
      s += sb * __expf(mb - m);
    }
  }
};

__device__ void block_online_sm(OnlineSM& o, float* smem /* 2*nwaves */) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    float mb = __shfl_xor(o.m, off);
    float sb = __shfl_xor(o.s, off);
    o.combine(mb, sb);
  }
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  if (lane == 0) {
    smem[2 * wid] = o.m;
    smem[2 * wid + 1] = o.s;
  }
  __syncthreads();
  OnlineSM total;
  for (int i = 0; i < nwaves; ++i) total.combine(smem[2 * i], smem[2 * i + 1]);
  __syncthreads();
  o = total;
}

// MODE: 0 = plain, 1 = additive bool mask [b,1,sq,sk], 2 = causal upper-tri
// VEC: 16-byte packed loads (8 x bf16/fp16 per lane — G13); the scalar
// variant (VEC=false) covers odd sk / unaligned rows.
template <typename T, int MODE, bool VEC>
__global__ void __launch_bounds__(SM_BLOCK) softmax_fwd_kernel(
    const T* __restrict__ in, T* __restrict__ out, const uint8_t* __restrict__ mask, float scale,
    long rows, long sk, long np, long sq) {
  __shared__ float smem[2 * (SM_BLOCK / WAVE_SIZE)];
  constexpr int W = VecPack<T>::width;
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* x = in + row * sk;
    T* y = out + row * sk;
    const uint8_t* mrow = nullptr;
    long limit = sk;
    if (MODE == 1) {
      // row = (b*np + h)*sq + q  →  mask row = b*sq + q
      const long q = row % sq;
      const long b = row / (np * sq);
      mrow = mask + (b * sq + q) * sk;
    } else if (MODE == 2) {
      // row = ab*sq + q ; causal keeps the first q+1 elements
      limit = (row % sq) + 1;
    }

    OnlineSM o;
    if (VEC) {
      for (long i = (long)threadIdx.x * W; i < sk; i += (long)blockDim.x * W) {
        VecPack<T> v;
        load_pack(v, x + i);
#pragma unroll
        for (int j = 0; j < W; ++j) {
          if (MODE == 2 && i + j >= limit) break;
          float f = to_float(v.a[j]) * scale;
          if (MODE == 1 && mrow[i + j]) f = -10000.0f;
          o.add(f);
        }
      }
    } else {
      for (long i = threadIdx.x; i < limit; i += blockDim.x) {
        float v = to_float(x[i]) * scale;
        if (MODE == 1 && mrow[i]) v = -10000.0f;
        o.add(v);
      }
    }
    block_online_sm(o, smem);
    const float inv_s = o.s > 0.f ? 1.f / o.s : 0.f;

    if (VEC) {
      for (long i = (long)threadIdx.x * W; i < sk; i += (long)blockDim.x * W) {
        VecPack<T> v, r;
        load_pack(v, x + i);
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float f = 0.f;
          if (MODE != 2 || i + j < limit) {
            float vv = to_float(v.a[j]) * scale;
            if (MODE == 1 && mrow[i + j]) vv = -10000.0f;
            f = __expf(vv - o.m) * inv_s;
          }
          r.a[j] = from_float<T>(f);
        }
        store_pack(y + i, r);
      }
    } else {
      for (long i = threadIdx.x; i < sk; i += blockDim.x) {
        float r = 0.f;
        if (i < limit) {
          float v = to_float(x[i]) * scale;
          if (MODE == 1 && mrow[i]) v = -10000.0f;
          r = __expf(v - o.m) * inv_s;
        }
        y[i] = from_float<T>(r);
      }
    }
    __syncthreads();
  }
}

// Wave-per-row variant for short rows (sk <= 64*W*NPACK): each wave64 owns
// one row held in registers — no LDS, no barriers, one global read per
// element (4 rows per 256-thread workgroup).
template <typename T, int MODE, int NPACK>
__global__ void __launch_bounds__(SM_BLOCK) softmax_fwd_wave_kernel(
    const T* __restrict__ in, T* __restrict__ out, const uint8_t* __restrict__ mask, float scale,
    long rows, long sk, long np, long sq) {
  constexpr int W = VecPack<T>::width;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int waves_per_block = SM_BLOCK / WAVE_SIZE;
  for (long row = (long)blockIdx.x * waves_per_block + wid; row < rows;
       row += (long)gridDim.x * waves_per_block) {
    const T* x = in + row * sk;
    T* y = out + row * sk;
    const uint8_t* mrow = nullptr;
    long limit = sk;
    if (MODE == 1) {
      const long q = row % sq;
      const long b = row / (np * sq);
      mrow = mask + (b * sq + q) * sk;
    } else if (MODE == 2) {
      limit = (row % sq) + 1;
    }

    OnlineSM o;
    VecPack<T> v[NPACK];
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i0 = (long)(k * WAVE_SIZE + lane) * W;
      if (i0 < sk) {
        load_pack(v[k], x + i0);
#pragma unroll
        for (int j = 0; j < W; ++j) {
          if (MODE == 2 && i0 + j >= limit) break;
          float f = to_float(v[k].a[j]) * scale;
          if (MODE == 1 && mrow[i0 + j]) f = -10000.0f;
          o.add(f);
        }
      }
    }
#pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
      o.combine(__shfl_xor(o.m, off), __shfl_xor(o.s, off));
    }
    const float inv_s = o.s > 0.f ? 1.f / o.s : 0.f;
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i0 = (long)(k * WAVE_SIZE + lane) * W;
      if (i0 < sk) {
        VecPack<T> r;
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float f = 0.f;
          if (MODE != 2 || i0 + j < limit) {
            float vv = to_float(v[k].a[j]) * scale;
            if (MODE == 1 && mrow[i0 + j]) vv = -10000.0f;
            f = __expf(vv - o.m) * inv_s;
          }
          r.a[j] = from_float<T>(f);
        }
        store_pack(y + i0, r);
      }
    }
  }
}

template <typename T, int NPACK>
__global__ void __launch_bounds__(SM_BLOCK) softmax_bwd_wave_kernel(
    const T* __restrict__ dy_ptr, const T* __restrict__ y_ptr, T* __restrict__ dx_ptr,
    float scale, long rows, long sk) {
  constexpr int W = VecPack<T>::width;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int waves_per_block = SM_BLOCK / WAVE_SIZE;
  for (long row = (long)blockIdx.x * waves_per_block + wid; row < rows;
       row += (long)gridDim.x * waves_per_block) {
    const T* dy = dy_ptr + row * sk;
    const T* y = y_ptr + row * sk;
    T* dx = dx_ptr + row * sk;
    VecPack<T> vd[NPACK], vy[NPACK];
    float acc = 0.f;
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i0 = (long)(k * WAVE_SIZE + lane) * W;
      if (i0 < sk) {
        load_pack(vd[k], dy + i0);
        load_pack(vy[k], y + i0);
#pragma unroll
        for (int j = 0; j < W; ++j) acc = fmaf(to_float(vd[k].a[j]), to_float(vy[k].a[j]), acc);
      }
    }
    float dot = wave_reduce_sum(acc);
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i0 = (long)(k * WAVE_SIZE + lane) * W;
      if (i0 < sk) {
        VecPack<T> r;
#pragma unroll
        for (int j = 0; j < W; ++j) {
          r.a[j] = from_float<T>(to_float(vy[k].a[j]) * (to_float(vd[k].a[j]) - dot) * scale);
        }
        store_pack(dx + i0, r);
      }
    }
  }
}

// grad = y * (dy - sum(dy*y)) * scale
template <typename T, bool VEC>
__global__ void __launch_bounds__(SM_BLOCK) softmax_bwd_kernel(
    const T* __restrict__ dy_ptr, const T* __restrict__ y_ptr, T* __restrict__ dx_ptr,
    float scale, long rows, long sk) {
  __shared__ float smem[SM_BLOCK / WAVE_SIZE];
  constexpr int W = VecPack<T>::width;
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dy = dy_ptr + row * sk;
    const T* y = y_ptr + row * sk;
    T* dx = dx_ptr + row * sk;
    float acc = 0.f;
    if (VEC) {
      for (long i = (long)threadIdx.x * W; i < sk; i += (long)blockDim.x * W) {
        VecPack<T> vd, vy;
        load_pack(vd, dy + i);
        load_pack(vy, y + i);
#pragma unroll
        for (int j = 0; j < W; ++j) acc = fmaf(to_float(vd.a[j]), to_float(vy.a[j]), acc);
      }
    } else {
      for (long i = threadIdx.x; i < sk; i += blockDim.x)
        acc = fmaf(to_float(dy[i]), to_float(y[i]), acc);
    }
    float dot = block_reduce_sum(acc, smem);
    if (VEC) {
      for (long i = (long)threadIdx.x * W; i < sk; i += (long)blockDim.x * W) {
        VecPack<T> vd, vy, r;
        load_pack(vd, dy + i);
        load_pack(vy, y + i);
#pragma unroll
        for (int j = 0; j < W; ++j) {
          r.a[j] = from_float<T>(to_float(vy.a[j]) * (to_float(vd.a[j]) - dot) * scale);
        }
        store_pack(dx + i, r);
      }
    } else {
      for (long i = threadIdx.x; i < sk; i += blockDim.x) {
        float yv = to_float(y[i]);
        dx[i] = from_float<T>(yv * (to_float(dy[i]) - dot) * scale);
      }
    }
    __syncthreads();
  }
}

inline int sm_grid(long rows) { return (int)std::min<long>(rows, 32768); }

template <int MODE>
at::Tensor fwd_impl(const at::Tensor& input, const c10::optional<at::Tensor>& mask, double scale,
                    long np, long sq) {
  auto x = input.contiguous();
  auto y = at::empty_like(x);
  const long sk = x.size(-1);
  const long rows = x.numel() / sk;
  at::Tensor m8;
  if (MODE == 1) m8 = mask->to(at::kByte).contiguous();
  APEX_DISPATCH_FLOAT_HALF_BF(x.scalar_type(), "scaled_softmax_forward", ([&] {
    constexpr int PW = 16 / sizeof(scalar_t);
    const bool vec = (sk % PW == 0) && is_pack_aligned<scalar_t>(x.data_ptr());
    const long wave_span = (long)WAVE_SIZE * PW;
    if (vec && sk <= wave_span * 4) {
      // short rows: one wave64 per row (row in registers), zero barriers
      const int wpb = SM_BLOCK / WAVE_SIZE;
      const int grid = (int)std::min<long>((rows + wpb - 1) / wpb, 32768);
      const int npack = (int)((sk + wave_span - 1) / wave_span);
      auto lw = [&](auto np_tag) {
        hipLaunchKernelGGL((softmax_fwd_wave_kernel<scalar_t, MODE, decltype(np_tag)::value>),
                           dim3(grid), dim3(SM_BLOCK), 0, current_stream(),
                           (const scalar_t*)x.data_ptr(), (scalar_t*)y.data_ptr(),
                           MODE == 1 ? m8.data_ptr<uint8_t>() : nullptr, (float)scale, rows, sk,
                           np, sq);
      };
      switch (npack) {
        case 1: lw(std::integral_constant<int, 1>{}); break;
        case 2: lw(std::integral_constant<int, 2>{}); break;
        case 3: lw(std::integral_constant<int, 3>{}); break;
        default: lw(std::integral_constant<int, 4>{}); break;
      }
    } else if (vec) {
      hipLaunchKernelGGL((softmax_fwd_kernel<scalar_t, MODE, true>), dim3(sm_grid(rows)),
                         dim3(SM_BLOCK), 0, current_stream(), (const scalar_t*)x.data_ptr(),
                         (scalar_t*)y.data_ptr(), MODE == 1 ? m8.data_ptr<uint8_t>() : nullptr,
                         (float)scale, rows, sk, np, sq);
    } else {
      hipLaunchKernelGGL((softmax_fwd_kernel<scalar_t, MODE, false>), dim3(sm_grid(rows)),
                         dim3(SM_BLOCK), 0, current_stream(), (const scalar_t*)x.data_ptr(),
                         (scalar_t*)y.data_ptr(), MODE == 1 ? m8.data_ptr<uint8_t>() : nullptr,
                         (float)scale, rows, sk, np, sq);
    }
  }()));
  HIP_CHECK(hipGetLastError());
  return y;
}

at::Tensor bwd_impl(const at::Tensor& grad_out, const at::Tensor& softmax_out, double scale) {
  auto dy = grad_out.contiguous();
  auto y = softmax_out.contiguous();
  auto dx = at::empty_like(dy);
  const long sk = y.size(-1);
  const long rows = y.numel() / sk;
  APEX_DISPATCH_FLOAT_HALF_BF(y.scalar_type(), "scaled_softmax_backward", ([&] {
    constexpr int PW = 16 / sizeof(scalar_t);
    const bool vec = (sk % PW == 0) &&
                     is_pack_aligned<scalar_t>(y.data_ptr()) &&
                     is_pack_aligned<scalar_t>(dy.data_ptr());
    const long wave_span = (long)WAVE_SIZE * PW;
    if (vec && sk <= wave_span * 2) {
      // NPACK capped at 2 for bwd (2 operand rows live in registers)
      const int wpb = SM_BLOCK / WAVE_SIZE;
      const int grid = (int)std::min<long>((rows + wpb - 1) / wpb, 32768);
      const int npack = (int)((sk + wave_span - 1) / wave_span);
      auto lw = [&](auto np_tag) {
        hipLaunchKernelGGL((softmax_bwd_wave_kernel<scalar_t, decltype(np_tag)::value>),
                           dim3(grid), dim3(SM_BLOCK), 0, current_stream(),
                           (const scalar_t*)dy.data_ptr(), (const scalar_t*)y.data_ptr(),
                           (scalar_t*)dx.data_ptr(), (float)scale, rows, sk);
      };
      if (npack == 1) lw(std::integral_constant<int, 1>{});
      else lw(std::integral_constant<int, 2>{});
    } else if (vec) {
      hipLaunchKernelGGL((softmax_bwd_kernel<scalar_t, true>), dim3(sm_grid(rows)),
                         dim3(SM_BLOCK), 0, current_stream(), (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)y.data_ptr(), (scalar_t*)dx.data_ptr(), (float)scale,
                         rows, sk);
    } else {
      hipLaunchKernelGGL((softmax_bwd_kernel<scalar_t, false>), dim3(sm_grid(rows)),
                         dim3(SM_BLOCK), 0, current_stream(), (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)y.data_ptr(), (scalar_t*)dx.data_ptr(), (float)scale,
                         rows, sk);
    }
  }()));
  HIP_CHECK(hipGetLastError());
  return dx;
}

}  // namespace

at::Tensor scaled_softmax_forward(at::Tensor input, double scale) {
  TORCH_CHECK(input.dim() == 4, "expected 4D [b, np, sq, sk]");
  return fwd_impl<0>(input, c10::nullopt, scale, input.size(1), input.size(2));
}

at::Tensor scaled_softmax_backward(at::Tensor grad_out, at::Tensor softmax_out, double scale) {
  return bwd_impl(grad_out, softmax_out, scale);
}

at::Tensor scaled_masked_softmax_forward(at::Tensor input, at::Tensor mask, double scale) {
  TORCH_CHECK(input.dim() == 4, "expected 4D [b, np, sq, sk]");
  TORCH_CHECK(mask.dim() == 4 && mask.size(1) == 1, "mask must be [b, 1, sq, sk]");
  return fwd_impl<1>(input, mask, scale, input.size(1), input.size(2));
}

at::Tensor scaled_masked_softmax_backward(at::Tensor grad_out, at::Tensor softmax_out,
                                          double scale) {
  return bwd_impl(grad_out, softmax_out, scale);
}

at::Tensor scaled_upper_triang_masked_softmax_forward(at::Tensor input, double scale) {
  TORCH_CHECK(input.dim() == 3 && input.size(1) == input.size(2),
              "expected [attn_batches, sq, sq]");
  return fwd_impl<2>(input, c10::nullopt, scale, 1, input.size(1));
}

at::Tensor scaled_upper_triang_masked_softmax_backward(at::Tensor grad_out,
                                                       at::Tensor softmax_out, double scale) {
  return bwd_impl(grad_out, softmax_out, scale);
}

// generic (arbitrary sk / non-pow2): same block-streaming kernels
at::Tensor generic_scaled_masked_softmax_forward(at::Tensor input, at::Tensor mask,
                                                 double scale) {
  return scaled_masked_softmax_forward(input, mask, scale);
}

at::Tensor generic_scaled_masked_softmax_backward(at::Tensor grad_out, at::Tensor softmax_out,
                                                  double scale) {
  return bwd_impl(grad_out, softmax_out, scale);
}

// Megatron launch-planning helper (reference: scaled_masked_softmax.cpp:74)
long get_batch_per_block(long sq, long sk, long b, long np) { return 1; }

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("scaled_softmax_forward", &scaled_softmax_forward);
  m.def("scaled_softmax_backward", &scaled_softmax_backward);
  m.def("scaled_masked_softmax_forward", &scaled_masked_softmax_forward);
  m.def("scaled_masked_softmax_backward", &scaled_masked_softmax_backward);
  m.def("scaled_upper_triang_masked_softmax_forward", &scaled_upper_triang_masked_softmax_forward);
  m.def("scaled_upper_triang_masked_softmax_backward", &scaled_upper_triang_masked_softmax_backward);
  m.def("generic_scaled_masked_softmax_forward", &generic_scaled_masked_softmax_forward);
  m.def("generic_scaled_masked_softmax_backward", &generic_scaled_masked_softmax_backward);
  m.def("get_batch_per_block", &get_batch_per_block);
}
