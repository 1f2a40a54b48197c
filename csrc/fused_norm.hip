// apex_amd._fused_norm — FusedLayerNorm / FusedRMSNorm kernels for gfx950.
//
// Reference API surface: csrc/layer_norm_cuda.cpp (forward/forward_affine/
// forward_affine_mixed_dtypes, backward/backward_affine, rms_*) and kernels
// in csrc/layer_norm_cuda_kernel.cu (Welford stats, two-pass gamma/beta
// grads, memory_efficient backward-from-output).
//
// MI355X design (not a port):
// * one 256-thread workgroup (4 waves) per ROW; lane-local streaming Welford
//   merged with a wave64 Chan combine + LDS tree across the 4 waves.
// * 8/16-byte vector loads whenever the row is vector-aligned (G13).
// * gamma/beta grads: deterministic two-stage column reduction — stage 1
//   tiles rows per workgroup into a [tiles, n2] fp32 partial buffer, stage 2
//   reduces columns in fixed order (bitwise-stable).
// * stats (mean/invvar) always fp32; mixed-dtype variants template input and
//   weight dtypes separately.
#include "common.h"
#include "multi_tensor_apply.h"  // Vec4 vector load/store helpers

#include <vector>

namespace {

constexpr int LN_BLOCK = 256;
constexpr int LN_BWD_ROWS_PER_BLOCK = 32;

struct Welford {
  float mean = 0.f, m2 = 0.f, count = 0.f;
  __device__ void add(float x) {
    count += 1.f;
    float delta = x - mean;
    mean += delta / count;
    m2 = fmaf(delta, x - mean, m2);
  }
  __device__ void combine(float mb, float m2b, float nb) {
    if (nb == 0.f) return;
    float n = count + nb;
    float delta = mb - mean;
    mean += delta * nb / n;
    m2 += m2b + delta * delta * count * nb / n;
    count = n;
  }
};

// full-block Welford reduce; every thread returns (mean, m2, count=n)
__device__ void block_welford(Welford& w, float* smem /* 3 * nwaves */) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    float mb = __shfl_xor(w.mean, off);
    float m2b = __shfl_xor(w.m2, off);
    float nb = __shfl_xor(w.count, off);
    w.combine(mb, m2b, nb);
  }
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  if (lane == 0) {
    smem[3 * wid] = w.mean;
    smem[3 * wid + 1] = w.m2;
    smem[3 * wid + 2] = w.count;
  }
  __syncthreads();
  Welford total;
  for (int i = 0; i < nwaves; ++i) total.combine(smem[3 * i], smem[3 * i + 1], smem[3 * i + 2]);
  __syncthreads();
  w = total;
}

template <typename T>
__device__ __forceinline__ bool row_vec_ok(const T* base, long n2) {
  return ((reinterpret_cast<uintptr_t>(base) & (sizeof(T) * 4 - 1)) == 0) && ((n2 & 3) == 0);
}

// ---------------- wave-per-row forward (narrow rows) ----------------
// One wave64 per row, the whole row held in registers (<= NPACK 16-byte
// packs per lane): single global read, zero barriers. Covers the common
// transformer hidden sizes (bf16: n2 <= 2048 at NPACK=4).
template <typename T, typename WT, bool RMS, bool AFFINE, int NPACK>
__global__ void __launch_bounds__(LN_BLOCK) ln_fwd_wave_kernel(
    const T* __restrict__ input, T* __restrict__ output, float* __restrict__ mean_out,
    float* __restrict__ invvar_out, const WT* __restrict__ gamma, const WT* __restrict__ beta,
    long n1, long n2, float eps) {
  constexpr int W = VecPack<T>::width;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  constexpr int WPB = LN_BLOCK / WAVE_SIZE;
  for (long row = (long)blockIdx.x * WPB + wid; row < n1; row += (long)gridDim.x * WPB) {
    const T* x = input + row * n2;
    T* y = output + row * n2;
    VecPack<T> xs[NPACK];
    Welford w;
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i = (long)(k * WAVE_SIZE + lane) * W;
      if (i < n2) {
        load_pack(xs[k], x + i);
#pragma unroll
        for (int j = 0; j < W; ++j) w.add(to_float(xs[k].a[j]));
      }
    }
#pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
      float mb = __shfl_xor(w.mean, off);
      float m2b = __shfl_xor(w.m2, off);
      float nb = __shfl_xor(w.count, off);
      w.combine(mb, m2b, nb);
    }
    const float mean = RMS ? 0.f : w.mean;
    // for RMS, m2 accumulated via Welford gives sum((x-mean)^2); recompute
    // sum(x^2) = m2 + n*mean^2
    const float sumsq = w.m2 + w.count * w.mean * w.mean;
    const float invvar = RMS ? rsqrtf(sumsq / n2 + eps) : rsqrtf(w.m2 / n2 + eps);
    if (lane == 0) {
      if (!RMS && mean_out) mean_out[row] = mean;
      invvar_out[row] = invvar;
    }
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i = (long)(k * WAVE_SIZE + lane) * W;
      if (i < n2) {
        VecPack<T> o;
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float xhat = (to_float(xs[k].a[j]) - mean) * invvar;
          float r = xhat;
          if (AFFINE) {
            r = xhat * to_float(gamma[i + j]);
            if (!RMS) r += to_float(beta[i + j]);
          }
          o.a[j] = from_float<T>(r);
        }
        store_pack(y + i, o);
      }
    }
  }
}

// fused residual-add + norm forward: z = x + res is written once and the
// norm is computed from the registers — one read of each input, one write of
// z and y, vs. separate add (2R+1W) + norm (1R+1W) passes.
template <typename T, typename WT, bool RMS, bool AFFINE, int NPACK>
__global__ void __launch_bounds__(LN_BLOCK) ln_add_fwd_wave_kernel(
    const T* __restrict__ input, const T* __restrict__ residual, T* __restrict__ sum_out,
    T* __restrict__ output, float* __restrict__ mean_out, float* __restrict__ invvar_out,
    const WT* __restrict__ gamma, const WT* __restrict__ beta, long n1, long n2, float eps) {
  constexpr int W = VecPack<T>::width;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  constexpr int WPB = LN_BLOCK / WAVE_SIZE;
  for (long row = (long)blockIdx.x * WPB + wid; row < n1; row += (long)gridDim.x * WPB) {
    const T* x = input + row * n2;
    const T* r = residual + row * n2;
    T* z = sum_out + row * n2;
    T* y = output + row * n2;
    VecPack<T> xs[NPACK];
    Welford w;
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i = (long)(k * WAVE_SIZE + lane) * W;
      if (i < n2) {
        VecPack<T> rv;
        load_pack(xs[k], x + i);
        load_pack(rv, r + i);
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float s = to_float(xs[k].a[j]) + to_float(rv.a[j]);
          xs[k].a[j] = from_float<T>(s);
          w.add(to_float(xs[k].a[j]));  // stats on the stored (rounded) sum
        }
        store_pack(z + i, xs[k]);
      }
    }
#pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
      float mb = __shfl_xor(w.mean, off);
      float m2b = __shfl_xor(w.m2, off);
      float nb = __shfl_xor(w.count, off);
      w.combine(mb, m2b, nb);
    }
    const float mean = RMS ? 0.f : w.mean;
    const float sumsq = w.m2 + w.count * w.mean * w.mean;
    const float invvar = RMS ? rsqrtf(sumsq / n2 + eps) : rsqrtf(w.m2 / n2 + eps);
    if (lane == 0) {
      if (!RMS && mean_out) mean_out[row] = mean;
      invvar_out[row] = invvar;
    }
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i = (long)(k * WAVE_SIZE + lane) * W;
      if (i < n2) {
        VecPack<T> o;
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float xhat = (to_float(xs[k].a[j]) - mean) * invvar;
          float rr = xhat;
          if (AFFINE) {
            rr = xhat * to_float(gamma[i + j]);
            if (!RMS) rr += to_float(beta[i + j]);
          }
          o.a[j] = from_float<T>(rr);
        }
        store_pack(y + i, o);
      }
    }
  }
}

// wave-per-row backward dx: dy and x(/y) rows in registers, two wave sums.
template <typename T, typename WT, bool RMS, bool AFFINE, bool MEMEFF, int NPACK>
__global__ void __launch_bounds__(LN_BLOCK) ln_bwd_dx_wave_kernel(
    const T* __restrict__ dy_ptr, const T* __restrict__ io, const float* __restrict__ mean_ptr,
    const float* __restrict__ invvar_ptr, const WT* __restrict__ gamma,
    const WT* __restrict__ beta, T* __restrict__ dx_ptr, long n1, long n2) {
  constexpr int W = VecPack<T>::width;
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  constexpr int WPB = LN_BLOCK / WAVE_SIZE;
  for (long row = (long)blockIdx.x * WPB + wid; row < n1; row += (long)gridDim.x * WPB) {
    const T* dy = dy_ptr + row * n2;
    const T* xr = io + row * n2;
    T* dx = dx_ptr + row * n2;
    const float invvar = invvar_ptr[row];
    const float mean = (RMS || MEMEFF) ? 0.f : mean_ptr[row];

    VecPack<T> vd[NPACK], vx[NPACK];
    float a1 = 0.f, a2 = 0.f;
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i = (long)(k * WAVE_SIZE + lane) * W;
      if (i < n2) {
        load_pack(vd[k], dy + i);
        load_pack(vx[k], xr + i);
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float g = AFFINE ? to_float(gamma[i + j]) : 1.f;
          float dyf = to_float(vd[k].a[j]) * g;
          float xhat;
          if (MEMEFF) {
            float yv = to_float(vx[k].a[j]);
            if (AFFINE) {
              float gv = to_float(gamma[i + j]);
              xhat = RMS ? yv / gv : (yv - to_float(beta[i + j])) / gv;
            } else {
              xhat = yv;
            }
          } else {
            xhat = (to_float(vx[k].a[j]) - mean) * invvar;
          }
          a1 = fmaf(dyf, xhat, a1);
          a2 += dyf;
        }
      }
    }
    float s1 = wave_reduce_sum(a1);
    float s2 = RMS ? 0.f : wave_reduce_sum(a2);
    const float inv_n = 1.f / (float)n2;
#pragma unroll
    for (int k = 0; k < NPACK; ++k) {
      const long i = (long)(k * WAVE_SIZE + lane) * W;
      if (i < n2) {
        VecPack<T> o;
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float g = AFFINE ? to_float(gamma[i + j]) : 1.f;
          float dyf = to_float(vd[k].a[j]) * g;
          float xhat;
          if (MEMEFF) {
            float yv = to_float(vx[k].a[j]);
            if (AFFINE) {
              float gv = to_float(gamma[i + j]);
              xhat = RMS ? yv / gv : (yv - to_float(beta[i + j])) / gv;
            } else {
              xhat = yv;
            }
          } else {
            xhat = (to_float(vx[k].a[j]) - mean) * invvar;
          }
          float r;
          if (RMS) {
            r = invvar * (dyf - xhat * s1 * inv_n);
          } else {
            r = invvar * (dyf - s2 * inv_n - xhat * s1 * inv_n);
          }
          o.a[j] = from_float<T>(r);
        }
        store_pack(dx + i, o);
      }
    }
  }
}

// ---------------- forward ----------------
// RMS: no mean; affine: gamma (and beta for LN) applied.
template <typename T, typename WT, bool RMS, bool AFFINE>
__global__ void __launch_bounds__(LN_BLOCK) ln_fwd_kernel(
    const T* __restrict__ input, T* __restrict__ output, float* __restrict__ mean_out,
    float* __restrict__ invvar_out, const WT* __restrict__ gamma, const WT* __restrict__ beta,
    long n1, long n2, float eps) {
  __shared__ float smem[3 * (LN_BLOCK / WAVE_SIZE)];
  for (long row = blockIdx.x; row < n1; row += gridDim.x) {
    const T* x = input + row * n2;
    T* y = output + row * n2;

    float mean, invvar;
    if (RMS) {
      float acc = 0.f;
      if (row_vec_ok(x, n2)) {
        for (long i = (long)threadIdx.x * 4; i < n2; i += (long)blockDim.x * 4) {
          Vec4<T> v;
          load_vec4(v, x + i);
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            float f = to_float(v.a[j]);
            acc = fmaf(f, f, acc);
          }
        }
      } else {
        for (long i = threadIdx.x; i < n2; i += blockDim.x) {
          float f = to_float(x[i]);
          acc = fmaf(f, f, acc);
        }
      }
      float tot = block_reduce_sum(acc, smem);
      mean = 0.f;
      invvar = rsqrtf(tot / n2 + eps);
    } else {
      Welford w;
      if (row_vec_ok(x, n2)) {
        for (long i = (long)threadIdx.x * 4; i < n2; i += (long)blockDim.x * 4) {
          Vec4<T> v;
          load_vec4(v, x + i);
#pragma unroll
          for (int j = 0; j < 4; ++j) w.add(to_float(v.a[j]));
        }
      } else {
        for (long i = threadIdx.x; i < n2; i += blockDim.x) w.add(to_float(x[i]));
      }
      block_welford(w, smem);
      mean = w.mean;
      invvar = rsqrtf(w.m2 / n2 + eps);
    }

    if (threadIdx.x == 0) {
      if (!RMS && mean_out) mean_out[row] = mean;
      invvar_out[row] = invvar;
    }

    if (row_vec_ok(x, n2) && row_vec_ok(y, n2)) {
      for (long i = (long)threadIdx.x * 4; i < n2; i += (long)blockDim.x * 4) {
        Vec4<T> v;
        load_vec4(v, x + i);
        Vec4<T> o;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float xhat = (to_float(v.a[j]) - mean) * invvar;
          float r = xhat;
          if (AFFINE) {
            r = xhat * to_float(gamma[i + j]);
            if (!RMS) r += to_float(beta[i + j]);
          }
          o.a[j] = from_float<T>(r);
        }
        store_vec4(y + i, o);
      }
    } else {
      for (long i = threadIdx.x; i < n2; i += blockDim.x) {
        float xhat = (to_float(x[i]) - mean) * invvar;
        float r = xhat;
        if (AFFINE) {
          r = xhat * to_float(gamma[i]);
          if (!RMS) r += to_float(beta[i]);
        }
        y[i] = from_float<T>(r);
      }
    }
    __syncthreads();
  }
}

// ---------------- backward: grad_input ----------------
// MEMEFF: `io` holds the forward OUTPUT; xhat is recovered as
// (y - beta)/gamma (LN) or y/gamma (RMS); x-mu = xhat/invvar.
template <typename T, typename WT, bool RMS, bool AFFINE, bool MEMEFF>
__global__ void __launch_bounds__(LN_BLOCK) ln_bwd_dx_kernel(
    const T* __restrict__ dy_ptr, const T* __restrict__ io, const float* __restrict__ mean_ptr,
    const float* __restrict__ invvar_ptr, const WT* __restrict__ gamma,
    const WT* __restrict__ beta, T* __restrict__ dx_ptr, long n1, long n2) {
  __shared__ float smem[LN_BLOCK / WAVE_SIZE];
  for (long row = blockIdx.x; row < n1; row += gridDim.x) {
    const T* dy = dy_ptr + row * n2;
    const T* xr = io + row * n2;
    T* dx = dx_ptr + row * n2;
    const float invvar = invvar_ptr[row];
    const float mean = (RMS || MEMEFF) ? 0.f : mean_ptr[row];

    // pass 1: s1 = sum(dy*gamma*xhat), s2 = sum(dy*gamma)
    float a1 = 0.f, a2 = 0.f;
    for (long i = threadIdx.x; i < n2; i += blockDim.x) {
      float g = AFFINE ? to_float(gamma[i]) : 1.f;
      float dyf = to_float(dy[i]) * g;
      float xhat;
      if (MEMEFF) {
        float yv = to_float(xr[i]);
        if (AFFINE) {
          float gv = to_float(gamma[i]);
          xhat = RMS ? yv / gv : (yv - to_float(beta[i])) / gv;
        } else {
          xhat = yv;
        }
      } else {
        xhat = (to_float(xr[i]) - mean) * invvar;
      }
      a1 = fmaf(dyf, xhat, a1);
      a2 += dyf;
    }
    float s1 = block_reduce_sum(a1, smem);
    float s2 = RMS ? 0.f : block_reduce_sum(a2, smem);

    const float inv_n = 1.f / (float)n2;
    for (long i = threadIdx.x; i < n2; i += blockDim.x) {
      float g = AFFINE ? to_float(gamma[i]) : 1.f;
      float dyf = to_float(dy[i]) * g;
      float xhat;
      if (MEMEFF) {
        float yv = to_float(xr[i]);
        if (AFFINE) {
          float gv = to_float(gamma[i]);
          xhat = RMS ? yv / gv : (yv - to_float(beta[i])) / gv;
        } else {
          xhat = yv;
        }
      } else {
        xhat = (to_float(xr[i]) - mean) * invvar;
      }
      float r;
      if (RMS) {
        r = invvar * (dyf - xhat * s1 * inv_n);
      } else {
        r = invvar * (dyf - s2 * inv_n - xhat * s1 * inv_n);
      }
      dx[i] = from_float<T>(r);
    }
    __syncthreads();
  }
}

// ---------------- backward: gamma/beta partials ----------------
// stage 1: each block owns a tile of LN_BWD_ROWS_PER_BLOCK rows; threads
// stride columns; per-column accumulate over the tile's rows.
template <typename T, typename WT, bool RMS, bool AFFINE, bool MEMEFF>
__global__ void __launch_bounds__(LN_BLOCK) ln_bwd_partials_kernel(
    const T* __restrict__ dy_ptr, const T* __restrict__ io, const float* __restrict__ mean_ptr,
    const float* __restrict__ invvar_ptr, const WT* __restrict__ gamma,
    const WT* __restrict__ beta, float* __restrict__ part_gw, float* __restrict__ part_gb,
    long n1, long n2) {
  const long row0 = (long)blockIdx.y * LN_BWD_ROWS_PER_BLOCK;
  const long row1 = min(row0 + LN_BWD_ROWS_PER_BLOCK, n1);
  for (long col = blockIdx.x * blockDim.x + threadIdx.x; col < n2;
       col += (long)gridDim.x * blockDim.x) {
    float gw = 0.f, gb = 0.f;
    for (long row = row0; row < row1; ++row) {
      const float invvar = invvar_ptr[row];
      const float mean = (RMS || MEMEFF) ? 0.f : mean_ptr[row];
      float dyf = to_float(dy_ptr[row * n2 + col]);
      float xhat;
      if (MEMEFF) {
        float yv = to_float(io[row * n2 + col]);
        if (AFFINE) {
          float gv = to_float(gamma[col]);
          xhat = RMS ? yv / gv : (yv - to_float(beta[col])) / gv;
        } else {
          xhat = yv;
        }
      } else {
        xhat = (to_float(io[row * n2 + col]) - mean) * invvar;
      }
      gw = fmaf(dyf, xhat, gw);
      gb += dyf;
    }
    part_gw[blockIdx.y * n2 + col] = gw;
    if (!RMS) part_gb[blockIdx.y * n2 + col] = gb;
  }
}

// stage 2: fixed-order column sum of the partial tiles.
template <typename WT>
__global__ void __launch_bounds__(LN_BLOCK) ln_bwd_colsum_kernel(
    const float* __restrict__ part_gw, const float* __restrict__ part_gb,
    WT* __restrict__ grad_gamma, WT* __restrict__ grad_beta, long tiles, long n2) {
  for (long col = blockIdx.x * blockDim.x + threadIdx.x; col < n2;
       col += (long)gridDim.x * blockDim.x) {
    float gw = 0.f, gb = 0.f;
    for (long ti = 0; ti < tiles; ++ti) {
      gw += part_gw[ti * n2 + col];
      if (part_gb) gb += part_gb[ti * n2 + col];
    }
    grad_gamma[col] = from_float<WT>(gw);
    if (part_gb && grad_beta) grad_beta[col] = from_float<WT>(gb);
  }
}

// ---------------- host ----------------

void shape_split(const at::Tensor& input, const std::vector<long>& normalized_shape, long& n1,
                 long& n2) {
  const int nd = (int)normalized_shape.size();
  const int idim = input.dim();
  TORCH_CHECK(nd <= idim, "normalized_shape rank too large");
  n2 = 1;
  for (int i = 0; i < nd; ++i) {
    TORCH_CHECK(input.size(idim - nd + i) == normalized_shape[i], "normalized_shape mismatch");
    n2 *= normalized_shape[i];
  }
  n1 = input.numel() / n2;
}

inline int fwd_grid(long n1) { return (int)std::min<long>(n1, 16384); }

template <bool RMS>
std::vector<at::Tensor> norm_fwd(const at::Tensor& input,
                                 const std::vector<long>& normalized_shape,
                                 const c10::optional<at::Tensor>& gamma,
                                 const c10::optional<at::Tensor>& beta, double eps) {
  long n1, n2;
  shape_split(input, normalized_shape, n1, n2);
  auto in = input.contiguous();
  auto out = at::empty_like(in);
  auto fopts = at::TensorOptions().dtype(at::kFloat).device(in.device());
  auto mean = RMS ? at::empty({0}, fopts) : at::empty({n1}, fopts);
  auto invvar = at::empty({n1}, fopts);
  const bool affine = gamma.has_value();
  auto stream = current_stream();

  APEX_DISPATCH_FLOAT_HALF_BF(in.scalar_type(), "fused_norm_fwd", ([&] {
    using in_t = scalar_t;
    const auto w_type = affine ? gamma->scalar_type() : in.scalar_type();
    APEX_DISPATCH_FLOAT_HALF_BF(w_type, "fused_norm_fwd", ([&] {
      using w_t = scalar_t;
      constexpr int W = VecPack<in_t>::width;
      const bool wave_ok = (n2 % W == 0) && is_pack_aligned<in_t>(in.data_ptr()) &&
                           n2 <= (long)WAVE_SIZE * W * 4;
      const int npack = wave_ok ? (int)((n2 + WAVE_SIZE * W - 1) / (WAVE_SIZE * W)) : 0;
      const w_t* g_ptr = affine ? (const w_t*)gamma->data_ptr() : nullptr;
      const w_t* b_ptr = (affine && !RMS) ? (const w_t*)beta->data_ptr() : nullptr;
      float* mean_ptr = RMS ? nullptr : mean.data_ptr<float>();

      auto launch_wave = [&](auto aff, auto np_tag) {
        constexpr bool AFF = decltype(aff)::value;
        constexpr int NP = decltype(np_tag)::value;
        constexpr int WPB = LN_BLOCK / WAVE_SIZE;
        const int grid = (int)std::min<long>((n1 + WPB - 1) / WPB, 32768);
        hipLaunchKernelGGL((ln_fwd_wave_kernel<in_t, w_t, RMS, AFF, NP>), dim3(grid),
                           dim3(LN_BLOCK), 0, stream, (const in_t*)in.data_ptr(),
                           (in_t*)out.data_ptr(), mean_ptr, invvar.data_ptr<float>(), g_ptr,
                           b_ptr, n1, n2, (float)eps);
      };
      auto launch_block = [&](auto aff) {
        constexpr bool AFF = decltype(aff)::value;
        hipLaunchKernelGGL((ln_fwd_kernel<in_t, w_t, RMS, AFF>), dim3(fwd_grid(n1)),
                           dim3(LN_BLOCK), 0, stream, (const in_t*)in.data_ptr(),
                           (in_t*)out.data_ptr(), mean_ptr, invvar.data_ptr<float>(), g_ptr,
                           b_ptr, n1, n2, (float)eps);
      };
      using Tt = std::true_type;
      using Ff = std::false_type;
      if (wave_ok) {
        auto dispatch_np = [&](auto aff) {
          switch (npack) {
            case 1: launch_wave(aff, std::integral_constant<int, 1>{}); break;
            case 2: launch_wave(aff, std::integral_constant<int, 2>{}); break;
            case 3: launch_wave(aff, std::integral_constant<int, 3>{}); break;
            default: launch_wave(aff, std::integral_constant<int, 4>{}); break;
          }
        };
        if (affine) dispatch_np(Tt{});
        else dispatch_np(Ff{});
      } else {
        if (affine) launch_block(Tt{});
        else launch_block(Ff{});
      }
      HIP_CHECK(hipGetLastError());
    }()));
  }()));
  if (RMS) return {out, invvar};
  return {out, mean, invvar};
}

// fused residual-add + norm forward; falls back to an eager add + the block
// kernel for rows too wide for the wave kernel's register budget.
template <bool RMS>
std::vector<at::Tensor> norm_add_fwd(const at::Tensor& input, const at::Tensor& residual,
                                     const std::vector<long>& normalized_shape,
                                     const c10::optional<at::Tensor>& gamma,
                                     const c10::optional<at::Tensor>& beta, double eps) {
  long n1, n2;
  shape_split(input, normalized_shape, n1, n2);
  auto in = input.contiguous();
  auto res = residual.contiguous();
  TORCH_CHECK(res.sizes() == in.sizes() && res.scalar_type() == in.scalar_type(),
              "residual must match input shape/dtype");
  const bool affine = gamma.has_value();

  bool wave_possible = false;
  APEX_DISPATCH_FLOAT_HALF_BF(in.scalar_type(), "fused_norm_add_fwd_probe", ([&] {
    constexpr int W = VecPack<scalar_t>::width;
    wave_possible = (n2 % W == 0) && is_pack_aligned<scalar_t>(in.data_ptr()) &&
                    is_pack_aligned<scalar_t>(res.data_ptr()) && n2 <= (long)WAVE_SIZE * W * 4;
  }()));
  if (!wave_possible) {
    auto z = in + res;
    auto out = norm_fwd<RMS>(z, normalized_shape, gamma, beta, eps);
    out.insert(out.begin() + 1, z);
    return out;
  }

  auto z = at::empty_like(in);
  auto out = at::empty_like(in);
  auto fopts = at::TensorOptions().dtype(at::kFloat).device(in.device());
  auto mean = RMS ? at::empty({0}, fopts) : at::empty({n1}, fopts);
  auto invvar = at::empty({n1}, fopts);
  auto stream = current_stream();

  APEX_DISPATCH_FLOAT_HALF_BF(in.scalar_type(), "fused_norm_add_fwd", ([&] {
    using in_t = scalar_t;
    const auto w_type = affine ? gamma->scalar_type() : in.scalar_type();
    APEX_DISPATCH_FLOAT_HALF_BF(w_type, "fused_norm_add_fwd", ([&] {
      using w_t = scalar_t;
      constexpr int W = VecPack<in_t>::width;
      const int npack = (int)((n2 + WAVE_SIZE * W - 1) / (WAVE_SIZE * W));
      const w_t* g_ptr = affine ? (const w_t*)gamma->data_ptr() : nullptr;
      const w_t* b_ptr = (affine && !RMS) ? (const w_t*)beta->data_ptr() : nullptr;
      float* mean_ptr = RMS ? nullptr : mean.data_ptr<float>();
      constexpr int WPB = LN_BLOCK / WAVE_SIZE;
      const int grid = (int)std::min<long>((n1 + WPB - 1) / WPB, 32768);
      auto lw = [&](auto aff, auto np_tag) {
        constexpr bool AFF = decltype(aff)::value;
        constexpr int NP = decltype(np_tag)::value;
        hipLaunchKernelGGL((ln_add_fwd_wave_kernel<in_t, w_t, RMS, AFF, NP>), dim3(grid),
                           dim3(LN_BLOCK), 0, stream, (const in_t*)in.data_ptr(),
                           (const in_t*)res.data_ptr(), (in_t*)z.data_ptr(),
                           (in_t*)out.data_ptr(), mean_ptr, invvar.data_ptr<float>(), g_ptr,
                           b_ptr, n1, n2, (float)eps);
      };
      auto dispatch_np = [&](auto aff) {
        switch (npack) {
          case 1: lw(aff, std::integral_constant<int, 1>{}); break;
          case 2: lw(aff, std::integral_constant<int, 2>{}); break;
          case 3: lw(aff, std::integral_constant<int, 3>{}); break;
          default: lw(aff, std::integral_constant<int, 4>{}); break;
        }
      };
      if (affine) dispatch_np(std::true_type{});
      else dispatch_np(std::false_type{});
      HIP_CHECK(hipGetLastError());
    }()));
  }()));
  if (RMS) return {out, z, invvar};
  return {out, z, mean, invvar};
}

template <typename in_t, typename w_t, bool RMS, bool AFF, bool MEFF>
void launch_bwd_impl(const at::Tensor& dy, const at::Tensor& io, const float* mean_ptr,
                     const at::Tensor& invvar, const w_t* g_ptr, const w_t* b_ptr,
                     at::Tensor& dx, at::Tensor& part_gw, at::Tensor& part_gb,
                     at::Tensor& grad_gamma, at::Tensor& grad_beta, long tiles, long n1, long n2,
                     bool affine, hipStream_t stream) {
  constexpr int W = VecPack<in_t>::width;
  const bool wave_ok = (n2 % W == 0) && is_pack_aligned<in_t>(dy.data_ptr()) &&
                       is_pack_aligned<in_t>(io.data_ptr()) && n2 <= (long)WAVE_SIZE * W * 4;
  if (wave_ok) {
    const int npack = (int)((n2 + WAVE_SIZE * W - 1) / (WAVE_SIZE * W));
    constexpr int WPB = LN_BLOCK / WAVE_SIZE;
    const int grid = (int)std::min<long>((n1 + WPB - 1) / WPB, 32768);
    auto lw = [&](auto np_tag) {
      constexpr int NP = decltype(np_tag)::value;
      hipLaunchKernelGGL((ln_bwd_dx_wave_kernel<in_t, w_t, RMS, AFF, MEFF, NP>), dim3(grid),
                         dim3(LN_BLOCK), 0, stream, (const in_t*)dy.data_ptr(),
                         (const in_t*)io.data_ptr(), mean_ptr, invvar.data_ptr<float>(), g_ptr,
                         b_ptr, (in_t*)dx.data_ptr(), n1, n2);
    };
    switch (npack) {
      case 1: lw(std::integral_constant<int, 1>{}); break;
      case 2: lw(std::integral_constant<int, 2>{}); break;
      case 3: lw(std::integral_constant<int, 3>{}); break;
      default: lw(std::integral_constant<int, 4>{}); break;
    }
  } else {
    hipLaunchKernelGGL((ln_bwd_dx_kernel<in_t, w_t, RMS, AFF, MEFF>), dim3(fwd_grid(n1)),
                       dim3(LN_BLOCK), 0, stream, (const in_t*)dy.data_ptr(),
                       (const in_t*)io.data_ptr(), mean_ptr, invvar.data_ptr<float>(), g_ptr,
                       b_ptr, (in_t*)dx.data_ptr(), n1, n2);
  }
  HIP_CHECK(hipGetLastError());
  if (affine) {
    dim3 pgrid((uint32_t)std::min<long>((n2 + LN_BLOCK - 1) / LN_BLOCK, 1024), (uint32_t)tiles);
    hipLaunchKernelGGL((ln_bwd_partials_kernel<in_t, w_t, RMS, AFF, MEFF>), pgrid,
                       dim3(LN_BLOCK), 0, stream, (const in_t*)dy.data_ptr(),
                       (const in_t*)io.data_ptr(), mean_ptr, invvar.data_ptr<float>(), g_ptr,
                       b_ptr, part_gw.data_ptr<float>(),
                       RMS ? nullptr : part_gb.data_ptr<float>(), n1, n2);
    HIP_CHECK(hipGetLastError());
    // column sum of the partial tiles: torch's deterministic reduction
    // parallelizes over n2*tiles (the hand-rolled single-stage colsum kernel
    // underfilled the GPU at small n2 — 30 ms/step on BERT-base)
    grad_gamma.copy_(part_gw.sum(0));
    if (!RMS) grad_beta.copy_(part_gb.sum(0));
  }
}

template <bool RMS>
std::vector<at::Tensor> norm_bwd(const at::Tensor& grad_out,
                                 const c10::optional<at::Tensor>& mean, const at::Tensor& invvar,
                                 const at::Tensor& input_or_output,
                                 const std::vector<long>& normalized_shape,
                                 const c10::optional<at::Tensor>& gamma,
                                 const c10::optional<at::Tensor>& beta, double eps,
                                 bool memory_efficient) {
  long n1, n2;
  shape_split(input_or_output, normalized_shape, n1, n2);
  auto dy = grad_out.contiguous();
  auto io = input_or_output.contiguous();
  auto dx = at::empty_like(io);
  const bool affine = gamma.has_value();
  auto stream = current_stream();

  at::Tensor grad_gamma, grad_beta, part_gw, part_gb;
  long tiles = 0;
  if (affine) {
    tiles = (n1 + LN_BWD_ROWS_PER_BLOCK - 1) / LN_BWD_ROWS_PER_BLOCK;
    auto fopts = at::TensorOptions().dtype(at::kFloat).device(io.device());
    part_gw = at::empty({tiles, n2}, fopts);
    part_gb = RMS ? at::empty({0}, fopts) : at::empty({tiles, n2}, fopts);
    grad_gamma = at::empty_like(*gamma);
    grad_beta = RMS ? at::Tensor() : at::empty_like(*gamma);
  }

  const float* mean_ptr = (RMS || memory_efficient || !mean.has_value())
                              ? nullptr
                              : mean->data_ptr<float>();

  APEX_DISPATCH_FLOAT_HALF_BF(io.scalar_type(), "fused_norm_bwd", ([&] {
    using in_t = scalar_t;
    const auto w_type = affine ? gamma->scalar_type() : io.scalar_type();
    APEX_DISPATCH_FLOAT_HALF_BF(w_type, "fused_norm_bwd", ([&] {
      using w_t = scalar_t;
      const w_t* g_ptr = affine ? (const w_t*)gamma->data_ptr() : nullptr;
      const w_t* b_ptr = (affine && !RMS) ? (const w_t*)beta->data_ptr() : nullptr;

      if (affine && memory_efficient)
        launch_bwd_impl<in_t, w_t, RMS, true, true>(dy, io, mean_ptr, invvar, g_ptr, b_ptr, dx,
                                                    part_gw, part_gb, grad_gamma, grad_beta,
                                                    tiles, n1, n2, affine, stream);
      else if (affine)
        launch_bwd_impl<in_t, w_t, RMS, true, false>(dy, io, mean_ptr, invvar, g_ptr, b_ptr, dx,
                                                     part_gw, part_gb, grad_gamma, grad_beta,
                                                     tiles, n1, n2, affine, stream);
      else if (memory_efficient)
        launch_bwd_impl<in_t, w_t, RMS, false, true>(dy, io, mean_ptr, invvar, g_ptr, b_ptr, dx,
                                                     part_gw, part_gb, grad_gamma, grad_beta,
                                                     tiles, n1, n2, affine, stream);
      else
        launch_bwd_impl<in_t, w_t, RMS, false, false>(dy, io, mean_ptr, invvar, g_ptr, b_ptr, dx,
                                                      part_gw, part_gb, grad_gamma, grad_beta,
                                                      tiles, n1, n2, affine, stream);
    }()));
  }()));

  if (!affine) return {dx};
  if (RMS) return {dx, grad_gamma};
  return {dx, grad_gamma, grad_beta};
}

}  // namespace

// ---------------- bindings ----------------

std::vector<at::Tensor> forward_affine(at::Tensor input, std::vector<long> normalized_shape,
                                       at::Tensor weight, at::Tensor bias, double eps) {
  return norm_fwd<false>(input, normalized_shape, weight, bias, eps);
}

std::vector<at::Tensor> forward_affine_mixed_dtypes(at::Tensor input,
                                                    std::vector<long> normalized_shape,
                                                    at::Tensor weight, at::Tensor bias,
                                                    double eps) {
  return norm_fwd<false>(input, normalized_shape, weight, bias, eps);
}

std::vector<at::Tensor> forward_plain(at::Tensor input, std::vector<long> normalized_shape,
                                      double eps) {
  return norm_fwd<false>(input, normalized_shape, c10::nullopt, c10::nullopt, eps);
}

std::vector<at::Tensor> backward_affine(at::Tensor grad_out, c10::optional<at::Tensor> mean,
                                        at::Tensor invvar, at::Tensor input_or_output,
                                        std::vector<long> normalized_shape, at::Tensor weight,
                                        at::Tensor bias, double eps, bool memory_efficient) {
  return norm_bwd<false>(grad_out, mean, invvar, input_or_output, normalized_shape, weight, bias,
                         eps, memory_efficient);
}

at::Tensor backward_plain(at::Tensor grad_out, c10::optional<at::Tensor> mean, at::Tensor invvar,
                          at::Tensor input_or_output, std::vector<long> normalized_shape,
                          double eps, bool memory_efficient) {
  return norm_bwd<false>(grad_out, mean, invvar, input_or_output, normalized_shape, c10::nullopt,
                         c10::nullopt, eps, memory_efficient)[0];
}

std::vector<at::Tensor> forward_add_affine(at::Tensor input, at::Tensor residual,
                                           std::vector<long> normalized_shape, at::Tensor weight,
                                           at::Tensor bias, double eps) {
  return norm_add_fwd<false>(input, residual, normalized_shape, weight, bias, eps);
}

std::vector<at::Tensor> rms_forward_add_affine(at::Tensor input, at::Tensor residual,
                                               std::vector<long> normalized_shape,
                                               at::Tensor weight, double eps) {
  return norm_add_fwd<true>(input, residual, normalized_shape, weight, c10::nullopt, eps);
}

std::vector<at::Tensor> rms_forward_affine(at::Tensor input, std::vector<long> normalized_shape,
                                           at::Tensor weight, double eps) {
  return norm_fwd<true>(input, normalized_shape, weight, c10::nullopt, eps);
}

std::vector<at::Tensor> rms_forward_affine_mixed_dtypes(at::Tensor input,
                                                        std::vector<long> normalized_shape,
                                                        at::Tensor weight, double eps) {
  return norm_fwd<true>(input, normalized_shape, weight, c10::nullopt, eps);
}

std::vector<at::Tensor> rms_forward_plain(at::Tensor input, std::vector<long> normalized_shape,
                                          double eps) {
  return norm_fwd<true>(input, normalized_shape, c10::nullopt, c10::nullopt, eps);
}

std::vector<at::Tensor> rms_backward_affine(at::Tensor grad_out, at::Tensor invvar,
                                            at::Tensor input_or_output,
                                            std::vector<long> normalized_shape, at::Tensor weight,
                                            double eps, bool memory_efficient) {
  return norm_bwd<true>(grad_out, c10::nullopt, invvar, input_or_output, normalized_shape, weight,
                        c10::nullopt, eps, memory_efficient);
}

at::Tensor rms_backward_plain(at::Tensor grad_out, at::Tensor invvar, at::Tensor input_or_output,
                              std::vector<long> normalized_shape, double eps,
                              bool memory_efficient) {
  return norm_bwd<true>(grad_out, c10::nullopt, invvar, input_or_output, normalized_shape,
                        c10::nullopt, c10::nullopt, eps, memory_efficient)[0];
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("forward_affine", &forward_affine, "LayerNorm fwd (affine)");
  m.def("forward_affine_mixed_dtypes", &forward_affine_mixed_dtypes,
        "LayerNorm fwd, low-precision input with fp32 params");
  m.def("forward", &forward_plain, "LayerNorm fwd (no affine)");
  m.def("backward_affine", &backward_affine, "LayerNorm bwd (affine)");
  m.def("backward", &backward_plain, "LayerNorm bwd (no affine)");
  m.def("forward_add_affine", &forward_add_affine,
        "fused residual-add + LayerNorm fwd: returns (y, z=x+res, mean, invvar)");
  m.def("rms_forward_add_affine", &rms_forward_add_affine,
        "fused residual-add + RMSNorm fwd: returns (y, z=x+res, invvar)");
  m.def("rms_forward_affine", &rms_forward_affine, "RMSNorm fwd (affine)");
  m.def("rms_forward_affine_mixed_dtypes", &rms_forward_affine_mixed_dtypes,
        "RMSNorm fwd, low-precision input with fp32 params");
  m.def("rms_forward", &rms_forward_plain, "RMSNorm fwd (no affine)");
  m.def("rms_backward_affine", &rms_backward_affine, "RMSNorm bwd (affine)");
  m.def("rms_backward", &rms_backward_plain, "RMSNorm bwd (no affine)");
}
