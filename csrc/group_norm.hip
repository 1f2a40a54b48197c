// apex_amd._group_norm — NHWC GroupNorm with optional fused SiLU (the
// diffusion-UNet norm). Reference surface: apex/contrib/group_norm
// (group_norm_nhwc fwd/bwd two-pass + one-pass persistent kernels,
// GroupNorm module with silent torch fallback for unsupported channel
// counts — group_norm.py:213-232).
//
// MI355X design: shape-generic two-pass kernels (no fixed channel-count
// registry — the reference's one-pass kernel list exists because sm90 LDS
// persistence needed per-shape tuning; on gfx950 the two-pass form is
// HBM-streaming-bound either way, so we keep one generic implementation):
//   pass 1: per-(n, group) mean/rstd — blocks slice HW rows, lanes own
//           channels within the group (fully coalesced NHWC), fixed-order
//           merge for determinism.
//   pass 2: elementwise normalize + affine + optional SiLU.
// Backward mirrors it; dgamma/dbeta use the NHWC column-reduction pattern.
#include "common.h"

#include <unordered_map>
#include <vector>

namespace {

constexpr int GN_BLOCK = 256;

// ---- forward stats: grid (N*G, S). Each block reduces a row-slice of its
// (n, g) group; partials merged by a second tiny kernel (deterministic).
template <typename T>
__global__ void __launch_bounds__(GN_BLOCK) gn_fwd_stats_kernel(
    const T* __restrict__ x, float* __restrict__ part /* [N*G, S, 2] */, long N, long HW, long C,
    long G, int S) {
  const long ng = blockIdx.x;
  const long n = ng / G, g = ng % G;
  const long cpg = C / G;
  const long c0 = g * cpg;
  const int s = blockIdx.y;
  const long per = (HW + S - 1) / S;
  const long r0 = (long)s * per, r1 = min(r0 + per, HW);

  float sum = 0.f, sq = 0.f;
  // threads sweep (row, channel-in-group) pairs; channel fastest → coalesced
  for (long i = threadIdx.x; i < (r1 - r0) * cpg; i += blockDim.x) {
    const long r = r0 + i / cpg;
    const long c = c0 + i % cpg;
    const float v = to_float(x[(n * HW + r) * C + c]);
    sum += v;
    sq = fmaf(v, v, sq);
  }
  __shared__ float smem[GN_BLOCK / WAVE_SIZE];
  sum = block_reduce_sum(sum, smem);
  sq = block_reduce_sum(sq, smem);
  if (threadIdx.x == 0) {
    part[(ng * S + s) * 2] = sum;
    part[(ng * S + s) * 2 + 1] = sq;
  }
}

__global__ void gn_fwd_finish_kernel(const float* __restrict__ part, float* __restrict__ mean,
                                     float* __restrict__ rstd, long NG, long count, int S,
                                     float eps) {
  const long ng = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (ng >= NG) return;
  float sum = 0.f, sq = 0.f;
  for (int s = 0; s < S; ++s) {
    sum += part[(ng * S + s) * 2];
    sq += part[(ng * S + s) * 2 + 1];
  }
  const float mu = sum / count;
  const float var = fmaxf(sq / count - mu * mu, 0.f);
  mean[ng] = mu;
  rstd[ng] = rsqrtf(var + eps);
}

template <typename T, bool AFFINE, bool SILU>
__global__ void __launch_bounds__(GN_BLOCK) gn_fwd_apply_kernel(
    const T* __restrict__ x, T* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ w, const float* __restrict__ b,
    long total, long HW, long C, long G) {
  const long cpg = C / G;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long c = i % C;
    const long n = i / (HW * C);
    const long ng = n * G + c / cpg;
    float v = (to_float(x[i]) - mean[ng]) * rstd[ng];
    if (AFFINE) v = fmaf(v, w[c], b[c]);
    if (SILU) v = v / (1.f + __expf(-v));
    y[i] = from_float<T>(v);
  }
}

// ---- backward ----
// With SILU, dy is first mapped through dsilu(z) where z is the
// pre-activation (recomputed from x, stats, affine).
template <typename T, bool AFFINE, bool SILU>
__device__ __forceinline__ float eff_dy(const T* x, const T* dy, long i, long c, float mu,
                                        float rs, const float* w, const float* b) {
  float d = to_float(dy[i]);
  if (SILU) {
    float z = (to_float(x[i]) - mu) * rs;
    if (AFFINE) z = fmaf(z, w[c], b[c]);
    const float sig = 1.f / (1.f + __expf(-z));
    d *= sig * (1.f + z * (1.f - sig));
  }
  return d;
}

// per-(n,g): s1 = sum(dyw * xhat), s2 = sum(dyw), dyw = eff_dy * gamma[c]
template <typename T, bool AFFINE, bool SILU>
__global__ void __launch_bounds__(GN_BLOCK) gn_bwd_stats_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, float* __restrict__ part, long N, long HW,
    long C, long G, int S, const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ w, const float* __restrict__ b) {
  const long ng = blockIdx.x;
  const long n = ng / G, g = ng % G;
  const long cpg = C / G;
  const long c0 = g * cpg;
  const int s = blockIdx.y;
  const long per = (HW + S - 1) / S;
  const long r0 = (long)s * per, r1 = min(r0 + per, HW);
  const float mu = mean[ng], rs = rstd[ng];

  float s1 = 0.f, s2 = 0.f;
  for (long i = threadIdx.x; i < (r1 - r0) * cpg; i += blockDim.x) {
    const long r = r0 + i / cpg;
    const long c = c0 + i % cpg;
    const long idx = (n * HW + r) * C + c;
    float d = eff_dy<T, AFFINE, SILU>(x, dy, idx, c, mu, rs, w, b);
    if (AFFINE) d *= w[c];
    const float xhat = (to_float(x[idx]) - mu) * rs;
    s1 = fmaf(d, xhat, s1);
    s2 += d;
  }
  __shared__ float smem[GN_BLOCK / WAVE_SIZE];
  s1 = block_reduce_sum(s1, smem);
  s2 = block_reduce_sum(s2, smem);
  if (threadIdx.x == 0) {
    part[(ng * S + s) * 2] = s1;
    part[(ng * S + s) * 2 + 1] = s2;
  }
}

__global__ void gn_bwd_finish_kernel(const float* __restrict__ part, float* __restrict__ sums,
                                     long NG, int S) {
  const long ng = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (ng >= NG) return;
  float s1 = 0.f, s2 = 0.f;
  for (int s = 0; s < S; ++s) {
    s1 += part[(ng * S + s) * 2];
    s2 += part[(ng * S + s) * 2 + 1];
  }
  sums[ng * 2] = s1;
  sums[ng * 2 + 1] = s2;
}

template <typename T, bool AFFINE, bool SILU>
__global__ void __launch_bounds__(GN_BLOCK) gn_bwd_apply_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ sums, const float* __restrict__ w, const float* __restrict__ b,
    long total, long HW, long C, long G) {
  const long cpg = C / G;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long c = i % C;
    const long n = i / (HW * C);
    const long ng = n * G + c / cpg;
    const float mu = mean[ng], rs = rstd[ng];
    const float cnt = (float)(HW * cpg);
    float d = eff_dy<T, AFFINE, SILU>(x, dy, i, c, mu, rs, w, b);
    if (AFFINE) d *= w[c];
    const float xhat = (to_float(x[i]) - mu) * rs;
    const float r = rs * (d - sums[ng * 2 + 1] / cnt - xhat * sums[ng * 2] / cnt);
    dx[i] = from_float<T>(r);
  }
}

// dgamma/dbeta: per-channel column reduction over (n, hw) with per-row stats
template <typename T, bool AFFINE, bool SILU>
__global__ void __launch_bounds__(GN_BLOCK) gn_bwd_wgrad_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, float* __restrict__ part_gw,
    float* __restrict__ part_gb, const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ w, const float* __restrict__ b, long N, long HW, long C, long G,
    int S) {
  const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const int s = blockIdx.y;
  if (c >= C) return;
  const long cpg = C / G;
  const long g = c / cpg;
  const long rows = N * HW;
  const long per = (rows + S - 1) / S;
  const long r0 = (long)s * per, r1 = min(r0 + per, rows);
  float gw = 0.f, gb = 0.f;
  for (long r = r0; r < r1; ++r) {
    const long n = r / HW;
    const long ng = n * G + g;
    const float mu = mean[ng], rs = rstd[ng];
    const long idx = r * C + c;
    const float d = eff_dy<T, AFFINE, SILU>(x, dy, idx, c, mu, rs, w, b);
    const float xhat = (to_float(x[idx]) - mu) * rs;
    gw = fmaf(d, xhat, gw);
    gb += d;
  }
  part_gw[s * C + c] = gw;
  part_gb[s * C + c] = gb;
}

__global__ void gn_bwd_wgrad_finish_kernel(const float* __restrict__ part_gw,
                                           const float* __restrict__ part_gb,
                                           float* __restrict__ gw, float* __restrict__ gb, long C,
                                           int S) {
  const long c = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float a = 0.f, bsum = 0.f;
  for (int s = 0; s < S; ++s) {
    a += part_gw[s * C + c];
    bsum += part_gb[s * C + c];
  }
  gw[c] = a;
  gb[c] = bsum;
}

int gn_splits(long work) {
  return (int)std::max<long>(1, std::min<long>(32, (work + 16383) / 16384));
}

// ---- one-pass forward (reference analogue:
// group_norm_nhwc_fwd_one_pass_kernel.cuh) ----
// One workgroup per (n, g): the slab (HW x cpg) is read for the stats, then
// re-read for the normalize pass. Unlike the two-pass form — where ALL
// slabs are swept between the two reads so the apply kernel always misses
// L2 — the re-read here happens while the block's own slab (<= ~1 MB) is
// still resident in its XCD's 4 MB L2, so HBM sees the data once. An LDS
// slab was tried first and was SLOWER (profiles/probe_group_norm.log r2
// call 5): 80-128 KB of LDS per workgroup caps the CU at 1 block / 4 waves,
// too little latency hiding for a streaming kernel. No LDS, full occupancy,
// L2 provides the persistence.
constexpr long GN_ONEPASS_SLAB_BYTES = 1 * 1024 * 1024;  // per-block L2 budget

template <typename T, int N>
struct VecN {
  alignas(sizeof(T) * N) T a[N];
};
template <typename T, int N>
__device__ __forceinline__ void load_vecn(VecN<T, N>& d, const T* s) {
  d = *reinterpret_cast<const VecN<T, N>*>(s);
}
template <typename T, int N>
__device__ __forceinline__ void store_vecn(T* d, const VecN<T, N>& s) {
  *reinterpret_cast<VecN<T, N>*>(d) = s;
}

// VW = elements per vector access (8 when the per-group channel count
// divides by 8, else 4/2/1) — the r2 scalar form measured 250-500 GB/s and
// LOST to two-pass; the vectorized form makes the single launch pay off at
// the launch-bound diffusion shapes.
template <typename T, bool AFFINE, bool SILU, int VW>
__global__ void __launch_bounds__(GN_BLOCK) gn_fwd_onepass_kernel(
    const T* __restrict__ x, T* __restrict__ y, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, const float* __restrict__ w, const float* __restrict__ b,
    long HW, long C, long G, float eps) {
  const long ng = blockIdx.x;
  const long n = ng / G, g = ng % G;
  const long cpg = C / G;
  const long c0 = g * cpg;
  const long count = HW * cpg;
  const long vpr = cpg / VW;          // vectors per row
  const long nvec = HW * vpr;
  const T* xp = x + n * HW * C + c0;

  // pass A: stats (VW-wide vector loads; consecutive threads take
  // consecutive vectors within a row)
  float sum = 0.f, sq = 0.f;
  for (long vi = threadIdx.x; vi < nvec; vi += blockDim.x) {
    const long r = vi / vpr;
    const long cc = (vi % vpr) * VW;
    VecN<T, VW> v;
    load_vecn(v, xp + r * C + cc);
#pragma unroll
    for (int j = 0; j < VW; ++j) {
      const float f = to_float(v.a[j]);
      sum += f;
      sq = fmaf(f, f, sq);
    }
  }
  __shared__ float smem[GN_BLOCK / WAVE_SIZE];
  sum = block_reduce_sum(sum, smem);
  sq = block_reduce_sum(sq, smem);
  __shared__ float s_mu, s_rs;
  if (threadIdx.x == 0) {
    const float mu = sum / count;
    const float var = fmaxf(sq / count - mu * mu, 0.f);
    s_mu = mu;
    s_rs = rsqrtf(var + eps);
    mean_out[ng] = mu;
    rstd_out[ng] = s_rs;
  }
  __syncthreads();
  const float mu = s_mu, rs = s_rs;

  // pass B: normalize from the (L2-resident) slab, vectorized both ways
  T* yp = y + n * HW * C + c0;
  for (long vi = threadIdx.x; vi < nvec; vi += blockDim.x) {
    const long r = vi / vpr;
    const long cc = (vi % vpr) * VW;
    VecN<T, VW> v, o;
    load_vecn(v, xp + r * C + cc);
#pragma unroll
    for (int j = 0; j < VW; ++j) {
      float f = (to_float(v.a[j]) - mu) * rs;
      if (AFFINE) f = fmaf(f, w[c0 + cc + j], b[c0 + cc + j]);
      if (SILU) f = f / (1.f + __expf(-f));
      o.a[j] = from_float<T>(f);
    }
    store_vecn(yp + r * C + cc, o);
  }
}

}  // namespace

// x: [N, H, W, C] NHWC. Returns (y, mean[N*G], rstd[N*G]).
// passes: 0 = auto heuristic, 1 = force one-pass (if eligible), 2 = force
// two-pass — mirrors the reference's `passes` knob (group_norm.py:213-232).
std::vector<at::Tensor> group_norm_nhwc_fwd(at::Tensor x, c10::optional<at::Tensor> weight,
                                            c10::optional<at::Tensor> bias, long G, double eps,
                                            bool silu, long passes) {
  auto xc = x.contiguous();
  const long N = xc.size(0), C = xc.size(-1);
  const long HW = xc.numel() / (N * C);
  const long cpg = C / G;
  TORCH_CHECK(C % G == 0, "channels not divisible by groups");
  auto fopts = xc.options().dtype(at::kFloat);
  auto y = at::empty_like(xc);
  auto mean = at::empty({N * G}, fopts);
  auto rstd = at::empty({N * G}, fopts);
  const int S = gn_splits(HW * cpg);
  auto part = at::empty({N * G, S, 2}, fopts);
  const bool affine = weight.has_value() && weight->defined();
  auto w32 = affine ? weight->to(at::kFloat).contiguous() : at::Tensor();
  auto b32 = affine ? bias->to(at::kFloat).contiguous() : at::Tensor();
  auto stream = current_stream();

  // MEASURED (profiles/probe_group_norm3.log): with VW-wide vector access
  // the one-pass form beats two-pass 1.4-2.7x when the per-group channel
  // count vectorizes 4+ wide and the slab stays L2-resident (C >= 512 at
  // G=32); narrow-vector shapes (C=320 -> cpg=10) still lose, so auto keeps
  // two-pass there. passes=1/2 force either form (reference knob surface).
  const long slab_bytes = HW * cpg * (long)xc.element_size();
  const int vw_elig_max = (int)(16 / xc.element_size());
  int vw_elig = 1;
  for (int cand : {8, 4, 2}) {
    if (cand <= vw_elig_max && (cpg % cand) == 0) { vw_elig = cand; break; }
  }
  const bool eligible = slab_bytes <= GN_ONEPASS_SLAB_BYTES;
  const bool one_pass = eligible &&
      (passes == 1 ||
       (passes == 0 && vw_elig >= 4 && slab_bytes <= 256 * 1024 && N * G >= 128));

  APEX_DISPATCH_FLOAT_HALF_BF(xc.scalar_type(), "group_norm_nhwc_fwd", ([&] {
    if (one_pass) {
      // widest vector dividing the per-group channel count, <= 16 bytes
      const int vw_max = (int)(16 / xc.element_size());
      int vw = 1;
      for (int cand : {8, 4, 2}) {
        if (cand <= vw_max && (cpg % cand) == 0) { vw = cand; break; }
      }
      auto launch1 = [&](auto aff, auto sl) {
        auto run = [&](auto vwc) {
          auto kfn = gn_fwd_onepass_kernel<scalar_t, decltype(aff)::value,
                                           decltype(sl)::value, decltype(vwc)::value>;
          hipLaunchKernelGGL(kfn, dim3((uint32_t)(N * G)), dim3(GN_BLOCK), 0, stream,
                             (const scalar_t*)xc.data_ptr(), (scalar_t*)y.data_ptr(),
                             mean.data_ptr<float>(), rstd.data_ptr<float>(),
                             affine ? w32.data_ptr<float>() : nullptr,
                             affine ? b32.data_ptr<float>() : nullptr, HW, C, G, (float)eps);
        };
        if (vw == 8) run(std::integral_constant<int, 8>{});
        else if (vw == 4) run(std::integral_constant<int, 4>{});
        else if (vw == 2) run(std::integral_constant<int, 2>{});
        else run(std::integral_constant<int, 1>{});
      };
      using Tt = std::true_type;
      using Ff = std::false_type;
      if (affine && silu) launch1(Tt{}, Tt{});
      else if (affine) launch1(Tt{}, Ff{});
      else if (silu) launch1(Ff{}, Tt{});
      else launch1(Ff{}, Ff{});
      HIP_CHECK(hipGetLastError());
      return;
    }
    hipLaunchKernelGGL((gn_fwd_stats_kernel<scalar_t>), dim3((uint32_t)(N * G), S),
                       dim3(GN_BLOCK), 0, stream, (const scalar_t*)xc.data_ptr(),
                       part.data_ptr<float>(), N, HW, C, G, S);
    HIP_CHECK(hipGetLastError());
    hipLaunchKernelGGL(gn_fwd_finish_kernel,
                       dim3((uint32_t)((N * G + GN_BLOCK - 1) / GN_BLOCK)), dim3(GN_BLOCK), 0,
                       stream, part.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), N * G, HW * cpg, S, (float)eps);
    HIP_CHECK(hipGetLastError());
    const long total = xc.numel();
    const int grid = (int)std::min<long>((total + GN_BLOCK - 1) / GN_BLOCK, 8192);
    auto launch = [&](auto aff, auto sl) {
      hipLaunchKernelGGL((gn_fwd_apply_kernel<scalar_t, decltype(aff)::value, decltype(sl)::value>),
                         dim3(grid), dim3(GN_BLOCK), 0, stream, (const scalar_t*)xc.data_ptr(),
                         (scalar_t*)y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         affine ? w32.data_ptr<float>() : nullptr,
                         affine ? b32.data_ptr<float>() : nullptr, total, HW, C, G);
    };
    using Tt = std::true_type;
    using Ff = std::false_type;
    if (affine && silu) launch(Tt{}, Tt{});
    else if (affine) launch(Tt{}, Ff{});
    else if (silu) launch(Ff{}, Tt{});
    else launch(Ff{}, Ff{});
    HIP_CHECK(hipGetLastError());
  }()));
  return {y, mean, rstd};
}

std::vector<at::Tensor> group_norm_nhwc_bwd(at::Tensor dy, at::Tensor x, at::Tensor mean,
                                            at::Tensor rstd, c10::optional<at::Tensor> weight,
                                            c10::optional<at::Tensor> bias, long G, bool silu) {
  auto xc = x.contiguous();
  auto dyc = dy.contiguous();
  const long N = xc.size(0), C = xc.size(-1);
  const long HW = xc.numel() / (N * C);
  const bool affine = weight.has_value() && weight->defined();
  auto fopts = xc.options().dtype(at::kFloat);
  auto dx = at::empty_like(xc);
  auto w32 = affine ? weight->to(at::kFloat).contiguous() : at::Tensor();
  auto b32 = affine ? bias->to(at::kFloat).contiguous() : at::Tensor();
  auto gw = at::empty({C}, fopts);
  auto gb = at::empty({C}, fopts);
  auto stream = current_stream();
  const long cpg = C / G;
  const int S = gn_splits(HW * cpg);
  auto part = at::empty({N * G, S, 2}, fopts);
  auto sums = at::empty({N * G, 2}, fopts);
  const int Sw = gn_splits(N * HW);
  auto part_gw = at::empty({Sw, C}, fopts);
  auto part_gb = at::empty({Sw, C}, fopts);

  APEX_DISPATCH_FLOAT_HALF_BF(xc.scalar_type(), "group_norm_nhwc_bwd", ([&] {
    auto launch = [&](auto aff, auto sl) {
      constexpr bool AF = decltype(aff)::value;
      constexpr bool SL = decltype(sl)::value;
      hipLaunchKernelGGL((gn_bwd_stats_kernel<scalar_t, AF, SL>), dim3((uint32_t)(N * G), S),
                         dim3(GN_BLOCK), 0, stream, (const scalar_t*)xc.data_ptr(),
                         (const scalar_t*)dyc.data_ptr(), part.data_ptr<float>(), N, HW, C, G, S,
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         AF ? w32.data_ptr<float>() : nullptr,
                         AF ? b32.data_ptr<float>() : nullptr);
      HIP_CHECK(hipGetLastError());
      hipLaunchKernelGGL(gn_bwd_finish_kernel,
                         dim3((uint32_t)((N * G + GN_BLOCK - 1) / GN_BLOCK)), dim3(GN_BLOCK), 0,
                         stream, part.data_ptr<float>(), sums.data_ptr<float>(), N * G, S);
      HIP_CHECK(hipGetLastError());
      const long total = xc.numel();
      const int grid = (int)std::min<long>((total + GN_BLOCK - 1) / GN_BLOCK, 8192);
      hipLaunchKernelGGL((gn_bwd_apply_kernel<scalar_t, AF, SL>), dim3(grid), dim3(GN_BLOCK), 0,
                         stream, (const scalar_t*)xc.data_ptr(), (const scalar_t*)dyc.data_ptr(),
                         (scalar_t*)dx.data_ptr(), mean.data_ptr<float>(),
                         rstd.data_ptr<float>(), sums.data_ptr<float>(),
                         AF ? w32.data_ptr<float>() : nullptr,
                         AF ? b32.data_ptr<float>() : nullptr, total, HW, C, G);
      HIP_CHECK(hipGetLastError());
      hipLaunchKernelGGL((gn_bwd_wgrad_kernel<scalar_t, AF, SL>),
                         dim3((uint32_t)((C + GN_BLOCK - 1) / GN_BLOCK), Sw), dim3(GN_BLOCK), 0,
                         stream, (const scalar_t*)xc.data_ptr(), (const scalar_t*)dyc.data_ptr(),
                         part_gw.data_ptr<float>(), part_gb.data_ptr<float>(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         AF ? w32.data_ptr<float>() : nullptr,
                         AF ? b32.data_ptr<float>() : nullptr, N, HW, C, G, Sw);
      HIP_CHECK(hipGetLastError());
      hipLaunchKernelGGL(gn_bwd_wgrad_finish_kernel,
                         dim3((uint32_t)((C + GN_BLOCK - 1) / GN_BLOCK)), dim3(GN_BLOCK), 0,
                         stream, part_gw.data_ptr<float>(), part_gb.data_ptr<float>(),
                         gw.data_ptr<float>(), gb.data_ptr<float>(), C, Sw);
      HIP_CHECK(hipGetLastError());
    };
    using Tt = std::true_type;
    using Ff = std::false_type;
    if (affine && silu) launch(Tt{}, Tt{});
    else if (affine) launch(Tt{}, Ff{});
    else if (silu) launch(Ff{}, Tt{});
    else launch(Ff{}, Ff{});
  }()));
  return {dx, gw, gb};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fwd", &group_norm_nhwc_fwd, "NHWC GroupNorm forward (+SiLU) -> (y, mean, rstd)",
        py::arg("x"), py::arg("weight"), py::arg("bias"), py::arg("G"), py::arg("eps"),
        py::arg("silu"), py::arg("passes") = 0);
  m.def("bwd", &group_norm_nhwc_bwd, "NHWC GroupNorm backward -> (dx, dgamma, dbeta)");
}
