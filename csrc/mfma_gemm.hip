// apex_amd._mfma — hand-written MFMA-tile GEMM with fused bias(+GELU)
// epilogue for gfx950, used where hipBLASLt has no fused algorithm (bf16
// GELU aux) and as the library's reference MFMA implementation.
//
// Structure (guide §5, the verified 128^2 ladder): 128x128 output tile per
// 256-thread workgroup (4 waves in 2x2, 64x64 per wave as 4x4 fragments of
// mfma_f32_16x16x32_bf16), K stepped by 32 with A/B tiles staged through
// LDS via 16-byte global_load_lds, 2 barriers per K-step, XCD-aware
// workgroup swizzle (guide T1). Epilogue applies bias and tanh-GELU in
// registers before one coalesced store — the fusion hipBLASLt cannot do for
// bf16.
//
// Fragment layouts verified on-device by mfma_tile_probe (csrc/mfma_probe.hip):
//   A[r][k]: lane l -> r = l%16, k = (l/16)*8 + j
//   B[k][c]: lane l -> c = l%16, k = (l/16)*8 + j
//   D[r][c]: lane l reg q -> r = (l>>4)*4 + q, c = l&15
#include "common.h"

#include <vector>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;
constexpr int NTHREADS = 256;

constexpr float kGeluC = 0.7978845608028654f;  // sqrt(2/pi)
constexpr float kGeluA = 0.044715f;

__device__ __forceinline__ float gelu_tanh(float v) {
  const float t = tanhf(kGeluC * (v + kGeluA * v * v * v));
  return 0.5f * v * (1.f + t);
}

// EPI: 0 = none, 1 = bias, 2 = bias+gelu (gelu_in saved when provided)
template <int EPI>
__global__ void __launch_bounds__(NTHREADS, 2) gemm_bt_mfma_kernel(
    const short* __restrict__ X /* [M,K] bf16 */, const short* __restrict__ W /* [N,K] bf16 */,
    const float* __restrict__ bias /* [N] fp32 */, short* __restrict__ out /* [M,N] bf16 */,
    short* __restrict__ gelu_in /* [M,N] bf16 or null */, int M, int N, int K) {
  // XCD-aware swizzle of the linear workgroup id (guide T1, bijective form)
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    const int nx = 8;
    const int q = nwg / nx, r = nwg % nx;
    const int xcd = wgid % nx, idx = wgid / nx;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntiles_n = N / BN;
  const int m0 = (wgid / ntiles_n) * BM;
  const int n0 = (wgid % ntiles_n) * BN;

  __shared__ short lds_a[BM * BK];  // [m][k] row-major
  __shared__ short lds_b[BN * BK];  // [n][k] row-major

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;  // wave row offset in tile
  const int wc = (wave & 1) * 64;   // wave col offset in tile

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // staging: BM*BK bf16 = 8192 B per tile = 512 x 16B → each of the 256
  // threads issues 2 16-byte global_load_lds per tile (A and B alike).
  // linear element index for this thread's two loads: tid*8 within rows.
  for (int kt = 0; kt < K; kt += BK) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      // elements [e0, e0+8) of the [128][32] tile, linear
      const int e0 = (half * NTHREADS + tid) * 8;
      const int row = e0 / BK;
      const int col = e0 % BK;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(X + (long)(m0 + row) * K + kt + col),
          (__attribute__((address_space(3))) uint32_t*)(lds_a + e0), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(W + (long)(n0 + row) * K + kt + col),
          (__attribute__((address_space(3))) uint32_t*)(lds_b + e0), 16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();

    // fragments: a[fr] = A rows (wr+fr*16), k-slice (lane/16)*8
    bf16x8 a[4], b[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const int ar = wr + f * 16 + (lane & 15);
      a[f] = *reinterpret_cast<const bf16x8*>(lds_a + ar * BK + (lane >> 4) * 8);
      const int bc = wc + f * 16 + (lane & 15);
      b[f] = *reinterpret_cast<const bf16x8*>(lds_b + bc * BK + (lane >> 4) * 8);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j], acc[i][j], 0, 0, 0);
    __syncthreads();
  }

  // epilogue: D[r][c], lane q -> row (lane>>4)*4+q, col lane&15
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = n0 + wc + j * 16 + (lane & 15);
      const float bv = EPI >= 1 ? bias[col] : 0.f;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const long row = m0 + wr + i * 16 + ((lane >> 4) * 4 + q);
        float v = acc[i][j][q] + bv;
        if (EPI == 2) {
          if (gelu_in) {
            __hip_bfloat16 gi = __float2bfloat16(v);
            gelu_in[row * N + col] = *reinterpret_cast<short*>(&gi);
          }
          v = gelu_tanh(v);
        }
        __hip_bfloat16 o = __float2bfloat16(v);
        out[row * N + col] = *reinterpret_cast<short*>(&o);
      }
    }
  }
}

bool mfma_shape_ok(long M, long N, long K) {
  return M % BM == 0 && N % BN == 0 && K % BK == 0 && M > 0 && N > 0 && K > 0;
}

// ---- v2: BK=64 with XOR-swizzled LDS (guide T2 / G4) ----
// Rows are 128 B (64 bf16); a wave's ds_read_b128 has 16 lanes hitting the
// same 16B column of consecutive rows → 8-way bank conflict unswizzled.
// byte ^= ((row&7)<<4) spreads the 8-row group across eight 16B slots
// (2-way residual aliasing is free on CDNA4). global_load_lds writes
// linearly, so the SOURCE address is pre-swizzled with the same involution
// and fragment reads re-apply it (both-sides-or-neither).
constexpr int BK2 = 64;

__device__ __forceinline__ int swz128(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

template <int EPI>
__global__ void __launch_bounds__(NTHREADS, 2) gemm_bt_mfma_v2_kernel(
    const short* __restrict__ X, const short* __restrict__ W, const float* __restrict__ bias,
    short* __restrict__ out, short* __restrict__ gelu_in, int M, int N, int K) {
  const int nwg = gridDim.x;
  int wgid = blockIdx.x;
  {
    const int nx = 8;
    const int q = nwg / nx, r = nwg % nx;
    const int xcd = wgid % nx, idx = wgid / nx;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int ntiles_n = N / BN;
  const int m0 = (wgid / ntiles_n) * BM;
  const int n0 = (wgid % ntiles_n) * BN;

  __shared__ short lds_a[BM * BK2];
  __shared__ short lds_b[BN * BK2];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int kt = 0; kt < K; kt += BK2) {
    // stage: 128x64 bf16 = 16 KiB per matrix → 4 x 16B per lane, linear LDS
    // dest, pre-swizzled global source.
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int o_l = (pass * NTHREADS + tid) * 16;   // linear LDS byte
      const int o_s = swz128(o_l);                    // swizzled source byte
      const int row = o_s >> 7;                       // /128 B per row
      const int col = (o_s & 127) >> 1;               // bf16 column
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(X + (long)(m0 + row) * K + kt + col),
          (__attribute__((address_space(3))) uint32_t*)((char*)lds_a + o_l), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)(W + (long)(n0 + row) * K + kt + col),
          (__attribute__((address_space(3))) uint32_t*)((char*)lds_b + o_l), 16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();

    bf16x8 a[4][2], b[4][2];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int ar = wr + f * 16 + (lane & 15);
        const int abyte = swz128(ar * 128 + (lane >> 4) * 16 + kk * 64);
        a[f][kk] = *reinterpret_cast<const bf16x8*>((const char*)lds_a + abyte);
        const int bc = wc + f * 16 + (lane & 15);
        const int bbyte = swz128(bc * 128 + (lane >> 4) * 16 + kk * 64);
        b[f][kk] = *reinterpret_cast<const bf16x8*>((const char*)lds_b + bbyte);
      }
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i][kk], b[j][kk], acc[i][j], 0, 0, 0);
    __syncthreads();
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int col = n0 + wc + j * 16 + (lane & 15);
      const float bv = EPI >= 1 ? bias[col] : 0.f;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const long row = m0 + wr + i * 16 + ((lane >> 4) * 4 + q);
        float v = acc[i][j][q] + bv;
        if (EPI == 2) {
          if (gelu_in) {
            __hip_bfloat16 gi = __float2bfloat16(v);
            gelu_in[row * N + col] = *reinterpret_cast<short*>(&gi);
          }
          v = gelu_tanh(v);
        }
        __hip_bfloat16 o = __float2bfloat16(v);
        out[row * N + col] = *reinterpret_cast<short*>(&o);
      }
    }
  }
}

}  // namespace

// out[m,n] = gelu(X[m,k] @ W[n,k]^T + bias), returns (out, gelu_in)
std::vector<at::Tensor> gemm_bias_gelu_mfma(at::Tensor X, at::Tensor W, at::Tensor bias,
                                            bool save_gelu_in) {
  TORCH_CHECK(X.scalar_type() == at::ScalarType::BFloat16 &&
              W.scalar_type() == at::ScalarType::BFloat16, "bf16 only");
  auto x = X.contiguous();
  auto w = W.contiguous();
  auto b32 = bias.to(at::kFloat).contiguous();
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(mfma_shape_ok(M, N, K), "gemm_bias_gelu_mfma: shape must tile by 128x128x32");
  auto out = at::empty({M, N}, x.options());
  auto gelu_in = save_gelu_in ? at::empty({M, N}, x.options()) : at::Tensor();
  const int grid = (int)((M / BM) * (N / BN));
  hipLaunchKernelGGL((gemm_bt_mfma_kernel<2>), dim3(grid), dim3(NTHREADS), 0, current_stream(),
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     b32.data_ptr<float>(), (short*)out.data_ptr(),
                     save_gelu_in ? (short*)gelu_in.data_ptr() : nullptr, (int)M, (int)N, (int)K);
  HIP_CHECK(hipGetLastError());
  return {out, gelu_in};
}

std::vector<at::Tensor> gemm_bias_gelu_mfma_v2(at::Tensor X, at::Tensor W, at::Tensor bias,
                                               bool save_gelu_in) {
  auto x = X.contiguous();
  auto w = W.contiguous();
  auto b32 = bias.to(at::kFloat).contiguous();
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK2 == 0, "v2: shape must tile by 128x128x64");
  auto out = at::empty({M, N}, x.options());
  auto gelu_in = save_gelu_in ? at::empty({M, N}, x.options()) : at::Tensor();
  const int grid = (int)((M / BM) * (N / BN));
  hipLaunchKernelGGL((gemm_bt_mfma_v2_kernel<2>), dim3(grid), dim3(NTHREADS), 0,
                     current_stream(), (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     b32.data_ptr<float>(), (short*)out.data_ptr(),
                     save_gelu_in ? (short*)gelu_in.data_ptr() : nullptr, (int)M, (int)N, (int)K);
  HIP_CHECK(hipGetLastError());
  return {out, gelu_in};
}

at::Tensor gemm_bias_mfma_v2(at::Tensor X, at::Tensor W, at::Tensor bias) {
  auto x = X.contiguous();
  auto w = W.contiguous();
  auto b32 = bias.to(at::kFloat).contiguous();
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK2 == 0, "v2: shape must tile by 128x128x64");
  auto out = at::empty({M, N}, x.options());
  const int grid = (int)((M / BM) * (N / BN));
  hipLaunchKernelGGL((gemm_bt_mfma_v2_kernel<1>), dim3(grid), dim3(NTHREADS), 0,
                     current_stream(), (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     b32.data_ptr<float>(), (short*)out.data_ptr(), nullptr, (int)M, (int)N,
                     (int)K);
  HIP_CHECK(hipGetLastError());
  return out;
}

at::Tensor gemm_bias_mfma(at::Tensor X, at::Tensor W, at::Tensor bias) {
  auto x = X.contiguous();
  auto w = W.contiguous();
  auto b32 = bias.to(at::kFloat).contiguous();
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(mfma_shape_ok(M, N, K), "gemm_bias_mfma: shape must tile by 128x128x32");
  auto out = at::empty({M, N}, x.options());
  const int grid = (int)((M / BM) * (N / BN));
  hipLaunchKernelGGL((gemm_bt_mfma_kernel<1>), dim3(grid), dim3(NTHREADS), 0, current_stream(),
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     b32.data_ptr<float>(), (short*)out.data_ptr(), nullptr, (int)M, (int)N,
                     (int)K);
  HIP_CHECK(hipGetLastError());
  return out;
}

at::Tensor mfma_tile_probe(at::Tensor A_bf16, at::Tensor B_bf16);

std::vector<at::Tensor> fmha_fwd(at::Tensor q, at::Tensor k, at::Tensor v, bool causal,
                                 double scale, double dropout_p, long seed);
std::vector<at::Tensor> fmha_bwd(at::Tensor dout, at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor out, at::Tensor lse, bool causal, double scale,
                                 double dropout_p, long seed);
at::Tensor fmha_delta(at::Tensor dout, at::Tensor out);
at::Tensor fmha_p(at::Tensor s, at::Tensor lse, double scale, bool causal);
at::Tensor fmha_ds(at::Tensor p, at::Tensor dp, at::Tensor delta, double scale);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fmha_fwd", &fmha_fwd,
        "flash-attention fwd (bf16, D=64/128, philox dropout) -> (out, lse)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("causal"), py::arg("scale"),
        py::arg("dropout_p") = 0.0, py::arg("seed") = 0);
  m.def("fmha_bwd", &fmha_bwd,
        "flash-attention MFMA bwd (mask-free dropout) -> (dq, dk, dv)",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"), py::arg("out"),
        py::arg("lse"), py::arg("causal"), py::arg("scale"),
        py::arg("dropout_p") = 0.0, py::arg("seed") = 0);
  m.def("fmha_delta", &fmha_delta, "rowsum(dout*out) fp32 (one pass)");
  m.def("fmha_p", &fmha_p, "p = exp(s*scale - lse[row]) (+causal mask), bf16 one pass");
  m.def("fmha_ds", &fmha_ds, "ds = p*(dp - delta[row])*scale, bf16 one pass");
  m.def("gemm_bias_gelu", &gemm_bias_gelu_mfma, "bf16 MFMA GEMM + bias + tanh-GELU (fused)");
  m.def("gemm_bias", &gemm_bias_mfma, "bf16 MFMA GEMM + bias");
  m.def("mfma_tile_probe", &mfma_tile_probe, "single-tile fragment-layout verification");
  m.def("gemm_bias_gelu_v2", &gemm_bias_gelu_mfma_v2, "BK=64 + swizzled-LDS variant");
  m.def("gemm_bias_v2", &gemm_bias_mfma_v2, "BK=64 + swizzled-LDS variant");
}
