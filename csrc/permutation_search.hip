// apex_amd._permutation_search — GPU stripe-pair scoring for the ASP 2:4
// channel-permutation search. Reference analogue:
// apex/contrib/sparsity/permutation_search_kernels/CUDA_kernels/
// permutation_search_kernels.cu:46-497 (permute_and_sum_after_2_to_4,
// swap_columns_sum_after_2_to_4, build_permute_map).
//
// MI355X design: ONE launch scores every stripe pair x every candidate
// repartition — grid = n_pairs blocks; each 256-thread block streams the
// rows of its pair's 2m columns (gathered through the current permutation),
// accumulates the kept magnitude of all P partitions in registers, and
// block-reduces per partition. The host then greedily applies the best
// non-overlapping improvements per sweep (one device->host copy per sweep,
// instead of the former per-pair sync).
#include "common.h"

#include <vector>

namespace {

constexpr int PS_BLOCK = 256;
constexpr int PS_M = 4;        // stripe width (2:4)
constexpr int PS_2M = 8;
constexpr int PS_P = 35;       // C(7,3) partitions of 8 cols into two stripes

// top-2 sum of 4 floats = max over the 6 pair sums
__device__ __forceinline__ float top2sum4(float a, float b, float c, float d) {
  float s = fmaxf(a + b, a + c);
  s = fmaxf(s, a + d);
  s = fmaxf(s, b + c);
  s = fmaxf(s, b + d);
  s = fmaxf(s, c + d);
  return s;
}

// scores[pair, p] = kept magnitude of partition p applied to stripe pair
// `pair` = (i, j) (enumerated i<j). cols8[pair] are the 8 source column
// indices (already permuted by the host).
__global__ void __launch_bounds__(PS_BLOCK) stripe_pair_scores_kernel(
    const float* __restrict__ w_abs, const int* __restrict__ cols8 /* [n_pairs, 8] */,
    const int* __restrict__ parts /* [P, 8] */, float* __restrict__ scores, long rows,
    long cols, int n_pairs) {
  const int pair = blockIdx.x;
  if (pair >= n_pairs) return;
  __shared__ int s_cols[PS_2M];
  __shared__ int s_parts[PS_P][PS_2M];
  if (threadIdx.x < PS_2M) s_cols[threadIdx.x] = cols8[pair * PS_2M + threadIdx.x];
  for (int i = threadIdx.x; i < PS_P * PS_2M; i += blockDim.x)
    s_parts[i / PS_2M][i % PS_2M] = parts[i];
  __syncthreads();

  float acc[PS_P];
#pragma unroll
  for (int p = 0; p < PS_P; ++p) acc[p] = 0.f;

  for (long r = threadIdx.x; r < rows; r += blockDim.x) {
    float v[PS_2M];
#pragma unroll
    for (int c = 0; c < PS_2M; ++c) v[c] = w_abs[r * cols + s_cols[c]];
#pragma unroll
    for (int p = 0; p < PS_P; ++p) {
      const int* pp = s_parts[p];
      acc[p] += top2sum4(v[pp[0]], v[pp[1]], v[pp[2]], v[pp[3]]) +
                top2sum4(v[pp[4]], v[pp[5]], v[pp[6]], v[pp[7]]);
    }
  }

  // block reduction per partition (deterministic fixed-order over waves)
  __shared__ float smem[PS_BLOCK / WAVE_SIZE];
#pragma unroll
  for (int p = 0; p < PS_P; ++p) {
    const float s = block_reduce_sum(acc[p], smem);
    if (threadIdx.x == 0) scores[(long)pair * PS_P + p] = s;
    __syncthreads();
  }
}

}  // namespace

// w_abs: [rows, cols] fp32; cols8: [n_pairs, 8] int32 gathered column ids;
// parts: [35, 8] int32. Returns [n_pairs, 35] fp32 kept magnitudes.
at::Tensor stripe_pair_scores(at::Tensor w_abs, at::Tensor cols8, at::Tensor parts) {
  TORCH_CHECK(w_abs.scalar_type() == at::ScalarType::Float, "w_abs must be fp32");
  TORCH_CHECK(parts.size(0) == PS_P && parts.size(1) == PS_2M, "parts must be [35, 8]");
  auto w = w_abs.contiguous();
  auto c8 = cols8.to(at::kInt).contiguous();
  auto pt = parts.to(at::kInt).contiguous();
  const long rows = w.size(0), cols = w.size(1);
  const int n_pairs = (int)c8.size(0);
  auto scores = at::empty({n_pairs, PS_P}, w.options());
  hipLaunchKernelGGL(stripe_pair_scores_kernel, dim3(n_pairs), dim3(PS_BLOCK), 0,
                     current_stream(), w.data_ptr<float>(), c8.data_ptr<int>(),
                     pt.data_ptr<int>(), scores.data_ptr<float>(), rows, cols, n_pairs);
  HIP_CHECK(hipGetLastError());
  return scores;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("stripe_pair_scores", &stripe_pair_scores,
        "kept-magnitude scores for all (stripe pair, repartition) candidates");
}
