// apex_amd._peer_memory — intra-node HIP-IPC peer memory pool + direct-store
// 1-D halo exchange over xGMI.
// Reference surface: apex/contrib/csrc/peer_memory/peer_memory_cuda.cu
// (allocate_raw / free_raw / get_raw_ipc_address / get_raw_peers / blob_view_*
// / push_pull_halos_1d). On MI355X the peer direct store rides the 7 xGMI
// p2p links — the native analogue of the reference's CUDA-IPC halo path.
#include "common.h"

#include <vector>

namespace {

constexpr int PM_BLOCK = 256;

// Direct-store halo push: copy my edge rows into the neighbours' recv
// buffers through IPC-mapped peer pointers. Contiguous [halo_elems] blobs.
template <typename T>
__global__ void __launch_bounds__(PM_BLOCK) push_halos_kernel(
    const T* __restrict__ top_out, const T* __restrict__ btm_out, T* __restrict__ top_peer_in,
    T* __restrict__ btm_peer_in, long n_top, long n_btm) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_top;
       i += (long)gridDim.x * blockDim.x) {
    top_peer_in[i] = top_out[i];
  }
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_btm;
       i += (long)gridDim.x * blockDim.x) {
    btm_peer_in[i] = btm_out[i];
  }
}

}  // namespace

int64_t allocate_raw(int64_t size) {
  void* ptr = nullptr;
  HIP_CHECK(hipMalloc(&ptr, (size_t)size));
  HIP_CHECK(hipMemset(ptr, 0, (size_t)size));
  return reinterpret_cast<int64_t>(ptr);
}

void free_raw(int64_t raw) {
  HIP_CHECK(hipFree(reinterpret_cast<void*>(raw)));
}

at::Tensor get_raw_ipc_address(int64_t raw) {
  hipIpcMemHandle_t handle;
  HIP_CHECK(hipIpcGetMemHandle(&handle, reinterpret_cast<void*>(raw)));
  auto t = at::empty({(long)sizeof(hipIpcMemHandle_t)}, at::TensorOptions().dtype(at::kByte));
  memcpy(t.data_ptr(), &handle, sizeof(handle));
  return t;
}

std::vector<int64_t> get_raw_peers(at::Tensor ipc_addresses, int64_t peer_rank, int64_t raw) {
  TORCH_CHECK(ipc_addresses.dim() == 2 && ipc_addresses.size(1) == (long)sizeof(hipIpcMemHandle_t),
              "ipc_addresses must be [world, handle_bytes] on CPU");
  auto cpu = ipc_addresses.to(at::kCPU).contiguous();
  const long world = cpu.size(0);
  std::vector<int64_t> out((size_t)world, 0);
  for (long r = 0; r < world; ++r) {
    if (r == peer_rank) {
      out[r] = raw;
      continue;
    }
    hipIpcMemHandle_t handle;
    memcpy(&handle, cpu[r].data_ptr(), sizeof(handle));
    void* p = nullptr;
    HIP_CHECK(hipIpcOpenMemHandle(&p, handle, hipIpcMemLazyEnablePeerAccess));
    out[r] = reinterpret_cast<int64_t>(p);
  }
  return out;
}

at::Tensor blob_view(int64_t raw, std::vector<long> shape, at::ScalarType dtype,
                     bool channels_last) {
  long numel = 1;
  for (auto s : shape) numel *= s;
  auto options = at::TensorOptions().dtype(dtype).device(at::kCUDA);
  auto t = at::from_blob(reinterpret_cast<void*>(raw), shape, options);
  if (channels_last && shape.size() == 4) t = t.contiguous(at::MemoryFormat::ChannelsLast);
  return t;
}

at::Tensor blob_view_half(int64_t raw, std::vector<long> shape, bool channels_last) {
  return blob_view(raw, shape, at::ScalarType::Half, channels_last);
}
at::Tensor blob_view_float(int64_t raw, std::vector<long> shape, bool channels_last) {
  return blob_view(raw, shape, at::ScalarType::Float, channels_last);
}
at::Tensor blob_view_bfloat16(int64_t raw, std::vector<long> shape, bool channels_last) {
  return blob_view(raw, shape, at::ScalarType::BFloat16, channels_last);
}
at::Tensor blob_view_int(int64_t raw, std::vector<long> shape, bool channels_last) {
  return blob_view(raw, shape, at::ScalarType::Int, channels_last);
}

// Push this rank's top/bottom halo slabs into neighbour recv buffers
// (peer pointers), then the caller signals/wait via its process group.
void push_pull_halos_1d(at::Tensor top_out, at::Tensor btm_out, int64_t top_peer_in_raw,
                        int64_t btm_peer_in_raw) {
  const long n_top = top_out.numel();
  const long n_btm = btm_out.numel();
  const long n = std::max(n_top, n_btm);
  const int grid = (int)std::min<long>((n + PM_BLOCK - 1) / PM_BLOCK, 2048);
  APEX_DISPATCH_FLOAT_HALF_BF(top_out.scalar_type(), "push_pull_halos_1d", ([&] {
    hipLaunchKernelGGL((push_halos_kernel<scalar_t>), dim3(grid), dim3(PM_BLOCK), 0,
                       current_stream(), (const scalar_t*)top_out.data_ptr(),
                       (const scalar_t*)btm_out.data_ptr(),
                       reinterpret_cast<scalar_t*>(top_peer_in_raw),
                       reinterpret_cast<scalar_t*>(btm_peer_in_raw), n_top, n_btm);
  }()));
  HIP_CHECK(hipGetLastError());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("allocate_raw", &allocate_raw);
  m.def("free_raw", &free_raw);
  m.def("get_raw_ipc_address", &get_raw_ipc_address);
  m.def("get_raw_peers", &get_raw_peers);
  m.def("blob_view_half", &blob_view_half);
  m.def("blob_view_float", &blob_view_float);
  m.def("blob_view_bfloat16", &blob_view_bfloat16);
  m.def("blob_view_int", &blob_view_int);
  m.def("push_pull_halos_1d", &push_pull_halos_1d);
}
