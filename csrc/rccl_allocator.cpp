// apex_amd._rccl_allocator — ncclMemAlloc-backed allocator for RCCL
// buffer-registered (zero-copy) collectives.
// Reference surface: apex/contrib/csrc/nccl_allocator/NCCLAllocator.cpp
// (ncclMemAlloc/ncclMemFree wrapped as a torch pluggable allocator; Python
// wraps it in torch.cuda.MemPool). NVLS has no xGMI analogue, but RCCL's
// user-buffer registration path still benefits from ncclMemAlloc'd buffers.
#include <torch/csrc/cuda/CUDAPluggableAllocator.h>
#include <torch/extension.h>

#include <rccl/rccl.h>

namespace {

void* rccl_alloc_fn(size_t size, int device, void* stream) {
  void* ptr = nullptr;
  auto r = ncclMemAlloc(&ptr, size);
  TORCH_CHECK(r == ncclSuccess, "ncclMemAlloc failed: ", ncclGetErrorString(r));
  return ptr;
}

void rccl_free_fn(void* ptr, size_t size, int device, void* stream) {
  auto r = ncclMemFree(ptr);
  TORCH_CHECK(r == ncclSuccess, "ncclMemFree failed: ", ncclGetErrorString(r));
}

}  // namespace

std::shared_ptr<c10::cuda::CUDACachingAllocator::CUDAAllocator> get_rccl_allocator() {
  return torch::cuda::CUDAPluggableAllocator::createCustomAllocator(rccl_alloc_fn, rccl_free_fn);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("get_rccl_allocator", &get_rccl_allocator, "ncclMemAlloc-backed pluggable allocator");
}
