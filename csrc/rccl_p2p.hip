// apex_amd._rccl_p2p — raw RCCL communicator for latency-critical halo
// exchange (bypasses torch.distributed's c10d layer).
// Reference surface: apex/contrib/csrc/nccl_p2p/nccl_p2p_cuda.cu
// (get_unique_nccl_id / init_nccl_comm / left_right_halo_exchange via
// ncclGroupStart/Send/Recv/GroupEnd).
#include "common.h"

#include <rccl/rccl.h>

#include <vector>

#define RCCL_CHECK(expr)                                                       \
  do {                                                                         \
    ncclResult_t _r = (expr);                                                  \
    TORCH_CHECK(_r == ncclSuccess, "RCCL error: ", ncclGetErrorString(_r));    \
  } while (0)

namespace {

ncclDataType_t rccl_dtype(at::ScalarType t) {
  switch (t) {
    case at::ScalarType::Float: return ncclFloat32;
    case at::ScalarType::Half: return ncclFloat16;
    case at::ScalarType::BFloat16: return ncclBfloat16;
    case at::ScalarType::Int: return ncclInt32;
    default: TORCH_CHECK(false, "rccl_p2p: unsupported dtype");
  }
}

std::vector<ncclComm_t> g_comms;

}  // namespace

at::Tensor get_unique_nccl_id(int64_t n) {
  ncclUniqueId id;
  RCCL_CHECK(ncclGetUniqueId(&id));
  auto t = at::zeros({n, (long)sizeof(ncclUniqueId)}, at::TensorOptions().dtype(at::kByte));
  memcpy(t[0].data_ptr(), &id, sizeof(id));
  return t;
}

int64_t init_nccl_comm(at::Tensor unique_nccl_id, int64_t my_rank, int64_t num_ranks) {
  ncclUniqueId id;
  auto cpu = unique_nccl_id.to(at::kCPU).contiguous();
  memcpy(&id, cpu.data_ptr(), sizeof(id));
  ncclComm_t comm;
  RCCL_CHECK(ncclCommInitRank(&comm, (int)num_ranks, id, (int)my_rank));
  g_comms.push_back(comm);
  return (int64_t)g_comms.size() - 1;
}

// ring halo exchange: send my left edge to left neighbour, right edge to
// right neighbour; receive their edges (reference :83-114).
void left_right_halo_exchange_inplace(int64_t handle, int64_t left_rank, int64_t right_rank,
                                      at::Tensor left_output_halo, at::Tensor right_output_halo,
                                      at::Tensor left_input_halo, at::Tensor right_input_halo) {
  auto comm = g_comms.at((size_t)handle);
  auto stream = current_stream();
  const auto dt = rccl_dtype(left_output_halo.scalar_type());
  RCCL_CHECK(ncclGroupStart());
  if (left_rank >= 0) {
    RCCL_CHECK(ncclSend(left_output_halo.data_ptr(), left_output_halo.numel(), dt,
                        (int)left_rank, comm, stream));
    RCCL_CHECK(ncclRecv(left_input_halo.data_ptr(), left_input_halo.numel(), dt, (int)left_rank,
                        comm, stream));
  }
  if (right_rank >= 0) {
    RCCL_CHECK(ncclSend(right_output_halo.data_ptr(), right_output_halo.numel(), dt,
                        (int)right_rank, comm, stream));
    RCCL_CHECK(ncclRecv(right_input_halo.data_ptr(), right_input_halo.numel(), dt,
                        (int)right_rank, comm, stream));
  }
  RCCL_CHECK(ncclGroupEnd());
}

std::vector<at::Tensor> left_right_halo_exchange(int64_t handle, int64_t left_rank,
                                                 int64_t right_rank, at::Tensor left_output_halo,
                                                 at::Tensor right_output_halo) {
  auto left_input_halo = at::empty_like(right_output_halo);
  auto right_input_halo = at::empty_like(left_output_halo);
  left_right_halo_exchange_inplace(handle, left_rank, right_rank, left_output_halo,
                                   right_output_halo, left_input_halo, right_input_halo);
  return {left_input_halo, right_input_halo};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("get_unique_nccl_id", &get_unique_nccl_id);
  m.def("init_nccl_comm", &init_nccl_comm);
  m.def("left_right_halo_exchange", &left_right_halo_exchange);
  m.def("left_right_halo_exchange_inplace", &left_right_halo_exchange_inplace);
}
