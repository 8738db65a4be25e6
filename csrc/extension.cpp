// Registration layer: TorchScript custom class + free ops + pybind module.
// Parity target: reference csrc/extension.cpp:1270-1436 (L3/L1 of
// SURVEY.md §1), with MPI lifecycle replaced by c10d/RCCL bootstrap that
// happens lazily inside the Communicator (no import-time rendezvous).

#include "ops.hpp"
#include "transport.hpp"

#include <torch/script.h>
#include <torch/extension.h>

#include <rccl/rccl.h>

namespace m4a {
namespace {

// World group name, set by the Python layer after torch.distributed init.
// Empty = no distributed context (local world of one).
std::string& world_group_name() {
  static std::string name;
  return name;
}

c10::intrusive_ptr<Communicator> comm_world() {
  return c10::make_intrusive<Communicator>(world_group_name());
}

c10::intrusive_ptr<Communicator> comm_from_group(std::string group_name) {
  return c10::make_intrusive<Communicator>(std::move(group_name));
}

at::Tensor join_dummies_op(const at::Tensor& loopthrough,
                           const std::vector<at::Tensor>& dummies) {
  return join_dummies(loopthrough, dummies);
}

// TorchScript custom class (reference :1270-1298). The pickle round-trip
// serializes the group name — and deserializes it with the condition the
// right way around (the reference's deserializer rejected its own valid
// payload, ref :1290-1296).
static auto communicator_class =
    torch::class_<Communicator>("mpi4torch_amd", "Communicator")
        .def(torch::init<std::string>())
        .def("GetRank", &Communicator::GetRank)
        .def("GetSize", &Communicator::GetSize)
        .def("Allreduce", &Communicator::Allreduce)
        .def("Bcast_", &Communicator::Bcast_)
        .def("Reduce_", &Communicator::Reduce_)
        .def("Gather", &Communicator::Gather)
        .def("Allgather", &Communicator::Allgather)
        .def("Reducescatter", &Communicator::Reducescatter)
        .def("Scatter", &Communicator::Scatter)
        .def("Alltoall", &Communicator::Alltoall)
        .def("Alltoallv", &Communicator::Alltoallv)
        .def("AlltoallPairwise", &Communicator::AlltoallPairwise)
        .def("Iallreduce", &Communicator::Iallreduce)
        .def("Ireducescatter", &Communicator::Ireducescatter)
        .def("Iallgather", &Communicator::Iallgather)
        .def("Isend", &Communicator::Isend)
        .def("Irecv", &Communicator::Irecv)
        .def("Wait", &Communicator::Wait)
        .def("GetGroupName",
             [](const c10::intrusive_ptr<Communicator>& self) {
               return self->group_name();
             })
        .def_pickle(
            [](const c10::intrusive_ptr<Communicator>& self) -> std::string {
              return self->group_name();
            },
            [](std::string state) -> c10::intrusive_ptr<Communicator> {
              return c10::make_intrusive<Communicator>(std::move(state));
            });

TORCH_LIBRARY(mpi4torch_amd, m) {
  m.def("comm_world", comm_world);
  m.def("comm_from_group", comm_from_group);
  m.def("JoinDummies", join_dummies_op);
}

} // namespace
} // namespace m4a

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "mpi4torch_amd native core: RCCL-over-xGMI autograd collectives";

  m.def("_set_world_group_name",
        [](const std::string& name) { m4a::world_group_name() = name; });
  m.def("_get_world_group_name",
        []() { return m4a::world_group_name(); });

  // Debug escape hatch mirroring deactivate_cuda_aware_mpi_support
  // (reference :1404-1414): route GPU tensors through host staging + gloo.
  m.def("force_host_staging",
        [](bool enabled) { m4a::config().force_host_staging = enabled; });
  m.def("host_staging_forced",
        []() { return m4a::config().force_host_staging; });

  // Disable world-size-1 fast paths (kernel benchmarking / native-path
  // verification on a single GPU); also via MPI4TORCH_AMD_FORCE_FULL_PATH=1.
  m.def("force_full_path",
        [](bool enabled) { m4a::config().force_full_path = enabled; });

  m.def("_pack_roundtrip", &m4a::debug_pack_roundtrip);
  m.def("_bitwise_reduce", &m4a::debug_bitwise_reduce);
  m.def("_fp8_reduce", &m4a::debug_fp8_reduce);
  m.def("_pairloc_reduce", &m4a::debug_pairloc_reduce);
  m.def("reload_config", &m4a::reload_config_from_env);

  m.def("_rccl_version", []() {
    int v = 0;
    ncclGetVersion(&v);
    return v;
  });

  // Reduction-op constants as plain ints (TorchScript rejects py::enum_;
  // same workaround as reference :1417-1435).
  m.attr("MPI_MAX") = (int64_t)m4a::kMax;
  m.attr("MPI_MIN") = (int64_t)m4a::kMin;
  m.attr("MPI_SUM") = (int64_t)m4a::kSum;
  m.attr("MPI_PROD") = (int64_t)m4a::kProd;
  m.attr("MPI_LAND") = (int64_t)m4a::kLAnd;
  m.attr("MPI_BAND") = (int64_t)m4a::kBAnd;
  m.attr("MPI_LOR") = (int64_t)m4a::kLOr;
  m.attr("MPI_BOR") = (int64_t)m4a::kBOr;
  m.attr("MPI_LXOR") = (int64_t)m4a::kLXor;
  m.attr("MPI_BXOR") = (int64_t)m4a::kBXor;
  m.attr("MPI_MINLOC") = (int64_t)m4a::kMinLoc;
  m.attr("MPI_MAXLOC") = (int64_t)m4a::kMaxLoc;
}
