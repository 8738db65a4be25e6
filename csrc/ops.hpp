// Communicator: the TorchScript custom class at the center of
// mpi4torch_amd. Parity target: MPI_Comm_Wrapper in the reference
// (helmholtz-analytics/mpi4torch csrc/extension.cpp:140-187), with the
// communicator identified by a c10d group name instead of a raw MPI_Comm
// (sub-communicators arrive through torch.distributed groups rather than
// mpi4py Fortran handles, reference :168-171).
#pragma once

#include "common.hpp"

#include <ATen/ATen.h>
#include <torch/custom_class.h>

#include <memory>
#include <mutex>
#include <string>
#include <vector>

namespace m4a {

struct Transport;

struct Communicator : torch::CustomClassHolder {
  explicit Communicator(std::string group_name);

  int64_t GetRank();
  int64_t GetSize();

  // Collectives (autograd-transparent; adjoints per SURVEY.md §3.3).
  at::Tensor Allreduce(const at::Tensor& input, int64_t op);
  at::Tensor Bcast_(const at::Tensor& input, int64_t root);
  at::Tensor Reduce_(const at::Tensor& input, int64_t op, int64_t root);
  at::Tensor Gather(const at::Tensor& input, int64_t gatheraxis, int64_t root);
  at::Tensor Allgather(const at::Tensor& input, int64_t gatheraxis);
  // Adjoint pair of Allgather, exposed as a public collective (MI355X
  // extension; not in the reference API): elementwise-SUM across ranks,
  // then this rank keeps `numelem` slices of `axis` (per-rank counts may
  // differ). Backward: Allgather of the gradient. Enables ZeRO-style
  // gradient/optimizer sharding (parallel/zero.py).
  at::Tensor Reducescatter(const at::Tensor& input, int64_t axis,
                           int64_t numelem);
  at::Tensor Scatter(const at::Tensor& input, int64_t scatteraxis,
                     int64_t numelem, int64_t root);
  at::Tensor Alltoall(const at::Tensor& input, int64_t gatheraxis,
                      int64_t scatteraxis, int64_t numelem);
  // Explicit-counts variant (MI355X extension; not in the reference):
  // skips the two host count exchanges when the caller already knows every
  // rank's counts (expert-parallel routing). target_counts[j] = what rank j
  // keeps along scatteraxis; source_sizes[j] = rank j's current
  // gather/partition-axis size.
  at::Tensor Alltoallv(const at::Tensor& input, int64_t gatheraxis,
                       int64_t scatteraxis,
                       std::vector<int64_t> target_counts,
                       std::vector<int64_t> source_sizes);
  at::Tensor AlltoallvImpl(const at::Tensor& input, int64_t gatheraxis,
                           int64_t scatteraxis, int64_t numelem,
                           const std::vector<int64_t>& target_counts,
                           const std::vector<int64_t>& source_sizes);
  // Pairwise-count alltoall (MI355X extension): rank r sends
  // send_counts[j] slices of `axis` to rank j (arbitrary P x P count
  // matrix — the expert-parallel token dispatch, inexpressible as the
  // reference's interval repartition). recv_counts, if empty, are
  // exchanged over the host channel. Backward sends every received slice
  // back (the exact adjoint).
  at::Tensor AlltoallPairwise(const at::Tensor& input, int64_t axis,
                              std::vector<int64_t> send_counts,
                              std::vector<int64_t> recv_counts);

  // Non-blocking allreduce (MI355X-first overlap primitive, not in the
  // reference API): returns a wait handle; no autograd through it — use
  // Allreduce for differentiable paths. Gradient bucketing (parallel/ddp)
  // is its main consumer.
  std::vector<at::Tensor> Iallreduce(const at::Tensor& input, int64_t op);
  // Non-blocking equal-count reduce-scatter (no autograd): this rank's
  // flat block of the elementwise sum. input.numel() must be
  // world_size * block; returns a wait handle for the [block] result.
  std::vector<at::Tensor> Ireducescatter(const at::Tensor& input, int64_t op);
  // Non-blocking equal-count flat allgather (no autograd): Wait yields the
  // rank-major concatenation [size * numel]. FSDP prefetch primitive.
  std::vector<at::Tensor> Iallgather(const at::Tensor& input);

  // Non-blocking p2p. Handle contract identical to the reference
  // (csrc/extension.cpp:1094-1107): [meta tensor, comm buffer, input].
  std::vector<at::Tensor> Isend(const at::Tensor& input, int64_t dest,
                                int64_t tag);
  std::vector<at::Tensor> Irecv(const at::Tensor& input, int64_t source,
                                int64_t tag);
  at::Tensor Wait(const std::vector<at::Tensor>& handle);

  // Channel-aware p2p used by the backward pass: adjoint transfers ride a
  // dedicated RCCL communicator/stream instead of the reference's tag+10
  // offset (csrc/extension.cpp:1161).
  std::vector<at::Tensor> IsendImpl(const at::Tensor& input, int64_t dest,
                                    int64_t tag, bool backward_channel);
  std::vector<at::Tensor> IrecvImpl(const at::Tensor& input, int64_t source,
                                    int64_t tag, bool backward_channel);

  const std::string& group_name() const { return group_name_; }

  // internal
  Transport& tr_for(const at::Tensor& t);
  Transport& cpu_tr();
  Transport& gpu_tr(int device);

 private:
  std::string group_name_;
  std::mutex mu_;
  std::shared_ptr<Transport> cpu_tr_;
  std::shared_ptr<Transport> gpu_tr_;
  int gpu_device_ = -1;
};

// JoinDummies free op (reference csrc/extension.cpp:989-1046).
at::Tensor join_dummies(const at::Tensor& loopthrough,
                        const std::vector<at::Tensor>& dummies);

// Debug/test entry points (tests/test_gpu.py): exercise the CDNA4 slab and
// bitwise-reduce kernels directly, without a multi-rank world.
at::Tensor debug_pack_roundtrip(const at::Tensor& input, int64_t axis,
                                std::vector<int64_t> counts);
at::Tensor debug_bitwise_reduce(const at::Tensor& stacked, int64_t op);
at::Tensor debug_fp8_reduce(const at::Tensor& stacked, int64_t op);
// stacked: [nranks, ..., 2] (value, location) pairs; op: 0=minloc 1=maxloc
at::Tensor debug_pairloc_reduce(const at::Tensor& stacked, int64_t op);

} // namespace m4a
