// Transport abstraction for mpi4torch_amd.
//
// The reference (helmholtz-analytics/mpi4torch) has exactly one transport:
// raw MPI with a CUDA-aware/host-staging device policy
// (csrc/extension.cpp:61-104). The MI355X-native design has
// three, selected per tensor device:
//   * RcclTransport  — RCCL over xGMI; one process per GPU; three RCCL
//     communicators per world (collectives / forward p2p / backward p2p),
//     each driven from its own non-blocking HIP stream and joined to the
//     caller's compute stream with hipEvents. The separate p2p channels
//     replace MPI tag matching (reference tag+10 scheme,
//     csrc/extension.cpp:1161): RCCL has no tags and requires a globally
//     consistent enqueue order per communicator, so independent traffic
//     classes get independent communicators.
//   * C10dTransport  — a c10d (gloo) backend for CPU tensors and for the
//     host-side metadata exchanges of the axis collectives. Replaces the
//     reference's MPI_Gather/Bcast of ints (csrc/extension.cpp:540,675,789).
//   * LocalTransport — world_size==1 fast paths with no runtime at all.
#pragma once

#include "common.hpp"
#include "kernels.hpp"

#include <ATen/ATen.h>

#include <memory>
#include <vector>

namespace c10d {
class Backend;
}

namespace m4a {

enum class Channel : int { Coll = 0, P2P = 1, P2PBwd = 2 };

struct Transport : std::enable_shared_from_this<Transport> {
  virtual ~Transport() = default;
  virtual int rank() const = 0;
  virtual int size() const = 0;
  virtual bool is_gpu() const = 0;

  // Launch any deferred point-to-point operations (see RcclTransport: p2p
  // enqueues are batched and issued as ONE ncclGroupStart/End at the first
  // Wait, c10d batch_isend_irecv style, so matched send/recv pairs can
  // rendezvous — serially-issued ncclSend/ncclRecv on one stream deadlock
  // in a ring once payloads exceed RCCL's internal buffering). No-op on
  // transports whose p2p is eagerly posted (gloo has a host-side matching
  // engine, like MPI's).
  virtual void flush_p2p() {}

  // All tensors must be contiguous and on this transport's device class.
  // GPU: stream-ordered with respect to the caller's current stream
  // (internally bracketed onto a dedicated side stream). CPU: blocking.
  virtual void allreduce(const at::Tensor& in, at::Tensor& out, RedOp op) = 0;
  virtual void broadcast(at::Tensor& t, int root) = 0;  // in place
  // In-place reduce; result valid only on root (caller zero-fills non-root,
  // matching reference csrc/extension.cpp:443-447).
  virtual void reduce(at::Tensor& t, RedOp op, int root) = 0;
  // out.numel() == size * in.numel(); rank-major blocks.
  virtual void allgather_equal(const at::Tensor& in, at::Tensor& out) = 0;
  // in.numel() == size * out.numel(); rank-major blocks, elementwise-reduced.
  virtual void reduce_scatter_equal(const at::Tensor& in, at::Tensor& out,
                                    RedOp op) = 0;
  // Grouped variable-count exchange: send sendbufs[i] to speers[i] and
  // receive recvbufs[j] from rpeers[j], all concurrently. Self pairs are
  // matched in order and turned into device/host copies. This single
  // primitive carries Gather/Scatter/Allgather/Alltoall marshaled blocks
  // (the role MPI_Gatherv/Scatterv/Allgatherv/derived datatypes play in the
  // reference, csrc/extension.cpp:584,719,869).
  virtual void exchange(const std::vector<at::Tensor>& sendbufs,
                        const std::vector<int>& speers,
                        std::vector<at::Tensor>& recvbufs,
                        const std::vector<int>& rpeers) = 0;
  // Non-blocking exchange: same grouped launch, but the caller's stream is
  // only joined when the returned request is waited. One phase of the
  // pipelined (chunked) axis collectives — phase k+1's pack and phase k's
  // unpack overlap phase k's wire time.
  virtual uint64_t iexchange(const std::vector<at::Tensor>& sendbufs,
                             const std::vector<int>& speers,
                             std::vector<at::Tensor>& recvbufs,
                             const std::vector<int>& rpeers) = 0;
  // Non-blocking p2p. Returns a request id resolvable via wait_request().
  // Matching contract (identical on ALL transports): FIFO per
  // (peer, channel) — the n-th send posted to a peer on a channel matches
  // the n-th recv posted from it on that channel. `tag` does NOT
  // disambiguate matching (RCCL has no tags); it is metadata, validated
  // against the peer's tag under MPI4TORCH_AMD_DEBUG=1 so crossed
  // transfers raise instead of silently swapping payloads.
  virtual uint64_t isend(const at::Tensor& buf, int peer, int tag,
                         Channel ch) = 0;
  virtual uint64_t irecv(at::Tensor& buf, int peer, int tag, Channel ch) = 0;
  // Non-blocking allreduce (no autograd; overlap primitive for gradient
  // bucketing — an MI355X-first extension beyond the reference's API).
  virtual uint64_t iallreduce(const at::Tensor& in, at::Tensor& out,
                              RedOp op) = 0;
  // Non-blocking equal-count reduce-scatter (no autograd): ZeRO-2 bucket
  // primitive. in.numel() == size * out.numel(), rank-major blocks.
  virtual uint64_t ireduce_scatter(const at::Tensor& in, at::Tensor& out,
                                   RedOp op) = 0;
  // Non-blocking equal-count allgather (no autograd): the FSDP parameter
  // prefetch primitive. out.numel() == size * in.numel(), rank-major.
  virtual uint64_t iallgather(const at::Tensor& in, at::Tensor& out) = 0;
  // Whether fp8 reductions run natively (RCCL probes at runtime; the cast
  // fallback is used otherwise). Non-GPU transports: false.
  virtual bool fp8_reduce_supported(at::ScalarType) { return false; }
};

// Completes a request: GPU → inserts a wait into the caller's current
// stream (fully async, no host sync); CPU → blocks.
void wait_request(uint64_t req);

// Factories. make_c10d() wraps an already-registered c10d group (by name).
std::shared_ptr<Transport> make_local_transport();
std::shared_ptr<Transport> make_c10d_transport(const std::string& group_name);
// Bootstraps three RCCL communicators for `device` using the given c10d
// (gloo) backend for the ncclUniqueId exchange.
std::shared_ptr<Transport> make_rccl_transport(const std::string& group_name,
                                               int device);

// Host-side metadata collectives over the gloo backend (tiny int64 vectors;
// replaces reference's MPI_Gather/MPI_Bcast of ints for count exchanges).
std::vector<int64_t> host_allgather_int64(const std::string& group_name,
                                          int64_t value);
// Allgather a fixed-length int64 vector: returns rank-major concatenation
// (P * len entries). Carries the per-pair count matrices of the pairwise
// alltoall.
std::vector<int64_t> host_allgather_int64_vec(const std::string& group_name,
                                              const std::vector<int64_t>& v);
std::vector<int64_t> host_broadcast_int64(const std::string& group_name,
                                          const std::vector<int64_t>& values,
                                          int root, int64_t fixed_len);

} // namespace m4a
