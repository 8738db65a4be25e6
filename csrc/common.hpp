// mpi4torch_amd — MI355X-native autodiff-transparent collectives for PyTorch.
//
// Shared declarations: reduction-op enum, dtype maps, error helpers, config.
//
// Design (vs the reference helmholtz-analytics/mpi4torch, csrc/extension.cpp):
// the reference speaks raw MPI from one C++ TU; we are MI355X-first:
//   * GPU tensors ride RCCL over xGMI on dedicated HIP streams
//     (csrc/transport.cpp), with hand-written CDNA4 pack/unpack kernels
//     (csrc/kernels.hip) replacing MPI derived datatypes
//     (reference csrc/extension.cpp:556-577).
//   * CPU tensors ride a c10d gloo backend (torchrun bootstrap, no mpirun).
// Parity targets are cited per-site as reference file:line.
#pragma once

#include <torch/types.h>
#include <c10/util/Exception.h>

#include <cstdint>
#include <string>

namespace m4a {

// Reduction op constants. Mirrors the 12 ops of the reference
// (csrc/extension.cpp:204-252, bound at :1424-1435). Values are our own
// stable ABI (the reference exported raw MPI_Op handles cast to int).
enum RedOp : int64_t {
  kMax = 0,
  kMin = 1,
  kSum = 2,
  kProd = 3,
  kLAnd = 4,
  kBAnd = 5,
  kLOr = 6,
  kBOr = 7,
  kLXor = 8,
  kBXor = 9,
  kMinLoc = 10,
  kMaxLoc = 11,
};

inline const char* red_op_name(int64_t op) {
  switch (op) {
    case kMax: return "MPI_MAX";
    case kMin: return "MPI_MIN";
    case kSum: return "MPI_SUM";
    case kProd: return "MPI_PROD";
    case kLAnd: return "MPI_LAND";
    case kBAnd: return "MPI_BAND";
    case kLOr: return "MPI_LOR";
    case kBOr: return "MPI_BOR";
    case kLXor: return "MPI_LXOR";
    case kBXor: return "MPI_BXOR";
    case kMinLoc: return "MPI_MINLOC";
    case kMaxLoc: return "MPI_MAXLOC";
    default: return "<invalid op>";
  }
}

// Global config toggles (see python layer mpi4torch_amd/utils/config.py).
struct Config {
  // Mirrors reference deactivate_cuda_aware_mpi_support()
  // (csrc/extension.cpp:54-59,1404-1414): when true, GPU tensors are staged
  // through host memory and communicated over the CPU (gloo) transport —
  // a debugging escape hatch, never the default on MI355X.
  bool force_host_staging = false;
  // hipEvent-based per-collective tracing (the reference has
  // none; we add it as a first-class aux subsystem).
  bool trace_enabled = false;
  // Collective-desync detector (MPI4TORCH_AMD_DEBUG=1): before every
  // collective, ranks exchange a hash of (op, shape, dtype, args) over the
  // host channel and raise on mismatch — turning would-be deadlocks (the
  // core SPMD hazard, reference doc/basic_usage.rst:184-322) into
  // immediate, attributed errors.
  bool debug_collectives = false;
  // MPI4TORCH_AMD_FORCE_FULL_PATH=1: disable world-size-1 fast paths so the
  // full pack -> exchange -> unpack machinery runs even alone (kernel
  // benchmarking / native-path verification on a single GPU).
  bool force_full_path = false;
  // MPI4TORCH_AMD_TIMEOUT_S: bound every blocking CPU-transport wait; a
  // desynchronized peer then raises instead of hanging forever. 0 = wait
  // indefinitely (default). (GPU-side progress is stream-ordered; use the
  // desync detector for pre-enqueue divergence.)
  int64_t op_timeout_ms = 0;
  // MPI4TORCH_AMD_FORCE_HIERARCHICAL=1 (testing): run the hierarchical
  // allreduce lowering (bitwise/fp8/pairloc) on the CPU transport too,
  // with torch-composite local reductions — lets the gloo SPMD suite
  // validate the multi-rank block-exchange geometry the GPU uses.
  bool force_hierarchical = false;
  // MPI4TORCH_AMD_PIPELINE_MB (float, default 256): chunk size for the
  // phased pack->wire pipelining of axis-marshaling collectives. Payloads
  // whose packed side exceeds one chunk are exchanged in up to 4 phases so
  // the CDNA4 pack/unpack kernels overlap the wire time of neighboring
  // phases. <=0 disables phasing. Default sized from measurement
  // (profiles/tune_pipeline_r2.log): each phase costs ~40 us of
  // event-chain latency and loses some copy occupancy, while the overlap
  // can hide ~2*payload/5TB/s of marshaling — net-positive only when the
  // wire is the slow side (multi-GPU) and payloads are >= ~1 GiB/rank.
  int64_t pipeline_chunk_bytes = 256ll << 20;
};

Config& config();
// Re-read the MPI4TORCH_AMD_* environment into the live config (tests and
// long-lived processes that toggle behavior between phases).
void reload_config_from_env();

} // namespace m4a
