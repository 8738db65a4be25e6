// Host-side declarations for the CDNA4 (gfx950) HIP kernels in kernels.hip.
// This header is torch-free so the device TU stays minimal.
#pragma once

#include <hip/hip_runtime_api.h>
#include <cstdint>

namespace m4a {

// Describes one slab of a batched strided copy: the MI355X-native
// replacement for the reference's MPI derived-datatype marshaling
// (MPI_Type_vector + MPI_Type_create_resized, reference
// csrc/extension.cpp:556-577, 691-712, 839-861).
//
// Copies, for b in [0,before), c in [0,count):
//   dst[b*dst_pitch_b + c*after_b + 0..after_b) =
//   src[b*src_pitch_b + c*after_b + 0..after_b)
// All quantities in BYTES except `before`/`count` (row counts). Note the
// c-stride is after_b on BOTH sides by definition, so (count, after)
// collapse into one contiguous row of count*after_b bytes per b — the
// kernel exploits this and vectorizes rows with phase-aligned 16/8/4/2-byte
// granules plus bytewise head/tail.
struct SlabDesc {
  const void* src;
  void* dst;
  int64_t before;
  int64_t count;
  int64_t after_b;
  int64_t src_pitch_b;  // byte stride between consecutive `b` on src
  int64_t dst_pitch_b;  // byte stride between consecutive `b` on dst
};

// Max slabs handled by a single kernel launch (descriptor array is passed
// by value in kernel args). Larger batches loop over launches.
constexpr int kMaxSlabsPerLaunch = 8;

// Batched strided copy of `n` slabs on `stream`. Picks the widest access
// (16B/4B/1B) every slab admits; one launch per <=8 slabs.
void launch_slab_copy(const SlabDesc* descs, int n, hipStream_t stream);

// Elementwise bitwise reduction across `nranks` contiguous chunks:
//   out[0..chunk_bytes) = op_{r<nranks} in[r*chunk_bytes ..]
// op: 0=AND 1=OR 2=XOR. Bitwise ops are byte-local, so this is
// dtype-independent (serves MPI_BAND/BOR/BXOR for every integer dtype;
// reference op table csrc/extension.cpp:204-252).
void launch_bitwise_reduce(const void* in, void* out, int64_t chunk_bytes,
                           int nranks, int op, hipStream_t stream);

// MINLOC/MAXLOC local arg-reduction across `nranks` contiguous chunks of
// (value, location) pairs (MPI pair-type semantics, reference op table
// csrc/extension.cpp:204-252): in is [nranks][n_pairs][2] (last axis =
// value, location in the same dtype), out is [n_pairs][2].
//   op 0 = MINLOC: smallest value, ties -> smallest location
//   op 1 = MAXLOC: largest value, ties -> smallest location
// dtype codes: 0=f32 1=f64 2=f16 3=bf16 4=i8 5=u8 6=i16 7=i32 8=i64
void launch_pairloc_reduce(const void* in, void* out, int64_t n_pairs,
                           int nranks, int op, int dtype, hipStream_t stream);

// Fused fp8 local reduction with fp32 accumulation: elementwise
//   out[i] = op_{r<nranks} fp32(in[r*n + i])  quantized to fp8 ONCE.
// This is the MI355X fp8 allreduce tail (allgather + this kernel): one
// quantization instead of one per ring hop, fp32 accumulation throughout.
// op: 0=sum 1=prod 2=min 3=max; e5m2 selects the encoding.
void launch_fp8_reduce(const void* in, void* out, int64_t n_elems,
                       int nranks, int op, bool e5m2, hipStream_t stream);

} // namespace m4a
