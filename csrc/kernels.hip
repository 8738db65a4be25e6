// CDNA4 (gfx950) kernels for mpi4torch_amd.
//
// These kernels are the MI355X-native replacement for the CPU-side MPI
// derived-datatype marshaling the reference uses for axis-aware collectives
// (reference csrc/extension.cpp:556-577, 691-712, 839-861): strided slab
// pack/unpack between tensors and contiguous RCCL staging buffers, plus a
// local bitwise reduction for the MPI_BAND/BOR/BXOR ops RCCL lacks.
//
// Written directly for gfx950: 64-wide wavefronts, 256-thread workgroups,
// 16-byte (dwordx4) vectorized global accesses, grids sized >> 256
// workgroups to fill all 8 XCDs. Pure streaming copies have no inter-block
// reuse, so no XCD-aware blockIdx remap is needed (it only pays when
// neighboring blocks share operand panels — cdna_hip_programming.md T1).

#include <hip/hip_runtime.h>
#include "kernels.hpp"

namespace m4a {

namespace {

// Slab geometry precomputed in *chunks* (one chunk = Granule bytes).
struct SlabArgs {
  const char* src;
  char* dst;
  long long chunks_per_row;   // after_b / granule
  long long rows_per_b;       // count
  long long total_chunks;     // before * count * chunks_per_row
  long long src_pitch_b;
  long long dst_pitch_b;
  long long after_b;
};

template <int N>
struct SlabPack {
  SlabArgs s[N];
};

template <typename VecT, int NSLABS>
__global__ __launch_bounds__(256) void slab_copy_kernel(SlabPack<NSLABS> pack) {
  const SlabArgs& a = pack.s[blockIdx.z];
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < a.total_chunks; i += stride) {
    // i -> (b, c, k): row-major over [before][count][chunks_per_row]
    const long long row = i / a.chunks_per_row;       // = b*count + c
    const long long k = i - row * a.chunks_per_row;
    const long long b = row / a.rows_per_b;
    const long long c = row - b * a.rows_per_b;
    const long long off = c * a.after_b + k * (long long)sizeof(VecT);
    const VecT* sp =
        reinterpret_cast<const VecT*>(a.src + b * a.src_pitch_b + off);
    VecT* dp = reinterpret_cast<VecT*>(a.dst + b * a.dst_pitch_b + off);
    *dp = *sp;
  }
}

inline bool aligned_to(const SlabDesc& d, int64_t g) {
  return (d.after_b % g == 0) &&
         (reinterpret_cast<uintptr_t>(d.src) % g == 0) &&
         (reinterpret_cast<uintptr_t>(d.dst) % g == 0) &&
         (d.src_pitch_b % g == 0) && (d.dst_pitch_b % g == 0);
}

template <typename VecT>
void launch_bucket(const SlabDesc* descs, int n, hipStream_t stream) {
  while (n > 0) {
    const int take = n < kMaxSlabsPerLaunch ? n : kMaxSlabsPerLaunch;
    SlabPack<kMaxSlabsPerLaunch> pack{};
    long long max_chunks = 0;
    for (int i = 0; i < take; ++i) {
      const SlabDesc& d = descs[i];
      SlabArgs& a = pack.s[i];
      a.src = static_cast<const char*>(d.src);
      a.dst = static_cast<char*>(d.dst);
      a.chunks_per_row = d.after_b / (int64_t)sizeof(VecT);
      a.rows_per_b = d.count;
      a.total_chunks = d.before * d.count * a.chunks_per_row;
      a.src_pitch_b = d.src_pitch_b;
      a.dst_pitch_b = d.dst_pitch_b;
      a.after_b = d.after_b;
      if (a.total_chunks > max_chunks) max_chunks = a.total_chunks;
    }
    for (int i = take; i < kMaxSlabsPerLaunch; ++i) {
      pack.s[i] = pack.s[0];
      pack.s[i].total_chunks = 0;
    }
    if (max_chunks > 0) {
      // >=2048 workgroups when there is enough work: fills 256 CUs / 8 XCDs
      // with several waves per CU; grid-stride loop handles the tail.
      long long blocks = (max_chunks + 255) / 256;
      if (blocks > 4096) blocks = 4096;
      if (blocks < 1) blocks = 1;
      dim3 grid((unsigned)blocks, 1, (unsigned)take);
      hipLaunchKernelGGL((slab_copy_kernel<VecT, kMaxSlabsPerLaunch>), grid,
                         dim3(256), 0, stream, pack);
    }
    descs += take;
    n -= take;
  }
}

struct OpAnd {
  __device__ static uint4 apply(uint4 a, uint4 b) {
    return make_uint4(a.x & b.x, a.y & b.y, a.z & b.z, a.w & b.w);
  }
  __device__ static unsigned char apply(unsigned char a, unsigned char b) {
    return a & b;
  }
};
struct OpOr {
  __device__ static uint4 apply(uint4 a, uint4 b) {
    return make_uint4(a.x | b.x, a.y | b.y, a.z | b.z, a.w | b.w);
  }
  __device__ static unsigned char apply(unsigned char a, unsigned char b) {
    return a | b;
  }
};
struct OpXor {
  __device__ static uint4 apply(uint4 a, uint4 b) {
    return make_uint4(a.x ^ b.x, a.y ^ b.y, a.z ^ b.z, a.w ^ b.w);
  }
  __device__ static unsigned char apply(unsigned char a, unsigned char b) {
    return a ^ b;
  }
};

template <typename T, typename Op>
__global__ __launch_bounds__(256) void bitwise_reduce_kernel(
    const T* __restrict__ in, T* __restrict__ out, long long chunk_elems,
    long long nranks) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < chunk_elems; i += stride) {
    T acc = in[i];
    for (long long r = 1; r < nranks; ++r) {
      acc = Op::apply(acc, in[r * chunk_elems + i]);
    }
    out[i] = acc;
  }
}

template <typename T, typename Op>
void launch_bitred(const void* in, void* out, int64_t chunk_elems, int nranks,
                   hipStream_t stream) {
  long long blocks = (chunk_elems + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL((bitwise_reduce_kernel<T, Op>), dim3((unsigned)blocks),
                     dim3(256), 0, stream, static_cast<const T*>(in),
                     static_cast<T*>(out), (long long)chunk_elems,
                     (long long)nranks);
}

} // namespace

void launch_slab_copy(const SlabDesc* descs, int n, hipStream_t stream) {
  // Bucket consecutive slabs by the widest granule they admit so each launch
  // is uniform. In practice all slabs of one collective share alignment.
  int i = 0;
  while (i < n) {
    const int64_t g = aligned_to(descs[i], 16) ? 16
                      : aligned_to(descs[i], 4) ? 4
                                                : 1;
    int j = i + 1;
    while (j < n) {
      const int64_t gj = aligned_to(descs[j], 16) ? 16
                         : aligned_to(descs[j], 4) ? 4
                                                   : 1;
      if (gj != g) break;
      ++j;
    }
    if (g == 16) {
      launch_bucket<uint4>(descs + i, j - i, stream);
    } else if (g == 4) {
      launch_bucket<unsigned int>(descs + i, j - i, stream);
    } else {
      launch_bucket<unsigned char>(descs + i, j - i, stream);
    }
    i = j;
  }
}

void launch_bitwise_reduce(const void* in, void* out, int64_t chunk_bytes,
                           int nranks, int op, hipStream_t stream) {
  const bool vec16 = (chunk_bytes % 16 == 0) &&
                     (reinterpret_cast<uintptr_t>(in) % 16 == 0) &&
                     (reinterpret_cast<uintptr_t>(out) % 16 == 0);
  if (vec16) {
    const int64_t elems = chunk_bytes / 16;
    switch (op) {
      case 0: launch_bitred<uint4, OpAnd>(in, out, elems, nranks, stream); break;
      case 1: launch_bitred<uint4, OpOr>(in, out, elems, nranks, stream); break;
      default: launch_bitred<uint4, OpXor>(in, out, elems, nranks, stream); break;
    }
  } else {
    switch (op) {
      case 0:
        launch_bitred<unsigned char, OpAnd>(in, out, chunk_bytes, nranks, stream);
        break;
      case 1:
        launch_bitred<unsigned char, OpOr>(in, out, chunk_bytes, nranks, stream);
        break;
      default:
        launch_bitred<unsigned char, OpXor>(in, out, chunk_bytes, nranks, stream);
        break;
    }
  }
}

} // namespace m4a
