// CDNA4 (gfx950) kernels for mpi4torch_amd.
//
// These kernels are the MI355X-native replacement for the CPU-side MPI
// derived-datatype marshaling the reference uses for axis-aware collectives
// (reference csrc/extension.cpp:556-577, 691-712, 839-861) and for the
// reduction ops RCCL lacks. Four families: strided slab pack/unpack
// between tensors and contiguous RCCL staging buffers; bitwise reduction
// (MPI_BAND/BOR/BXOR); fused fp8 reduction with fp32 accumulation and a
// single quantization; and the MINLOC/MAXLOC pair arg-reduce. The latter
// three are the local tails of the hierarchical allreduce lowering
// (csrc/ops.cpp hierarchical_allreduce).
//
// Written directly for gfx950: 64-wide wavefronts, 256-thread workgroups,
// 16-byte (dwordx4) vectorized global accesses, grids sized >> 256
// workgroups to fill all 8 XCDs. Pure streaming copies have no inter-block
// reuse, so no XCD-aware blockIdx remap is needed (it only pays when
// neighboring blocks share operand panels — cdna_hip_programming.md T1).
//
// Slab geometry: a slab is [before] rows of [row_b] contiguous bytes each
// (the (count, after) dims of an axis slice collapse into one row — both
// sides advance by after_b per axis element, so consecutive axis elements
// are contiguous on BOTH sides). Rows are vectorized at the widest granule
// whose alignment PHASE matches between src and dst ((src - dst) % G == 0):
// each row gets a bytewise head up to dst 16/4/2-alignment, a granule body,
// and a bytewise tail, so odd element counts and odd displacements still
// stream at near-HBM rate instead of falling to a byte loop.

#include <hip/hip_runtime.h>
#include <hip/hip_fp8.h>
#include "kernels.hpp"

namespace m4a {

namespace {

struct SlabArgs {
  const char* src;
  char* dst;
  long long row_b;          // bytes per row (collapsed count*after)
  long long chunks_per_row; // body granules + 1 (head/tail chunk)
  long long total_chunks;   // before * chunks_per_row
  long long src_pitch_b;
  long long dst_pitch_b;
};

template <int N>
struct SlabPack {
  SlabArgs s[N];
};

// The nontemporal builtins want native scalar/vector types; HIP's uint4 is
// a class. Map each copy granule to an equivalent native type.
typedef unsigned int v4u __attribute__((ext_vector_type(4)));
template <typename T>
struct NtVec {
  using type = T;
};
template <>
struct NtVec<uint4> {
  using type = v4u;
};

// One chunk index c in [0, chunks_per_row): c < n_body copies granule c of
// the phase-aligned body; c == n_body copies the head and tail bytes.
template <typename VecT, int NSLABS>
__global__ __launch_bounds__(256) void slab_copy_kernel(SlabPack<NSLABS> pack) {
  constexpr long long G = (long long)sizeof(VecT);
  const SlabArgs& a = pack.s[blockIdx.z];
  const long long stride = (long long)gridDim.x * blockDim.x;
  // NOTE: a 4x partial unroll of this loop measured null (±1%) — the
  // kernel is HBM-bound with ample wave-level parallelism, not ILP-bound.
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < a.total_chunks; i += stride) {
    const long long b = i / a.chunks_per_row;
    const long long c = i - b * a.chunks_per_row;
    const char* srow = a.src + b * a.src_pitch_b;
    char* drow = a.dst + b * a.dst_pitch_b;
    // head: bytes until drow reaches G alignment (phase equality makes srow
    // aligned at the same point); clamped for rows shorter than the head
    long long head =
        (G - ((long long)(uintptr_t)drow & (G - 1))) & (G - 1);
    if (head > a.row_b) head = a.row_b;
    const long long body = (a.row_b - head) / G;
    if (c < body) {
      using NT = typename NtVec<VecT>::type;
      const NT* sp = reinterpret_cast<const NT*>(srow + head) + c;
      NT* dp = reinterpret_cast<NT*>(drow + head) + c;
      // nontemporal: each byte is touched exactly once (pure streaming),
      // so bypass the L1/L2 allocation (cdna_hip_programming.md nt-weights:
      // right for once-read streams)
      __builtin_nontemporal_store(__builtin_nontemporal_load(sp), dp);
    } else {
      // head + tail bytes, done by the one extra chunk per row
      for (long long k = 0; k < head; ++k) drow[k] = srow[k];
      for (long long k = head + body * G; k < a.row_b; ++k) drow[k] = srow[k];
    }
  }
}

template <typename VecT>
void launch_bucket(const SlabDesc* descs, int n, hipStream_t stream) {
  constexpr long long G = (long long)sizeof(VecT);
  while (n > 0) {
    const int take = n < kMaxSlabsPerLaunch ? n : kMaxSlabsPerLaunch;
    SlabPack<kMaxSlabsPerLaunch> pack{};
    long long max_chunks = 0;
    for (int i = 0; i < take; ++i) {
      const SlabDesc& d = descs[i];
      SlabArgs& a = pack.s[i];
      a.src = static_cast<const char*>(d.src);
      a.dst = static_cast<char*>(d.dst);
      a.row_b = d.count * d.after_b;  // collapsed row
      a.src_pitch_b = d.src_pitch_b;
      a.dst_pitch_b = d.dst_pitch_b;
      // worst-case head is G-1 bytes; one extra chunk per row covers
      // head+tail. Rows shorter than G go entirely through that chunk.
      a.chunks_per_row = (a.row_b >= G ? a.row_b / G : 0) + 1;
      a.total_chunks = d.before * a.chunks_per_row;
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

// Misaligned-phase path: when src and dst rows disagree mod 4, narrow
// granules (<=2B) are issue-bound (~2.2 TB/s measured). Instead: chunks are
// 16B-aligned on the DST side; the src bytes come from TWO aligned uint4
// loads recombined by a byte window (each src 16B line is read twice, by
// neighboring chunks) — ~1.5x traffic instead of 8x instruction count.

template <int Q>
__device__ inline uint4 byte_window_q(const unsigned int d[8], int rbits) {
  // bytes [4*Q + rbits/8, +16) of the 32-byte dword array d
  if (rbits == 0) return make_uint4(d[Q], d[Q + 1], d[Q + 2], d[Q + 3]);
  auto take = [&](int j) {
    unsigned long long w =
        ((unsigned long long)d[Q + j + 1] << 32) | d[Q + j];
    return (unsigned int)(w >> rbits);
  };
  return make_uint4(take(0), take(1), take(2), take(3));
}

__device__ inline uint4 byte_window(const uint4& lo, const uint4& hi, int s) {
  // bytes [s, s+16) of concat(lo, hi), 0 <= s < 16. The dword start is
  // selected by an unrolled switch so the array stays in registers
  // (runtime-indexed register arrays spill to scratch —
  // cdna_hip_programming.md §5.4 rule 20).
  const unsigned int d[8] = {lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
  const int q = s >> 2;
  const int rbits = (s & 3) * 8;
  switch (q) {
    case 0: return byte_window_q<0>(d, rbits);
    case 1: return byte_window_q<1>(d, rbits);
    case 2: return byte_window_q<2>(d, rbits);
    default: return byte_window_q<3>(d, rbits);
  }
}

template <int NSLABS>
__global__ __launch_bounds__(256) void slab_copy_shift_kernel(
    SlabPack<NSLABS> pack) {
  const SlabArgs& a = pack.s[blockIdx.z];
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < a.total_chunks; i += stride) {
    const long long b = i / a.chunks_per_row;
    const long long c = i - b * a.chunks_per_row;
    const char* srow = a.src + b * a.src_pitch_b;
    char* drow = a.dst + b * a.dst_pitch_b;
    long long head = (16 - ((long long)(uintptr_t)drow & 15)) & 15;
    if (head > a.row_b) head = a.row_b;
    const long long body = (a.row_b - head) / 16;
    const char* sbase = srow + head;
    const int s = (int)((uintptr_t)sbase & 15);
    // Nonzero shift: the last vector chunk would read up to 15 bytes past
    // the row, and chunk 0's aligned base (sbase - s) sits up to 15 bytes
    // BEFORE the row when s > head — both go to the bytewise worker.
    const long long bodyv = (s == 0) ? body : (body > 0 ? body - 1 : 0);
    const long long vs = (s == 0 || s <= head) ? 0 : 1;  // first vector chunk
    if (c >= vs && c < bodyv) {
      const char* sal = sbase - s + 16 * c;  // 16-aligned, in-bounds
      if (s == 0) {
        __builtin_nontemporal_store(
            __builtin_nontemporal_load(reinterpret_cast<const v4u*>(sal)),
            reinterpret_cast<v4u*>(drow + head + 16 * c));
      } else {
        // shared 16B lines between neighboring chunks: keep these cached
        const uint4 lo = *reinterpret_cast<const uint4*>(sal);
        const uint4 hi = *reinterpret_cast<const uint4*>(sal + 16);
        const uint4 w = byte_window(lo, hi, s);
        __builtin_nontemporal_store(
            *reinterpret_cast<const v4u*>(&w),
            reinterpret_cast<v4u*>(drow + head + 16 * c));
      }
    } else if (c == bodyv) {
      // bytewise worker: head (+ the excluded chunk 0, if any) and tail
      if (bodyv <= vs) {
        for (long long k = 0; k < a.row_b; ++k) drow[k] = srow[k];
      } else {
        const long long lo_end = head + vs * 16;
        for (long long k = 0; k < lo_end; ++k) drow[k] = srow[k];
        for (long long k = head + bodyv * 16; k < a.row_b; ++k) {
          drow[k] = srow[k];
        }
      }
    }
  }
}

void launch_shift_bucket(const SlabDesc* descs, int n, hipStream_t stream) {
  while (n > 0) {
    const int take = n < kMaxSlabsPerLaunch ? n : kMaxSlabsPerLaunch;
    SlabPack<kMaxSlabsPerLaunch> pack{};
    long long max_chunks = 0;
    for (int i = 0; i < take; ++i) {
      const SlabDesc& d = descs[i];
      SlabArgs& a = pack.s[i];
      a.src = static_cast<const char*>(d.src);
      a.dst = static_cast<char*>(d.dst);
      a.row_b = d.count * d.after_b;
      a.src_pitch_b = d.src_pitch_b;
      a.dst_pitch_b = d.dst_pitch_b;
      a.chunks_per_row = (a.row_b >= 16 ? a.row_b / 16 : 0) + 1;
      a.total_chunks = d.before * a.chunks_per_row;
      if (a.total_chunks > max_chunks) max_chunks = a.total_chunks;
    }
    for (int i = take; i < kMaxSlabsPerLaunch; ++i) {
      pack.s[i] = pack.s[0];
      pack.s[i].total_chunks = 0;
    }
    if (max_chunks > 0) {
      long long blocks = (max_chunks + 255) / 256;
      if (blocks > 4096) blocks = 4096;
      if (blocks < 1) blocks = 1;
      dim3 grid((unsigned)blocks, 1, (unsigned)take);
      hipLaunchKernelGGL((slab_copy_shift_kernel<kMaxSlabsPerLaunch>), grid,
                         dim3(256), 0, stream, pack);
    }
    descs += take;
    n -= take;
  }
}

// Widest granule whose alignment phase matches on both sides for every row:
// needs (src - dst) % G == 0 and (pitch difference) % G == 0.
inline int slab_granule(const SlabDesc& d) {
  const long long delta =
      (long long)((uintptr_t)d.src - (uintptr_t)d.dst);
  const long long pdelta = d.src_pitch_b - d.dst_pitch_b;
  for (int g = 16; g > 1; g >>= 1) {
    if ((delta & (g - 1)) == 0 && (pdelta & (g - 1)) == 0) return g;
  }
  return 1;
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
  using NT = typename NtVec<T>::type;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < chunk_elems; i += stride) {
    // every element is touched once: nontemporal streaming
    NT av = __builtin_nontemporal_load(reinterpret_cast<const NT*>(in) + i);
    T acc = *reinterpret_cast<const T*>(&av);
    for (long long r = 1; r < nranks; ++r) {
      NT bv = __builtin_nontemporal_load(reinterpret_cast<const NT*>(in) +
                                         r * chunk_elems + i);
      acc = Op::apply(acc, *reinterpret_cast<const T*>(&bv));
    }
    __builtin_nontemporal_store(*reinterpret_cast<const NT*>(&acc),
                                reinterpret_cast<NT*>(out) + i);
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
  // Bucket consecutive slabs by alignment-phase granule so each launch is
  // uniform. In practice all slabs of one collective share a granule.
  int i = 0;
  while (i < n) {
    const int g = slab_granule(descs[i]);
    int j = i + 1;
    while (j < n && slab_granule(descs[j]) == g) ++j;
    switch (g) {
      case 16: launch_bucket<uint4>(descs + i, j - i, stream); break;
      case 8: launch_bucket<unsigned long long>(descs + i, j - i, stream); break;
      case 4: launch_bucket<unsigned int>(descs + i, j - i, stream); break;
      default:
        // sub-dword phase difference: byte-window recombination path
        launch_shift_bucket(descs + i, j - i, stream);
        break;
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


namespace {

// OCP fp8 <-> f32 through the gfx950 hardware conversion ops exposed by
// hip_fp8.h (__hip_fp8_e4m3 / __hip_fp8_e5m2 are the OCP encodings —
// NOT the MI300X fnuz variants; cdna_hip_programming.md §4).
template <bool E5M2>
__device__ inline float fp8_to_f32(unsigned char b) {
  if (E5M2) {
    __hip_fp8_e5m2 v;
    v.__x = b;
    return float(v);
  }
  __hip_fp8_e4m3 v;
  v.__x = b;
  return float(v);
}

template <bool E5M2>
__device__ inline unsigned char f32_to_fp8(float f) {
  if (E5M2) {
    __hip_fp8_e5m2 v(f);
    return v.__x;
  }
  __hip_fp8_e4m3 v(f);
  return v.__x;
}

struct FAdd {
  __device__ static float apply(float a, float b) { return a + b; }
};
struct FMul {
  __device__ static float apply(float a, float b) { return a * b; }
};
struct FMin {
  __device__ static float apply(float a, float b) { return fminf(a, b); }
};
struct FMax {
  __device__ static float apply(float a, float b) { return fmaxf(a, b); }
};

// 16 elements per thread via 16-byte (dwordx4) loads; fp32 accumulators.
// (An 8-byte variant measured 5.0-5.1 TB/s; the wider granule lifts the
// loads to the same dwordx4 form the 6.1 TB/s bitwise kernel uses.)
template <typename Op, bool E5M2>
__global__ __launch_bounds__(256) void fp8_reduce_kernel(
    const unsigned char* __restrict__ in, unsigned char* __restrict__ out,
    long long n, long long nranks) {
  const long long nvec = n / 16;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvec; i += stride) {
    const v4u* base = reinterpret_cast<const v4u*>(in);
    v4u w = __builtin_nontemporal_load(base + i);
    float acc[16];
    for (int d = 0; d < 4; ++d) {
      for (int k = 0; k < 4; ++k) {
        acc[4 * d + k] =
            fp8_to_f32<E5M2>((unsigned char)(w[d] >> (8 * k)));
      }
    }
    for (long long r = 1; r < nranks; ++r) {
      v4u wr = __builtin_nontemporal_load(base + r * nvec + i);
      for (int d = 0; d < 4; ++d) {
        for (int k = 0; k < 4; ++k) {
          acc[4 * d + k] = Op::apply(
              acc[4 * d + k],
              fp8_to_f32<E5M2>((unsigned char)(wr[d] >> (8 * k))));
        }
      }
    }
    v4u o;
    for (int d = 0; d < 4; ++d) {
      unsigned int od = 0;
      for (int k = 0; k < 4; ++k) {
        od |= (unsigned int)f32_to_fp8<E5M2>(acc[4 * d + k]) << (8 * k);
      }
      o[d] = od;
    }
    __builtin_nontemporal_store(o, reinterpret_cast<v4u*>(out) + i);
  }
}

// scalar tail/general path (n not divisible by 8, or unaligned)
template <typename Op, bool E5M2>
__global__ __launch_bounds__(256) void fp8_reduce_kernel_scalar(
    const unsigned char* __restrict__ in, unsigned char* __restrict__ out,
    long long n, long long nranks) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    float acc = fp8_to_f32<E5M2>(in[i]);
    for (long long r = 1; r < nranks; ++r) {
      acc = Op::apply(acc, fp8_to_f32<E5M2>(in[r * n + i]));
    }
    out[i] = f32_to_fp8<E5M2>(acc);
  }
}

template <typename Op, bool E5M2>
void launch_fp8_reduce_t(const void* in, void* out, int64_t n, int nranks,
                         hipStream_t stream) {
  const bool vec = (n % 16 == 0) &&
                   (reinterpret_cast<uintptr_t>(in) % 16 == 0) &&
                   (reinterpret_cast<uintptr_t>(out) % 16 == 0);
  const long long work = vec ? n / 16 : n;
  long long blocks = (work + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  if (vec) {
    hipLaunchKernelGGL((fp8_reduce_kernel<Op, E5M2>), dim3((unsigned)blocks),
                       dim3(256), 0, stream,
                       static_cast<const unsigned char*>(in),
                       static_cast<unsigned char*>(out), (long long)n,
                       (long long)nranks);
  } else {
    hipLaunchKernelGGL((fp8_reduce_kernel_scalar<Op, E5M2>),
                       dim3((unsigned)blocks), dim3(256), 0, stream,
                       static_cast<const unsigned char*>(in),
                       static_cast<unsigned char*>(out), (long long)n,
                       (long long)nranks);
  }
}

} // namespace

void launch_fp8_reduce(const void* in, void* out, int64_t n_elems, int nranks,
                       int op, bool e5m2, hipStream_t stream) {
  auto dispatch = [&](auto opv) {
    using Op = decltype(opv);
    if (e5m2) {
      launch_fp8_reduce_t<Op, true>(in, out, n_elems, nranks, stream);
    } else {
      launch_fp8_reduce_t<Op, false>(in, out, n_elems, nranks, stream);
    }
  };
  switch (op) {
    case 0: dispatch(FAdd{}); break;
    case 1: dispatch(FMul{}); break;
    case 2: dispatch(FMin{}); break;
    default: dispatch(FMax{}); break;
  }
}

// ---------------------------------------------------------------------------
// MINLOC/MAXLOC pair arg-reduction. RCCL (like NCCL) has no MPI pair types;
// the op layer allgathers the (value, location) pairs and this kernel does
// the arg-reduce locally — completing the reference's 12-op table
// (reference csrc/extension.cpp:204-252). Tie-break: smallest location
// (MPI-defined).
// ---------------------------------------------------------------------------

#include <hip/hip_fp16.h>
#include <hip/hip_bfloat16.h>

namespace {

template <typename T>
__device__ inline bool pl_lt(T a, T b) { return a < b; }
template <>
__device__ inline bool pl_lt<__half>(__half a, __half b) {
  return (float)a < (float)b;
}
template <>
__device__ inline bool pl_lt<hip_bfloat16>(hip_bfloat16 a, hip_bfloat16 b) {
  return (float)a < (float)b;
}
template <typename T>
__device__ inline bool pl_eq(T a, T b) { return !pl_lt(a, b) && !pl_lt(b, a); }

template <typename T, bool MAXLOC>
__global__ __launch_bounds__(256) void pairloc_reduce_kernel(
    const T* __restrict__ in, T* __restrict__ out, long long n,
    long long nranks) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    T bv = in[2 * i];
    T bl = in[2 * i + 1];
    for (long long r = 1; r < nranks; ++r) {
      const T v = in[r * 2 * n + 2 * i];
      const T l = in[r * 2 * n + 2 * i + 1];
      const bool better = MAXLOC ? pl_lt(bv, v) : pl_lt(v, bv);
      if (better || (pl_eq(v, bv) && pl_lt(l, bl))) {
        bv = v;
        bl = l;
      }
    }
    out[2 * i] = bv;
    out[2 * i + 1] = bl;
  }
}

template <typename T>
void launch_pairloc_t(const void* in, void* out, int64_t n, int nranks,
                      int op, hipStream_t stream) {
  long long blocks = (n + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  if (op == 0) {
    hipLaunchKernelGGL((pairloc_reduce_kernel<T, false>),
                       dim3((unsigned)blocks), dim3(256), 0, stream,
                       static_cast<const T*>(in), static_cast<T*>(out),
                       (long long)n, (long long)nranks);
  } else {
    hipLaunchKernelGGL((pairloc_reduce_kernel<T, true>),
                       dim3((unsigned)blocks), dim3(256), 0, stream,
                       static_cast<const T*>(in), static_cast<T*>(out),
                       (long long)n, (long long)nranks);
  }
}

} // namespace

void launch_pairloc_reduce(const void* in, void* out, int64_t n_pairs,
                           int nranks, int op, int dtype,
                           hipStream_t stream) {
  switch (dtype) {
    case 0: launch_pairloc_t<float>(in, out, n_pairs, nranks, op, stream); break;
    case 1: launch_pairloc_t<double>(in, out, n_pairs, nranks, op, stream); break;
    case 2: launch_pairloc_t<__half>(in, out, n_pairs, nranks, op, stream); break;
    case 3: launch_pairloc_t<hip_bfloat16>(in, out, n_pairs, nranks, op, stream); break;
    case 4: launch_pairloc_t<signed char>(in, out, n_pairs, nranks, op, stream); break;
    case 5: launch_pairloc_t<unsigned char>(in, out, n_pairs, nranks, op, stream); break;
    case 6: launch_pairloc_t<short>(in, out, n_pairs, nranks, op, stream); break;
    case 7: launch_pairloc_t<int>(in, out, n_pairs, nranks, op, stream); break;
    default: launch_pairloc_t<long long>(in, out, n_pairs, nranks, op, stream); break;
  }
}

} // namespace m4a
