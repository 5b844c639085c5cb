// CDNA4 (gfx950) max-min quantization kernels for the cgx backend.
//
// Design (MI355X-native, not a port):
//  * wavefront = 64 lanes; one WAVE per quantization bucket — for buckets up
//    to 2048 every lane's packs stay in registers between the meta and
//    encode phases (single HBM read), the min/max reduction is a DPP row-op
//    sequence; 16-bit dtypes reduce with packed v_pk_min/max; small
//    power-of-two buckets (< 512) pack 64/(bucket/8) buckets per wave with
//    segmented DPP reductions (QuantSub).
//  * REGISTER-ISOLATED kernel family (the round-2 lesson, measured three
//    times over): the register allocator sizes a kernel for its worst path,
//    so the hot paths live in their own lean launches —
//      k_quantize_fast  (53-89 VGPR; QuantRun, compile-time groups/lane)
//      k_quantize_sub   (small buckets, multi-bucket-per-wave)
//      k_quantize       (generic: error feedback, unaligned, partial-bucket
//                        tails re-entering via kFlagTailOnly, bucket%8!=0
//                        meta pass)
//      k_dequant_fast   (56/33 VGPR; branch-free, incremental bucket index)
//      k_dequant        (generic + ragged tails via kFlagTailOnly)
//    The host (engine.cc run_quantize/run_dequant, bindings) routes slices.
//  * memory-bound workload: 16-byte vectorized loads/stores wherever the
//    slice base is 16B-aligned; decode uses 4 elems/thread for fp32 so every
//    store is one coalesced float4; no integer division or per-access
//    alignment branches in any hot loop (incremental bucket trackers,
//    compile-time alignment/accumulate template flags).
//  * stochastic rounding via a stateless splitmix64 hash per pack (no RNG
//    state buffers, deterministic given the per-launch seed; the hash key is
//    the GLOBAL pack index, so the kernel split is byte-transparent);
//    optional error-feedback residual update fused into the encode.
//
// Wire-format parity with the reference implementation is defined by
// torch_cgx_amd/ops/golden.py (see reference
// src/common/compression/cuda_compression_operations.cu:68-96,219-285).
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <type_traits>

#include "compress.h"

namespace cgx {
namespace {

constexpr float kEps = 1e-10f;
constexpr int kThreads = 256;  // 4 waves
constexpr int kWave = 64;
constexpr int kMaxBlocks = 4096;  // 256 CU * 8 blocks/CU * 2

// One splitmix64-style hash per 8-value pack; each element draws 8 bits of
// uniform randomness ([0,1) in steps of 1/256 -- rounding bias <= 1/512 of a
// quantization unit, negligible vs. the unit-sized quantization error).
__device__ __forceinline__ uint64_t rand_pack(uint64_t seed, uint64_t key) {
  uint64_t z = seed + key * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

__device__ __forceinline__ float rand_lane(uint64_t pack_rand, int j) {
  return static_cast<float>((pack_rand >> (8 * j)) & 0xFF) * (1.0f / 256.0f);
}

// Full-wave min/max reduction via DPP row operations (register path, ~2
// cycles/step) instead of __shfl_xor, which hipcc lowers to six dependent
// ds_bpermute_b32 (LDS latency) per value -- the dominant cost of the
// original wave-per-bucket kernel.
// Sequence: row_shr 1/2/4/8 then row_bcast15/31; lane 63 holds the result.
#define CGX_DPP_STEP(CTRL, OP)                                                t = __builtin_amdgcn_update_dpp(iid, __builtin_bit_cast(int, v), (CTRL),                                    0xf, 0xf, false);                           v = OP(v, __builtin_bit_cast(float, t));

__device__ __forceinline__ void wave_minmax(float& vmin, float& vmax) {
  int t;
  {
    const int iid = __builtin_bit_cast(int, INFINITY);
    float v = vmin;
    CGX_DPP_STEP(0x111, fminf)  // row_shr:1
    CGX_DPP_STEP(0x112, fminf)  // row_shr:2
    CGX_DPP_STEP(0x114, fminf)  // row_shr:4
    CGX_DPP_STEP(0x118, fminf)  // row_shr:8
    CGX_DPP_STEP(0x142, fminf)  // row_bcast:15
    CGX_DPP_STEP(0x143, fminf)  // row_bcast:31
    vmin = __builtin_bit_cast(
        float, __builtin_amdgcn_readlane(__builtin_bit_cast(int, v), 63));
  }
  {
    const int iid = __builtin_bit_cast(int, -INFINITY);
    float v = vmax;
    CGX_DPP_STEP(0x111, fmaxf)
    CGX_DPP_STEP(0x112, fmaxf)
    CGX_DPP_STEP(0x114, fmaxf)
    CGX_DPP_STEP(0x118, fmaxf)
    CGX_DPP_STEP(0x142, fmaxf)
    CGX_DPP_STEP(0x143, fmaxf)
    vmax = __builtin_bit_cast(
        float, __builtin_amdgcn_readlane(__builtin_bit_cast(int, v), 63));
  }
}
#undef CGX_DPP_STEP

template <typename To, typename From>
__device__ __forceinline__ To bitcast(From f) {
  static_assert(sizeof(To) == sizeof(From));
  To t;
  __builtin_memcpy(&t, &f, sizeof(To));
  return t;
}

template <typename T>
struct RawOf;
template <>
struct RawOf<float> {
  using type = uint32_t;
};
template <>
struct RawOf<__half> {
  using type = uint16_t;
};
template <>
struct RawOf<__hip_bfloat16> {
  using type = uint16_t;
};

// All element math runs on raw bit patterns + fp32; conversions below are the
// only dtype-specific pieces.  T-precision ops (decode mul/add, accumulate)
// are expressed as fp32 ops + round-to-T, which is bitwise identical to
// native T arithmetic because products/sums of two T values are exact in
// fp32 for all of {fp32, fp16, bf16}.
template <typename T>
__device__ __forceinline__ float raw2f(uint32_t r);
template <>
__device__ __forceinline__ float raw2f<float>(uint32_t r) {
  return bitcast<float>(r);
}
template <>
__device__ __forceinline__ float raw2f<__half>(uint32_t r) {
  return __half2float(bitcast<__half>(static_cast<uint16_t>(r)));
}
template <>
__device__ __forceinline__ float raw2f<__hip_bfloat16>(uint32_t r) {
  return __bfloat162float(bitcast<__hip_bfloat16>(static_cast<uint16_t>(r)));
}

template <typename T>
__device__ __forceinline__ uint32_t f2raw(float f);
template <>
__device__ __forceinline__ uint32_t f2raw<float>(float f) {
  return bitcast<uint32_t>(f);
}
template <>
__device__ __forceinline__ uint32_t f2raw<__half>(float f) {
  return bitcast<uint16_t>(__float2half(f));
}
template <>
__device__ __forceinline__ uint32_t f2raw<__hip_bfloat16>(float f) {
  return bitcast<uint16_t>(__float2bfloat16(f));
}

// Packed-pair min/max for 16-bit dtypes (v_pk_min/max_f16|bf16): phase A of
// the fp16/bf16 quantize was 87% VALU-busy with per-element convert+min+max;
// packed ops process 2 elements per instruction without unpacking.
template <typename T>
struct Pk2;
template <>
struct Pk2<__half> {
  using P = _Float16 __attribute__((ext_vector_type(2)));
  static constexpr uint32_t kInf = 0x7C007C00u;   // +inf, +inf
  static constexpr uint32_t kNInf = 0xFC00FC00u;  // -inf, -inf
  static __device__ __forceinline__ P min(P a, P b) {
    return __builtin_elementwise_min(a, b);
  }
  static __device__ __forceinline__ P max(P a, P b) {
    return __builtin_elementwise_max(a, b);
  }
  static __device__ __forceinline__ float lo(P v) {
    return static_cast<float>(v[0]);
  }
  static __device__ __forceinline__ float hi(P v) {
    return static_cast<float>(v[1]);
  }
};
template <>
struct Pk2<__hip_bfloat16> {
  using P = __bf16 __attribute__((ext_vector_type(2)));
  static constexpr uint32_t kInf = 0x7F807F80u;
  static constexpr uint32_t kNInf = 0xFF80FF80u;
  static __device__ __forceinline__ P min(P a, P b) {
    return __builtin_elementwise_min(a, b);
  }
  static __device__ __forceinline__ P max(P a, P b) {
    return __builtin_elementwise_max(a, b);
  }
  static __device__ __forceinline__ float lo(P v) {
    return static_cast<float>(v[0]);
  }
  static __device__ __forceinline__ float hi(P v) {
    return static_cast<float>(v[1]);
  }
};

// 8-element vector load/store (one pack).  aligned16 is wave-uniform for the
// quantize kernel (slice bases are pack-aligned) and per-thread for dequant.
template <typename T>
__device__ __forceinline__ void load8(const T* p, bool aligned16,
                                      uint32_t (&r)[8]) {
  if constexpr (sizeof(T) == 4) {
    if (aligned16) {
      int4 a = *reinterpret_cast<const int4*>(p);
      int4 b = *reinterpret_cast<const int4*>(p + 4);
      r[0] = a.x; r[1] = a.y; r[2] = a.z; r[3] = a.w;
      r[4] = b.x; r[5] = b.y; r[6] = b.z; r[7] = b.w;
      return;
    }
  } else {
    if (aligned16) {
      int4 a = *reinterpret_cast<const int4*>(p);
      const uint32_t w[4] = {static_cast<uint32_t>(a.x),
                             static_cast<uint32_t>(a.y),
                             static_cast<uint32_t>(a.z),
                             static_cast<uint32_t>(a.w)};
#pragma unroll
      for (int j = 0; j < 8; j++) r[j] = (w[j >> 1] >> ((j & 1) * 16)) & 0xFFFF;
      return;
    }
  }
  using R = typename RawOf<T>::type;
  const R* q = reinterpret_cast<const R*>(p);
#pragma unroll
  for (int j = 0; j < 8; j++) r[j] = q[j];
}

template <typename T>
__device__ __forceinline__ void store8(T* p, bool aligned16,
                                       const uint32_t (&r)[8]) {
  if constexpr (sizeof(T) == 4) {
    if (aligned16) {
      int4 a = {static_cast<int>(r[0]), static_cast<int>(r[1]),
                static_cast<int>(r[2]), static_cast<int>(r[3])};
      int4 b = {static_cast<int>(r[4]), static_cast<int>(r[5]),
                static_cast<int>(r[6]), static_cast<int>(r[7])};
      *reinterpret_cast<int4*>(p) = a;
      *reinterpret_cast<int4*>(p + 4) = b;
      return;
    }
  } else {
    if (aligned16) {
      int4 a;
      a.x = static_cast<int>(r[0] | (r[1] << 16));
      a.y = static_cast<int>(r[2] | (r[3] << 16));
      a.z = static_cast<int>(r[4] | (r[5] << 16));
      a.w = static_cast<int>(r[6] | (r[7] << 16));
      *reinterpret_cast<int4*>(p) = a;
      return;
    }
  }
  using R = typename RawOf<T>::type;
  R* q = reinterpret_cast<R*>(p);
#pragma unroll
  for (int j = 0; j < 8; j++) q[j] = static_cast<R>(r[j]);
}

__device__ __forceinline__ void load4f(const float* p, bool aligned16,
                                       uint32_t (&r)[4]) {
  if (aligned16) {
    const int4 a = *reinterpret_cast<const int4*>(p);
    r[0] = a.x; r[1] = a.y; r[2] = a.z; r[3] = a.w;
    return;
  }
  const uint32_t* q = reinterpret_cast<const uint32_t*>(p);
#pragma unroll
  for (int j = 0; j < 4; j++) r[j] = q[j];
}

__device__ __forceinline__ void store4f(float* p, bool aligned16,
                                        const uint32_t (&r)[4]) {
  if (aligned16) {
    const int4 a = {static_cast<int>(r[0]), static_cast<int>(r[1]),
                    static_cast<int>(r[2]), static_cast<int>(r[3])};
    *reinterpret_cast<int4*>(p) = a;
    return;
  }
  uint32_t* q = reinterpret_cast<uint32_t*>(p);
#pragma unroll
  for (int j = 0; j < 4; j++) q[j] = r[j];
}

// Write the low `nb` bytes of v to p (little-endian), widest aligned stores.
__device__ __forceinline__ void store_bytes(uint8_t* p, uint64_t v, int nb) {
  const uintptr_t a = reinterpret_cast<uintptr_t>(p);
  if (nb == 8 && (a & 7) == 0) {
    *reinterpret_cast<uint64_t*>(p) = v;
    return;
  }
  if (nb == 4 && (a & 3) == 0) {
    *reinterpret_cast<uint32_t*>(p) = static_cast<uint32_t>(v);
    return;
  }
  if (nb == 2 && (a & 1) == 0) {
    *reinterpret_cast<uint16_t*>(p) = static_cast<uint16_t>(v);
    return;
  }
  for (int i = 0; i < nb; i++) p[i] = static_cast<uint8_t>(v >> (8 * i));
}

__device__ __forceinline__ uint64_t load_bytes(const uint8_t* p, int nb) {
  const uintptr_t a = reinterpret_cast<uintptr_t>(p);
  if (nb == 8) {
    if ((a & 7) == 0) return *reinterpret_cast<const uint64_t*>(p);
    if ((a & 3) == 0) {
      const uint32_t lo = reinterpret_cast<const uint32_t*>(p)[0];
      const uint32_t hi = reinterpret_cast<const uint32_t*>(p)[1];
      return static_cast<uint64_t>(lo) | (static_cast<uint64_t>(hi) << 32);
    }
  }
  if (nb == 4) {
    if ((a & 3) == 0) return *reinterpret_cast<const uint32_t*>(p);
    if ((a & 1) == 0) {
      const uint16_t lo = reinterpret_cast<const uint16_t*>(p)[0];
      const uint16_t hi = reinterpret_cast<const uint16_t*>(p)[1];
      return static_cast<uint64_t>(lo) | (static_cast<uint64_t>(hi) << 16);
    }
  }
  if (nb == 2 && (a & 1) == 0) return *reinterpret_cast<const uint16_t*>(p);
  uint64_t v = 0;
  for (int i = 0; i < nb; i++) v |= static_cast<uint64_t>(p[i]) << (8 * i);
  return v;
}

// Register-resident quantize of a RUN of full buckets (bucket % 8 == 0, 16B
// aligned, no error feedback) with a COMPILE-TIME groups-per-lane count G.
// Two design points vs the round-1 per-bucket fast path:
//  * G is a template parameter, so the stash loops fully unroll (the
//    runtime-trip form compiled into s_set_gpr_idx register-indexed access);
//  * the wave pipelines bucket i+1's HBM loads under bucket i's
//    reduce+encode (two stash banks, statically unrolled 2x steady state) —
//    the PMC profile showed SQ_WAIT_ANY ~17x SQ_BUSY on the serial
//    load->reduce->encode chain.
// 16-bit dtypes stash the RAW dwords (w[G][4], half the registers) and run
// packed v_pk_min/max on them; elements are extracted at encode time.
template <typename T, int BITS, int G>
struct QuantRun {
  using R = typename RawOf<T>::type;
  static constexpr int kWords = sizeof(T) == 4 ? 8 : 4;
  struct Stash {
    uint32_t w[G][kWords];
  };

  const T* __restrict__ base;
  R* __restrict__ meta;
  uint8_t* __restrict__ packed;
  int lane;
  int ngroups;  // per bucket
  float divisor;
  uint64_t seed;
  int stochastic;
  int slice_idx;

  __device__ __forceinline__ void load(Stash& s, int64_t lb) const {
    load_from(s, base + lb * (int64_t)(ngroups * 8));
  }

  // arbitrary 16B-aligned source (global or LDS via generic pointer)
  __device__ __forceinline__ void load_from(Stash& s, const T* in) const {
#pragma unroll
    for (int k = 0; k < G; k++) {
      const int g = lane + k * kWave;
      if (g >= ngroups) continue;  // buckets < 512: extra lanes idle
      const int4 a = *reinterpret_cast<const int4*>(
          __builtin_assume_aligned(in + g * 8, 16));
      if constexpr (sizeof(T) == 4) {
        const int4 b = *reinterpret_cast<const int4*>(
            __builtin_assume_aligned(in + g * 8 + 4, 16));
        s.w[k][0] = a.x; s.w[k][1] = a.y; s.w[k][2] = a.z; s.w[k][3] = a.w;
        s.w[k][4] = b.x; s.w[k][5] = b.y; s.w[k][6] = b.z; s.w[k][7] = b.w;
      } else {
        s.w[k][0] = a.x; s.w[k][1] = a.y; s.w[k][2] = a.z; s.w[k][3] = a.w;
      }
    }
  }

  __device__ __forceinline__ uint32_t elem(const Stash& s, int k,
                                           int j) const {
    if constexpr (sizeof(T) == 4) {
      return s.w[k][j];
    } else {
      return (s.w[k][j >> 1] >> ((j & 1) * 16)) & 0xFFFF;
    }
  }

  __device__ __forceinline__ void encode(const Stash& s, int64_t lb) const {
    float lmin = INFINITY, lmax = -INFINITY;
    if constexpr (sizeof(T) == 2) {
      using PK = Pk2<T>;
      typename PK::P pmin = bitcast<typename PK::P>(PK::kInf);
      typename PK::P pmax = bitcast<typename PK::P>(PK::kNInf);
#pragma unroll
      for (int k = 0; k < G; k++) {
        const int g = lane + k * kWave;
        if (g >= ngroups) continue;
#pragma unroll
        for (int q = 0; q < 4; q++) {
          const auto v = bitcast<typename PK::P>(s.w[k][q]);
          pmin = PK::min(pmin, v);
          pmax = PK::max(pmax, v);
        }
      }
      lmin = fminf(PK::lo(pmin), PK::hi(pmin));
      lmax = fmaxf(PK::lo(pmax), PK::hi(pmax));
    } else {
#pragma unroll
      for (int k = 0; k < G; k++) {
        const int g = lane + k * kWave;
        if (g >= ngroups) continue;
#pragma unroll
        for (int j = 0; j < 8; j++) {
          const float f = raw2f<T>(s.w[k][j]);
          lmin = fminf(lmin, f);
          lmax = fmaxf(lmax, f);
        }
      }
    }
    wave_minmax(lmin, lmax);
    const uint32_t unit_raw = f2raw<T>((lmax - lmin) / divisor);
    const float unitf = raw2f<T>(unit_raw);
    const float minf = lmin;
    if (lane == 0) {
      meta[2 * lb] = static_cast<R>(unit_raw);
      meta[2 * lb + 1] = static_cast<R>(f2raw<T>(lmin));
    }
    const bool live = unitf >= kEps;
    const float rinv = live ? 1.0f / unitf : 0.0f;
    const int64_t gbase = lb * (int64_t)ngroups;
    using acc_t =
        typename std::conditional<(BITS <= 4), uint32_t, uint64_t>::type;
#pragma unroll
    for (int k = 0; k < G; k++) {
      const int g = lane + k * kWave;
      if (g >= ngroups) continue;
      acc_t value = 0;
      if (live) {
        const uint64_t pr =
            stochastic
                ? rand_pack(seed, (static_cast<uint64_t>(slice_idx) << 44) |
                                      static_cast<uint64_t>(gbase + g))
                : 0;
#pragma unroll
        for (int j = 0; j < 8; j++) {
          const float rnd = stochastic ? rand_lane(pr, j) : 0.5f;
          const float dd = (raw2f<T>(elem(s, k, j)) - minf) * rinv + rnd;
          const uint32_t level =
              static_cast<uint32_t>(fminf(floorf(dd), divisor));
          value |=
              static_cast<acc_t>(level & ((1u << BITS) - 1)) << (j * BITS);
        }
      }
      store_bytes(packed + (gbase + g) * BITS, static_cast<uint64_t>(value),
                  BITS);
    }
  }

  // count buckets at lb0, lb0+nw, ...  Single stash bank (used by the
  // MAXGT>=2 kernel variants, where a second bank costs occupancy — the
  // two-bank pipeline was measured and rejected twice, see RESULTS.md).
  __device__ __forceinline__ void run(int64_t lb0, int64_t count,
                                      int64_t nw) const {
    Stash A;
    for (int64_t i = 0; i < count; i++) {
      load(A, lb0 + i * nw);
      encode(A, lb0 + i * nw);
    }
  }

};

// Sub-wave quantize for SMALL buckets (bucket/8 = L lanes per bucket, a
// power of two < 64): the wave-per-bucket layout leaves (64-L) lanes idle
// and serializes 64/L x more buckets per wave; here the wave processes
// W = 64/L buckets at once, reducing each L-lane segment with the DPP
// ladder truncated at L and broadcasting the segment result back with one
// ds_bpermute per value.  Bucket 256 measured 1.3-2.4 TB/s on the
// wave-per-bucket path; this recovers most of the bucket-1024 rate.
template <int L>
__device__ __forceinline__ void seg_minmax(float& vmin, float& vmax,
                                           int lane) {
  if constexpr (L > 1) {
    int t;
#define CGX_DPP_SEG(CTRL, OP, IDENT)                                       \
  t = __builtin_amdgcn_update_dpp(IDENT, __builtin_bit_cast(int, v),       \
                                  (CTRL), 0xf, 0xf, false);                \
  v = OP(v, __builtin_bit_cast(float, t));
    {
      const int iid = __builtin_bit_cast(int, INFINITY);
      float v = vmin;
      CGX_DPP_SEG(0x111, fminf, iid)
      if constexpr (L >= 4) CGX_DPP_SEG(0x112, fminf, iid)
      if constexpr (L >= 8) CGX_DPP_SEG(0x114, fminf, iid)
      if constexpr (L >= 16) CGX_DPP_SEG(0x118, fminf, iid)
      if constexpr (L >= 32) CGX_DPP_SEG(0x142, fminf, iid)
      vmin = v;
    }
    {
      const int iid = __builtin_bit_cast(int, -INFINITY);
      float v = vmax;
      CGX_DPP_SEG(0x111, fmaxf, iid)
      if constexpr (L >= 4) CGX_DPP_SEG(0x112, fmaxf, iid)
      if constexpr (L >= 8) CGX_DPP_SEG(0x114, fmaxf, iid)
      if constexpr (L >= 16) CGX_DPP_SEG(0x118, fmaxf, iid)
      if constexpr (L >= 32) CGX_DPP_SEG(0x142, fmaxf, iid)
      vmax = v;
    }
#undef CGX_DPP_SEG
    // segment result sits at lane (s+1)*L-1; broadcast it to the segment
    const int src = ((lane / L) * L + (L - 1)) * 4;
    vmin = __builtin_bit_cast(
        float, __builtin_amdgcn_ds_bpermute(src,
                                            __builtin_bit_cast(int, vmin)));
    vmax = __builtin_bit_cast(
        float, __builtin_amdgcn_ds_bpermute(src,
                                            __builtin_bit_cast(int, vmax)));
  }
}

template <typename T, int BITS, int L>
struct QuantSub {
  using R = typename RawOf<T>::type;
  const T* __restrict__ base;
  R* __restrict__ meta;
  uint8_t* __restrict__ packed;
  int lane;
  float divisor;
  uint64_t seed;
  int stochastic;
  int slice_idx;

  // buckets lb0, lb0+nw, ..., count of them; W = 64/L per wave iteration
  __device__ void run(int64_t lb0, int64_t count, int64_t nw) const {
    constexpr int W = kWave / L;
    const int s = lane / L;   // my segment = which of the W buckets
    const int li = lane % L;  // my 8-elem group within the bucket
    for (int64_t i = 0; i < count; i += W) {
      const bool active = (i + s) < count;
      const int64_t lb = lb0 + (i + s) * nw;
      uint32_t r[8];
      float lmin = INFINITY, lmax = -INFINITY;
      if (active) {
        const T* in = base + lb * (int64_t)(L * 8) + li * 8;
        load8<T>(in, true, r);
#pragma unroll
        for (int j = 0; j < 8; j++) {
          const float f = raw2f<T>(r[j]);
          lmin = fminf(lmin, f);
          lmax = fmaxf(lmax, f);
        }
      }
      seg_minmax<L>(lmin, lmax, lane);
      if (!active) continue;
      const uint32_t unit_raw = f2raw<T>((lmax - lmin) / divisor);
      const float unitf = raw2f<T>(unit_raw);
      const float minf = lmin;
      if (li == 0) {
        meta[2 * lb] = static_cast<R>(unit_raw);
        meta[2 * lb + 1] = static_cast<R>(f2raw<T>(lmin));
      }
      const int64_t g = lb * L + li;  // global group index in the slice
      using acc_t =
          typename std::conditional<(BITS <= 4), uint32_t, uint64_t>::type;
      acc_t value = 0;
      if (unitf >= kEps) {
        const float rinv = 1.0f / unitf;
        const uint64_t pr =
            stochastic
                ? rand_pack(seed, (static_cast<uint64_t>(slice_idx) << 44) |
                                      static_cast<uint64_t>(g))
                : 0;
#pragma unroll
        for (int j = 0; j < 8; j++) {
          const float rnd = stochastic ? rand_lane(pr, j) : 0.5f;
          const float dd = (raw2f<T>(r[j]) - minf) * rinv + rnd;
          const uint32_t level =
              static_cast<uint32_t>(fminf(floorf(dd), divisor));
          value |=
              static_cast<acc_t>(level & ((1u << BITS) - 1)) << (j * BITS);
        }
      }
      store_bytes(packed + g * BITS, static_cast<uint64_t>(value), BITS);
    }
  }
};

// Lean fast-path kernel (see compress.h launch_quantize_fast): only the
// QuantRun register path is instantiated, so the register allocator is not
// forced to the generic/EF paths' footprint (k_quantize<float,4,true>
// allocates 139 VGPRs = 3 waves/SIMD; this kernel ~half that).
template <typename T, int BITS, int MAXGT>
__global__ __launch_bounds__(kThreads) void k_quantize_fast(
    const QuantDesc* __restrict__ descs, const int64_t* __restrict__ cum,
    int nslices, int64_t total_buckets, uint64_t seed, int stochastic) {
  using R = typename RawOf<T>::type;
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wid = static_cast<int64_t>(blockIdx.x) * (kThreads / kWave) +
                      (threadIdx.x / kWave);
  const int64_t nw = static_cast<int64_t>(gridDim.x) * (kThreads / kWave);
  constexpr float divisor = static_cast<float>((1 << BITS) - 1);
  int64_t b = wid;
  while (b < total_buckets) {
    int lo = 0, hi = nslices;
    while (hi - lo > 1) {
      const int mid = (lo + hi) >> 1;
      if (cum[mid] <= b) lo = mid; else hi = mid;
    }
    const QuantDesc d = descs[lo];
    const bool skip = (d.flags & kFlagSkipIncomplete) != 0;
    const int64_t nb_slice =
        skip ? d.n / d.bucket
             : (d.n + d.bucket - 1) / static_cast<int64_t>(d.bucket);
    const int64_t nb_full = cum[lo + 1] - cum[lo];
    const int64_t lb = b - cum[lo];
    const int ngroups = d.bucket >> 3;
    const int64_t count = (nb_full - lb + nw - 1) / nw;
#define CGX_QRUNF(GV)                                                      \
  do {                                                                     \
    QuantRun<T, BITS, GV> qr{reinterpret_cast<const T*>(d.in),             \
                             reinterpret_cast<R*>(d.out),                  \
                             reinterpret_cast<uint8_t*>(d.out) +           \
                                 2 * sizeof(R) * nb_slice,                 \
                             lane,    ngroups, divisor,                    \
                             seed,    stochastic, lo};                     \
    qr.run(lb, count, nw);                                                 \
  } while (0)
    if constexpr (MAXGT <= 2) {
      // all buckets <= 1024: half the stash registers -> higher occupancy.
      // (A two-bank pipelined run was retried here after the lean-kernel
      // split: VGPR 53 -> 94, quantize 0.072 -> 0.077 ms — the second bank
      // still costs more occupancy than its prefetch hides.  Rejected.)
      if (((ngroups + kWave - 1) >> 6) <= 1) CGX_QRUNF(1);
      else CGX_QRUNF(2);
    } else {
      switch ((ngroups + kWave - 1) >> 6) {
        case 1: CGX_QRUNF(1); break;
        case 2: CGX_QRUNF(2); break;
        case 3: CGX_QRUNF(3); break;
        default: CGX_QRUNF(4); break;
      }
    }
#undef CGX_QRUNF
    b += count * nw;
  }
}

// Dedicated small-bucket kernel: ONLY QuantSub is instantiated (folding it
// into k_quantize_fast measured VGPR 53 -> 107/143, regressing the large
// buckets -- the same register-allocation coupling the lean split fixed).
template <typename T, int BITS>
__global__ __launch_bounds__(kThreads) void k_quantize_sub(
    const QuantDesc* __restrict__ descs, const int64_t* __restrict__ cum,
    int nslices, int64_t total_buckets, uint64_t seed, int stochastic) {
  using R = typename RawOf<T>::type;
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wid = static_cast<int64_t>(blockIdx.x) * (kThreads / kWave) +
                      (threadIdx.x / kWave);
  const int64_t nw = static_cast<int64_t>(gridDim.x) * (kThreads / kWave);
  constexpr float divisor = static_cast<float>((1 << BITS) - 1);
  int64_t b = wid;
  while (b < total_buckets) {
    int lo = 0, hi = nslices;
    while (hi - lo > 1) {
      const int mid = (lo + hi) >> 1;
      if (cum[mid] <= b) lo = mid; else hi = mid;
    }
    const QuantDesc d = descs[lo];
    const bool skip = (d.flags & kFlagSkipIncomplete) != 0;
    const int64_t nb_slice =
        skip ? d.n / d.bucket
             : (d.n + d.bucket - 1) / static_cast<int64_t>(d.bucket);
    const int64_t nb_full = cum[lo + 1] - cum[lo];
    const int64_t lb = b - cum[lo];
    const int ngroups = d.bucket >> 3;
    const int64_t count = (nb_full - lb + nw - 1) / nw;
#define CGX_QSUB(LV)                                                       \
  do {                                                                     \
    QuantSub<T, BITS, LV> q{reinterpret_cast<const T*>(d.in),              \
                            reinterpret_cast<R*>(d.out),                   \
                            reinterpret_cast<uint8_t*>(d.out) +            \
                                2 * sizeof(R) * nb_slice,                  \
                            lane, divisor, seed, stochastic, lo};          \
    q.run(lb, count, nw);                                                  \
  } while (0)
    switch (ngroups) {
      case 1: CGX_QSUB(1); break;
      case 2: CGX_QSUB(2); break;
      case 4: CGX_QSUB(4); break;
      case 8: CGX_QSUB(8); break;
      case 16: CGX_QSUB(16); break;
      default: CGX_QSUB(32); break;
    }
#undef CGX_QSUB
    b += count * nw;
  }
}

// ---------------------------------------------------------------------------
// Quantize: one wave per bucket.  ENCODE=true requires bucket % 8 == 0 for
// every slice (host guarantees); ENCODE=false computes meta only (the
// arbitrary-bucket path then packs with k_pack_generic).
// ---------------------------------------------------------------------------
template <typename T, int BITS, bool ENCODE>
__global__ __launch_bounds__(kThreads) void k_quantize(
    const QuantDesc* __restrict__ descs, const int64_t* __restrict__ cum,
    int nslices, int64_t total_buckets, uint64_t seed, int stochastic) {
  using R = typename RawOf<T>::type;
  const int lane = threadIdx.x & (kWave - 1);
  const int64_t wid =
      static_cast<int64_t>(blockIdx.x) * (kThreads / kWave) + (threadIdx.x / kWave);
  const int64_t nw = static_cast<int64_t>(gridDim.x) * (kThreads / kWave);
  constexpr float divisor = static_cast<float>((1 << BITS) - 1);
  constexpr int MAXG = 4;  // register-stashed packs per lane (bucket <= 2048)

  for (int64_t b = wid; b < total_buckets; b += nw) {
    int lo = 0, hi = nslices;
    while (hi - lo > 1) {
      const int mid = (lo + hi) >> 1;
      if (cum[mid] <= b) lo = mid; else hi = mid;
    }
    const QuantDesc d = descs[lo];
    const bool skip = (d.flags & kFlagSkipIncomplete) != 0;
    const int64_t nq =
        skip ? (d.n / d.bucket) * static_cast<int64_t>(d.bucket) : d.n;
    int64_t nb_slice = cum[lo + 1] - cum[lo];
    int64_t lb = b - cum[lo];
    if (d.flags & kFlagTailOnly) {
      // full buckets were handled by k_quantize_fast; this desc's single
      // cum entry maps to the trailing partial bucket
      lb += d.n / d.bucket;
      nb_slice = (d.n + d.bucket - 1) / static_cast<int64_t>(d.bucket);
    }
    const int64_t bstart = lb * static_cast<int64_t>(d.bucket);
    const int cur = static_cast<int>(min(static_cast<int64_t>(d.bucket), nq - bstart));
    const T* in = reinterpret_cast<const T*>(d.in) + bstart;
    const bool al16 = (reinterpret_cast<uintptr_t>(in) & 15) == 0;
    const int ngroups = (cur + 7) >> 3;

    uint32_t stash[MAXG][8];
    float lmin = INFINITY, lmax = -INFINITY;
    // fast path: full bucket, whole groups per lane (wave-uniform branch)
    const bool full = cur == d.bucket && (cur & 7) == 0 && al16;
    T* const fbp = d.fb ? reinterpret_cast<T*>(d.fb) + bstart : nullptr;
    if (full && ngroups <= MAXG * kWave && !fbp) {
      if constexpr (sizeof(T) == 2) {
        using PK = Pk2<T>;
        typename PK::P pmin = bitcast<typename PK::P>(PK::kInf);
        typename PK::P pmax = bitcast<typename PK::P>(PK::kNInf);
        for (int g = lane, gi = 0; g < ngroups; g += kWave, gi++) {
          const int4 a = *reinterpret_cast<const int4*>(in + g * 8);
          const uint32_t w[4] = {static_cast<uint32_t>(a.x),
                                 static_cast<uint32_t>(a.y),
                                 static_cast<uint32_t>(a.z),
                                 static_cast<uint32_t>(a.w)};
#pragma unroll
          for (int k = 0; k < 4; k++) {
            const auto v = bitcast<typename PK::P>(w[k]);
            pmin = PK::min(pmin, v);
            pmax = PK::max(pmax, v);
          }
#pragma unroll
          for (int j = 0; j < 8; j++)
            stash[gi][j] = (w[j >> 1] >> ((j & 1) * 16)) & 0xFFFF;
        }
        lmin = fminf(PK::lo(pmin), PK::hi(pmin));
        lmax = fmaxf(PK::lo(pmax), PK::hi(pmax));
      } else {
        for (int g = lane, gi = 0; g < ngroups; g += kWave, gi++) {
          load8<T>(in + g * 8, true, stash[gi]);
#pragma unroll
          for (int j = 0; j < 8; j++) {
            const float f = raw2f<T>(stash[gi][j]);
            lmin = fminf(lmin, f);
            lmax = fmaxf(lmax, f);
          }
        }
      }
    } else if (full && ngroups <= MAXG * kWave && fbp) {
      // error-feedback: stash the T-rounded x+fb and reduce over it.
      // fb alignment is independent of the input's (the engine rebases the
      // phase-2 feedback base by a partition offset), so probe it separately.
      const bool al16_fb = (reinterpret_cast<uintptr_t>(fbp) & 15) == 0;
      for (int g = lane, gi = 0; g < ngroups; g += kWave, gi++) {
        uint32_t xr[8], fr[8];
        load8<T>(in + g * 8, true, xr);
        load8<T>(fbp + g * 8, al16_fb, fr);
#pragma unroll
        for (int j = 0; j < 8; j++) {
          const uint32_t xe = f2raw<T>(raw2f<T>(xr[j]) + raw2f<T>(fr[j]));
          stash[gi][j] = xe;
          const float f = raw2f<T>(xe);
          lmin = fminf(lmin, f);
          lmax = fmaxf(lmax, f);
        }
      }
    } else {
      for (int g = lane, gi = 0; g < ngroups; g += kWave, gi++) {
        uint32_t r[8];
        const int m = min(8, cur - g * 8);
        if (m == 8) {
          load8<T>(in + g * 8, al16, r);
        } else {
          const R* q = reinterpret_cast<const R*>(in) + g * 8;
          for (int j = 0; j < m; j++) r[j] = q[j];
          for (int j = m; j < 8; j++) r[j] = 0;
        }
        if (fbp) {
          const R* q = reinterpret_cast<const R*>(fbp) + g * 8;
          for (int j = 0; j < m; j++)
            r[j] = f2raw<T>(raw2f<T>(r[j]) + raw2f<T>(q[j]));
        }
        if (gi < MAXG) {
#pragma unroll
          for (int j = 0; j < 8; j++) stash[gi][j] = r[j];
        }
        for (int j = 0; j < m; j++) {
          const float f = raw2f<T>(r[j]);
          lmin = fminf(lmin, f);
          lmax = fmaxf(lmax, f);
        }
      }
    }
    wave_minmax(lmin, lmax);
    const uint32_t unit_raw = f2raw<T>((lmax - lmin) / divisor);
    const float unitf = raw2f<T>(unit_raw);
    const float minf = lmin;
    R* meta = reinterpret_cast<R*>(d.out);
    if (lane == 0) {
      meta[2 * lb] = static_cast<R>(unit_raw);
      meta[2 * lb + 1] = static_cast<R>(f2raw<T>(lmin));
    }

    if constexpr (ENCODE) {
      uint8_t* packed =
          reinterpret_cast<uint8_t*>(d.out) + 2 * sizeof(R) * nb_slice;
      const int64_t num_char = (nq * BITS + 7) >> 3;
      const int64_t gbase = bstart >> 3;
      const bool live = unitf >= kEps;
      const float rinv = 1.0f / unitf;  // hoisted: fp32 div is 1/4 VALU rate
      if (full && ngroups <= MAXG * kWave && live && !fbp) {
        // 8 levels fit a uint32 when BITS <= 4: halves the shift/or cost
        using acc_t =
            typename std::conditional<(BITS <= 4), uint32_t, uint64_t>::type;
        for (int g = lane, gi = 0; g < ngroups; g += kWave, gi++) {
          const uint64_t pr =
              stochastic
                  ? rand_pack(seed, (static_cast<uint64_t>(lo) << 44) |
                                        static_cast<uint64_t>(gbase + g))
                  : 0;
          acc_t value = 0;
#pragma unroll
          for (int j = 0; j < 8; j++) {
            const float rnd = stochastic ? rand_lane(pr, j) : 0.5f;
            const float dd = (raw2f<T>(stash[gi][j]) - minf) * rinv + rnd;
            const uint32_t level =
                static_cast<uint32_t>(fminf(floorf(dd), divisor));
            value |= static_cast<acc_t>(level & ((1u << BITS) - 1))
                     << (j * BITS);
          }
          store_bytes(packed + (gbase + g) * BITS,
                      static_cast<uint64_t>(value), BITS);
        }
        continue;
      }
      for (int g = lane, gi = 0; g < ngroups; g += kWave, gi++) {
        uint32_t r[8];
        const int m = min(8, cur - g * 8);
        if (gi < MAXG) {
#pragma unroll
          for (int j = 0; j < 8; j++) r[j] = stash[gi][j];
        } else if (m == 8 && !fbp) {
          load8<T>(in + g * 8, al16, r);
        } else {
          const R* q = reinterpret_cast<const R*>(in) + g * 8;
          for (int j = 0; j < m; j++) r[j] = q[j];
          if (fbp) {
            const R* f = reinterpret_cast<const R*>(fbp) + g * 8;
            for (int j = 0; j < m; j++)
              r[j] = f2raw<T>(raw2f<T>(r[j]) + raw2f<T>(f[j]));
          }
        }
        uint64_t value = 0;
        uint32_t lv[8] = {};
        if (live) {
          const uint64_t pr =
              stochastic ? rand_pack(seed, (static_cast<uint64_t>(lo) << 44) |
                                               static_cast<uint64_t>(gbase + g))
                         : 0;
          for (int j = 0; j < m; j++) {
            const float rnd = stochastic ? rand_lane(pr, j) : 0.5f;
            const float dd = (raw2f<T>(r[j]) - minf) * rinv + rnd;
            const uint32_t level =
                static_cast<uint32_t>(fminf(floorf(dd), divisor));
            lv[j] = level;
            value |= static_cast<uint64_t>(level & ((1u << BITS) - 1))
                     << (j * BITS);
          }
        }
        if (fbp) {
          // residual: (x+fb) - decode(level), rounded to T per the spec
          R* f = reinterpret_cast<R*>(fbp) + g * 8;
          for (int j = 0; j < m; j++) {
            const uint32_t prod =
                f2raw<T>(unitf * static_cast<float>(live ? lv[j] : 0));
            const uint32_t dec = f2raw<T>(minf + raw2f<T>(prod));
            f[j] = static_cast<R>(
                f2raw<T>(raw2f<T>(r[j]) - raw2f<T>(dec)));
          }
        }
        const int64_t gb = (gbase + g) * BITS;
        const int nbytes =
            static_cast<int>(min(static_cast<int64_t>(BITS), num_char - gb));
        store_bytes(packed + gb, value, nbytes);
      }
    }
  }
}

// Generic pack phase for bucket % 8 != 0 (meta already computed): grid-stride
// threads over packs, per-element bucket lookup.
template <typename T, int BITS>
__global__ __launch_bounds__(kThreads) void k_pack_generic(
    const QuantDesc* __restrict__ descs, int nslices, uint64_t seed,
    int stochastic) {
  using R = typename RawOf<T>::type;
  const int64_t t0 =
      static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  constexpr float divisor = static_cast<float>((1 << BITS) - 1);
  for (int s = 0; s < nslices; s++) {
    const QuantDesc d = descs[s];
    const bool skip = (d.flags & kFlagSkipIncomplete) != 0;
    const int64_t nq =
        skip ? (d.n / d.bucket) * static_cast<int64_t>(d.bucket) : d.n;
    const int64_t ngroups = (nq + 7) >> 3;
    const int64_t nb_slice =
        skip ? d.n / d.bucket : (d.n + d.bucket - 1) / d.bucket;
    const R* meta = reinterpret_cast<const R*>(d.out);
    uint8_t* packed =
        reinterpret_cast<uint8_t*>(d.out) + 2 * sizeof(R) * nb_slice;
    const int64_t num_char = (nq * BITS + 7) >> 3;
    const R* in = reinterpret_cast<const R*>(d.in);
    const bool small = d.n < (int64_t(1) << 31);
    for (int64_t g = t0; g < ngroups; g += stride) {
      const int m = static_cast<int>(min(static_cast<int64_t>(8), nq - g * 8));
      uint64_t value = 0;
      const uint64_t pr =
          stochastic ? rand_pack(seed, (static_cast<uint64_t>(s) << 44) |
                                           static_cast<uint64_t>(g))
                     : 0;
      for (int j = 0; j < m; j++) {
        const int64_t eidx = g * 8 + j;
        const int64_t bk =
            small ? static_cast<int64_t>(static_cast<uint32_t>(eidx) /
                                         static_cast<uint32_t>(d.bucket))
                  : eidx / d.bucket;
        const float unitf = raw2f<T>(meta[2 * bk]);
        if (unitf < kEps) continue;
        const float minf = raw2f<T>(meta[2 * bk + 1]);
        const float rinv = 1.0f / unitf;
        const float rnd = stochastic ? rand_lane(pr, j) : 0.5f;
        const float dd = (raw2f<T>(in[eidx]) - minf) * rinv + rnd;
        const uint32_t level = static_cast<uint32_t>(fminf(floorf(dd), divisor));
        value |= static_cast<uint64_t>(level & ((1u << BITS) - 1)) << (j * BITS);
      }
      const int64_t gb = g * BITS;
      const int nbytes =
          static_cast<int>(min(static_cast<int64_t>(BITS), num_char - gb));
      store_bytes(packed + gb, value, nbytes);
    }
  }
}

// Load a (unit, min) meta pair with ONE aligned load (pairs are naturally
// 2*sizeof(R)-aligned: the slice base is 8-byte aligned).
template <typename R>
__device__ __forceinline__ void load_meta_pair(const R* meta, int64_t bk,
                                               uint32_t& unit_raw,
                                               uint32_t& min_raw) {
  if constexpr (sizeof(R) == 4) {
    const uint64_t p = *reinterpret_cast<const uint64_t*>(meta + 2 * bk);
    unit_raw = static_cast<uint32_t>(p);
    min_raw = static_cast<uint32_t>(p >> 32);
  } else {
    const uint32_t p = *reinterpret_cast<const uint32_t*>(meta + 2 * bk);
    unit_raw = p & 0xFFFF;
    min_raw = p >> 16;
  }
}

// Branch-free packed loads for the hot BITS values.  The compressed stream
// base is always 8-byte aligned (align8 slice sizes + 2*sizeof(R)*nb meta),
// so the typed loads below are aligned by construction.
// half: 4 elements (fp32 path; h selects the half-pack).
template <int BITS>
__device__ __forceinline__ uint64_t load_pack_half(
    const uint8_t* __restrict__ src, int64_t g, int h) {
  if constexpr (BITS == 1) {
    return static_cast<uint64_t>(src[g]) >> (h * 4);
  } else if constexpr (BITS == 2) {
    return src[g * 2 + h];
  } else if constexpr (BITS == 4) {
    return *reinterpret_cast<const uint16_t*>(src + g * 4 + h * 2);
  } else if constexpr (BITS == 8) {
    return *reinterpret_cast<const uint32_t*>(src + g * 8 + h * 4);
  } else if constexpr ((BITS & 1) == 0) {
    return load_bytes(src + g * BITS + h * (BITS / 2), BITS / 2);
  } else {
    return load_bytes(src + g * BITS, BITS) >> (h * 4 * BITS);
  }
}

// full: 8 elements (16-bit dtype path).
template <int BITS>
__device__ __forceinline__ uint64_t load_pack_full(
    const uint8_t* __restrict__ src, int64_t g) {
  if constexpr (BITS == 1) {
    return src[g];
  } else if constexpr (BITS == 2) {
    return *reinterpret_cast<const uint16_t*>(src + g * 2);
  } else if constexpr (BITS == 4) {
    return *reinterpret_cast<const uint32_t*>(src + g * 4);
  } else if constexpr (BITS == 8) {
    return *reinterpret_cast<const uint64_t*>(src + g * 8);
  } else {
    return load_bytes(src + g * BITS, BITS);
  }
}

using v4u = uint32_t __attribute__((ext_vector_type(4)));

// Compile-time-aligned 16-byte load/store (the runtime-bool forms compile
// into per-dword flat accesses inside the hot loop).
template <bool AL16>
__device__ __forceinline__ void load16B(const void* p, uint32_t (&r)[4]) {
  if constexpr (AL16) {
    const v4u a = *reinterpret_cast<const v4u*>(
        __builtin_assume_aligned(p, 16));
    r[0] = a.x; r[1] = a.y; r[2] = a.z; r[3] = a.w;
  } else {
    const uint32_t* q = reinterpret_cast<const uint32_t*>(p);
#pragma unroll
    for (int j = 0; j < 4; j++) r[j] = q[j];
  }
}

template <bool AL16>
__device__ __forceinline__ void store16B(void* p, const uint32_t (&r)[4]) {
  if constexpr (AL16) {
    v4u a;
    a.x = r[0]; a.y = r[1]; a.z = r[2]; a.w = r[3];
    *reinterpret_cast<v4u*>(__builtin_assume_aligned(p, 16)) = a;
  } else {
    uint32_t* q = reinterpret_cast<uint32_t*>(p);
#pragma unroll
    for (int j = 0; j < 4; j++) q[j] = r[j];
  }
}

// Incremental bucket-index tracker: replaces the per-iteration integer
// division (a ~40-instruction software sequence that dominated the round-1
// hot loop) with one uniform quotient add + carry per iteration.  Valid
// while indexes fit in u32 (`small` slices) and the per-iteration step is
// uniform.
struct BkTrack {
  uint32_t bk, rem, q, r, div;
  __device__ __forceinline__ void init(uint32_t x, uint32_t step,
                                       uint32_t d) {
    div = d;
    bk = x / d;
    rem = x % d;
    q = step / d;
    r = step % d;
  }
  __device__ __forceinline__ void advance() {
    bk += q;
    rem += r;
    if (rem >= div) {
      rem -= div;
      bk += 1;
    }
  }
};

// fp32 fast path: bucket % 8 == 0, u32-indexable slice.  4 elems/lane, two
// independent chains, no division, no per-access branches.  (U=4 in the
// lean kernel: VGPR 56->106 for a wash — the fifth occupancy-vs-ILP
// datapoint on this kernel family, all negative.)
template <int BITS, bool AL16, bool HAVE>
__device__ void deq_f32_fast(const DequantDesc& d,
                             const uint8_t* __restrict__ in0,
                             int64_t meta_bytes, int64_t full_subs,
                             uint32_t B4, int64_t t0, int64_t stride) {
  constexpr int U = 2;
  BkTrack bk[U];
#pragma unroll
  for (int u = 0; u < U; u++)
    bk[u].init(static_cast<uint32_t>(t0 + u * stride),
               static_cast<uint32_t>(U * stride), B4);
  int64_t w = t0;
  for (; w + (U - 1) * stride < full_subs; w += U * stride) {
    uint32_t v[U][4];
    float* outp[U];
#pragma unroll
    for (int u = 0; u < U; u++) {
      outp[u] = reinterpret_cast<float*>(d.out) + (w + u * stride) * 4;
      if constexpr (HAVE) load16B<AL16>(outp[u], v[u]);
    }
    for (int sidx = 0; sidx < d.nsrc; sidx++) {
      const uint8_t* __restrict__ src = in0 + sidx * d.src_stride;
      const float* __restrict__ meta =
          reinterpret_cast<const float*>(src - meta_bytes);
      uint64_t value[U];
      uint32_t ur[U], mr[U];
#pragma unroll
      for (int u = 0; u < U; u++) {
        const int64_t ws = w + u * stride;
        value[u] = load_pack_half<BITS>(src, ws >> 1,
                                        static_cast<int>(ws & 1));
        load_meta_pair<uint32_t>(reinterpret_cast<const uint32_t*>(meta),
                                 bk[u].bk, ur[u], mr[u]);
      }
#pragma unroll
      for (int u = 0; u < U; u++) {
        const float unitf = bitcast<float>(ur[u]);
        const float minf = bitcast<float>(mr[u]);
#pragma unroll
        for (int j = 0; j < 4; j++) {
          const uint32_t lvl = static_cast<uint32_t>(
              (value[u] >> (j * BITS)) & ((1u << BITS) - 1));
          const float dec = minf + unitf * static_cast<float>(lvl);
          if (!HAVE && sidx == 0) {
            v[u][j] = bitcast<uint32_t>(dec);
          } else {
            v[u][j] = bitcast<uint32_t>(bitcast<float>(v[u][j]) + dec);
          }
        }
      }
    }
#pragma unroll
    for (int u = 0; u < U; u++) {
      store16B<AL16>(outp[u], v[u]);
      bk[u].advance();
    }
  }
  // at most one trailing sub per thread
  for (; w < full_subs; w += stride) {
    uint32_t v[4];
    float* outp = reinterpret_cast<float*>(d.out) + w * 4;
    if constexpr (HAVE) load16B<AL16>(outp, v);
    const uint32_t bkw = static_cast<uint32_t>(w) / B4;
    for (int sidx = 0; sidx < d.nsrc; sidx++) {
      const uint8_t* __restrict__ src = in0 + sidx * d.src_stride;
      const uint32_t* __restrict__ meta =
          reinterpret_cast<const uint32_t*>(src - meta_bytes);
      const uint64_t value =
          load_pack_half<BITS>(src, w >> 1, static_cast<int>(w & 1));
      uint32_t ur, mr;
      load_meta_pair<uint32_t>(meta, bkw, ur, mr);
      const float unitf = bitcast<float>(ur);
      const float minf = bitcast<float>(mr);
#pragma unroll
      for (int j = 0; j < 4; j++) {
        const uint32_t lvl = static_cast<uint32_t>((value >> (j * BITS)) &
                                                   ((1u << BITS) - 1));
        const float dec = minf + unitf * static_cast<float>(lvl);
        if (!HAVE && sidx == 0) {
          v[j] = bitcast<uint32_t>(dec);
        } else {
          v[j] = bitcast<uint32_t>(bitcast<float>(v[j]) + dec);
        }
      }
    }
    store16B<AL16>(outp, v);
  }
}

// 16-bit fast path: 8 elems/lane (one int4 store), incremental bucket index.
template <typename T, int BITS, bool AL16, bool HAVE>
__device__ void deq_16_fast(const DequantDesc& d,
                            const uint8_t* __restrict__ in0,
                            int64_t meta_bytes, int64_t full_groups,
                            uint32_t B8, int64_t t0, int64_t stride) {
  using R = typename RawOf<T>::type;
  BkTrack bk;
  bk.init(static_cast<uint32_t>(t0), static_cast<uint32_t>(stride), B8);
  for (int64_t g = t0; g < full_groups; g += stride) {
    uint32_t w4[4];
    T* outp = reinterpret_cast<T*>(d.out) + g * 8;
    uint32_t v[8];
    if constexpr (HAVE) {
      if constexpr (AL16) {
        load16B<true>(outp, w4);
#pragma unroll
        for (int j = 0; j < 8; j++)
          v[j] = (w4[j >> 1] >> ((j & 1) * 16)) & 0xFFFF;
      } else {
        // 2-byte-aligned output: element loads (u32 would be misaligned)
        const R* q = reinterpret_cast<const R*>(outp);
#pragma unroll
        for (int j = 0; j < 8; j++) v[j] = q[j];
      }
    }
    for (int sidx = 0; sidx < d.nsrc; sidx++) {
      const uint8_t* __restrict__ src = in0 + sidx * d.src_stride;
      const R* __restrict__ meta = reinterpret_cast<const R*>(src - meta_bytes);
      const uint64_t value = load_pack_full<BITS>(src, g);
      uint32_t ur, mr;
      load_meta_pair<R>(meta, bk.bk, ur, mr);
      const float unitf = raw2f<T>(ur);
      const float minf = raw2f<T>(mr);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const uint32_t lvl = static_cast<uint32_t>((value >> (j * BITS)) &
                                                   ((1u << BITS) - 1));
        const uint32_t prod = f2raw<T>(unitf * static_cast<float>(lvl));
        const uint32_t dec = f2raw<T>(minf + raw2f<T>(prod));
        if (!HAVE && sidx == 0) {
          v[j] = dec;
        } else {
          v[j] = f2raw<T>(raw2f<T>(v[j]) + raw2f<T>(dec));
        }
      }
    }
    if constexpr (AL16) {
#pragma unroll
      for (int j = 0; j < 4; j++) w4[j] = v[2 * j] | (v[2 * j + 1] << 16);
      store16B<true>(outp, w4);
    } else {
      R* q = reinterpret_cast<R*>(outp);
#pragma unroll
      for (int j = 0; j < 8; j++) q[j] = static_cast<R>(v[j]);
    }
    bk.advance();
  }
}

// Lean dequantize kernel: only the branch-free fast paths (bucket%8==0,
// u32-indexable slices) are instantiated — the fused k_dequant allocates
// 102 VGPRs (fp32) because the register allocator must also cover the
// generic/tail paths; ragged tails re-enter k_dequant with kFlagTailOnly.
template <typename T, int BITS>
__global__ __launch_bounds__(kThreads) void k_dequant_fast(
    const DequantDesc* __restrict__ descs, int nslices) {
  using R = typename RawOf<T>::type;
  const int64_t t0 =
      static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int si = 0; si < nslices; si++) {
    const DequantDesc d = descs[si];
    const bool skip = (d.flags & kFlagSkipIncomplete) != 0;
    const int64_t nq =
        skip ? (d.n / d.bucket) * static_cast<int64_t>(d.bucket) : d.n;
    const int64_t nb_slice =
        skip ? d.n / d.bucket : (d.n + d.bucket - 1) / d.bucket;
    const int64_t meta_bytes = 2 * sizeof(R) * nb_slice;
    const bool al16 = (reinterpret_cast<uintptr_t>(d.out) & 15) == 0;
    const uint8_t* const in0 = d.in + meta_bytes;
    const bool have = d.add != 0;
    if constexpr (sizeof(T) == 4) {
      const int64_t full_subs = nq >> 2;
      const uint32_t B4 = static_cast<uint32_t>(d.bucket >> 2);
      if (al16) {
        if (have)
          deq_f32_fast<BITS, true, true>(d, in0, meta_bytes, full_subs, B4,
                                         t0, stride);
        else
          deq_f32_fast<BITS, true, false>(d, in0, meta_bytes, full_subs, B4,
                                          t0, stride);
      } else {
        if (have)
          deq_f32_fast<BITS, false, true>(d, in0, meta_bytes, full_subs, B4,
                                          t0, stride);
        else
          deq_f32_fast<BITS, false, false>(d, in0, meta_bytes, full_subs, B4,
                                           t0, stride);
      }
    } else {
      const int64_t full_groups = nq >> 3;
      const uint32_t B8 = static_cast<uint32_t>(d.bucket >> 3);
      if (al16) {
        if (have)
          deq_16_fast<T, BITS, true, true>(d, in0, meta_bytes, full_groups,
                                           B8, t0, stride);
        else
          deq_16_fast<T, BITS, true, false>(d, in0, meta_bytes, full_groups,
                                            B8, t0, stride);
      } else {
        if (have)
          deq_16_fast<T, BITS, false, true>(d, in0, meta_bytes, full_groups,
                                            B8, t0, stride);
        else
          deq_16_fast<T, BITS, false, false>(d, in0, meta_bytes, full_groups,
                                             B8, t0, stride);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Dequantize(+multi-source accumulate): grid-stride threads over packs.
// Sums d.nsrc compressed streams in T precision in stream order (matching the
// CPU simulation's per-source sequential accumulate), optionally on top of
// the existing output values (d.add).
// ---------------------------------------------------------------------------
template <typename T, int BITS>
__global__ __launch_bounds__(kThreads) void k_dequant(
    const DequantDesc* __restrict__ descs, const int64_t* __restrict__ cum,
    int nslices, int64_t total_groups) {
  // Outer loop over slices with every per-slice quantity hoisted; the inner
  // grid-stride loop over FULL groups is branch-free (no tail/alignment
  // checks, no exec-mask divergence) -- the original per-group binary-search
  // form spent >80% of its instructions on repeated int64 address math.
  using R = typename RawOf<T>::type;
  (void)cum;
  (void)total_groups;
  const int64_t t0 =
      static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int si = 0; si < nslices; si++) {
    const DequantDesc d = descs[si];
    const bool skip = (d.flags & kFlagSkipIncomplete) != 0;
    const int64_t nq =
        skip ? (d.n / d.bucket) * static_cast<int64_t>(d.bucket) : d.n;
    const int64_t nb_slice =
        skip ? d.n / d.bucket : (d.n + d.bucket - 1) / d.bucket;
    const int64_t meta_bytes = 2 * sizeof(R) * nb_slice;
    const int64_t full_groups = nq >> 3;
    const bool oneb = (d.bucket & 7) == 0;
    const uint32_t B8 = oneb ? static_cast<uint32_t>(d.bucket >> 3) : 1u;
    const bool al16 = (reinterpret_cast<uintptr_t>(d.out) & 15) == 0;
    const bool small = nq < (int64_t(1) << 31);
    T* const out_base = reinterpret_cast<T*>(d.out);
    const uint8_t* const in0 = d.in + meta_bytes;  // packed base, source 0

    if constexpr (sizeof(T) == 4) {
      // fp32: 4 elements per thread -- ONE coalesced float4 per lane
      // (8-wide fp32 needs two 16B stores at 32B lane stride, which measured
      // 3x slower per element than the fp16 path's single int4)
      const int64_t full_subs = nq >> 2;
      const uint32_t B4 = oneb ? static_cast<uint32_t>(d.bucket >> 2) : 1u;
      // kFlagTailOnly: k_dequant_fast already decoded the full 4-elem subs
      if (d.flags & kFlagTailOnly) goto f32_tail;
      {
      // two grid-stride iterations in flight (U=4 measured 0.170 ms vs
      // 0.142 at U=2 for 64M/4-bit: the extra registers cost more occupancy
      // than the added chains buy in latency hiding)
      constexpr int U = 2;
      int64_t w = t0;
      for (; w + (U - 1) * stride < full_subs; w += U * stride) {
        int64_t ws2[U];
        uint32_t v[U][4];
        const bool have = d.add != 0;
        float* outp[U];
        uint32_t bk0[U];
#pragma unroll
        for (int u = 0; u < U; u++) {
          ws2[u] = w + u * stride;
          outp[u] = reinterpret_cast<float*>(d.out) + ws2[u] * 4;
          if (have) load4f(outp[u], al16, v[u]);
          bk0[u] = oneb
                       ? (small ? static_cast<uint32_t>(ws2[u]) / B4
                                : static_cast<uint32_t>(
                                      ws2[u] / static_cast<int64_t>(B4)))
                       : 0;
        }
        for (int sidx = 0; sidx < d.nsrc; sidx++) {
          const uint8_t* src = in0 + sidx * d.src_stride;
          const R* meta = reinterpret_cast<const R*>(src - meta_bytes);
          uint64_t value[U];
#pragma unroll
          for (int u = 0; u < U; u++) {
            const int64_t g = ws2[u] >> 1;
            const int h = static_cast<int>(ws2[u] & 1);
            // half-pack loads for even BITS (a full-pack load + register
            // shift, which reads each pack twice, measured 0.156 ms vs
            // 0.142 for 64M/4-bit: the doubled L1 traffic costs more than
            // the narrower load saves)
            if constexpr ((BITS & 1) == 0) {
              value[u] = load_bytes(src + g * BITS + h * (BITS / 2), BITS / 2);
            } else {
              value[u] = load_bytes(src + g * BITS, BITS) >> (h * 4 * BITS);
            }
          }
#pragma unroll
          for (int u = 0; u < U; u++) {
#pragma unroll
            for (int j = 0; j < 4; j++) {
              const int64_t bk = oneb ? bk0[u] : (ws2[u] * 4 + j) / d.bucket;
              const uint32_t lvl = static_cast<uint32_t>(
                  (value[u] >> (j * BITS)) & ((1u << BITS) - 1));
              const uint32_t prod =
                  f2raw<T>(raw2f<T>(meta[2 * bk]) * static_cast<float>(lvl));
              const uint32_t dec =
                  f2raw<T>(raw2f<T>(meta[2 * bk + 1]) + raw2f<T>(prod));
              if (!have && sidx == 0) {
                v[u][j] = dec;
              } else {
                v[u][j] = f2raw<T>(raw2f<T>(v[u][j]) + raw2f<T>(dec));
              }
            }
          }
        }
#pragma unroll
        for (int u = 0; u < U; u++) store4f(outp[u], al16, v[u]);
      }
      for (; w < full_subs; w += stride) {
        const int64_t g = w >> 1;
        const int h = static_cast<int>(w & 1);
        uint32_t v[4];
        bool have = d.add != 0;
        float* outp = reinterpret_cast<float*>(d.out) + w * 4;
        if (have) load4f(outp, al16, v);
        const uint32_t bk0 =
            oneb ? (small ? static_cast<uint32_t>(w) / B4
                          : static_cast<uint32_t>(
                                w / static_cast<int64_t>(B4)))
                 : 0;
        for (int sidx = 0; sidx < d.nsrc; sidx++) {
          const uint8_t* src = in0 + sidx * d.src_stride;
          const R* meta = reinterpret_cast<const R*>(src - meta_bytes);
          uint64_t value;
          if constexpr ((BITS & 1) == 0) {
            value = load_bytes(src + g * BITS + h * (BITS / 2), BITS / 2);
          } else {
            value = load_bytes(src + g * BITS, BITS) >> (h * 4 * BITS);
          }
          uint32_t u0 = 0, m0 = 0;
          if (oneb) load_meta_pair<R>(meta, bk0, u0, m0);
#pragma unroll
          for (int j = 0; j < 4; j++) {
            uint32_t ur = u0, mr = m0;
            if (!oneb) {
              const int64_t bk = (w * 4 + j) / d.bucket;
              load_meta_pair<R>(meta, bk, ur, mr);
            }
            const uint32_t lvl = static_cast<uint32_t>(
                (value >> (j * BITS)) & ((1u << BITS) - 1));
            const uint32_t prod =
                f2raw<T>(raw2f<T>(ur) * static_cast<float>(lvl));
            const uint32_t dec = f2raw<T>(raw2f<T>(mr) + raw2f<T>(prod));
            if (!have && sidx == 0) {
              v[j] = dec;
            } else {
              v[j] = f2raw<T>(raw2f<T>(v[j]) + raw2f<T>(dec));
            }
          }
        }
        store4f(outp, al16, v);
      }
      }  // generic fp32 path
    f32_tail:
      // scalar tail: elements [full_subs*4, nq)
      if ((nq & 3) && t0 == 0) {
        R* outr = reinterpret_cast<R*>(d.out);
        for (int64_t eidx = full_subs * 4; eidx < nq; eidx++) {
          const int64_t g = eidx >> 3;
          const int j = static_cast<int>(eidx & 7);
          const int64_t num_char = (nq * BITS + 7) >> 3;
          const int nbytes = static_cast<int>(
              min(static_cast<int64_t>(BITS), num_char - g * BITS));
          const int64_t bk = eidx / d.bucket;
          bool first = true;
          for (int sidx = 0; sidx < d.nsrc; sidx++) {
            const uint8_t* src = in0 + sidx * d.src_stride;
            const R* meta = reinterpret_cast<const R*>(src - meta_bytes);
            const uint64_t value = load_bytes(src + g * BITS, nbytes);
            const uint32_t lvl = static_cast<uint32_t>(
                (value >> (j * BITS)) & ((1u << BITS) - 1));
            const uint32_t prod = f2raw<T>(raw2f<T>(meta[2 * bk]) *
                                           static_cast<float>(lvl));
            const uint32_t dec =
                f2raw<T>(raw2f<T>(meta[2 * bk + 1]) + raw2f<T>(prod));
            if (first && !d.add) {
              outr[eidx] = static_cast<R>(dec);
            } else {
              outr[eidx] = static_cast<R>(
                  f2raw<T>(raw2f<T>(outr[eidx]) + raw2f<T>(dec)));
            }
            first = false;
          }
        }
      }
      continue;  // next slice (16-bit loop below is for 2-byte dtypes)
    }
    // kFlagTailOnly: k_dequant_fast already decoded the full 8-elem groups
    if (d.flags & kFlagTailOnly) goto tail16;
    for (int64_t g = t0; g < full_groups; g += stride) {
      uint32_t v[8];
      bool have = d.add != 0;
      T* outp = out_base + g * 8;
      if (have) load8<T>(outp, al16, v);
      const uint32_t bk0 =
          oneb ? (small ? static_cast<uint32_t>(g) / B8
                        : static_cast<uint32_t>(g / static_cast<int64_t>(B8)))
               : 0;
      for (int sidx = 0; sidx < d.nsrc; sidx++) {
        const uint8_t* src = in0 + sidx * d.src_stride;
        const R* meta = reinterpret_cast<const R*>(src - meta_bytes);
        const uint64_t value = load_bytes(src + g * BITS, BITS);
        uint32_t u0 = 0, m0 = 0;
        if (oneb) load_meta_pair<R>(meta, bk0, u0, m0);
#pragma unroll
        for (int j = 0; j < 8; j++) {
          uint32_t ur = u0, mr = m0;
          if (!oneb) {
            const int64_t bk = (g * 8 + j) / d.bucket;
            load_meta_pair<R>(meta, bk, ur, mr);
          }
          const uint32_t lvl = static_cast<uint32_t>((value >> (j * BITS)) &
                                                     ((1u << BITS) - 1));
          const uint32_t prod = f2raw<T>(raw2f<T>(ur) * static_cast<float>(lvl));
          const uint32_t dec = f2raw<T>(raw2f<T>(mr) + raw2f<T>(prod));
          if (!have && sidx == 0) {
            v[j] = dec;
          } else {
            v[j] = f2raw<T>(raw2f<T>(v[j]) + raw2f<T>(dec));
          }
        }
      }
      store8<T>(outp, al16, v);
    }

  tail16:
    // tail group (fewer than 8 elements): one thread, scalar
    const int mtail = static_cast<int>(nq - full_groups * 8);
    if (mtail > 0 && t0 == 0) {
      const int64_t g = full_groups;
      const int64_t num_char = (nq * BITS + 7) >> 3;
      const int nbytes = static_cast<int>(
          min(static_cast<int64_t>(BITS), num_char - g * BITS));
      R* outp = reinterpret_cast<R*>(d.out) + g * 8;
      for (int sidx = 0; sidx < d.nsrc; sidx++) {
        const uint8_t* src = in0 + sidx * d.src_stride;
        const R* meta = reinterpret_cast<const R*>(src - meta_bytes);
        const uint64_t value = load_bytes(src + g * BITS, nbytes);
        for (int j = 0; j < mtail; j++) {
          const int64_t bk = (g * 8 + j) / d.bucket;
          const uint32_t lvl = static_cast<uint32_t>((value >> (j * BITS)) &
                                                     ((1u << BITS) - 1));
          const float unitf = raw2f<T>(meta[2 * bk]);
          const float minf = raw2f<T>(meta[2 * bk + 1]);
          const uint32_t prod = f2raw<T>(unitf * static_cast<float>(lvl));
          const uint32_t dec = f2raw<T>(minf + raw2f<T>(prod));
          if (sidx == 0 && !d.add) {
            outp[j] = static_cast<R>(dec);
          } else {
            outp[j] = static_cast<R>(
                f2raw<T>(raw2f<T>(outp[j]) + raw2f<T>(dec)));
          }
        }
      }
    }
  }
}

// skip_incomplete residual handling: the trailing partial bucket travels as
// raw T values after the aligned packed region (reference
// MaxMinQuantizer::Compress/DecompressBuffer, compressor.cc:332-341,384-400).
template <typename T>
__global__ __launch_bounds__(kThreads) void k_residual_q(
    const QuantDesc* __restrict__ descs, int nslices, int bits) {
  using R = typename RawOf<T>::type;
  const int64_t t0 =
      static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int s = 0; s < nslices; s++) {
    const QuantDesc d = descs[s];
    if (!(d.flags & kFlagSkipIncomplete)) continue;
    const int64_t nb = d.n / d.bucket;
    const int64_t nq = nb * static_cast<int64_t>(d.bucket);
    const int64_t r = d.n - nq;
    if (r == 0) continue;
    const R* src = reinterpret_cast<const R*>(d.in) + nq;
    R* dst = reinterpret_cast<R*>(
        d.out + 2 * sizeof(R) * nb + ((nq * bits + 7) / 8 + 7) / 8 * 8);
    if (d.fb) {
      // error feedback: the raw residual tail carries x+fb exactly, so the
      // new residual is zero
      R* f = reinterpret_cast<R*>(d.fb) + nq;
      for (int64_t i = t0; i < r; i += stride) {
        dst[i] = static_cast<R>(
            f2raw<T>(raw2f<T>(src[i]) + raw2f<T>(f[i])));
        f[i] = 0;
      }
    } else {
      for (int64_t i = t0; i < r; i += stride) dst[i] = src[i];
    }
  }
}

template <typename T>
__global__ __launch_bounds__(kThreads) void k_residual_d(
    const DequantDesc* __restrict__ descs, int nslices, int bits) {
  using R = typename RawOf<T>::type;
  const int64_t t0 =
      static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int s = 0; s < nslices; s++) {
    const DequantDesc d = descs[s];
    if (!(d.flags & kFlagSkipIncomplete)) continue;
    const int64_t nb = d.n / d.bucket;
    const int64_t nq = nb * static_cast<int64_t>(d.bucket);
    const int64_t r = d.n - nq;
    if (r == 0) continue;
    const int64_t comp = 2 * sizeof(R) * nb + ((nq * bits + 7) / 8 + 7) / 8 * 8;
    R* outp = reinterpret_cast<R*>(d.out) + nq;
    for (int64_t i = t0; i < r; i += stride) {
      uint32_t v;
      bool have = d.add != 0;
      if (have) v = outp[i];
      for (int src = 0; src < d.nsrc; src++) {
        const R* rp = reinterpret_cast<const R*>(d.in + src * d.src_stride +
                                                 comp);
        const uint32_t dec = rp[i];
        if (!have && src == 0) {
          v = dec;
          have = true;
        } else {
          v = f2raw<T>(raw2f<T>(v) + raw2f<T>(dec));
        }
      }
      outp[i] = static_cast<R>(v);
    }
  }
}

template <typename T>
__global__ __launch_bounds__(kThreads) void k_add(const T* __restrict__ x,
                                                  T* __restrict__ y,
                                                  int64_t n) {
  const int64_t t0 =
      static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  using R = typename RawOf<T>::type;
  const R* xr = reinterpret_cast<const R*>(x);
  R* yr = reinterpret_cast<R*>(y);
  for (int64_t i = t0; i < n; i += stride) {
    yr[i] = static_cast<R>(f2raw<T>(raw2f<T>(xr[i]) + raw2f<T>(yr[i])));
  }
}

inline int max_blocks() {
  // CGX_MAX_BLOCKS: experimentation knob for grid-size sweeps (default
  // kMaxBlocks = 4096, i.e. 2 resident workgroup generations on 256 CUs)
  static const int v = [] {
    const char* e = std::getenv("CGX_MAX_BLOCKS");
    const int x = e && *e ? std::atoi(e) : 0;
    return x > 0 ? x : kMaxBlocks;
  }();
  return v;
}

inline int grid_for(int64_t work, int per_block) {
  const int64_t blocks = (work + per_block - 1) / per_block;
  const int64_t cap = max_blocks();
  return static_cast<int>(blocks < 1 ? 1 : (blocks > cap ? cap : blocks));
}

#define CGX_DISPATCH_BITS(BITS_VAL, ...)                        \
  switch (BITS_VAL) {                                           \
    case 1: { constexpr int BITS = 1; __VA_ARGS__; break; }     \
    case 2: { constexpr int BITS = 2; __VA_ARGS__; break; }     \
    case 3: { constexpr int BITS = 3; __VA_ARGS__; break; }     \
    case 4: { constexpr int BITS = 4; __VA_ARGS__; break; }     \
    case 5: { constexpr int BITS = 5; __VA_ARGS__; break; }     \
    case 6: { constexpr int BITS = 6; __VA_ARGS__; break; }     \
    case 7: { constexpr int BITS = 7; __VA_ARGS__; break; }     \
    case 8: { constexpr int BITS = 8; __VA_ARGS__; break; }     \
    default:                                                    \
      printf("cgx: invalid bits %d\n", BITS_VAL);               \
      abort();                                                  \
  }

#define CGX_DISPATCH_T(DT, ...)                                       \
  switch (DT) {                                                       \
    case DType::F32: { using T = float; __VA_ARGS__; break; }         \
    case DType::F16: { using T = __half; __VA_ARGS__; break; }        \
    case DType::BF16: { using T = __hip_bfloat16; __VA_ARGS__; break; } \
  }

}  // namespace

void launch_quantize_batch(const QuantDesc* descs, const int64_t* cum,
                           int nslices, int64_t total_buckets, DType dt,
                           int bits, uint64_t seed, bool stochastic,
                           hipStream_t stream, bool buckets_mult8,
                           bool any_residual) {
  if (nslices <= 0) return;
  const int grid = grid_for(std::max<int64_t>(total_buckets, 1),
                            kThreads / kWave);
  CGX_DISPATCH_T(dt, CGX_DISPATCH_BITS(bits, {
    if (total_buckets > 0) {
      if (buckets_mult8) {
        hipLaunchKernelGGL((k_quantize<T, BITS, true>), dim3(grid),
                           dim3(kThreads), 0, stream, descs, cum, nslices,
                           total_buckets, seed, (int)stochastic);
      } else {
        hipLaunchKernelGGL((k_quantize<T, BITS, false>), dim3(grid),
                           dim3(kThreads), 0, stream, descs, cum, nslices,
                           total_buckets, seed, (int)stochastic);
        hipLaunchKernelGGL((k_pack_generic<T, BITS>), dim3(grid),
                           dim3(kThreads), 0, stream, descs, nslices, seed,
                           (int)stochastic);
      }
    }
    if (any_residual) {
      hipLaunchKernelGGL((k_residual_q<T>), dim3(32), dim3(kThreads), 0,
                         stream, descs, nslices, bits);
    }
  }));
}

void launch_quantize_fast(const QuantDesc* descs, const int64_t* cum,
                          int nslices, int64_t total_buckets, DType dt,
                          int bits, uint64_t seed, bool stochastic,
                          hipStream_t stream, bool any_residual,
                          int max_gpl) {
  if (nslices <= 0) return;
  const int grid = grid_for(std::max<int64_t>(total_buckets, 1),
                            kThreads / kWave);
  CGX_DISPATCH_T(dt, CGX_DISPATCH_BITS(bits, {
    if (total_buckets > 0) {
      if (max_gpl <= 2) {
        hipLaunchKernelGGL((k_quantize_fast<T, BITS, 2>), dim3(grid),
                           dim3(kThreads), 0, stream, descs, cum, nslices,
                           total_buckets, seed, (int)stochastic);
      } else {
        hipLaunchKernelGGL((k_quantize_fast<T, BITS, 4>), dim3(grid),
                           dim3(kThreads), 0, stream, descs, cum, nslices,
                           total_buckets, seed, (int)stochastic);
      }
    }
    if (any_residual) {
      hipLaunchKernelGGL((k_residual_q<T>), dim3(32), dim3(kThreads), 0,
                         stream, descs, nslices, bits);
    }
  }));
}

void launch_quantize_sub(const QuantDesc* descs, const int64_t* cum,
                         int nslices, int64_t total_buckets, DType dt,
                         int bits, uint64_t seed, bool stochastic,
                         hipStream_t stream, bool any_residual) {
  if (nslices <= 0) return;
  const int grid = grid_for(std::max<int64_t>(total_buckets, 1),
                            kThreads / kWave);
  CGX_DISPATCH_T(dt, CGX_DISPATCH_BITS(bits, {
    if (total_buckets > 0) {
      hipLaunchKernelGGL((k_quantize_sub<T, BITS>), dim3(grid),
                         dim3(kThreads), 0, stream, descs, cum, nslices,
                         total_buckets, seed, (int)stochastic);
    }
    if (any_residual) {
      hipLaunchKernelGGL((k_residual_q<T>), dim3(32), dim3(kThreads), 0,
                         stream, descs, nslices, bits);
    }
  }));
}

void launch_dequantize_batch(const DequantDesc* descs, const int64_t* cum,
                             int nslices, int64_t total_groups, DType dt,
                             int bits, hipStream_t stream,
                             bool any_residual) {
  if (nslices <= 0) return;
  const int grid = grid_for(std::max<int64_t>(total_groups, 1), kThreads);
  CGX_DISPATCH_T(dt, CGX_DISPATCH_BITS(bits, {
    if (total_groups > 0) {
      hipLaunchKernelGGL((k_dequant<T, BITS>), dim3(grid), dim3(kThreads), 0,
                         stream, descs, cum, nslices, total_groups);
    }
    if (any_residual) {
      hipLaunchKernelGGL((k_residual_d<T>), dim3(32), dim3(kThreads), 0,
                         stream, descs, nslices, bits);
    }
  }));
}

void launch_dequantize_fast(const DequantDesc* descs, int nslices,
                            int64_t total_groups, DType dt, int bits,
                            hipStream_t stream, bool any_residual) {
  if (nslices <= 0) return;
  const int grid = grid_for(std::max<int64_t>(total_groups, 1), kThreads);
  CGX_DISPATCH_T(dt, CGX_DISPATCH_BITS(bits, {
    if (total_groups > 0) {
      hipLaunchKernelGGL((k_dequant_fast<T, BITS>), dim3(grid),
                         dim3(kThreads), 0, stream, descs, nslices);
    }
    if (any_residual) {
      hipLaunchKernelGGL((k_residual_d<T>), dim3(32), dim3(kThreads), 0,
                         stream, descs, nslices, bits);
    }
  }));
}

void launch_add(const void* x, void* y, int64_t n, DType dt,
                hipStream_t stream) {
  if (n <= 0) return;
  const int grid = grid_for(n, kThreads);
  CGX_DISPATCH_T(dt, {
    hipLaunchKernelGGL((k_add<T>), dim3(grid), dim3(kThreads), 0, stream,
                       reinterpret_cast<const T*>(x), reinterpret_cast<T*>(y),
                       n);
  });
}

}  // namespace cgx
