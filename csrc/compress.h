// Host-side API of the CDNA4 max-min quantization kernels.
//
// Wire format (see torch_cgx_amd/ops/golden.py, the single source of truth;
// parity with reference cuda_compression_operations.cu:68-96,219-285):
// per slice: [2*num_buckets values of T, interleaved (unit, min)]
//            [ceil(n*bits/8) packed bytes, groups of 8 values little-endian]
// total size align8-padded (buffer_size()).
#pragma once

#include <hip/hip_runtime.h>

#include <cstddef>
#include <cstdint>

namespace cgx {

enum class DType : int { F32 = 0, F16 = 1, BF16 = 2 };

inline int elem_size(DType d) { return d == DType::F32 ? 4 : 2; }

inline int64_t align8(int64_t x) { return (x + 7) / 8 * 8; }

// Compressed byte size of an n-element slice (golden.buffer_size parity).
// skip_incomplete: the trailing partial bucket is stored as raw T values
// after the aligned packed region instead of being quantized.
inline int64_t buffer_size(int64_t n, DType dt, int bits, int bucket_size,
                           bool skip_incomplete = false) {
  if (n == 0) return 0;
  int64_t nb = (n + bucket_size - 1) / bucket_size;
  int64_t residuals = 0;
  if (skip_incomplete) {
    nb = n / bucket_size;
    residuals = n % bucket_size;
    n = nb * bucket_size;
  }
  return 2 * nb * elem_size(dt) + align8((n * bits + 7) / 8) +
         residuals * elem_size(dt);
}

constexpr int32_t kFlagSkipIncomplete = 1;
// The slice's full buckets were already quantized by the lean fast kernel;
// this (generic-kernel) desc covers only the trailing partial bucket.
constexpr int32_t kFlagTailOnly = 2;

// One quantize work item: compress `n` elems at `in` into `out` bytes.
// fb (optional): error-feedback residual of the same length/dtype as `in`;
// the encoded value is in[i]+fb[i] (rounded to T) and fb is updated in place
// to the new residual.  Wire format is unchanged.  (The reference carried
// this plumbing but never enabled it, cuda_compression_operations.cu:713.)
struct QuantDesc {
  const void* in;
  uint8_t* out;
  void* fb;
  int64_t n;
  int32_t bucket;
  int32_t flags;  // kFlagSkipIncomplete
};

// One dequantize work item: decode `n` elems from `nsrc` compressed streams
// (at in + s*src_stride for s < nsrc), summing them in T precision in stream
// order, then write into `out` (add=1: accumulate into existing values).
struct DequantDesc {
  const uint8_t* in;
  void* out;
  int64_t n;
  int64_t src_stride;
  int32_t bucket;
  int32_t nsrc;
  int32_t add;
  int32_t flags;  // kFlagSkipIncomplete
};

// Batched launchers. descs/cum live in DEVICE memory. `cum` is the exclusive
// prefix over per-slice work-unit counts with a trailing total:
//   quantize: work unit = bucket,     cum[i+1]-cum[i] = ceil(n_i/bucket_i)
//   dequant:  work unit = 8-elem group, cum[i+1]-cum[i] = ceil(n_i/8)
// All slices in one launch share `bits` (the engine groups by bits).
// buckets_mult8: caller guarantees every slice's bucket_size % 8 == 0 (the
// fused single-read path); otherwise a meta pass + generic pack pass run.
// any_residual: at least one slice carries kFlagSkipIncomplete with a
// non-empty raw residual tail (adds one small copy/accumulate pass).
void launch_quantize_batch(const QuantDesc* descs, const int64_t* cum,
                           int nslices, int64_t total_buckets, DType dt,
                           int bits, uint64_t seed, bool stochastic,
                           hipStream_t stream, bool buckets_mult8,
                           bool any_residual = false);

// Lean fast-path kernel: every slice has bucket % 8 == 0, a 16B-aligned
// input base, no error feedback; cum counts FULL buckets per slice (the
// partial tail travels through launch_quantize_batch with kFlagTailOnly).
// Separate launch so its register footprint (and thus occupancy) is not
// inflated by the generic/EF code paths.
// max_gpl: max over slices of ceil((bucket/8)/64) — <=2 selects the
// half-stash kernel variant (higher occupancy for buckets <= 1024).
void launch_quantize_fast(const QuantDesc* descs, const int64_t* cum,
                          int nslices, int64_t total_buckets, DType dt,
                          int bits, uint64_t seed, bool stochastic,
                          hipStream_t stream, bool any_residual = false,
                          int max_gpl = 4);

// Small-bucket variant of the fast kernel: every slice has bucket/8 a
// power of two < 64; the wave packs 64/(bucket/8) buckets per iteration
// (segmented DPP reductions).  Separate kernel for register isolation.
void launch_quantize_sub(const QuantDesc* descs, const int64_t* cum,
                         int nslices, int64_t total_buckets, DType dt,
                         int bits, uint64_t seed, bool stochastic,
                         hipStream_t stream, bool any_residual = false);

void launch_dequantize_batch(const DequantDesc* descs, const int64_t* cum,
                             int nslices, int64_t total_groups, DType dt,
                             int bits, hipStream_t stream,
                             bool any_residual = false);

// Lean dequant kernel: every slice has bucket % 8 == 0 and fits u32
// indexing (n < 2^31; per-thread indices are 4/8-element units); ragged tails (n % 4 fp32 / n % 8 16-bit) re-enter
// launch_dequantize_batch with kFlagTailOnly.
void launch_dequantize_fast(const DequantDesc* descs, int nslices,
                            int64_t total_groups, DType dt, int bits,
                            hipStream_t stream, bool any_residual = false);

// y[i] += x[i] elementwise (T precision), n elements.
void launch_add(const void* x, void* y, int64_t n, DType dt,
                hipStream_t stream);

}  // namespace cgx
