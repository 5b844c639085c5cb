// Implementation of the compression engine (see engine.h).
#include "engine.h"

#include <ATen/hip/impl/HIPCachingAllocatorMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <c10/util/Exception.h>

#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <map>

namespace cgx {

// ---------------------------------------------------------------------------
// Registry
// ---------------------------------------------------------------------------
Registry& Registry::get() {
  static Registry r;
  return r;
}

void Registry::register_layer(int bucket_idx, int layer_idx, int64_t numel,
                              int bits, int bucket_size) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = by_idx_.find(bucket_idx);
  if (it == by_idx_.end()) {
    by_idx_[bucket_idx] = order_.size();
    order_.emplace_back();
    order_.back().idx = bucket_idx;
    it = by_idx_.find(bucket_idx);
  }
  BucketInfo& b = order_[it->second];
  if (layer_idx == 0) {  // re-registration resets the bucket
    b.numels.clear();
    b.cfgs.clear();
    b.total = 0;
  }
  b.numels.push_back(numel);
  LayerConfig c;
  c.bits = bits;
  c.bucket_size = bucket_size > 0 ? bucket_size : 512;
  b.cfgs.push_back(c);
  b.total += numel;
  cursor_ = 0;
}

void Registry::set_bits(int bucket_idx, int layer_idx, int bits) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = by_idx_.find(bucket_idx);
  if (it == by_idx_.end()) return;
  auto& b = order_[it->second];
  if (layer_idx >= 0 && (size_t)layer_idx < b.cfgs.size())
    b.cfgs[layer_idx].bits = bits;
}

void Registry::set_bucket_size(int bucket_idx, int layer_idx, int bucket_size) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = by_idx_.find(bucket_idx);
  if (it == by_idx_.end()) return;
  auto& b = order_[it->second];
  if (layer_idx >= 0 && (size_t)layer_idx < b.cfgs.size())
    b.cfgs[layer_idx].bucket_size = bucket_size;
}

void Registry::clear() {
  std::lock_guard<std::mutex> g(mu_);
  order_.clear();
  by_idx_.clear();
  bound_.clear();
  cursor_ = 0;
}

bool Registry::empty() {
  std::lock_guard<std::mutex> g(mu_);
  return order_.empty();
}

std::vector<Registry::BucketInfo> Registry::snapshot() {
  std::lock_guard<std::mutex> g(mu_);
  return order_;
}

bool Registry::next(int64_t numel, const void* key, BucketInfo* out) {
  std::lock_guard<std::mutex> g(mu_);
  if (order_.empty()) return false;
  // pointer identity first: a previously seen flat-bucket storage resolves
  // to the bucket it matched before, regardless of cursor position
  if (key) {
    auto it = bound_.find(key);
    if (it != bound_.end()) {
      const size_t i = it->second;
      if (i < order_.size() && order_[i].total == numel) {
        *out = order_[i];
        cursor_ = i + 1;
        return true;
      }
      bound_.erase(it);  // bucket rebuilt or storage reused: unlearn
    }
  }
  const size_t n = order_.size();
  const size_t start = cursor_ % n;
  // prefer the cursor position; otherwise any bucket whose total matches
  for (size_t k = 0; k < n; k++) {
    const size_t i = (start + k) % n;
    if (order_[i].total == numel) {
      *out = order_[i];
      cursor_ = i + 1;
      if (key) bound_[key] = i;
      return true;
    }
  }
  return false;
}

// ---------------------------------------------------------------------------
// EngineConfig
// ---------------------------------------------------------------------------
static int64_t env_int(const char* name, int64_t dflt) {
  const char* v = std::getenv(name);
  if (!v || !*v) return dflt;
  return std::atoll(v);
}

EngineConfig EngineConfig::from_env() {
  EngineConfig c;
  c.fusion_bytes = env_int("CGX_FUSION_BUFFER_SIZE_MB", 64) << 20;
  if (c.fusion_bytes < 2048) c.fusion_bytes = 2048;
  c.min_elems = env_int("CGX_COMPRESSION_MINIMAL_SIZE", 16);
  c.default_bits = (int)env_int("CGX_COMPRESSION_QUANTIZATION_BITS", 32);
  c.default_bucket = (int)env_int("CGX_COMPRESSION_BUCKET_SIZE", 512);
  c.stochastic = env_int("CGX_STOCHASTIC_ROUNDING", 1) != 0;
  c.skip_incomplete =
      env_int("CGX_COMPRESSION_SKIP_INCOMPLETE_BUCKETS", 0) != 0;
  c.dummy = env_int("CGX_DEBUG_DUMMY_COMPRESSION", 0) != 0;
  c.intra_compress = env_int("CGX_INTRA_COMPRESS", 1) != 0;
  c.error_feedback = env_int("CGX_ERROR_FEEDBACK", 0) != 0;
  c.debug_a2a = env_int("CGX_DEBUG_ALL_TO_ALL_REDUCTION", 0) != 0;
  const char* fr = std::getenv("CGX_COMPRESSION_FAKE_RATIO");
  if (fr && *fr) c.fake_ratio = std::atof(fr);
  if (!(c.fake_ratio > 0.0 && c.fake_ratio <= 1.0)) c.fake_ratio = 1.0;
  // Reduction algorithm selection, independently for the intra-node and
  // cross-node engines (reference mpi_allreduce_operations.cc:90-115:
  // intra default SRA, cross default Ring).  The legacy CGX_REDUCTION_TYPE
  // alias drives whichever specific variable is unset.
  auto is_ring = [](const char* v, bool dflt) {
    if (!v || !*v) return dflt;
    return std::strcmp(v, "Ring") == 0 || std::strcmp(v, "ring") == 0 ||
           std::strcmp(v, "RING") == 0;
  };
  const char* legacy = std::getenv("CGX_REDUCTION_TYPE");
  const char* inner = std::getenv("CGX_INNER_REDUCTION_TYPE");
  const char* cross = std::getenv("CGX_CROSS_REDUCTION_TYPE");
  c.ring = is_ring(inner && *inner ? inner : legacy, false);
  c.cross_ring = is_ring(cross && *cross ? cross : legacy, true);
  return c;
}

// ---------------------------------------------------------------------------
// DescRing
// ---------------------------------------------------------------------------
DescRing::~DescRing() {
  for (auto& s : slots_) {
    if (s.host) (void)hipHostFree(s.host);
    if (s.dev) (void)hipFree(s.dev);
    if (s.ev) (void)hipEventDestroy(s.ev);
  }
}

void* DescRing::acquire(size_t bytes) {
  Slot& s = slots_[cur_];
  if (!s.ev) CGX_HIP_CHECK(hipEventCreateWithFlags(&s.ev, hipEventDisableTiming));
  if (s.ev_recorded) CGX_HIP_CHECK(hipEventSynchronize(s.ev));
  if (s.cap < bytes) {
    if (s.host) CGX_HIP_CHECK(hipHostFree(s.host));
    if (s.dev) CGX_HIP_CHECK(hipFree(s.dev));
    size_t cap = 4096;
    while (cap < bytes) cap *= 2;
    CGX_HIP_CHECK(hipHostMalloc(&s.host, cap));
    CGX_HIP_CHECK(hipMalloc(&s.dev, cap));
    s.cap = cap;
  }
  return s.host;
}

void* DescRing::commit(size_t bytes, hipStream_t stream) {
  Slot& s = slots_[cur_];
  CGX_HIP_CHECK(
      hipMemcpyAsync(s.dev, s.host, bytes, hipMemcpyHostToDevice, stream));
  CGX_HIP_CHECK(hipEventRecord(s.ev, stream));
  s.ev_recorded = true;
  void* dev = s.dev;
  cur_ = (cur_ + 1) % kSlots;
  return dev;
}

// ---------------------------------------------------------------------------
// dtype maps
// ---------------------------------------------------------------------------
DType dtype_of(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return DType::F32;
    case at::kHalf: return DType::F16;
    case at::kBFloat16: return DType::BF16;
    default:
      TORCH_CHECK(false, "cgx: unsupported compression dtype ", t.scalar_type());
  }
}

ncclDataType_t nccl_dtype(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return ncclFloat32;
    case at::kDouble: return ncclFloat64;
    case at::kHalf: return ncclFloat16;
    case at::kBFloat16: return ncclBfloat16;
    case at::kInt: return ncclInt32;
    case at::kLong: return ncclInt64;
    case at::kChar: return ncclInt8;
    case at::kByte: return ncclUint8;
    case at::kBool: return ncclUint8;
    default:
      TORCH_CHECK(false, "cgx: unsupported dtype for RCCL ", t.scalar_type());
  }
}

// ---------------------------------------------------------------------------
// Engine
// ---------------------------------------------------------------------------
Engine::Engine(int rank, int size, bool is_cross)
    : rank_(rank), size_(size), is_cross_(is_cross) {
  seed_ = 0x9E3779B97F4A7C15ull ^ (0xD1B54A32D192ED03ull * (uint64_t)(rank + 1));
  // constructed under the backend's device guard (lazyInit)
  int least = 0, greatest = 0;
  CGX_HIP_CHECK(hipDeviceGetStreamPriorityRange(&least, &greatest));
  CGX_HIP_CHECK(hipStreamCreateWithPriority(&comm_stream_,
                                            hipStreamNonBlocking, greatest));
  CGX_HIP_CHECK(hipStreamCreateWithPriority(&deq_stream_,
                                            hipStreamNonBlocking, greatest));
  for (auto& e : evs_)
    CGX_HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
  for (auto& s : slots_)
    CGX_HIP_CHECK(hipEventCreateWithFlags(&s.done_ev, hipEventDisableTiming));
}

Engine::~Engine() {
  timer_drain();
  for (auto& s : slots_)
    if (s.done_ev) (void)hipEventDestroy(s.done_ev);
  for (auto& e : evs_)
    if (e) (void)hipEventDestroy(e);
  if (comm_stream_) (void)hipStreamDestroy(comm_stream_);
  if (deq_stream_) (void)hipStreamDestroy(deq_stream_);
  if (timer_.inited) {
    for (int r = 0; r < PhaseTimer::kRing; r++)
      for (int e = 0; e < PhaseTimer::kEv; e++)
        if (timer_.ev[r][e]) (void)hipEventDestroy(timer_.ev[r][e]);
  }
}

void Engine::timer_drain() {
  // collect any in-flight timing slots and print the final averages (short
  // runs would otherwise never reach the 50-chunk periodic print)
  PhaseTimer& t = timer_;
  if (!t.inited) return;
  for (int r = 0; r < PhaseTimer::kRing; r++) {
    if (!t.pending[r]) continue;
    if (hipEventSynchronize(t.ev[r][PhaseTimer::kEv - 1]) != hipSuccess)
      continue;
    for (int p = 0; p + 1 < PhaseTimer::kEv; p++) {
      float ms = 0.f;
      if (hipEventElapsedTime(&ms, t.ev[r][p], t.ev[r][p + 1]) == hipSuccess)
        t.sum_ms[p] += ms;
    }
    t.count++;
    t.pending[r] = false;
  }
  if (t.count > 0) {
    fprintf(stderr,
            "[cgx timings rank %d, %lld chunks total] quantize %.3f ms | "
            "comm1 %.3f | decode+requant %.3f | comm2 %.3f | decode2 %.3f\n",
            rank_, (long long)t.count, t.sum_ms[0] / t.count,
            t.sum_ms[1] / t.count, t.sum_ms[2] / t.count,
            t.sum_ms[3] / t.count, t.sum_ms[4] / t.count);
  }
}

hipEvent_t Engine::next_ev() {
  hipEvent_t e = evs_[ev_cur_];
  ev_cur_ = (ev_cur_ + 1) % kEvents;
  return e;
}

void Engine::chain(hipStream_t from, hipStream_t to) {
  hipEvent_t e = next_ev();
  CGX_HIP_CHECK(hipEventRecord(e, from));
  CGX_HIP_CHECK(hipStreamWaitEvent(to, e, 0));
}

void Engine::partition(int64_t num_elements, int ws,
                       const std::vector<int64_t>& lnumels, int esize,
                       std::vector<int64_t>* offsets,
                       std::vector<int64_t>* sizes) {
  // Mirror of torch_cgx_amd.parallel.partition.partition (tested against it).
  int64_t offset = 0;
  size_t li = 0;
  int64_t n_elem = lnumels.empty() ? 0 : std::min(lnumels[0], num_elements);
  const int64_t unit = (esize == 2) ? 8 : 4;
  int64_t remaining = num_elements;
  for (int r = 0; r < ws; r++) {
    const int64_t per = remaining / (ws - r);
    int64_t cur = 0;
    while (cur < per) {
      if (n_elem <= per - cur) {
        cur += n_elem;
        li++;
        if (li == lnumels.size()) break;
        n_elem = std::min(lnumels[li], num_elements);
      } else {
        const int64_t want = per - cur;
        const int64_t aligned =
            std::min(((want + unit - 1) / unit) * unit, n_elem);
        cur += aligned;
        n_elem -= aligned;
      }
    }
    remaining -= cur;
    sizes->push_back(cur);
    offsets->push_back(offset);
    offset += cur;
  }
}

uint8_t* Engine::staging(int64_t bytes, hipStream_t user) {
  if (bytes <= 0) bytes = 1;
  if (!staging_.defined() || staging_.numel() < bytes) {
    if (staging_.defined()) {
      // the old buffer may still be read by queued work on the user/comm/deq
      // streams; let the caching allocator defer reuse past those streams
      for (hipStream_t s : {user, comm_stream_, deq_stream_}) {
        if (!s) continue;
        c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::
            recordStreamMasqueradingAsCUDA(
                staging_.storage().data_ptr(),
                c10::hip::getStreamFromExternalMasqueradingAsCUDA(
                    s, staging_.device().index()));
      }
    }
    int64_t cap = 1 << 20;
    while (cap < bytes) cap *= 2;
    staging_ = at::empty({cap}, at::TensorOptions()
                                    .dtype(at::kByte)
                                    .device(at::kCUDA));
  }
  return staging_.data_ptr<uint8_t>();
}

uint8_t* Engine::slot_bytes(StagingSlot& slot, int64_t bytes) {
  if (bytes <= 0) bytes = 1;
  if (!slot.buf.defined() || slot.buf.numel() < bytes) {
    if (slot.buf.defined()) {
      // the old buffer may still be referenced by in-flight comm/deq work;
      // let the caching allocator defer reuse until those streams pass
      auto rec = [&](hipStream_t s) {
        c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::
            recordStreamMasqueradingAsCUDA(
                slot.buf.storage().data_ptr(),
                c10::hip::getStreamFromExternalMasqueradingAsCUDA(
                    s, slot.buf.device().index()));
      };
      rec(comm_stream_);
      rec(deq_stream_);
    }
    int64_t cap = 1 << 20;
    while (cap < bytes) cap *= 2;
    slot.buf = at::empty({cap}, at::TensorOptions()
                                    .dtype(at::kByte)
                                    .device(at::kCUDA));
  }
  return slot.buf.data_ptr<uint8_t>();
}

at::Tensor& Engine::feedback_buf(int64_t key, int phase, int64_t numel,
                                 at::ScalarType st) {
  // bounded: residuals are a convergence aid; resetting them once (on a cap
  // overflow that only a pathological caller can trigger) is benign,
  // unbounded growth is not
  constexpr size_t kMaxEntries = 4096;
  if (fb_bufs_.size() >= kMaxEntries) {
    TORCH_WARN_ONCE("cgx: error-feedback store exceeded ", kMaxEntries,
                    " entries; resetting residuals");
    fb_bufs_.clear();
  }
  auto k = std::make_pair(key, phase);
  auto it = fb_bufs_.find(k);
  if (it != fb_bufs_.end() &&
      (it->second.numel() < numel || it->second.scalar_type() != st)) {
    fb_bufs_.erase(it);  // chunk grew or dtype changed: restart residual
    it = fb_bufs_.end();
  }
  if (it == fb_bufs_.end()) {
    auto t = at::zeros({numel}, at::TensorOptions().dtype(st).device(
                                    at::kCUDA));
    it = fb_bufs_.emplace(k, std::move(t)).first;
  }
  return it->second;
}

void Engine::run_quantize(const std::vector<Slice>& slices, uint8_t* out_base,
                          DType dt, hipStream_t stream, bool stochastic,
                          char* fb_base, int64_t fb_rebase) {
  // Three launch kinds per bits value:
  //   0 = lean fast kernel (bucket%8==0, 16B-aligned input, no error
  //       feedback, >=1 full bucket) — low register footprint, cum counts
  //       FULL buckets; a partial tail re-enters kind 1 with kFlagTailOnly
  //   1 = generic fused kernel (EF, unaligned, tails)
  //   2 = meta + pack two-pass (bucket%8 != 0)
  struct Ent {
    const Slice* s;
    void* fb;
    int32_t flags;
  };
  std::map<std::pair<int, int>, std::vector<Ent>> groups;
  for (const auto& s : slices) {
    if (s.n <= 0) continue;
    void* fb =
        fb_base ? fb_base + (s.fb_off - fb_rebase) * elem_size(dt) : nullptr;
    TORCH_CHECK(!fb_base || s.fb_off >= fb_rebase,
                "cgx: error-feedback slice offset below buffer base");
    TORCH_CHECK(!fb || s.bucket % 8 == 0,
                "cgx: CGX_ERROR_FEEDBACK requires bucket_size % 8 == 0");
    const int32_t flags = s.skip_incomplete ? kFlagSkipIncomplete : 0;
    if ((s.bucket % 8) != 0) {
      groups[{s.bits, 2}].push_back(Ent{&s, fb, flags});
      continue;
    }
    const bool aligned = (reinterpret_cast<uintptr_t>(s.data) & 15) == 0;
    // bucket <= 2048: whole bucket fits the register stash (4 groups/lane);
    // small power-of-two buckets (bucket < 512) pack several buckets per
    // wave in their own kernel (kind 3, QuantSub)
    if (!fb && aligned && s.n >= s.bucket && s.bucket <= 2048) {
      const int ng = s.bucket >> 3;
      const bool small_pow2 = ng < 64 && (ng & (ng - 1)) == 0;
      groups[{s.bits, small_pow2 ? 3 : 0}].push_back(Ent{&s, nullptr, flags});
      if (!s.skip_incomplete && (s.n % s.bucket) != 0)
        groups[{s.bits, 1}].push_back(Ent{&s, nullptr,
                                          flags | kFlagTailOnly});
    } else {
      groups[{s.bits, 1}].push_back(Ent{&s, fb, flags});
    }
  }
  for (auto& [key, list] : groups) {
    const auto [bits, kind] = key;
    const int nsl = (int)list.size();
    const size_t desc_bytes = sizeof(QuantDesc) * nsl;
    const size_t cum_bytes = sizeof(int64_t) * (nsl + 1);
    char* host = (char*)ring_.acquire(desc_bytes + cum_bytes);
    auto* qd = reinterpret_cast<QuantDesc*>(host);
    auto* cum = reinterpret_cast<int64_t*>(host + desc_bytes);
    cum[0] = 0;
    bool any_residual = false;
    for (int i = 0; i < nsl; i++) {
      const Slice& s = *list[i].s;
      qd[i] = QuantDesc{s.data, out_base + s.comp_off, list[i].fb, s.n,
                        s.bucket, list[i].flags};
      int64_t nb;
      if (kind == 0 || kind == 3) {
        nb = s.n / s.bucket;  // full buckets only
        any_residual |= s.skip_incomplete && (s.n % s.bucket) != 0;
      } else if (list[i].flags & kFlagTailOnly) {
        nb = 1;
      } else if (s.skip_incomplete) {
        nb = s.n / s.bucket;
        any_residual |= (s.n % s.bucket) != 0;
      } else {
        nb = (s.n + s.bucket - 1) / s.bucket;
      }
      cum[i + 1] = cum[i] + nb;
    }
    char* dev = (char*)ring_.commit(desc_bytes + cum_bytes, stream);
    auto* ddesc = reinterpret_cast<QuantDesc*>(dev);
    auto* dcum = reinterpret_cast<int64_t*>(dev + desc_bytes);
    if (kind == 3) {
      launch_quantize_sub(ddesc, dcum, nsl, cum[nsl], dt, bits, seed_++,
                          stochastic, stream, any_residual);
    } else if (kind == 0) {
      int max_gpl = 1;
      for (const auto& e : list)
        max_gpl = std::max(max_gpl,
                           ((e.s->bucket >> 3) + 63) >> 6);
      launch_quantize_fast(ddesc, dcum, nsl, cum[nsl], dt, bits, seed_++,
                           stochastic, stream, any_residual, max_gpl);
    } else {
      launch_quantize_batch(ddesc, dcum, nsl, cum[nsl], dt, bits, seed_++,
                            stochastic, stream, /*buckets_mult8=*/kind == 1,
                            any_residual);
    }
  }
}

void Engine::run_dequant(const std::vector<Slice>& slices,
                         const uint8_t* in_base, int64_t src_stride, int nsrc,
                         bool add, DType dt, hipStream_t stream) {
  // kind 0: lean fast kernel (bucket%8==0, u32-indexable); a ragged tail
  // re-enters kind 1 with kFlagTailOnly.  kind 1: generic kernel.
  struct Ent {
    const Slice* s;
    int32_t flags;
  };
  const int es = elem_size(dt);
  std::map<std::pair<int, int>, std::vector<Ent>> groups;
  for (const auto& s : slices) {
    if (s.n <= 0) continue;
    const int32_t flags = s.skip_incomplete ? kFlagSkipIncomplete : 0;
    const int64_t nq =
        s.skip_incomplete ? s.n / s.bucket * (int64_t)s.bucket : s.n;
    const bool fast = (s.bucket % 8) == 0 && nq < (int64_t(1) << 31);
    if (fast) {
      groups[{s.bits, 0}].push_back(Ent{&s, flags});
      const bool ragged = es == 4 ? (nq & 3) != 0 : (nq & 7) != 0;
      if (ragged)
        groups[{s.bits, 1}].push_back(Ent{&s, flags | kFlagTailOnly});
    } else {
      groups[{s.bits, 1}].push_back(Ent{&s, flags});
    }
  }
  for (auto& [key, list] : groups) {
    const auto [bits, kind] = key;
    const int nsl = (int)list.size();
    const size_t desc_bytes = sizeof(DequantDesc) * nsl;
    const size_t cum_bytes = sizeof(int64_t) * (nsl + 1);
    char* host = (char*)ring_.acquire(desc_bytes + cum_bytes);
    auto* dd = reinterpret_cast<DequantDesc*>(host);
    auto* cum = reinterpret_cast<int64_t*>(host + desc_bytes);
    cum[0] = 0;
    bool any_residual = false;
    for (int i = 0; i < nsl; i++) {
      const Slice& s = *list[i].s;
      dd[i] = DequantDesc{in_base + s.comp_off, s.data,    s.n, src_stride,
                          s.bucket,             nsrc, add ? 1 : 0,
                          list[i].flags};
      int64_t work;
      if (list[i].flags & kFlagTailOnly) {
        work = 1;
      } else if (s.skip_incomplete) {
        const int64_t nq = s.n / s.bucket * (int64_t)s.bucket;
        work = (nq + 7) / 8;
        any_residual |= (s.n % s.bucket) != 0;
      } else {
        work = (s.n + 7) / 8;
      }
      cum[i + 1] = cum[i] + work;
    }
    char* dev = (char*)ring_.commit(desc_bytes + cum_bytes, stream);
    auto* ddesc = reinterpret_cast<DequantDesc*>(dev);
    auto* dcum = reinterpret_cast<int64_t*>(dev + desc_bytes);
    if (kind == 0) {
      launch_dequantize_fast(ddesc, nsl, cum[nsl], dt, bits, stream,
                             any_residual);
    } else {
      launch_dequantize_batch(ddesc, dcum, nsl, cum[nsl], dt, bits, stream,
                              any_residual);
    }
  }
}

Engine::ChunkPlan Engine::plan(const std::vector<LayerView>& views,
                               DType dt, bool skip_incomplete) {
  ChunkPlan pl;
  const int ws = size_;
  const int es = elem_size(dt);
  std::vector<int64_t> lnumels;
  lnumels.reserve(views.size());
  for (const auto& v : views) {
    lnumels.push_back(v.numel);
    pl.n += v.numel;
  }
  if (pl.n == 0) return pl;
  partition(pl.n, ws, lnumels, es, &pl.offs, &pl.szs);
  pl.rs.resize(ws);
  pl.comp.assign(ws, 0);
  for (int r = 0; r < ws; r++) {
    const int64_t start = pl.offs[r], end = pl.offs[r] + pl.szs[r];
    int64_t pos = 0, coff = 0;
    for (const auto& v : views) {
      const int64_t lo = std::max(pos, start);
      const int64_t hi = std::min(pos + v.numel, end);
      if (hi > lo) {
        pl.rs[r].push_back(Slice{v.data + (lo - pos) * es, hi - lo, v.bits,
                                 v.bucket_size, coff, skip_incomplete, lo});
        coff += buffer_size(hi - lo, dt, v.bits, v.bucket_size,
                            skip_incomplete);
      }
      pos += v.numel;
      if (pos >= end) break;
    }
    pl.comp[r] = coff;
  }
  return pl;
}

void Engine::timer_begin(hipStream_t qs, bool enabled) {
  timing_ = enabled;
  timer_slot_ = -1;
  if (!enabled) return;
  PhaseTimer& t = timer_;
  if (!t.inited) {
    for (int r = 0; r < PhaseTimer::kRing; r++)
      for (int e = 0; e < PhaseTimer::kEv; e++)
        CGX_HIP_CHECK(hipEventCreate(&t.ev[r][e]));  // timing-capable
    t.inited = true;
  }
  const int slot = t.cur;
  t.cur = (t.cur + 1) % PhaseTimer::kRing;
  if (t.pending[slot]) {
    if (hipEventQuery(t.ev[slot][PhaseTimer::kEv - 1]) == hipSuccess) {
      for (int p = 0; p + 1 < PhaseTimer::kEv; p++) {
        float ms = 0.f;
        if (hipEventElapsedTime(&ms, t.ev[slot][p], t.ev[slot][p + 1]) ==
            hipSuccess)
          t.sum_ms[p] += ms;
      }
      t.count++;
      if (t.count % 50 == 0) {
        fprintf(stderr,
                "[cgx timings rank %d, %lld chunks] quantize %.3f ms | "
                "comm1 %.3f | decode+requant %.3f | comm2 %.3f | decode2 "
                "%.3f\n",
                rank_, (long long)t.count, t.sum_ms[0] / t.count,
                t.sum_ms[1] / t.count, t.sum_ms[2] / t.count,
                t.sum_ms[3] / t.count, t.sum_ms[4] / t.count);
      }
      t.pending[slot] = false;
    } else {
      timing_ = false;  // ring full of in-flight chunks: skip this one
      return;
    }
  }
  timer_slot_ = slot;
  CGX_HIP_CHECK(hipEventRecord(t.ev[slot][0], qs));
}

void Engine::timer_mark(int idx, hipStream_t stream) {
  if (!timing_ || timer_slot_ < 0) return;
  CGX_HIP_CHECK(hipEventRecord(timer_.ev[timer_slot_][idx], stream));
}

void Engine::timer_finish() {
  if (timing_ && timer_slot_ >= 0) timer_.pending[timer_slot_] = true;
}

void Engine::sra_chunk(const std::vector<LayerView>& views, DType dt,
                       Transport* tr, hipStream_t qs,
                       const EngineConfig& cfg, int64_t fb_key) {
  const int ws = size_;
  ChunkPlan pl = plan(views, dt, cfg.skip_incomplete);
  if (pl.n == 0) return;
  auto& rs = pl.rs;
  auto& comp = pl.comp;
  const int64_t mycomp = comp[rank_];

  int64_t send1_total = 0;
  for (int p = 0; p < ws; p++)
    if (p != rank_) send1_total += comp[p];
  const int64_t recv1_total = (int64_t)(ws - 1) * mycomp;

  StagingSlot& sslot = slots_[slot_cur_];
  slot_cur_ ^= 1;
  if (sslot.recorded) {
    // don't overwrite staging another chunk's comm/decode still reads
    CGX_HIP_CHECK(hipStreamWaitEvent(qs, sslot.done_ev, 0));
  }
  uint8_t* base =
      slot_bytes(sslot, send1_total + recv1_total + mycomp + send1_total);
  uint8_t* send1 = base;
  uint8_t* recv1 = send1 + send1_total;
  uint8_t* send2 = recv1 + recv1_total;
  uint8_t* recv2 = send2 + mycomp;

  std::vector<int64_t> peer_off(ws, 0);
  {
    int64_t o = 0;
    for (int p = 0; p < ws; p++) {
      if (p == rank_) continue;
      peer_off[p] = o;
      o += comp[p];
    }
  }
  auto slot = [&](int p) { return p < rank_ ? p : p - 1; };

  // round 1 quantize of every peer chunk, on the quantize stream
  {
    std::vector<Slice> all;
    for (int p = 0; p < ws; p++) {
      if (p == rank_) continue;
      for (const auto& sl : rs[p]) {
        Slice t = sl;
        t.comp_off += peer_off[p];
        all.push_back(t);
      }
    }
    char* fb1 = nullptr;
    if (cfg.error_feedback) {
      at::Tensor& t =
          feedback_buf(fb_key, /*phase=*/1, pl.n,
                       dt == DType::F32 ? at::kFloat
                       : dt == DType::F16 ? at::kHalf : at::kBFloat16);
      fb1 = static_cast<char*>(t.data_ptr());
    }
    run_quantize(all, send1, dt, qs, cfg.stochastic, fb1);
  }
  timer_mark(1, qs);

  // round 1 exchange on the comm stream (grouped p2p drives all xGMI links)
  chain(qs, comm_stream_);
  {
    std::vector<Transport::Op> sends, recvs;
    for (int p = 0; p < ws; p++) {
      if (p == rank_) continue;
      sends.push_back(Transport::Op{send1 + peer_off[p], comp[p], p});
      recvs.push_back(
          Transport::Op{recv1 + (int64_t)slot(p) * mycomp, mycomp, p});
    }
    tr->exchange(sends, recvs, comm_stream_);
  }
  timer_mark(2, comm_stream_);

  // decode-accumulate + self-quantize on the deq stream
  chain(comm_stream_, deq_stream_);
  if (mycomp > 0) {
    run_dequant(rs[rank_], recv1, mycomp, ws - 1, /*add=*/true, dt,
                deq_stream_);
    // self-quantize the reduced chunk; the same bytes go to every peer and
    // through my own decode so all ranks end bit-identical
    char* fb2 = nullptr;
    int64_t fb2_rebase = 0;
    if (cfg.error_feedback && pl.szs[rank_] > 0) {
      at::Tensor& t =
          feedback_buf(fb_key, /*phase=*/2, pl.szs[rank_],
                       dt == DType::F32 ? at::kFloat
                       : dt == DType::F16 ? at::kHalf : at::kBFloat16);
      // slice fb_off is in chunk element space; the phase-2 buffer covers
      // only this rank's partition, so rebase by its start (host-side per
      // slice — no out-of-range pointer arithmetic)
      fb2 = static_cast<char*>(t.data_ptr());
      fb2_rebase = pl.offs[rank_];
    }
    run_quantize(rs[rank_], send2, dt, deq_stream_, cfg.stochastic, fb2,
                 fb2_rebase);
  }
  timer_mark(3, deq_stream_);

  // round 2 exchange on the comm stream
  chain(deq_stream_, comm_stream_);
  {
    std::vector<Transport::Op> sends, recvs;
    for (int p = 0; p < ws; p++) {
      if (p == rank_) continue;
      sends.push_back(Transport::Op{send2, mycomp, p});
      recvs.push_back(Transport::Op{recv2 + peer_off[p], comp[p], p});
    }
    tr->exchange(sends, recvs, comm_stream_);
  }
  timer_mark(4, comm_stream_);

  // final decode: own chunk from send2, peers' chunks from recv2
  chain(comm_stream_, deq_stream_);
  if (mycomp > 0)
    run_dequant(rs[rank_], send2, 0, 1, /*add=*/false, dt, deq_stream_);
  {
    std::vector<Slice> all;
    for (int p = 0; p < ws; p++) {
      if (p == rank_) continue;
      for (const auto& sl : rs[p]) {
        Slice t = sl;
        t.comp_off += peer_off[p];
        all.push_back(t);
      }
    }
    run_dequant(all, recv2, 0, 1, /*add=*/false, dt, deq_stream_);
  }
  timer_mark(5, deq_stream_);
  timer_finish();
  CGX_HIP_CHECK(hipEventRecord(sslot.done_ev, deq_stream_));
  sslot.recorded = true;
}

void Engine::ring_chunk(const std::vector<LayerView>& views, DType dt,
                        Transport* tr, hipStream_t stream,
                        const EngineConfig& cfg) {
  // Compressed ring allreduce (reference MPI_Allreduce_Ring semantics,
  // ring.cc:139-226): ws-1 reduce-scatter steps with per-hop requantize of
  // the running partial sum, then ws-1 allgather steps FORWARDING the
  // once-quantized reduced segments, final batch decompress.
  TORCH_CHECK(!cfg.error_feedback,
              "cgx: CGX_ERROR_FEEDBACK is not supported with Ring reduction "
              "(the ring requantizes running partial sums, which have no "
              "stable per-step residual); use SRA or disable error feedback");
  const int ws = size_;
  ChunkPlan pl = plan(views, dt, cfg.skip_incomplete);
  if (pl.n == 0) return;
  auto& rs = pl.rs;
  auto& comp = pl.comp;
  const int next = (rank_ + 1) % ws;
  const int prev = (rank_ + ws - 1) % ws;

  int64_t seg_total = 0, maxc = 0;
  std::vector<int64_t> seg_off(ws, 0);
  for (int k = 0; k < ws; k++) {
    seg_off[k] = seg_total;
    seg_total += comp[k];
    maxc = std::max(maxc, comp[k]);
  }
  uint8_t* base = staging(seg_total + 2 * maxc, stream);
  uint8_t* segs = base;
  uint8_t* tmp_send = base + seg_total;
  uint8_t* tmp_recv = tmp_send + maxc;

  // reduce-scatter: accumulate into local chunks hop by hop
  for (int s = 0; s < ws - 1; s++) {
    const int send_c = (rank_ - s + ws) % ws;
    const int recv_c = (rank_ - s - 1 + ws) % ws;
    if (comp[send_c] > 0)
      run_quantize(rs[send_c], tmp_send, dt, stream, cfg.stochastic);
    tr->exchange({Transport::Op{tmp_send, comp[send_c], next}},
                 {Transport::Op{tmp_recv, comp[recv_c], prev}}, stream);
    if (comp[recv_c] > 0)
      run_dequant(rs[recv_c], tmp_recv, 0, 1, /*add=*/true, dt, stream);
  }

  // own fully-reduced chunk after ws-1 hops
  const int own = (rank_ + 1) % ws;
  if (comp[own] > 0)
    run_quantize(rs[own], segs + seg_off[own], dt, stream, cfg.stochastic);

  // allgather: forward the quantized reduced segments around the ring
  for (int s = 0; s < ws - 1; s++) {
    const int send_c = (own - s + ws) % ws;
    const int recv_c = (own - s - 1 + ws) % ws;
    tr->exchange({Transport::Op{segs + seg_off[send_c], comp[send_c], next}},
                 {Transport::Op{segs + seg_off[recv_c], comp[recv_c], prev}},
                 stream);
  }

  // final decode of every segment (own included, for bit-identical results)
  std::vector<Slice> all;
  for (int k = 0; k < ws; k++) {
    for (const auto& sl : rs[k]) {
      Slice t = sl;
      t.comp_off += seg_off[k];
      all.push_back(t);
    }
  }
  run_dequant(all, segs, 0, 1, /*add=*/false, dt, stream);
}

void Engine::a2a_chunk(const std::vector<LayerView>& views, DType dt,
                       Transport* tr, hipStream_t qs,
                       const EngineConfig& cfg) {
  // Debug brute-force reduction (reference AllReduceAlltoAllCompressed,
  // scatter_reduce_allgather.cc:269-306): every rank quantizes its ENTIRE
  // chunk once and sends it to every peer; every rank then decodes the same
  // ws compressed streams (self first, then peers in rank order).  Each
  // rank's data is quantized exactly once, but the per-rank ACCUMULATION
  // order differs (self-first), so at ws>2 results agree to fp rounding,
  // not bitwise — same as the reference's arrival-order accumulate.  ws x
  // the wire bytes of SRA, but no partitioning machinery in the fault
  // surface.
  TORCH_CHECK(!cfg.error_feedback,
              "cgx: CGX_DEBUG_ALL_TO_ALL_REDUCTION does not support "
              "CGX_ERROR_FEEDBACK");
  const int ws = size_;
  std::vector<Slice> sl;
  int64_t coff = 0, pos = 0;
  for (const auto& v : views) {
    if (v.numel <= 0) continue;
    sl.push_back(Slice{v.data, v.numel, v.bits, v.bucket_size, coff,
                       cfg.skip_incomplete, pos});
    coff +=
        buffer_size(v.numel, dt, v.bits, v.bucket_size, cfg.skip_incomplete);
    pos += v.numel;
  }
  if (coff == 0) return;
  const int64_t C = coff;
  uint8_t* base = staging(C * ws, qs);
  uint8_t* send = base;          // own compressed stream
  uint8_t* recv = base + C;      // ws-1 peer streams, stride C
  run_quantize(sl, send, dt, qs, cfg.stochastic);
  chain(qs, comm_stream_);
  {
    std::vector<Transport::Op> sends, recvs;
    for (int p = 0; p < ws; p++) {
      if (p == rank_) continue;
      const int s = p < rank_ ? p : p - 1;
      sends.push_back(Transport::Op{send, C, p});
      recvs.push_back(Transport::Op{recv + (int64_t)s * C, C, p});
    }
    tr->exchange(sends, recvs, comm_stream_);
  }
  chain(comm_stream_, deq_stream_);
  run_dequant(sl, send, 0, 1, /*add=*/false, dt, deq_stream_);
  if (ws > 1)
    run_dequant(sl, recv, C, ws - 1, /*add=*/true, dt, deq_stream_);
  // the shared staging() buffer is reused by the next chunk's quantize on
  // qs: fence it behind this chunk's decode
  chain(deq_stream_, qs);
}

hipStream_t Engine::broadcast(at::Tensor t, int root, Transport* tr,
                              hipStream_t qs) {
  const EngineConfig cfg = EngineConfig::from_env();
  const int64_t n = t.numel();
  const bool compress = !cfg.dummy && cfg.intra_compress &&
                        cfg.default_bits <= 8 && n > cfg.min_elems &&
                        (t.scalar_type() == at::kFloat ||
                         t.scalar_type() == at::kHalf ||
                         t.scalar_type() == at::kBFloat16);
  if (!compress || size_ <= 1) {
    if (size_ > 1) {
      tr->broadcast(t.data_ptr(), n, (int)t.element_size(), nccl_dtype(t),
                    root, qs);
    }
    return qs;
  }
  const DType dt = dtype_of(t);
  const int64_t bytes =
      buffer_size(n, dt, cfg.default_bits, cfg.default_bucket,
                  cfg.skip_incomplete);
  uint8_t* buf = staging(bytes, qs);
  std::vector<Slice> sl{Slice{static_cast<char*>(t.data_ptr()), n,
                              cfg.default_bits, cfg.default_bucket, 0,
                              cfg.skip_incomplete}};
  if (rank_ == root) {
    run_quantize(sl, buf, dt, qs, cfg.stochastic);
  }
  tr->broadcast(buf, bytes, 1, ncclUint8, root, qs);
  // every rank (root included) decodes the same bytes -> bit-identical
  run_dequant(sl, buf, 0, 1, /*add=*/false, dt, qs);
  return qs;
}

hipStream_t Engine::allreduce(at::Tensor bucket, Transport* tr,
                              hipStream_t qs,
                              const Registry::BucketInfo* forced,
                              bool forced_match) {
  if (size_ <= 1) return qs;
  TORCH_CHECK(bucket.is_contiguous(), "cgx: bucket must be contiguous");
  EngineConfig cfg = EngineConfig::from_env();
  if (is_cross_) cfg.ring = cfg.cross_ring;
  const DType dt = dtype_of(bucket);
  const int es = elem_size(dt);
  char* base = static_cast<char*>(bucket.data_ptr());
  const int64_t numel = bucket.numel();

  Registry::BucketInfo info;
  bool matched;
  if (forced) {
    matched = forced_match;
    if (matched) info = *forced;
  } else {
    matched = Registry::get().next(numel, bucket.data_ptr(), &info);
  }
  std::vector<LayerView> views;
  if (matched) {
    int64_t off = 0;
    for (size_t i = 0; i < info.numels.size(); i++) {
      views.push_back(LayerView{base + off * es, info.numels[i],
                                info.cfgs[i].bits, info.cfgs[i].bucket_size});
      off += info.numels[i];
    }
  } else {
    views.push_back(LayerView{base, numel, cfg.default_bits,
                              cfg.default_bucket});
  }

  // split into compressible layers and uncompressed merged ranges
  std::vector<LayerView> comp_views;
  std::vector<std::pair<char*, int64_t>> uncomp;
  for (const auto& v : views) {
    const bool c = !cfg.dummy && v.bits <= 8 && v.numel > cfg.min_elems;
    if (c) {
      comp_views.push_back(v);
    } else {
      if (!uncomp.empty() &&
          uncomp.back().first + uncomp.back().second * es == v.data) {
        uncomp.back().second += v.numel;
      } else {
        uncomp.emplace_back(v.data, v.numel);
      }
    }
  }

  if (!uncomp.empty()) {
    const ncclDataType_t ndt = nccl_dtype(bucket);
    chain(qs, comm_stream_);
    std::vector<std::pair<void*, int64_t>> bufs;
    for (auto& [ptr, cnt] : uncomp) bufs.emplace_back(ptr, cnt);
    tr->allreduce_sum_group(bufs, ndt, comm_stream_);
  }

  // tensor-fusion chunking over the compressible layers
  const int64_t fusion_elems = std::max<int64_t>(256, cfg.fusion_bytes / es);
  std::vector<LayerView> cur;
  int64_t cur_n = 0;
  static const bool timings_on = env_int("CGX_TIMINGS", 0) != 0;
  bool any_comp = false;
  int chunk_seq = 0;
  auto run_chunk = [&](const std::vector<LayerView>& vs_in) {
    any_comp = true;
    // error-feedback residual identity for this (bucket, chunk): registry
    // bucket index + chunk ordinal when matched; pointer-derived negative
    // fallback key otherwise (DDP flat buffers are pointer-stable)
    const int64_t fb_key =
        matched ? (((int64_t)info.idx << 20) | chunk_seq)
                : -(int64_t)(reinterpret_cast<uintptr_t>(vs_in[0].data) >> 3);
    chunk_seq++;
    timer_begin(qs, timings_on && !cfg.ring);
    // CGX_COMPRESSION_FAKE_RATIO < 1: reduce only a fraction of each chunk
    // (bandwidth experiments; intentionally lossy -- reference
    // mpi_allreduce_operations.cc:143-144)
    std::vector<LayerView> trimmed;
    const std::vector<LayerView>* vsp = &vs_in;
    if (cfg.fake_ratio < 1.0) {
      int64_t total = 0;
      for (const auto& v : vs_in) total += v.numel;
      int64_t keep = std::max<int64_t>(16, (int64_t)(total * cfg.fake_ratio));
      for (const auto& v : vs_in) {
        if (keep <= 0) break;
        LayerView w = v;
        w.numel = std::min(v.numel, keep);
        keep -= w.numel;
        trimmed.push_back(w);
      }
      vsp = &trimmed;
    }
    const std::vector<LayerView>& vs = *vsp;
    if (cfg.debug_a2a) {
      a2a_chunk(vs, dt, tr, qs, cfg);
    } else if (cfg.ring && size_ > 2) {
      // ring is hop-serial and runs on one stream (qs).  Its NCCL ops must
      // be ordered behind any uncompressed ncclAllReduce group queued on
      // comm_stream_ above: unordered concurrent ops on one communicator
      // are UB in NCCL/RCCL.
      chain(comm_stream_, qs);
      ring_chunk(vs, dt, tr, qs, cfg);
      chain(qs, deq_stream_);             // keep completion on deq stream
    } else {
      sra_chunk(vs, dt, tr, qs, cfg, fb_key);
    }
  };
  auto flush = [&]() {
    if (!cur.empty()) {
      run_chunk(cur);
      cur.clear();
      cur_n = 0;
    }
  };
  for (const auto& v : comp_views) {
    if (v.numel >= fusion_elems) {
      flush();
      for (int64_t o = 0; o < v.numel; o += fusion_elems) {
        std::vector<LayerView> w{LayerView{v.data + o * es,
                                           std::min(fusion_elems, v.numel - o),
                                           v.bits, v.bucket_size}};
        run_chunk(w);
      }
    } else {
      if (cur_n + v.numel > fusion_elems) flush();
      cur.push_back(v);
      cur_n += v.numel;
    }
  }
  flush();

  // completion stream: all compressed chunks end on deq_stream_; the
  // uncompressed group ran (first) on comm_stream_, which every later chunk
  // already chained through -- but cover the uncomp-only case explicitly.
  if (any_comp) {
    if (!uncomp.empty() && comp_views.empty()) chain(comm_stream_, deq_stream_);
    return deq_stream_;
  }
  if (!uncomp.empty()) return comm_stream_;
  return qs;
}

}  // namespace cgx
