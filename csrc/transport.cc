// Transport implementations (see transport.h).
#include "transport.h"

#include <ATen/ATen.h>

#include "engine.h"  // CGX_HIP_CHECK / CGX_NCCL_CHECK

namespace cgx {

// ---------------------------------------------------------------------------
// RcclTransport
// ---------------------------------------------------------------------------
void RcclTransport::exchange(const std::vector<Op>& sends,
                             const std::vector<Op>& recvs,
                             hipStream_t stream) {
  CGX_NCCL_CHECK(ncclGroupStart());
  for (const auto& s : sends) {
    if (s.bytes <= 0) continue;
    CGX_NCCL_CHECK(
        ncclSend(s.ptr, s.bytes, ncclUint8, s.peer, comm_, stream));
  }
  for (const auto& r : recvs) {
    if (r.bytes <= 0) continue;
    CGX_NCCL_CHECK(
        ncclRecv(r.ptr, r.bytes, ncclUint8, r.peer, comm_, stream));
  }
  CGX_NCCL_CHECK(ncclGroupEnd());
}

void RcclTransport::broadcast(void* ptr, int64_t count, int /*esize*/,
                              ncclDataType_t dt, int root,
                              hipStream_t stream) {
  CGX_NCCL_CHECK(ncclBroadcast(ptr, ptr, count, dt, root, comm_, stream));
}

void RcclTransport::allreduce_sum(void* ptr, int64_t count, ncclDataType_t dt,
                                  hipStream_t stream) {
  CGX_NCCL_CHECK(ncclAllReduce(ptr, ptr, count, dt, ncclSum, comm_, stream));
}

void RcclTransport::allreduce_sum_group(
    const std::vector<std::pair<void*, int64_t>>& bufs, ncclDataType_t dt,
    hipStream_t stream) {
  CGX_NCCL_CHECK(ncclGroupStart());
  for (const auto& [ptr, cnt] : bufs) {
    CGX_NCCL_CHECK(ncclAllReduce(ptr, ptr, cnt, dt, ncclSum, comm_, stream));
  }
  CGX_NCCL_CHECK(ncclGroupEnd());
}

// ---------------------------------------------------------------------------
// LoopbackHub
// ---------------------------------------------------------------------------
LoopbackHub::~LoopbackHub() {
  for (hipEvent_t e : events_)
    if (e) (void)hipEventDestroy(e);
  for (Msg* m : msgs_) delete m;
}

hipEvent_t LoopbackHub::new_event() {
  hipEvent_t e = nullptr;
  CGX_HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
  std::lock_guard<std::mutex> g(mu_);
  events_.push_back(e);
  return e;
}

LoopbackHub::Msg* LoopbackHub::post_send(int src, int dst, const void* ptr,
                                         int64_t bytes, hipEvent_t ready) {
  auto* m = new Msg{ptr, bytes, ready, nullptr, false};
  {
    std::lock_guard<std::mutex> g(mu_);
    msgs_.push_back(m);
    boxes_[{src, dst}].push_back(m);
  }
  cv_.notify_all();
  return m;
}

LoopbackHub::Msg* LoopbackHub::take_send(int src, int dst) {
  std::unique_lock<std::mutex> g(mu_);
  auto& box = boxes_[{src, dst}];
  cv_.wait(g, [&] { return aborted_ || !box.empty(); });
  TORCH_CHECK(!aborted_, "cgx loopback: hub aborted (peer rank failed)");
  Msg* m = box.front();
  box.pop_front();
  return m;
}

void LoopbackHub::mark_consumed(Msg* m, hipEvent_t consumed) {
  {
    std::lock_guard<std::mutex> g(mu_);
    m->consumed = consumed;
    m->done = true;
  }
  cv_.notify_all();
}

void LoopbackHub::wait_consumed(Msg* m) {
  std::unique_lock<std::mutex> g(mu_);
  cv_.wait(g, [&] { return aborted_ || m->done; });
  TORCH_CHECK(!aborted_, "cgx loopback: hub aborted (peer rank failed)");
}

void LoopbackHub::abort() {
  {
    std::lock_guard<std::mutex> g(mu_);
    aborted_ = true;
  }
  cv_.notify_all();
}

// ---------------------------------------------------------------------------
// LoopbackTransport
// ---------------------------------------------------------------------------
void LoopbackTransport::exchange(const std::vector<Op>& sends,
                                 const std::vector<Op>& recvs,
                                 hipStream_t stream) {
  // phase 1: post every send
  std::vector<LoopbackHub::Msg*> mine;
  mine.reserve(sends.size());
  for (const auto& s : sends) {
    if (s.bytes <= 0) continue;
    hipEvent_t ev = hub_->new_event();
    CGX_HIP_CHECK(hipEventRecord(ev, stream));
    mine.push_back(hub_->post_send(rank_, s.peer, s.ptr, s.bytes, ev));
  }
  // phase 2: resolve every recv with a D2D copy on MY stream
  for (const auto& r : recvs) {
    if (r.bytes <= 0) continue;
    LoopbackHub::Msg* m = hub_->take_send(r.peer, rank_);
    TORCH_CHECK(m->bytes == r.bytes, "cgx loopback: size mismatch (send ",
                m->bytes, " vs recv ", r.bytes, " bytes, peer ", r.peer, ")");
    CGX_HIP_CHECK(hipStreamWaitEvent(stream, m->ready, 0));
    CGX_HIP_CHECK(hipMemcpyAsync(r.ptr, m->src, r.bytes,
                                 hipMemcpyDeviceToDevice, stream));
    hipEvent_t done = hub_->new_event();
    CGX_HIP_CHECK(hipEventRecord(done, stream));
    hub_->mark_consumed(m, done);
  }
  // phase 3: my stream resumes only after every peer copied my sends out
  for (LoopbackHub::Msg* m : mine) {
    hub_->wait_consumed(m);
    CGX_HIP_CHECK(hipStreamWaitEvent(stream, m->consumed, 0));
  }
}

void LoopbackTransport::broadcast(void* ptr, int64_t count, int esize,
                                  ncclDataType_t /*dt*/, int root,
                                  hipStream_t stream) {
  const int64_t bytes = count * esize;
  std::vector<Op> sends, recvs;
  if (rank_ == root) {
    for (int p = 0; p < size_; p++)
      if (p != root) sends.push_back(Op{ptr, bytes, p});
  } else {
    recvs.push_back(Op{ptr, bytes, root});
  }
  exchange(sends, recvs, stream);
}

namespace {
DType loopback_dtype(ncclDataType_t dt) {
  switch (dt) {
    case ncclFloat32: return DType::F32;
    case ncclFloat16: return DType::F16;
    case ncclBfloat16: return DType::BF16;
    default:
      TORCH_CHECK(false,
                  "cgx loopback: unsupported allreduce dtype (only "
                  "fp32/fp16/bf16)");
  }
}
int nccl_esize(ncclDataType_t dt) { return dt == ncclFloat32 ? 4 : 2; }
}  // namespace

void LoopbackTransport::allreduce_sum(void* ptr, int64_t count,
                                      ncclDataType_t dt, hipStream_t stream) {
  // brute-force: allgather every peer's buffer into scratch, then sum
  // locally with the engine's add kernel.  The exchange's phase-3 fence
  // guarantees peers copied MY original data before the adds mutate it.
  const DType t = loopback_dtype(dt);
  const int es = nccl_esize(dt);
  const int64_t bytes = count * es;
  auto scratch = at::empty({std::max<int64_t>((size_ - 1) * bytes, 1)},
                           at::TensorOptions()
                               .dtype(at::kByte)
                               .device(at::kCUDA));
  std::vector<Op> sends, recvs;
  int slot = 0;
  for (int p = 0; p < size_; p++) {
    if (p == rank_) continue;
    sends.push_back(Op{ptr, bytes, p});
    recvs.push_back(
        Op{scratch.data_ptr<uint8_t>() + (int64_t)slot * bytes, bytes, p});
    slot++;
  }
  exchange(sends, recvs, stream);
  for (int s = 0; s < slot; s++) {
    launch_add(scratch.data_ptr<uint8_t>() + (int64_t)s * bytes, ptr, count,
               t, stream);
  }
  // scratch is a temporary; its storage must not be reused by the caching
  // allocator while the stream still reads it — a stream sync here is the
  // simple correct choice for a test transport
  CGX_HIP_CHECK(hipStreamSynchronize(stream));
}

void LoopbackTransport::allreduce_sum_group(
    const std::vector<std::pair<void*, int64_t>>& bufs, ncclDataType_t dt,
    hipStream_t stream) {
  // sequential exchanges in list order: every rank passes the same list, so
  // the hub rendezvous stays matched
  for (const auto& [ptr, cnt] : bufs) allreduce_sum(ptr, cnt, dt, stream);
}

}  // namespace cgx
