// Implementation of ProcessGroupCGX (see backend.h).
#include "backend.h"

#include <ATen/hip/impl/HIPCachingAllocatorMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>

#include <cstring>
#include <unistd.h>

namespace cgx {

namespace {

ncclRedOp_t to_nccl_op(const c10d::ReduceOp& op) {
  switch (op) {
    case c10d::ReduceOp::SUM: return ncclSum;
    case c10d::ReduceOp::AVG: return ncclAvg;
    case c10d::ReduceOp::PRODUCT: return ncclProd;
    case c10d::ReduceOp::MIN: return ncclMin;
    case c10d::ReduceOp::MAX: return ncclMax;
    default:
      TORCH_CHECK(false, "cgx: unsupported reduce op");
  }
}

bool env_flag(const char* name) {
  const char* v = std::getenv(name);
  return v && *v && std::strcmp(v, "0") != 0;
}

void check_single(const std::vector<at::Tensor>& ts) {
  TORCH_CHECK(ts.size() == 1, "cgx: expected exactly one tensor per rank");
  // contiguity is only required on the RCCL/kernel path; the gloo CPU
  // delegate handles arbitrary layouts itself
  TORCH_CHECK(!ts[0].is_cuda() || ts[0].is_contiguous(),
              "cgx: CUDA tensor must be contiguous");
}

}  // namespace

// ---------------------------------------------------------------------------
// WorkCGX
// ---------------------------------------------------------------------------
WorkCGX::WorkCGX(int rank, c10d::OpType op, at::Device device,
                 std::vector<at::Tensor> outputs,
                 const char* profiling_title)
    : c10d::Work(rank, op, profiling_title), device_(device),
      outputs_(std::move(outputs)) {
  future_ = c10::make_intrusive<c10::ivalue::Future>(
      c10::ListType::ofTensors(), std::vector<c10::Device>{device_});
}

WorkCGX::~WorkCGX() {
  if (ev_) (void)hipEventDestroy(ev_);
}

void WorkCGX::recordEnd(const c10::hip::HIPStreamMasqueradingAsCUDA& stream) {
  CGX_HIP_CHECK(hipEventCreateWithFlags(&ev_, hipEventDisableTiming));
  CGX_HIP_CHECK(hipEventRecord(ev_, stream.stream()));
  recorded_ = true;
  // stream is current here (caller holds a stream guard), so the future's
  // CUDA hooks capture the side stream for callback chaining.
  future_->markCompleted(c10::IValue(outputs_));
}

bool WorkCGX::isCompleted() {
  if (!recorded_) return true;
  return hipEventQuery(ev_) == hipSuccess;
}

bool WorkCGX::isSuccess() const { return true; }

bool WorkCGX::wait(std::chrono::milliseconds /*timeout*/) {
  if (recorded_) {
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_.index());
    CGX_HIP_CHECK(hipStreamWaitEvent(cur.stream(), ev_, 0));
    if (env_flag("CGX_BLOCKING_WAIT")) CGX_HIP_CHECK(hipEventSynchronize(ev_));
  }
  return true;
}

void WorkCGX::synchronize() { (void)wait(std::chrono::milliseconds(0)); }

c10::intrusive_ptr<c10::ivalue::Future> WorkCGX::getFuture() {
  return future_;
}

// ---------------------------------------------------------------------------
// Topology
// ---------------------------------------------------------------------------
Topology compute_topology(const c10::intrusive_ptr<c10d::Store>& store,
                          int rank, int size, const std::string& hostname) {
  store->set("cgx/host/" + std::to_string(rank),
             std::vector<uint8_t>(hostname.begin(), hostname.end()));
  std::vector<std::string> hosts(size);
  for (int r = 0; r < size; r++) {
    auto v = store->get("cgx/host/" + std::to_string(r));  // blocks until set
    hosts[r] = std::string(v.begin(), v.end());
  }
  Topology t;
  std::vector<std::string> node_order;  // node hostnames by first appearance
  std::vector<int> node_of(size);
  for (int r = 0; r < size; r++) {
    int id = -1;
    for (size_t i = 0; i < node_order.size(); i++)
      if (node_order[i] == hosts[r]) { id = (int)i; break; }
    if (id < 0) {
      id = (int)node_order.size();
      node_order.push_back(hosts[r]);
    }
    node_of[r] = id;
  }
  t.n_nodes = (int)node_order.size();
  t.node_id = node_of[rank];
  std::vector<int> counts(t.n_nodes, 0);
  for (int r = 0; r < size; r++) {
    if (r == rank) t.local_rank = counts[node_of[r]];
    counts[node_of[r]]++;
  }
  t.local_size = counts[t.node_id];
  for (int c : counts)
    if (c != counts[0]) t.uniform = false;
  return t;
}

// ---------------------------------------------------------------------------
// ProcessGroupCGX
// ---------------------------------------------------------------------------
ProcessGroupCGX::ProcessGroupCGX(c10::intrusive_ptr<c10d::Store> store,
                                 int rank, int size,
                                 c10::intrusive_ptr<c10d::Backend> cpu_delegate)
    : c10d::Backend(rank, size), store_(std::move(store)),
      cpu_(std::move(cpu_delegate)) {}

ProcessGroupCGX::~ProcessGroupCGX() {
  // drain pending work before tearing communicators down
  if (stream_) (void)hipStreamSynchronize(stream_->stream());
  for (Engine* e : {engine_.get(), intra_engine_.get(),
                    cross_engine_.get()}) {
    if (!e) continue;
    (void)hipStreamSynchronize(e->comm_stream());
    (void)hipStreamSynchronize(e->deq_stream());
  }
  if (intra_comm_) (void)ncclCommDestroy(intra_comm_);
  if (cross_comm_) (void)ncclCommDestroy(cross_comm_);
  if (comm_) (void)ncclCommDestroy(comm_);
  if (start_ev_) (void)hipEventDestroy(start_ev_);
}

void ProcessGroupCGX::lazyInit(at::Device device) {
  TORCH_CHECK(device.is_cuda(), "cgx: GPU path requires a CUDA/HIP tensor");
  if (comm_) {
    TORCH_CHECK(device.index() == device_index_,
                "cgx: one process drives one GPU (got device ", device.index(),
                ", initialized with ", device_index_, ")");
    return;
  }
  device_index_ = device.index();
  c10::hip::HIPGuardMasqueradingAsCUDA guard(device_index_);
  ncclUniqueId id;
  if (rank_ == 0) {
    CGX_NCCL_CHECK(ncclGetUniqueId(&id));
    store_->set("cgx/nccl_uid",
                std::vector<uint8_t>(reinterpret_cast<uint8_t*>(&id),
                                     reinterpret_cast<uint8_t*>(&id) +
                                         sizeof(id)));
  } else {
    auto v = store_->get("cgx/nccl_uid");
    TORCH_CHECK(v.size() == sizeof(id), "cgx: bad nccl uid in store");
    std::memcpy(&id, v.data(), sizeof(id));
  }
  CGX_NCCL_CHECK(ncclCommInitRank(&comm_, size_, id, rank_));
  tr_ = std::make_unique<RcclTransport>(comm_, rank_, size_);
  stream_ = c10::hip::getStreamFromPoolMasqueradingAsCUDA(
      /*isHighPriority=*/true, device_index_);
  engine_ = std::make_unique<Engine>(rank_, size_);
  CGX_HIP_CHECK(hipEventCreateWithFlags(&start_ev_, hipEventDisableTiming));

  // multi-node: build intra/cross communicators for the hierarchical path
  // (CGX_HIERARCHICAL=0 disables; CGX_INTRA_BROADCAST=0 falls back flat,
  // matching the reference's mode switch)
  char host[256] = {0};
  gethostname(host, sizeof(host) - 1);
  topo_ = compute_topology(store_, rank_, size_, host);
  const char* henv = std::getenv("CGX_HIERARCHICAL");
  const bool want_hier = !(henv && std::strcmp(henv, "0") == 0);
  if (topo_.n_nodes > 1 && topo_.uniform && topo_.local_size > 1 &&
      want_hier) {
    // intra communicator: ranks on this node
    const std::string ikey = "cgx/intra_uid/" + std::to_string(topo_.node_id);
    ncclUniqueId iid;
    if (topo_.local_rank == 0) {
      CGX_NCCL_CHECK(ncclGetUniqueId(&iid));
      store_->set(ikey, std::vector<uint8_t>(
                            reinterpret_cast<uint8_t*>(&iid),
                            reinterpret_cast<uint8_t*>(&iid) + sizeof(iid)));
    } else {
      auto v = store_->get(ikey);
      std::memcpy(&iid, v.data(), sizeof(iid));
    }
    CGX_NCCL_CHECK(ncclCommInitRank(&intra_comm_, topo_.local_size, iid,
                                    topo_.local_rank));
    intra_tr_ = std::make_unique<RcclTransport>(intra_comm_, topo_.local_rank,
                                                topo_.local_size);
    // cross communicator: peers with the same local_rank on every node
    const std::string ckey =
        "cgx/cross_uid/" + std::to_string(topo_.local_rank);
    ncclUniqueId cid;
    if (topo_.node_id == 0) {
      CGX_NCCL_CHECK(ncclGetUniqueId(&cid));
      store_->set(ckey, std::vector<uint8_t>(
                            reinterpret_cast<uint8_t*>(&cid),
                            reinterpret_cast<uint8_t*>(&cid) + sizeof(cid)));
    } else {
      auto v = store_->get(ckey);
      std::memcpy(&cid, v.data(), sizeof(cid));
    }
    CGX_NCCL_CHECK(
        ncclCommInitRank(&cross_comm_, topo_.n_nodes, cid, topo_.node_id));
    cross_tr_ = std::make_unique<RcclTransport>(cross_comm_, topo_.node_id,
                                                topo_.n_nodes);
    intra_engine_ = std::make_unique<Engine>(topo_.local_rank,
                                             topo_.local_size);
    cross_engine_ = std::make_unique<Engine>(topo_.node_id, topo_.n_nodes,
                                             /*is_cross=*/true);
    hierarchical_ = true;
  }
  if (env_flag("CGX_VERBOSE") && rank_ == 0) {
    fprintf(stderr,
            "[cgx] init: world=%d nodes=%d local_size=%d hierarchical=%d "
            "device=%d\n",
            size_, topo_.n_nodes, topo_.local_size, (int)hierarchical_,
            device_index_);
  }
}

namespace {
void check_async_error(ncclComm_t comm) {
  if (!comm) return;
  ncclResult_t e = ncclSuccess;
  (void)ncclCommGetAsyncError(comm, &e);
  TORCH_CHECK(e == ncclSuccess,
              "cgx: RCCL async error on communicator: ",
              ncclGetErrorString(e));
}
}  // namespace

template <typename Fn>
c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::collective(
    std::vector<at::Tensor> outputs, at::Device device, c10d::OpType op,
    Fn&& fn) {
  std::lock_guard<std::mutex> lock(mu_);
  lazyInit(device);
  check_async_error(comm_);
  c10::hip::HIPGuardMasqueradingAsCUDA dguard(device_index_);
  auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_index_);
  CGX_HIP_CHECK(hipEventRecord(start_ev_, cur.stream()));
  CGX_HIP_CHECK(hipStreamWaitEvent(stream_->stream(), start_ev_, 0));
  // RCCL forbids unordered concurrent ops on one communicator: fence this
  // passthrough op behind any compressed-allreduce traffic still queued on
  // the engine's comm/decode streams (the compressed path itself chains the
  // other direction through start_ev_, so ordering is total).
  for (Engine* e : {engine_.get(), intra_engine_.get(),
                    cross_engine_.get()}) {
    if (!e) continue;
    for (hipStream_t s : {e->comm_stream(), e->deq_stream()}) {
      CGX_HIP_CHECK(hipEventRecord(start_ev_, s));
      CGX_HIP_CHECK(hipStreamWaitEvent(stream_->stream(), start_ev_, 0));
    }
  }
  c10::hip::HIPStreamGuardMasqueradingAsCUDA sguard(stream_->unwrap());
  fn(stream_->stream());
  for (const auto& t : outputs) {
    if (t.defined() && t.is_cuda()) {
      c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::
          recordStreamMasqueradingAsCUDA(t.storage().data_ptr(), *stream_);
    }
  }
  auto work = c10::make_intrusive<WorkCGX>(rank_, op, device,
                                           std::move(outputs),
                                           c10d::opTypeToString(op).c_str());
  work->recordEnd(*stream_);
  return work;
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::allreduce(
    std::vector<at::Tensor>& tensors, const c10d::AllreduceOptions& opts) {
  check_single(tensors);
  at::Tensor t = tensors[0];
  if (!t.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->allreduce(tensors, opts);
  }
  const bool compressible =
      opts.reduceOp == c10d::ReduceOp::SUM &&
      (t.scalar_type() == at::kFloat || t.scalar_type() == at::kHalf ||
       t.scalar_type() == at::kBFloat16);
  auto op = opts.reduceOp;
  if (compressible) {
    // pipelined path: engine fans work over quantize/comm/deq streams and
    // returns the stream carrying the final op; the Work's end event and
    // future live there so the NEXT bucket's quantize can overlap this
    // bucket's xGMI traffic.
    std::lock_guard<std::mutex> lock(mu_);
    lazyInit(t.device());
    check_async_error(comm_);
    if (hierarchical_) {
      check_async_error(intra_comm_);
      check_async_error(cross_comm_);
    }
    c10::hip::HIPGuardMasqueradingAsCUDA dguard(device_index_);
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_index_);
    CGX_HIP_CHECK(hipEventRecord(start_ev_, cur.stream()));
    CGX_HIP_CHECK(hipStreamWaitEvent(stream_->stream(), start_ev_, 0));
    hipStream_t fin = stream_->stream();
    if (size_ > 1) {
      const char* ib = std::getenv("CGX_INTRA_BROADCAST");
      const bool hier = hierarchical_ && !(ib && std::strcmp(ib, "0") == 0);
      if (hier) {
        // intra-node compressed allreduce over xGMI, cross-node reduction on
        // node leaders, then intra broadcast of the result.  The layer
        // registry is consulted exactly ONCE per bucket so the cursor stays
        // in lockstep on every rank (leaders run two engine passes).
        cgx::Registry::BucketInfo info;
        const bool matched =
            cgx::Registry::get().next(t.numel(), t.data_ptr(), &info);
        fin = intra_engine_->allreduce(t, intra_tr_.get(),
                                       stream_->stream(), &info, matched);
        if (topo_.local_rank == 0)
          fin = cross_engine_->allreduce(t, cross_tr_.get(), fin, &info,
                                         matched);
        fin = intra_engine_->broadcast(t, /*root=*/0, intra_tr_.get(), fin);
      } else {
        fin = engine_->allreduce(t, tr_.get(), stream_->stream());
      }
    }
    auto fin_masq = c10::hip::getStreamFromExternalMasqueradingAsCUDA(
        fin, device_index_);
    c10::hip::HIPStreamGuardMasqueradingAsCUDA sguard(fin_masq.unwrap());
    std::vector<hipStream_t> touched{stream_->stream()};
    for (Engine* e : {engine_.get(), intra_engine_.get(),
                      cross_engine_.get()}) {
      if (!e) continue;
      touched.push_back(e->comm_stream());
      touched.push_back(e->deq_stream());
    }
    for (hipStream_t s : touched) {
      if (!s) continue;
      c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::
          recordStreamMasqueradingAsCUDA(
              t.storage().data_ptr(),
              c10::hip::getStreamFromExternalMasqueradingAsCUDA(
                  s, device_index_));
    }
    auto work = c10::make_intrusive<WorkCGX>(rank_, c10d::OpType::ALLREDUCE,
                                             t.device(), tensors,
                                             "cgx:allreduce_compressed");
    work->recordEnd(fin_masq);
    return work;
  }
  return collective(tensors, t.device(), c10d::OpType::ALLREDUCE,
                    [this, t, op](hipStream_t s) {
                      if (size_ == 1) return;
                      CGX_NCCL_CHECK(ncclAllReduce(
                          t.data_ptr(), t.data_ptr(), t.numel(),
                          nccl_dtype(t), to_nccl_op(op), comm_, s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::allreduce_coalesced(
    std::vector<at::Tensor>& tensors,
    const c10d::AllreduceCoalescedOptions& opts) {
  TORCH_CHECK(!tensors.empty());
  if (!tensors[0].is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->allreduce_coalesced(tensors, opts);
  }
  auto op = opts.reduceOp;
  return collective(tensors, tensors[0].device(), c10d::OpType::COALESCED,
                    [this, tensors, op](hipStream_t s) {
                      if (size_ == 1) return;
                      CGX_NCCL_CHECK(ncclGroupStart());
                      for (const auto& t : tensors) {
                        CGX_NCCL_CHECK(ncclAllReduce(
                            t.data_ptr(), t.data_ptr(), t.numel(),
                            nccl_dtype(t), to_nccl_op(op), comm_, s));
                      }
                      CGX_NCCL_CHECK(ncclGroupEnd());
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::broadcast(
    std::vector<at::Tensor>& tensors, const c10d::BroadcastOptions& opts) {
  check_single(tensors);
  at::Tensor t = tensors[0];
  if (!t.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->broadcast(tensors, opts);
  }
  const int root = static_cast<int>(opts.rootRank);
  return collective(tensors, t.device(), c10d::OpType::BROADCAST,
                    [this, t, root](hipStream_t s) {
                      if (size_ == 1) return;
                      CGX_NCCL_CHECK(ncclBroadcast(t.data_ptr(), t.data_ptr(),
                                                   t.numel(), nccl_dtype(t),
                                                   root, comm_, s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::reduce(
    std::vector<at::Tensor>& tensors, const c10d::ReduceOptions& opts) {
  check_single(tensors);
  at::Tensor t = tensors[0];
  if (!t.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->reduce(tensors, opts);
  }
  const int root = static_cast<int>(opts.rootRank);
  auto op = opts.reduceOp;
  return collective(tensors, t.device(), c10d::OpType::REDUCE,
                    [this, t, root, op](hipStream_t s) {
                      if (size_ == 1) return;
                      CGX_NCCL_CHECK(ncclReduce(t.data_ptr(), t.data_ptr(),
                                                t.numel(), nccl_dtype(t),
                                                to_nccl_op(op), root, comm_,
                                                s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::allgather(
    std::vector<std::vector<at::Tensor>>& outputs,
    std::vector<at::Tensor>& inputs, const c10d::AllgatherOptions& opts) {
  check_single(inputs);
  TORCH_CHECK(outputs.size() == 1 && (int)outputs[0].size() == size_);
  at::Tensor in = inputs[0];
  if (!in.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->allgather(outputs, inputs, opts);
  }
  auto outs = outputs[0];
  return collective(outs, in.device(), c10d::OpType::ALLGATHER,
                    [this, in, outs](hipStream_t s) {
                      CGX_NCCL_CHECK(ncclGroupStart());
                      for (int p = 0; p < size_; p++) {
                        if (p == rank_) continue;
                        CGX_NCCL_CHECK(ncclSend(in.data_ptr(), in.numel(),
                                                nccl_dtype(in), p, comm_, s));
                        CGX_NCCL_CHECK(ncclRecv(outs[p].data_ptr(),
                                                outs[p].numel(),
                                                nccl_dtype(outs[p]), p, comm_,
                                                s));
                      }
                      CGX_NCCL_CHECK(ncclGroupEnd());
                      CGX_HIP_CHECK(hipMemcpyAsync(
                          outs[rank_].data_ptr(), in.data_ptr(),
                          in.numel() * in.element_size(),
                          hipMemcpyDeviceToDevice, s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::_allgather_base(
    at::Tensor& output, at::Tensor& input,
    const c10d::AllgatherOptions& opts) {
  if (!input.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->_allgather_base(output, input, opts);
  }
  TORCH_CHECK(output.numel() == input.numel() * size_);
  at::Tensor in = input, out = output;
  return collective({output}, input.device(), c10d::OpType::_ALLGATHER_BASE,
                    [this, in, out](hipStream_t s) {
                      CGX_NCCL_CHECK(ncclAllGather(in.data_ptr(),
                                                   out.data_ptr(), in.numel(),
                                                   nccl_dtype(in), comm_, s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::gather(
    std::vector<std::vector<at::Tensor>>& outputs,
    std::vector<at::Tensor>& inputs, const c10d::GatherOptions& opts) {
  check_single(inputs);
  at::Tensor in = inputs[0];
  if (!in.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->gather(outputs, inputs, opts);
  }
  const int root = static_cast<int>(opts.rootRank);
  std::vector<at::Tensor> outs =
      rank_ == root ? outputs[0] : std::vector<at::Tensor>{};
  return collective(outs.empty() ? inputs : outs, in.device(),
                    c10d::OpType::GATHER, [this, in, outs, root](hipStream_t s) {
                      if (rank_ == root) {
                        CGX_NCCL_CHECK(ncclGroupStart());
                        for (int p = 0; p < size_; p++) {
                          if (p == root) continue;
                          CGX_NCCL_CHECK(ncclRecv(outs[p].data_ptr(),
                                                  outs[p].numel(),
                                                  nccl_dtype(outs[p]), p,
                                                  comm_, s));
                        }
                        CGX_NCCL_CHECK(ncclGroupEnd());
                        CGX_HIP_CHECK(hipMemcpyAsync(
                            outs[root].data_ptr(), in.data_ptr(),
                            in.numel() * in.element_size(),
                            hipMemcpyDeviceToDevice, s));
                      } else {
                        CGX_NCCL_CHECK(ncclSend(in.data_ptr(), in.numel(),
                                                nccl_dtype(in), root, comm_,
                                                s));
                      }
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::scatter(
    std::vector<at::Tensor>& outputs,
    std::vector<std::vector<at::Tensor>>& inputs,
    const c10d::ScatterOptions& opts) {
  check_single(outputs);
  at::Tensor out = outputs[0];
  if (!out.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->scatter(outputs, inputs, opts);
  }
  const int root = static_cast<int>(opts.rootRank);
  std::vector<at::Tensor> ins =
      rank_ == root ? inputs[0] : std::vector<at::Tensor>{};
  return collective(outputs, out.device(), c10d::OpType::SCATTER,
                    [this, out, ins, root](hipStream_t s) {
                      if (rank_ == root) {
                        CGX_NCCL_CHECK(ncclGroupStart());
                        for (int p = 0; p < size_; p++) {
                          if (p == root) continue;
                          CGX_NCCL_CHECK(ncclSend(ins[p].data_ptr(),
                                                  ins[p].numel(),
                                                  nccl_dtype(ins[p]), p,
                                                  comm_, s));
                        }
                        CGX_NCCL_CHECK(ncclGroupEnd());
                        CGX_HIP_CHECK(hipMemcpyAsync(
                            out.data_ptr(), ins[root].data_ptr(),
                            out.numel() * out.element_size(),
                            hipMemcpyDeviceToDevice, s));
                      } else {
                        CGX_NCCL_CHECK(ncclRecv(out.data_ptr(), out.numel(),
                                                nccl_dtype(out), root, comm_,
                                                s));
                      }
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::reduce_scatter(
    std::vector<at::Tensor>& outputs,
    std::vector<std::vector<at::Tensor>>& inputs,
    const c10d::ReduceScatterOptions& opts) {
  check_single(outputs);
  at::Tensor out = outputs[0];
  if (!out.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->reduce_scatter(outputs, inputs, opts);
  }
  TORCH_CHECK((int)inputs[0].size() == size_);
  // flatten inputs into one contiguous buffer, then ncclReduceScatter
  at::Tensor flat = at::cat(inputs[0]).contiguous();
  auto op = opts.reduceOp;
  return collective(outputs, out.device(), c10d::OpType::REDUCE_SCATTER,
                    [this, flat, out, op](hipStream_t s) {
                      // flat is a temporary destroyed when this call returns;
                      // keep its storage out of the allocator's reuse pool
                      // until the async collective on the side stream is done
                      c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::
                          recordStreamMasqueradingAsCUDA(
                              flat.storage().data_ptr(),
                              c10::hip::getStreamFromExternalMasqueradingAsCUDA(
                                  s, flat.device().index()));
                      CGX_NCCL_CHECK(ncclReduceScatter(
                          flat.data_ptr(), out.data_ptr(), out.numel(),
                          nccl_dtype(out), to_nccl_op(op), comm_, s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::_reduce_scatter_base(
    at::Tensor& output, at::Tensor& input,
    const c10d::ReduceScatterOptions& opts) {
  if (!input.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->_reduce_scatter_base(output, input, opts);
  }
  TORCH_CHECK(input.numel() == output.numel() * size_);
  at::Tensor in = input, out = output;
  auto op = opts.reduceOp;
  return collective({output}, input.device(),
                    c10d::OpType::_REDUCE_SCATTER_BASE,
                    [this, in, out, op](hipStream_t s) {
                      CGX_NCCL_CHECK(ncclReduceScatter(
                          in.data_ptr(), out.data_ptr(), out.numel(),
                          nccl_dtype(out), to_nccl_op(op), comm_, s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::alltoall_base(
    at::Tensor& output, at::Tensor& input,
    std::vector<int64_t>& outputSplitSizes,
    std::vector<int64_t>& inputSplitSizes, const c10d::AllToAllOptions&) {
  if (!input.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    std::vector<int64_t> os = outputSplitSizes, is = inputSplitSizes;
    c10d::AllToAllOptions o;
    return cpu_->alltoall_base(output, input, os, is, o);
  }
  at::Tensor in = input, out = output;
  std::vector<int64_t> osplit = outputSplitSizes, isplit = inputSplitSizes;
  if (osplit.empty()) {
    TORCH_CHECK(out.numel() % size_ == 0 && in.numel() % size_ == 0,
                "cgx: alltoall_base tensor not divisible by world size");
    osplit.assign(size_, out.numel() / size_);
    isplit.assign(size_, in.numel() / size_);
  } else {
    // splits are in units of dim-0 rows
    int64_t orow = out.dim() > 0 && out.size(0) > 0 ? out.numel() / out.size(0) : 1;
    int64_t irow = in.dim() > 0 && in.size(0) > 0 ? in.numel() / in.size(0) : 1;
    for (auto& v : osplit) v *= orow;
    for (auto& v : isplit) v *= irow;
  }
  return collective({output}, input.device(), c10d::OpType::ALLTOALL_BASE,
                    [this, in, out, osplit, isplit](hipStream_t s) {
                      const int es = in.element_size();
                      std::vector<int64_t> ooff(size_, 0), ioff(size_, 0);
                      for (int p = 1; p < size_; p++) {
                        ooff[p] = ooff[p - 1] + osplit[p - 1];
                        ioff[p] = ioff[p - 1] + isplit[p - 1];
                      }
                      CGX_NCCL_CHECK(ncclGroupStart());
                      for (int p = 0; p < size_; p++) {
                        if (p == rank_) continue;
                        CGX_NCCL_CHECK(ncclSend(
                            static_cast<char*>(in.data_ptr()) + ioff[p] * es,
                            isplit[p], nccl_dtype(in), p, comm_, s));
                        CGX_NCCL_CHECK(ncclRecv(
                            static_cast<char*>(out.data_ptr()) + ooff[p] * es,
                            osplit[p], nccl_dtype(out), p, comm_, s));
                      }
                      CGX_NCCL_CHECK(ncclGroupEnd());
                      CGX_HIP_CHECK(hipMemcpyAsync(
                          static_cast<char*>(out.data_ptr()) + ooff[rank_] * es,
                          static_cast<char*>(in.data_ptr()) + ioff[rank_] * es,
                          isplit[rank_] * es, hipMemcpyDeviceToDevice, s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::alltoall(
    std::vector<at::Tensor>& outputs, std::vector<at::Tensor>& inputs,
    const c10d::AllToAllOptions&) {
  TORCH_CHECK((int)outputs.size() == size_ && (int)inputs.size() == size_);
  if (!inputs[0].is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    // gloo has no alltoall; compose it from ordered pairwise gloo p2p
    // (deadlock-free: the lower rank of each pair sends first; the
    // reference used MPI_Alltoall here, ProcessGroupCGX.cc:659)
    outputs[rank_].copy_(inputs[rank_]);
    for (int p = 0; p < size_; p++) {
      if (p == rank_) continue;
      std::vector<at::Tensor> st{inputs[p].contiguous()};
      std::vector<at::Tensor> rt{outputs[p]};
      if (rank_ < p) {
        cpu_->send(st, p, 0)->wait();
        cpu_->recv(rt, p, 0)->wait();
      } else {
        cpu_->recv(rt, p, 0)->wait();
        cpu_->send(st, p, 0)->wait();
      }
    }
    c10d::BarrierOptions bo;
    return cpu_->barrier(bo);
  }
  std::vector<at::Tensor> ins = inputs, outs = outputs;
  return collective(outputs, inputs[0].device(), c10d::OpType::ALLTOALL,
                    [this, ins, outs](hipStream_t s) {
                      CGX_NCCL_CHECK(ncclGroupStart());
                      for (int p = 0; p < size_; p++) {
                        if (p == rank_) continue;
                        CGX_NCCL_CHECK(ncclSend(ins[p].data_ptr(),
                                                ins[p].numel(),
                                                nccl_dtype(ins[p]), p, comm_,
                                                s));
                        CGX_NCCL_CHECK(ncclRecv(outs[p].data_ptr(),
                                                outs[p].numel(),
                                                nccl_dtype(outs[p]), p, comm_,
                                                s));
                      }
                      CGX_NCCL_CHECK(ncclGroupEnd());
                      CGX_HIP_CHECK(hipMemcpyAsync(
                          outs[rank_].data_ptr(), ins[rank_].data_ptr(),
                          ins[rank_].numel() * ins[rank_].element_size(),
                          hipMemcpyDeviceToDevice, s));
                    });
}

namespace {
// CGX_P2P_ANYSOURCE=1 enables MPI_ANY_SOURCE-style GPU receives (reference
// ProcessGroupCGX.cc:765-785): RCCL p2p needs a known peer, so every GPU
// send also posts a tiny gloo "announcement" carrying the source rank, and
// every GPU receive consumes one announcement first (recvAnysource learns
// its peer from it).  Off by default — it adds a host-side gloo message to
// every GPU send, and mixing announced and plain p2p would desynchronize
// the pairing, so the flag must be set consistently on ALL ranks.
bool p2p_anysource() {
  static const bool v = env_flag("CGX_P2P_ANYSOURCE");
  return v;
}
}  // namespace

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::send(
    std::vector<at::Tensor>& tensors, int dstRank, int tag) {
  check_single(tensors);
  at::Tensor t = tensors[0];
  if (!t.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->send(tensors, dstRank, tag);
  }
  if (p2p_anysource()) {
    TORCH_CHECK(cpu_, "cgx: CGX_P2P_ANYSOURCE requires the CPU delegate");
    std::vector<at::Tensor> ann{
        at::full({1}, rank_, at::TensorOptions().dtype(at::kInt))};
    // fire-and-forget: waiting here would host-block until the peer posts
    // its receive (gloo p2p is unbuffered), deadlocking send/send-then-
    // recv/recv patterns that plain async sends allow.  The Work (and the
    // announcement tensor it references) stays queued until completed.
    {
      std::lock_guard<std::mutex> g(ann_mu_);
      pending_ann_.push_back(cpu_->send(ann, dstRank, tag));
      while (!pending_ann_.empty() && pending_ann_.front()->isCompleted())
        pending_ann_.pop_front();
    }
  }
  return collective(tensors, t.device(), c10d::OpType::SEND,
                    [this, t, dstRank](hipStream_t s) {
                      CGX_NCCL_CHECK(ncclSend(t.data_ptr(), t.numel(),
                                              nccl_dtype(t), dstRank, comm_,
                                              s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::recv(
    std::vector<at::Tensor>& tensors, int srcRank, int tag) {
  check_single(tensors);
  at::Tensor t = tensors[0];
  if (!t.is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->recv(tensors, srcRank, tag);
  }
  if (p2p_anysource()) {
    // consume this sender's announcement to keep the pairing in lockstep
    std::vector<at::Tensor> ann{
        at::zeros({1}, at::TensorOptions().dtype(at::kInt))};
    cpu_->recv(ann, srcRank, tag)->wait();
  }
  return collective(tensors, t.device(), c10d::OpType::RECV,
                    [this, t, srcRank](hipStream_t s) {
                      CGX_NCCL_CHECK(ncclRecv(t.data_ptr(), t.numel(),
                                              nccl_dtype(t), srcRank, comm_,
                                              s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::recvAnysource(
    std::vector<at::Tensor>& tensors, int tag) {
  check_single(tensors);
  if (!tensors[0].is_cuda()) {
    TORCH_CHECK(cpu_, "cgx: no CPU delegate backend available");
    return cpu_->recvAnysource(tensors, tag);
  }
  TORCH_CHECK(p2p_anysource(),
              "cgx: recvAnysource for GPU tensors requires "
              "CGX_P2P_ANYSOURCE=1 on every rank (RCCL p2p needs a known "
              "source; the flag adds a gloo source announcement to each "
              "GPU send)");
  at::Tensor t = tensors[0];
  // learn the source from the announcement, then RCCL-receive from it
  std::vector<at::Tensor> ann{
      at::zeros({1}, at::TensorOptions().dtype(at::kInt))};
  auto w = cpu_->recvAnysource(ann, tag);
  w->wait();
  const int src = w->sourceRank() >= 0
                      ? w->sourceRank()
                      : ann[0].item<int>();
  TORCH_CHECK(src >= 0 && src < size_ && src != rank_,
              "cgx: bad announced source rank ", src);
  return collective(tensors, t.device(), c10d::OpType::RECVANYSOURCE,
                    [this, t, src](hipStream_t s) {
                      CGX_NCCL_CHECK(ncclRecv(t.data_ptr(), t.numel(),
                                              nccl_dtype(t), src, comm_,
                                              s));
                    });
}

c10::intrusive_ptr<c10d::Work> ProcessGroupCGX::barrier(
    const c10d::BarrierOptions& opts) {
  // Host-level rendezvous; if the GPU side is active, drain the side stream
  // first so the barrier also orders pending GPU collectives.
  if (comm_) {
    std::lock_guard<std::mutex> lock(mu_);
    CGX_HIP_CHECK(hipStreamSynchronize(stream_->stream()));
    for (Engine* e : {engine_.get(), intra_engine_.get(),
                      cross_engine_.get()}) {
      if (!e) continue;
      CGX_HIP_CHECK(hipStreamSynchronize(e->comm_stream()));
      CGX_HIP_CHECK(hipStreamSynchronize(e->deq_stream()));
    }
  }
  if (cpu_) {
    c10d::BarrierOptions o = opts;
    o.device_ids.clear();
    return cpu_->barrier(o);
  }
  TORCH_CHECK(comm_, "cgx: barrier before any collective and no CPU delegate");
  // RCCL barrier: 1-element allreduce + host sync
  auto t = at::zeros({1}, at::TensorOptions()
                              .dtype(at::kFloat)
                              .device(at::kCUDA, device_index_));
  std::vector<at::Tensor> ts{t};
  c10d::AllreduceOptions aopts;
  auto w = allreduce(ts, aopts);
  w->wait(std::chrono::milliseconds(0));
  CGX_HIP_CHECK(hipStreamSynchronize(stream_->stream()));
  return w;
}

}  // namespace cgx
