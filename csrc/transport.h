// Byte-transport seam under the compression engine.
//
// The engine's reducers (SRA / Ring / debug all-to-all, engine.cc) speak a
// tiny transport vocabulary: a grouped p2p exchange (ncclGroupStart/End
// semantics), a byte broadcast, and an uncompressed SUM-allreduce fallback.
// Production uses RcclTransport (grouped ncclSend/ncclRecv over xGMI — the
// MI355X replacement for the reference's MPI/SHM/NCCL communicator trio,
// /root/reference/src/common/communicator.h:28-52).  LoopbackTransport lets
// N engine instances inside ONE process exchange through device-to-device
// copies on a single GPU, so the real multi-rank orchestration (streams,
// events, staging reuse, kernels) can execute and be tested at ws>=2 with
// one MI355X — RCCL itself refuses two ranks on one device.
#pragma once

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <condition_variable>
#include <cstdint>
#include <deque>
#include <map>
#include <mutex>
#include <vector>

#include "compress.h"

namespace cgx {

struct Transport {
  struct Op {
    void* ptr;
    int64_t bytes;
    int peer;
  };
  virtual ~Transport() = default;

  // All sends and recvs logically concurrent, stream-ordered on `stream`:
  // work queued after the call on `stream` sees every recv completed and
  // every send buffer free for reuse.
  virtual void exchange(const std::vector<Op>& sends,
                        const std::vector<Op>& recvs, hipStream_t stream) = 0;

  // Broadcast `count` elements of element size `esize` from `root`.
  virtual void broadcast(void* ptr, int64_t count, int esize,
                         ncclDataType_t dt, int root, hipStream_t stream) = 0;

  // In-place SUM-allreduce (the uncompressed fallback path).
  virtual void allreduce_sum(void* ptr, int64_t count, ncclDataType_t dt,
                             hipStream_t stream) = 0;

  // Group several allreduce_sum calls (RCCL: one ncclGroup).
  virtual void allreduce_sum_group(
      const std::vector<std::pair<void*, int64_t>>& bufs, ncclDataType_t dt,
      hipStream_t stream) = 0;
};

class RcclTransport final : public Transport {
 public:
  RcclTransport(ncclComm_t comm, int rank, int size) : comm_(comm) {
    (void)rank;
    (void)size;
  }
  void exchange(const std::vector<Op>& sends, const std::vector<Op>& recvs,
                hipStream_t stream) override;
  void broadcast(void* ptr, int64_t count, int esize, ncclDataType_t dt,
                 int root, hipStream_t stream) override;
  void allreduce_sum(void* ptr, int64_t count, ncclDataType_t dt,
                     hipStream_t stream) override;
  void allreduce_sum_group(const std::vector<std::pair<void*, int64_t>>& bufs,
                           ncclDataType_t dt, hipStream_t stream) override;
  ncclComm_t comm() const { return comm_; }

 private:
  ncclComm_t comm_;
};

// Shared rendezvous for N loopback transports in one process.  Each
// (src,dst) pair has a FIFO mailbox.  exchange() performs a host-side
// three-phase handshake per call:
//   1. post every send (event recorded on the sender's stream),
//   2. resolve every recv (host-block until the matching send is posted,
//      then stream-wait + hipMemcpyAsync D2D on the receiver's stream and
//      record a "consumed" event),
//   3. host-block until every posted send is consumed, then stream-wait the
//      consumed events so the sender's buffer is provably reusable.
// Deadlock-free as long as all ranks issue the same exchange sequence —
// which the engine guarantees (identical chunk schedules).
class LoopbackHub {
 public:
  explicit LoopbackHub(int world) : world_(world) {}
  ~LoopbackHub();

  struct Msg {
    const void* src = nullptr;
    int64_t bytes = 0;
    hipEvent_t ready = nullptr;     // recorded on sender's stream
    hipEvent_t consumed = nullptr;  // recorded on receiver's stream
    bool done = false;              // consumed event recorded
  };

  Msg* post_send(int src, int dst, const void* ptr, int64_t bytes,
                 hipEvent_t ready);
  Msg* take_send(int src, int dst);         // blocks until posted
  void mark_consumed(Msg* m, hipEvent_t consumed);
  void wait_consumed(Msg* m);               // blocks until marked

  hipEvent_t new_event();  // owned by the hub, freed at teardown

  // Poison the hub: every blocked or future wait throws instead of hanging
  // (used when one rank thread fails so its peers can unwind).
  void abort();

  int world() const { return world_; }

 private:
  int world_;
  bool aborted_ = false;
  std::mutex mu_;
  std::condition_variable cv_;
  std::map<std::pair<int, int>, std::deque<Msg*>> boxes_;  // pending sends
  std::vector<Msg*> msgs_;
  std::vector<hipEvent_t> events_;
};

class LoopbackTransport final : public Transport {
 public:
  LoopbackTransport(LoopbackHub* hub, int rank)
      : hub_(hub), rank_(rank), size_(hub->world()) {}
  void exchange(const std::vector<Op>& sends, const std::vector<Op>& recvs,
                hipStream_t stream) override;
  void broadcast(void* ptr, int64_t count, int esize, ncclDataType_t dt,
                 int root, hipStream_t stream) override;
  void allreduce_sum(void* ptr, int64_t count, ncclDataType_t dt,
                     hipStream_t stream) override;
  void allreduce_sum_group(const std::vector<std::pair<void*, int64_t>>& bufs,
                           ncclDataType_t dt, hipStream_t stream) override;

 private:
  LoopbackHub* hub_;
  int rank_, size_;
};

}  // namespace cgx
