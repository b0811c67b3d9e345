// TCP transport Context: the pair set of one communicator plus the
// recv-from-any arbitration and self-loopback queues.
//
// Capability parity with reference gloo/transport/tcp/context.cc:48-360
// (store-based full-mesh bootstrap, recv-from-any tally, exception
// fan-out). Re-designed: a single context mutex guards all protocol
// state (see transport/tcp/pair.h header comment).
#pragma once

#include <condition_variable>
#include <list>
#include <memory>
#include <mutex>

#include "common/store.h"
#include "transport/tcp/pair.h"
#include "transport/transport.h"

namespace glooamd {
namespace tcp {

class TcpUnboundBuffer : public transport::UnboundBuffer {
 public:
  TcpUnboundBuffer(TcpContext* ctx, void* ptr, size_t size)
      : UnboundBuffer(ptr, size), ctx_(ctx) {}
  ~TcpUnboundBuffer() override;

  void send(int dstRank, uint64_t slot, size_t offset, size_t nbytes) override;
  void recv(int srcRank, uint64_t slot, size_t offset, size_t nbytes) override;
  void recv(
      const std::vector<int>& srcRanks,
      uint64_t slot,
      size_t offset,
      size_t nbytes) override;
  bool waitRecv(int* srcRank, std::chrono::milliseconds timeout) override;
  bool tryWaitRecv(int* srcRank, std::chrono::milliseconds timeout) override;
  bool waitSend(std::chrono::milliseconds timeout) override;
  void abortWaitRecv() override;
  void abortWaitSend() override;

 private:
  friend class TcpContext;
  friend class TcpPair;
  TcpContext* ctx_;
  // guarded by context mutex
  std::deque<int> recvCompletions_; // src ranks
  uint64_t sendCompletions_{0};
  bool abortRecv_{false};
  bool abortSend_{false};
  std::exception_ptr error_;
};

class TcpContext : public transport::Context,
                   public std::enable_shared_from_this<TcpContext> {
 public:
  TcpContext(std::shared_ptr<TcpDevice> device, int rank, int size);
  ~TcpContext() override;

  transport::Pair* getPair(int rank) override;
  transport::Pair* createPair(int rank) override;
  std::unique_ptr<transport::UnboundBuffer> createUnboundBuffer(
      void* ptr,
      size_t size) override;

  // Full-mesh bootstrap through a rendezvous store: publish one key per
  // rank containing the per-peer pair addresses, then connect each pair
  // (lower rank dials).  Reference protocol: tcp/context.cc:48-165.
  void connectFullMesh(IStore& store);

  // Close all pairs and fail every pending operation with IoException.
  void signalException(const std::string& msg);
  void dumpStateLocked(const char* why); // GLOO_AMD_DUMP_ON_TIMEOUT aid

  TcpDevice* device() {
    return device_.get();
  }

 protected:
  // For transport subclasses (tls) overriding createPair.
  void setPair(int rank, std::unique_ptr<TcpPair> pair) {
    pairs_[rank] = std::move(pair);
  }

 private:
  friend class TcpPair;
  friend class TcpUnboundBuffer;
  friend class TcpBuffer;

  struct AnyRecv {
    TcpUnboundBuffer* buf;
    uint64_t slot;
    size_t offset;
    size_t nbytes;
    std::vector<int> srcRanks;
  };

  // Called on SEND_READY arrival creating an excess notified send.
  // Returns true if an any-recv waiter matched (recv posted inside).
  bool offerSendReadyLocked(TcpPair* pair, uint64_t slot);
  // Pop the oldest any-recv waiter matching (slot, srcRank) into *out.
  bool claimAnyRecvLocked(int srcRank, uint64_t slot, AnyRecv* out);
  // Consume one unclaimed send-ready entry for (slot, rank) if present.
  bool consumeUnclaimedLocked(uint64_t slot, int rank);

  // Self-loopback (rank -> itself) send/recv matching.
  void selfSendLocked(TcpUnboundBuffer* buf, uint64_t slot, size_t off, size_t nb);
  void selfRecvLocked(TcpUnboundBuffer* buf, uint64_t slot, size_t off, size_t nb);

  std::shared_ptr<TcpDevice> device_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::vector<std::unique_ptr<TcpPair>> pairs_;

  std::unordered_map<uint64_t, std::deque<int>> unclaimedSendReady_;
  std::list<AnyRecv> anyRecvs_;

  struct SelfOp {
    TcpUnboundBuffer* buf;
    size_t offset;
    size_t nbytes;
  };
  std::unordered_map<uint64_t, std::deque<SelfOp>> selfSends_;
  std::unordered_map<uint64_t, std::deque<SelfOp>> selfRecvs_;
};

} // namespace tcp
} // namespace glooamd
