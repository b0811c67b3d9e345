// TCP Pair: one connected point-to-point channel.
//
// Capability parity with reference gloo/transport/tcp/pair.{h,cc} (the
// 48-byte preamble wire protocol, send/recv-ready handshake for unbound
// buffers, bound-buffer one-sided writes, partial read/write state
// machines, exception fan-out) — re-designed from scratch:
//
//  * 32-byte preamble {opcode, srcRank, slot, length, roffset}; opcodes
//    BOUND_DATA / UNBOUND_DATA / RECV_READY / SEND_READY / EAGER_DATA.
//  * Unbound rendezvous: recv() posts RECV_READY; send() consumes a
//    recv-credit (payload immediately), pushes the payload eagerly when
//    small with no older queued send (receiver stashes if no recv is
//    posted yet), or queues + posts SEND_READY.
//  * recv-from-any arbitration by cumulative counters: a SEND_READY is
//    "available" for any-recv matching iff sendReadySeen > recvsPosted
//    on that (pair, slot) — see TcpContext.
//  * All protocol state guarded by the owning TcpContext's single mutex;
//    socket reads happen only on the device loop thread, socket writes on
//    whichever thread holds the lock (inline fast path + EPOLLOUT drain).
#pragma once

#include <sys/uio.h>

#include <atomic>
#include <deque>
#include <string>
#include <unordered_map>
#include <vector>

#include "transport/tcp/address.h"
#include "transport/tcp/loop.h"
#include "transport/transport.h"

namespace glooamd {
namespace tcp {

class TcpContext;
class TcpDevice;
class TcpPair;
class TcpUnboundBuffer;

enum WireOp : uint32_t {
  BOUND_DATA = 1,
  UNBOUND_DATA = 2,
  RECV_READY = 3,
  SEND_READY = 4,
  // Small unbound send pushed without rendezvous: payload travels
  // immediately; receiver delivers to the front pending recv, a matching
  // any-recv waiter, or stashes until a recv is posted.
  EAGER_DATA = 5,
};

struct Preamble {
  uint32_t opcode;
  uint32_t srcRank;
  uint64_t slot;
  uint64_t length; // payload bytes (or op size for READY messages)
  uint64_t roffset; // BOUND_DATA: offset into remote registered buffer
};
static_assert(sizeof(Preamble) == 32, "wire preamble must be 32 bytes");

// Bound buffer: pre-registered region addressed by slot; sends are
// one-sided-style writes into the peer's same-slot region.
class TcpBuffer : public transport::Buffer {
 public:
  TcpBuffer(TcpPair* pair, uint64_t slot, void* ptr, size_t size, bool isSend)
      : Buffer(slot, ptr, size), pair_(pair), isSend_(isSend) {}
  ~TcpBuffer() override;

  void send(size_t offset, size_t length, size_t roffset = 0) override;
  void waitRecv() override;
  void waitSend() override;

 private:
  friend class TcpPair;
  TcpPair* pair_;
  bool isSend_;
  // guarded by context mutex
  uint64_t recvCount_{0};
  uint64_t sendCount_{0};
  std::exception_ptr error_;
};

class TcpPair : public transport::Pair, public Handler {
 public:
  TcpPair(TcpContext* ctx, TcpDevice* dev, int peerRank);
  ~TcpPair() override;

  const transport::Address& address() const override {
    return self_;
  }
  void connect(const std::vector<char>& peerAddressBytes) override;
  void close() override;

  std::unique_ptr<transport::Buffer>
  createSendBuffer(uint64_t slot, void* ptr, size_t size) override;
  std::unique_ptr<transport::Buffer>
  createRecvBuffer(uint64_t slot, void* ptr, size_t size) override;

  bool isConnected() const override {
    return state_ == CONNECTED;
  }
  std::string str() const override;

  int peerRank() const {
    return peerRank_;
  }

  // --- called by TcpContext / TcpUnboundBuffer (context mutex held) ---
  void ubufSend(TcpUnboundBuffer* buf, uint64_t slot, size_t off, size_t nb);
  // fromClaim: the caller already consumed the matching unclaimed
  // send-ready entry (recv-from-any path).
  void postRecv(
      TcpUnboundBuffer* buf,
      uint64_t slot,
      size_t off,
      size_t nb,
      bool fromClaim);
  void boundSend(TcpBuffer* buf, size_t off, size_t len, size_t roff);
  // Deliver the oldest stashed eager/credit payload for slot into buf, if
  // any. Completes the recv locally (no RECV_READY). Context mutex held.
  bool takeStashLocked(TcpUnboundBuffer* buf, uint64_t slot, size_t off, size_t nb);
  void failLocked(std::exception_ptr e); // fan exception to all pending ops
  bool referencesBuffer(const void* buf) const; // in-flight rx/tx check
  void detachBuffer(TcpBuffer* buf); // bound buffer destructor support
  void detachUnbound(TcpUnboundBuffer* buf); // drop queued (unsent) ops

  // loop thread entry
  void handleEvents(uint32_t events) override;

 protected:
  // Byte-level I/O, overridable by the TLS pair. Semantics match
  // read(2)/writev(2): >0 bytes, 0 EOF (read), -1 with errno
  // (EAGAIN/EWOULDBLOCK = retry later).
  virtual ssize_t ioRead(char* buf, size_t len);
  virtual ssize_t ioWritev(const struct iovec* iov, int iovcnt);
  // Called from connect() once the raw socket is established (blocking
  // allowed; fd_ is still in blocking mode for the initiator path).
  virtual void ioHandshake(bool initiator) {}
  virtual void ioClose() {}

  int fd() const {
    return fd_;
  }

 private:
  struct TxOp {
    Preamble pre;
    size_t preWritten{0};
    const char* payload{nullptr};
    size_t payloadLen{0};
    size_t payloadWritten{0};
    TcpBuffer* bbuf{nullptr};
    TcpUnboundBuffer* ubuf{nullptr};
  };

  static size_t eagerMaxBytes(); // GLOO_AMD_EAGER_MAX, default 8 KiB
  void checkInvariantLocked(const char* where, uint64_t slot);

  void enqueueTxLocked(TxOp op);
  void flushTxLocked();
  void teardownFdLocked();
  void armEpollOutLocked();
  void readLoop(); // loop thread; takes/releases ctx lock internally
  bool readPreamble(); // no lock
  void dispatchPreamble(); // takes ctx lock
  void finishRx(); // takes ctx lock
  void completeTxLocked(TxOp& op);

  TcpContext* ctx_;
  TcpDevice* dev_;
  int peerRank_;
  TcpAddress self_;
  TcpAddress peer_;
  // Written under the context mutex; read lock-free by the loop thread's
  // readLoop fast-path checks, hence atomic (relaxed is sufficient: the
  // epoll unregister + tick-wait in close() is the real synchronizer).
  std::atomic<int> fd_{-1};
  enum State { INIT, CONNECTED, CLOSED };
  std::atomic<State> state_{INIT};
  std::exception_ptr error_;

  // --- protocol state (context mutex) ---
  struct PendingSend {
    TcpUnboundBuffer* buf;
    size_t offset;
    size_t nbytes;
  };
  struct PendingRecv {
    TcpUnboundBuffer* buf;
    size_t offset;
    size_t nbytes;
  };
  std::unordered_map<uint64_t, std::deque<PendingSend>> localPendingSend_;
  std::unordered_map<uint64_t, std::deque<PendingRecv>> localPendingRecv_;
  std::unordered_map<uint64_t, uint64_t> remoteRecvCredits_;
  std::unordered_map<uint64_t, uint64_t> recvsPosted_; // cumulative R
  std::unordered_map<uint64_t, uint64_t> sendReadySeen_; // cumulative Sn
  std::unordered_map<uint64_t, TcpBuffer*> sendBufs_;
  std::unordered_map<uint64_t, TcpBuffer*> recvBufs_;
  // BOUND_DATA that arrived before createRecvBuffer: slot -> (roffset, data)
  std::unordered_map<uint64_t, std::deque<std::pair<uint64_t, std::string>>>
      earlyBoundData_;
  // Unbound payloads (eager, or credit sends that overtook a stash-satisfied
  // recv's RECV_READY) that arrived with no posted recv. FIFO per slot;
  // paired with localPendingRecv_: at most one of the two is non-empty.
  std::unordered_map<uint64_t, std::deque<std::string>> eagerStash_;

  std::deque<TxOp> tx_;
  bool epollOutArmed_{false};
  // True while one thread drains tx_ with the context mutex dropped
  // around the writev syscall; guards tx_.front() stability and defers
  // fd teardown (see flushTxLocked / failLocked / close).
  bool txBusy_{false};
  bool teardownDeferred_{false};
  bool everRegistered_{false}; // gated loop barrier in the destructor

  // --- rx state machine (loop thread only, except targets set under lock) ---
  Preamble rxPre_;
  size_t rxPreRead_{0};
  char* rxDst_{nullptr};
  size_t rxLen_{0};
  size_t rxRead_{0};
  bool rxActive_{false};
  TcpUnboundBuffer* rxUbuf_{nullptr};
  TcpBuffer* rxBbuf_{nullptr};
  std::string rxSpill_; // payload for not-yet-registered bound slots
  bool rxIsSpill_{false};
  bool rxIsEagerSpill_{false}; // rxSpill_ holds unbound data for eagerStash_

  friend class TcpContext;
  friend class TcpBuffer;
};

} // namespace tcp
} // namespace glooamd
