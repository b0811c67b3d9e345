#include "transport/tcp/pair.h"

#include <netinet/tcp.h>
#include <poll.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <sys/uio.h>
#include <unistd.h>

#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <limits>

#include "common/logging.h"
#include "transport/tcp/context.h"
#include "transport/tcp/device.h"

namespace glooamd {
namespace tcp {

// ---------------------------------------------------------------------------
// TcpBuffer (bound)
// ---------------------------------------------------------------------------

TcpBuffer::~TcpBuffer() {
  pair_->detachBuffer(this);
}

void TcpBuffer::send(size_t offset, size_t length, size_t roffset) {
  pair_->boundSend(this, offset, length, roffset);
}

void TcpBuffer::waitRecv() {
  auto* ctx = pair_->ctx_;
  std::unique_lock<std::mutex> lock(ctx->mu_);
  auto timeout = ctx->getTimeout();
  auto pred = [&] { return recvCount_ > 0 || error_ != nullptr; };
  if (timeout.count() < 0) {
    ctx->cv_.wait(lock, pred);
  } else if (!ctx->cv_.wait_for(lock, timeout, pred)) {
    lock.unlock();
    ctx->signalException("waitRecv timeout on bound buffer");
    throw TimeoutException("bound buffer waitRecv timed out");
  }
  if (error_ && recvCount_ == 0) {
    std::rethrow_exception(error_);
  }
  recvCount_--;
}

void TcpBuffer::waitSend() {
  auto* ctx = pair_->ctx_;
  std::unique_lock<std::mutex> lock(ctx->mu_);
  auto timeout = ctx->getTimeout();
  auto pred = [&] { return sendCount_ > 0 || error_ != nullptr; };
  if (timeout.count() < 0) {
    ctx->cv_.wait(lock, pred);
  } else if (!ctx->cv_.wait_for(lock, timeout, pred)) {
    lock.unlock();
    ctx->signalException("waitSend timeout on bound buffer");
    throw TimeoutException("bound buffer waitSend timed out");
  }
  if (error_ && sendCount_ == 0) {
    std::rethrow_exception(error_);
  }
  sendCount_--;
}

// ---------------------------------------------------------------------------
// TcpPair
// ---------------------------------------------------------------------------

TcpPair::TcpPair(TcpContext* ctx, TcpDevice* dev, int peerRank)
    : ctx_(ctx), dev_(dev), peerRank_(peerRank) {
  self_ = dev_->addressForSeq(dev_->nextSeq());
}

TcpPair::~TcpPair() {
  close();
  // If the pair was torn down earlier via failLocked (unregisterNoWait +
  // deferred ::close), close() above returned immediately — but a poll
  // callback dispatched before the deferred removal may still hold this
  // handler. Wait one loop barrier before the memory goes away.
  if (everRegistered_) {
    dev_->loop().barrier();
  }
}

std::string TcpPair::str() const {
  return "pair[" + std::to_string(ctx_->rank) + "<->" +
      std::to_string(peerRank_) + " " + self_.str() + "]";
}

ssize_t TcpPair::ioRead(char* buf, size_t len) {
  return ::read(fd_, buf, len);
}

ssize_t TcpPair::ioWritev(const struct iovec* iov, int iovcnt) {
  return ::writev(fd_, iov, iovcnt);
}

void TcpPair::connect(const std::vector<char>& peerAddressBytes) {
  peer_ = TcpAddress::fromBytes(peerAddressBytes);
  auto timeout = ctx_->getTimeout();
  bool initiator = ctx_->rank < peerRank_;

  if (initiator) {
    // Initiator: dial the peer's listener and write the peer pair's seq.
    // Retry until the context timeout deadline (mirroring the listener
    // side's waitForConnection) — a fixed attempt cap spuriously fails
    // slow or large-scale rendezvous.
    const auto deadline = std::chrono::steady_clock::now() +
        (timeout.count() > 0 ? timeout : std::chrono::milliseconds(30000));
    int fd = -1;
    for (;;) {
      fd = socket(peer_.sockaddr().ss_family, SOCK_STREAM | SOCK_CLOEXEC, 0);
      GA_ENFORCE_GE(fd, 0, "socket: ", strerror(errno));
      int rv = ::connect(
          fd,
          reinterpret_cast<const struct sockaddr*>(&peer_.sockaddr()),
          peer_.sockaddrLen());
      if (rv == 0) {
        break;
      }
      ::close(fd);
      fd = -1;
      if (std::chrono::steady_clock::now() >= deadline) {
        GA_THROW_IO(
            "connect to ", peer_.str(), " failed: ", strerror(errno));
      }
      usleep(100 * 1000); // peer's listener may not be up in cross-process
                          // rendezvous races; retry until deadline
    }
    uint64_t seq = peer_.seq();
    size_t written = 0;
    auto* p = reinterpret_cast<const char*>(&seq);
    while (written < sizeof(seq)) {
      ssize_t n = write(fd, p + written, sizeof(seq) - written);
      if (n < 0 && errno == EINTR) {
        continue;
      }
      GA_ENFORCE_GT(n, 0, "seq write: ", strerror(errno));
      written += n;
    }
    setSocketOptions(fd);
    fd_ = fd;
  } else {
    // Listener: wait for the peer to dial with our seq.
    fd_ = dev_->waitForConnection(self_.seq(), timeout);
  }

  ioHandshake(initiator); // e.g. TLS; fd may be blocking here
  setNonBlocking(fd_);

  {
    std::lock_guard<std::mutex> lock(ctx_->mu_);
    state_ = CONNECTED;
  }
  everRegistered_ = true;
  dev_->loop().registerDescriptor(fd_, EPOLLIN, this);
}

void TcpPair::close() {
  int fd;
  {
    std::unique_lock<std::mutex> lock(ctx_->mu_);
    if (state_ == CLOSED) {
      return;
    }
    state_ = CLOSED; // a mid-drain flusher aborts at its next relock
    ctx_->cv_.wait(lock, [&] { return !txBusy_; });
    fd = fd_;
    fd_ = -1;
  }
  if (fd >= 0) {
    dev_->loop().unregisterDescriptor(fd);
    ioClose();
    // Graceful shutdown: send FIN, then drain incoming bytes until the
    // peer's EOF. Closing a TCP socket with unread rx data raises RST,
    // and Linux discards the peer's unread receive queue on RST — which
    // destroys in-flight final notifications on the peer (observed as a
    // rare "closed by peer" failure in the benchmark's final barrier).
    (void)::shutdown(fd, SHUT_WR);
    char drainBuf[4096];
    const auto drainDeadline =
        std::chrono::steady_clock::now() + std::chrono::milliseconds(200);
    while (std::chrono::steady_clock::now() < drainDeadline) {
      ssize_t n = ::read(fd, drainBuf, sizeof(drainBuf));
      if (n == 0) {
        break; // EOF: the peer closed its side too
      }
      if (n < 0) {
        if (errno == EAGAIN || errno == EWOULDBLOCK) {
          usleep(1000);
          continue;
        }
        break;
      }
    }
    ::close(fd);
  }
}

std::unique_ptr<transport::Buffer>
TcpPair::createSendBuffer(uint64_t slot, void* ptr, size_t size) {
  auto buf = std::make_unique<TcpBuffer>(this, slot, ptr, size, true);
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  GA_ENFORCE_EQ(
      sendBufs_.count(slot), 0ul, "send buffer already registered for slot");
  sendBufs_[slot] = buf.get();
  return buf;
}

std::unique_ptr<transport::Buffer>
TcpPair::createRecvBuffer(uint64_t slot, void* ptr, size_t size) {
  auto buf = std::make_unique<TcpBuffer>(this, slot, ptr, size, false);
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  GA_ENFORCE_EQ(
      recvBufs_.count(slot), 0ul, "recv buffer already registered for slot");
  recvBufs_[slot] = buf.get();
  // Replay early-arrived one-sided writes for this slot.
  auto it = earlyBoundData_.find(slot);
  if (it != earlyBoundData_.end()) {
    for (auto& entry : it->second) {
      GA_ENFORCE_LE(entry.first + entry.second.size(), size);
      std::memcpy(
          static_cast<char*>(ptr) + entry.first,
          entry.second.data(),
          entry.second.size());
      buf->recvCount_++;
    }
    earlyBoundData_.erase(it);
    ctx_->cv_.notify_all();
  }
  return buf;
}

void TcpPair::detachBuffer(TcpBuffer* buf) {
  std::unique_lock<std::mutex> lock(ctx_->mu_);
  // Wait for any in-flight rx/tx referencing this buffer to drain.
  ctx_->cv_.wait(lock, [&] { return !referencesBuffer(buf); });
  for (auto it = sendBufs_.begin(); it != sendBufs_.end();) {
    it = (it->second == buf) ? sendBufs_.erase(it) : std::next(it);
  }
  for (auto it = recvBufs_.begin(); it != recvBufs_.end();) {
    it = (it->second == buf) ? recvBufs_.erase(it) : std::next(it);
  }
}

bool TcpPair::referencesBuffer(const void* buf) const {
  if (rxUbuf_ == buf || rxBbuf_ == buf) {
    return true;
  }
  for (const auto& op : tx_) {
    if (op.bbuf == buf || op.ubuf == buf) {
      return true;
    }
  }
  return false;
}

void TcpPair::detachUnbound(TcpUnboundBuffer* buf) {
  // context mutex held by caller (TcpUnboundBuffer destructor)
  for (auto& kv : localPendingSend_) {
    auto& q = kv.second;
    for (auto it = q.begin(); it != q.end();) {
      it = (it->buf == buf) ? q.erase(it) : std::next(it);
    }
  }
  for (auto& kv : localPendingRecv_) {
    auto& q = kv.second;
    for (auto it = q.begin(); it != q.end();) {
      it = (it->buf == buf) ? q.erase(it) : std::next(it);
    }
  }
}

// --- send paths (context mutex held) ---------------------------------------

void TcpPair::ubufSend(
    TcpUnboundBuffer* buf,
    uint64_t slot,
    size_t off,
    size_t nb) {
  if (error_) {
    std::rethrow_exception(error_);
  }
  GA_ENFORCE_EQ(state_, CONNECTED, "pair not connected: ", str());
  auto& credits = remoteRecvCredits_[slot];
  if (credits > 0) {
    credits--;
    TxOp op{};
    op.pre = {UNBOUND_DATA,
              static_cast<uint32_t>(ctx_->rank),
              slot,
              nb,
              0};
    op.payload = static_cast<const char*>(buf->ptr) + off;
    op.payloadLen = nb;
    op.ubuf = buf;
    enqueueTxLocked(std::move(op));
  } else if (nb <= eagerMaxBytes() && localPendingSend_[slot].empty()) {
    // Small message, no credit, and no older notified send whose payload
    // this one must not overtake: push the data without rendezvous.
    TxOp op{};
    op.pre = {EAGER_DATA, static_cast<uint32_t>(ctx_->rank), slot, nb, 0};
    op.payload = static_cast<const char*>(buf->ptr) + off;
    op.payloadLen = nb;
    op.ubuf = buf;
    enqueueTxLocked(std::move(op));
  } else {
    localPendingSend_[slot].push_back({buf, off, nb});
    TxOp op{};
    op.pre = {SEND_READY, static_cast<uint32_t>(ctx_->rank), slot, nb, 0};
    enqueueTxLocked(std::move(op));
  }
}

void TcpPair::checkInvariantLocked(const char* where, uint64_t slot) {
  static const bool on = getenv("GLOO_AMD_CHECK_INVARIANTS") != nullptr;
  if (!on) {
    return;
  }
  auto qi = localPendingRecv_.find(slot);
  auto si = eagerStash_.find(slot);
  bool qne = qi != localPendingRecv_.end() && !qi->second.empty();
  bool sne = si != eagerStash_.end() && !si->second.empty();
  if (qne && sne) {
    GA_ERROR << "INVARIANT VIOLATION at " << where << " " << str()
             << " slot=" << slot
             << " pendRecv=" << qi->second.size()
             << " stash=" << si->second.size()
             << " rxActive=" << rxActive_
             << " rxIsEagerSpill=" << rxIsEagerSpill_
             << " rxSlot=" << rxPre_.slot
             << " rxOpcode=" << rxPre_.opcode;
    abort();
  }
}

size_t TcpPair::eagerMaxBytes() {
  static size_t v = [] {
    if (const char* e = getenv("GLOO_AMD_EAGER_MAX")) {
      return static_cast<size_t>(atoll(e));
    }
    return static_cast<size_t>(8192);
  }();
  return v;
}

bool TcpPair::takeStashLocked(
    TcpUnboundBuffer* buf,
    uint64_t slot,
    size_t off,
    size_t nb) {
  auto it = eagerStash_.find(slot);
  if (it == eagerStash_.end() || it->second.empty()) {
    return false;
  }
  std::string payload = std::move(it->second.front());
  it->second.pop_front();
  if (it->second.empty()) {
    eagerStash_.erase(it);
  }
  GA_ENFORCE_LE(payload.size(), nb, "stashed payload larger than posted recv");
  std::memcpy(static_cast<char*>(buf->ptr) + off, payload.data(), payload.size());
  buf->recvCompletions_.push_back(peerRank_);
  ctx_->cv_.notify_all();
  return true;
}

void TcpPair::postRecv(
    TcpUnboundBuffer* buf,
    uint64_t slot,
    size_t off,
    size_t nb,
    bool fromClaim) {
  if (!fromClaim && takeStashLocked(buf, slot, off, nb)) {
    // Satisfied by already-arrived eager/credit data: complete locally.
    // No RECV_READY (the sender never queued this payload) and no counter
    // changes (the stash never corresponds to a notified send). Checked
    // before the error check: a peer that eagerly sent its final payload
    // may have exited (closing the pair) before this recv was posted —
    // fully-arrived data is still valid.
    return;
  }
  if (error_) {
    std::rethrow_exception(error_);
  }
  GA_ENFORCE_EQ(state_, CONNECTED, "pair not connected: ", str());
  recvsPosted_[slot]++;
  if (!fromClaim) {
    // A targeted recv consumes one excess notified send, if any, so the
    // unclaimed tally stays consistent (len == max(0, Sn - R)).
    ctx_->consumeUnclaimedLocked(slot, peerRank_);
  }
  localPendingRecv_[slot].push_back({buf, off, nb});
  checkInvariantLocked("postRecv", slot);
  TxOp op{};
  op.pre = {RECV_READY, static_cast<uint32_t>(ctx_->rank), slot, nb, 0};
  enqueueTxLocked(std::move(op));
}

void TcpPair::boundSend(
    TcpBuffer* buf,
    size_t off,
    size_t len,
    size_t roff) {
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  if (error_) {
    std::rethrow_exception(error_);
  }
  GA_ENFORCE_EQ(state_, CONNECTED, "pair not connected: ", str());
  GA_ENFORCE_LE(off + len, buf->size());
  TxOp op{};
  op.pre = {BOUND_DATA,
            static_cast<uint32_t>(ctx_->rank),
            buf->slot(),
            len,
            roff};
  op.payload = static_cast<const char*>(buf->ptr()) + off;
  op.payloadLen = len;
  op.bbuf = buf;
  enqueueTxLocked(std::move(op));
}

// --- tx machinery (context mutex held) -------------------------------------

void TcpPair::enqueueTxLocked(TxOp op) {
  tx_.push_back(std::move(op));
  flushTxLocked();
}

void TcpPair::completeTxLocked(TxOp& op) {
  if (op.ubuf != nullptr &&
      (op.pre.opcode == UNBOUND_DATA || op.pre.opcode == EAGER_DATA)) {
    op.ubuf->sendCompletions_++;
    ctx_->cv_.notify_all();
  }
  if (op.bbuf != nullptr) {
    op.bbuf->sendCount_++;
    ctx_->cv_.notify_all();
  }
}

void TcpPair::flushTxLocked() {
  if (state_ != CONNECTED || fd_ < 0 || txBusy_) {
    // txBusy_: another thread is mid-drain (it re-checks tx_ after its
    // unlocked writev returns, so ops enqueued meanwhile are not lost).
    return;
  }
  // Cap the bytes pushed per call so a loop-thread EPOLLOUT drain cannot
  // monopolize the loop for an unbounded large payload; user threads get
  // their RX handled concurrently regardless because the writev below
  // runs with the context mutex RELEASED (reference: per-pair mutex in
  // gloo/transport/tcp/pair.cc:658-688 serves the same goal).
  static const size_t kMaxFlushBytes = [] {
    if (const char* e = getenv("GLOO_AMD_MAX_FLUSH")) {
      long v = atol(e);
      return v <= 0 ? std::numeric_limits<size_t>::max()
                    : static_cast<size_t>(v);
    }
    return static_cast<size_t>(4 << 20);
  }();
  txBusy_ = true;
  size_t flushed = 0;
  bool arm = false;
  while (true) {
    if (error_) {
      // failLocked ran (possibly while we were unlocked): it fans the
      // error out but defers queue/fd teardown to us (teardownDeferred_).
      tx_.clear();
      if (teardownDeferred_) {
        teardownDeferred_ = false;
        teardownFdLocked();
      }
      break;
    }
    if (state_ != CONNECTED || fd_ < 0 || tx_.empty()) {
      break;
    }
    if (flushed >= kMaxFlushBytes) {
      arm = true;
      break;
    }
    // tx_.front() is stable while txBusy_: only this thread pops, and
    // deque push_back never invalidates references to existing elements.
    TxOp& op = tx_.front();
    struct iovec iov[2];
    int iovcnt = 0;
    if (op.preWritten < sizeof(Preamble)) {
      iov[iovcnt].iov_base =
          reinterpret_cast<char*>(&op.pre) + op.preWritten;
      iov[iovcnt].iov_len = sizeof(Preamble) - op.preWritten;
      iovcnt++;
    }
    if (op.payloadWritten < op.payloadLen) {
      iov[iovcnt].iov_base =
          const_cast<char*>(op.payload) + op.payloadWritten;
      iov[iovcnt].iov_len = std::min(
          op.payloadLen - op.payloadWritten, kMaxFlushBytes - flushed);
      iovcnt++;
    }
    if (iovcnt == 0) {
      completeTxLocked(op);
      tx_.pop_front();
      continue;
    }
    // The syscall (a memcpy of up to the quota into the kernel sndbuf)
    // runs unlocked so it never serializes other pairs of this context.
    // fd lifetime: close()/failLocked defer the ::close while txBusy_.
    ctx_->mu_.unlock();
    ssize_t n = ioWritev(iov, iovcnt);
    int savedErrno = errno;
    ctx_->mu_.lock();
    if (error_ || state_ != CONNECTED || fd_ < 0) {
      continue; // top of loop handles teardown/exit
    }
    if (n < 0) {
      if (savedErrno == EAGAIN || savedErrno == EWOULDBLOCK) {
        arm = true;
        break;
      }
      if (savedErrno == EINTR) {
        continue;
      }
      failLocked(std::make_exception_ptr(IoException(
          "writev to " + str() + ": " + strerror(savedErrno))));
      continue; // top of loop performs the deferred teardown
    }
    // Account written bytes across preamble then payload.
    flushed += static_cast<size_t>(n);
    size_t rem = static_cast<size_t>(n);
    size_t preLeft = sizeof(Preamble) - op.preWritten;
    size_t take = std::min(rem, preLeft);
    op.preWritten += take;
    rem -= take;
    op.payloadWritten += rem;
    if (op.preWritten == sizeof(Preamble) &&
        op.payloadWritten == op.payloadLen) {
      completeTxLocked(op);
      tx_.pop_front();
    }
    // else: partial; loop retries writev (kernel buffer may have space)
  }
  txBusy_ = false;
  if (arm) {
    armEpollOutLocked();
  }
  // Wake close()/detachBuffer/~TcpUnboundBuffer waiting on !txBusy_ or
  // on tx_ references draining.
  ctx_->cv_.notify_all();
}

void TcpPair::armEpollOutLocked() {
  if (!epollOutArmed_ && fd_ >= 0) {
    dev_->loop().modifyDescriptor(fd_, EPOLLIN | EPOLLOUT, this);
    epollOutArmed_ = true;
  }
}

// --- loop thread: events ----------------------------------------------------

void TcpPair::handleEvents(uint32_t events) {
  if (events & EPOLLIN) {
    // Drain readable data FIRST: when the peer closes right after
    // sending, EPOLLHUP arrives together with the final payload bytes;
    // failing before reading would drop them. read()==0 inside the loop
    // handles the eventual EOF.
    readLoop();
  }
  if (events & (EPOLLERR | EPOLLHUP)) {
    std::lock_guard<std::mutex> lock(ctx_->mu_);
    failLocked(std::make_exception_ptr(
        IoException("connection to " + str() + " closed (EPOLLERR/HUP)")));
    return;
  }
  if (events & EPOLLOUT) {
    std::lock_guard<std::mutex> lock(ctx_->mu_);
    flushTxLocked();
    if (tx_.empty() && epollOutArmed_ && fd_ >= 0) {
      dev_->loop().modifyDescriptor(fd_, EPOLLIN, this);
      epollOutArmed_ = false;
    }
  }
}

bool TcpPair::readPreamble() {
  while (rxPreRead_ < sizeof(Preamble)) {
    ssize_t n = ioRead(
        reinterpret_cast<char*>(&rxPre_) + rxPreRead_,
        sizeof(Preamble) - rxPreRead_);
    if (n > 0) {
      rxPreRead_ += n;
      continue;
    }
    if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
      return false;
    }
    if (n < 0 && errno == EINTR) {
      continue;
    }
    // EOF or hard error
    std::lock_guard<std::mutex> lock(ctx_->mu_);
    failLocked(std::make_exception_ptr(IoException(
        n == 0 ? "connection to " + str() + " closed by peer"
               : "read from " + str() + ": " + strerror(errno))));
    return false;
  }
  return true;
}

void TcpPair::readLoop() {
  for (;;) {
    if (fd_ < 0 || state_ != CONNECTED) {
      return;
    }
    if (!rxActive_) {
      if (!readPreamble()) {
        return;
      }
      rxPreRead_ = 0;
      dispatchPreamble();
      continue;
    }
    // Payload phase: read into the destination chosen by dispatch.
    while (rxRead_ < rxLen_) {
      ssize_t n = ioRead(rxDst_ + rxRead_, rxLen_ - rxRead_);
      if (n > 0) {
        rxRead_ += n;
        continue;
      }
      if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
        return;
      }
      if (n < 0 && errno == EINTR) {
        continue;
      }
      std::lock_guard<std::mutex> lock(ctx_->mu_);
      failLocked(std::make_exception_ptr(IoException(
          n == 0 ? "connection to " + str() + " closed mid-payload"
                 : "read from " + str() + ": " + strerror(errno))));
      return;
    }
    finishRx();
  }
}

void TcpPair::dispatchPreamble() {
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  const auto& pre = rxPre_;
  switch (pre.opcode) {
    case RECV_READY: {
      auto& q = localPendingSend_[pre.slot];
      if (!q.empty()) {
        PendingSend ps = q.front();
        q.pop_front();
        TxOp op{};
        op.pre = {UNBOUND_DATA,
                  static_cast<uint32_t>(ctx_->rank),
                  pre.slot,
                  ps.nbytes,
                  0};
        op.payload = static_cast<const char*>(ps.buf->ptr) + ps.offset;
        op.payloadLen = ps.nbytes;
        op.ubuf = ps.buf;
        enqueueTxLocked(std::move(op));
      } else {
        remoteRecvCredits_[pre.slot]++;
      }
      return;
    }
    case SEND_READY: {
      sendReadySeen_[pre.slot]++;
      if (sendReadySeen_[pre.slot] > recvsPosted_[pre.slot]) {
        if (!ctx_->offerSendReadyLocked(this, pre.slot)) {
          ctx_->unclaimedSendReady_[pre.slot].push_back(peerRank_);
        }
      }
      return;
    }
    case UNBOUND_DATA:
    case EAGER_DATA: {
      // A posted recv takes the data directly; otherwise a matching
      // any-recv waiter does; otherwise it is stashed until one is
      // posted. UNBOUND_DATA can hit the stash path too: a credit send
      // whose RECV_READY's recv was itself satisfied from the stash.
      auto& q = localPendingRecv_[pre.slot];
      if (!q.empty()) {
        PendingRecv pr = q.front();
        q.pop_front();
        GA_ENFORCE_LE(pre.length, pr.nbytes, "unbound payload overflow");
        rxUbuf_ = pr.buf;
        rxDst_ = static_cast<char*>(pr.buf->ptr) + pr.offset;
      } else {
        TcpContext::AnyRecv ar;
        if (ctx_->claimAnyRecvLocked(peerRank_, pre.slot, &ar)) {
          GA_ENFORCE_LE(pre.length, ar.nbytes, "unbound payload overflow");
          rxUbuf_ = ar.buf;
          rxDst_ = static_cast<char*>(ar.buf->ptr) + ar.offset;
        } else {
          rxSpill_.resize(pre.length);
          rxDst_ = rxSpill_.empty() ? nullptr : &rxSpill_[0];
          rxIsEagerSpill_ = true;
        }
      }
      rxLen_ = pre.length;
      rxRead_ = 0;
      rxActive_ = true;
      if (rxLen_ == 0) {
        // complete inline (finishRx would re-lock)
        if (rxUbuf_ != nullptr) {
          rxUbuf_->recvCompletions_.push_back(peerRank_);
          rxUbuf_ = nullptr;
          ctx_->cv_.notify_all();
        } else {
          eagerStash_[pre.slot].emplace_back();
          rxIsEagerSpill_ = false;
        }
        rxActive_ = false;
      }
      checkInvariantLocked("dispatchData", pre.slot);
      return;
    }
    case BOUND_DATA: {
      auto it = recvBufs_.find(pre.slot);
      if (it != recvBufs_.end()) {
        TcpBuffer* buf = it->second;
        GA_ENFORCE_LE(
            pre.roffset + pre.length, buf->size(), "bound write overflow");
        rxBbuf_ = buf;
        rxDst_ = static_cast<char*>(buf->ptr()) + pre.roffset;
        rxIsSpill_ = false;
      } else {
        // Not registered yet: spill to heap, replay at registration.
        rxSpill_.resize(pre.length);
        rxDst_ = rxSpill_.empty() ? nullptr : &rxSpill_[0];
        rxIsSpill_ = true;
      }
      rxLen_ = pre.length;
      rxRead_ = 0;
      rxActive_ = true;
      if (rxLen_ == 0 && !rxIsSpill_) {
        rxBbuf_->recvCount_++;
        rxBbuf_ = nullptr;
        rxActive_ = false;
        ctx_->cv_.notify_all();
      } else if (rxLen_ == 0 && rxIsSpill_) {
        earlyBoundData_[pre.slot].emplace_back(pre.roffset, std::string());
        rxActive_ = false;
        rxIsSpill_ = false;
      }
      return;
    }
    default:
      failLocked(std::make_exception_ptr(IoException(
          "protocol violation: bad opcode " + std::to_string(pre.opcode))));
  }
}

void TcpPair::finishRx() {
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  if (rxUbuf_ != nullptr) {
    rxUbuf_->recvCompletions_.push_back(peerRank_);
    rxUbuf_ = nullptr;
  } else if (rxIsEagerSpill_) {
    // A recv may have been posted while the payload was being read;
    // deliver directly in that case, matching the bound-data replay rule.
    auto& q = localPendingRecv_[rxPre_.slot];
    TcpContext::AnyRecv ar;
    if (!q.empty()) {
      PendingRecv pr = q.front();
      q.pop_front();
      GA_ENFORCE_LE(rxSpill_.size(), pr.nbytes, "unbound payload overflow");
      std::memcpy(
          static_cast<char*>(pr.buf->ptr) + pr.offset,
          rxSpill_.data(),
          rxSpill_.size());
      pr.buf->recvCompletions_.push_back(peerRank_);
    } else if (ctx_->claimAnyRecvLocked(peerRank_, rxPre_.slot, &ar)) {
      GA_ENFORCE_LE(rxSpill_.size(), ar.nbytes, "unbound payload overflow");
      std::memcpy(
          static_cast<char*>(ar.buf->ptr) + ar.offset,
          rxSpill_.data(),
          rxSpill_.size());
      ar.buf->recvCompletions_.push_back(peerRank_);
    } else {
      eagerStash_[rxPre_.slot].push_back(std::move(rxSpill_));
    }
    checkInvariantLocked("finishRxEager", rxPre_.slot);
    rxSpill_ = std::string();
    rxIsEagerSpill_ = false;
  } else if (rxBbuf_ != nullptr) {
    rxBbuf_->recvCount_++;
    rxBbuf_ = nullptr;
  } else if (rxIsSpill_) {
    // The slot may have been registered while we were reading the
    // payload; deliver directly in that case (otherwise the stash would
    // never be replayed).
    auto it = recvBufs_.find(rxPre_.slot);
    if (it != recvBufs_.end()) {
      TcpBuffer* buf = it->second;
      GA_ENFORCE_LE(rxPre_.roffset + rxSpill_.size(), buf->size());
      std::memcpy(
          static_cast<char*>(buf->ptr()) + rxPre_.roffset,
          rxSpill_.data(),
          rxSpill_.size());
      buf->recvCount_++;
    } else {
      earlyBoundData_[rxPre_.slot].emplace_back(
          rxPre_.roffset, std::move(rxSpill_));
    }
    rxSpill_ = std::string();
    rxIsSpill_ = false;
  }
  rxActive_ = false;
  rxDst_ = nullptr;
  rxLen_ = rxRead_ = 0;
  ctx_->cv_.notify_all();
}

// --- failure fan-out (context mutex held) -----------------------------------

void TcpPair::failLocked(std::exception_ptr e) {
  if (error_) {
    return; // already failed
  }
  error_ = e;
  // Fail every queued / pending operation's owner.
  for (auto& op : tx_) {
    if (op.ubuf && !op.ubuf->error_) {
      op.ubuf->error_ = e;
    }
    if (op.bbuf && !op.bbuf->error_) {
      op.bbuf->error_ = e;
    }
  }
  tx_.clear();
  for (auto& kv : localPendingSend_) {
    for (auto& ps : kv.second) {
      if (!ps.buf->error_) {
        ps.buf->error_ = e;
      }
    }
  }
  localPendingSend_.clear();
  for (auto& kv : localPendingRecv_) {
    for (auto& pr : kv.second) {
      if (!pr.buf->error_) {
        pr.buf->error_ = e;
      }
    }
  }
  localPendingRecv_.clear();
  for (auto& kv : sendBufs_) {
    if (!kv.second->error_) {
      kv.second->error_ = e;
    }
  }
  for (auto& kv : recvBufs_) {
    if (!kv.second->error_) {
      kv.second->error_ = e;
    }
  }
  if (rxUbuf_ && !rxUbuf_->error_) {
    rxUbuf_->error_ = e;
  }
  rxUbuf_ = nullptr;
  rxBbuf_ = nullptr;
  rxActive_ = false;
  rxIsSpill_ = false;
  rxIsEagerSpill_ = false;
  // eagerStash_ is deliberately kept: it holds fully-received payloads,
  // which stay deliverable after the peer closes.

  // Tear the socket down. If a flusher thread is mid-writev on this fd
  // with the mutex released, defer the teardown to it (closing now could
  // let the kernel recycle the fd under the in-flight syscall).
  if (fd_ >= 0) {
    if (txBusy_) {
      teardownDeferred_ = true;
    } else {
      teardownFdLocked();
    }
  }
  ctx_->cv_.notify_all();
}

void TcpPair::teardownFdLocked() {
  // Without waiting for a loop tick: this can run with the context mutex
  // held while the loop thread is blocked on that same mutex (tick-wait
  // would deadlock). epoll_ctl(DEL) stops new dispatches; the close
  // itself is deferred to the loop thread so it runs strictly after any
  // handler that may still be inside read().
  if (fd_ >= 0) {
    int fd = fd_;
    fd_ = -1;
    state_ = CLOSED;
    dev_->loop().unregisterNoWait(fd);
    dev_->loop().defer([fd] { ::close(fd); });
  }
}

} // namespace tcp
} // namespace glooamd
