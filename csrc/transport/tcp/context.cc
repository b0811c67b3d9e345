#include "transport/tcp/context.h"

#include <cstring>

#include "common/logging.h"
#include "common/utils.h"
#include "transport/tcp/device.h"

namespace glooamd {
namespace tcp {

// ---------------------------------------------------------------------------
// TcpContext
// ---------------------------------------------------------------------------

TcpContext::TcpContext(std::shared_ptr<TcpDevice> device, int rank, int size)
    : transport::Context(rank, size), device_(std::move(device)) {
  pairs_.resize(size);
}

TcpContext::~TcpContext() {
  for (auto& p : pairs_) {
    if (p) {
      p->close();
    }
  }
}

transport::Pair* TcpContext::getPair(int rank) {
  GA_ENFORCE_GE(rank, 0);
  GA_ENFORCE_LT(rank, size);
  return pairs_[rank].get();
}

transport::Pair* TcpContext::createPair(int rank) {
  GA_ENFORCE_NE(rank, this->rank, "no pair to self");
  pairs_[rank] = std::make_unique<TcpPair>(this, device_.get(), rank);
  return pairs_[rank].get();
}

std::unique_ptr<transport::UnboundBuffer> TcpContext::createUnboundBuffer(
    void* ptr,
    size_t size) {
  return std::make_unique<TcpUnboundBuffer>(this, ptr, size);
}

namespace {
// Serialization of one rank's pair addresses: u32 count, then per entry
// u32 len + bytes (self entry empty).
std::vector<char> packAddresses(const std::vector<std::vector<char>>& addrs) {
  size_t total = 4;
  for (const auto& a : addrs) {
    total += 4 + a.size();
  }
  std::vector<char> out(total);
  char* p = out.data();
  uint32_t n = addrs.size();
  std::memcpy(p, &n, 4);
  p += 4;
  for (const auto& a : addrs) {
    uint32_t len = a.size();
    std::memcpy(p, &len, 4);
    p += 4;
    std::memcpy(p, a.data(), a.size());
    p += a.size();
  }
  return out;
}

std::vector<std::vector<char>> unpackAddresses(const std::vector<char>& blob) {
  GA_ENFORCE_GE(blob.size(), 4ul);
  const char* p = blob.data();
  uint32_t n;
  std::memcpy(&n, p, 4);
  p += 4;
  std::vector<std::vector<char>> out(n);
  for (uint32_t i = 0; i < n; i++) {
    uint32_t len;
    std::memcpy(&len, p, 4);
    p += 4;
    out[i].assign(p, p + len);
    p += len;
  }
  return out;
}
} // namespace

void TcpContext::connectFullMesh(IStore& store) {
  std::vector<std::vector<char>> addrs(size);
  for (int i = 0; i < size; i++) {
    if (i == rank) {
      continue;
    }
    createPair(i);
    addrs[i] = pairs_[i]->address().bytes();
  }
  store.set(std::to_string(rank), packAddresses(addrs));

  for (int i = 0; i < size; i++) {
    if (i == rank) {
      continue;
    }
    store.wait({std::to_string(i)}, getTimeout());
    auto peerAddrs = unpackAddresses(store.get(std::to_string(i)));
    GA_ENFORCE_EQ(peerAddrs.size(), static_cast<size_t>(size));
    pairs_[i]->connect(peerAddrs[rank]);
  }

  // Connectivity report (reference tcp/context.cc:241-256 parity).
  if (logThreshold() <= LogLevel::DEBUG) {
    std::string report = "rank " + std::to_string(rank) + " connected to:";
    for (int i = 0; i < size; i++) {
      if (i != rank && pairs_[i] && pairs_[i]->isConnected()) {
        report += " " + pairs_[i]->str();
      }
    }
    GA_DEBUG << report;
  }
}

static std::string detail2(const std::unordered_map<uint64_t, uint64_t>& m) {
  std::string d;
  for (auto& kv : m) {
    if (kv.second != 0) {
      d += " " + std::to_string(kv.first) + ":" + std::to_string(kv.second);
    }
  }
  return d.empty() ? std::string(" -") : d;
}

void TcpContext::dumpStateLocked(const char* why) {
  // GLOO_AMD_DUMP_ON_TIMEOUT aid: one line per pair with queue depths.
  std::string out = std::string("ctx rank ") + std::to_string(rank) +
      " dump (" + why + ")\n";
  for (auto& p : pairs_) {
    if (!p) {
      continue;
    }
    out += "  " + p->str() + " tx=" + std::to_string(p->tx_.size());
    out += p->error_ ? " FAILED" : "";
    auto sum = [](auto& m) {
      size_t t = 0;
      for (auto& kv : m) {
        t += kv.second.size();
      }
      return t;
    };
    auto detail = [](auto& m) {
      std::string d;
      for (auto& kv : m) {
        if (!kv.second.empty()) {
          d += " " + std::to_string(kv.first) + ":" +
              std::to_string(kv.second.size());
        }
      }
      return d.empty() ? std::string(" -") : d;
    };
    out += " pendSend[" + detail(p->localPendingSend_) + " ]";
    out += " pendRecv[" + detail(p->localPendingRecv_) + " ]";
    out += " stash[" + detail(p->eagerStash_) + " ]";
    out += " credits[" + detail2(p->remoteRecvCredits_) + " ]";
    out += " Sn[" + detail2(p->sendReadySeen_) + " ]";
    out += " R[" + detail2(p->recvsPosted_) + " ]";
    out += " epollOut=" + std::to_string(p->epollOutArmed_ ? 1 : 0);
    out += " rxActive=" + std::to_string(p->rxActive_ ? 1 : 0);
    out += "\n";
  }
  out += "  anyRecvs=" + std::to_string(anyRecvs_.size());
  GA_ERROR << out;
}

void TcpContext::signalException(const std::string& msg) {
  std::lock_guard<std::mutex> lock(mu_);
  if (getEnvFlag("GLOO_AMD_DUMP_ON_TIMEOUT")) {
    dumpStateLocked(msg.c_str());
  }
  for (auto& p : pairs_) {
    if (p && !p->eagerStash_.empty()) {
      GA_WARN << "rank " << rank << ": timeout with undelivered payloads "
              << "from " << p->str() << " — a peer likely advanced to a "
              << "different collective (ranks must invoke collectives "
              << "equally; GLOO_AMD_DUMP_ON_TIMEOUT=1 shows queue state)";
      break;
    }
  }
  auto e = std::make_exception_ptr(IoException(msg));
  for (auto& p : pairs_) {
    if (p) {
      p->failLocked(e);
    }
  }
  // Fail any-recv waiters and self queues too.
  for (auto& ar : anyRecvs_) {
    if (!ar.buf->error_) {
      ar.buf->error_ = e;
    }
  }
  anyRecvs_.clear();
  for (auto* qmap : {&selfSends_, &selfRecvs_}) {
    for (auto& kv : *qmap) {
      for (auto& op : kv.second) {
        if (!op.buf->error_) {
          op.buf->error_ = e;
        }
      }
    }
    qmap->clear();
  }
  cv_.notify_all();
}

bool TcpContext::claimAnyRecvLocked(int srcRank, uint64_t slot, AnyRecv* out) {
  for (auto it = anyRecvs_.begin(); it != anyRecvs_.end(); ++it) {
    if (it->slot != slot) {
      continue;
    }
    bool match = false;
    for (int r : it->srcRanks) {
      if (r == srcRank) {
        match = true;
        break;
      }
    }
    if (!match) {
      continue;
    }
    *out = *it;
    anyRecvs_.erase(it);
    return true;
  }
  return false;
}

bool TcpContext::offerSendReadyLocked(TcpPair* pair, uint64_t slot) {
  AnyRecv ar;
  if (!claimAnyRecvLocked(pair->peerRank(), slot, &ar)) {
    return false;
  }
  pair->postRecv(ar.buf, ar.slot, ar.offset, ar.nbytes, /*fromClaim=*/true);
  return true;
}

bool TcpContext::consumeUnclaimedLocked(uint64_t slot, int rank) {
  auto it = unclaimedSendReady_.find(slot);
  if (it == unclaimedSendReady_.end()) {
    return false;
  }
  auto& q = it->second;
  for (auto qi = q.begin(); qi != q.end(); ++qi) {
    if (*qi == rank) {
      q.erase(qi);
      if (q.empty()) {
        unclaimedSendReady_.erase(it);
      }
      return true;
    }
  }
  return false;
}

// --- self loopback ----------------------------------------------------------

void TcpContext::selfSendLocked(
    TcpUnboundBuffer* buf,
    uint64_t slot,
    size_t off,
    size_t nb) {
  auto& recvs = selfRecvs_[slot];
  if (!recvs.empty()) {
    SelfOp r = recvs.front();
    recvs.pop_front();
    GA_ENFORCE_LE(nb, r.nbytes, "self send larger than posted recv");
    std::memcpy(
        static_cast<char*>(r.buf->ptr) + r.offset,
        static_cast<const char*>(buf->ptr) + off,
        nb);
    r.buf->recvCompletions_.push_back(rank);
    buf->sendCompletions_++;
    cv_.notify_all();
    return;
  }
  // Any-recv waiter including self?
  for (auto it = anyRecvs_.begin(); it != anyRecvs_.end(); ++it) {
    if (it->slot != slot) {
      continue;
    }
    bool match = false;
    for (int r : it->srcRanks) {
      if (r == rank) {
        match = true;
        break;
      }
    }
    if (!match) {
      continue;
    }
    AnyRecv ar = *it;
    anyRecvs_.erase(it);
    GA_ENFORCE_LE(nb, ar.nbytes, "self send larger than posted recv");
    std::memcpy(
        static_cast<char*>(ar.buf->ptr) + ar.offset,
        static_cast<const char*>(buf->ptr) + off,
        nb);
    ar.buf->recvCompletions_.push_back(rank);
    buf->sendCompletions_++;
    cv_.notify_all();
    return;
  }
  selfSends_[slot].push_back({buf, off, nb});
}

void TcpContext::selfRecvLocked(
    TcpUnboundBuffer* buf,
    uint64_t slot,
    size_t off,
    size_t nb) {
  auto& sends = selfSends_[slot];
  if (!sends.empty()) {
    SelfOp s = sends.front();
    sends.pop_front();
    GA_ENFORCE_LE(s.nbytes, nb, "self send larger than posted recv");
    std::memcpy(
        static_cast<char*>(buf->ptr) + off,
        static_cast<const char*>(s.buf->ptr) + s.offset,
        s.nbytes);
    buf->recvCompletions_.push_back(rank);
    s.buf->sendCompletions_++;
    cv_.notify_all();
    return;
  }
  selfRecvs_[slot].push_back({buf, off, nb});
}

// ---------------------------------------------------------------------------
// TcpUnboundBuffer
// ---------------------------------------------------------------------------

TcpUnboundBuffer::~TcpUnboundBuffer() {
  std::unique_lock<std::mutex> lock(ctx_->mu_);
  // Drop queued-but-unsent protocol entries referencing this buffer.
  for (auto& p : ctx_->pairs_) {
    if (p) {
      p->detachUnbound(this);
    }
  }
  for (auto it = ctx_->anyRecvs_.begin(); it != ctx_->anyRecvs_.end();) {
    it = (it->buf == this) ? ctx_->anyRecvs_.erase(it) : std::next(it);
  }
  for (auto* qmap : {&ctx_->selfSends_, &ctx_->selfRecvs_}) {
    for (auto& kv : *qmap) {
      auto& q = kv.second;
      for (auto qi = q.begin(); qi != q.end();) {
        qi = (qi->buf == this) ? q.erase(qi) : std::next(qi);
      }
    }
  }
  // Wait for in-flight socket I/O referencing this buffer to drain.
  ctx_->cv_.wait(lock, [&] {
    for (auto& p : ctx_->pairs_) {
      if (p && p->referencesBuffer(this)) {
        return false;
      }
    }
    return true;
  });
}

void TcpUnboundBuffer::send(
    int dstRank,
    uint64_t slot,
    size_t offset,
    size_t nbytes) {
  if (nbytes == transport::kUnspecified) {
    nbytes = size - offset;
  }
  GA_ENFORCE_LE(offset + nbytes, size, "send out of bounds");
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  if (error_) {
    std::rethrow_exception(error_);
  }
  if (dstRank == ctx_->rank) {
    ctx_->selfSendLocked(this, slot, offset, nbytes);
    return;
  }
  auto* pair = static_cast<TcpPair*>(ctx_->getPair(dstRank));
  GA_ENFORCE(pair != nullptr, "no pair for rank ", dstRank);
  pair->ubufSend(this, slot, offset, nbytes);
}

void TcpUnboundBuffer::recv(
    int srcRank,
    uint64_t slot,
    size_t offset,
    size_t nbytes) {
  if (nbytes == transport::kUnspecified) {
    nbytes = size - offset;
  }
  GA_ENFORCE_LE(offset + nbytes, size, "recv out of bounds");
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  if (error_) {
    std::rethrow_exception(error_);
  }
  if (srcRank == ctx_->rank) {
    ctx_->selfRecvLocked(this, slot, offset, nbytes);
    return;
  }
  auto* pair = static_cast<TcpPair*>(ctx_->getPair(srcRank));
  GA_ENFORCE(pair != nullptr, "no pair for rank ", srcRank);
  pair->postRecv(this, slot, offset, nbytes, /*fromClaim=*/false);
}

void TcpUnboundBuffer::recv(
    const std::vector<int>& srcRanks,
    uint64_t slot,
    size_t offset,
    size_t nbytes) {
  if (nbytes == transport::kUnspecified) {
    nbytes = size - offset;
  }
  GA_ENFORCE_LE(offset + nbytes, size, "recv out of bounds");
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  if (error_) {
    std::rethrow_exception(error_);
  }
  // Claim a pending self send first, then any excess notified send.
  for (int r : srcRanks) {
    if (r == ctx_->rank) {
      auto& sends = ctx_->selfSends_[slot];
      if (!sends.empty()) {
        auto s = sends.front();
        sends.pop_front();
        GA_ENFORCE_LE(s.nbytes, nbytes);
        std::memcpy(
            static_cast<char*>(ptr) + offset,
            static_cast<const char*>(s.buf->ptr) + s.offset,
            s.nbytes);
        recvCompletions_.push_back(ctx_->rank);
        s.buf->sendCompletions_++;
        ctx_->cv_.notify_all();
        return;
      }
    }
  }
  // Already-arrived eager/credit payloads are the oldest available data
  // on their pair; take them before claiming merely-notified sends.
  for (int r : srcRanks) {
    if (r != ctx_->rank) {
      auto* pair = static_cast<TcpPair*>(ctx_->getPair(r));
      if (pair != nullptr &&
          pair->takeStashLocked(this, slot, offset, nbytes)) {
        return;
      }
    }
  }
  for (int r : srcRanks) {
    if (r != ctx_->rank && ctx_->consumeUnclaimedLocked(slot, r)) {
      auto* pair = static_cast<TcpPair*>(ctx_->getPair(r));
      pair->postRecv(this, slot, offset, nbytes, /*fromClaim=*/true);
      return;
    }
  }
  ctx_->anyRecvs_.push_back({this, slot, offset, nbytes, srcRanks});
}

bool TcpUnboundBuffer::waitRecv(
    int* srcRank,
    std::chrono::milliseconds timeout) {
  std::unique_lock<std::mutex> lock(ctx_->mu_);
  if (timeout.count() < 0) {
    timeout = ctx_->getTimeout();
  }
  auto pred = [&] {
    return !recvCompletions_.empty() || error_ != nullptr || abortRecv_;
  };
  if (timeout.count() < 0) {
    ctx_->cv_.wait(lock, pred);
  } else if (!ctx_->cv_.wait_for(lock, timeout, pred)) {
    lock.unlock();
    ctx_->signalException("waitRecv timeout");
    throw TimeoutException("unbound buffer waitRecv timed out");
  }
  if (!recvCompletions_.empty()) {
    if (srcRank != nullptr) {
      *srcRank = recvCompletions_.front();
    }
    recvCompletions_.pop_front();
    return true;
  }
  if (abortRecv_) {
    abortRecv_ = false;
    return false;
  }
  std::rethrow_exception(error_);
}

bool TcpUnboundBuffer::waitSend(std::chrono::milliseconds timeout) {
  std::unique_lock<std::mutex> lock(ctx_->mu_);
  if (timeout.count() < 0) {
    timeout = ctx_->getTimeout();
  }
  auto pred = [&] {
    return sendCompletions_ > 0 || error_ != nullptr || abortSend_;
  };
  if (timeout.count() < 0) {
    ctx_->cv_.wait(lock, pred);
  } else if (!ctx_->cv_.wait_for(lock, timeout, pred)) {
    lock.unlock();
    ctx_->signalException("waitSend timeout");
    throw TimeoutException("unbound buffer waitSend timed out");
  }
  if (sendCompletions_ > 0) {
    sendCompletions_--;
    return true;
  }
  if (abortSend_) {
    abortSend_ = false;
    return false;
  }
  std::rethrow_exception(error_);
}

bool TcpUnboundBuffer::tryWaitRecv(
    int* srcRank,
    std::chrono::milliseconds timeout) {
  std::unique_lock<std::mutex> lock(ctx_->mu_);
  if (timeout.count() < 0) {
    timeout = ctx_->getTimeout();
  }
  auto pred = [&] {
    return !recvCompletions_.empty() || error_ != nullptr || abortRecv_;
  };
  if (timeout.count() < 0) {
    ctx_->cv_.wait(lock, pred);
  } else if (!ctx_->cv_.wait_for(lock, timeout, pred)) {
    return false; // timed out; context left intact (probe semantics)
  }
  if (!recvCompletions_.empty()) {
    if (srcRank != nullptr) {
      *srcRank = recvCompletions_.front();
    }
    recvCompletions_.pop_front();
    return true;
  }
  if (abortRecv_) {
    abortRecv_ = false;
    return false;
  }
  std::rethrow_exception(error_);
}

void TcpUnboundBuffer::abortWaitRecv() {
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  abortRecv_ = true;
  ctx_->cv_.notify_all();
}

void TcpUnboundBuffer::abortWaitSend() {
  std::lock_guard<std::mutex> lock(ctx_->mu_);
  abortSend_ = true;
  ctx_->cv_.notify_all();
}

} // namespace tcp
} // namespace glooamd
