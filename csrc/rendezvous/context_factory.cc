#include "rendezvous/context_factory.h"

#include <cstring>

#include "common/logging.h"
#include "transport/tcp/context.h"
#include "types.h"

namespace glooamd {

namespace {
constexpr size_t kMaxAddr = 512; // fixed-size address slot on the wire
} // namespace

ContextFactory::ContextFactory(std::shared_ptr<Context> backingContext)
    : backing_(std::move(backingContext)) {
  GA_ENFORCE(backing_->transportContext() != nullptr, "backing not connected");
}

std::shared_ptr<Context> ContextFactory::makeContext(
    std::shared_ptr<transport::Device> device) {
  const int rank = backing_->rank;
  const int size = backing_->size;
  auto transportCtx = device->createContext(rank, size);
  transportCtx->setTimeout(backing_->getTimeout());
  auto* tcpCtx = dynamic_cast<tcp::TcpContext*>(transportCtx.get());
  GA_ENFORCE(tcpCtx != nullptr, "unsupported transport for ContextFactory");

  // Create all pairs, then swap addresses with each peer over the
  // backing context.
  std::vector<std::vector<char>> myAddr(size);
  for (int i = 0; i < size; i++) {
    if (i == rank) {
      continue;
    }
    myAddr[i] = transportCtx->createPair(i)->address().bytes();
    GA_ENFORCE_LE(myAddr[i].size() + 8, kMaxAddr);
  }

  const uint64_t slot =
      Slot::build(SlotPrefix::kContextFactory, backing_->nextSlot());
  std::vector<char> sendBlob(static_cast<size_t>(size) * kMaxAddr, 0);
  std::vector<char> recvBlob(static_cast<size_t>(size) * kMaxAddr, 0);
  for (int i = 0; i < size; i++) {
    if (i == rank) {
      continue;
    }
    uint64_t len = myAddr[i].size();
    std::memcpy(&sendBlob[i * kMaxAddr], &len, 8);
    std::memcpy(&sendBlob[i * kMaxAddr + 8], myAddr[i].data(), len);
  }
  auto sendBuf =
      backing_->createUnboundBuffer(sendBlob.data(), sendBlob.size());
  auto recvBuf =
      backing_->createUnboundBuffer(recvBlob.data(), recvBlob.size());
  for (int i = 0; i < size; i++) {
    if (i == rank) {
      continue;
    }
    recvBuf->recv(i, slot, static_cast<size_t>(i) * kMaxAddr, kMaxAddr);
  }
  for (int i = 0; i < size; i++) {
    if (i == rank) {
      continue;
    }
    sendBuf->send(i, slot, static_cast<size_t>(i) * kMaxAddr, kMaxAddr);
  }
  for (int i = 0; i < size - 1; i++) {
    recvBuf->waitRecv();
  }
  for (int i = 0; i < size - 1; i++) {
    sendBuf->waitSend();
  }

  for (int i = 0; i < size; i++) {
    if (i == rank) {
      continue;
    }
    uint64_t len;
    std::memcpy(&len, &recvBlob[i * kMaxAddr], 8);
    GA_ENFORCE_LE(len, kMaxAddr - 8);
    std::vector<char> addr(
        &recvBlob[i * kMaxAddr + 8], &recvBlob[i * kMaxAddr + 8] + len);
    transportCtx->getPair(i)->connect(addr);
  }

  auto ctx = std::make_shared<Context>(rank, size, backing_->base);
  ctx->setTimeout(backing_->getTimeout());
  ctx->setTransportContext(std::move(transportCtx));
  return ctx;
}

} // namespace glooamd
