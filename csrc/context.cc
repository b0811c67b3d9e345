#include "context.h"

#include "common/logging.h"
#include "transport/tcp/context.h"

namespace glooamd {

namespace hip {
// Defined in hip/core.cc (always linked); walks an empty pool when no
// device work ever ran.
void releasePooledStreams(const void* key);
} // namespace hip

Context::~Context() {
  hip::releasePooledStreams(this);
}

Context::Context(int rank, int size, int base)
    : rank(rank), size(size), base(base) {
  GA_ENFORCE_GE(rank, 0);
  GA_ENFORCE_LT(rank, size);
  GA_ENFORCE_GE(size, 1);
}

void Context::connectFullMesh(
    IStore& store,
    std::shared_ptr<transport::Device> device) {
  auto transportCtx = device->createContext(rank, size);
  transportCtx->setTimeout(timeout_);
  // Full-mesh bootstrap is implemented by the concrete transport context.
  auto* tcpCtx = dynamic_cast<tcp::TcpContext*>(transportCtx.get());
  GA_ENFORCE(tcpCtx != nullptr, "unsupported transport context type");
  tcpCtx->connectFullMesh(store);
  transportContext_ = std::move(transportCtx);
}

transport::Pair* Context::getPair(int rank) {
  GA_ENFORCE(transportContext_ != nullptr, "context not connected");
  return transportContext_->getPair(rank);
}

std::unique_ptr<transport::UnboundBuffer> Context::createUnboundBuffer(
    void* ptr,
    size_t size) {
  GA_ENFORCE(transportContext_ != nullptr, "context not connected");
  return transportContext_->createUnboundBuffer(ptr, size);
}

void Context::setTimeout(std::chrono::milliseconds timeout) {
  timeout_ = timeout;
  if (transportContext_) {
    transportContext_->setTimeout(timeout);
  }
}

void Context::closeConnections() {
  if (!transportContext_) {
    return;
  }
  for (int i = 0; i < size; i++) {
    auto* p = transportContext_->getPair(i);
    if (p != nullptr) {
      p->close();
    }
  }
}

} // namespace glooamd
