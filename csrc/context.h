// Top-level communicator Context.
//
// Capability parity with reference gloo/context.h:27-65 +
// gloo/rendezvous/context.cc:25-35: rank/size/base, the transport context
// with its full mesh of pairs, the slot allocator for concurrent
// collectives, and the default operation timeout. connectFullMesh is a
// member here (the reference splits it into rendezvous::Context).
#pragma once

#include <atomic>
#include <chrono>
#include <memory>

#include "common/store.h"
#include "transport/transport.h"

namespace glooamd {

class Context {
 public:
  Context(int rank, int size, int base = 2);
  virtual ~Context(); // returns pooled HIP streams to the freelist

  const int rank;
  const int size;
  int base; // branching factor for bcube/tree algorithms

  // Rendezvous: create pairs for every peer, exchange addresses through
  // the store, connect the full mesh.
  void connectFullMesh(
      IStore& store,
      std::shared_ptr<transport::Device> device);

  // Adopt an existing transport context (used by tests / factories).
  void setTransportContext(std::shared_ptr<transport::Context> ctx) {
    transportContext_ = std::move(ctx);
  }
  const std::shared_ptr<transport::Context>& transportContext() const {
    return transportContext_;
  }

  transport::Pair* getPair(int rank);
  std::unique_ptr<transport::UnboundBuffer> createUnboundBuffer(
      void* ptr,
      size_t size);

  // Monotonic tag allocator so concurrent collectives on one context use
  // disjoint slot ranges.
  uint32_t nextSlot(uint32_t numToSkip = 1) {
    return slotCounter_.fetch_add(numToSkip);
  }
  // Resync support: after a rank-divergent failure (e.g. a collective
  // constructor threw mid-setup), ranks agree on max(slotCounter) out of
  // band and jump every counter there so tag allocation realigns.
  uint32_t slotCounter() const {
    return slotCounter_.load();
  }
  void resetSlotCounter(uint32_t v) {
    slotCounter_.store(v);
  }

  void setTimeout(std::chrono::milliseconds timeout);
  std::chrono::milliseconds getTimeout() const {
    return timeout_;
  }

  void closeConnections();

 private:
  std::shared_ptr<transport::Context> transportContext_;
  std::atomic<uint32_t> slotCounter_{1u << 20}; // above user tag space
  std::chrono::milliseconds timeout_{transport::kDefaultTimeout};
};

} // namespace glooamd
