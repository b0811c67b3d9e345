// ContextFactory: re-rendezvous over an existing connected context,
// without a store — pair addresses travel over the backing context's
// tagged send/recv. Capability parity with reference
// gloo/rendezvous/context.cc:37-162 (used by the benchmark tool to build
// per-thread contexts).
#pragma once

#include <memory>

#include "context.h"

namespace glooamd {

class ContextFactory {
 public:
  explicit ContextFactory(std::shared_ptr<Context> backingContext);

  // Collective: every rank calls makeContext the same number of times.
  std::shared_ptr<Context> makeContext(
      std::shared_ptr<transport::Device> device);

 private:
  std::shared_ptr<Context> backing_;
};

} // namespace glooamd
