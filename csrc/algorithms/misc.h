// Legacy PairwiseExchange (benchmark traffic pattern), AllreduceBcube
// and AllreduceLocal.
// Capability parity with reference gloo/pairwise_exchange.h,
// gloo/allreduce_bcube.h, gloo/allreduce_local.{h,cc}.
#pragma once

#include <cstring>
#include <vector>

#include "algorithms/algorithm.h"
#include "collectives/collectives.h"
#include "collectives/reduce_fns.h"
#include "common/utils.h"
#include "math.h"
#include "types.h"

namespace glooamd {

// Halving-doubling-shaped traffic generator: numSteps exchanges of
// bytes/numSteps with peers rank ^ 2^i. No computation.
class PairwiseExchange : public Algorithm {
 public:
  PairwiseExchange(
      const std::shared_ptr<Context>& context,
      size_t bytes,
      int numSteps)
      : Algorithm(context), numSteps_(numSteps) {
    GA_ENFORCE_GT(numSteps_, 0);
    GA_ENFORCE_LE(
        (1 << numSteps_), context_->size, "too many steps for size");
    chunkBytes_ = std::max<size_t>(bytes / numSteps_, 1);
    sendBuf_ = makeAligned(chunkBytes_);
    recvBuf_ = makeAligned(chunkBytes_);
    const uint64_t slot =
        Slot::build(SlotPrefix::kSendRecv, context_->nextSlot());
    for (int i = 0; i < numSteps_; i++) {
      const int peer = context_->rank ^ (1 << i);
      auto* pair = context_->getPair(peer);
      sendBufs_.push_back(
          pair->createSendBuffer(slot + i, sendBuf_.get(), chunkBytes_));
      recvBufs_.push_back(
          pair->createRecvBuffer(slot + i, recvBuf_.get(), chunkBytes_));
    }
  }

  void run() override {
    for (int i = 0; i < numSteps_; i++) {
      sendBufs_[i]->send(0, chunkBytes_);
      recvBufs_[i]->waitRecv();
      sendBufs_[i]->waitSend();
    }
  }

 private:
  int numSteps_;
  size_t chunkBytes_;
  AlignedPtr sendBuf_;
  AlignedPtr recvBuf_;
  std::vector<std::unique_ptr<transport::Buffer>> sendBufs_;
  std::vector<std::unique_ptr<transport::Buffer>> recvBufs_;
};

// Base-b hypercube allreduce; wraps the v2 bcube engine (the legacy
// class surface over the same schedule).
template <typename T>
class AllreduceBcube : public Algorithm {
 public:
  AllreduceBcube(
      const std::shared_ptr<Context>& context,
      const std::vector<T*>& ptrs,
      size_t count,
      typename ReductionFunction<T>::Function* fn =
          &ReductionFunction<T>::sumFn)
      : Algorithm(context), ptrs_(ptrs), count_(count), fn_(fn) {
    tag_ = context_->nextSlot();
  }

  void run() override {
    AllreduceOptions opts(context_);
    for (auto* p : ptrs_) {
      opts.outputs.push_back(p);
    }
    opts.elements = count_;
    opts.elementSize = sizeof(T);
    auto* fn = fn_;
    opts.reduce = [fn](void* dst, const void* a, const void* b, size_t n) {
      if (dst != a) {
        std::memcpy(dst, a, n * sizeof(T));
      }
      fn(static_cast<T*>(dst), static_cast<const T*>(b), n);
    };
    opts.algorithm = AllreduceOptions::Algorithm::BCUBE;
    opts.tag = tag_;
    allreduce(opts);
  }

 private:
  std::vector<T*> ptrs_;
  size_t count_;
  typename ReductionFunction<T>::Function* fn_;
  uint32_t tag_;
};

// Single-process multi-pointer reduce + broadcast.
template <typename T>
class AllreduceLocal : public Algorithm {
 public:
  AllreduceLocal(
      const std::shared_ptr<Context>& context,
      const std::vector<T*>& ptrs,
      size_t count,
      typename ReductionFunction<T>::Function* fn =
          &ReductionFunction<T>::sumFn)
      : Algorithm(context), ptrs_(ptrs), count_(count), fn_(fn) {}

  void run() override {
    for (size_t i = 1; i < ptrs_.size(); i++) {
      fn_(ptrs_[0], ptrs_[i], count_);
    }
    for (size_t i = 1; i < ptrs_.size(); i++) {
      std::memcpy(ptrs_[i], ptrs_[0], count_ * sizeof(T));
    }
  }

 private:
  std::vector<T*> ptrs_;
  size_t count_;
  typename ReductionFunction<T>::Function* fn_;
};

} // namespace glooamd
