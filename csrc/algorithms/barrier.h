// Legacy barrier algorithms over bound buffers.
// Capability parity with reference gloo/barrier_all_to_all.h and
// gloo/barrier_all_to_one.h.
#pragma once

#include <vector>

#include "algorithms/algorithm.h"
#include "types.h"

namespace glooamd {

// Every rank sends a 1-byte notification to every other rank.
class BarrierAllToAll : public Algorithm {
 public:
  explicit BarrierAllToAll(const std::shared_ptr<Context>& context)
      : Algorithm(context) {
    const uint64_t slot =
        Slot::build(SlotPrefix::kBarrier, context_->nextSlot());
    for (int i = 0; i < context_->size; i++) {
      if (i == context_->rank) {
        continue;
      }
      auto* pair = context_->getPair(i);
      sendBufs_.push_back(
          pair->createSendBuffer(slot, &byte_, sizeof(byte_)));
      recvBufs_.push_back(
          pair->createRecvBuffer(slot, &byte_, sizeof(byte_)));
    }
  }

  void run() override {
    for (auto& b : sendBufs_) {
      b->send(0, sizeof(byte_));
    }
    for (auto& b : recvBufs_) {
      b->waitRecv();
    }
    for (auto& b : sendBufs_) {
      b->waitSend();
    }
  }

 private:
  char byte_{0};
  std::vector<std::unique_ptr<transport::Buffer>> sendBufs_;
  std::vector<std::unique_ptr<transport::Buffer>> recvBufs_;
};

// Star barrier through a root rank: gather notifications, then release.
class BarrierAllToOne : public Algorithm {
 public:
  BarrierAllToOne(const std::shared_ptr<Context>& context, int root = 0)
      : Algorithm(context), root_(root) {
    const uint64_t slotIn =
        Slot::build(SlotPrefix::kBarrier, context_->nextSlot());
    const uint64_t slotOut =
        Slot::build(SlotPrefix::kBarrier, context_->nextSlot());
    if (context_->rank == root_) {
      for (int i = 0; i < context_->size; i++) {
        if (i == root_) {
          continue;
        }
        auto* pair = context_->getPair(i);
        recvBufs_.push_back(
            pair->createRecvBuffer(slotIn, &byte_, sizeof(byte_)));
        sendBufs_.push_back(
            pair->createSendBuffer(slotOut, &byte_, sizeof(byte_)));
      }
    } else {
      auto* pair = context_->getPair(root_);
      sendBufs_.push_back(
          pair->createSendBuffer(slotIn, &byte_, sizeof(byte_)));
      recvBufs_.push_back(
          pair->createRecvBuffer(slotOut, &byte_, sizeof(byte_)));
    }
  }

  void run() override {
    if (context_->rank == root_) {
      for (auto& b : recvBufs_) {
        b->waitRecv();
      }
      for (auto& b : sendBufs_) {
        b->send(0, sizeof(byte_));
      }
      for (auto& b : sendBufs_) {
        b->waitSend();
      }
    } else {
      sendBufs_[0]->send(0, sizeof(byte_));
      sendBufs_[0]->waitSend();
      recvBufs_[0]->waitRecv();
    }
  }

 private:
  int root_;
  char byte_{0};
  std::vector<std::unique_ptr<transport::Buffer>> sendBufs_;
  std::vector<std::unique_ptr<transport::Buffer>> recvBufs_;
};

} // namespace glooamd
