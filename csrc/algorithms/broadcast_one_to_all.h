// Legacy BroadcastOneToAll<T>: root writes the full buffer into every
// peer's registered buffer; clear-to-send notifications make the
// algorithm reusable across runs.
// Capability parity with reference gloo/broadcast_one_to_all.h.
#pragma once

#include <cstring>
#include <vector>

#include "algorithms/algorithm.h"
#include "types.h"

namespace glooamd {

template <typename T>
class BroadcastOneToAll : public Algorithm {
 public:
  BroadcastOneToAll(
      const std::shared_ptr<Context>& context,
      const std::vector<T*>& ptrs,
      size_t count,
      int rootRank = 0)
      : Algorithm(context), ptrs_(ptrs), count_(count), root_(rootRank) {
    const size_t bytes = count_ * sizeof(T);
    const uint64_t slotData =
        Slot::build(SlotPrefix::kBroadcast, context_->nextSlot());
    const uint64_t slotNotify =
        Slot::build(SlotPrefix::kBroadcast, context_->nextSlot());
    if (context_->rank == root_) {
      for (int i = 0; i < context_->size; i++) {
        if (i == root_) {
          continue;
        }
        auto* pair = context_->getPair(i);
        dataBufs_.push_back(pair->createSendBuffer(slotData, ptrs_[0], bytes));
        notifyBufs_.push_back(
            pair->createRecvBuffer(slotNotify, &byte_, sizeof(byte_)));
      }
    } else {
      auto* pair = context_->getPair(root_);
      dataBufs_.push_back(pair->createRecvBuffer(slotData, ptrs_[0], bytes));
      notifyBufs_.push_back(
          pair->createSendBuffer(slotNotify, &byte_, sizeof(byte_)));
    }
  }

  void run() override {
    if (context_->size > 1) {
      if (context_->rank == root_) {
        if (!first_) {
          // Clear-to-send: peers consumed the previous run.
          for (auto& b : notifyBufs_) {
            b->waitRecv();
          }
        }
        first_ = false;
        for (auto& b : dataBufs_) {
          b->send(0, count_ * sizeof(T));
        }
        for (auto& b : dataBufs_) {
          b->waitSend();
        }
      } else {
        dataBufs_[0]->waitRecv();
        notifyBufs_[0]->send(0, sizeof(byte_));
        notifyBufs_[0]->waitSend();
      }
    }
    for (size_t i = 1; i < ptrs_.size(); i++) {
      std::memcpy(ptrs_[i], ptrs_[0], count_ * sizeof(T));
    }
  }

 private:
  std::vector<T*> ptrs_;
  size_t count_;
  int root_;
  bool first_{true};
  char byte_{0};
  std::vector<std::unique_ptr<transport::Buffer>> dataBufs_;
  std::vector<std::unique_ptr<transport::Buffer>> notifyBufs_;
};

} // namespace glooamd
