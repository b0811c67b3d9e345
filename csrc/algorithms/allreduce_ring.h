// Legacy AllreduceRing<T>: P-1 rounds of full-buffer ring with
// inbox/outbox bound buffers and a notification "localized barrier".
// Capability parity with reference gloo/allreduce_ring.h:68-112,
// re-derived: data flows left->right, notifications right->left.
#pragma once

#include <cstring>
#include <vector>

#include "algorithms/algorithm.h"
#include "common/utils.h"
#include "types.h"

namespace glooamd {

template <typename T>
class AllreduceRing : public Algorithm {
 public:
  AllreduceRing(
      const std::shared_ptr<Context>& context,
      const std::vector<T*>& ptrs,
      size_t count,
      typename ReductionFunction<T>::Function* fn =
          &ReductionFunction<T>::sumFn)
      : Algorithm(context),
        ptrs_(ptrs),
        count_(count),
        bytes_(count * sizeof(T)),
        fn_(fn) {
    if (context_->size == 1) {
      return;
    }
    inbox_ = makeAligned(bytes_);
    outbox_ = makeAligned(bytes_);
    const uint64_t slotData =
        Slot::build(SlotPrefix::kAllreduce, context_->nextSlot());
    const uint64_t slotNotify =
        Slot::build(SlotPrefix::kAllreduce, context_->nextSlot());
    sendDataBuf_ =
        getRightPair()->createSendBuffer(slotData, outbox_.get(), bytes_);
    recvDataBuf_ =
        getLeftPair()->createRecvBuffer(slotData, inbox_.get(), bytes_);
    sendNotificationBuf_ = getLeftPair()->createSendBuffer(
        slotNotify, &notifyByte_, sizeof(notifyByte_));
    recvNotificationBuf_ = getRightPair()->createRecvBuffer(
        slotNotify, &notifyByte_, sizeof(notifyByte_));
  }

  void run() override {
    // Local reduction across input pointers.
    for (size_t i = 1; i < ptrs_.size(); i++) {
      fn_(ptrs_[0], ptrs_[i], count_);
    }
    const int P = context_->size;
    if (P > 1 && count_ > 0) {
      std::memcpy(outbox_.get(), ptrs_[0], bytes_);
      for (int i = 0; i < P - 1; i++) {
        if (i > 0) {
          // Right must have freed its inbox before we overwrite it.
          recvNotificationBuf_->waitRecv();
        }
        sendDataBuf_->send(0, bytes_);
        recvDataBuf_->waitRecv();
        fn_(ptrs_[0], static_cast<const T*>(
                          static_cast<void*>(inbox_.get())), count_);
        sendDataBuf_->waitSend(); // outbox flushed; safe to overwrite
        if (i < P - 2) {
          std::memcpy(outbox_.get(), inbox_.get(), bytes_);
        }
        sendNotificationBuf_->send(0, sizeof(notifyByte_));
        sendNotificationBuf_->waitSend();
      }
      // Absorb the final notification so the next run starts clean.
      recvNotificationBuf_->waitRecv();
    }
    // Local broadcast to remaining pointers.
    for (size_t i = 1; i < ptrs_.size(); i++) {
      std::memcpy(ptrs_[i], ptrs_[0], bytes_);
    }
  }

 private:
  std::vector<T*> ptrs_;
  size_t count_;
  size_t bytes_;
  typename ReductionFunction<T>::Function* fn_;
  AlignedPtr inbox_;
  AlignedPtr outbox_;
  char notifyByte_{0};
  std::unique_ptr<transport::Buffer> sendDataBuf_;
  std::unique_ptr<transport::Buffer> recvDataBuf_;
  std::unique_ptr<transport::Buffer> sendNotificationBuf_;
  std::unique_ptr<transport::Buffer> recvNotificationBuf_;
};

} // namespace glooamd
