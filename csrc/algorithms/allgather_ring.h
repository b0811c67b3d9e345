// Legacy AllgatherRing<T>: ring allgather into per-rank output segments
// using one-sided bound-buffer writes with remote offsets.
// Capability parity with reference gloo/allgather_ring.h.
#pragma once

#include <cstring>
#include <vector>

#include "algorithms/algorithm.h"
#include "types.h"

namespace glooamd {

template <typename T>
class AllgatherRing : public Algorithm {
 public:
  // out must hold size * ptrs.size() * count elements; rank r's inputs
  // occupy block r (in input-pointer order).
  AllgatherRing(
      const std::shared_ptr<Context>& context,
      const std::vector<const T*>& inPtrs,
      T* outPtr,
      size_t count)
      : Algorithm(context),
        inPtrs_(inPtrs),
        outPtr_(outPtr),
        count_(count),
        blockElems_(count * inPtrs.size()) {
    const int P = context_->size;
    if (P == 1) {
      return;
    }
    const size_t outBytes = blockElems_ * P * sizeof(T);
    const uint64_t slotData =
        Slot::build(SlotPrefix::kAllgather, context_->nextSlot());
    const uint64_t slotNotify =
        Slot::build(SlotPrefix::kAllgather, context_->nextSlot());
    sendDataBuf_ =
        getRightPair()->createSendBuffer(slotData, outPtr_, outBytes);
    recvDataBuf_ =
        getLeftPair()->createRecvBuffer(slotData, outPtr_, outBytes);
    sendNotificationBuf_ = getLeftPair()->createSendBuffer(
        slotNotify, &notifyByte_, sizeof(notifyByte_));
    recvNotificationBuf_ = getRightPair()->createRecvBuffer(
        slotNotify, &notifyByte_, sizeof(notifyByte_));
  }

  void run() override {
    const int P = context_->size;
    const int r = context_->rank;
    const size_t blockBytes = blockElems_ * sizeof(T);
    // Place own inputs.
    for (size_t i = 0; i < inPtrs_.size(); i++) {
      std::memcpy(
          outPtr_ + r * blockElems_ + i * count_,
          inPtrs_[i],
          count_ * sizeof(T));
    }
    if (P == 1 || blockBytes == 0) {
      return;
    }
    for (int i = 0; i < P - 1; i++) {
      if (i > 0) {
        recvNotificationBuf_->waitRecv();
      }
      const int sendBlock = (r - i + 2 * P) % P;
      // One-sided write into the same block offset of the peer's output.
      sendDataBuf_->send(
          sendBlock * blockBytes, blockBytes, sendBlock * blockBytes);
      recvDataBuf_->waitRecv();
      sendDataBuf_->waitSend();
      sendNotificationBuf_->send(0, sizeof(notifyByte_));
      sendNotificationBuf_->waitSend();
    }
    recvNotificationBuf_->waitRecv();
  }

 private:
  std::vector<const T*> inPtrs_;
  T* outPtr_;
  size_t count_;
  size_t blockElems_;
  char notifyByte_{0};
  std::unique_ptr<transport::Buffer> sendDataBuf_;
  std::unique_ptr<transport::Buffer> recvDataBuf_;
  std::unique_ptr<transport::Buffer> sendNotificationBuf_;
  std::unique_ptr<transport::Buffer> recvNotificationBuf_;
};

} // namespace glooamd
