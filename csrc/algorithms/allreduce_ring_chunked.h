// Legacy AllreduceRingChunked<T>: pipelined segmented ring over bound
// buffers — two segments per rank in flight through double-buffered
// inboxes (capability parity with reference gloo/allreduce_ring_chunked.h:
// 101-236; schedule re-derived as a unified reduce-scatter + allgather
// step list, two wire segments per block).
#pragma once

#include <cstring>
#include <vector>

#include "algorithms/algorithm.h"
#include "collectives/schedule.h"
#include "common/utils.h"
#include "types.h"

namespace glooamd {

template <typename T>
class AllreduceRingChunked : public Algorithm {
 public:
  AllreduceRingChunked(
      const std::shared_ptr<Context>& context,
      const std::vector<T*>& ptrs,
      size_t count,
      typename ReductionFunction<T>::Function* fn =
          &ReductionFunction<T>::sumFn)
      : Algorithm(context),
        ptrs_(ptrs),
        count_(count),
        fn_(fn) {
    const int P = context_->size;
    if (P == 1) {
      return;
    }
    // Two segments per block: 2P wire chunks, as in the reference.
    segCapBytes_ = (((count_ + P - 1) / P + 1) / 2 + 1) * sizeof(T);
    for (int par = 0; par < 2; par++) {
      inbox_[par] = makeAligned(segCapBytes_);
      const uint64_t slotData =
          Slot::build(SlotPrefix::kAllreduce, context_->nextSlot());
      const uint64_t slotNotify =
          Slot::build(SlotPrefix::kAllreduce, context_->nextSlot());
      sendDataBuf_[par] = getRightPair()->createSendBuffer(
          slotData, ptrs_[0], count_ * sizeof(T));
      recvDataBuf_[par] = getLeftPair()->createRecvBuffer(
          slotData, inbox_[par].get(), segCapBytes_);
      sendNotificationBuf_[par] = getLeftPair()->createSendBuffer(
          slotNotify, &notifyByte_, sizeof(notifyByte_));
      recvNotificationBuf_[par] = getRightPair()->createRecvBuffer(
          slotNotify, &notifyByte_, sizeof(notifyByte_));
    }
  }

  void run() override {
    for (size_t i = 1; i < ptrs_.size(); i++) {
      fn_(ptrs_[0], ptrs_[i], count_);
    }
    const int P = context_->size;
    const int r = context_->rank;
    const size_t es = sizeof(T);
    if (P > 1 && count_ > 0) {
      constexpr int S = 2;
      const int K1 = (P - 1) * S;
      char* out = reinterpret_cast<char*>(ptrs_[0]);
      bool firstUse[2] = {true, true};
      for (int k = 0; k < 2 * K1; k++) {
        const bool reduceStep = k < K1;
        const int kk = reduceStep ? k : k - K1;
        const int i = kk / S;
        const int s = kk % S;
        const int par = k & 1;
        const sched::Seg send = reduceStep
            ? sched::segmentOf(count_, P, (r - i + P) % P, s, S)
            : sched::segmentOf(count_, P, (r + 1 - i + P) % P, s, S);
        const sched::Seg recv = reduceStep
            ? sched::segmentOf(count_, P, (r - i - 1 + 2 * P) % P, s, S)
            : sched::segmentOf(count_, P, (r - i + P) % P, s, S);
        if (!firstUse[par]) {
          recvNotificationBuf_[par]->waitRecv();
        }
        firstUse[par] = false;
        sendDataBuf_[par]->send(send.off * es, send.len * es, 0);
        recvDataBuf_[par]->waitRecv();
        if (recv.len > 0) {
          if (reduceStep) {
            fn_(reinterpret_cast<T*>(out + recv.off * es),
                reinterpret_cast<const T*>(inbox_[par].get()),
                recv.len);
          } else {
            std::memcpy(out + recv.off * es, inbox_[par].get(),
                        recv.len * es);
          }
        }
        sendDataBuf_[par]->waitSend();
        sendNotificationBuf_[par]->send(0, sizeof(notifyByte_));
        sendNotificationBuf_[par]->waitSend();
      }
      // Drain one trailing notification per parity.
      recvNotificationBuf_[0]->waitRecv();
      recvNotificationBuf_[1]->waitRecv();
    }
    for (size_t i = 1; i < ptrs_.size(); i++) {
      std::memcpy(ptrs_[i], ptrs_[0], count_ * sizeof(T));
    }
  }

 private:
  std::vector<T*> ptrs_;
  size_t count_;
  typename ReductionFunction<T>::Function* fn_;
  size_t segCapBytes_{0};
  AlignedPtr inbox_[2];
  char notifyByte_{0};
  std::unique_ptr<transport::Buffer> sendDataBuf_[2];
  std::unique_ptr<transport::Buffer> recvDataBuf_[2];
  std::unique_ptr<transport::Buffer> sendNotificationBuf_[2];
  std::unique_ptr<transport::Buffer> recvNotificationBuf_[2];
};

} // namespace glooamd
