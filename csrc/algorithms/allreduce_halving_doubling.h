// Legacy AllreduceHalvingDoubling<T>: recursive vector-halving /
// distance-doubling reduce-scatter with a mirrored allgather.
// Capability parity with reference gloo/allreduce_halving_doubling.h.
// Non-power-of-2 sizes use pre/post folding (extra ranks fold their
// contribution into a partner before the power-of-2 exchange and
// receive the result afterwards) instead of the reference's binary
// blocks — same asymptotics, far less machinery.
#pragma once

#include <cstring>
#include <vector>

#include "algorithms/algorithm.h"
#include "collectives/schedule.h"
#include "common/utils.h"
#include "types.h"

namespace glooamd {

template <typename T>
class AllreduceHalvingDoubling : public Algorithm {
 public:
  AllreduceHalvingDoubling(
      const std::shared_ptr<Context>& context,
      const std::vector<T*>& ptrs,
      size_t count,
      typename ReductionFunction<T>::Function* fn =
          &ReductionFunction<T>::sumFn)
      : Algorithm(context), ptrs_(ptrs), count_(count), fn_(fn) {
    const int P = context_->size;
    pow2_ = 1;
    steps_ = 0;
    while (pow2_ * 2 <= P) {
      pow2_ *= 2;
      steps_++;
    }
    slot_ = Slot::build(SlotPrefix::kAllreduce, context_->nextSlot());
    tmp_ = makeAligned(std::max<size_t>(count_ * sizeof(T), 64));
    buf_ = context_->createUnboundBuffer(
        ptrs_.empty() ? nullptr : ptrs_[0], count_ * sizeof(T));
    tmpBuf_ = context_->createUnboundBuffer(
        tmp_.get(), std::max<size_t>(count_ * sizeof(T), 64));
  }

  void run() override {
    for (size_t i = 1; i < ptrs_.size(); i++) {
      fn_(ptrs_[0], ptrs_[i], count_);
    }
    const int P = context_->size;
    const int r = context_->rank;
    const size_t es = sizeof(T);
    const int extras = P - pow2_;
    if (P > 1 && count_ > 0) {
      // Pre-fold: ranks >= pow2 send everything to (r - pow2), which
      // reduces it in before the exchange.
      if (r >= pow2_) {
        buf_->send(r - pow2_, slot_, 0, count_ * es);
        buf_->waitSend();
      } else if (r < extras) {
        tmpBuf_->recv(r + pow2_, slot_, 0, count_ * es);
        tmpBuf_->waitRecv();
        fn_(ptrs_[0], reinterpret_cast<const T*>(tmp_.get()), count_);
      }

      if (r < pow2_) {
        sched::Seg span{0, count_};
        std::vector<sched::Seg> spanAt(steps_);
        // Reduce-scatter: halve the span each step.
        for (int t = 0; t < steps_; t++) {
          spanAt[t] = span;
          const int peer = r ^ (1 << t);
          const int bit = (r >> t) & 1;
          sched::Seg kp = sched::subspanOf(span, bit, 2);
          sched::Seg gv = sched::subspanOf(span, 1 - bit, 2);
          tmpBuf_->recv(peer, slot_ + 1 + t, kp.off * es, kp.len * es);
          buf_->send(peer, slot_ + 1 + t, gv.off * es, gv.len * es);
          tmpBuf_->waitRecv();
          if (kp.len > 0) {
            fn_(ptrs_[0] + kp.off,
                reinterpret_cast<const T*>(tmp_.get() + kp.off * es),
                kp.len);
          }
          buf_->waitSend();
          span = kp;
        }
        // Allgather mirror.
        for (int t = steps_ - 1; t >= 0; t--) {
          const int peer = r ^ (1 << t);
          const int bit = (r >> t) & 1;
          sched::Seg other = sched::subspanOf(spanAt[t], 1 - bit, 2);
          buf_->recv(peer, slot_ + 1 + steps_ + t, other.off * es,
                     other.len * es);
          buf_->send(peer, slot_ + 1 + steps_ + t, span.off * es,
                     span.len * es);
          buf_->waitRecv();
          buf_->waitSend();
          span = spanAt[t];
        }
      }

      // Post-fold: partners return the full result to the extra ranks.
      if (r < extras) {
        buf_->send(r + pow2_, slot_ + 1 + 2 * steps_, 0, count_ * es);
        buf_->waitSend();
      } else if (r >= pow2_) {
        buf_->recv(r - pow2_, slot_ + 1 + 2 * steps_, 0, count_ * es);
        buf_->waitRecv();
      }
    }
    for (size_t i = 1; i < ptrs_.size(); i++) {
      std::memcpy(ptrs_[i], ptrs_[0], count_ * sizeof(T));
    }
  }

 private:
  std::vector<T*> ptrs_;
  size_t count_;
  typename ReductionFunction<T>::Function* fn_;
  int pow2_;
  int steps_;
  uint64_t slot_;
  AlignedPtr tmp_;
  std::unique_ptr<transport::UnboundBuffer> buf_;
  std::unique_ptr<transport::UnboundBuffer> tmpBuf_;
};

} // namespace glooamd
