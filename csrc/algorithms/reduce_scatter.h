// Legacy ReduceScatterHalvingDoubling<T>: halving-doubling
// reduce-scatter followed by redistribution to the caller's per-rank
// element counts. Capability parity with reference gloo/reduce_scatter.h
// (re-derived: the bit-reversed block bookkeeping is replaced by an
// explicit overlap-and-forward redistribution pass).
#pragma once

#include <cstring>
#include <vector>

#include "algorithms/algorithm.h"
#include "collectives/schedule.h"
#include "common/utils.h"
#include "types.h"

namespace glooamd {

template <typename T>
class ReduceScatterHalvingDoubling : public Algorithm {
 public:
  // After run(), rank r holds recvElems[r] reduced elements (the global
  // prefix block) at the start of ptrs[0].
  ReduceScatterHalvingDoubling(
      const std::shared_ptr<Context>& context,
      const std::vector<T*>& ptrs,
      size_t count,
      const std::vector<int>& recvElems,
      typename ReductionFunction<T>::Function* fn =
          &ReductionFunction<T>::sumFn)
      : Algorithm(context),
        ptrs_(ptrs),
        count_(count),
        recvElems_(recvElems),
        fn_(fn) {
    const int P = context_->size;
    pow2_ = 1;
    steps_ = 0;
    while (pow2_ * 2 <= P) {
      pow2_ *= 2;
      steps_++;
    }
    slot_ = Slot::build(SlotPrefix::kReduceScatter, context_->nextSlot());
    tmp_ = makeAligned(std::max<size_t>(count_ * sizeof(T), 64));
    scratch_ = makeAligned(std::max<size_t>(count_ * sizeof(T), 64));
    buf_ = context_->createUnboundBuffer(
        scratch_.get(), std::max<size_t>(count_ * sizeof(T), 64));
    tmpBuf_ = context_->createUnboundBuffer(
        tmp_.get(), std::max<size_t>(count_ * sizeof(T), 64));
    outBuf_ = context_->createUnboundBuffer(
        ptrs_.empty() ? nullptr : ptrs_[0], count_ * sizeof(T));
  }

  void run() override {
    for (size_t i = 1; i < ptrs_.size(); i++) {
      fn_(ptrs_[0], ptrs_[i], count_);
    }
    const int P = context_->size;
    const int r = context_->rank;
    const size_t es = sizeof(T);
    const int extras = P - pow2_;
    if (P == 1 || count_ == 0) {
      return;
    }
    T* scratch = reinterpret_cast<T*>(scratch_.get());
    std::memcpy(scratch, ptrs_[0], count_ * es);

    // Pre-fold extras into partners (as in AllreduceHalvingDoubling).
    if (r >= pow2_) {
      buf_->send(r - pow2_, slot_, 0, count_ * es);
      buf_->waitSend();
    } else if (r < extras) {
      tmpBuf_->recv(r + pow2_, slot_, 0, count_ * es);
      tmpBuf_->waitRecv();
      fn_(scratch, reinterpret_cast<const T*>(tmp_.get()), count_);
    }

    sched::Seg span{0, count_};
    if (r < pow2_) {
      for (int t = 0; t < steps_; t++) {
        const int peer = r ^ (1 << t);
        const int bit = (r >> t) & 1;
        sched::Seg kp = sched::subspanOf(span, bit, 2);
        sched::Seg gv = sched::subspanOf(span, 1 - bit, 2);
        tmpBuf_->recv(peer, slot_ + 1 + t, kp.off * es, kp.len * es);
        buf_->send(peer, slot_ + 1 + t, gv.off * es, gv.len * es);
        tmpBuf_->waitRecv();
        if (kp.len > 0) {
          fn_(scratch + kp.off,
              reinterpret_cast<const T*>(tmp_.get() + kp.off * es),
              kp.len);
        }
        buf_->waitSend();
        span = kp;
      }
    } else {
      span = {0, 0}; // extras own nothing after the exchange
    }

    // Redistribution: rank d's target range is the prefix block
    // [pre[d], pre[d] + recvElems[d]); every owner forwards the overlap
    // of its span with each target range; targets receive into ptrs[0].
    std::vector<size_t> pre(P + 1, 0);
    for (int d = 0; d < P; d++) {
      pre[d + 1] = pre[d] + static_cast<size_t>(recvElems_[d]);
    }
    const uint64_t rslot = slot_ + 1 + steps_;

    // Post receives for my target range from every owner whose span
    // overlaps it. Spans after HD-RS partition [0, count): owner o < pow2
    // holds spanOfRank(o).
    size_t myLo = pre[r];
    size_t myHi = std::min(pre[r + 1], count_);
    int expected = 0;
    for (int o = 0; o < pow2_; o++) {
      sched::Seg os = spanOfRank(o);
      size_t lo = std::max(myLo, os.off);
      size_t hi = std::min(myHi, os.off + os.len);
      if (lo < hi) {
        if (o == r) {
          std::memcpy(ptrs_[0] + (lo - myLo), scratch + lo, (hi - lo) * es);
        } else {
          outBuf_->recv(o, rslot, (lo - myLo) * es, (hi - lo) * es);
          expected++;
        }
      }
    }
    // Send my span's overlap with every target.
    for (int d = 0; d < P; d++) {
      if (d == r) {
        continue;
      }
      size_t dLo = pre[d];
      size_t dHi = std::min(pre[d + 1], count_);
      size_t lo = std::max(dLo, span.off);
      size_t hi = std::min(dHi, span.off + span.len);
      if (lo < hi) {
        buf_->send(d, rslot, lo * es, (hi - lo) * es);
      }
    }
    for (int i = 0; i < expected; i++) {
      outBuf_->waitRecv();
    }
    // Drain sends.
    for (int d = 0; d < P; d++) {
      if (d == r) {
        continue;
      }
      size_t dLo = pre[d];
      size_t dHi = std::min(pre[d + 1], count_);
      size_t lo = std::max(dLo, span.off);
      size_t hi = std::min(dHi, span.off + span.len);
      if (lo < hi) {
        buf_->waitSend();
      }
    }
  }

 private:
  // Span owned by rank o (< pow2) after the reduce-scatter stage.
  sched::Seg spanOfRank(int o) const {
    sched::Seg span{0, count_};
    for (int t = 0; t < steps_; t++) {
      span = sched::subspanOf(span, (o >> t) & 1, 2);
    }
    return span;
  }

  std::vector<T*> ptrs_;
  size_t count_;
  std::vector<int> recvElems_;
  typename ReductionFunction<T>::Function* fn_;
  int pow2_;
  int steps_;
  uint64_t slot_;
  AlignedPtr tmp_;
  AlignedPtr scratch_;
  std::unique_ptr<transport::UnboundBuffer> buf_;
  std::unique_ptr<transport::UnboundBuffer> tmpBuf_;
  std::unique_ptr<transport::UnboundBuffer> outBuf_;
};

} // namespace glooamd
