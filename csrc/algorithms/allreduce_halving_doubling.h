// Legacy AllreduceHalvingDoubling<T>: recursive vector-halving /
// distance-doubling reduce-scatter with a mirrored allgather.
// Capability parity with reference gloo/allreduce_halving_doubling.h,
// including the BINARY BLOCKS scheme for non-power-of-2 sizes
// (reference :38-64): ranks split into power-of-2 blocks (binary
// decomposition of P, largest block first), each block reduce-scatters
// internally, smaller blocks cascade their owned segments into the next
// larger block (piecewise, by segment overlap), and the mirror reverses
// the cascade before each block's internal allgather. Unlike pre/post
// folding, every rank participates in the exchange and no single
// partner absorbs a full extra buffer.
//
// Re-derivation note: the reference maps owned segments with
// bit-reversal arithmetic; here each member's final segment is obtained
// by REPLAYING the halving recursion (ownedSpan), which any rank can
// evaluate for any other rank — overlap pieces then follow from plain
// interval intersection.
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
    // Binary decomposition of P, largest block first from rank 0:
    // P=11 -> blocks [0..7](8), [8..9](2), [10](1).
    const int P = context_->size;
    int off = 0;
    for (int bit = 30; bit >= 0; bit--) {
      if (P & (1 << bit)) {
        blocks_.push_back({off, 1 << bit});
        off += 1 << bit;
      }
    }
    for (size_t i = 0; i < blocks_.size(); i++) {
      if (context_->rank >= blocks_[i].off &&
          context_->rank < blocks_[i].off + blocks_[i].size) {
        myBlock_ = static_cast<int>(i);
      }
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
    if (P > 1 && count_ > 0) {
      runBlocks();
    }
    for (size_t i = 1; i < ptrs_.size(); i++) {
      std::memcpy(ptrs_[i], ptrs_[0], count_ * sizeof(T));
    }
  }

 private:
  struct Block {
    int off;
    int size;
  };

  static int log2of(int v) {
    int s = 0;
    while ((1 << s) < v) {
      s++;
    }
    return s;
  }

  // The segment member m of a B-sized block owns after the halving
  // recursion (any rank can evaluate this for any member).
  sched::Seg ownedSpan(int blockSize, int m) const {
    sched::Seg span{0, count_};
    for (int t = 0; t < log2of(blockSize); t++) {
      span = sched::subspanOf(span, (m >> t) & 1, 2);
    }
    return span;
  }

  static sched::Seg intersect(sched::Seg a, sched::Seg b) {
    const size_t lo = std::max(a.off, b.off);
    const size_t hi = std::min(a.off + a.len, b.off + b.len);
    return {lo, hi > lo ? hi - lo : 0};
  }

  void runBlocks() {
    const int r = context_->rank;
    const size_t es = sizeof(T);
    const Block me = blocks_[myBlock_];
    const int m = r - me.off; // rank within my block
    const int steps = log2of(me.size);

    // ---- phase 1: intra-block reduce-scatter ----
    sched::Seg span{0, count_};
    std::vector<sched::Seg> spanAt(steps);
    for (int t = 0; t < steps; t++) {
      spanAt[t] = span;
      const int peer = me.off + ((m ^ (1 << t)));
      const int bit = (m >> t) & 1;
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

    // ---- phase 2: cascade up (smallest block into the next larger,
    // sequentially, so contributions accumulate toward block 0) ----
    // Slot bases must be computed from a block-independent quantity
    // (sender and receiver live in different-sized blocks).
    const int maxSteps = log2of(blocks_[0].size);
    const uint64_t upBase = slot_ + 1 + 2 * maxSteps + 8;
    for (int i = static_cast<int>(blocks_.size()) - 1; i >= 1; i--) {
      const Block& src = blocks_[i];
      const Block& dst = blocks_[i - 1];
      const uint64_t s = upBase + i;
      if (myBlock_ == i) {
        // Send each overlap piece of my owned span to its dst owner;
        // wait sends only after all posts (no mutation in between).
        int posted = 0;
        for (int k = 0; k < dst.size; k++) {
          sched::Seg piece = intersect(span, ownedSpan(dst.size, k));
          if (piece.len > 0) {
            buf_->send(dst.off + k, s, piece.off * es, piece.len * es);
            posted++;
          }
        }
        while (posted-- > 0) {
          buf_->waitSend();
        }
      } else if (myBlock_ == i - 1) {
        // Receive overlap pieces from every src member, reduce in.
        std::vector<sched::Seg> pieces;
        for (int q = 0; q < src.size; q++) {
          sched::Seg piece = intersect(span, ownedSpan(src.size, q));
          if (piece.len > 0) {
            tmpBuf_->recv(src.off + q, s, piece.off * es, piece.len * es);
            pieces.push_back(piece);
          }
        }
        for (size_t w = 0; w < pieces.size(); w++) {
          tmpBuf_->waitRecv();
        }
        for (const auto& piece : pieces) {
          fn_(ptrs_[0] + piece.off,
              reinterpret_cast<const T*>(tmp_.get() + piece.off * es),
              piece.len);
        }
      }
    }

    // ---- phase 3: reverse cascade (final segment values flow back
    // down the chain, largest block first) ----
    const uint64_t downBase = upBase + 64;
    for (size_t i = 1; i < blocks_.size(); i++) {
      const Block& src = blocks_[i - 1];
      const Block& dst = blocks_[i];
      const uint64_t s = downBase + i;
      if (myBlock_ == static_cast<int>(i) - 1) {
        int posted = 0;
        for (int k = 0; k < dst.size; k++) {
          sched::Seg piece = intersect(span, ownedSpan(dst.size, k));
          if (piece.len > 0) {
            buf_->send(dst.off + k, s, piece.off * es, piece.len * es);
            posted++;
          }
        }
        while (posted-- > 0) {
          buf_->waitSend();
        }
      } else if (myBlock_ == static_cast<int>(i)) {
        int posted = 0;
        for (int q = 0; q < src.size; q++) {
          sched::Seg piece = intersect(span, ownedSpan(src.size, q));
          if (piece.len > 0) {
            buf_->recv(src.off + q, s, piece.off * es, piece.len * es);
            posted++;
          }
        }
        while (posted-- > 0) {
          buf_->waitRecv();
        }
      }
    }

    // ---- phase 4: intra-block allgather mirror ----
    for (int t = steps - 1; t >= 0; t--) {
      const int peer = me.off + (m ^ (1 << t));
      const int bit = (m >> t) & 1;
      sched::Seg other = sched::subspanOf(spanAt[t], 1 - bit, 2);
      buf_->recv(peer, slot_ + 1 + steps + t, other.off * es,
                 other.len * es);
      buf_->send(peer, slot_ + 1 + steps + t, span.off * es,
                 span.len * es);
      buf_->waitRecv();
      buf_->waitSend();
      span = spanAt[t];
    }
  }

  std::vector<T*> ptrs_;
  size_t count_;
  typename ReductionFunction<T>::Function* fn_;
  std::vector<Block> blocks_;
  int myBlock_{0};
  uint64_t slot_;
  AlignedPtr tmp_;
  std::unique_ptr<transport::UnboundBuffer> buf_;
  std::unique_ptr<transport::UnboundBuffer> tmpBuf_;
};

} // namespace glooamd
