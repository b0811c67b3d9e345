// Shared block/segment math for ring and bcube schedules (used by the
// CPU v2 collectives and the hip_* device algorithms so both sides of a
// wire agree on every chunk boundary). Counterpart of the reference's
// per-collective segment arithmetic (gloo/allreduce.cc:209-218 segment
// sizing, gloo/allgather.cc chunk walk), centralized instead of
// re-derived in every algorithm.
#pragma once

#include <algorithm>
#include <cstddef>

namespace glooamd {
namespace sched {

struct Seg {
  size_t off; // element offset
  size_t len; // element count
};

// Block b of an N-element buffer split across P ranks (tail-clamped).
inline Seg blockOf(size_t N, int P, int b) {
  size_t perRank = (N + P - 1) / P;
  size_t start = std::min<size_t>(static_cast<size_t>(b) * perRank, N);
  size_t end = std::min<size_t>(start + perRank, N);
  return {start, end - start};
}

// Segment s (of S) within block b.
inline Seg segmentOf(size_t N, int P, int b, int s, int S) {
  Seg blk = blockOf(N, P, b);
  size_t perSeg = (blk.len + S - 1) / S;
  size_t start =
      std::min(blk.off + static_cast<size_t>(s) * perSeg, blk.off + blk.len);
  size_t end = std::min(start + perSeg, blk.off + blk.len);
  return {start, end - start};
}

// Sub-span j of `base` equal parts of span (remainder spread left).
inline Seg subspanOf(Seg span, int j, int base) {
  size_t q = span.len / base;
  size_t r = span.len % base;
  size_t start =
      span.off + static_cast<size_t>(j) * q + std::min<size_t>(j, r);
  size_t len = q + (static_cast<size_t>(j) < r ? 1 : 0);
  return {start, len};
}

// Chunk c (of nc) within a span.
inline Seg chunkOf(Seg span, int c, int nc) {
  size_t per = (span.len + nc - 1) / nc;
  size_t start = std::min(span.off + static_cast<size_t>(c) * per,
                          span.off + span.len);
  size_t end = std::min(start + per, span.off + span.len);
  return {start, end - start};
}

inline size_t alignUp(size_t v, size_t a) {
  return a <= 1 ? v : ((v + a - 1) / a) * a;
}
inline size_t alignDown(size_t v, size_t a) {
  return a <= 1 ? v : (v / a) * a;
}

// Aligned variants for the DEVICE engines: every split boundary is a
// multiple of alignElems (elements worth of 16 bytes), so reduction
// kernels stay on the vectorized 16-B/lane path and D2D copies stay
// dwordx4-aligned; only the clamped tail piece may be shorter. All
// ranks compute these from collective arguments, so wire peers agree.
inline Seg blockOfA(size_t N, int P, int b, size_t A) {
  size_t perRank = alignUp((N + P - 1) / P, A);
  size_t start = std::min<size_t>(static_cast<size_t>(b) * perRank, N);
  size_t end = std::min<size_t>(start + perRank, N);
  return {start, end - start};
}

inline Seg segmentOfA(size_t N, int P, int b, int s, int S, size_t A) {
  Seg blk = blockOfA(N, P, b, A);
  size_t perSeg = alignUp((blk.len + S - 1) / S, A);
  size_t start =
      std::min(blk.off + static_cast<size_t>(s) * perSeg, blk.off + blk.len);
  size_t end = std::min(start + perSeg, blk.off + blk.len);
  return {start, end - start};
}

inline Seg subspanOfA(Seg span, int j, int parts, size_t A) {
  size_t per = alignUp((span.len + parts - 1) / parts, A);
  size_t start = std::min(span.off + static_cast<size_t>(j) * per,
                          span.off + span.len);
  size_t end = std::min(start + per, span.off + span.len);
  return {start, end - start};
}

inline Seg chunkOfA(Seg span, int c, int nc, size_t A) {
  size_t per = alignUp((span.len + nc - 1) / nc, A);
  size_t start = std::min(span.off + static_cast<size_t>(c) * per,
                          span.off + span.len);
  size_t end = std::min(start + per, span.off + span.len);
  return {start, end - start};
}

} // namespace sched
} // namespace glooamd
