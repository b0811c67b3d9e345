// v2 collective functions: options structs + entry points.
//
// Capability parity with the reference's new-style collectives
// (gloo/allreduce.h:193, gloo/allgather.cc, gloo/alltoall.cc,
// gloo/barrier.cc, gloo/broadcast.cc, gloo/gather.cc, gloo/reduce.cc,
// gloo/scatter.cc and the *v variants). All operate on type-erased byte
// buffers; reductions are std::function over (dst, a, b, n) element
// pointers like gloo/allreduce.h:36. Algorithms are re-derived:
//   allreduce: segmented ring reduce-scatter + ring allgather
//              (<=maxSegmentSize segments, >=2 per rank, two in flight)
//              or bcube (recursive grouped exchange, base from context).
//   allgather[v]: ring, two blocks in flight.
//   alltoall[v]: pairwise exchange, all recvs posted up front.
//   barrier: dissemination.
//   broadcast: binomial tree on virtual ranks.
//   gather[v]/scatter: direct to/from root.
//   reduce: ring reduce-scatter + block gather to root.
#pragma once

#include <chrono>
#include <cstring>
#include <functional>
#include <memory>
#include <vector>

#include "context.h"
#include "types.h"

namespace glooamd {

// dst[i] = a[i] op b[i] for n elements (dst may alias a).
using ReduceFn =
    std::function<void(void*, const void*, const void*, size_t)>;

constexpr size_t kDefaultMaxSegmentSize = 1 << 20; // 1 MiB wire segments

namespace detail {
struct CollectiveOptionsBase {
  explicit CollectiveOptionsBase(std::shared_ptr<Context> ctx)
      : context(std::move(ctx)) {}
  std::shared_ptr<Context> context;
  uint32_t tag = 0;
  std::chrono::milliseconds timeout{0}; // 0 -> context default
};
} // namespace detail

struct AllreduceOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  enum class Algorithm { UNSPECIFIED, RING, BCUBE };

  // Multiple I/O pointers supported (local reduction across them first).
  std::vector<void*> inputs; // optional; empty -> in-place on outputs
  std::vector<void*> outputs;
  size_t elements = 0;
  size_t elementSize = 0;
  ReduceFn reduce;
  Algorithm algorithm = Algorithm::RING;
  size_t maxSegmentSize = kDefaultMaxSegmentSize;

  template <typename T>
  void setOutput(T* ptr, size_t n) {
    outputs = {ptr};
    elements = n;
    elementSize = sizeof(T);
  }
  template <typename T>
  void setInput(T* ptr, size_t n) {
    inputs = {ptr};
    elements = n;
    elementSize = sizeof(T);
  }
};
void allreduce(AllreduceOptions& opts);

struct AllgatherOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr; // optional (in-place at out + rank*inElements)
  void* output = nullptr; // size*inElements elements
  size_t inElements = 0;
  size_t elementSize = 0;
};
void allgather(AllgatherOptions& opts);

struct AllgathervOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr;
  void* output = nullptr;
  std::vector<size_t> counts; // per-rank element counts
  size_t elementSize = 0;
};
void allgatherv(AllgathervOptions& opts);

struct AlltoallOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr; // size*perRankElements
  void* output = nullptr;
  size_t perRankElements = 0;
  size_t elementSize = 0;
};
void alltoall(AlltoallOptions& opts);

struct AlltoallvOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr;
  void* output = nullptr;
  std::vector<size_t> inCounts; // per-destination element counts
  std::vector<size_t> outCounts; // per-source element counts
  size_t elementSize = 0;
};
void alltoallv(AlltoallvOptions& opts);

struct BarrierOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
};
void barrier(BarrierOptions& opts);

struct BroadcastOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr; // root only (optional; out used if null)
  void* output = nullptr;
  size_t elements = 0;
  size_t elementSize = 0;
  int root = 0;
};
void broadcast(BroadcastOptions& opts);

struct GatherOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr;
  void* output = nullptr; // root only: size*inElements
  size_t inElements = 0;
  size_t elementSize = 0;
  int root = 0;
};
void gather(GatherOptions& opts);

struct GathervOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr;
  void* output = nullptr; // root only
  std::vector<size_t> counts; // per-rank element counts (all ranks know)
  size_t elementSize = 0;
  int root = 0;
};
void gatherv(GathervOptions& opts);

struct ReduceOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr; // optional
  void* output = nullptr;
  size_t elements = 0;
  size_t elementSize = 0;
  ReduceFn reduce;
  int root = 0;
  size_t maxSegmentSize = kDefaultMaxSegmentSize;
};
void reduce(ReduceOptions& opts);

struct ScatterOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr; // root only: size*outElements
  void* output = nullptr;
  size_t outElements = 0;
  size_t elementSize = 0;
  int root = 0;
};
void scatter(ScatterOptions& opts);

struct ReduceScatterOptions : detail::CollectiveOptionsBase {
  using CollectiveOptionsBase::CollectiveOptionsBase;
  void* input = nullptr; // size*recvElements elements (optional, in-place)
  void* output = nullptr; // recvElements elements
  size_t recvElements = 0;
  size_t elementSize = 0;
  ReduceFn reduce;
};
void reduce_scatter(ReduceScatterOptions& opts);

} // namespace glooamd
