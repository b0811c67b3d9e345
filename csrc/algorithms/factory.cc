#include "algorithms/factory.h"

#include "algorithms/allgather_ring.h"
#include "algorithms/allreduce_halving_doubling.h"
#include "algorithms/allreduce_ring.h"
#include "algorithms/allreduce_ring_chunked.h"
#include "algorithms/barrier.h"
#include "algorithms/broadcast_one_to_all.h"
#include "algorithms/misc.h"
#include "algorithms/reduce_scatter.h"
#include "common/logging.h"

namespace glooamd {

namespace {

template <typename T>
typename ReductionFunction<T>::Function* fnFor(ReduceOp op) {
  switch (op) {
    case ReduceOp::SUM:
      return &ReductionFunction<T>::sumFn;
    case ReduceOp::PRODUCT:
      return &ReductionFunction<T>::productFn;
    case ReduceOp::MIN:
      return &ReductionFunction<T>::minFn;
    case ReduceOp::MAX:
      return &ReductionFunction<T>::maxFn;
  }
  GA_THROW("bad op");
}

template <typename T>
std::unique_ptr<Algorithm> createTyped(
    const std::string& name,
    std::shared_ptr<Context> ctx,
    const std::vector<void*>& rawPtrs,
    size_t count,
    ReduceOp op,
    int root,
    const std::vector<int>& recvElems) {
  std::vector<T*> ptrs;
  for (auto* p : rawPtrs) {
    ptrs.push_back(static_cast<T*>(p));
  }
  auto* fn = fnFor<T>(op);
  if (name == "allreduce_ring") {
    return std::make_unique<AllreduceRing<T>>(ctx, ptrs, count, fn);
  }
  if (name == "allreduce_ring_chunked") {
    return std::make_unique<AllreduceRingChunked<T>>(ctx, ptrs, count, fn);
  }
  if (name == "allreduce_halving_doubling") {
    return std::make_unique<AllreduceHalvingDoubling<T>>(ctx, ptrs, count, fn);
  }
  if (name == "allreduce_bcube") {
    return std::make_unique<AllreduceBcube<T>>(ctx, ptrs, count, fn);
  }
  if (name == "allreduce_local") {
    return std::make_unique<AllreduceLocal<T>>(ctx, ptrs, count, fn);
  }
  if (name == "allgather_ring") {
    GA_ENFORCE_GE(ptrs.size(), 2ul, "allgather_ring: inputs... + output");
    std::vector<const T*> ins(ptrs.begin(), ptrs.end() - 1);
    return std::make_unique<AllgatherRing<T>>(ctx, ins, ptrs.back(), count);
  }
  if (name == "broadcast_one_to_all") {
    return std::make_unique<BroadcastOneToAll<T>>(ctx, ptrs, count, root);
  }
  if (name == "reduce_scatter_halving_doubling") {
    return std::make_unique<ReduceScatterHalvingDoubling<T>>(
        ctx, ptrs, count, recvElems, fn);
  }
  GA_THROW("unknown algorithm: ", name);
}

} // namespace

std::unique_ptr<Algorithm> createAlgorithm(
    const std::string& name,
    std::shared_ptr<Context> context,
    const std::vector<void*>& ptrs,
    size_t count,
    DType dtype,
    ReduceOp op,
    int root,
    const std::vector<int>& recvElems,
    size_t bytes,
    int steps) {
  if (name == "barrier_all_to_all") {
    return std::make_unique<BarrierAllToAll>(context);
  }
  if (name == "barrier_all_to_one") {
    return std::make_unique<BarrierAllToOne>(context, root);
  }
  if (name == "pairwise_exchange") {
    return std::make_unique<PairwiseExchange>(context, bytes, steps);
  }
  switch (dtype) {
    case DType::F32:
      return createTyped<float>(name, context, ptrs, count, op, root,
                                recvElems);
    case DType::F64:
      return createTyped<double>(name, context, ptrs, count, op, root,
                                 recvElems);
    case DType::F16:
      return createTyped<float16>(name, context, ptrs, count, op, root,
                                  recvElems);
    case DType::BF16:
      return createTyped<bfloat16>(name, context, ptrs, count, op, root,
                                   recvElems);
    case DType::I8:
      return createTyped<int8_t>(name, context, ptrs, count, op, root,
                                 recvElems);
    case DType::U8:
      return createTyped<uint8_t>(name, context, ptrs, count, op, root,
                                  recvElems);
    case DType::I32:
      return createTyped<int32_t>(name, context, ptrs, count, op, root,
                                  recvElems);
    case DType::I64:
      return createTyped<int64_t>(name, context, ptrs, count, op, root,
                                  recvElems);
    case DType::U64:
      return createTyped<uint64_t>(name, context, ptrs, count, op, root,
                                   recvElems);
  }
  GA_THROW("bad dtype");
}

} // namespace glooamd
