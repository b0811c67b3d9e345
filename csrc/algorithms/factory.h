// Name-based factory over the legacy Algorithm classes (used by the
// Python bindings and the benchmark tool). Mirrors the reference
// benchmark's name -> Algorithm registration table
// (gloo/benchmark/main.cc:920-1068) as a reusable component.
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "algorithms/algorithm.h"
#include "collectives/reduce_fns.h"

namespace glooamd {

// names: allreduce_ring, allreduce_ring_chunked,
// allreduce_halving_doubling, allreduce_bcube, allreduce_local,
// allgather_ring (ptrs = inputs..., last = output),
// broadcast_one_to_all, barrier_all_to_all, barrier_all_to_one,
// pairwise_exchange (bytes/steps args), reduce_scatter_halving_doubling
// (recvElems arg).
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
    int steps);

} // namespace glooamd
