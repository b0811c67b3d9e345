// Dtype/op registry: maps (DType, ReduceOp) to the CPU elementwise
// reduction templates in math.h. Parity with the reference's
// ReductionFunction<T> built-ins (gloo/math.h:15-73 instantiated over
// the type set in gloo/cuda.cu:321-409).
#pragma once

#include <cstddef>

#include "collectives/collectives.h"

namespace glooamd {

enum class DType : int {
  F32 = 0,
  F64 = 1,
  F16 = 2,
  BF16 = 3,
  I8 = 4,
  U8 = 5,
  I32 = 6,
  I64 = 7,
  U64 = 8,
};

enum class ReduceOp : int {
  SUM = 0,
  PRODUCT = 1,
  MIN = 2,
  MAX = 3,
};

size_t dtypeSize(DType d);

// Elementwise dst = a op b over n elements.
ReduceFn cpuReduceFn(DType d, ReduceOp op);

} // namespace glooamd
