#include "collectives/reduce_fns.h"

#include "common/logging.h"
#include "math.h"
#include "types.h"

namespace glooamd {

size_t dtypeSize(DType d) {
  switch (d) {
    case DType::F32:
      return 4;
    case DType::F64:
      return 8;
    case DType::F16:
    case DType::BF16:
      return 2;
    case DType::I8:
    case DType::U8:
      return 1;
    case DType::I32:
      return 4;
    case DType::I64:
    case DType::U64:
      return 8;
  }
  GA_THROW("bad dtype");
}

namespace {

template <typename T>
ReduceFn makeFn(ReduceOp op) {
  switch (op) {
    case ReduceOp::SUM:
      return [](void* d, const void* a, const void* b, size_t n) {
        sum<T>(static_cast<T*>(d), static_cast<const T*>(a),
               static_cast<const T*>(b), n);
      };
    case ReduceOp::PRODUCT:
      return [](void* d, const void* a, const void* b, size_t n) {
        product<T>(static_cast<T*>(d), static_cast<const T*>(a),
                   static_cast<const T*>(b), n);
      };
    case ReduceOp::MIN:
      return [](void* d, const void* a, const void* b, size_t n) {
        min<T>(static_cast<T*>(d), static_cast<const T*>(a),
               static_cast<const T*>(b), n);
      };
    case ReduceOp::MAX:
      return [](void* d, const void* a, const void* b, size_t n) {
        max<T>(static_cast<T*>(d), static_cast<const T*>(a),
               static_cast<const T*>(b), n);
      };
  }
  GA_THROW("bad reduce op");
}

} // namespace

ReduceFn cpuReduceFn(DType d, ReduceOp op) {
  switch (d) {
    case DType::F32:
      return makeFn<float>(op);
    case DType::F64:
      return makeFn<double>(op);
    case DType::F16:
      return makeFn<float16>(op);
    case DType::BF16:
      return makeFn<bfloat16>(op);
    case DType::I8:
      return makeFn<int8_t>(op);
    case DType::U8:
      return makeFn<uint8_t>(op);
    case DType::I32:
      return makeFn<int32_t>(op);
    case DType::I64:
      return makeFn<int64_t>(op);
    case DType::U64:
      return makeFn<uint64_t>(op);
  }
  GA_THROW("bad dtype");
}

} // namespace glooamd
