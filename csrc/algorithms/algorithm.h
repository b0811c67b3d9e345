// Legacy class-based Algorithm API.
//
// Capability parity with reference gloo/algorithm.{h,cc}: an Algorithm is
// constructed once against a Context with fixed buffer pointers and then
// run() repeatedly; ring-style algorithms use the left/right pair helpers
// and pre-registered ("bound") buffers with one-sided writes.
#pragma once

#include <memory>

#include "context.h"
#include "math.h"
#include "transport/transport.h"

namespace glooamd {

class Algorithm {
 public:
  explicit Algorithm(const std::shared_ptr<Context>& context)
      : context_(context) {}
  virtual ~Algorithm() = default;

  virtual void run() = 0;

 protected:
  std::shared_ptr<Context> context_;

  transport::Pair* getLeftPair() {
    const int r = (context_->rank - 1 + context_->size) % context_->size;
    return context_->getPair(r);
  }
  transport::Pair* getRightPair() {
    const int r = (context_->rank + 1) % context_->size;
    return context_->getPair(r);
  }
};

// In-place reduction: dst[i] = op(dst[i], src[i]).
template <typename T>
struct ReductionFunction {
  using Function = void(T*, const T*, size_t);

  static void sumFn(T* dst, const T* src, size_t n) {
    for (size_t i = 0; i < n; i++) {
      dst[i] = dst[i] + src[i];
    }
  }
  static void productFn(T* dst, const T* src, size_t n) {
    for (size_t i = 0; i < n; i++) {
      dst[i] = dst[i] * src[i];
    }
  }
  static void minFn(T* dst, const T* src, size_t n) {
    for (size_t i = 0; i < n; i++) {
      if (src[i] < dst[i]) {
        dst[i] = src[i];
      }
    }
  }
  static void maxFn(T* dst, const T* src, size_t n) {
    for (size_t i = 0; i < n; i++) {
      if (dst[i] < src[i]) {
        dst[i] = src[i];
      }
    }
  }
};

} // namespace glooamd
