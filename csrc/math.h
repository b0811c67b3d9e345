// CPU elementwise reduction templates + integer helpers.
//
// Capability parity with reference gloo/math.h:15-95 (sum/product/min/max
// plus roundUp/log2ceil). The hot fp32 path is written so clang/gcc -O3
// auto-vectorizes to AVX on the host cores of an MI355X node; float16 and
// bfloat16 go through float conversion (clang emits F16C for _Float16).
#pragma once

#include <algorithm>
#include <cstddef>
#include <cstdint>

#include "types.h"

namespace glooamd {

template <typename T>
void sum(T* __restrict dst, const T* __restrict a, const T* __restrict b, size_t n) {
  for (size_t i = 0; i < n; i++) {
    dst[i] = a[i] + b[i];
  }
}

template <typename T>
void product(T* __restrict dst, const T* __restrict a, const T* __restrict b, size_t n) {
  for (size_t i = 0; i < n; i++) {
    dst[i] = a[i] * b[i];
  }
}

template <typename T>
void min(T* __restrict dst, const T* __restrict a, const T* __restrict b, size_t n) {
  for (size_t i = 0; i < n; i++) {
    dst[i] = b[i] < a[i] ? b[i] : a[i];
  }
}

template <typename T>
void max(T* __restrict dst, const T* __restrict a, const T* __restrict b, size_t n) {
  for (size_t i = 0; i < n; i++) {
    dst[i] = a[i] < b[i] ? b[i] : a[i];
  }
}

inline size_t roundUp(size_t value, size_t multiple) {
  if (multiple == 0) {
    return value;
  }
  size_t rem = value % multiple;
  return rem == 0 ? value : value + multiple - rem;
}

inline uint32_t log2ceil(uint64_t value) {
  uint32_t dim = 0;
  for (uint64_t size = 1; size < value; size <<= 1) {
    dim++;
  }
  return dim;
}

} // namespace glooamd
