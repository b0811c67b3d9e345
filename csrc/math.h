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

#if defined(__x86_64__)
#include <immintrin.h>
#endif

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

#if defined(__x86_64__)
// F16C-accelerated float16 reductions (reference gloo/math.cc:17-100
// AVX specializations): 8 halfs per iteration through hardware
// cvtph/cvtps instead of the software-converted scalar path. Guarded at
// runtime (__builtin_cpu_supports) so generic builds still work.
namespace detail {

#define GA_F16C_OP(NAME, COMBINE)                                        \
  __attribute__((target("f16c,avx"))) inline void NAME##F16C(            \
      uint16_t* d, const uint16_t* a, const uint16_t* b, size_t n) {     \
    size_t i = 0;                                                        \
    for (; i + 8 <= n; i += 8) {                                         \
      __m256 va = _mm256_cvtph_ps(                                       \
          _mm_loadu_si128(reinterpret_cast<const __m128i*>(a + i)));     \
      __m256 vb = _mm256_cvtph_ps(                                       \
          _mm_loadu_si128(reinterpret_cast<const __m128i*>(b + i)));     \
      _mm_storeu_si128(                                                  \
          reinterpret_cast<__m128i*>(d + i),                             \
          _mm256_cvtps_ph(COMBINE(va, vb), _MM_FROUND_TO_NEAREST_INT));  \
    }                                                                    \
    for (; i < n; i++) {                                                 \
      const float16* fa = reinterpret_cast<const float16*>(a + i);       \
      const float16* fb = reinterpret_cast<const float16*>(b + i);       \
      reinterpret_cast<float16*>(d + i)[0] = float16(NAME##Scalar(       \
          static_cast<float>(*fa), static_cast<float>(*fb)));            \
    }                                                                    \
  }

inline float sumScalar(float x, float y) {
  return x + y;
}
inline float productScalar(float x, float y) {
  return x * y;
}
inline float minScalar(float x, float y) {
  return y < x ? y : x;
}
inline float maxScalar(float x, float y) {
  return x < y ? y : x;
}
GA_F16C_OP(sum, _mm256_add_ps)
GA_F16C_OP(product, _mm256_mul_ps)
GA_F16C_OP(min, _mm256_min_ps)
GA_F16C_OP(max, _mm256_max_ps)
#undef GA_F16C_OP

inline bool haveF16C() {
  static const bool v =
      __builtin_cpu_supports("f16c") && __builtin_cpu_supports("avx");
  return v;
}
} // namespace detail

#define GA_F16_SPECIALIZE(NAME)                                          \
  template <>                                                            \
  inline void NAME<float16>(                                             \
      float16* __restrict dst, const float16* __restrict a,              \
      const float16* __restrict b, size_t n) {                           \
    if (detail::haveF16C()) {                                            \
      detail::NAME##F16C(                                                \
          reinterpret_cast<uint16_t*>(dst),                              \
          reinterpret_cast<const uint16_t*>(a),                          \
          reinterpret_cast<const uint16_t*>(b), n);                      \
      return;                                                            \
    }                                                                    \
    for (size_t i = 0; i < n; i++) {                                     \
      dst[i] = float16(detail::NAME##Scalar(                             \
          static_cast<float>(a[i]), static_cast<float>(b[i])));          \
    }                                                                    \
  }

GA_F16_SPECIALIZE(sum)
GA_F16_SPECIALIZE(product)
GA_F16_SPECIALIZE(min)
GA_F16_SPECIALIZE(max)
#undef GA_F16_SPECIALIZE
#endif // __x86_64__

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
