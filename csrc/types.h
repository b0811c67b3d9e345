// Slot/tag system + host-side half-precision scalar types.
//
// Capability parity with reference gloo/types.h:40-91 (Slot::build with
// 8-bit collective prefix, user tag, bounded per-op delta so concurrent
// collectives on one context never collide) and gloo/types.h:97-335
// (software float16). Re-designed layout:
//
//   64-bit slot = [ prefix:8 | user tag:32 | op delta:24 ]
//
// The 24-bit delta space (vs the reference's 8) lets deeply-segmented
// ring schedules (large HBM3E-resident buffers -> many chunks) use one
// slot per in-flight segment without wrap-around.
#pragma once

#include <cstdint>
#include <cstring>

#include "common/logging.h"

namespace glooamd {

// One prefix per collective family; a (prefix, tag) pair identifies a
// collective "channel" on a context, and per-op deltas distinguish the
// messages inside one collective invocation.
enum class SlotPrefix : uint8_t {
  kAllreduce = 0x01,
  kAllgather = 0x02,
  kAlltoall = 0x03,
  kBarrier = 0x04,
  kBroadcast = 0x05,
  kGather = 0x06,
  kReduce = 0x07,
  kScatter = 0x08,
  kSendRecv = 0x09,
  kReduceScatter = 0x0a,
  kContextFactory = 0x0b,
  kHip = 0x0c,
};

class Slot {
 public:
  static constexpr uint64_t kDeltaBits = 24;
  static constexpr uint64_t kMaxDelta = (uint64_t(1) << kDeltaBits) - 1;

  static Slot build(SlotPrefix prefix, uint32_t tag) {
    uint64_t base = (uint64_t(static_cast<uint8_t>(prefix)) << 56) |
        (uint64_t(tag) << kDeltaBits);
    return Slot(base, 0);
  }

  operator uint64_t() const {
    return base_ + delta_;
  }

  Slot operator+(uint64_t delta) const {
    GA_ENFORCE_LE(delta, kMaxDelta, "slot delta out of range");
    return Slot(base_, delta);
  }

 private:
  Slot(uint64_t base, uint64_t delta) : base_(base), delta_(delta) {}
  uint64_t base_;
  uint64_t delta_;
};

// ---------------------------------------------------------------------------
// Host scalar types for reduced precision. The wire and kernels treat these
// as opaque 16-bit payloads; host math converts through float.
// hipcc/clang provide _Float16 natively on x86-64.
// ---------------------------------------------------------------------------

namespace detail {
// IEEE binary16 <-> binary32 conversion in plain integer math (round to
// nearest even), so the type builds identically under g++ and hipcc.
inline uint16_t f32_to_f16_bits(float f) {
  uint32_t u;
  std::memcpy(&u, &f, 4);
  uint32_t sign = (u >> 16) & 0x8000u;
  int32_t exp = static_cast<int32_t>((u >> 23) & 0xff) - 127 + 15;
  uint32_t mant = u & 0x7fffffu;
  if (((u >> 23) & 0xff) == 0xff) { // inf / nan
    return sign | 0x7c00u | (mant ? 0x200u | (mant >> 13) : 0);
  }
  if (exp >= 31) { // overflow -> inf
    return sign | 0x7c00u;
  }
  if (exp <= 0) { // subnormal or zero
    if (exp < -10) {
      return sign;
    }
    mant |= 0x800000u;
    uint32_t shift = 14 - exp;
    uint32_t half = mant >> shift;
    uint32_t rem = mant & ((1u << shift) - 1);
    uint32_t mid = 1u << (shift - 1);
    if (rem > mid || (rem == mid && (half & 1))) {
      half++;
    }
    return sign | half;
  }
  uint32_t half = (static_cast<uint32_t>(exp) << 10) | (mant >> 13);
  uint32_t rem = mant & 0x1fffu;
  if (rem > 0x1000u || (rem == 0x1000u && (half & 1))) {
    half++; // may carry into exponent: correct behaviour
  }
  return sign | half;
}

inline float f16_bits_to_f32(uint16_t h) {
  uint32_t sign = (h & 0x8000u) << 16;
  uint32_t exp = (h >> 10) & 0x1f;
  uint32_t mant = h & 0x3ffu;
  uint32_t u;
  if (exp == 0) {
    if (mant == 0) {
      u = sign;
    } else { // subnormal: normalize
      int e = -1;
      do {
        e++;
        mant <<= 1;
      } while ((mant & 0x400u) == 0);
      u = sign | ((127 - 15 - e) << 23) | ((mant & 0x3ffu) << 13);
    }
  } else if (exp == 0x1f) {
    u = sign | 0x7f800000u | (mant << 13);
  } else {
    u = sign | ((exp - 15 + 127) << 23) | (mant << 13);
  }
  float f;
  std::memcpy(&f, &u, 4);
  return f;
}
} // namespace detail

struct float16 {
  uint16_t bits;

  float16() : bits(0) {}
  explicit float16(float f) : bits(detail::f32_to_f16_bits(f)) {}
  explicit operator float() const {
    return detail::f16_bits_to_f32(bits);
  }
  float16& operator+=(const float16& o) {
    *this = float16(float(*this) + float(o));
    return *this;
  }
  bool operator==(const float16& o) const {
    return bits == o.bits;
  }
};

struct bfloat16 {
  uint16_t bits;

  bfloat16() : bits(0) {}
  explicit bfloat16(float f) {
    uint32_t u;
    std::memcpy(&u, &f, sizeof(u));
    // round-to-nearest-even on the dropped 16 bits
    uint32_t rounding = 0x7fff + ((u >> 16) & 1);
    bits = static_cast<uint16_t>((u + rounding) >> 16);
  }
  explicit operator float() const {
    uint32_t u = uint32_t(bits) << 16;
    float f;
    std::memcpy(&f, &u, sizeof(f));
    return f;
  }
  bfloat16& operator+=(const bfloat16& o) {
    *this = bfloat16(float(*this) + float(o));
    return *this;
  }
  bool operator==(const bfloat16& o) const {
    return bits == o.bits;
  }
};

inline float16 operator+(float16 a, float16 b) {
  return float16(float(a) + float(b));
}
inline float16 operator*(float16 a, float16 b) {
  return float16(float(a) * float(b));
}
inline bool operator<(float16 a, float16 b) {
  return float(a) < float(b);
}
inline bfloat16 operator+(bfloat16 a, bfloat16 b) {
  return bfloat16(float(a) + float(b));
}
inline bfloat16 operator*(bfloat16 a, bfloat16 b) {
  return bfloat16(float(a) * float(b));
}
inline bool operator<(bfloat16 a, bfloat16 b) {
  return float(a) < float(b);
}

} // namespace glooamd
