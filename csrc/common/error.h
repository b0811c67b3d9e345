// Exception taxonomy (capability parity with reference gloo/common/error.h:23-48).
//
//   Exception          - base for all gloo_amd errors (recoverable by
//                        rebuilding contexts per docs).
//   EnforceNotMet      - failed GA_ENFORCE (programming/contract error).
//   IoException        - transport-level I/O failure (peer died, reset...).
//   TimeoutException   - an operation exceeded its deadline (subtype of
//                        IoException so existing IoException handlers fire).
#pragma once

#include <stdexcept>
#include <string>

namespace glooamd {

struct Exception : public std::runtime_error {
  explicit Exception(const std::string& msg) : std::runtime_error(msg) {}
};

struct EnforceNotMet : public Exception {
  explicit EnforceNotMet(const std::string& msg) : Exception(msg) {}
};

struct IoException : public Exception {
  explicit IoException(const std::string& msg) : Exception(msg) {}
};

struct TimeoutException : public IoException {
  explicit TimeoutException(const std::string& msg) : IoException(msg) {}
};

} // namespace glooamd
