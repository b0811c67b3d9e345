// Logging + enforce macros for gloo_amd.
//
// Provides the same capability surface as the reference's
// gloo/common/logging.h:30-208 (GLOO_ENFORCE* family, leveled stderr
// logging with env-controlled verbosity) re-designed from scratch:
// a single stream-based LogMessage sink plus GA_ENFORCE* macros that
// throw EnforceNotMet with file:line context.
#pragma once

#include <sstream>
#include <string>

#include "common/error.h"

namespace glooamd {

enum class LogLevel : int { DEBUG = 0, INFO = 1, WARNING = 2, ERROR = 3 };

// Minimum level actually emitted; read once from GLOO_AMD_LOG_LEVEL
// (accepts 0-3 or DEBUG/INFO/WARNING/ERROR). Default: WARNING.
LogLevel logThreshold();

class LogMessage {
 public:
  LogMessage(const char* file, int line, LogLevel level);
  ~LogMessage(); // emits to stderr if level >= threshold
  std::ostream& stream() {
    return stream_;
  }

 private:
  std::ostringstream stream_;
  LogLevel level_;
};

#define GA_LOG(level)                                                \
  ::glooamd::LogMessage(__FILE__, __LINE__, ::glooamd::LogLevel::level) \
      .stream()
#define GA_DEBUG GA_LOG(DEBUG)
#define GA_INFO GA_LOG(INFO)
#define GA_WARN GA_LOG(WARNING)
#define GA_ERROR GA_LOG(ERROR)

namespace detail {
template <typename... Args>
std::string makeString(const Args&... args) {
  std::ostringstream oss;
  (void)std::initializer_list<int>{((oss << args), 0)...};
  return oss.str();
}
inline std::string makeString() {
  return std::string();
}

[[noreturn]] void enforceFail(
    const char* file,
    int line,
    const char* cond,
    const std::string& msg);
} // namespace detail

// GA_ENFORCE(cond, msg...) -> throws EnforceNotMet on failure.
#define GA_ENFORCE(cond, ...)                          \
  do {                                                 \
    if (__builtin_expect(!(cond), 0)) {                \
      ::glooamd::detail::enforceFail(                  \
          __FILE__,                                    \
          __LINE__,                                    \
          #cond,                                       \
          ::glooamd::detail::makeString(__VA_ARGS__)); \
    }                                                  \
  } while (0)

#define GA_ENFORCE_BINOP(x, y, op, ...)                            \
  do {                                                             \
    const auto& ga_x_ = (x);                                       \
    const auto& ga_y_ = (y);                                       \
    if (__builtin_expect(!(ga_x_ op ga_y_), 0)) {                  \
      ::glooamd::detail::enforceFail(                              \
          __FILE__,                                                \
          __LINE__,                                                \
          #x " " #op " " #y,                                       \
          ::glooamd::detail::makeString(                           \
              "(", ga_x_, " vs ", ga_y_, ") ", ##__VA_ARGS__));    \
    }                                                              \
  } while (0)

#define GA_ENFORCE_EQ(x, y, ...) GA_ENFORCE_BINOP(x, y, ==, ##__VA_ARGS__)
#define GA_ENFORCE_NE(x, y, ...) GA_ENFORCE_BINOP(x, y, !=, ##__VA_ARGS__)
#define GA_ENFORCE_LT(x, y, ...) GA_ENFORCE_BINOP(x, y, <, ##__VA_ARGS__)
#define GA_ENFORCE_LE(x, y, ...) GA_ENFORCE_BINOP(x, y, <=, ##__VA_ARGS__)
#define GA_ENFORCE_GT(x, y, ...) GA_ENFORCE_BINOP(x, y, >, ##__VA_ARGS__)
#define GA_ENFORCE_GE(x, y, ...) GA_ENFORCE_BINOP(x, y, >=, ##__VA_ARGS__)

#define GA_THROW(...)                                \
  throw ::glooamd::Exception(                        \
      ::glooamd::detail::makeString(                 \
          __FILE__, ":", __LINE__, ": ", ##__VA_ARGS__))

#define GA_THROW_IO(...)                             \
  throw ::glooamd::IoException(                      \
      ::glooamd::detail::makeString(                 \
          __FILE__, ":", __LINE__, ": ", ##__VA_ARGS__))

} // namespace glooamd
