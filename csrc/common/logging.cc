#include "common/logging.h"

#include <cstdlib>
#include <cstring>
#include <iostream>
#include <mutex>

namespace glooamd {

LogLevel logThreshold() {
  static LogLevel level = [] {
    const char* env = std::getenv("GLOO_AMD_LOG_LEVEL");
    if (env == nullptr) {
      return LogLevel::WARNING;
    }
    if (std::strcmp(env, "DEBUG") == 0 || std::strcmp(env, "0") == 0) {
      return LogLevel::DEBUG;
    }
    if (std::strcmp(env, "INFO") == 0 || std::strcmp(env, "1") == 0) {
      return LogLevel::INFO;
    }
    if (std::strcmp(env, "WARNING") == 0 || std::strcmp(env, "2") == 0) {
      return LogLevel::WARNING;
    }
    return LogLevel::ERROR;
  }();
  return level;
}

LogMessage::LogMessage(const char* file, int line, LogLevel level)
    : level_(level) {
  const char* base = std::strrchr(file, '/');
  stream_ << "[gloo_amd " << (base ? base + 1 : file) << ":" << line << "] ";
}

LogMessage::~LogMessage() {
  if (level_ >= logThreshold()) {
    static std::mutex mu;
    std::lock_guard<std::mutex> lock(mu);
    std::cerr << stream_.str() << std::endl;
  }
}

namespace detail {

[[noreturn]] void enforceFail(
    const char* file,
    int line,
    const char* cond,
    const std::string& msg) {
  std::ostringstream oss;
  const char* base = std::strrchr(file, '/');
  oss << "[enforce fail at " << (base ? base + 1 : file) << ":" << line
      << "] " << cond;
  if (!msg.empty()) {
    oss << ". " << msg;
  }
  throw EnforceNotMet(oss.str());
}

} // namespace detail
} // namespace glooamd
