#include "common/utils.h"

#include <unistd.h>
#include <cstring>
#include <strings.h>

#include "common/logging.h"

namespace glooamd {

bool getEnvFlag(const char* name, bool defaultValue) {
  const char* env = std::getenv(name);
  if (env == nullptr || env[0] == '\0') {
    return defaultValue;
  }
  return !(std::strcmp(env, "0") == 0 || strcasecmp(env, "false") == 0 ||
           strcasecmp(env, "no") == 0 || strcasecmp(env, "off") == 0);
}

long getEnvInt(const char* name, long defaultValue) {
  const char* env = std::getenv(name);
  if (env == nullptr || env[0] == '\0') {
    return defaultValue;
  }
  return std::strtol(env, nullptr, 10);
}

std::string getHostname() {
  char buf[256];
  if (gethostname(buf, sizeof(buf)) != 0) {
    return "localhost";
  }
  buf[sizeof(buf) - 1] = '\0';
  return std::string(buf);
}

void* alignedAlloc(size_t bytes) {
  if (bytes == 0) {
    bytes = kBufferAlignment;
  }
  void* ptr = nullptr;
  int rv = posix_memalign(&ptr, kBufferAlignment, bytes);
  GA_ENFORCE_EQ(rv, 0, "posix_memalign failed");
  return ptr;
}

void alignedFree(void* ptr) {
  free(ptr);
}

} // namespace glooamd
