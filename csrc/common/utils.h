// Small host utilities: env flags, hostname, aligned allocation.
// Capability parity with reference gloo/common/utils.cc:40-57 (env flags)
// and gloo/common/aligned_allocator.h.
#pragma once

#include <cstdlib>
#include <memory>
#include <string>
#include <vector>

namespace glooamd {

// True iff env var is set to a truthy value (1/true/yes/on).
bool getEnvFlag(const char* name, bool defaultValue = false);

// Integer env with default.
long getEnvInt(const char* name, long defaultValue);

std::string getHostname();

// 64-byte (cacheline) aligned allocation for collective staging buffers.
constexpr size_t kBufferAlignment = 64;

void* alignedAlloc(size_t bytes);
void alignedFree(void* ptr);

struct AlignedDeleter {
  void operator()(void* p) const {
    alignedFree(p);
  }
};
using AlignedPtr = std::unique_ptr<char[], AlignedDeleter>;

inline AlignedPtr makeAligned(size_t bytes) {
  return AlignedPtr(static_cast<char*>(alignedAlloc(bytes)));
}

} // namespace glooamd
