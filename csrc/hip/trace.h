// rocprof-friendly named ranges around the device collectives
// (the reference has no tracer; the rebuild adds
// roctx ranges so rocprofv3 --marker-trace attributes kernels/copies to
// collective invocations).
#pragma once

#include <roctracer/roctx.h>

namespace glooamd {
namespace hip {

struct TraceRange {
  explicit TraceRange(const char* name) {
    roctxRangePushA(name);
  }
  ~TraceRange() {
    roctxRangePop();
  }
};

} // namespace hip
} // namespace glooamd
