// Race-detection stress harness: two ranks in one process, fixed
// contexts, concurrent v2 collectives (large-ring + tiny-eager paths).
//
// Build with `make SANITIZE=thread race_stress` and run under TSan.
// NOTE: waits must be untimed for a faithful TSan model: libstdc++-11's
// condition_variable::wait_for calls pthread_cond_clockwait, which this
// toolchain's libtsan does not intercept, so every timed wait corrupts
// TSan's lock-ownership state and produces hundreds of false "double
// lock"/"data race" reports. setTimeout(-1) below routes all transport
// waits through plain pthread_cond_wait (intercepted). Verified clean
// (0 warnings, 3 runs) as of the eager-send + busy-poll protocol.
#include <cstdlib>
#include <thread>
#include <vector>
#include <cstdio>
#include "collectives/collectives.h"
#include "collectives/reduce_fns.h"
#include "context.h"
#include "rendezvous/stores.h"
#include "transport/tcp/device.h"

using namespace glooamd;

int main() {
  auto store = std::make_shared<HashStore>();
  tcp::TcpAttr attr;
  if (getenv("GLOO_AMD_STRESS_UV")) {
    attr.useLibuv = true; // exercise the libuv loop under the sanitizer
  }
  auto dev = tcp::createTcpDevice(attr);
  const int P = 2;
  std::vector<std::thread> ths;
  for (int r = 0; r < P; r++) {
    ths.emplace_back([&, r] {
      auto ctx = std::make_shared<Context>(r, P);
      ctx->setTimeout(std::chrono::milliseconds(-1));
      ctx->connectFullMesh(*store, dev);
      std::vector<float> x(10000);
      for (int it = 0; it < 300; it++) {
        for (size_t i = 0; i < x.size(); i++) x[i] = float(i % 7 + r);
        AllreduceOptions o(ctx);
        o.setOutput(x.data(), x.size());
        o.reduce = cpuReduceFn(DType::F32, ReduceOp::SUM);
        o.tag = 1;
        allreduce(o);
      }
      // tiny path (eager + recursive doubling) on a second slot
      for (int it = 0; it < 300; it++) {
        std::vector<float> y(64, float(r + 1));
        AllreduceOptions o(ctx);
        o.setOutput(y.data(), y.size());
        o.reduce = cpuReduceFn(DType::F32, ReduceOp::SUM);
        o.tag = 2;
        allreduce(o);
      }
      std::printf("rank %d done\n", r);
    });
  }
  for (auto& t : ths) t.join();
  return 0;
}
