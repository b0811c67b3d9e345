// gloo_amd benchmark tool.
//
// Capability parity with the reference benchmark harness
// (gloo/benchmark/main.cc:920-1068, runner.cc:259-369, options.cc,
// cuda_main.cc:167-224): named benchmarks over the legacy algorithms,
// the v2 collectives, send/recv, and the hip_* device algorithms;
// store-based rendezvous (file or tcp); warmup + time-calibrated
// iteration counts; min/p50/p99/max latency distribution per element
// count; element sweep with --elements -1.
//
// Usage (per rank):
//   gloo_amd_bench --size N --rank R [--store-file DIR | --store-tcp
//   HOST:PORT] --benchmark allreduce_ring_chunked --elements -1
//   [--gpu] [--half-precision] [--iteration-time-ms 2000]
#include <getopt.h>

#include <thread>

#include <cstdio>
#include <cstring>
#include <functional>
#include <memory>
#include <string>
#include <vector>

#include "algorithms/factory.h"
#include "bench/timer.h"
#include "collectives/collectives.h"
#include "collectives/reduce_fns.h"
#include "common/utils.h"
#include "context.h"
#include "hip/algorithms.h"
#include "hip/kernels.h"
#include "rendezvous/context_factory.h"
#include "rendezvous/stores.h"
#include "transport/tcp/device.h"
#include "transport/tcp/tls.h"

using namespace glooamd;

namespace {

struct Options {
  int size = 1;
  int rank = 0;
  std::string storeFile;
  std::string storeTcp; // host:port
  std::string benchmark = "allreduce_ring_chunked";
  long elements = -1; // -1: sweep
  long iterationTimeMs = 2000;
  int warmupIters = 5;
  int inputs = 1;
  int threads = 1;
  bool gpu = false;
  bool halfPrecision = false;
  int base = 2;
  bool verify = true;
  std::string transport = "tcp"; // tcp | tls
  std::string tlsCert, tlsKey, tlsCa;
};

[[noreturn]] void usage() {
  fprintf(stderr,
          "gloo_amd_bench --size N --rank R [--store-file DIR | "
      "[--transport tcp|tls --tls-cert F --tls-key F [--tls-ca F]] "
          "--store-tcp HOST:PORT]\n"
          "  --benchmark NAME --elements N|-1 [--gpu] [--half-precision]\n"
          "  [--iteration-time-ms MS] [--warmup-iters N] [--inputs N] "
          "[--base B] [--no-verify]\n"
          "benchmarks: allreduce_ring, allreduce_ring_chunked,\n"
          "  allreduce_halving_doubling, allreduce_bcube, allgather_ring,\n"
          "  barrier_all_to_all, barrier_all_to_one, broadcast_one_to_all,\n"
          "  pairwise_exchange, reduce_scatter_halving_doubling,\n"
          "  new_allreduce_ring, new_allreduce_bcube, sendrecv_roundtrip,\n"
          "  hip_allreduce_ring, hip_allreduce_ring_chunked,\n"
          "  hip_allreduce_halving_doubling, hip_broadcast_one_to_all,\n"
          "  hip_allgather_ring, hip_reduce_scatter, hip_alltoall,\n"
          "  hip_allreduce_direct, hip_allreduce_bcube\n");
  exit(1);
}

Options parse(int argc, char** argv) {
  Options o;
  static struct option longOpts[] = {
      {"size", required_argument, nullptr, 's'},
      {"rank", required_argument, nullptr, 'r'},
      {"store-file", required_argument, nullptr, 'f'},
      {"store-tcp", required_argument, nullptr, 't'},
      {"benchmark", required_argument, nullptr, 'b'},
      {"elements", required_argument, nullptr, 'e'},
      {"iteration-time-ms", required_argument, nullptr, 'i'},
      {"warmup-iters", required_argument, nullptr, 'w'},
      {"inputs", required_argument, nullptr, 'n'},
      {"threads", required_argument, nullptr, 'T'},
      {"gpu", no_argument, nullptr, 'g'},
      {"half-precision", no_argument, nullptr, 'h'},
      {"base", required_argument, nullptr, 'B'},
      {"no-verify", no_argument, nullptr, 'V'},
      {"transport", required_argument, nullptr, 'X'},
      {"tls-cert", required_argument, nullptr, 'C'},
      {"tls-key", required_argument, nullptr, 'K'},
      {"tls-ca", required_argument, nullptr, 'A'},
      {nullptr, 0, nullptr, 0},
  };
  int c;
  while ((c = getopt_long(argc, argv, "", longOpts, nullptr)) != -1) {
    switch (c) {
      case 's':
        o.size = atoi(optarg);
        break;
      case 'r':
        o.rank = atoi(optarg);
        break;
      case 'f':
        o.storeFile = optarg;
        break;
      case 't':
        o.storeTcp = optarg;
        break;
      case 'b':
        o.benchmark = optarg;
        break;
      case 'e':
        o.elements = atol(optarg);
        break;
      case 'i':
        o.iterationTimeMs = atol(optarg);
        break;
      case 'w':
        o.warmupIters = atoi(optarg);
        break;
      case 'n':
        o.inputs = atoi(optarg);
        break;
      case 'T':
        o.threads = atoi(optarg);
        break;
      case 'g':
        o.gpu = true;
        break;
      case 'h':
        o.halfPrecision = true;
        break;
      case 'B':
        o.base = atoi(optarg);
        break;
      case 'X':
        o.transport = optarg;
        break;
      case 'C':
        o.tlsCert = optarg;
        break;
      case 'K':
        o.tlsKey = optarg;
        break;
      case 'A':
        o.tlsCa = optarg;
        break;
      case 'V':
        o.verify = false;
        break;
      default:
        usage();
    }
  }
  return o;
}

template <typename T>
static size_t v0size(const std::vector<std::vector<T>>& d) {
  return d.empty() ? 0 : d[0].size();
}

// One benchmark instance for a fixed element count: setup() allocates and
// returns a run closure; verify() checks the closed-form fixture.
struct Bench {
  std::function<void()> run;
  std::function<bool()> verify; // may be null
  std::function<void()> teardown; // may be null
};

template <typename T>
void fillFixture(T* p, size_t n, int rank, int input, int numInputs) {
  // Reference pattern (gloo/benchmark/benchmark.h:26-86):
  // value = rank*inputs + input; mem[j] = j * stride + value, with
  // stride = size * numInputs. Modulo keeps fp16 exact.
  for (size_t j = 0; j < n; j++) {
    p[j] = T(float((j % 29) + rank * numInputs + input));
  }
}

template <typename T>
bool checkAllreduce(const T* p, size_t n, int size, int numInputs) {
  for (size_t j = 0; j < n; j++) {
    float expected = 0;
    for (int r = 0; r < size; r++) {
      for (int i = 0; i < numInputs; i++) {
        expected += float((j % 29) + r * numInputs + i);
      }
    }
    if (float(p[j]) != expected) {
      fprintf(stderr, "verify fail at %zu: %f != %f\n", j, double(float(p[j])),
              double(expected));
      return false;
    }
  }
  return true;
}

template <typename T>
Bench makeCpuBench(
    const Options& o,
    std::shared_ptr<Context> ctx,
    size_t elements,
    DType dtype) {
  Bench b;
  const std::string& name = o.benchmark;

  if (name == "barrier_all_to_all" || name == "barrier_all_to_one" ||
      name == "pairwise_exchange") {
    auto algo = std::shared_ptr<Algorithm>(createAlgorithm(
        name, ctx, {}, 0, dtype, ReduceOp::SUM, 0, {},
        elements * sizeof(T), std::max(1, int(log2ceil(ctx->size)))));
    b.run = [algo] { algo->run(); };
    return b;
  }

  if (name == "sendrecv_roundtrip") {
    auto data = std::make_shared<std::vector<T>>(elements);
    auto buf = std::shared_ptr<transport::UnboundBuffer>(
        ctx->createUnboundBuffer(data->data(), elements * sizeof(T)));
    int next = (ctx->rank + 1) % ctx->size;
    int prev = (ctx->rank - 1 + ctx->size) % ctx->size;
    auto ctxp = ctx;
    b.run = [ctxp, buf, next, prev, data] {
      uint64_t slot = Slot::build(SlotPrefix::kSendRecv, 0x77);
      if (ctxp->rank == 0) {
        buf->send(next, slot);
        buf->waitSend();
        buf->recv(prev, slot);
        buf->waitRecv();
      } else {
        buf->recv(prev, slot);
        buf->waitRecv();
        buf->send(next, slot);
        buf->waitSend();
      }
    };
    return b;
  }

  if (name.rfind("new_", 0) == 0) {
    // v2 function-based collectives.
    auto data = std::make_shared<std::vector<std::vector<T>>>();
    for (int i = 0; i < o.inputs; i++) {
      data->emplace_back(elements);
    }
    auto reset = [data, &o, ctx] {
      for (int i = 0; i < int(data->size()); i++) {
        fillFixture((*data)[i].data(), (*data)[i].size(), ctx->rank, i,
                    int(data->size()));
      }
    };
    reset();
    auto algoKind = name == "new_allreduce_bcube"
        ? AllreduceOptions::Algorithm::BCUBE
        : AllreduceOptions::Algorithm::RING;
    auto ctxp = ctx;
    DType dt = dtype;
    auto runOnce = [ctxp, data, algoKind, dt] {
      AllreduceOptions opts(ctxp);
      for (auto& v : *data) {
        opts.outputs.push_back(v.data());
      }
      opts.elements = v0size(*data);
      opts.elementSize = dtypeSize(dt);
      opts.reduce = cpuReduceFn(dt, ReduceOp::SUM);
      opts.algorithm = algoKind;
      allreduce(opts);
    };
    b.run = runOnce;
    int sz = ctx->size, ni = o.inputs;
    b.verify = [data, reset, runOnce, sz, ni] {
      reset();
      runOnce();
      bool ok = checkAllreduce((*data)[0].data(), (*data)[0].size(), sz, ni);
      reset();
      return ok;
    };
    return b;
  }

  // Legacy Algorithm classes.
  auto data = std::make_shared<std::vector<std::vector<T>>>();
  int numInputs = o.inputs;
  for (int i = 0; i < numInputs; i++) {
    data->emplace_back(elements);
  }
  std::vector<void*> ptrs;
  for (auto& v : *data) {
    ptrs.push_back(v.data());
  }
  std::shared_ptr<std::vector<T>> out;
  if (name == "allgather_ring") {
    out = std::make_shared<std::vector<T>>(
        elements * numInputs * ctx->size);
    ptrs.push_back(out->data());
  }
  std::vector<int> recvElems;
  if (name == "reduce_scatter_halving_doubling") {
    int base = elements / ctx->size;
    for (int r = 0; r < ctx->size; r++) {
      recvElems.push_back(
          base + (r < int(elements % ctx->size) ? 1 : 0));
    }
  }
  auto reset = [data, ctx, numInputs] {
    for (int i = 0; i < numInputs; i++) {
      fillFixture((*data)[i].data(), (*data)[i].size(), ctx->rank, i,
                  numInputs);
    }
  };
  reset();
  auto algo = std::shared_ptr<Algorithm>(createAlgorithm(
      name, ctx, ptrs, elements, dtype, ReduceOp::SUM, 0, recvElems, 0, 1));
  bool isAllreduce = name.rfind("allreduce", 0) == 0 &&
      name != "allreduce_local";
  // Like the reference, the timed loop re-reduces the running values
  // (no per-iteration reset; verification uses fresh fixtures).
  // data/out are captured to keep the buffers alive for algo's lifetime.
  b.run = [algo, data, out] { algo->run(); };
  int sz = ctx->size;
  if (isAllreduce && o.verify) {
    b.verify = [algo, data, reset, sz, numInputs] {
      reset();
      algo->run();
      bool ok =
          checkAllreduce((*data)[0].data(), (*data)[0].size(), sz, numInputs);
      reset();
      return ok;
    };
  }
  return b;
}

Bench makeHipBench(
    const Options& o,
    std::shared_ptr<Context> ctx,
    size_t elements,
    DType dtype) {
  Bench b;
  const std::string& name = o.benchmark;
  const int device = 0; // one process per GPU: HIP_VISIBLE_DEVICES selects
  const size_t es = dtypeSize(dtype);
  void* devPtr = nullptr;
  GA_HIP_CHECK(hipSetDevice(device));
  GA_HIP_CHECK(hipMalloc(&devPtr, std::max<size_t>(elements * es, 16)));
  hip::launchFillPattern(devPtr, elements, dtype, ctx->rank, 1.0, nullptr);
  GA_HIP_CHECK(hipDeviceSynchronize());

  if (name == "hip_allreduce_ring" || name == "hip_allreduce_ring_chunked") {
    auto algo = std::make_shared<hip::HipAllreduceRing>(
        ctx, device, name == "hip_allreduce_ring_chunked");
    DType dt = dtype;
    b.run = [algo, devPtr, elements, dt] {
      algo->run(devPtr, elements, dt, ReduceOp::SUM);
    };
  } else if (name == "hip_allreduce_direct") {
    auto algo = std::make_shared<hip::HipAllreduceDirect>(ctx, device);
    DType dt = dtype;
    b.run = [algo, devPtr, elements, dt] {
      algo->run(devPtr, elements, dt, ReduceOp::SUM);
    };
  } else if (name == "hip_allreduce_bcube") {
    auto algo = std::make_shared<hip::HipAllreduceBcube>(ctx, device, o.base);
    DType dt = dtype;
    b.run = [algo, devPtr, elements, dt] {
      algo->run(devPtr, elements, dt, ReduceOp::SUM);
    };
  } else if (name == "hip_allreduce_halving_doubling") {
    auto algo =
        std::make_shared<hip::HipAllreduceHalvingDoubling>(ctx, device);
    DType dt = dtype;
    b.run = [algo, devPtr, elements, dt] {
      algo->run(devPtr, elements, dt, ReduceOp::SUM);
    };
  } else if (name == "hip_broadcast_one_to_all") {
    auto algo = std::make_shared<hip::HipBroadcastOneToAll>(ctx, device, 0);
    b.run = [algo, devPtr, elements, es] {
      algo->run(devPtr, elements * es);
    };
  } else if (name == "hip_allgather_ring") {
    void* outPtr = nullptr;
    GA_HIP_CHECK(hipMalloc(
        &outPtr, std::max<size_t>(elements * es * ctx->size, 16)));
    auto algo = std::make_shared<hip::HipAllgatherRing>(ctx, device);
    b.run = [algo, devPtr, outPtr, elements, es] {
      algo->run(devPtr, outPtr, elements, es);
    };
    b.teardown = [outPtr, devPtr] {
      (void)hipFree(outPtr);
      (void)hipFree(devPtr);
    };
    return b;
  } else if (name == "hip_reduce_scatter") {
    // devPtr holds size*elements; out holds elements
    void* outPtr = nullptr;
    GA_HIP_CHECK(hipFree(devPtr));
    GA_HIP_CHECK(hipMalloc(
        &devPtr,
        std::max<size_t>(elements * es * ctx->size, 16)));
    hip::launchFillPattern(devPtr, elements * ctx->size, dtype, ctx->rank,
                           1.0, nullptr);
    GA_HIP_CHECK(hipMalloc(&outPtr, std::max<size_t>(elements * es, 16)));
    GA_HIP_CHECK(hipDeviceSynchronize());
    auto algo = std::make_shared<hip::HipReduceScatterRing>(ctx, device);
    DType dt = dtype;
    b.run = [algo, devPtr, outPtr, elements, dt] {
      algo->run(devPtr, outPtr, elements, dt, ReduceOp::SUM);
    };
    b.teardown = [outPtr, devPtr] {
      (void)hipFree(outPtr);
      (void)hipFree(devPtr);
    };
    return b;
  } else if (name == "hip_alltoall") {
    void* outPtr = nullptr;
    GA_HIP_CHECK(hipFree(devPtr));
    GA_HIP_CHECK(hipMalloc(
        &devPtr,
        std::max<size_t>(elements * es * ctx->size, 16)));
    hip::launchFillPattern(devPtr, elements * ctx->size, dtype, ctx->rank,
                           1.0, nullptr);
    GA_HIP_CHECK(
        hipMalloc(&outPtr, std::max<size_t>(elements * es * ctx->size, 16)));
    GA_HIP_CHECK(hipDeviceSynchronize());
    auto algo = std::make_shared<hip::HipAlltoall>(ctx, device);
    b.run = [algo, devPtr, outPtr, elements, es] {
      algo->run(devPtr, outPtr, elements, es);
    };
    b.teardown = [outPtr, devPtr] {
      (void)hipFree(outPtr);
      (void)hipFree(devPtr);
    };
    return b;
  } else {
    GA_THROW("unknown hip benchmark: ", name);
  }
  b.teardown = [devPtr] { (void)hipFree(devPtr); };
  return b;
}

void printHeader(const Options& o) {
  printf("%-14s %-12s %8s\n", "benchmark:", o.benchmark.c_str(), "");
  printf("%-14s %-12d\n", "size:", o.size);
  printf("%11s %11s %11s %11s %11s %11s\n", "elements", "min (us)",
         "p50 (us)", "p99 (us)", "max (us)", "samples");
}

void runOne(
    const Options& o,
    std::shared_ptr<Context> ctx,
    size_t elements) {
  DType dtype = o.halfPrecision ? DType::F16 : DType::F32;
  if (o.gpu && o.halfPrecision) {
    dtype = DType::BF16; // MI355X-native half type for device benches
  }
  Bench b;
  if (o.gpu) {
    b = makeHipBench(o, ctx, elements, dtype);
  } else if (dtype == DType::F16) {
    b = makeCpuBench<float16>(o, ctx, elements, dtype);
  } else {
    b = makeCpuBench<float>(o, ctx, elements, dtype);
  }

  if (b.verify && o.verify) {
    GA_ENFORCE(b.verify(), "verification failed for ", o.benchmark);
  }

  // Warmup.
  for (int i = 0; i < o.warmupIters; i++) {
    b.run();
  }

  // Calibrate iteration count from the median of a short probe, so the
  // timed region lasts ~iterationTimeMs (reference runner.cc:308-366).
  bench::Timer t;
  bench::Samples probe;
  for (int i = 0; i < 5; i++) {
    t.start();
    b.run();
    probe.add(t.lapNs());
  }
  bench::Distribution probeDist(probe);
  double med = std::max<double>(double(probeDist.percentile(0.5)), 100.0);
  long iters = std::max<long>(1, long(o.iterationTimeMs * 1e6 / med));
  iters = std::min<long>(iters, 100000);
  // All ranks must agree: broadcast rank 0's count.
  {
    BroadcastOptions bo(ctx);
    bo.output = &iters;
    bo.input = &iters;
    bo.elements = 1;
    bo.elementSize = sizeof(iters);
    bo.root = 0;
    bo.tag = ctx->nextSlot();
    broadcast(bo);
  }

  // Inter-rank sync, then the timed loop.
  {
    BarrierOptions barOpts(ctx);
    barOpts.tag = ctx->nextSlot();
    barrier(barOpts);
  }
  bench::Samples samples;
  for (long i = 0; i < iters; i++) {
    t.start();
    b.run();
    samples.add(t.lapNs());
  }
  {
    BarrierOptions barOpts(ctx);
    barOpts.tag = ctx->nextSlot();
    barrier(barOpts);
  }

  if (o.rank == 0) {
    bench::Distribution d(samples);
    printf("%11zu %11.0f %11.0f %11.0f %11.0f %11zu\n", elements,
           d.min() / 1e3, d.percentile(0.5) / 1e3, d.percentile(0.99) / 1e3,
           d.max() / 1e3, d.size());
    fflush(stdout);
  }
  if (b.teardown) {
    b.teardown();
  }
}

} // namespace

int main(int argc, char** argv) {
  // One HW queue per pooled stream (see csrc/hip/core.h pooledStream).
  setenv("GPU_MAX_HW_QUEUES", "8", 0);
  Options o = parse(argc, argv);
  if (o.benchmark.rfind("hip_", 0) == 0) {
    o.gpu = true; // hip_* benchmarks imply the device path
  }
  GA_ENFORCE_GE(o.rank, 0);
  GA_ENFORCE_LT(o.rank, o.size);

  std::shared_ptr<IStore> store;
  if (!o.storeFile.empty()) {
    store = std::make_shared<FileStore>(o.storeFile);
  } else if (!o.storeTcp.empty()) {
    auto colon = o.storeTcp.find(':');
    GA_ENFORCE_NE(colon, std::string::npos, "--store-tcp needs host:port");
    store = std::make_shared<TcpStore>(
        o.storeTcp.substr(0, colon), atoi(o.storeTcp.c_str() + colon + 1),
        o.rank == 0);
  } else {
    GA_ENFORCE(o.size == 1, "need --store-file or --store-tcp for size > 1");
    store = std::make_shared<HashStore>();
  }

  std::shared_ptr<transport::Device> device;
  if (o.transport == "tls") {
    tcp::tls::TlsAttr tattr;
    tattr.certFile = o.tlsCert;
    tattr.pkeyFile = o.tlsKey;
    tattr.caFile = o.tlsCa;
    device = tcp::tls::createTlsDevice(tattr);
  } else if (o.transport == "uds") {
    tcp::TcpAttr attr;
    attr.useUds = true;
    device = tcp::createTcpDevice(attr);
  } else {
    GA_ENFORCE(o.transport == "tcp", "unknown --transport ", o.transport);
    tcp::TcpAttr attr;
    device = tcp::createTcpDevice(attr);
  }
  auto ctx = std::make_shared<Context>(o.rank, o.size, o.base);
  ctx->connectFullMesh(*store, device);

  // Per-thread contexts via store-less re-rendezvous (reference
  // Runner/ContextFactory parity, runner.cc:151-156). Thread 0 reports.
  std::vector<std::shared_ptr<Context>> ctxs{ctx};
  if (o.threads > 1) {
    ContextFactory factory(ctx);
    for (int t = 1; t < o.threads; t++) {
      ctxs.push_back(factory.makeContext(device));
    }
  }

  if (o.rank == 0) {
    printHeader(o);
  }
  auto runAll = [&](size_t n) {
    if (o.threads == 1) {
      runOne(o, ctx, n);
      return;
    }
    std::vector<std::thread> ths;
    for (int t = 0; t < o.threads; t++) {
      Options ot = o;
      if (t != 0) {
        ot.rank = -1; // only thread 0 of rank 0 prints
      }
      ths.emplace_back([ot, &ctxs, t, n] { runOne(ot, ctxs[t], n); });
    }
    for (auto& th : ths) {
      th.join();
    }
  };
  if (o.elements >= 0) {
    runAll(o.elements);
  } else {
    // Reference README sweep: 1 ... 5,000,000.
    const long sweep[] = {1, 2, 5, 10, 20, 50, 100, 200, 500,
                          1000, 2000, 5000, 10000, 20000, 50000,
                          100000, 200000, 500000, 1000000, 2000000, 5000000};
    for (long n : sweep) {
      runAll(n);
    }
  }
  // Final sync so no rank tears down while a peer is mid-collective.
  {
    BarrierOptions barOpts(ctx);
    barOpts.tag = ctx->nextSlot();
    barrier(barOpts);
  }
  return 0;
}
