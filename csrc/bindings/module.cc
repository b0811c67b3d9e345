// pybind11 bindings for gloo_amd.
//
// Exposes devices, stores, contexts, the v2 collectives, unbound buffers
// (tagged send/recv incl. recv-from-any) and bound buffers, so the Python
// test-suite and the torch.distributed ProcessGroup wrapper can drive the
// C++ core. Buffers are passed as raw addresses (tensor.data_ptr()); all
// blocking entry points release the GIL.
#include <pybind11/chrono.h>
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "algorithms/factory.h"
#include "collectives/collectives.h"
#include "collectives/schedule.h"
#include "collectives/reduce_fns.h"
#include "common/linux.h"
#include "common/store.h"
#include "context.h"
#include "hip/algorithms.h"
#include "hip/kernels.h"
#include "rendezvous/context_factory.h"
#include "rendezvous/stores.h"
#include "transport/tcp/context.h"
#include "transport/tcp/device.h"
#include "transport/tcp/tls.h"

namespace py = pybind11;
using namespace glooamd;

namespace {

std::chrono::milliseconds ms(long v) {
  return std::chrono::milliseconds(v);
}

AllreduceOptions::Algorithm algoFromString(const std::string& s) {
  if (s == "ring") {
    return AllreduceOptions::Algorithm::RING;
  }
  if (s == "bcube") {
    return AllreduceOptions::Algorithm::BCUBE;
  }
  GA_THROW("unknown allreduce algorithm: ", s);
}

} // namespace

PYBIND11_MODULE(_C, m) {
  m.doc() = "gloo_amd: MI355X-native collective communications";

  // --- exceptions -----------------------------------------------------------
  static py::exception<Exception> excBase(m, "GlooAmdError");
  static py::exception<EnforceNotMet> excEnforce(
      m, "EnforceNotMet", excBase.ptr());
  static py::exception<IoException> excIo(m, "IoError", excBase.ptr());
  static py::exception<TimeoutException> excTimeout(
      m, "TimeoutError", excIo.ptr());
  py::register_exception_translator([](std::exception_ptr p) {
    try {
      if (p) {
        std::rethrow_exception(p);
      }
    } catch (const TimeoutException& e) {
      py::set_error(excTimeout, e.what());
    } catch (const IoException& e) {
      py::set_error(excIo, e.what());
    } catch (const EnforceNotMet& e) {
      py::set_error(excEnforce, e.what());
    } catch (const Exception& e) {
      py::set_error(excBase, e.what());
    }
  });

  // --- enums ----------------------------------------------------------------
  py::enum_<DType>(m, "DType")
      .value("f32", DType::F32)
      .value("f64", DType::F64)
      .value("f16", DType::F16)
      .value("bf16", DType::BF16)
      .value("i8", DType::I8)
      .value("u8", DType::U8)
      .value("i32", DType::I32)
      .value("i64", DType::I64)
      .value("u64", DType::U64);

  py::enum_<ReduceOp>(m, "ReduceOp")
      .value("sum", ReduceOp::SUM)
      .value("product", ReduceOp::PRODUCT)
      .value("min", ReduceOp::MIN)
      .value("max", ReduceOp::MAX);

  m.def("dtype_size", &dtypeSize);

  // schedule math (shared by CPU collectives and device engines); bound
  // for property tests — both sides of a wire must agree on boundaries.
  m.def("_block_of", [](size_t n, int p, int b) {
    auto s = sched::blockOf(n, p, b);
    return std::make_pair(s.off, s.len);
  });
  m.def("_segment_of", [](size_t n, int p, int b, int q, int S) {
    auto s = sched::segmentOf(n, p, b, q, S);
    return std::make_pair(s.off, s.len);
  });
  m.def("_subspan_of", [](size_t off, size_t len, int j, int base) {
    auto s = sched::subspanOf({off, len}, j, base);
    return std::make_pair(s.off, s.len);
  });
  m.def("block_of_a", [](size_t n, int p, int b, size_t a) {
    auto s = sched::blockOfA(n, p, b, a);
    return std::make_pair(s.off, s.len);
  });
  m.def("segment_of_a", [](size_t n, int p, int b, int q, int S, size_t a) {
    auto s = sched::segmentOfA(n, p, b, q, S, a);
    return std::make_pair(s.off, s.len);
  });
  m.def("subspan_of_a", [](size_t off, size_t len, int j, int parts,
                           size_t a) {
    auto s = sched::subspanOfA({off, len}, j, parts, a);
    return std::make_pair(s.off, s.len);
  });

  // topology helpers (common/linux)
  m.def("list_interfaces", &listInterfaces);
  m.def("interface_speed", &getInterfaceSpeedByName);
  m.def("interface_to_bus_id", &interfaceToBusID);
  m.def("pci_distance", &pciDistance);
  m.def("gpu_pci_bus_id", &hip::gpuPCIBusID);
  // NIC nearest a GPU: reference findCudaDevicePointerClosestToDevice
  // inverted for the one-NIC-per-GPU-process launch pattern.
  m.def("closest_interface_to_gpu", [](int device) {
    std::string gpuBus = hip::gpuPCIBusID(device);
    std::string best;
    int bestDist = 1 << 30;
    for (const auto& ifname : listInterfaces()) {
      std::string bus = interfaceToBusID(ifname);
      if (bus.empty()) {
        continue;
      }
      int d = pciDistance(bus, gpuBus);
      if (d < bestDist) {
        bestDist = d;
        best = ifname;
      }
    }
    return best;
  });

  // --- stores ---------------------------------------------------------------
  // Trampoline so Python classes (e.g. a torch.distributed Store adapter)
  // can implement the rendezvous interface consumed by connectFullMesh.
  class PyStore : public IStore {
   public:
    using IStore::IStore;
    void set(const std::string& key, const std::vector<char>& data) override {
      py::gil_scoped_acquire gil;
      py::function f = py::get_override(this, "set");
      GA_ENFORCE(f, "Store.set not implemented");
      f(key, py::bytes(data.data(), data.size()));
    }
    std::vector<char> get(const std::string& key) override {
      py::gil_scoped_acquire gil;
      py::function f = py::get_override(this, "get");
      GA_ENFORCE(f, "Store.get not implemented");
      std::string s = f(key).cast<std::string>();
      return std::vector<char>(s.begin(), s.end());
    }
    void wait(
        const std::vector<std::string>& keys,
        const std::chrono::milliseconds& timeout) override {
      py::gil_scoped_acquire gil;
      py::function f = py::get_override(this, "wait");
      GA_ENFORCE(f, "Store.wait not implemented");
      f(keys, timeout.count());
    }
  };

  py::class_<IStore, PyStore, std::shared_ptr<IStore>>(m, "Store")
      .def(py::init<>())
      .def(
          "set",
          [](IStore& s, const std::string& key, py::bytes data) {
            std::string d = data;
            s.set(key, std::vector<char>(d.begin(), d.end()));
          })
      .def(
          "get",
          [](IStore& s, const std::string& key) {
            std::vector<char> v;
            {
              py::gil_scoped_release rel;
              v = s.get(key);
            }
            return py::bytes(v.data(), v.size());
          })
      .def(
          "wait",
          [](IStore& s, const std::vector<std::string>& keys, long timeoutMs) {
            py::gil_scoped_release rel;
            s.wait(keys, ms(timeoutMs));
          },
          py::arg("keys"),
          py::arg("timeout_ms") = 30000)
      .def("has_v2", &IStore::hasV2)
      .def(
          "append",
          [](IStore& s, const std::string& key, py::bytes data) {
            std::string d = data;
            s.append(key, std::vector<char>(d.begin(), d.end()));
          })
      .def(
          "add",
          [](IStore& s, const std::string& key, int64_t delta) {
            py::gil_scoped_release rel;
            return s.add(key, delta);
          })
      .def(
          "multi_set",
          [](IStore& s,
             const std::vector<std::string>& keys,
             const std::vector<py::bytes>& values) {
            std::vector<std::vector<char>> vals;
            vals.reserve(values.size());
            for (const auto& v : values) {
              std::string d = v;
              vals.emplace_back(d.begin(), d.end());
            }
            py::gil_scoped_release rel;
            s.multiSet(keys, vals);
          })
      .def(
          "multi_get",
          [](IStore& s, const std::vector<std::string>& keys) {
            std::vector<std::vector<char>> vals;
            {
              py::gil_scoped_release rel;
              vals = s.multiGet(keys);
            }
            py::list out;
            for (const auto& v : vals) {
              out.append(py::bytes(v.data(), v.size()));
            }
            return out;
          });

  py::class_<HashStore, IStore, std::shared_ptr<HashStore>>(m, "HashStore")
      .def(py::init<>());
  py::class_<FileStore, IStore, std::shared_ptr<FileStore>>(m, "FileStore")
      .def(py::init<const std::string&>());
  py::class_<PrefixStore, IStore, std::shared_ptr<PrefixStore>>(
      m, "PrefixStore")
      .def(py::init<const std::string&, std::shared_ptr<IStore>>());
  py::class_<TcpStore, IStore, std::shared_ptr<TcpStore>>(m, "TcpStore")
      .def(
          py::init([](const std::string& host, int port, bool isServer,
                      long timeoutMs) {
            py::gil_scoped_release rel;
            return std::make_shared<TcpStore>(host, port, isServer,
                                              ms(timeoutMs));
          }),
          py::arg("host"),
          py::arg("port"),
          py::arg("is_server"),
          py::arg("timeout_ms") = 60000);

  // --- device ---------------------------------------------------------------
  py::class_<transport::Device, std::shared_ptr<transport::Device>>(
      m, "Device")
      .def("__str__", &transport::Device::str);

  m.def(
      "create_tcp_device",
      [](const std::string& hostname, bool useLibuv, bool useUds) {
        tcp::TcpAttr attr;
        attr.hostname = hostname;
        attr.useLibuv = useLibuv;
        attr.useUds = useUds;
        return std::static_pointer_cast<transport::Device>(
            tcp::createTcpDevice(attr));
      },
      py::arg("hostname") = std::string(),
      py::arg("use_libuv") = false,
      py::arg("use_uds") = false);

  m.def(
      "create_tls_device",
      [](const std::string& hostname, const std::string& pkey,
         const std::string& cert, const std::string& caFile) {
        tcp::tls::TlsAttr attr;
        attr.tcp.hostname = hostname;
        attr.pkeyFile = pkey;
        attr.certFile = cert;
        attr.caFile = caFile;
        return std::static_pointer_cast<transport::Device>(
            tcp::tls::createTlsDevice(attr));
      },
      py::arg("hostname") = std::string(),
      py::arg("pkey") = std::string(),
      py::arg("cert") = std::string(),
      py::arg("ca_file") = std::string());

  // --- unbound + bound buffers ---------------------------------------------
  py::class_<transport::UnboundBuffer>(m, "UnboundBuffer")
      .def(
          "send",
          [](transport::UnboundBuffer& b, int dst, uint64_t slot, size_t off,
             long nbytes) {
            py::gil_scoped_release rel;
            b.send(dst, slot, off,
                   nbytes < 0 ? transport::kUnspecified : size_t(nbytes));
          },
          py::arg("dst"),
          py::arg("slot"),
          py::arg("offset") = 0,
          py::arg("nbytes") = -1)
      .def(
          "recv",
          [](transport::UnboundBuffer& b, int src, uint64_t slot, size_t off,
             long nbytes) {
            py::gil_scoped_release rel;
            b.recv(src, slot, off,
                   nbytes < 0 ? transport::kUnspecified : size_t(nbytes));
          },
          py::arg("src"),
          py::arg("slot"),
          py::arg("offset") = 0,
          py::arg("nbytes") = -1)
      .def(
          "recv_any",
          [](transport::UnboundBuffer& b, const std::vector<int>& srcs,
             uint64_t slot, size_t off, long nbytes) {
            py::gil_scoped_release rel;
            b.recv(srcs, slot, off,
                   nbytes < 0 ? transport::kUnspecified : size_t(nbytes));
          },
          py::arg("srcs"),
          py::arg("slot"),
          py::arg("offset") = 0,
          py::arg("nbytes") = -1)
      .def(
          "wait_recv",
          [](transport::UnboundBuffer& b, long timeoutMs) {
            int src = -1;
            bool ok;
            {
              py::gil_scoped_release rel;
              ok = b.waitRecv(&src, ms(timeoutMs));
            }
            return py::make_tuple(ok, src);
          },
          py::arg("timeout_ms") = -1)
      .def(
          "wait_send",
          [](transport::UnboundBuffer& b, long timeoutMs) {
            py::gil_scoped_release rel;
            return b.waitSend(ms(timeoutMs));
          },
          py::arg("timeout_ms") = -1)
      .def(
          "try_wait_recv",
          [](transport::UnboundBuffer& b, long timeoutMs) {
            int src = -1;
            bool ok;
            {
              py::gil_scoped_release rel;
              ok = b.tryWaitRecv(&src, ms(timeoutMs));
            }
            return py::make_tuple(ok, src);
          },
          py::arg("timeout_ms") = -1,
          "Probing wait: returns (False, -1) on timeout without "
          "poisoning the context (monitored_barrier liveness probes).")
      .def("abort_wait_recv", &transport::UnboundBuffer::abortWaitRecv)
      .def("abort_wait_send", &transport::UnboundBuffer::abortWaitSend);

  py::class_<transport::Buffer>(m, "Buffer")
      .def(
          "send",
          [](transport::Buffer& b, size_t off, long length, size_t roff) {
            py::gil_scoped_release rel;
            b.send(off, length < 0 ? b.size() - off : size_t(length), roff);
          },
          py::arg("offset") = 0,
          py::arg("length") = -1,
          py::arg("roffset") = 0)
      .def("wait_recv",
           [](transport::Buffer& b) {
             py::gil_scoped_release rel;
             b.waitRecv();
           })
      .def("wait_send", [](transport::Buffer& b) {
        py::gil_scoped_release rel;
        b.waitSend();
      });

  py::class_<transport::Pair>(m, "Pair")
      .def(
          "create_send_buffer",
          [](transport::Pair& p, uint64_t slot, uintptr_t ptr, size_t size) {
            return p.createSendBuffer(slot, reinterpret_cast<void*>(ptr),
                                      size);
          },
          py::keep_alive<0, 1>())
      .def(
          "create_recv_buffer",
          [](transport::Pair& p, uint64_t slot, uintptr_t ptr, size_t size) {
            return p.createRecvBuffer(slot, reinterpret_cast<void*>(ptr),
                                      size);
          },
          py::keep_alive<0, 1>())
      .def("is_connected", &transport::Pair::isConnected)
      .def("__str__", &transport::Pair::str);

  // --- context --------------------------------------------------------------
  py::class_<Context, std::shared_ptr<Context>>(m, "Context")
      .def(py::init<int, int, int>(), py::arg("rank"), py::arg("size"),
           py::arg("base") = 2)
      .def_readonly("rank", &Context::rank)
      .def_readonly("size", &Context::size)
      .def_readwrite("base", &Context::base)
      .def(
          "connect_full_mesh",
          [](Context& ctx, std::shared_ptr<IStore> store,
             std::shared_ptr<transport::Device> dev) {
            py::gil_scoped_release rel;
            ctx.connectFullMesh(*store, dev);
          },
          py::arg("store"),
          py::arg("device"))
      .def("set_timeout",
           [](Context& ctx, long timeoutMs) { ctx.setTimeout(ms(timeoutMs)); })
      .def("close", &Context::closeConnections,
           py::call_guard<py::gil_scoped_release>())
      .def("next_slot", &Context::nextSlot, py::arg("num_to_skip") = 1)
      .def("slot_counter", &Context::slotCounter)
      .def("reset_slot_counter", &Context::resetSlotCounter, py::arg("value"))
      .def(
          "get_pair",
          [](Context& ctx, int rank) { return ctx.getPair(rank); },
          py::return_value_policy::reference_internal)
      .def(
          "create_unbound_buffer",
          [](Context& ctx, uintptr_t ptr, size_t size) {
            return ctx.createUnboundBuffer(reinterpret_cast<void*>(ptr),
                                           size);
          },
          py::keep_alive<0, 1>());

  py::class_<ContextFactory>(m, "ContextFactory")
      .def(py::init<std::shared_ptr<Context>>(), py::arg("backing_context"))
      .def(
          "make_context",
          [](ContextFactory& f, std::shared_ptr<transport::Device> dev) {
            py::gil_scoped_release rel;
            return f.makeContext(dev);
          },
          py::arg("device"));

  // --- collectives ----------------------------------------------------------
  m.def(
      "allreduce",
      [](std::shared_ptr<Context> ctx, std::vector<uintptr_t> outputs,
         size_t elements, DType dtype, ReduceOp op,
         std::vector<uintptr_t> inputs, uint32_t tag,
         const std::string& algorithm, size_t maxSegmentSize,
         long timeoutMs) {
        AllreduceOptions opts(ctx);
        for (auto p : outputs) {
          opts.outputs.push_back(reinterpret_cast<void*>(p));
        }
        for (auto p : inputs) {
          opts.inputs.push_back(reinterpret_cast<void*>(p));
        }
        opts.elements = elements;
        opts.elementSize = dtypeSize(dtype);
        opts.reduce = cpuReduceFn(dtype, op);
        opts.tag = tag;
        opts.algorithm = algoFromString(algorithm);
        opts.maxSegmentSize = maxSegmentSize;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        allreduce(opts);
      },
      py::arg("context"),
      py::arg("outputs"),
      py::arg("elements"),
      py::arg("dtype") = DType::F32,
      py::arg("op") = ReduceOp::SUM,
      py::arg("inputs") = std::vector<uintptr_t>(),
      py::arg("tag") = 0,
      py::arg("algorithm") = "ring",
      py::arg("max_segment_size") = kDefaultMaxSegmentSize,
      py::arg("timeout_ms") = 0);

  m.def(
      "allgather",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         size_t inElements, DType dtype, uint32_t tag, long timeoutMs) {
        AllgatherOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.inElements = inElements;
        opts.elementSize = dtypeSize(dtype);
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        allgather(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("in_elements"),
      py::arg("dtype") = DType::F32,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "allgatherv",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         std::vector<size_t> counts, DType dtype, uint32_t tag,
         long timeoutMs) {
        AllgathervOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.counts = std::move(counts);
        opts.elementSize = dtypeSize(dtype);
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        allgatherv(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("counts"),
      py::arg("dtype") = DType::F32,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "alltoall",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         size_t perRankElements, DType dtype, uint32_t tag, long timeoutMs) {
        AlltoallOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.perRankElements = perRankElements;
        opts.elementSize = dtypeSize(dtype);
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        alltoall(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("per_rank_elements"),
      py::arg("dtype") = DType::F32,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "alltoallv",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         std::vector<size_t> inCounts, std::vector<size_t> outCounts,
         DType dtype, uint32_t tag, long timeoutMs) {
        AlltoallvOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.inCounts = std::move(inCounts);
        opts.outCounts = std::move(outCounts);
        opts.elementSize = dtypeSize(dtype);
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        alltoallv(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("in_counts"),
      py::arg("out_counts"),
      py::arg("dtype") = DType::F32,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "barrier",
      [](std::shared_ptr<Context> ctx, uint32_t tag, long timeoutMs) {
        BarrierOptions opts(ctx);
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        barrier(opts);
      },
      py::arg("context"),
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "broadcast",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         size_t elements, DType dtype, int root, uint32_t tag,
         long timeoutMs) {
        BroadcastOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.elements = elements;
        opts.elementSize = dtypeSize(dtype);
        opts.root = root;
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        broadcast(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("elements"),
      py::arg("dtype") = DType::F32,
      py::arg("root") = 0,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "gather",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         size_t inElements, DType dtype, int root, uint32_t tag,
         long timeoutMs) {
        GatherOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.inElements = inElements;
        opts.elementSize = dtypeSize(dtype);
        opts.root = root;
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        gather(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("in_elements"),
      py::arg("dtype") = DType::F32,
      py::arg("root") = 0,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "gatherv",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         std::vector<size_t> counts, DType dtype, int root, uint32_t tag,
         long timeoutMs) {
        GathervOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.counts = std::move(counts);
        opts.elementSize = dtypeSize(dtype);
        opts.root = root;
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        gatherv(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("counts"),
      py::arg("dtype") = DType::F32,
      py::arg("root") = 0,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "reduce",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         size_t elements, DType dtype, ReduceOp op, int root, uint32_t tag,
         long timeoutMs) {
        ReduceOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.elements = elements;
        opts.elementSize = dtypeSize(dtype);
        opts.reduce = cpuReduceFn(dtype, op);
        opts.root = root;
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        reduce(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("elements"),
      py::arg("dtype") = DType::F32,
      py::arg("op") = ReduceOp::SUM,
      py::arg("root") = 0,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "scatter",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         size_t outElements, DType dtype, int root, uint32_t tag,
         long timeoutMs) {
        ScatterOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.outElements = outElements;
        opts.elementSize = dtypeSize(dtype);
        opts.root = root;
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        scatter(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("out_elements"),
      py::arg("dtype") = DType::F32,
      py::arg("root") = 0,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  m.def(
      "reduce_scatter",
      [](std::shared_ptr<Context> ctx, uintptr_t output, uintptr_t input,
         size_t recvElements, DType dtype, ReduceOp op, uint32_t tag,
         long timeoutMs) {
        ReduceScatterOptions opts(ctx);
        opts.output = reinterpret_cast<void*>(output);
        opts.input = reinterpret_cast<void*>(input);
        opts.recvElements = recvElements;
        opts.elementSize = dtypeSize(dtype);
        opts.reduce = cpuReduceFn(dtype, op);
        opts.tag = tag;
        opts.timeout = ms(timeoutMs);
        py::gil_scoped_release rel;
        reduce_scatter(opts);
      },
      py::arg("context"),
      py::arg("output"),
      py::arg("input"),
      py::arg("recv_elements"),
      py::arg("dtype") = DType::F32,
      py::arg("op") = ReduceOp::SUM,
      py::arg("tag") = 0,
      py::arg("timeout_ms") = 0);

  // --- legacy Algorithm classes --------------------------------------------
  py::class_<Algorithm>(m, "Algorithm")
      .def("run", &Algorithm::run, py::call_guard<py::gil_scoped_release>());

  m.def(
      "create_algorithm",
      [](const std::string& name, std::shared_ptr<Context> ctx,
         std::vector<uintptr_t> ptrs, size_t count, DType dtype, ReduceOp op,
         int root, std::vector<int> recvElems, size_t bytes, int steps) {
        std::vector<void*> p;
        for (auto v : ptrs) {
          p.push_back(reinterpret_cast<void*>(v));
        }
        py::gil_scoped_release rel;
        return createAlgorithm(
            name, ctx, p, count, dtype, op, root, recvElems, bytes, steps);
      },
      py::arg("name"),
      py::arg("context"),
      py::arg("ptrs") = std::vector<uintptr_t>(),
      py::arg("count") = 0,
      py::arg("dtype") = DType::F32,
      py::arg("op") = ReduceOp::SUM,
      py::arg("root") = 0,
      py::arg("recv_elems") = std::vector<int>(),
      py::arg("bytes") = 0,
      py::arg("steps") = 1,
      py::keep_alive<0, 2>());

  // --- HIP / xGMI device collectives ---------------------------------------
  m.def("hip_available", &hip::available);
  m.def("hip_device_count", [] {
    return hip::available() ? hip::deviceCount() : 0;
  });

  // Raw kernel entry points for numerics tests (device pointers; blocks
  // until complete on the default-constructed stream semantics: caller
  // synchronizes via torch).
  m.def(
      "hip_reduce2",
      [](uintptr_t dst, uintptr_t a, uintptr_t b, size_t n, DType dt,
         ReduceOp op, uintptr_t stream) {
        py::gil_scoped_release rel;
        hip::launchReduce2(
            reinterpret_cast<void*>(dst),
            reinterpret_cast<const void*>(a),
            reinterpret_cast<const void*>(b),
            n,
            dt,
            op,
            reinterpret_cast<hipStream_t>(stream));
      },
      py::arg("dst"),
      py::arg("a"),
      py::arg("b"),
      py::arg("n"),
      py::arg("dtype"),
      py::arg("op"),
      py::arg("stream") = 0);

  m.def(
      "hip_allreduce_local",
      [](std::vector<uintptr_t> ptrs, size_t n, DType dt, ReduceOp op,
         int device, uintptr_t stream) {
        std::vector<void*> p;
        for (auto v : ptrs) {
          p.push_back(reinterpret_cast<void*>(v));
        }
        py::gil_scoped_release rel;
        hip::hipAllreduceLocal(p, n, dt, op, device,
                               reinterpret_cast<hipStream_t>(stream));
      },
      py::arg("ptrs"),
      py::arg("n"),
      py::arg("dtype"),
      py::arg("op"),
      py::arg("device") = 0,
      py::arg("stream") = 0);

  py::class_<hip::HipAllreduceRing>(m, "HipAllreduceRing")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device, bool chunked,
                      size_t inboxCap) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipAllreduceRing>(
                ctx, device, chunked, inboxCap);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("chunked") = true,
          py::arg("inbox_cap") = 0)
      .def(
          "run",
          [](hip::HipAllreduceRing& a, uintptr_t ptr, size_t n, DType dt,
             ReduceOp op, uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<void*>(ptr), n, dt, op,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("ptr"),
          py::arg("elements"),
          py::arg("dtype") = DType::F32,
          py::arg("op") = ReduceOp::SUM,
          py::arg("stream") = 0)
      .def(
          "run_multi",
          [](hip::HipAllreduceRing& a, const std::vector<uintptr_t>& ptrs,
             size_t n, DType dt, ReduceOp op, uintptr_t stream) {
            std::vector<void*> vp;
            vp.reserve(ptrs.size());
            for (auto p : ptrs) {
              vp.push_back(reinterpret_cast<void*>(p));
            }
            py::gil_scoped_release rel;
            a.run(vp, n, dt, op, reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("ptrs"),
          py::arg("elements"),
          py::arg("dtype") = DType::F32,
          py::arg("op") = ReduceOp::SUM,
          py::arg("stream") = 0);

  py::class_<hip::HipAllreduceHalvingDoubling>(m, "HipAllreduceHalvingDoubling")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device,
                      size_t inboxCap) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipAllreduceHalvingDoubling>(
                ctx, device, inboxCap);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("inbox_cap") = 0)
      .def(
          "run",
          [](hip::HipAllreduceHalvingDoubling& a, uintptr_t ptr, size_t n,
             DType dt, ReduceOp op, uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<void*>(ptr), n, dt, op,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("ptr"),
          py::arg("elements"),
          py::arg("dtype") = DType::F32,
          py::arg("op") = ReduceOp::SUM,
          py::arg("stream") = 0);

  py::class_<hip::HipAllreduceBcube>(m, "HipAllreduceBcube")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device, int base) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipAllreduceBcube>(ctx, device,
                                                            base);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("base") = 0)
      .def(
          "run",
          [](hip::HipAllreduceBcube& a, uintptr_t ptr, size_t n, DType dt,
             ReduceOp op, uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<void*>(ptr), n, dt, op,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("ptr"),
          py::arg("elements"),
          py::arg("dtype") = DType::F32,
          py::arg("op") = ReduceOp::SUM,
          py::arg("stream") = 0);

  py::class_<hip::HipAllreduceDirect>(m, "HipAllreduceDirect")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device,
                      int numStreams) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipAllreduceDirect>(ctx, device,
                                                             numStreams);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("num_streams") = 7)
      .def(
          "run",
          [](hip::HipAllreduceDirect& a, uintptr_t ptr, size_t n, DType dt,
             ReduceOp op, uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<void*>(ptr), n, dt, op,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("ptr"),
          py::arg("elements"),
          py::arg("dtype") = DType::F32,
          py::arg("op") = ReduceOp::SUM,
          py::arg("stream") = 0)
      .def(
          "run_multi",
          [](hip::HipAllreduceDirect& a, const std::vector<uintptr_t>& ptrs,
             size_t n, DType dt, ReduceOp op, uintptr_t stream) {
            std::vector<void*> vp;
            vp.reserve(ptrs.size());
            for (auto p : ptrs) {
              vp.push_back(reinterpret_cast<void*>(p));
            }
            py::gil_scoped_release rel;
            a.run(vp, n, dt, op, reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("ptrs"),
          py::arg("elements"),
          py::arg("dtype") = DType::F32,
          py::arg("op") = ReduceOp::SUM,
          py::arg("stream") = 0);

  py::class_<hip::HipAllgatherRing>(m, "HipAllgatherRing")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device,
                      size_t inboxCap) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipAllgatherRing>(
                ctx, device, inboxCap);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("inbox_cap") = 0)
      .def(
          "run",
          [](hip::HipAllgatherRing& a, uintptr_t in, uintptr_t out,
             size_t inElements, size_t es, uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<const void*>(in),
                  reinterpret_cast<void*>(out), inElements, es,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("in_ptr"),
          py::arg("out_ptr"),
          py::arg("in_elements"),
          py::arg("element_size") = 4,
          py::arg("stream") = 0);

  py::class_<hip::HipReduceScatterRing>(m, "HipReduceScatterRing")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device,
                      size_t inboxCap) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipReduceScatterRing>(
                ctx, device, inboxCap);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("inbox_cap") = 0)
      .def(
          "run",
          [](hip::HipReduceScatterRing& a, uintptr_t in, uintptr_t out,
             size_t recvElements, DType dt, ReduceOp op, uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<const void*>(in),
                  reinterpret_cast<void*>(out), recvElements, dt, op,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("in_ptr"),
          py::arg("out_ptr"),
          py::arg("recv_elements"),
          py::arg("dtype") = DType::F32,
          py::arg("op") = ReduceOp::SUM,
          py::arg("stream") = 0);

  py::class_<hip::HipAlltoall>(m, "HipAlltoall")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device,
                      int numStreams) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipAlltoall>(ctx, device,
                                                      numStreams);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("num_streams") = 4)
      .def(
          "run",
          [](hip::HipAlltoall& a, uintptr_t in, uintptr_t out,
             size_t perRankElements, size_t es, uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<const void*>(in),
                  reinterpret_cast<void*>(out), perRankElements, es,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("in_ptr"),
          py::arg("out_ptr"),
          py::arg("per_rank_elements"),
          py::arg("element_size") = 4,
          py::arg("stream") = 0);

  py::class_<hip::HipP2P>(m, "HipP2P")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device,
                      size_t chunkCap) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipP2P>(ctx, device, chunkCap);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("chunk_cap") = 0)
      .def(
          "post_send",
          [](hip::HipP2P& p, int dst, uintptr_t ptr, size_t bytes,
             uintptr_t stream) {
            py::gil_scoped_release rel;
            p.postSend(dst, reinterpret_cast<const void*>(ptr), bytes,
                       reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("dst"),
          py::arg("ptr"),
          py::arg("bytes"),
          py::arg("stream") = 0)
      .def(
          "post_recv",
          [](hip::HipP2P& p, int src, uintptr_t ptr, size_t bytes,
             uintptr_t stream) {
            py::gil_scoped_release rel;
            p.postRecv(src, reinterpret_cast<void*>(ptr), bytes,
                       reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("src"),
          py::arg("ptr"),
          py::arg("bytes"),
          py::arg("stream") = 0)
      .def("flush_sends",
           [](hip::HipP2P& p) {
             py::gil_scoped_release rel;
             p.flushSends();
           })
      .def("flush_recvs", [](hip::HipP2P& p) {
        py::gil_scoped_release rel;
        p.flushRecvs();
      });

  py::class_<hip::HipBroadcastOneToAll>(m, "HipBroadcastOneToAll")
      .def(
          py::init([](std::shared_ptr<Context> ctx, int device, int root,
                      int numStreams) {
            py::gil_scoped_release rel;
            return std::make_unique<hip::HipBroadcastOneToAll>(
                ctx, device, root, numStreams);
          }),
          py::arg("context"),
          py::arg("device"),
          py::arg("root") = 0,
          py::arg("num_streams") = 4)
      .def(
          "run",
          [](hip::HipBroadcastOneToAll& a, uintptr_t ptr, size_t bytes,
             uintptr_t stream) {
            py::gil_scoped_release rel;
            a.run(reinterpret_cast<void*>(ptr), bytes,
                  reinterpret_cast<hipStream_t>(stream));
          },
          py::arg("ptr"),
          py::arg("bytes"),
          py::arg("stream") = 0)
      .def("debug_flags", &hip::HipBroadcastOneToAll::debugFlags);
}
