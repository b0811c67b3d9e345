#include "hip/algorithms.h"

#include <atomic>
#include <chrono>
#include <thread>

#include "collectives/collectives.h"
#include "collectives/schedule.h"
#include "common/utils.h"
#include "hip/kernels.h"
#include "hip/trace.h"

namespace glooamd {
namespace hip {

using sched::Seg;
using sched::chunkOf;
using sched::segmentOf;
using sched::subspanOf;

size_t onDeviceThreshold() {
  static size_t v = static_cast<size_t>(
      getEnvInt("GLOO_AMD_DEVICE_THRESHOLD", 256 * 1024));
  return v;
}

namespace {
constexpr size_t kDefaultInboxCap = 4 << 20; // 4 MiB per inbox

// Wait for `ev` with a deadline; on expiry poison the mesh flags so the
// GPU-side waits drain, then throw. Fail-fast model like the reference's
// pair exception fan-out.
void watchdogWait(
    HipEvent& ev,
    XgmiMesh& mesh,
    std::chrono::milliseconds timeout,
    const char* what) {
  auto deadline = std::chrono::steady_clock::now() + timeout;
  while (!ev.query()) {
    if (timeout.count() > 0 && std::chrono::steady_clock::now() > deadline) {
      std::string flags;
      if (getEnvFlag("GLOO_AMD_FLAG_DEBUG")) {
        uint64_t vals[16] = {0};
        (void)hipMemcpy(vals, mesh.flag(0), sizeof(vals),
                        hipMemcpyDeviceToHost);
        for (int i = 0; i < 16; i++) {
          flags += " " + std::to_string(vals[i]);
        }
      }
      mesh.poisonFlags();
      // Bounded drain: a wedged hardware queue can make
      // hipDeviceSynchronize hang forever; give it a few seconds on a
      // helper thread, then fail fast (a leaked sync thread on a
      // broken device is acceptable in the fail-fast model — the
      // throw is what keeps the job from hanging).
      {
        auto done = std::make_shared<std::atomic<bool>>(false);
        std::thread([done] {
          (void)hipDeviceSynchronize();
          done->store(true);
        }).detach();
        const auto dl =
            std::chrono::steady_clock::now() + std::chrono::seconds(8);
        while (!done->load() && std::chrono::steady_clock::now() < dl) {
          std::this_thread::sleep_for(std::chrono::milliseconds(10));
        }
      }
      GA_THROW_IO("hip collective timed out in ", what, " flags:", flags);
    }
    std::this_thread::yield();
  }
}


// Bounded teardown drain: never hang a destructor on a wedged stream.
void drainBounded(HipStream* s, int ms = 2000) {
  if (s == nullptr) {
    return;
  }
  const auto dl = std::chrono::steady_clock::now() +
      std::chrono::milliseconds(ms);
  while (hipStreamQuery(s->stream()) == hipErrorNotReady &&
         std::chrono::steady_clock::now() < dl) {
    std::this_thread::sleep_for(std::chrono::milliseconds(1));
  }
  (void)hipGetLastError();
}

// Phase tracing for hang localization (GLOO_AMD_PHASE_TRACE=1): the
// last line printed before a stall names the blocking call.
bool phaseTrace() {
  static bool v = getEnvFlag("GLOO_AMD_PHASE_TRACE");
  return v;
}
#define GA_PHASE(msg)                                        \
  do {                                                       \
    if (phaseTrace()) {                                      \
      GA_ERROR << "[phase r" << ctx_->rank << "] " << msg;   \
    }                                                        \
  } while (0)

// Order our streams after the caller's stream (where the input tensors
// were produced). nullptr = legacy default stream, which covers torch's
// default current stream. The gate event must be PERSISTENT (owned by
// the engine): destroying an event while a hipStreamWaitEvent on it is
// still pending on a busy stream can wedge that stream permanently
// (observed as flaky engine-restart deadlocks; re-recording a
// persistent event is well-defined — pending waits keep the snapshot
// they were enqueued with).
// GLOO_AMD_GATE_SYNC=1: instead of recording a gate event on the
// caller's stream, host-synchronize it (isolation knob for legacy
// null-stream event-record hazards in multi-threaded processes; costs
// compute/comm overlap, so event gating stays the default).
bool gateBySync() {
  static bool v = getEnvFlag("GLOO_AMD_GATE_SYNC");
  return v;
}

void gateStreams(
    HipEvent& gateEv,
    hipStream_t caller,
    std::initializer_list<hipStream_t> gated) {
  if (gateBySync()) {
    GA_HIP_CHECK(hipStreamSynchronize(caller));
    return;
  }
  gateEv.record(caller);
  for (auto s : gated) {
    gateEv.streamWait(s);
  }
}
} // namespace

// ===========================================================================
// HipAllreduceRing
// ===========================================================================

HipAllreduceRing::~HipAllreduceRing() {
  for (size_t j = 0; j < cs_.size(); j++) {
    drainBounded(cs_[j]);
    drainBounded(ks_[j]);
  }
  for (auto& kv : graphs_) {
    if (kv.second.exec != nullptr) {
      (void)hipGraphExecDestroy(kv.second.exec);
    }
  }
  if (seqBaseDev_ != nullptr) {
    (void)hipFree(seqBaseDev_);
  }
}

HipAllreduceRing::HipAllreduceRing(
    std::shared_ptr<Context> ctx,
    int device,
    bool chunked,
    size_t inboxCap,
    int numRings)
    : ctx_(std::move(ctx)),
      device_(device),
      chunked_(chunked),
      inboxCap_(inboxCap == 0 ? kDefaultInboxCap : inboxCap) {
  GA_HIP_CHECK(hipSetDevice(device_));
  mesh_ = std::make_unique<XgmiMesh>(ctx_, device_, 0, inboxCap_);

  // Ring strides coprime to size: each ring's neighbor hop uses a
  // different xGMI link, so the rings' wire traffic runs concurrently.
  const int P = ctx_->size;
  if (numRings == 0) {
    // tunable without a rebuild for link-level experiments on hardware
    numRings = static_cast<int>(getEnvInt("GLOO_AMD_NUM_RINGS", 4));
  }
  const int maxRings =
      std::max(1, std::min(numRings, kStreamPoolSize / 2));
  for (int st = 1; st < std::max(P, 2) &&
       static_cast<int>(strides_.size()) < maxRings;
       st++) {
    int a = st, b = P;
    while (b != 0) {
      int t = a % b;
      a = b;
      b = t;
    }
    if (a == 1) { // gcd(st, P) == 1 -> single cycle
      strides_.push_back(st);
    }
  }
  if (strides_.empty()) {
    strides_.push_back(1);
  }
  const int R = static_cast<int>(strides_.size());
  cs_.resize(R);
  ks_.resize(R);
  events_.resize(R);
  for (int j = 0; j < R; j++) {
    cs_[j] = pooledStream(ctx_.get(), device_, 2 * j);
    ks_[j] = pooledStream(ctx_.get(), device_, 2 * j + 1);
    initEvent_.push_back(std::make_unique<HipEvent>(device_));
    doneEvent_.push_back(std::make_unique<HipEvent>(device_));
    fDATA_.push_back(mesh_->allocFlags(2));
    fACK_.push_back(mesh_->allocFlags(2));
  }
  gateEv_ = std::make_unique<HipEvent>(device_);
  graphFork_ = std::make_unique<HipEvent>(device_);
  lastAckPerRing_.assign(R, {0, 0});
}

void HipAllreduceRing::run(
    void* devPtr,
    size_t elements,
    DType dtype,
    ReduceOp op,
    hipStream_t callerStream) {
  std::vector<void*> ptrs{devPtr};
  run(ptrs, elements, dtype, op, callerStream);
}

void HipAllreduceRing::run(
    const std::vector<void*>& ptrs,
    size_t elements,
    DType dtype,
    ReduceOp op,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_allreduce_ring");
  GA_HIP_CHECK(hipSetDevice(device_));
  GA_ENFORCE(!ptrs.empty());
  const size_t es = dtypeSize(dtype);
  const size_t bytes = elements * es;
  if (elements == 0 || (ctx_->size == 1 && ptrs.size() == 1)) {
    return;
  }
  if (gateBySync()) {
    GA_HIP_CHECK(hipStreamSynchronize(callerStream));
  } else {
    gateEv_->record(callerStream);
    for (size_t j = 0; j < cs_.size(); j++) {
      gateEv_->streamWait(cs_[j]->stream());
      gateEv_->streamWait(ks_[j]->stream());
    }
  }
  char* buf = static_cast<char*>(ptrs[0]);
  if (ptrs.size() > 1) {
    // Fused local reduction of the caller's pointers into ptrs[0] as the
    // ring's copy-in stage (reference cuda_allreduce_ring.cc:72-120
    // reduces N local pointers before the wire phase; here one fused
    // k-way kernel replaces its k-1 sweeps).
    const int k = static_cast<int>(std::min<size_t>(ptrs.size(), 8));
    launchReduceN(
        buf, const_cast<const void* const*>(ptrs.data()), k, elements,
        dtype, op, cs_[0]->stream());
    for (size_t i = 8; i < ptrs.size(); i++) {
      launchReduce2(buf, buf, ptrs[i], elements, dtype, op,
                    cs_[0]->stream());
    }
    // Re-record the persistent gate event: the earlier caller-gating
    // waits keep their snapshot; destroying a transient event with
    // pending waits is what the persistent event exists to avoid.
    gateEv_->record(cs_[0]->stream());
    for (size_t j = 0; j < cs_.size(); j++) {
      gateEv_->streamWait(cs_[j]->stream());
      gateEv_->streamWait(ks_[j]->stream());
    }
  }
  if (ctx_->size > 1) {
    if (bytes < onDeviceThreshold()) {
      runHostStaged(buf, bytes, elements, dtype, op);
    } else {
      runDevice(buf, bytes, elements, dtype, op);
    }
  }
  if (ptrs.size() > 1) {
    // Broadcast the reduced result back to every caller pointer.
    for (size_t i = 1; i < ptrs.size(); i++) {
      GA_HIP_CHECK(hipMemcpyAsync(
          ptrs[i], buf, bytes, hipMemcpyDeviceToDevice, cs_[0]->stream()));
    }
    cs_[0]->synchronize();
  }
}

void HipAllreduceRing::runHostStaged(
    char* buf,
    size_t bytes,
    size_t n,
    DType dt,
    ReduceOp op) {
  if (hostStageCap_ < bytes) {
    if (hostStage_ != nullptr) {
      (void)hipHostFree(hostStage_);
    }
    GA_HIP_CHECK(hipHostMalloc(&hostStage_, bytes));
    hostStageCap_ = bytes;
  }
  GA_HIP_CHECK(hipMemcpyAsync(
      hostStage_, buf, bytes, hipMemcpyDeviceToHost, cs_[0]->stream()));
  cs_[0]->synchronize();
  AllreduceOptions opts(ctx_);
  opts.outputs = {hostStage_};
  opts.elements = n;
  opts.elementSize = dtypeSize(dt);
  opts.reduce = cpuReduceFn(dt, op);
  opts.tag = 0x7fff0000u; // reserved tag band for hip host staging
  allreduce(opts);
  GA_HIP_CHECK(hipMemcpyAsync(
      buf, hostStage_, bytes, hipMemcpyHostToDevice, cs_[0]->stream()));
  cs_[0]->synchronize();
}

// Segment count for one ring's per-rank block: sized to the per-ring
// inbox slot, with split boundaries rounded to 16 B (so a segment can
// grow by up to 16 B past the naive ceil — hence the fit loop).
int HipAllreduceRing::ringSegments(size_t partLen, size_t es) const {
  const int P = ctx_->size;
  const int R = static_cast<int>(strides_.size());
  const size_t A = std::max<size_t>(1, 16 / es);
  const size_t subCap = inboxCap_ / R;
  const size_t perRank = sched::alignUp((partLen + P - 1) / P, A);
  int S = chunked_
      ? std::max<int>(
            2, static_cast<int>((perRank * es + subCap - 1) / subCap))
      : std::max<int>(
            1, static_cast<int>((perRank * es + subCap - 1) / subCap));
  while (sched::alignUp((perRank + S - 1) / S, A) * es > subCap) {
    S++;
  }
  return S;
}

int HipAllreduceRing::enqueueRing(
    int j,
    char* work,
    size_t elemOff,
    size_t n,
    size_t es,
    DType dt,
    ReduceOp op,
    bool rel,
    int64_t crossRunOff) {
  const int P = ctx_->size;
  const int r = ctx_->rank;
  const int stride = strides_[j];
  const int right = (r + stride) % P;
  const int left = (r - stride + P) % P;
  // Virtual position of this rank along the stride-ring.
  int vr = 0;
  for (int t = 0; t < P; t++) {
    if ((t * stride) % P == r) {
      vr = t;
      break;
    }
  }

  const int R = static_cast<int>(strides_.size());
  const size_t subCap = inboxCap_ / R; // per-(ring,parity) inbox bytes
  const size_t A = std::max<size_t>(1, 16 / es); // 16-B split alignment
  const int S = ringSegments(n, es);
  auto ringInbox = [&](char* base, int par) {
    return base + (static_cast<size_t>(j) * 2 + par) * subCap;
  };
  (void)subCap;

  struct Step {
    Seg send;
    Seg recv;
    bool reduceStep;
  };
  const int K1 = (P - 1) * S;
  std::vector<Step> steps;
  steps.reserve(2 * K1);
  for (int k = 0; k < K1; k++) {
    int i = k / S, q = k % S;
    steps.push_back({sched::segmentOfA(n, P, (vr - i + P) % P, q, S, A),
                     sched::segmentOfA(n, P, (vr - i - 1 + 2 * P) % P, q, S, A),
                     true});
  }
  for (int k = 0; k < K1; k++) {
    int i = k / S, q = k % S;
    steps.push_back({sched::segmentOfA(n, P, (vr + 1 - i + P) % P, q, S, A),
                     sched::segmentOfA(n, P, (vr - i + P) % P, q, S, A),
                     false});
  }
  const int K = static_cast<int>(steps.size());
  ringSteps_ = std::max<uint64_t>(ringSteps_, K);

  const int pool = S + 2;
  auto& evs = events_[j];
  while (static_cast<int>(evs.size()) < pool) {
    evs.push_back(std::make_unique<HipEvent>(device_));
  }
  auto csm = cs_[j]->stream();
  auto ksm = ks_[j]->stream();
  char* base = work + elemOff * es;
  auto seqOf = [&](int k) { return seqBase_ + k + 1; };

  for (int k = 0; k < K; k++) {
    const Step& st = steps[k];
    const int par = k & 1;
    if (rel) {
      // Graph mode: doorbell targets are *seqBaseDev_ + offset at
      // EXECUTION time. Within-run gates are exact; the k<2 cross-run
      // inbox-reuse gate needs the previous run's same-key shape
      // (runDeviceGraph only replays when the previous run had this
      // key), corrected per ring for K_j < max K (crossRunOff).
      const int64_t off = (k >= 2)
          ? static_cast<int64_t>(k) - 1
          : crossRunOff + static_cast<int64_t>(k) - 1;
      launchWaitFlagGteRel(mesh_->flag(fACK_[j] + par), seqBaseDev_, off,
                           ksm);
    } else {
      const uint64_t prevAck =
          (k >= 2) ? seqOf(k - 2) : lastAckPerRing_[j][par];
      if (prevAck > 0) {
        launchWaitFlagGte(mesh_->flag(fACK_[j] + par), prevAck, ksm);
      }
    }
    if (k >= S) {
      evs[(k - S) % pool]->streamWait(ksm);
    } else {
      initEvent_[j]->streamWait(ksm);
    }
    if (st.send.len > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          ringInbox(mesh_->peerInbox(right, 0), par),
          base + st.send.off * es,
          st.send.len * es,
          hipMemcpyDeviceToDevice,
          ksm));
    }
    if (rel) {
      launchWriteFlagRel(
          mesh_->peerFlag(right, fDATA_[j] + par), seqBaseDev_, k + 1, ksm);
      launchWaitFlagGteRel(
          mesh_->flag(fDATA_[j] + par), seqBaseDev_, k + 1, csm);
    } else {
      launchWriteFlag(mesh_->peerFlag(right, fDATA_[j] + par), seqOf(k),
                      ksm);
      launchWaitFlagGte(mesh_->flag(fDATA_[j] + par), seqOf(k), csm);
    }
    if (st.recv.len > 0) {
      if (st.reduceStep) {
        launchReduce2(
            base + st.recv.off * es,
            base + st.recv.off * es,
            ringInbox(mesh_->inbox(0), par),
            st.recv.len,
            dt,
            op,
            csm);
      } else {
        GA_HIP_CHECK(hipMemcpyAsync(
            base + st.recv.off * es,
            ringInbox(mesh_->inbox(0), par),
            st.recv.len * es,
            hipMemcpyDeviceToDevice,
            csm));
      }
    }
    if (rel) {
      launchWriteFlagRel(
          mesh_->peerFlag(left, fACK_[j] + par), seqBaseDev_, k + 1, csm);
    } else {
      launchWriteFlag(mesh_->peerFlag(left, fACK_[j] + par), seqOf(k), csm);
    }
    evs[k % pool]->record(csm);
  }
  if (!rel) {
    for (int k = std::max(0, K - 2); k < K; k++) {
      lastAckPerRing_[j][k & 1] = seqOf(k);
    }
  }
  return K;
}

namespace {
bool graphsEnabled() {
  static int v = [] {
    const char* e = std::getenv("GLOO_AMD_GRAPH");
    return (e != nullptr && e[0] == '0') ? 0 : 1;
  }();
  return v == 1;
}
} // namespace

void HipAllreduceRing::runDevice(
    char* buf,
    size_t bytes,
    size_t n,
    DType dt,
    ReduceOp op) {
  const size_t es = bytes / n;
  // In-place on the user buffer: the ring schedule only ever writes peer
  // INBOXES (never a peer's work region), so no staging copy is needed —
  // r01 staged through mesh work both ways, 2*bytes of extra HBM traffic
  // per allreduce.
  mesh_->ensureCapacity(0, inboxCap_);
  const int R = static_cast<int>(strides_.size());

  // hipGraph replay: the many-segment multi-ring schedule is hundreds of
  // host enqueues per run (the host becomes the bottleneck at 8 ranks x
  // 4 rings); a captured graph replays it as one launch. Valid only in
  // steady state: the previous run must have the same shape, so the
  // first run of a shape goes eager (exact cross-run gates) and capture
  // happens from the second run on.
  GraphKey key{buf, n, static_cast<int>(dt), static_cast<int>(op)};
  if (graphsEnabled() && !graphBroken_ && lastKey_ == key) {
    if (runDeviceGraph(buf, n, es, dt, op)) {
      return;
    }
  }
  lastKey_ = key;

  for (int j = 0; j < R; j++) {
    // Carries the caller-stream dependency (gated in run()) into the
    // k<S steps of each ring's schedule.
    initEvent_[j]->record(cs_[j]->stream());
  }

  // Partition the buffer across the rings (16-B-aligned splits so the
  // reductions stay on the vectorized path) and enqueue each schedule.
  const size_t A = std::max<size_t>(1, 16 / es);
  ringSteps_ = 0;
  for (int j = 0; j < R; j++) {
    Seg part = sched::subspanOfA({0, n}, j, R, A);
    if (part.len > 0) {
      ringSteps_ = std::max<uint64_t>(
          ringSteps_,
          enqueueRing(j, buf, part.off, part.len, es, dt, op));
    }
  }

  for (int j = 1; j < R; j++) {
    doneEvent_[j]->record(cs_[j]->stream());
    doneEvent_[j]->streamWait(cs_[0]->stream());
  }
  doneEvent_[0]->record(cs_[0]->stream());

  auto timeout = ctx_->getTimeout();
  watchdogWait(*doneEvent_[0], *mesh_, timeout, "hip_allreduce_ring (cs)");
  for (int j = 0; j < R; j++) {
    HipEvent drain(device_);
    drain.record(ks_[j]->stream());
    watchdogWait(drain, *mesh_, timeout, "hip_allreduce_ring (ks)");
    ks_[j]->synchronize();
    cs_[j]->synchronize();
  }
  seqBase_ += ringSteps_;
}

bool HipAllreduceRing::runDeviceGraph(
    char* buf,
    size_t n,
    size_t es,
    DType dt,
    ReduceOp op) {
  const int R = static_cast<int>(strides_.size());
  if (seqBaseDev_ == nullptr) {
    if (hipMalloc(reinterpret_cast<void**>(&seqBaseDev_), 8) !=
        hipSuccess) {
      graphBroken_ = true;
      return false;
    }
  }
  GraphKey key{buf, n, static_cast<int>(dt), static_cast<int>(op)};
  auto it = graphs_.find(key);
  if (it == graphs_.end()) {
    if (graphs_.size() >= 16) {
      // Shape churn: run new shapes eagerly, keep replaying cached ones.
      return false;
    }
    // Per-ring step counts (needed for the k<2 cross-run gate when a
    // ring has fewer steps than the longest one).
    GraphEntry entry;
    entry.ringK.assign(R, 0);
    int maxK = 0;
    // Capture the whole multi-ring schedule rooted at cs_[0].
    if (hipStreamBeginCapture(
            cs_[0]->stream(), hipStreamCaptureModeThreadLocal) !=
        hipSuccess) {
      (void)hipGetLastError();
      graphBroken_ = true;
      return false;
    }
    bool ok = true;
    try {
      graphFork_->record(cs_[0]->stream());
      for (int j = 0; j < R; j++) {
        if (j > 0) {
          graphFork_->streamWait(cs_[j]->stream());
        }
        graphFork_->streamWait(ks_[j]->stream());
        initEvent_[j]->record(cs_[j]->stream());
      }
      // Dry pass for per-ring K (pure shape math, no enqueue): K_j =
      // 2 * (P-1) * S_j with S_j from the ring's partition length.
      const size_t A = std::max<size_t>(1, 16 / es);
      std::vector<Seg> parts(R);
      for (int j = 0; j < R; j++) {
        parts[j] = sched::subspanOfA({0, n}, j, R, A);
      }
      for (int j = 0; j < R; j++) {
        if (parts[j].len == 0) {
          continue;
        }
        entry.ringK[j] =
            2 * (ctx_->size - 1) * ringSegments(parts[j].len, es);
        maxK = std::max(maxK, entry.ringK[j]);
      }
      for (int j = 0; j < R; j++) {
        if (parts[j].len > 0) {
          const int K = enqueueRing(
              j, buf, parts[j].off, parts[j].len, es, dt, op,
              /*rel=*/true,
              /*crossRunOff=*/static_cast<int64_t>(entry.ringK[j]) - maxK);
          GA_ENFORCE_EQ(K, entry.ringK[j], "graph shape math diverged");
        }
      }
      // Join every stream back into cs_[0] (initEvent_[j] doubles as
      // the ks join marker; re-recording inside the capture is fine).
      for (int j = 0; j < R; j++) {
        if (j > 0) {
          doneEvent_[j]->record(cs_[j]->stream());
          doneEvent_[j]->streamWait(cs_[0]->stream());
        }
        initEvent_[j]->record(ks_[j]->stream());
        initEvent_[j]->streamWait(cs_[0]->stream());
      }
    } catch (...) {
      ok = false;
    }
    entry.steps = maxK;
    hipGraph_t g = nullptr;
    hipError_t ec = hipStreamEndCapture(cs_[0]->stream(), &g);
    if (!ok || ec != hipSuccess || g == nullptr) {
      (void)hipGetLastError();
      if (g != nullptr) {
        (void)hipGraphDestroy(g);
      }
      GA_WARN << "hipGraph capture of the ring schedule failed; "
                 "falling back to eager enqueue";
      graphBroken_ = true;
      return false;
    }
    hipGraphExec_t exec = nullptr;
    hipError_t ei = hipGraphInstantiate(&exec, g, nullptr, nullptr, 0);
    (void)hipGraphDestroy(g);
    if (ei != hipSuccess || exec == nullptr) {
      (void)hipGetLastError();
      GA_WARN << "hipGraph instantiate failed; eager fallback";
      graphBroken_ = true;
      return false;
    }
    entry.exec = exec;
    it = graphs_.emplace(key, entry).first;
    GA_INFO << "captured ring schedule graph: " << R << " rings, "
            << maxK << " steps";
  }

  // Replay: bump the device seq base, launch, watchdog.
  const GraphEntry& entry = it->second;
  GA_HIP_CHECK(hipMemcpyAsync(
      seqBaseDev_, &seqBase_, 8, hipMemcpyHostToDevice, cs_[0]->stream()));
  GA_HIP_CHECK(hipGraphLaunch(entry.exec, cs_[0]->stream()));
  doneEvent_[0]->record(cs_[0]->stream());
  watchdogWait(
      *doneEvent_[0], *mesh_, ctx_->getTimeout(),
      "hip_allreduce_ring (graph)");
  cs_[0]->synchronize();
  // Keep the eager bookkeeping exact so eager and graph runs interleave.
  for (int j = 0; j < R; j++) {
    for (int k = std::max(0, entry.ringK[j] - 2); k < entry.ringK[j]; k++) {
      lastAckPerRing_[j][k & 1] = seqBase_ + k + 1;
    }
  }
  seqBase_ += entry.steps;
  return true;
}

// ===========================================================================
// HipAllreduceHalvingDoubling
// ===========================================================================

HipAllreduceHalvingDoubling::HipAllreduceHalvingDoubling(
    std::shared_ptr<Context> ctx,
    int device,
    size_t inboxCap)
    : ctx_(std::move(ctx)),
      device_(device),
      inboxCap_(inboxCap == 0 ? kDefaultInboxCap : inboxCap) {
  const int P = ctx_->size;
  GA_ENFORCE_GT(P, 0);
  // Non-power-of-2 sizes fold the extra ranks into partners before and
  // after a power-of-2 exchange (the CPU legacy HD's scheme,
  // csrc/algorithms/allreduce_halving_doubling.h; reference instead uses
  // binary blocks, gloo/allreduce_halving_doubling.h:38-64).
  pow2_ = 1;
  log2P_ = 0;
  while (pow2_ * 2 <= P) {
    pow2_ *= 2;
    log2P_++;
  }
  GA_HIP_CHECK(hipSetDevice(device_));
  mesh_ = std::make_unique<XgmiMesh>(ctx_, device_, 0, inboxCap_);
  cs_ = pooledStream(ctx_.get(), device_, 0);
  ks_ = pooledStream(ctx_.get(), device_, 1);
  stepEvent_ = std::make_unique<HipEvent>(device_);
  initEvent_ = std::make_unique<HipEvent>(device_);
  doneEvent_ = std::make_unique<HipEvent>(device_);
  gateEv_ = std::make_unique<HipEvent>(device_);
  const int T = std::max(1, log2P_);
  fDATA_ = mesh_->allocFlags(T * 2);
  fACK_ = mesh_->allocFlags(T * 2);
  fAGD_ = mesh_->allocFlags(T);
  if (P != pow2_) {
    fFOLD_ = mesh_->allocFlags(2); // pre-fold chunk data, x2 parity
    fFACK_ = mesh_->allocFlags(2); // pre-fold chunk consumed
    fPOST_ = mesh_->allocFlags(1); // post-fold result landed
    fPACK_ = mesh_->allocFlags(1); // extra's copy-out done
  }
}

void HipAllreduceHalvingDoubling::run(
    void* devPtr,
    size_t elements,
    DType dtype,
    ReduceOp op,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_allreduce_halving_doubling");
  GA_HIP_CHECK(hipSetDevice(device_));
  const int P = ctx_->size;
  const int r = ctx_->rank;
  const size_t es = dtypeSize(dtype);
  const size_t bytes = elements * es;
  if (P == 1 || elements == 0) {
    return;
  }
  GA_PHASE("hd:enter");
  gateStreams(*gateEv_, callerStream, {cs_->stream(), ks_->stream()});
  GA_PHASE("hd:gated");
  char* buf = static_cast<char*>(devPtr);
  const int T = std::max(1, log2P_);
  const int extras = P - pow2_;
  const bool isExtra = r >= pow2_;
  mesh_->ensureCapacity(bytes, inboxCap_);
  GA_PHASE("hd:capacity-ok");
  // Sub-inbox layout: the 2*inboxCap inbox region split into rows x 2
  // parities; rows = T steps plus (when non-pow2) one fold row.
  const int rows = T + (extras > 0 ? 1 : 0);
  const size_t A = std::max<size_t>(1, 16 / es); // 16-B split alignment
  const size_t subBytes =
      sched::alignDown((2 * mesh_->inboxCap()) / (2 * rows), 16);
  GA_ENFORCE_GE(subBytes, size_t(16), "inbox too small for the step count");
  auto subInbox = [&](int t, int par) {
    return mesh_->inbox(0) + (static_cast<size_t>(t) * 2 + par) * subBytes;
  };
  auto peerSubInbox = [&](int rank, int t, int par) {
    return mesh_->peerInbox(rank, 0) +
        (static_cast<size_t>(t) * 2 + par) * subBytes;
  };

  char* work = mesh_->work();
  auto csm = cs_->stream();
  auto ksm = ks_->stream();

  // Fold-chunk geometry (element-aligned), identical on both sides.
  const size_t foldChunkElems = subBytes / es;
  const int foldNc = static_cast<int>(
      (elements + foldChunkElems - 1) / foldChunkElems);
  auto foldChunk = [&](int c) {
    const size_t off = static_cast<size_t>(c) * foldChunkElems;
    return Seg{off, std::min(foldChunkElems, elements - off)};
  };

  if (isExtra) {
    // ---- extra rank: fold into the partner, then receive the result.
    const int partner = r - pow2_;
    for (int c = 0; c < foldNc; c++) {
      const int par = c & 1;
      foldSeq_++;
      if (lastFold_[par] > 0) {
        launchWaitFlagGte(mesh_->flag(fFACK_ + par), lastFold_[par], ksm);
      }
      Seg ch = foldChunk(c);
      GA_HIP_CHECK(hipMemcpyAsync(
          peerSubInbox(partner, T, par),
          buf + ch.off * es,
          ch.len * es,
          hipMemcpyDeviceToDevice,
          ksm));
      launchWriteFlag(mesh_->peerFlag(partner, fFOLD_ + par), foldSeq_, ksm);
      lastFold_[par] = foldSeq_;
    }
    // Post-fold: partner writes the final result into our work region.
    postSeq_++;
    launchWaitFlagGte(mesh_->flag(fPOST_), postSeq_, csm);
    GA_HIP_CHECK(
        hipMemcpyAsync(buf, work, bytes, hipMemcpyDeviceToDevice, csm));
    launchWriteFlag(mesh_->peerFlag(partner, fPACK_), postSeq_, csm);
    doneEvent_->record(csm);
    auto timeout = ctx_->getTimeout();
    watchdogWait(*doneEvent_, *mesh_, timeout, "hip_allreduce_hd (fold cs)");
    initEvent_->record(ksm);
    watchdogWait(*initEvent_, *mesh_, timeout, "hip_allreduce_hd (fold ks)");
    cs_->synchronize();
    ks_->synchronize();
    return;
  }

  GA_HIP_CHECK(
      hipMemcpyAsync(work, buf, bytes, hipMemcpyDeviceToDevice, csm));

  // ---- pre-fold: reduce the extra rank's chunks into work (cs) ----
  if (r < extras) {
    for (int c = 0; c < foldNc; c++) {
      const int par = c & 1;
      foldSeq_++;
      launchWaitFlagGte(mesh_->flag(fFOLD_ + par), foldSeq_, csm);
      Seg ch = foldChunk(c);
      launchReduce2(
          work + ch.off * es,
          work + ch.off * es,
          subInbox(T, par),
          ch.len,
          dtype,
          op,
          csm);
      launchWriteFlag(
          mesh_->peerFlag(r + pow2_, fFACK_ + par), foldSeq_, csm);
    }
  }
  initEvent_->record(csm);

  // Host-tracked per-(step,parity) last-issued ack/data seqs; identical
  // on every rank (symmetric schedule), persistent across runs.
  if (lastAck_.empty()) {
    lastAck_.assign(T * 2, 0);
    lastAgd_.assign(T, 0);
  }

  Seg span{0, elements};
  std::vector<Seg> spanAt(T);
  std::vector<std::unique_ptr<HipEvent>>& evs = stepEvents_;
  while (static_cast<int>(evs.size()) < T) {
    evs.push_back(std::make_unique<HipEvent>(device_));
  }

  // ---- reduce-scatter (vector halving, distance doubling) ----
  for (int t = 0; t < T; t++) {
    spanAt[t] = span;
    const int peer = r ^ (1 << t);
    const int bit = (r >> t) & 1;
    Seg kp = sched::subspanOfA(span, bit, 2, A);
    Seg gv = sched::subspanOfA(span, 1 - bit, 2, A);
    const size_t maxHalf = std::max(kp.len, gv.len);
    int nc = std::max<int>(
        1, static_cast<int>((maxHalf * es + subBytes - 1) / subBytes));
    while (sched::alignUp((maxHalf + nc - 1) / nc, A) * es > subBytes) {
      nc++;
    }

    // gv must be fully produced (previous step's reduces on cs).
    if (t == 0) {
      initEvent_->streamWait(ksm);
    } else {
      evs[t - 1]->streamWait(ksm);
    }
    for (int c = 0; c < nc; c++) {
      const int par = c & 1;
      seq_++;
      const uint64_t dseq = seq_;
      if (lastAck_[t * 2 + par] > 0) {
        launchWaitFlagGte(
            mesh_->flag(fACK_ + t * 2 + par), lastAck_[t * 2 + par], ksm);
      }
      Seg gch = sched::chunkOfA(gv, c, nc, A);
      if (gch.len > 0) {
        GA_HIP_CHECK(hipMemcpyAsync(
            peerSubInbox(peer, t, par),
            work + gch.off * es,
            gch.len * es,
            hipMemcpyDeviceToDevice,
            ksm));
      }
      launchWriteFlag(mesh_->peerFlag(peer, fDATA_ + t * 2 + par), dseq, ksm);

      launchWaitFlagGte(mesh_->flag(fDATA_ + t * 2 + par), dseq, csm);
      Seg kch = sched::chunkOfA(kp, c, nc, A);
      if (kch.len > 0) {
        launchReduce2(
            work + kch.off * es,
            work + kch.off * es,
            subInbox(t, par),
            kch.len,
            dtype,
            op,
            csm);
      }
      launchWriteFlag(mesh_->peerFlag(peer, fACK_ + t * 2 + par), dseq, csm);
      lastAck_[t * 2 + par] = dseq;
    }
    evs[t]->record(csm);
    span = kp;
  }

  // ---- allgather (mirror; direct writes into peer's work) ----
  Seg cur = span;
  for (int t = T - 1; t >= 0; t--) {
    const int peer = r ^ (1 << t);
    if (t == T - 1) {
      evs[T - 1]->streamWait(ksm);
    } else {
      // Halves received in step t+1 are part of cur now.
      if (lastAgd_[t + 1] > 0) {
        launchWaitFlagGte(mesh_->flag(fAGD_ + t + 1), lastAgd_[t + 1], ksm);
      }
    }
    seq_++;
    const uint64_t aseq = seq_;
    if (cur.len > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          mesh_->peerWork(peer) + cur.off * es,
          work + cur.off * es,
          cur.len * es,
          hipMemcpyDeviceToDevice,
          ksm));
    }
    launchWriteFlag(mesh_->peerFlag(peer, fAGD_ + t), aseq, ksm);
    lastAgd_[t] = aseq;
    cur = spanAt[t];
  }

  // Final: wait all incoming allgather halves, stage out.
  for (int t = 0; t < T; t++) {
    launchWaitFlagGte(mesh_->flag(fAGD_ + t), lastAgd_[t], csm);
  }
  if (r < extras) {
    // Post-fold: push the final result into the extra rank's work
    // region (single writer). Run n+1's write is gated on the extra's
    // copy-out ack from run n so we never clobber an in-flight read.
    postSeq_++;
    const int extra = r + pow2_;
    if (postSeq_ > 1) {
      launchWaitFlagGte(mesh_->flag(fPACK_), postSeq_ - 1, csm);
    }
    GA_HIP_CHECK(hipMemcpyAsync(
        mesh_->peerWork(extra), work, bytes, hipMemcpyDeviceToDevice, csm));
    launchWriteFlag(mesh_->peerFlag(extra, fPOST_), postSeq_, csm);
  }
  GA_HIP_CHECK(
      hipMemcpyAsync(buf, work, bytes, hipMemcpyDeviceToDevice, csm));
  doneEvent_->record(csm);

  GA_PHASE("hd:enqueued");
  auto timeout = ctx_->getTimeout();
  watchdogWait(*doneEvent_, *mesh_, timeout, "hip_allreduce_hd (cs)");
  GA_PHASE("hd:cs-watchdog-ok");
  initEvent_->record(ksm);
  watchdogWait(*initEvent_, *mesh_, timeout, "hip_allreduce_hd (ks)");
  GA_PHASE("hd:ks-watchdog-ok");
  cs_->synchronize();
  ks_->synchronize();
  GA_PHASE("hd:done");
}


// Drain this engine's streams before members (mesh: flags + data pages)
// destruct: a failed, poisoned run can leave released-but-still-
// executing work on the POOLED streams, which outlive the engine.
HipAllreduceHalvingDoubling::~HipAllreduceHalvingDoubling() {
  drainBounded(cs_);
  drainBounded(ks_);
}

// ===========================================================================
// HipBroadcastOneToAll
// ===========================================================================

HipBroadcastOneToAll::HipBroadcastOneToAll(
    std::shared_ptr<Context> ctx,
    int device,
    int root,
    int numStreams)
    : ctx_(std::move(ctx)), device_(device), root_(root) {
  GA_HIP_CHECK(hipSetDevice(device_));
  mesh_ = std::make_unique<XgmiMesh>(ctx_, device_, 0, 4096);
  cs_ = pooledStream(ctx_.get(), device_, 0);
  gateEv_ = std::make_unique<HipEvent>(device_);
  const int nf = std::min(
      {numStreams, std::max(1, ctx_->size - 1), kStreamPoolSize - 1});
  for (int i = 0; i < nf; i++) {
    fanout_.push_back(pooledStream(ctx_.get(), device_, 1 + i));
  }
  fBDATA_ = mesh_->allocFlags(1);
  fBACK_ = mesh_->allocFlags(ctx_->size);
}

void HipBroadcastOneToAll::run(
    void* devPtr,
    size_t bytes,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_broadcast_one_to_all");
  GA_HIP_CHECK(hipSetDevice(device_));
  const int P = ctx_->size;
  const int r = ctx_->rank;
  if (P == 1 || bytes == 0) {
    return;
  }
  mesh_->ensureCapacity(bytes, 4096);
  gateStreams(*gateEv_, callerStream, {cs_->stream()});
  char* work = mesh_->work();
  char* buf = static_cast<char*>(devPtr);
  seq_++;

  auto timeout = ctx_->getTimeout();
  if (r == root_) {
    GA_HIP_CHECK(hipMemcpyAsync(
        work, buf, bytes, hipMemcpyDeviceToDevice, cs_->stream()));
    HipEvent staged(device_);
    staged.record(cs_->stream());
    int si = 0;
    for (int i = 0; i < P; i++) {
      if (i == root_) {
        continue;
      }
      auto& st = *fanout_[si++ % fanout_.size()];
      staged.streamWait(st.stream());
      if (seq_ > 1) {
        launchWaitFlagGte(mesh_->flag(fBACK_ + i), seq_ - 1, st.stream());
      }
      GA_HIP_CHECK(hipMemcpyAsync(
          mesh_->peerWork(i),
          work,
          bytes,
          hipMemcpyDeviceToDevice,
          st.stream()));
      launchWriteFlag(mesh_->peerFlag(i, fBDATA_), seq_, st.stream());
    }
    cs_->synchronize();
    for (auto& st : fanout_) {
      HipEvent done(device_);
      done.record(st->stream());
      watchdogWait(done, *mesh_, timeout, "hip_broadcast (fanout)");
      st->synchronize();
    }
  } else {
    launchWaitFlagGte(mesh_->flag(fBDATA_), seq_, cs_->stream());
    GA_HIP_CHECK(hipMemcpyAsync(
        buf, work, bytes, hipMemcpyDeviceToDevice, cs_->stream()));
    launchWriteFlag(mesh_->peerFlag(root_, fBACK_ + r), seq_, cs_->stream());
    HipEvent done(device_);
    done.record(cs_->stream());
    watchdogWait(done, *mesh_, timeout, "hip_broadcast (recv)");
    cs_->synchronize();
  }
}

HipBroadcastOneToAll::~HipBroadcastOneToAll() {
  drainBounded(cs_);
  for (auto* st : fanout_) {
    drainBounded(st);
  }
}

// ===========================================================================
// HipAllreduceDirect
// ===========================================================================

HipAllreduceDirect::HipAllreduceDirect(
    std::shared_ptr<Context> ctx,
    int device,
    int numStreams)
    : ctx_(std::move(ctx)), device_(device) {
  GA_ENFORCE_LE(ctx_->size, 8, "direct allreduce supports <= 8 ranks");
  GA_HIP_CHECK(hipSetDevice(device_));
  mesh_ = std::make_unique<XgmiMesh>(ctx_, device_, 0, kDefaultInboxCap);
  cs_ = pooledStream(ctx_.get(), device_, 0);
  const int nf = std::min(
      {numStreams, std::max(1, ctx_->size - 1), kStreamPoolSize - 1});
  for (int i = 0; i < nf; i++) {
    fanout_.push_back(pooledStream(ctx_.get(), device_, 1 + i));
  }
  doneEvent_ = std::make_unique<HipEvent>(device_);
  gateEv_ = std::make_unique<HipEvent>(device_);
  for (int i = 0; i < 4; i++) {
    chunkEvents_.push_back(std::make_unique<HipEvent>(device_));
  }
  fRS_ = mesh_->allocFlags(ctx_->size);
  fAG_ = mesh_->allocFlags(ctx_->size);
  fACK_ = mesh_->allocFlags(ctx_->size);
  fDONE_ = mesh_->allocFlags(ctx_->size);
  lastAck_.assign(ctx_->size, {0, 0});
}

void HipAllreduceDirect::run(
    void* devPtr,
    size_t elements,
    DType dtype,
    ReduceOp op,
    hipStream_t callerStream) {
  std::vector<void*> ptrs{devPtr};
  run(ptrs, elements, dtype, op, callerStream);
}

void HipAllreduceDirect::run(
    const std::vector<void*>& ptrs,
    size_t elements,
    DType dtype,
    ReduceOp op,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_allreduce_direct");
  GA_HIP_CHECK(hipSetDevice(device_));
  GA_ENFORCE(!ptrs.empty());
  const int P = ctx_->size;
  const int r = ctx_->rank;
  const size_t es = dtypeSize(dtype);
  if (elements == 0) {
    return;
  }
  if (P == 1 || ptrs.size() > 1) {
    // Fused local reduction of the caller's pointers into ptrs[0] before
    // the wire phase (reference cuda_allreduce_ring.cc:72-120 role).
    if (ptrs.size() > 1) {
      gateStreams(*gateEv_, callerStream, {cs_->stream()});
      const int k = static_cast<int>(std::min<size_t>(ptrs.size(), 8));
      launchReduceN(
          ptrs[0], const_cast<const void* const*>(ptrs.data()), k, elements,
          dtype, op, cs_->stream());
      for (size_t i = 8; i < ptrs.size(); i++) {
        launchReduce2(ptrs[0], ptrs[0], ptrs[i], elements, dtype, op,
                      cs_->stream());
      }
      callerStream = cs_->stream(); // wire phase ordered after the fuse
    }
    if (P == 1) {
      for (size_t i = 1; i < ptrs.size(); i++) {
        GA_HIP_CHECK(hipMemcpyAsync(
            ptrs[i], ptrs[0], elements * es, hipMemcpyDeviceToDevice,
            cs_->stream()));
      }
      cs_->synchronize();
      return;
    }
  }
  char* user = static_cast<char*>(ptrs[0]);
  // 16-B-aligned block splits keep the fused reduction vectorized.
  const size_t perRank = sched::alignUp(
      (elements + P - 1) / P, std::max<size_t>(1, 16 / es));
  // Chunk the per-rank block so scatter / reduce / broadcast pipeline
  // for large payloads. Inbox layout: P sources x 2 parities x chunkCap.
  const size_t chunkCap = std::min<size_t>(
      std::max<size_t>(perRank * es, 1), size_t(8) << 20);
  const int C =
      std::max<int>(1, static_cast<int>((perRank * es + chunkCap - 1) /
                                        chunkCap));
  // The (src, parity) slot grid needs 2*P*chunkCap bytes. The mesh lays
  // its two inboxes out contiguously (mesh.h: work | inbox0 | inbox1),
  // so asking for P*chunkCap PER inbox and indexing the grid from
  // inbox(0) uses both regions as one arena. This mesh is private to
  // this algorithm instance, so inbox(1) has no other user.
  mesh_->ensureCapacity(elements * es, P * chunkCap);
  char* work = mesh_->work();
  const size_t slotStride = chunkCap; // one chunk per (src, parity) slot
  auto subInbox = [&](char* base, int src, int par) {
    return base +
        (static_cast<size_t>(src) * 2 + par) * slotStride;
  };
  auto blockOff = [&](int b) {
    return std::min(static_cast<size_t>(b) * perRank, elements) * es;
  };
  auto blockLen = [&](int b) {
    return std::min(static_cast<size_t>(b + 1) * perRank, elements) * es -
        blockOff(b);
  };
  auto chunkOffIn = [&](int b, int c) {
    return std::min<size_t>(static_cast<size_t>(c) * chunkCap, blockLen(b));
  };
  auto chunkLen = [&](int b, int c) {
    return std::min<size_t>(
               static_cast<size_t>(c + 1) * chunkCap, blockLen(b)) -
        chunkOffIn(b, c);
  };

  if (gateBySync()) {
    GA_HIP_CHECK(hipStreamSynchronize(callerStream));
  } else {
    gateEv_->record(callerStream);
    gateEv_->streamWait(cs_->stream());
    for (auto* st : fanout_) {
      gateEv_->streamWait(st->stream());
    }
  }

  const uint64_t runSeq = ++runSeq_;
  auto cseq = [&](int c) { return chunkSeqBase_ + c + 1; };

  // --- chunk-major pipeline: for each chunk, enqueue scatter (fanout),
  // fused reduce (cs), broadcast (fanout). Flags gate cross-rank deps,
  // so chunk c+1's wire traffic overlaps chunk c's reduction. ---
  for (int j = 1; j < P; j++) {
    const int d = (r + j) % P;
    auto& st = *fanout_[(j - 1) % fanout_.size()];
    if (runSeq > 1) {
      // d finished its previous copy-out before we touch its memory.
      launchWaitFlagGte(mesh_->flag(fDONE_ + d), runSeq - 1, st.stream());
    }
  }
  for (int c = 0; c < C; c++) {
    const int par = c & 1;
    // scatter chunk c of every destination block
    for (int j = 1; j < P; j++) {
      const int d = (r + j) % P;
      auto& st = *fanout_[(j - 1) % fanout_.size()];
      const uint64_t gate = (c >= 2) ? cseq(c - 2) : lastAck_[d][par];
      if (gate > 0) {
        launchWaitFlagGte(mesh_->flag(fACK_ + d), gate, st.stream());
      }
      const size_t len = chunkLen(d, c);
      if (len > 0) {
        GA_HIP_CHECK(hipMemcpyAsync(
            subInbox(mesh_->peerInbox(d, 0), r, par),
            user + blockOff(d) + chunkOffIn(d, c),
            len,
            hipMemcpyDeviceToDevice,
            st.stream()));
      }
      launchWriteFlag(mesh_->peerFlag(d, fRS_ + r), cseq(c), st.stream());
    }
    // fused reduction of chunk c of my block
    for (int src = 0; src < P; src++) {
      if (src != r) {
        launchWaitFlagGte(mesh_->flag(fRS_ + src), cseq(c), cs_->stream());
      }
    }
    const size_t rlen = chunkLen(r, c);
    if (rlen > 0) {
      const void* srcs[8];
      srcs[0] = user + blockOff(r) + chunkOffIn(r, c);
      int k = 1;
      for (int src = 0; src < P; src++) {
        if (src != r) {
          srcs[k++] = subInbox(mesh_->inbox(0), src, par);
        }
      }
      launchReduceN(
          work + blockOff(r) + chunkOffIn(r, c), srcs, P, rlen / es, dtype,
          op, cs_->stream());
    }
    for (int src = 0; src < P; src++) {
      if (src != r) {
        launchWriteFlag(mesh_->peerFlag(src, fACK_ + r), cseq(c),
                        cs_->stream());
      }
    }
    chunkEvents_[c % 4]->record(cs_->stream());
    // broadcast reduced chunk c over all links
    for (int j = 1; j < P; j++) {
      const int d = (r + j) % P;
      auto& st = *fanout_[(j - 1) % fanout_.size()];
      chunkEvents_[c % 4]->streamWait(st.stream());
      if (rlen > 0) {
        GA_HIP_CHECK(hipMemcpyAsync(
            mesh_->peerWork(d) + blockOff(r) + chunkOffIn(r, c),
            work + blockOff(r) + chunkOffIn(r, c),
            rlen,
            hipMemcpyDeviceToDevice,
            st.stream()));
      }
      launchWriteFlag(mesh_->peerFlag(d, fAG_ + r), cseq(c), st.stream());
    }
  }

  // --- collect, stage out, publish copy-out-done (cs) ---
  for (int src = 0; src < P; src++) {
    if (src != r) {
      launchWaitFlagGte(mesh_->flag(fAG_ + src), cseq(C - 1),
                        cs_->stream());
    }
  }
  GA_HIP_CHECK(hipMemcpyAsync(
      user, work, elements * es, hipMemcpyDeviceToDevice, cs_->stream()));
  for (int src = 0; src < P; src++) {
    if (src != r) {
      launchWriteFlag(mesh_->peerFlag(src, fDONE_ + r), runSeq,
                      cs_->stream());
    }
  }
  doneEvent_->record(cs_->stream());
  auto timeout = ctx_->getTimeout();
  watchdogWait(*doneEvent_, *mesh_, timeout, "hip_allreduce_direct (cs)");
  cs_->synchronize();
  for (auto* st : fanout_) {
    HipEvent done(device_);
    done.record(st->stream());
    watchdogWait(done, *mesh_, timeout, "hip_allreduce_direct (fanout)");
    st->synchronize();
  }
  for (int d = 0; d < P; d++) {
    for (int c = std::max(0, C - 2); c < C; c++) {
      lastAck_[d][c & 1] = cseq(c);
    }
  }
  chunkSeqBase_ += C;
  if (ptrs.size() > 1) {
    for (size_t i = 1; i < ptrs.size(); i++) {
      GA_HIP_CHECK(hipMemcpyAsync(
          ptrs[i], user, elements * es, hipMemcpyDeviceToDevice,
          cs_->stream()));
    }
    cs_->synchronize();
  }
}

HipAllreduceDirect::~HipAllreduceDirect() {
  drainBounded(cs_);
  for (auto* st : fanout_) {
    drainBounded(st);
  }
}

// ===========================================================================
// HipAllreduceBcube
// ===========================================================================

HipAllreduceBcube::HipAllreduceBcube(
    std::shared_ptr<Context> ctx,
    int device,
    int base) {
  const int P = ctx->size;
  const int b = base > 0 ? base : std::max(2, ctx->base);
  if (P <= 8 && (b >= P || P > 2)) {
    direct_ = std::make_unique<HipAllreduceDirect>(ctx, device);
  } else if (b == 2 && (P & (P - 1)) == 0) {
    hd_ = std::make_unique<HipAllreduceHalvingDoubling>(ctx, device);
  } else {
    ring_ = std::make_unique<HipAllreduceRing>(ctx, device);
  }
}

void HipAllreduceBcube::run(
    void* devPtr,
    size_t elements,
    DType dtype,
    ReduceOp op,
    hipStream_t callerStream) {
  if (direct_) {
    direct_->run(devPtr, elements, dtype, op, callerStream);
  } else if (hd_) {
    hd_->run(devPtr, elements, dtype, op, callerStream);
  } else {
    ring_->run(devPtr, elements, dtype, op, callerStream);
  }
}

// ===========================================================================
// HipAllgatherRing
// ===========================================================================

HipAllgatherRing::HipAllgatherRing(
    std::shared_ptr<Context> ctx,
    int device,
    size_t inboxCap)
    : ctx_(std::move(ctx)),
      device_(device),
      inboxCap_(inboxCap == 0 ? kDefaultInboxCap : inboxCap) {
  GA_HIP_CHECK(hipSetDevice(device_));
  mesh_ = std::make_unique<XgmiMesh>(ctx_, device_, 0, inboxCap_);
  cs_ = pooledStream(ctx_.get(), device_, 0);
  ks_ = pooledStream(ctx_.get(), device_, 1);
  initEvent_ = std::make_unique<HipEvent>(device_);
  doneEvent_ = std::make_unique<HipEvent>(device_);
  gateEv_ = std::make_unique<HipEvent>(device_);
  fDATA_ = mesh_->allocFlags(2);
  fACK_ = mesh_->allocFlags(2);
}

void HipAllgatherRing::run(
    const void* devIn,
    void* devOut,
    size_t inElements,
    size_t es,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_allgather_ring");
  GA_HIP_CHECK(hipSetDevice(device_));
  const int P = ctx_->size;
  const int r = ctx_->rank;
  const size_t blockBytes = inElements * es;
  if (P == 1) {
    if (devOut != devIn) {
      GA_HIP_CHECK(hipMemcpy(devOut, devIn, blockBytes,
                             hipMemcpyDeviceToDevice));
    }
    return;
  }
  const size_t totalBytes = blockBytes * P;
  const size_t A = std::max<size_t>(1, 16 / es);
  const int S = std::max<int>(
      1, static_cast<int>((blockBytes + inboxCap_ - 1) / inboxCap_));
  const size_t segCapBytes =
      sched::alignUp((inElements + S - 1) / S, A) * es;
  mesh_->ensureCapacity(totalBytes, segCapBytes);
  gateStreams(*gateEv_, callerStream, {cs_->stream(), ks_->stream()});

  const int right = (r + 1) % P;
  const int left = (r - 1 + P) % P;
  const int pool = S + 2;
  while (static_cast<int>(events_.size()) < pool) {
    events_.push_back(std::make_unique<HipEvent>(device_));
  }
  char* work = mesh_->work();
  auto csm = cs_->stream();
  auto ksm = ks_->stream();

  GA_HIP_CHECK(hipMemcpyAsync(
      work + r * blockBytes, devIn, blockBytes, hipMemcpyDeviceToDevice,
      csm));
  initEvent_->record(csm);

  auto seqOf = [&](int k) { return seqBase_ + k + 1; };
  const int K = (P - 1) * S;
  for (int k = 0; k < K; k++) {
    const int i = k / S, s = k % S;
    const int par = k & 1;
    const sched::Seg send =
        sched::segmentOfA(inElements, 1, 0, s, S, A); // within-block segment
    const size_t sendBlock = (r - i + 2 * P) % P;
    const size_t recvBlock = (r - i - 1 + 2 * P) % P;

    const uint64_t prevAck = (k >= 2) ? seqOf(k - 2) : lastAck_[par];
    if (prevAck > 0) {
      launchWaitFlagGte(mesh_->flag(fACK_ + par), prevAck, ksm);
    }
    if (k >= S) {
      events_[(k - S) % pool]->streamWait(ksm);
    } else {
      initEvent_->streamWait(ksm);
    }
    if (send.len > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          mesh_->peerInbox(right, par),
          work + sendBlock * blockBytes + send.off * es,
          send.len * es,
          hipMemcpyDeviceToDevice,
          ksm));
    }
    launchWriteFlag(mesh_->peerFlag(right, fDATA_ + par), seqOf(k), ksm);

    launchWaitFlagGte(mesh_->flag(fDATA_ + par), seqOf(k), csm);
    if (send.len > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          work + recvBlock * blockBytes + send.off * es,
          mesh_->inbox(par),
          send.len * es,
          hipMemcpyDeviceToDevice,
          csm));
    }
    launchWriteFlag(mesh_->peerFlag(left, fACK_ + par), seqOf(k), csm);
    events_[k % pool]->record(csm);
  }

  GA_HIP_CHECK(hipMemcpyAsync(
      devOut, work, totalBytes, hipMemcpyDeviceToDevice, csm));
  doneEvent_->record(csm);
  auto timeout = ctx_->getTimeout();
  watchdogWait(*doneEvent_, *mesh_, timeout, "hip_allgather_ring (cs)");
  initEvent_->record(ksm);
  watchdogWait(*initEvent_, *mesh_, timeout, "hip_allgather_ring (ks)");
  cs_->synchronize();
  ks_->synchronize();
  for (int k = std::max(0, K - 2); k < K; k++) {
    lastAck_[k & 1] = seqOf(k);
  }
  seqBase_ += K;
}

HipAllgatherRing::~HipAllgatherRing() {
  drainBounded(cs_);
  drainBounded(ks_);
}

// ===========================================================================
// HipReduceScatterRing
// ===========================================================================

HipReduceScatterRing::HipReduceScatterRing(
    std::shared_ptr<Context> ctx,
    int device,
    size_t inboxCap)
    : ctx_(std::move(ctx)),
      device_(device),
      inboxCap_(inboxCap == 0 ? kDefaultInboxCap : inboxCap) {
  GA_HIP_CHECK(hipSetDevice(device_));
  mesh_ = std::make_unique<XgmiMesh>(ctx_, device_, 0, inboxCap_);
  cs_ = pooledStream(ctx_.get(), device_, 0);
  ks_ = pooledStream(ctx_.get(), device_, 1);
  initEvent_ = std::make_unique<HipEvent>(device_);
  doneEvent_ = std::make_unique<HipEvent>(device_);
  gateEv_ = std::make_unique<HipEvent>(device_);
  fDATA_ = mesh_->allocFlags(2);
  fACK_ = mesh_->allocFlags(2);
}

void HipReduceScatterRing::run(
    const void* devIn,
    void* devOut,
    size_t recvElements,
    DType dtype,
    ReduceOp op,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_reduce_scatter_ring");
  GA_HIP_CHECK(hipSetDevice(device_));
  const int P = ctx_->size;
  const int r = ctx_->rank;
  const size_t es = dtypeSize(dtype);
  const size_t blockBytes = recvElements * es;
  if (P == 1) {
    if (devOut != devIn) {
      GA_HIP_CHECK(hipMemcpy(devOut, devIn, blockBytes,
                             hipMemcpyDeviceToDevice));
    }
    return;
  }
  const size_t totalBytes = blockBytes * P;
  const size_t A = std::max<size_t>(1, 16 / es);
  const int S = std::max<int>(
      1, static_cast<int>((blockBytes + inboxCap_ - 1) / inboxCap_));
  const size_t segCapBytes =
      sched::alignUp((recvElements + S - 1) / S, A) * es;
  mesh_->ensureCapacity(totalBytes, segCapBytes);
  gateStreams(*gateEv_, callerStream, {cs_->stream(), ks_->stream()});

  const int right = (r + 1) % P;
  const int left = (r - 1 + P) % P;
  const int pool = S + 2;
  while (static_cast<int>(events_.size()) < pool) {
    events_.push_back(std::make_unique<HipEvent>(device_));
  }
  char* work = mesh_->work();
  auto csm = cs_->stream();
  auto ksm = ks_->stream();

  GA_HIP_CHECK(hipMemcpyAsync(
      work, devIn, totalBytes, hipMemcpyDeviceToDevice, csm));
  initEvent_->record(csm);

  // Ring reduce-scatter ending with rank r owning block r:
  // iteration i: send block (r-i-1), reduce block (r-i-2).
  auto seqOf = [&](int k) { return seqBase_ + k + 1; };
  const int K = (P - 1) * S;
  for (int k = 0; k < K; k++) {
    const int i = k / S, s = k % S;
    const int par = k & 1;
    const sched::Seg seg = sched::segmentOfA(recvElements, 1, 0, s, S, A);
    const size_t sendBlock = (r - i - 1 + 2 * P) % P;
    const size_t recvBlock = (r - i - 2 + 2 * P) % P;

    const uint64_t prevAck = (k >= 2) ? seqOf(k - 2) : lastAck_[par];
    if (prevAck > 0) {
      launchWaitFlagGte(mesh_->flag(fACK_ + par), prevAck, ksm);
    }
    if (k >= S) {
      events_[(k - S) % pool]->streamWait(ksm);
    } else {
      initEvent_->streamWait(ksm);
    }
    if (seg.len > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          mesh_->peerInbox(right, par),
          work + sendBlock * blockBytes + seg.off * es,
          seg.len * es,
          hipMemcpyDeviceToDevice,
          ksm));
    }
    launchWriteFlag(mesh_->peerFlag(right, fDATA_ + par), seqOf(k), ksm);

    launchWaitFlagGte(mesh_->flag(fDATA_ + par), seqOf(k), csm);
    if (seg.len > 0) {
      launchReduce2(
          work + recvBlock * blockBytes + seg.off * es,
          work + recvBlock * blockBytes + seg.off * es,
          mesh_->inbox(par),
          seg.len,
          dtype,
          op,
          csm);
    }
    launchWriteFlag(mesh_->peerFlag(left, fACK_ + par), seqOf(k), csm);
    events_[k % pool]->record(csm);
  }

  GA_HIP_CHECK(hipMemcpyAsync(
      devOut, work + r * blockBytes, blockBytes, hipMemcpyDeviceToDevice,
      csm));
  doneEvent_->record(csm);
  auto timeout = ctx_->getTimeout();
  watchdogWait(*doneEvent_, *mesh_, timeout, "hip_reduce_scatter (cs)");
  initEvent_->record(ksm);
  watchdogWait(*initEvent_, *mesh_, timeout, "hip_reduce_scatter (ks)");
  cs_->synchronize();
  ks_->synchronize();
  for (int k = std::max(0, K - 2); k < K; k++) {
    lastAck_[k & 1] = seqOf(k);
  }
  seqBase_ += K;
}

HipReduceScatterRing::~HipReduceScatterRing() {
  drainBounded(cs_);
  drainBounded(ks_);
}

// ===========================================================================
// HipAlltoall
// ===========================================================================

HipAlltoall::HipAlltoall(
    std::shared_ptr<Context> ctx,
    int device,
    int numStreams)
    : ctx_(std::move(ctx)), device_(device) {
  GA_HIP_CHECK(hipSetDevice(device_));
  mesh_ = std::make_unique<XgmiMesh>(ctx_, device_, 0, 4096);
  cs_ = pooledStream(ctx_.get(), device_, 0);
  gateEv_ = std::make_unique<HipEvent>(device_);
  const int nf = std::min(
      {numStreams, std::max(1, ctx_->size - 1), kStreamPoolSize - 1});
  for (int i = 0; i < nf; i++) {
    fanout_.push_back(pooledStream(ctx_.get(), device_, 1 + i));
  }
  fDATA_ = mesh_->allocFlags(ctx_->size);
  fACK_ = mesh_->allocFlags(ctx_->size);
}

void HipAlltoall::run(
    const void* devIn,
    void* devOut,
    size_t perRankElements,
    size_t es,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_alltoall");
  GA_HIP_CHECK(hipSetDevice(device_));
  const int P = ctx_->size;
  const int r = ctx_->rank;
  const size_t blockBytes = perRankElements * es;
  const char* in = static_cast<const char*>(devIn);
  char* out = static_cast<char*>(devOut);
  if (P == 1) {
    if (devOut != devIn) {
      GA_HIP_CHECK(hipMemcpy(out, in, blockBytes, hipMemcpyDeviceToDevice));
    }
    return;
  }
  mesh_->ensureCapacity(blockBytes * P, 4096);
  if (gateBySync()) {
    GA_HIP_CHECK(hipStreamSynchronize(callerStream));
  } else {
    gateEv_->record(callerStream);
    gateEv_->streamWait(cs_->stream());
    for (auto& st : fanout_) {
      gateEv_->streamWait(st->stream());
    }
  }
  char* work = mesh_->work();
  const uint64_t seq = ++seq_;

  // Push my block for each destination straight into its work region.
  int si = 0;
  for (int d = 0; d < P; d++) {
    if (d == r) {
      continue;
    }
    auto& st = *fanout_[si++ % fanout_.size()];
    if (seq > 1) {
      // d must have copied out the previous run's block from its work.
      launchWaitFlagGte(mesh_->flag(fACK_ + d), seq - 1, st.stream());
    }
    if (blockBytes > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          mesh_->peerWork(d) + static_cast<size_t>(r) * blockBytes,
          in + static_cast<size_t>(d) * blockBytes,
          blockBytes,
          hipMemcpyDeviceToDevice,
          st.stream()));
    }
    launchWriteFlag(mesh_->peerFlag(d, fDATA_ + r), seq, st.stream());
  }
  // Local block.
  GA_HIP_CHECK(hipMemcpyAsync(
      out + static_cast<size_t>(r) * blockBytes,
      in + static_cast<size_t>(r) * blockBytes,
      blockBytes,
      hipMemcpyDeviceToDevice,
      cs_->stream()));
  // Collect incoming blocks as they land.
  for (int s = 0; s < P; s++) {
    if (s == r) {
      continue;
    }
    launchWaitFlagGte(mesh_->flag(fDATA_ + s), seq, cs_->stream());
    GA_HIP_CHECK(hipMemcpyAsync(
        out + static_cast<size_t>(s) * blockBytes,
        work + static_cast<size_t>(s) * blockBytes,
        blockBytes,
        hipMemcpyDeviceToDevice,
        cs_->stream()));
    launchWriteFlag(mesh_->peerFlag(s, fACK_ + r), seq, cs_->stream());
  }
  {
    auto timeout = ctx_->getTimeout();
    HipEvent done(device_);
    done.record(cs_->stream());
    watchdogWait(done, *mesh_, timeout, "hip_alltoall (cs)");
  }
  cs_->synchronize();
  for (auto& st : fanout_) {
    HipEvent done(device_);
    done.record(st->stream());
    watchdogWait(done, *mesh_, ctx_->getTimeout(), "hip_alltoall (fanout)");
    st->synchronize();
  }
}

HipAlltoall::~HipAlltoall() {
  drainBounded(cs_);
  for (auto* st : fanout_) {
    drainBounded(st);
  }
}

// ===========================================================================
// HipP2P
// ===========================================================================

HipP2P::HipP2P(std::shared_ptr<Context> ctx, int device, size_t chunkCap)
    : ctx_(std::move(ctx)),
      device_(device),
      chunkCap_(chunkCap == 0 ? (4 << 20) : chunkCap) {
  GA_HIP_CHECK(hipSetDevice(device_));
  // Arena: P lanes x 2 parity slots x chunkCap, laid out over the
  // mesh's two contiguous inboxes (work | inbox0 | inbox1 in mesh.h).
  mesh_ = std::make_unique<XgmiMesh>(
      ctx_, device_, 0, static_cast<size_t>(ctx_->size) * chunkCap_);
  ss_ = std::make_unique<HipStream>(device_);
  rs_ = std::make_unique<HipStream>(device_);
  gateSendEv_ = std::make_unique<HipEvent>(device_);
  gateRecvEv_ = std::make_unique<HipEvent>(device_);
  fDATA_ = mesh_->allocFlags(ctx_->size);
  fACK_ = mesh_->allocFlags(ctx_->size);
  sendChunkSeq_.assign(ctx_->size, 0);
  recvChunkSeq_.assign(ctx_->size, 0);
}

void HipP2P::postSend(
    int dst,
    const void* devPtr,
    size_t bytes,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_p2p_send");
  std::lock_guard<std::mutex> lock(mu_);
  GA_HIP_CHECK(hipSetDevice(device_));
  const int r = ctx_->rank;
  GA_ENFORCE(dst != r, "p2p send to self");
  gateStreams(*gateSendEv_, callerStream, {ss_->stream()});
  const char* src = static_cast<const char*>(devPtr);
  const size_t nc = (bytes + chunkCap_ - 1) / chunkCap_;
  for (size_t c = 0; c < std::max<size_t>(nc, 1); c++) {
    const uint64_t seq = ++sendChunkSeq_[dst];
    const int par = static_cast<int>(seq & 1);
    if (seq > 2) {
      // Same-parity slot reuse: dst consumed chunk seq-2.
      launchWaitFlagGte(mesh_->flag(fACK_ + dst), seq - 2, ss_->stream());
    }
    const size_t off = c * chunkCap_;
    const size_t len = std::min(chunkCap_, bytes - std::min(bytes, off));
    if (len > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          slotOf(mesh_->peerInbox(dst, 0), r, par),
          src + off,
          len,
          hipMemcpyDeviceToDevice,
          ss_->stream()));
    }
    launchWriteFlag(mesh_->peerFlag(dst, fDATA_ + r), seq, ss_->stream());
  }
}

void HipP2P::postRecv(
    int src,
    void* devPtr,
    size_t bytes,
    hipStream_t callerStream) {
  TraceRange tr("gloo_amd::hip_p2p_recv");
  std::lock_guard<std::mutex> lock(mu_);
  GA_HIP_CHECK(hipSetDevice(device_));
  GA_ENFORCE(src != ctx_->rank, "p2p recv from self");
  gateStreams(*gateRecvEv_, callerStream, {rs_->stream()});
  char* dstPtr = static_cast<char*>(devPtr);
  const size_t nc = (bytes + chunkCap_ - 1) / chunkCap_;
  for (size_t c = 0; c < std::max<size_t>(nc, 1); c++) {
    const uint64_t seq = ++recvChunkSeq_[src];
    const int par = static_cast<int>(seq & 1);
    launchWaitFlagGte(mesh_->flag(fDATA_ + src), seq, rs_->stream());
    const size_t off = c * chunkCap_;
    const size_t len = std::min(chunkCap_, bytes - std::min(bytes, off));
    if (len > 0) {
      GA_HIP_CHECK(hipMemcpyAsync(
          dstPtr + off,
          slotOf(mesh_->inbox(0), src, par),
          len,
          hipMemcpyDeviceToDevice,
          rs_->stream()));
    }
    launchWriteFlag(
        mesh_->peerFlag(src, fACK_ + ctx_->rank), seq, rs_->stream());
  }
}

void HipP2P::flushSends() {
  GA_HIP_CHECK(hipSetDevice(device_));
  HipEvent done(device_);
  done.record(ss_->stream());
  watchdogWait(done, *mesh_, ctx_->getTimeout(), "hip_p2p (sends)");
  ss_->synchronize();
}

void HipP2P::flushRecvs() {
  GA_HIP_CHECK(hipSetDevice(device_));
  HipEvent done(device_);
  done.record(rs_->stream());
  watchdogWait(done, *mesh_, ctx_->getTimeout(), "hip_p2p (recvs)");
  rs_->synchronize();
}

// ===========================================================================
// hipAllreduceLocal
// ===========================================================================

void hipAllreduceLocal(
    const std::vector<void*>& ptrs,
    size_t elements,
    DType dtype,
    ReduceOp op,
    int device,
    hipStream_t callerStream) {
  GA_ENFORCE(!ptrs.empty());
  if (ptrs.size() == 1 || elements == 0) {
    return;
  }
  GA_HIP_CHECK(hipSetDevice(device));
  // Cached per-thread stream: creation costs ~300us, far more than the
  // kernels themselves for MB-scale buffers.
  static thread_local std::unique_ptr<HipStream> cached;
  static thread_local std::unique_ptr<HipEvent> cachedEv;
  if (!cached || cached->device() != device) {
    cached = std::make_unique<HipStream>(device);
    cachedEv = std::make_unique<HipEvent>(device);
  }
  HipStream& s = *cached;
  gateStreams(*cachedEv, callerStream, {s.stream()});
  const size_t es = dtypeSize(dtype);
  if (ptrs.size() <= 8) {
    // Fused: one pass reduces and broadcasts (every pointer gets the
    // result) — 2k*N traffic instead of (2k+1)*N plus copies.
    launchReduceNAll(
        ptrs.data(), static_cast<int>(ptrs.size()), elements, dtype, op,
        s.stream());
  } else {
    launchReduceN(
        ptrs[0], const_cast<const void* const*>(ptrs.data()), 8, elements,
        dtype, op, s.stream());
    for (size_t i = 8; i < ptrs.size(); i++) {
      launchReduce2(ptrs[0], ptrs[0], ptrs[i], elements, dtype, op,
                    s.stream());
    }
    for (size_t i = 1; i < ptrs.size(); i++) {
      GA_HIP_CHECK(hipMemcpyAsync(
          ptrs[i], ptrs[0], elements * es, hipMemcpyDeviceToDevice,
          s.stream()));
    }
  }
  s.synchronize();
}

} // namespace hip
} // namespace glooamd
