// Device-side collective algorithms over the xGMI/IPC mesh.
//
// Reference parity set (gloo/cuda_allreduce_ring.{h,cc},
// cuda_allreduce_ring_chunked.{h,cc}, cuda_allreduce_halving_doubling,
// cuda_allreduce_local, cuda_broadcast_one_to_all), re-derived for
// MI355X (see csrc/hip/mesh.h header for the transport re-design):
//
//  * HipAllreduceRing (chunked=false -> whole-block ring like
//    cuda_allreduce_ring; chunked=true -> >=2 pipelined segments per
//    block, the reference's ring_chunked schedule). The entire
//    schedule — peer SDMA copies, doorbells, reduction kernels — is
//    enqueued on two HIP streams with GPU-side flag waits; the host
//    leaves the critical path (the reference pays a host round trip
//    per chunk, cuda_allreduce_ring_chunked.cc:168-213).
//  * HipAllreduceHalvingDoubling: recursive vector halving / distance
//    doubling for power-of-2 sizes; each step's half moves in
//    inbox-capacity sub-chunks (double-buffered); the allgather mirror
//    writes straight into the peer's work region (single writer per
//    region, so no inbox hop is needed).
//  * HipBroadcastOneToAll: root fans out over several streams so
//    multiple xGMI links run concurrently.
//  * hipAllreduceLocal: single-process multi-pointer fused reduce +
//    broadcast (reference cuda_allreduce_local.cc).
//
// Buffers below kOnDeviceThreshold take the host-staged path (pinned
// D2H -> CPU tcp/shm collective -> H2D), mirroring the reference's
// kOnDeviceThreshold host/device workspace split (gloo/algorithm.cc:16)
// with MI355X-tuned default (256 KiB, env GLOO_AMD_DEVICE_THRESHOLD).
#pragma once

#include <array>
#include <map>
#include <memory>
#include <mutex>
#include <tuple>
#include <vector>

#include "collectives/reduce_fns.h"
#include "hip/core.h"
#include "hip/mesh.h"

namespace glooamd {
namespace hip {

size_t onDeviceThreshold(); // bytes; below this, host-staged path

class HipAllreduceRing {
 public:
  // numRings 0 -> auto: partition the buffer across rings whose neighbor
  // strides are coprime to size (8 ranks -> strides 1,3,5,7), so several
  // xGMI links carry traffic concurrently instead of one. Each ring gets
  // its own flag slots, inbox region and stream pair from the pool.
  HipAllreduceRing(
      std::shared_ptr<Context> ctx,
      int device,
      bool chunked = true,
      size_t inboxCap = 0 /*0 -> default 4 MiB*/,
      int numRings = 0);

  // In-place allreduce of a device buffer. Blocking (streams synced).
  // callerStream: the stream on which the caller produced devPtr (e.g.
  // torch.cuda.current_stream()); our streams are event-ordered after it.
  void run(
      void* devPtr,
      size_t elements,
      DType dtype,
      ReduceOp op,
      hipStream_t callerStream = nullptr);
  // Multi-input form (reference cuda benchmark --inputs sweep parity):
  // fused k-way local reduction into ptrs[0] as the copy-in stage, then
  // the wire phase in place on ptrs[0], result broadcast to every ptr.
  void run(
      const std::vector<void*>& ptrs,
      size_t elements,
      DType dtype,
      ReduceOp op,
      hipStream_t callerStream = nullptr);

  XgmiMesh* mesh() {
    return mesh_.get();
  }

  ~HipAllreduceRing();

 private:
  void runDevice(char* buf, size_t bytes, size_t n, DType dt, ReduceOp op);
  void runHostStaged(char* buf, size_t bytes, size_t n, DType dt, ReduceOp op);
  // hipGraph steady-state replay of the whole multi-ring schedule (one
  // launch instead of hundreds of enqueues). Returns false (and may set
  // graphBroken_) when the eager path must run instead.
  bool runDeviceGraph(char* buf, size_t n, size_t es, DType dt, ReduceOp op);
  int ringSegments(size_t partLen, size_t es) const;
  // Enqueue one ring's full schedule (reduce-scatter + allgather over
  // the element range [elemOff, elemOff+elems) with neighbor stride
  // strides_[j]) on that ring's stream pair; returns the step count.
  // rel=true uses *seqBaseDev_-relative doorbells (graph capture).
  int enqueueRing(
      int j,
      char* work,
      size_t elemOff,
      size_t elems,
      size_t es,
      DType dt,
      ReduceOp op,
      bool rel = false,
      int64_t crossRunOff = 0);

  std::shared_ptr<Context> ctx_;
  int device_;
  bool chunked_;
  size_t inboxCap_;
  std::unique_ptr<XgmiMesh> mesh_;
  // per-ring resources (ring j uses pooled streams 2j / 2j+1)
  std::vector<int> strides_;
  std::vector<HipStream*> cs_;
  std::vector<HipStream*> ks_;
  std::vector<std::vector<std::unique_ptr<HipEvent>>> events_;
  std::vector<std::unique_ptr<HipEvent>> initEvent_;
  std::vector<std::unique_ptr<HipEvent>> doneEvent_;
  // Persistent gate events: a transient event destroyed with pending
  // stream waits can wedge a stream (see gateStreams in algorithms.cc).
  std::unique_ptr<HipEvent> gateEv_;
  std::unique_ptr<HipEvent> graphFork_;
  std::vector<int> fDATA_, fACK_; // flag indices (x2 parity each, per ring)
  std::vector<std::array<uint64_t, 2>> lastAckPerRing_;
  uint64_t seqBase_{0};
  uint64_t ringSteps_{0}; // steps per ring in the last run
  // pinned staging for the host path
  void* hostStage_{nullptr};
  size_t hostStageCap_{0};
  // hipGraph replay state (see runDeviceGraph)
  struct GraphKey {
    const void* ptr;
    size_t n;
    int dt;
    int op;
    bool operator<(const GraphKey& o) const {
      return std::tie(ptr, n, dt, op) < std::tie(o.ptr, o.n, o.dt, o.op);
    }
    bool operator==(const GraphKey& o) const {
      return ptr == o.ptr && n == o.n && dt == o.dt && op == o.op;
    }
  };
  struct GraphEntry {
    hipGraphExec_t exec{nullptr};
    int steps{0};
    std::vector<int> ringK;
  };
  std::map<GraphKey, GraphEntry> graphs_;
  GraphKey lastKey_{nullptr, 0, -1, -1};
  uint64_t* seqBaseDev_{nullptr};
  bool graphBroken_{false};
};

class HipAllreduceHalvingDoubling {
 public:
  HipAllreduceHalvingDoubling(
      std::shared_ptr<Context> ctx,
      int device,
      size_t inboxCap = 0);

  void run(
      void* devPtr,
      size_t elements,
      DType dtype,
      ReduceOp op,
      hipStream_t callerStream = nullptr);
  ~HipAllreduceHalvingDoubling();

 private:
  std::shared_ptr<Context> ctx_;
  int device_;
  size_t inboxCap_;
  int log2P_;
  int pow2_; // largest power of 2 <= size; ranks >= pow2_ fold
  std::unique_ptr<XgmiMesh> mesh_;
  HipStream* cs_;
  HipStream* ks_;
  std::unique_ptr<HipEvent> stepEvent_;
  std::unique_ptr<HipEvent> initEvent_;
  std::unique_ptr<HipEvent> doneEvent_;
  std::unique_ptr<HipEvent> gateEv_;
  std::vector<std::unique_ptr<HipEvent>> stepEvents_;
  int fDATA_, fACK_, fAGD_; // per step t: [t*2 + parity]
  // Non-pow2 folding flags (allocated only when size != pow2_).
  int fFOLD_{-1}, fFACK_{-1}, fPOST_{-1}, fPACK_{-1};
  uint64_t seq_{0};
  uint64_t foldSeq_{0}; // private partner<->extra chunk counter
  uint64_t postSeq_{0}; // per-run post-fold handshake counter
  uint64_t lastFold_[2] = {0, 0};
  // Last-issued doorbell seq per (step, parity) / per step; identical on
  // all ranks (symmetric schedule) and persistent across runs so inbox
  // reuse is safe across run boundaries.
  std::vector<uint64_t> lastAck_;
  std::vector<uint64_t> lastAgd_;
};

class HipBroadcastOneToAll {
 public:
  HipBroadcastOneToAll(
      std::shared_ptr<Context> ctx,
      int device,
      int root = 0,
      int numStreams = 4);

  void run(void* devPtr, size_t bytes, hipStream_t callerStream = nullptr);
  ~HipBroadcastOneToAll();

  std::vector<uint64_t> debugFlags() {
    return mesh_->readFlags();
  }

 private:
  std::shared_ptr<Context> ctx_;
  int device_;
  int root_;
  std::unique_ptr<XgmiMesh> mesh_;
  HipStream* cs_;
  std::unique_ptr<HipEvent> gateEv_;
  std::vector<HipStream*> fanout_;
  int fBDATA_;
  int fBACK_;
  uint64_t seq_{0};
};

// Fully-connected "direct" allreduce: every MI355X pair shares an xGMI
// link, so the optimal schedule is one-shot — each rank scatters its P
// blocks to their owners over P-1 DIFFERENT links concurrently (fanout
// streams -> separate SDMA engines), fuses a (P)-way reduction of its
// own block, then broadcasts the result over all links at once. Wire
// time ~ 2*S*(P-1)/P divided by ~(P-1) concurrent links, vs the ring's
// per-link serialization. No reference counterpart: NVSwitch hides this
// choice; point-to-point xGMI rewards it.
class HipAllreduceDirect {
 public:
  HipAllreduceDirect(
      std::shared_ptr<Context> ctx,
      int device,
      int numStreams = 7);

  void run(
      void* devPtr,
      size_t elements,
      DType dtype,
      ReduceOp op,
      hipStream_t callerStream = nullptr);
  // Multi-input form: fused local reduceN into ptrs[0], then the one-shot
  // wire phase, result broadcast to every ptr.
  void run(
      const std::vector<void*>& ptrs,
      size_t elements,
      DType dtype,
      ReduceOp op,
      hipStream_t callerStream = nullptr);
  ~HipAllreduceDirect();

 private:
  std::shared_ptr<Context> ctx_;
  int device_;
  std::unique_ptr<XgmiMesh> mesh_;
  HipStream* cs_;
  std::vector<HipStream*> fanout_;
  std::unique_ptr<HipEvent> doneEvent_;
  std::unique_ptr<HipEvent> gateEv_;
  std::vector<std::unique_ptr<HipEvent>> chunkEvents_;
  int fRS_; // [src] scatter chunk arrived (monotonic per-chunk seq)
  int fAG_; // [src] reduced chunk arrived
  int fACK_; // [src] inbox slot consumed (per-chunk seq)
  int fDONE_; // [src] peer's copy-out done (per-run seq)
  uint64_t chunkSeqBase_{0};
  uint64_t runSeq_{0};
  std::vector<std::array<uint64_t, 2>> lastAck_; // per src, per parity
};

// Base-b hypercube allreduce (reference cuda_allreduce_bcube parity).
// On a fully-connected xGMI node, bcube's purpose — engaging multiple
// links per step — is achieved optimally by the one-shot direct
// schedule (bcube with base == size IS direct; base == 2 is halving
// doubling). This class therefore dispatches to the strongest matching
// engine instead of re-implementing the grouped multi-step exchange.
class HipAllreduceBcube {
 public:
  HipAllreduceBcube(
      std::shared_ptr<Context> ctx,
      int device,
      int base = 0 /*0 -> context base*/);

  void run(
      void* devPtr,
      size_t elements,
      DType dtype,
      ReduceOp op,
      hipStream_t callerStream = nullptr);

 private:
  std::unique_ptr<HipAllreduceDirect> direct_;
  std::unique_ptr<HipAllreduceHalvingDoubling> hd_;
  std::unique_ptr<HipAllreduceRing> ring_;
};

// Device-native allgather ring: out (size*inElements) assembled via the
// same segmented inbox pipeline as the allreduce ring's allgather phase.
class HipAllgatherRing {
 public:
  HipAllgatherRing(
      std::shared_ptr<Context> ctx,
      int device,
      size_t inboxCap = 0);
  // out must hold size * inElements elements.
  void run(
      const void* devIn,
      void* devOut,
      size_t inElements,
      size_t es,
      hipStream_t callerStream = nullptr);
  ~HipAllgatherRing();

 private:
  std::shared_ptr<Context> ctx_;
  int device_;
  size_t inboxCap_;
  std::unique_ptr<XgmiMesh> mesh_;
  HipStream* cs_;
  HipStream* ks_;
  std::vector<std::unique_ptr<HipEvent>> events_;
  std::unique_ptr<HipEvent> initEvent_;
  std::unique_ptr<HipEvent> doneEvent_;
  std::unique_ptr<HipEvent> gateEv_;
  int fDATA_, fACK_;
  uint64_t seqBase_{0};
  uint64_t lastAck_[2] = {0, 0};
};

// Device-native ring reduce-scatter: rank r ends with the reduced
// block r (recvElements) of the size*recvElements input.
class HipReduceScatterRing {
 public:
  HipReduceScatterRing(
      std::shared_ptr<Context> ctx,
      int device,
      size_t inboxCap = 0);
  void run(
      const void* devIn,
      void* devOut,
      size_t recvElements,
      DType dtype,
      ReduceOp op,
      hipStream_t callerStream = nullptr);
  ~HipReduceScatterRing();

 private:
  std::shared_ptr<Context> ctx_;
  int device_;
  size_t inboxCap_;
  std::unique_ptr<XgmiMesh> mesh_;
  HipStream* cs_;
  HipStream* ks_;
  std::vector<std::unique_ptr<HipEvent>> events_;
  std::unique_ptr<HipEvent> initEvent_;
  std::unique_ptr<HipEvent> doneEvent_;
  std::unique_ptr<HipEvent> gateEv_;
  int fDATA_, fACK_;
  uint64_t seqBase_{0};
  uint64_t lastAck_[2] = {0, 0};
};

// Device-native all-to-all: every rank writes its block for rank d
// directly into d's work region over xGMI (single writer per region;
// fan-out over several streams so multiple links run concurrently).
class HipAlltoall {
 public:
  HipAlltoall(
      std::shared_ptr<Context> ctx,
      int device,
      int numStreams = 4);
  void run(
      const void* devIn,
      void* devOut,
      size_t perRankElements,
      size_t es,
      hipStream_t callerStream = nullptr);
  ~HipAlltoall();

 private:
  std::shared_ptr<Context> ctx_;
  int device_;
  std::unique_ptr<XgmiMesh> mesh_;
  HipStream* cs_;
  std::unique_ptr<HipEvent> gateEv_;
  std::vector<HipStream*> fanout_;
  int fDATA_; // [src]
  int fACK_; // [src]
  uint64_t seq_{0};
};

// Device-native tagged-order point-to-point engine (pipeline-parallel
// p2p + gather/scatter staging). Each (src->dst) lane is a
// double-buffered chunk slot in the receiver's inbox arena; posts
// enqueue the full chunked schedule (copies + doorbell waits) on
// dedicated streams and return immediately, so a host thread can post a
// send AND a recv before waiting either (batch_isend_irecv safety).
// Ordering per lane is post order on both sides (c10d's matching rule).
// Replaces the r01 PG host-staged path for CUDA send/recv/gather/
// scatter (reference role: gloo tagged unbound send/recv over pairs).
class HipP2P {
 public:
  HipP2P(
      std::shared_ptr<Context> ctx,
      int device,
      size_t chunkCap = 0 /*0 -> 4 MiB per lane slot*/);

  // Enqueue a chunked device-to-device send/recv (async; returns once
  // enqueued). The caller's tensor must stay alive until the matching
  // flush returns.
  void postSend(
      int dst,
      const void* devPtr,
      size_t bytes,
      hipStream_t callerStream = nullptr);
  void postRecv(
      int src,
      void* devPtr,
      size_t bytes,
      hipStream_t callerStream = nullptr);
  // Block until every posted send / recv completed (watchdog-guarded).
  void flushSends();
  void flushRecvs();

 private:
  char* slotOf(char* inboxBase, int src, int par) {
    return inboxBase + (static_cast<size_t>(src) * 2 + par) * chunkCap_;
  }

  std::shared_ptr<Context> ctx_;
  int device_;
  size_t chunkCap_;
  std::unique_ptr<XgmiMesh> mesh_;
  // Dedicated streams (not pooled): p2p waits must never interleave
  // with a collective engine's schedule on a shared stream.
  std::unique_ptr<HipStream> ss_;
  std::unique_ptr<HipStream> rs_;
  std::unique_ptr<HipEvent> gateSendEv_;
  std::unique_ptr<HipEvent> gateRecvEv_;
  int fDATA_; // [src]: chunks sent to me by src (monotonic)
  int fACK_; // [dst]: chunks dst consumed of my sends (monotonic)
  std::vector<uint64_t> sendChunkSeq_; // per dst lane
  std::vector<uint64_t> recvChunkSeq_; // per src lane
  std::mutex mu_; // posts may come from several host threads
};

// Single-process multi-pointer allreduce: fused k-way reduction into
// ptrs[0] then broadcast copies (all on one device/stream; blocking).
void hipAllreduceLocal(
    const std::vector<void*>& ptrs,
    size_t elements,
    DType dtype,
    ReduceOp op,
    int device,
    hipStream_t callerStream = nullptr);

} // namespace hip
} // namespace glooamd
