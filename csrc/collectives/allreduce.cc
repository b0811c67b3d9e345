// v2 allreduce: segmented ring (default) and bcube.
//
// Re-derived from the algorithm shapes described by the reference
// (gloo/allreduce.cc:147-392 ring with <=1MiB segments, >=2 per rank,
// double-buffered tmp; gloo/allreduce.cc:428-676 bcube), implemented
// fresh. See csrc/collectives/collectives.h for the schedule summary.
#include <cstring>

#include "collectives/collectives.h"
#include "collectives/schedule.h"
#include "common/logging.h"
#include "common/utils.h"
#include "types.h"

namespace glooamd {

namespace {

using sched::Seg;
using sched::blockOf;
using sched::segmentOf;
using sched::subspanOf;

void localReduceInputs(const AllreduceOptions& opts, char* out) {
  const size_t nbytes = opts.elements * opts.elementSize;
  if (!opts.inputs.empty()) {
    if (opts.inputs[0] != out) {
      std::memcpy(out, opts.inputs[0], nbytes);
    }
    for (size_t i = 1; i < opts.inputs.size(); i++) {
      opts.reduce(out, out, opts.inputs[i], opts.elements);
    }
  } else {
    // in-place on outputs[0]; fold extra outputs in as inputs
    for (size_t i = 1; i < opts.outputs.size(); i++) {
      opts.reduce(out, out, opts.outputs[i], opts.elements);
    }
  }
}

void broadcastOutputs(const AllreduceOptions& opts, const char* out) {
  const size_t nbytes = opts.elements * opts.elementSize;
  for (size_t i = 1; i < opts.outputs.size(); i++) {
    if (opts.outputs[i] != out) {
      std::memcpy(opts.outputs[i], out, nbytes);
    }
  }
}

void ringAllreduce(AllreduceOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t N = opts.elements;
  const size_t es = opts.elementSize;
  char* out = static_cast<char*>(opts.outputs[0]);

  const size_t perRank = (N + P - 1) / P;
  const int S = std::max<int>(
      2, static_cast<int>((perRank * es + opts.maxSegmentSize - 1) /
                          opts.maxSegmentSize));
  const size_t segCapBytes = ((perRank + S - 1) / S) * es;

  auto tmp = makeAligned(2 * segCapBytes + 64);
  auto outBuf = ctx->createUnboundBuffer(out, N * es);
  auto tmpBuf = ctx->createUnboundBuffer(tmp.get(), 2 * segCapBytes + 64);

  const uint64_t slot = Slot::build(SlotPrefix::kAllreduce, opts.tag);
  const int sendRank = (r + 1) % P;
  const int recvRank = (r - 1 + P) % P;

  // Unified step schedule: (P-1)*S reduce-scatter steps then (P-1)*S
  // allgather steps. RS step k receives into tmp[k&1] and reduces;
  // AG steps receive directly into the output.
  struct Step {
    Seg send;
    Seg recv;
    bool reduceStep;
  };
  const int K1 = (P - 1) * S;
  std::vector<Step> steps;
  steps.reserve(2 * K1);
  for (int k = 0; k < K1; k++) {
    int i = k / S, s = k % S;
    steps.push_back({segmentOf(N, P, (r - i + P) % P, s, S),
                     segmentOf(N, P, (r - i - 1 + 2 * P) % P, s, S),
                     true});
  }
  for (int k = 0; k < K1; k++) {
    int i = k / S, s = k % S;
    steps.push_back({segmentOf(N, P, (r + 1 - i + P) % P, s, S),
                     segmentOf(N, P, (r - i + P) % P, s, S),
                     false});
  }

  auto postRecv = [&](int k) {
    const Step& st = steps[k];
    if (st.reduceStep) {
      tmpBuf->recv(recvRank, slot, (k & 1) * segCapBytes, st.recv.len * es);
    } else {
      outBuf->recv(recvRank, slot, st.recv.off * es, st.recv.len * es);
    }
  };

  const int K = static_cast<int>(steps.size());
  const auto timeout = opts.timeout.count() > 0 ? opts.timeout
                                                : ctx->getTimeout();
  int outstandingSends = 0;
  for (int k = 0; k < std::min(2, K); k++) {
    postRecv(k);
  }
  for (int k = 0; k < K; k++) {
    const Step& st = steps[k];
    if (outstandingSends >= 2) {
      outBuf->waitSend(timeout);
      outstandingSends--;
    }
    outBuf->send(sendRank, slot, st.send.off * es, st.send.len * es);
    outstandingSends++;
    if (st.reduceStep) {
      tmpBuf->waitRecv(timeout);
      if (st.recv.len > 0) {
        opts.reduce(
            out + st.recv.off * es,
            out + st.recv.off * es,
            tmp.get() + (k & 1) * segCapBytes,
            st.recv.len);
      }
    } else {
      outBuf->waitRecv(timeout);
    }
    if (k + 2 < K) {
      postRecv(k + 2);
    }
  }
  while (outstandingSends-- > 0) {
    outBuf->waitSend(timeout);
  }
}

// Latency-optimized path for small payloads: full-buffer recursive
// doubling (log2 P rounds of concurrent exchange) instead of the ring's
// 4(P-1) serialized segment hops. Non-power-of-2 sizes fold the extra
// ranks into partners before/after the exchange.
void smallAllreduce(AllreduceOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t bytes = opts.elements * opts.elementSize;
  char* out = static_cast<char*>(opts.outputs[0]);
  const uint64_t slot = Slot::build(SlotPrefix::kAllreduce, opts.tag);
  const auto timeout = opts.timeout.count() > 0 ? opts.timeout
                                                : ctx->getTimeout();
  int pow2 = 1, steps = 0;
  while (pow2 * 2 <= P) {
    pow2 *= 2;
    steps++;
  }
  const int extras = P - pow2;

  auto tmp = makeAligned(bytes + 64);
  auto outBuf = ctx->createUnboundBuffer(out, bytes);
  auto tmpBuf = ctx->createUnboundBuffer(tmp.get(), bytes + 64);

  if (r >= pow2) {
    outBuf->send(r - pow2, slot, 0, bytes);
    outBuf->waitSend(timeout);
  } else if (r < extras) {
    tmpBuf->recv(r + pow2, slot, 0, bytes);
    tmpBuf->waitRecv(timeout);
    opts.reduce(out, out, tmp.get(), opts.elements);
  }
  if (r < pow2) {
    for (int t = 0; t < steps; t++) {
      const int peer = r ^ (1 << t);
      tmpBuf->recv(peer, slot + 1 + t, 0, bytes);
      outBuf->send(peer, slot + 1 + t, 0, bytes);
      tmpBuf->waitRecv(timeout);
      // The transport serializes send payloads lazily (the TxOp holds a
      // pointer into `out`); the in-place reduce below would corrupt a
      // partially-flushed send. Wait for the send first — deadlock-free
      // because both peers posted their recvs before sending.
      outBuf->waitSend(timeout);
      opts.reduce(out, out, tmp.get(), opts.elements);
    }
  }
  if (r < extras) {
    outBuf->send(r + pow2, slot + 1 + steps, 0, bytes);
    outBuf->waitSend(timeout);
  } else if (r >= pow2) {
    outBuf->recv(r - pow2, slot + 1 + steps, 0, bytes);
    outBuf->waitRecv(timeout);
  }
}

// Factorize P into group sizes per step: repeated `base` factors, then
// the remainder as one final (possibly larger) factor — reference
// computeGroupSizePerStep semantics (gloo/allreduce.cc:397-408): a prime
// P degenerates to one step = direct reduce-scatter + allgather.
std::vector<int> bcubeFactors(int P, int base) {
  std::vector<int> f;
  while (P % base == 0) {
    f.push_back(base);
    P /= base;
  }
  if (P > 1) {
    f.push_back(P);
  }
  return f;
}

void bcubeAllreduce(AllreduceOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t N = opts.elements;
  const size_t es = opts.elementSize;
  const int base = std::max(2, ctx->base);
  char* out = static_cast<char*>(opts.outputs[0]);

  const std::vector<int> factors = bcubeFactors(P, base);
  const int k = static_cast<int>(factors.size());
  int maxFactor = 2;
  for (int f : factors) {
    maxFactor = std::max(maxFactor, f);
  }

  const uint64_t slot = Slot::build(SlotPrefix::kAllreduce, opts.tag);
  const auto timeout = opts.timeout.count() > 0 ? opts.timeout
                                                : ctx->getTimeout();

  // tmp: (maxFactor-1) slots sized for the largest piece we ever receive
  // (step 0 receives a 1/factors[0] sub-span of the full buffer).
  const size_t pieceCap = ((N + factors[0] - 1) / factors[0] + 1) * es;
  auto tmp = makeAligned((maxFactor - 1) * pieceCap + 64);
  auto outBuf = ctx->createUnboundBuffer(out, N * es);
  auto tmpBuf =
      ctx->createUnboundBuffer(tmp.get(), (maxFactor - 1) * pieceCap + 64);

  Seg span{0, N};
  std::vector<Seg> spanAtStep(k);

  // Reduce-scatter stage.
  int stride = 1;
  for (int t = 0; t < k; t++) {
    const int b = factors[t];
    spanAtStep[t] = span;
    const int myIdx = (r / stride) % b;
    // Post recvs of my sub-span from each peer.
    Seg mine = subspanOf(span, myIdx, b);
    GA_ENFORCE_LE(mine.len * es, pieceCap);
    int nrecv = 0;
    for (int j = 0; j < b; j++) {
      if (j == myIdx) {
        continue;
      }
      const int peer = r + (j - myIdx) * stride;
      tmpBuf->recv(peer, slot + t, nrecv * pieceCap, mine.len * es);
      nrecv++;
    }
    // Send each peer its sub-span.
    for (int j = 0; j < b; j++) {
      if (j == myIdx) {
        continue;
      }
      const int peer = r + (j - myIdx) * stride;
      Seg theirs = subspanOf(span, j, b);
      outBuf->send(peer, slot + t, theirs.off * es, theirs.len * es);
    }
    // Reduce received pieces (arrival order unknown across peers, but each
    // landed in its own tmp slot; reduce sequentially).
    for (int j = 0; j < nrecv; j++) {
      tmpBuf->waitRecv(timeout);
    }
    for (int j = 0; j < nrecv; j++) {
      if (mine.len > 0) {
        opts.reduce(
            out + mine.off * es,
            out + mine.off * es,
            tmp.get() + j * pieceCap,
            mine.len);
      }
    }
    for (int j = 0; j < b - 1; j++) {
      outBuf->waitSend(timeout);
    }
    span = mine;
    stride *= b;
  }

  // Allgather stage (mirror).
  for (int t = k - 1; t >= 0; t--) {
    const int b = factors[t];
    stride /= b;
    const int myIdx = (r / stride) % b;
    const Seg stepSpan = spanAtStep[t];
    // Post recvs for every other sub-span directly into out.
    for (int j = 0; j < b; j++) {
      if (j == myIdx) {
        continue;
      }
      const int peer = r + (j - myIdx) * stride;
      Seg theirs = subspanOf(stepSpan, j, b);
      outBuf->recv(peer, slot + k + t, theirs.off * es, theirs.len * es);
    }
    // Send my sub-span to every peer.
    Seg mine = subspanOf(stepSpan, myIdx, b);
    for (int j = 0; j < b; j++) {
      if (j == myIdx) {
        continue;
      }
      const int peer = r + (j - myIdx) * stride;
      outBuf->send(peer, slot + k + t, mine.off * es, mine.len * es);
    }
    for (int j = 0; j < b - 1; j++) {
      outBuf->waitRecv(timeout);
    }
    for (int j = 0; j < b - 1; j++) {
      outBuf->waitSend(timeout);
    }
    span = stepSpan;
  }
}

} // namespace

void allreduce(AllreduceOptions& opts) {
  GA_ENFORCE(opts.context != nullptr);
  GA_ENFORCE(!opts.outputs.empty(), "allreduce needs at least one output");
  GA_ENFORCE(opts.reduce != nullptr, "allreduce needs a reduction function");
  GA_ENFORCE_GT(opts.elementSize, 0ul);

  char* out = static_cast<char*>(opts.outputs[0]);
  localReduceInputs(opts, out);

  const int P = opts.context->size;
  if (P > 1 && opts.elements > 0) {
    // bcube handles any size: P factorizes into repeated `base` factors
    // plus a remainder step (prime P = one direct exchange step).
    const bool bcubeOk =
        opts.algorithm == AllreduceOptions::Algorithm::BCUBE;
    static const size_t smallThreshold = static_cast<size_t>(
        getEnvInt("GLOO_AMD_SMALL_ALLREDUCE", 16384));
    if (opts.elements * opts.elementSize <= smallThreshold &&
        opts.algorithm == AllreduceOptions::Algorithm::RING) {
      smallAllreduce(opts);
    } else if (bcubeOk) {
      bcubeAllreduce(opts);
    } else {
      ringAllreduce(opts);
    }
  }
  broadcastOutputs(opts, out);
}

void reduce_scatter(ReduceScatterOptions& opts) {
  GA_ENFORCE(opts.context != nullptr);
  GA_ENFORCE(opts.reduce != nullptr);
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t B = opts.recvElements; // block size per rank
  const size_t es = opts.elementSize;

  if (P == 1) {
    if (opts.output != opts.input && opts.input != nullptr) {
      std::memcpy(opts.output, opts.input, B * es);
    }
    return;
  }

  // Scratch copy of the input so the caller's buffer is preserved.
  const size_t N = B * P;
  auto scratch = makeAligned(N * es + 64);
  GA_ENFORCE(opts.input != nullptr, "reduce_scatter needs an input");
  std::memcpy(scratch.get(), opts.input, N * es);
  char* buf = scratch.get();

  auto tmp = makeAligned(2 * B * es + 64);
  auto bufUb = ctx->createUnboundBuffer(buf, N * es);
  auto tmpUb = ctx->createUnboundBuffer(tmp.get(), 2 * B * es + 64);
  const uint64_t slot = Slot::build(SlotPrefix::kReduceScatter, opts.tag);
  const auto timeout = opts.timeout.count() > 0 ? opts.timeout
                                                : ctx->getTimeout();
  const int sendRank = (r + 1) % P;
  const int recvRank = (r - 1 + P) % P;

  // Ring reduce-scatter ending with rank r owning block r:
  // iteration i: send block (r-i-1), recv+reduce block (r-i-2).
  auto blockOff = [&](int b) { return static_cast<size_t>(b) * B * es; };
  int outstanding = 0;
  for (int i = 0; i < std::min(2, P - 1); i++) {
    tmpUb->recv(recvRank, slot, (i & 1) * B * es, B * es);
  }
  for (int i = 0; i < P - 1; i++) {
    int sendBlock = (r - i - 1 + 2 * P) % P;
    int recvBlock = (r - i - 2 + 2 * P) % P;
    if (outstanding >= 2) {
      bufUb->waitSend(timeout);
      outstanding--;
    }
    bufUb->send(sendRank, slot, blockOff(sendBlock), B * es);
    outstanding++;
    tmpUb->waitRecv(timeout);
    if (B > 0) {
      opts.reduce(
          buf + blockOff(recvBlock),
          buf + blockOff(recvBlock),
          tmp.get() + (i & 1) * B * es,
          B);
    }
    if (i + 2 < P - 1) {
      tmpUb->recv(recvRank, slot, (i & 1) * B * es, B * es);
    }
  }
  while (outstanding-- > 0) {
    bufUb->waitSend(timeout);
  }
  std::memcpy(opts.output, buf + blockOff(r), B * es);
}

} // namespace glooamd
