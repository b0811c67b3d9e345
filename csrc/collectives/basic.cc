// v2 collectives: allgather(v), alltoall(v), barrier, broadcast,
// gather(v), scatter, reduce.
//
// Algorithm shapes re-derived from the reference inventory
// (gloo/allgather.cc:27-96 ring,
// gloo/alltoall.cc:43-53 pairwise, gloo/barrier.cc:30-35 dissemination,
// gloo/broadcast.cc:44-96 binomial tree, gloo/gather.cc, gloo/scatter.cc,
// gloo/reduce.cc ring reduce-scatter + gather-to-root).
#include <cstring>

#include "collectives/collectives.h"
#include "common/logging.h"
#include "common/utils.h"
#include "math.h"
#include "types.h"

namespace glooamd {

namespace {
std::chrono::milliseconds resolveTimeout(
    const detail::CollectiveOptionsBase& opts) {
  return opts.timeout.count() > 0 ? opts.timeout : opts.context->getTimeout();
}
} // namespace

// ---------------------------------------------------------------------------
// allgather / allgatherv — ring, two blocks in flight
// ---------------------------------------------------------------------------

void allgatherv(AllgathervOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t es = opts.elementSize;
  GA_ENFORCE_EQ(opts.counts.size(), static_cast<size_t>(P));
  char* out = static_cast<char*>(opts.output);

  std::vector<size_t> offs(P + 1, 0);
  for (int i = 0; i < P; i++) {
    offs[i + 1] = offs[i] + opts.counts[i];
  }
  const size_t totalBytes = offs[P] * es;

  // Place own contribution.
  if (opts.input != nullptr && opts.input != out + offs[r] * es) {
    std::memcpy(out + offs[r] * es, opts.input, opts.counts[r] * es);
  }
  if (P == 1) {
    return;
  }

  auto outBuf = ctx->createUnboundBuffer(out, totalBytes);
  const uint64_t slot = Slot::build(SlotPrefix::kAllgather, opts.tag);
  const auto timeout = resolveTimeout(opts);
  const int sendRank = (r + 1) % P;
  const int recvRank = (r - 1 + P) % P;

  auto blockAt = [&](int i) { return (r - i + 2 * P) % P; };
  // iteration i in 0..P-2: send block (r-i), recv block (r-i-1)
  for (int i = 0; i < std::min(2, P - 1); i++) {
    int b = blockAt(i + 1);
    outBuf->recv(recvRank, slot, offs[b] * es, opts.counts[b] * es);
  }
  int outstanding = 0;
  for (int i = 0; i < P - 1; i++) {
    int sb = blockAt(i);
    if (outstanding >= 2) {
      outBuf->waitSend(timeout);
      outstanding--;
    }
    outBuf->send(sendRank, slot, offs[sb] * es, opts.counts[sb] * es);
    outstanding++;
    outBuf->waitRecv(timeout);
    if (i + 2 < P - 1) {
      int b = blockAt(i + 3);
      outBuf->recv(recvRank, slot, offs[b] * es, opts.counts[b] * es);
    }
  }
  while (outstanding-- > 0) {
    outBuf->waitSend(timeout);
  }
}

void allgather(AllgatherOptions& opts) {
  AllgathervOptions v(opts.context);
  v.tag = opts.tag;
  v.timeout = opts.timeout;
  v.input = opts.input;
  v.output = opts.output;
  v.elementSize = opts.elementSize;
  v.counts.assign(opts.context->size, opts.inElements);
  allgatherv(v);
}

// ---------------------------------------------------------------------------
// alltoall / alltoallv — pairwise exchange
// ---------------------------------------------------------------------------

void alltoallv(AlltoallvOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t es = opts.elementSize;
  GA_ENFORCE_EQ(opts.inCounts.size(), static_cast<size_t>(P));
  GA_ENFORCE_EQ(opts.outCounts.size(), static_cast<size_t>(P));
  char* in = static_cast<char*>(opts.input);
  char* out = static_cast<char*>(opts.output);

  std::vector<size_t> inOffs(P + 1, 0), outOffs(P + 1, 0);
  for (int i = 0; i < P; i++) {
    inOffs[i + 1] = inOffs[i] + opts.inCounts[i];
    outOffs[i + 1] = outOffs[i] + opts.outCounts[i];
  }

  // Self block.
  std::memcpy(out + outOffs[r] * es, in + inOffs[r] * es,
              opts.inCounts[r] * es);
  if (P == 1) {
    return;
  }

  auto inBuf = ctx->createUnboundBuffer(in, inOffs[P] * es);
  auto outBuf = ctx->createUnboundBuffer(out, outOffs[P] * es);
  const uint64_t slot = Slot::build(SlotPrefix::kAlltoall, opts.tag);
  const auto timeout = resolveTimeout(opts);

  // Post all recvs, then send to (r+i), receiving from (r-i).
  for (int i = 1; i < P; i++) {
    int src = (r - i + P) % P;
    outBuf->recv(src, slot, outOffs[src] * es, opts.outCounts[src] * es);
  }
  for (int i = 1; i < P; i++) {
    int dst = (r + i) % P;
    inBuf->send(dst, slot, inOffs[dst] * es, opts.inCounts[dst] * es);
  }
  for (int i = 1; i < P; i++) {
    outBuf->waitRecv(timeout);
  }
  for (int i = 1; i < P; i++) {
    inBuf->waitSend(timeout);
  }
}

void alltoall(AlltoallOptions& opts) {
  AlltoallvOptions v(opts.context);
  v.tag = opts.tag;
  v.timeout = opts.timeout;
  v.input = opts.input;
  v.output = opts.output;
  v.elementSize = opts.elementSize;
  v.inCounts.assign(opts.context->size, opts.perRankElements);
  v.outCounts.assign(opts.context->size, opts.perRankElements);
  alltoallv(v);
}

// ---------------------------------------------------------------------------
// barrier — dissemination
// ---------------------------------------------------------------------------

void barrier(BarrierOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  if (P == 1) {
    return;
  }
  char byte = 1;
  auto buf = ctx->createUnboundBuffer(&byte, 1);
  const uint64_t slot = Slot::build(SlotPrefix::kBarrier, opts.tag);
  const auto timeout = resolveTimeout(opts);
  int round = 0;
  for (int d = 1; d < P; d <<= 1, round++) {
    buf->recv((r - d + 2 * P) % P, slot + round, 0, 0);
    buf->send((r + d) % P, slot + round, 0, 0);
    buf->waitRecv(timeout);
    buf->waitSend(timeout);
  }
}

// ---------------------------------------------------------------------------
// broadcast — binomial tree on virtual ranks (root -> vrank 0)
// ---------------------------------------------------------------------------

void broadcast(BroadcastOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t nbytes = opts.elements * opts.elementSize;
  char* out = static_cast<char*>(opts.output);
  GA_ENFORCE(out != nullptr);

  if (r == opts.root && opts.input != nullptr && opts.input != out) {
    std::memcpy(out, opts.input, nbytes);
  }
  if (P == 1) {
    return;
  }

  auto buf = ctx->createUnboundBuffer(out, nbytes);
  const uint64_t slot = Slot::build(SlotPrefix::kBroadcast, opts.tag);
  const auto timeout = resolveTimeout(opts);
  const int vr = (r - opts.root + P) % P;
  auto real = [&](int v) { return (v + opts.root) % P; };

  const uint32_t rounds = log2ceil(P);
  // Round k (0-based): vranks < 2^k hold the data and send to vr + 2^k.
  bool have = (vr == 0);
  for (uint32_t k = 0; k < rounds; k++) {
    const int half = 1 << k;
    if (vr < half) {
      if (vr + half < P) {
        buf->send(real(vr + half), slot + k, 0, nbytes);
        buf->waitSend(timeout);
      }
    } else if (vr < 2 * half) {
      GA_ENFORCE(!have);
      buf->recv(real(vr - half), slot + k, 0, nbytes);
      buf->waitRecv(timeout);
      have = true;
    }
  }
}

// ---------------------------------------------------------------------------
// gather / gatherv / scatter — direct to/from root
// ---------------------------------------------------------------------------

void gatherv(GathervOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t es = opts.elementSize;
  GA_ENFORCE_EQ(opts.counts.size(), static_cast<size_t>(P));
  const uint64_t slot = Slot::build(SlotPrefix::kGather, opts.tag);
  const auto timeout = resolveTimeout(opts);

  std::vector<size_t> offs(P + 1, 0);
  for (int i = 0; i < P; i++) {
    offs[i + 1] = offs[i] + opts.counts[i];
  }

  if (r == opts.root) {
    char* out = static_cast<char*>(opts.output);
    if (opts.input != nullptr) {
      std::memcpy(out + offs[r] * es, opts.input, opts.counts[r] * es);
    }
    auto outBuf = ctx->createUnboundBuffer(out, offs[P] * es);
    for (int i = 0; i < P; i++) {
      if (i == r) {
        continue;
      }
      outBuf->recv(i, slot, offs[i] * es, opts.counts[i] * es);
    }
    for (int i = 0; i < P - 1; i++) {
      outBuf->waitRecv(timeout);
    }
  } else {
    auto inBuf = ctx->createUnboundBuffer(opts.input, opts.counts[r] * es);
    inBuf->send(opts.root, slot, 0, opts.counts[r] * es);
    inBuf->waitSend(timeout);
  }
}

void gather(GatherOptions& opts) {
  GathervOptions v(opts.context);
  v.tag = opts.tag;
  v.timeout = opts.timeout;
  v.input = opts.input;
  v.output = opts.output;
  v.elementSize = opts.elementSize;
  v.root = opts.root;
  v.counts.assign(opts.context->size, opts.inElements);
  gatherv(v);
}

void scatter(ScatterOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t es = opts.elementSize;
  const size_t nb = opts.outElements * es;
  const uint64_t slot = Slot::build(SlotPrefix::kScatter, opts.tag);
  const auto timeout = resolveTimeout(opts);

  if (r == opts.root) {
    char* in = static_cast<char*>(opts.input);
    std::memcpy(opts.output, in + static_cast<size_t>(r) * nb, nb);
    auto inBuf = ctx->createUnboundBuffer(in, static_cast<size_t>(P) * nb);
    for (int i = 0; i < P; i++) {
      if (i == r) {
        continue;
      }
      inBuf->send(i, slot, static_cast<size_t>(i) * nb, nb);
    }
    for (int i = 0; i < P - 1; i++) {
      inBuf->waitSend(timeout);
    }
  } else {
    auto outBuf = ctx->createUnboundBuffer(opts.output, nb);
    outBuf->recv(opts.root, slot, 0, nb);
    outBuf->waitRecv(timeout);
  }
}

// ---------------------------------------------------------------------------
// reduce — ring reduce-scatter on a scratch copy + block gather to root
// ---------------------------------------------------------------------------

void reduce(ReduceOptions& opts) {
  auto& ctx = opts.context;
  const int P = ctx->size;
  const int r = ctx->rank;
  const size_t N = opts.elements;
  const size_t es = opts.elementSize;
  GA_ENFORCE(opts.reduce != nullptr);
  const uint64_t slot = Slot::build(SlotPrefix::kReduce, opts.tag);
  const auto timeout = resolveTimeout(opts);

  const char* in = static_cast<const char*>(
      opts.input != nullptr ? opts.input : opts.output);
  GA_ENFORCE(in != nullptr);

  if (P == 1) {
    if (opts.output != in) {
      std::memcpy(opts.output, in, N * es);
    }
    return;
  }

  // Scratch so the caller's input survives.
  auto scratch = makeAligned(N * es + 64);
  std::memcpy(scratch.get(), in, N * es);
  char* buf = scratch.get();

  // Block layout with tail clamp.
  const size_t perRank = (N + P - 1) / P;
  auto blockOff = [&](int b) {
    return std::min(static_cast<size_t>(b) * perRank, N);
  };
  auto blockLen = [&](int b) { return blockOff(b + 1) - blockOff(b); };

  auto bufUb = ctx->createUnboundBuffer(buf, N * es);
  auto tmp = makeAligned(2 * perRank * es + 64);
  auto tmpUb = ctx->createUnboundBuffer(tmp.get(), 2 * perRank * es + 64);
  const int sendRank = (r + 1) % P;
  const int recvRank = (r - 1 + P) % P;

  // Ring reduce-scatter: iteration i sends block (r-i), reduces (r-i-1);
  // rank r ends owning block (r+1) mod P.
  int outstanding = 0;
  for (int i = 0; i < std::min(2, P - 1); i++) {
    int b = (r - i - 1 + 2 * P) % P;
    tmpUb->recv(recvRank, slot, (i & 1) * perRank * es, blockLen(b) * es);
  }
  for (int i = 0; i < P - 1; i++) {
    int sb = (r - i + 2 * P) % P;
    int rb = (r - i - 1 + 2 * P) % P;
    if (outstanding >= 2) {
      bufUb->waitSend(timeout);
      outstanding--;
    }
    bufUb->send(sendRank, slot, blockOff(sb) * es, blockLen(sb) * es);
    outstanding++;
    tmpUb->waitRecv(timeout);
    if (blockLen(rb) > 0) {
      opts.reduce(
          buf + blockOff(rb) * es,
          buf + blockOff(rb) * es,
          tmp.get() + (i & 1) * perRank * es,
          blockLen(rb));
    }
    if (i + 2 < P - 1) {
      int b = (r - i - 3 + 2 * P) % P;
      tmpUb->recv(recvRank, slot, (i & 1) * perRank * es, blockLen(b) * es);
    }
  }
  while (outstanding-- > 0) {
    bufUb->waitSend(timeout);
  }

  // Gather the owned blocks at the root.
  const int owned = (r + 1) % P;
  if (r == opts.root) {
    char* out = static_cast<char*>(opts.output);
    std::memcpy(out + blockOff(owned) * es, buf + blockOff(owned) * es,
                blockLen(owned) * es);
    auto outUb = ctx->createUnboundBuffer(out, N * es);
    for (int i = 0; i < P; i++) {
      if (i == r) {
        continue;
      }
      int b = (i + 1) % P;
      outUb->recv(i, slot + 1000, blockOff(b) * es, blockLen(b) * es);
    }
    for (int i = 0; i < P - 1; i++) {
      outUb->waitRecv(timeout);
    }
  } else {
    bufUb->send(opts.root, slot + 1000, blockOff(owned) * es,
                blockLen(owned) * es);
    bufUb->waitSend(timeout);
  }
}

} // namespace glooamd
