"""Host-side model checking of the device ring allreduce's doorbell
protocol (csrc/hip/algorithms.cc enqueueRing/runDeviceGraph).

The simulator mirrors the enqueue logic op for op: per (rank, ring) a
copy stream (ks) and a compute stream (cs); ops execute in stream
order; a wait blocks its stream until the flag value arrives. ALL runs
are enqueued up front with a per-rank run barrier between them (the
real host syncs a rank's streams between runs, but OTHER ranks skew
arbitrarily), then executed under several adversarial stream orders.

Checks:
  1. progress — every op executes under every探 explored schedule (no
     deadlock in the dependency graph);
  2. overwrite safety — a `put` into a peer inbox slot asserts the
     previous token was consumed (cross-run lastAck gates / graph-mode
     crossRunOff gates are exactly what make this true under rank skew);
  3. token integrity — each reduce consumes exactly the expected seq.
"""
import math

import pytest


def align_up(v, a):
    return v if a <= 1 else ((v + a - 1) // a) * a


def subspan_of_a(off, length, j, parts, a):
    per = align_up((length + parts - 1) // parts, a)
    start = min(off + j * per, off + length)
    end = min(start + per, off + length)
    return start, end - start


def ring_segments(part_len, es, P, R, inbox_cap, chunked=True):
    A = max(1, 16 // es)
    sub_cap = inbox_cap // R
    per_rank = align_up((part_len + P - 1) // P, A)
    S = max(2 if chunked else 1, (per_rank * es + sub_cap - 1) // sub_cap)
    while align_up((per_rank + S - 1) // S, A) * es > sub_cap:
        S += 1
    return S


class Sim:
    def __init__(self):
        self.flags = {}
        self.streams = {}
        self.inbox = {}  # (owner, ring, par) -> [token, consumed]

    def stream(self, key):
        return self.streams.setdefault(key, [])

    def step_stream(self, ops):
        """Execute ops from the front until blocked; return ops run."""
        ran = 0
        while ops:
            op = ops[0]
            kind = op[0]
            if kind == "wait":
                _, owner, fid, val = op
                if self.flags.get((owner, fid), 0) < val:
                    break
            elif kind == "write":
                _, owner, fid, val = op
                self.flags[(owner, fid)] = max(
                    self.flags.get((owner, fid), 0), val)
            elif kind == "put":
                _, owner, ring, par, token = op
                slot = self.inbox.get((owner, ring, par))
                assert slot is None or slot[1], (
                    f"OVERWRITE: rank {owner} ring {ring} par {par} "
                    f"token {slot and slot[0]} not consumed before "
                    f"token {token} landed")
                self.inbox[(owner, ring, par)] = [token, False]
            elif kind == "consume":
                _, owner, ring, par, expect = op
                slot = self.inbox.get((owner, ring, par))
                assert slot is not None and slot[0] == expect, (
                    f"rank {owner} ring {ring} par {par}: got "
                    f"{slot and slot[0]}, expected {expect}")
                slot[1] = True
            else:
                raise AssertionError(kind)
            ops.pop(0)
            ran += 1
        return ran

    def run_to_completion(self, order=None):
        keys = sorted(self.streams.keys())
        if order is not None:
            keys = sorted(keys, key=order)
        progressed = True
        while progressed:
            progressed = False
            for k in keys:
                if self.step_stream(self.streams[k]) > 0:
                    progressed = True
        left = {k: len(v) for k, v in self.streams.items() if v}
        assert not left, f"deadlock; blocked streams: {left}"


def enqueue_run(sim, P, strides, n, es, inbox_cap, seq_base, last_ack,
                run_idx, rel=False, ring_k=None, max_k=0):
    """Mirror of enqueueRing for every rank/ring + the per-rank run
    barrier the blocking run() provides."""
    R = len(strides)
    A = max(1, 16 // es)
    ks_of = {}
    for r in range(P):
        # per-rank run barrier: every stream of rank r waits until every
        # stream of rank r finished the previous run
        sids = [f"cs{j}" for j in range(R)] + [f"ks{j}" for j in range(R)]
        if run_idx > 0:
            for sid in sids:
                for sid2 in sids:
                    sim.stream((r, sid)).append(
                        ("wait", r, ("RB", sid2), run_idx))
        for j, stride in enumerate(strides):
            off, part_len = subspan_of_a(0, n, j, R, A)
            if part_len == 0:
                ks_of[(r, j)] = 0
                continue
            S = ring_segments(part_len, es, P, R, inbox_cap)
            K = 2 * (P - 1) * S
            ks_of[(r, j)] = K
            right = (r + stride) % P
            left = (r - stride + P) % P
            cs = sim.stream((r, f"cs{j}"))
            ks = sim.stream((r, f"ks{j}"))

            for k in range(K):
                par = k & 1
                seq = seq_base + k + 1
                if rel:
                    offv = (k - 1) if k >= 2 else (
                        (ring_k[j] - max_k) + k - 1)
                    gate = seq_base + offv
                    if gate > 0:
                        ks.append(("wait", r, ("ACK", j, par), gate))
                else:
                    prev = (seq_base + k - 1) if k >= 2 else \
                        last_ack[(r, j, par)]
                    if prev > 0:
                        ks.append(("wait", r, ("ACK", j, par), prev))
                ks.append(("put", right, j, par, seq))
                ks.append(("write", right, ("DATA", j, par), seq))
                cs.append(("wait", r, ("DATA", j, par), seq))
                cs.append(("consume", r, j, par, seq))
                cs.append(("write", left, ("ACK", j, par), seq))
        # publish this rank's run completion per stream
        for sid in sids:
            sim.stream((r, sid)).append(
                ("write", r, ("RB", sid), run_idx + 1))
    # eager/graph both maintain the same lastAck bookkeeping
    for r in range(P):
        for j in range(R):
            K = ks_of[(r, j)]
            for k in range(max(0, K - 2), K):
                last_ack[(r, j, k & 1)] = seq_base + k + 1
    return ks_of


ORDERS = [
    None,
    lambda k: (k[0], k[1]),            # rank-major: rank 0 races ahead
    lambda k: (-k[0], k[1]),           # last rank races ahead
    lambda k: (k[1], k[0]),            # all ks streams first
]


def run_model(P, R, n_list, es=4, inbox_cap=4 << 20, graph_after=None):
    strides = [st for st in range(1, max(P, 2))
               if math.gcd(st, P) == 1][:R]
    if not strides:
        strides = [1]
    for order in ORDERS:
        sim = Sim()
        seq_base = 0
        last_ack = {}
        R2 = len(strides)
        for r in range(P):
            for j in range(R2):
                for par in (0, 1):
                    last_ack[(r, j, par)] = 0
        prev_n = None
        for i, n in enumerate(n_list):
            rel = (graph_after is not None and i >= graph_after
                   and prev_n == n)
            ring_k, max_k = None, 0
            if rel:
                A = max(1, 16 // es)
                ring_k = []
                for j in range(R2):
                    _, plen = subspan_of_a(0, n, j, R2, A)
                    K = (2 * (P - 1) *
                         ring_segments(plen, es, P, R2, inbox_cap)
                         if plen else 0)
                    ring_k.append(K)
                max_k = max(ring_k)
            ks_of = enqueue_run(
                sim, P, strides, n, es, inbox_cap, seq_base, last_ack,
                run_idx=i, rel=rel, ring_k=ring_k, max_k=max_k)
            seq_base += max(ks_of.values()) if ks_of else 0
            prev_n = n
        sim.run_to_completion(order)


@pytest.mark.parametrize("P,R", [(2, 1), (3, 2), (4, 2), (8, 4)])
def test_ring_protocol_eager_multirun(P, R):
    """Four eager runs enqueued with arbitrary cross-rank skew: the
    cross-run inbox gates must prevent overwrites and deadlocks."""
    run_model(P, R, [1_000_000] * 4)


@pytest.mark.parametrize("P,R", [(2, 1), (4, 2), (8, 4)])
def test_ring_protocol_graph_relative(P, R):
    """Run 1 eager, runs 2-5 graph-relative gating (incl. the per-ring
    crossRunOff correction when a ring has fewer steps than the max)."""
    run_model(P, R, [999_999] * 5, graph_after=1)


@pytest.mark.parametrize("P,R", [(4, 2), (8, 4)])
def test_ring_protocol_mixed_shapes(P, R):
    """Shape switches force eager runs between graph replays; eager and
    graph bookkeeping must interleave exactly."""
    run_model(P, R, [1_000_000, 700_001, 700_001, 1_000_000, 1_000_000,
                     700_001, 700_001], graph_after=1)


def test_ring_protocol_small_inbox_deep_pipeline():
    """64 KiB inbox -> dozens of segments; deep double-buffer chains."""
    run_model(4, 2, [500_000] * 3, inbox_cap=64 * 1024)


def _run_with_cross_run_bias(bias):
    """3 runs, graph-relative from run 2, with the k<2 cross-run gate
    offset biased by `bias` steps (0 = correct protocol)."""
    P, es, cap = 4, 4, 4 << 20
    strides = [1, 3]
    n = 1_000_000
    R = len(strides)
    A = max(1, 16 // es)
    ring_k = []
    for j in range(R):
        _, plen = subspan_of_a(0, n, j, R, A)
        ring_k.append(2 * (P - 1) *
                      ring_segments(plen, es, P, R, cap) if plen else 0)
    max_k = max(ring_k)
    for order in ORDERS:
        sim = Sim()
        last_ack = {(r, j, p): 0 for r in range(P) for j in range(R)
                    for p in (0, 1)}
        seq = 0
        for i in range(3):
            ks_of = enqueue_run(
                sim, P, strides, n, es, cap, seq, last_ack, run_idx=i,
                rel=(i >= 1),
                ring_k=[k + bias for k in ring_k],  # biases the gate
                max_k=max_k)
            seq += max(ks_of.values())
        sim.run_to_completion(order)


def test_ring_protocol_detects_too_early_gate():
    """Negative control: a cross-run gate 2 steps too permissive (the
    class of bug the per-ring crossRunOff correction prevents when ring
    step counts differ) must trip the overwrite assertion under at
    least one rank-skew schedule."""
    _run_with_cross_run_bias(0)  # correct protocol passes
    with pytest.raises(AssertionError):
        _run_with_cross_run_bias(-2)


def test_ring_protocol_detects_too_strict_gate():
    """A gate 2 steps too strict waits for acks that never come:
    detected as a deadlock."""
    with pytest.raises(AssertionError, match="deadlock"):
        _run_with_cross_run_bias(+2)


# ---------------------------------------------------------------------------
# Halving-doubling model (csrc/hip/algorithms.cc HipAllreduceHalvingDoubling)
# incl. the non-pow2 fold lanes and the allgather mirror's direct writes
# into peer work regions.
# ---------------------------------------------------------------------------


class HdSim(Sim):
    def __init__(self):
        super().__init__()
        self.work = {}  # (owner, t) -> [token, consumed] mirror-landing

    def step_stream(self, ops):
        ran = 0
        while ops:
            op = ops[0]
            kind = op[0]
            if kind == "wait":
                _, owner, fid, val = op
                if self.flags.get((owner, fid), 0) < val:
                    break
            elif kind == "write":
                _, owner, fid, val = op
                self.flags[(owner, fid)] = max(
                    self.flags.get((owner, fid), 0), val)
            elif kind == "put":
                _, owner, ring, par, token = op
                slot = self.inbox.get((owner, ring, par))
                assert slot is None or slot[1], (
                    f"OVERWRITE: rank {owner} slot ({ring},{par}) token "
                    f"{slot and slot[0]} not consumed before {token}")
                self.inbox[(owner, ring, par)] = [token, False]
            elif kind == "consume":
                _, owner, ring, par, expect = op
                slot = self.inbox.get((owner, ring, par))
                assert slot is not None and slot[0] == expect, (
                    f"rank {owner} slot ({ring},{par}): got "
                    f"{slot and slot[0]}, expected {expect}")
                slot[1] = True
            elif kind == "put_work":
                _, owner, t, token = op
                slot = self.work.get((owner, t))
                assert slot is None or slot[1], (
                    f"WORK OVERWRITE at rank {owner} step {t}: token "
                    f"{slot and slot[0]} not staged out before {token}")
                self.work[(owner, t)] = [token, False]
            elif kind == "stage_out":
                _, owner, expects = op
                for t, token in expects:
                    slot = self.work.get((owner, t))
                    assert slot is not None and slot[0] == token, (
                        f"rank {owner} stage-out step {t}: got "
                        f"{slot and slot[0]}, expected {token}")
                    slot[1] = True
            else:
                raise AssertionError(kind)
            ops.pop(0)
            ran += 1
        return ran


def hd_chunks(max_half, es, sub_bytes):
    A = max(1, 16 // es)
    nc = max(1, (max_half * es + sub_bytes - 1) // sub_bytes)
    while align_up((max_half + nc - 1) // nc, A) * es > sub_bytes:
        nc += 1
    return nc


def enqueue_hd_run(sim, P, n, es, inbox_cap, state, run_idx,
                   drop_fack_gate=False):
    """Mirror of HipAllreduceHalvingDoubling::run for every rank:
    pow2 exchange + fold lanes for extras + mirror + stage-out."""
    pow2 = 1
    T = 0
    while pow2 * 2 <= P:
        pow2 *= 2
        T += 1
    T = max(1, T)
    extras = P - pow2
    rows = T + (1 if extras else 0)
    sub_bytes = ((2 * inbox_cap) // (2 * rows)) // 16 * 16
    A = max(1, 16 // es)
    fold_elems = sub_bytes // es
    fold_nc = (n + fold_elems - 1) // fold_elems

    for r in range(P):
        cs = sim.stream((r, "cs"))
        ks = sim.stream((r, "ks"))
        if run_idx > 0:  # per-rank run barrier (blocking run())
            for sid in ("cs", "ks"):
                for sid2 in ("cs", "ks"):
                    sim.stream((r, sid)).append(
                        ("wait", r, ("RB", sid2), run_idx))
        if r >= pow2:  # extra: fold lanes + post-fold
            partner = r - pow2
            for c in range(fold_nc):
                par = c & 1
                state["foldSeq", r] = state.get(("foldSeq", r), 0) + 1
                seq = state["foldSeq", r]
                last = state.get(("lastFold", r, par), 0)
                if last > 0 and not drop_fack_gate:
                    ks.append(("wait", r, ("FACK", par), last))
                ks.append(("put", partner, "FOLD", par, seq))
                ks.append(("write", partner, ("FOLD", par), seq))
                state["lastFold", r, par] = seq
            state["postSeq", r] = state.get(("postSeq", r), 0) + 1
            ps = state["postSeq", r]
            cs.append(("wait", r, ("POST",), ps))
            cs.append(("stage_out", r, [("post", ps)]))
            cs.append(("write", partner, ("PACK",), ps))
        else:
            span_len = n
            if r < extras:  # partner: fold reduces
                extra = r + pow2
                for c in range(fold_nc):
                    par = c & 1
                    state["foldSeq", r] = state.get(("foldSeq", r), 0) + 1
                    seq = state["foldSeq", r]
                    cs.append(("wait", r, ("FOLD", par), seq))
                    cs.append(("consume", r, "FOLD", par, seq))
                    cs.append(("write", extra, ("FACK", par), seq))
            # reduce-scatter (cs records a per-step event the mirror
            # waits on — the real evs[t] / evs[T-1]->streamWait(ksm))
            for t in range(T):
                peer_global = (r ^ (1 << t))
                half = align_up((span_len + 1) // 2, A)
                nc = hd_chunks(half, es, sub_bytes)
                for c in range(nc):
                    par = c & 1
                    state["seq", r] = state.get(("seq", r), 0) + 1
                    seq = state["seq", r]
                    last = state.get(("lastAck", r, t, par), 0)
                    if last > 0:
                        ks.append(("wait", r, ("ACK", t, par), last))
                    ks.append(("put", peer_global, t, par, seq))
                    ks.append(("write", peer_global, ("DATA", t, par), seq))
                    cs.append(("wait", r, ("DATA", t, par), seq))
                    cs.append(("consume", r, t, par, seq))
                    cs.append(("write", peer_global, ("ACK", t, par), seq))
                    state["lastAck", r, t, par] = seq
                cs.append(("write", r, ("EV", t), run_idx + 1))
                span_len = half
            # allgather mirror: one direct write into peer work per step
            for t in range(T - 1, -1, -1):
                peer_global = (r ^ (1 << t))
                state["seq", r] = state.get(("seq", r), 0) + 1
                aseq = state["seq", r]
                if t == T - 1:
                    ks.append(("wait", r, ("EV", T - 1), run_idx + 1))
                else:
                    la = state.get(("lastAgd", r, t + 1), 0)
                    if la > 0:
                        ks.append(("wait", r, ("AGD", t + 1), la))
                ks.append(("put_work", peer_global, t, aseq))
                ks.append(("write", peer_global, ("AGD", t), aseq))
                state["lastAgd", r, t] = aseq
            # final: wait all AGD, stage out, post-fold push
            expects = []
            for t in range(T):
                la = state.get(("lastAgd", r, t), 0)
                cs.append(("wait", r, ("AGD", t), la))
                expects.append((t, la))
            cs.append(("stage_out", r, expects))
            if r < extras:
                extra = r + pow2
                state["postSeq", r] = state.get(("postSeq", r), 0) + 1
                ps = state["postSeq", r]
                if ps > 1:
                    cs.append(("wait", r, ("PACK",), ps - 1))
                cs.append(("put_work", extra, "post", ps))
                cs.append(("write", extra, ("POST",), ps))
        for sid in ("cs", "ks"):
            sim.stream((r, sid)).append(("write", r, ("RB", sid),
                                         run_idx + 1))


def run_hd_model(P, n_list, es=4, inbox_cap=4 << 20):
    for order in ORDERS:
        sim = HdSim()
        state = {}
        for i, n in enumerate(n_list):
            enqueue_hd_run(sim, P, n, es, inbox_cap, state, i)
        sim.run_to_completion(order)


@pytest.mark.parametrize("P", [2, 3, 4, 5, 6, 7, 8])
def test_hd_protocol_multirun(P):
    """Three HD runs under adversarial rank skew, incl. non-pow2 fold
    lanes, mirror work-region writes and post-fold handshakes."""
    run_hd_model(P, [1_500_000] * 3)


@pytest.mark.parametrize("P", [3, 6])
def test_hd_protocol_small_inbox(P):
    """Chunked fold + sub-chunked halves (deep fold/step pipelines)."""
    run_hd_model(P, [800_000] * 3, inbox_cap=256 * 1024)


def test_hd_protocol_detects_missing_fold_ack_gate():
    """Negative control: without the fFACK slot-reuse gate the extra
    rank's chunk c overwrites chunk c-2's fold slot before the partner
    consumed it (needs a chunked fold: small inbox)."""
    P, es, cap = 3, 4, 256 * 1024
    failed = False
    for order in ORDERS:
        sim = HdSim()
        state = {}
        for i in range(2):
            enqueue_hd_run(sim, P, 800_000, es, cap, state, i,
                           drop_fack_gate=True)
        try:
            sim.run_to_completion(order)
        except AssertionError:
            failed = True
    assert failed, "checker failed to catch the dropped fFACK gate"


# ---------------------------------------------------------------------------
# Direct (one-shot) allreduce model (HipAllreduceDirect): scatter /
# fused-reduce / broadcast chunk pipeline with fRS/fACK/fAG/fDONE and
# the cross-run fDONE gate protecting peer work regions.
# ---------------------------------------------------------------------------


def enqueue_direct_run(sim, P, C, state, run_idx, nf=7,
                       drop_ack_gate=False, drop_ce_gate=False):
    run_seq = run_idx + 1
    base = state.get("chunkSeqBase", 0)

    def cseq(c):
        return base + c + 1

    for r in range(P):
        sids = ["cs"] + [f"f{i}" for i in range(min(nf, P - 1))]
        if run_idx > 0:
            for sid in sids:
                for sid2 in sids:
                    sim.stream((r, sid)).append(
                        ("wait", r, ("RB", sid2), run_idx))
        cs = sim.stream((r, "cs"))
        fan = [sim.stream((r, f"f{i}"))
               for i in range(min(nf, P - 1))]
        # peers must have copied out the previous run before we write
        # their work regions again
        for j in range(1, P):
            d = (r + j) % P
            st = fan[(j - 1) % len(fan)]
            if run_seq > 1:
                st.append(("wait", r, ("DONE", d), run_seq - 1))
        for c in range(C):
            par = c & 1
            for j in range(1, P):
                d = (r + j) % P
                st = fan[(j - 1) % len(fan)]
                gate = (cseq(c - 2) if c >= 2
                        else state.get(("lastAck", r, d, par), 0))
                if gate > 0 and not drop_ack_gate:
                    st.append(("wait", r, ("ACK", d), gate))
                st.append(("put", d, ("in", r), par, cseq(c)))
                st.append(("write", d, ("RS", r), cseq(c)))
            for src in range(P):
                if src != r:
                    cs.append(("wait", r, ("RS", src), cseq(c)))
            for src in range(P):
                if src != r:
                    cs.append(("consume", r, ("in", src), par, cseq(c)))
            for src in range(P):
                if src != r:
                    cs.append(("write", src, ("ACK", r), cseq(c)))
            cs.append(("write", r, ("CE", c % 4), cseq(c)))
            for j in range(1, P):
                d = (r + j) % P
                st = fan[(j - 1) % len(fan)]
                if not drop_ce_gate:
                    st.append(("wait", r, ("CE", c % 4), cseq(c)))
                st.append(("put_work", d, (r, c), cseq(c)))
                st.append(("write", d, ("AG", r), cseq(c)))
        expects = []
        for src in range(P):
            if src != r:
                cs.append(("wait", r, ("AG", src), cseq(C - 1)))
                for c in range(C):
                    expects.append(((src, c), cseq(c)))
        cs.append(("stage_out", r, expects))
        for src in range(P):
            if src != r:
                cs.append(("write", src, ("DONE", r), run_seq))
        for sid in sids:
            sim.stream((r, sid)).append(("write", r, ("RB", sid),
                                         run_idx + 1))
    for r in range(P):
        for d in range(P):
            for c in range(max(0, C - 2), C):
                state["lastAck", r, d, c & 1] = cseq(c)
    state["chunkSeqBase"] = base + C


def run_direct_model(P, C, runs=3, nf=7, drop_ack_gate=False,
                     drop_ce_gate=False):
    for order in ORDERS:
        sim = HdSim()
        state = {}
        for i in range(runs):
            enqueue_direct_run(sim, P, C, state, i, nf=nf,
                               drop_ack_gate=drop_ack_gate,
                               drop_ce_gate=drop_ce_gate)
        sim.run_to_completion(order)


@pytest.mark.parametrize("P,C", [(2, 1), (2, 3), (4, 2), (8, 4), (8, 1)])
def test_direct_protocol_multirun(P, C):
    """Chunk-pipelined direct allreduce, 3 runs under rank skew:
    double-buffered inbox slots, chunkEvent broadcast gating, and the
    cross-run fDONE work-region gate."""
    run_direct_model(P, C)


def test_direct_protocol_fewer_fanout_streams():
    run_direct_model(8, 3, nf=3)


def test_direct_protocol_defense_in_depth_and_teeth():
    """Model FINDING: with the per-chunk interleaved enqueue (scatter,
    reduce-gated broadcast per fanout stream), the pipeline is
    self-throttling — dropping the fACK slot-reuse gate alone cannot
    produce an overwrite (the gate is defense in depth, load-bearing
    only if the enqueue shape changes). Dropping BOTH the chunkEvent
    broadcast gate and the ack gate breaks the throttle and the checker
    catches the overwrite — proving the model has teeth."""
    run_direct_model(4, 4, runs=2, drop_ack_gate=True)  # belt holds
    with pytest.raises(AssertionError):
        run_direct_model(4, 4, runs=2, drop_ack_gate=True,
                         drop_ce_gate=True)


# ---------------------------------------------------------------------------
# HipP2P lane model: per-(src->dst) double-buffered chunk lanes with
# post/flush split (posts enqueue, flush is the only sync point).
# ---------------------------------------------------------------------------


def enqueue_p2p_message(sim, state, src, dst, nchunks):
    ss = sim.stream((src, "ss"))
    rs = sim.stream((dst, "rs"))
    for _ in range(nchunks):
        state["s", src, dst] = state.get(("s", src, dst), 0) + 1
        seq = state["s", src, dst]
        par = seq & 1
        if seq > 2:
            ss.append(("wait", src, ("ACK", dst), seq - 2))
        ss.append(("put", dst, ("lane", src), par, seq))
        ss.append(("write", dst, ("DATA", src), seq))
    for _ in range(nchunks):
        state["r", dst, src] = state.get(("r", dst, src), 0) + 1
        seq = state["r", dst, src]
        par = seq & 1
        rs.append(("wait", dst, ("DATA", src), seq))
        rs.append(("consume", dst, ("lane", src), par, seq))
        rs.append(("write", src, ("ACK", dst), seq))


@pytest.mark.parametrize("nchunks", [1, 2, 5, 9])
def test_p2p_lane_protocol(nchunks):
    """Bidirectional multi-message exchange + a gather-shaped fan-in:
    lane seqs are per-direction, so concurrent traffic cannot cross."""
    for order in ORDERS:
        sim = Sim()
        state = {}
        # bidirectional pair traffic, several messages
        for _ in range(3):
            enqueue_p2p_message(sim, state, 0, 1, nchunks)
            enqueue_p2p_message(sim, state, 1, 0, nchunks)
        # gather fan-in to rank 0 from 3 senders
        for src in (1, 2, 3):
            enqueue_p2p_message(sim, state, src, 0, nchunks)
        sim.run_to_completion(order)


def test_p2p_lane_detects_missing_ack():
    """Negative control: without the seq-2 ack gate a >2-chunk message
    overwrites the un-consumed lane slot."""
    sim = Sim()
    state = {}
    ss = sim.stream((0, "ss"))
    rs = sim.stream((1, "rs"))
    for seq in (1, 2, 3):
        ss.append(("put", 1, ("lane", 0), seq & 1, seq))
        ss.append(("write", 1, ("DATA", 0), seq))
    for seq in (1, 2, 3):
        rs.append(("wait", 1, ("DATA", 0), seq))
        rs.append(("consume", 1, ("lane", 0), seq & 1, seq))
        rs.append(("write", 0, ("ACK", 1), seq))
    with pytest.raises(AssertionError):
        sim.run_to_completion(lambda k: 0 if k[1] == "ss" else 1)


def test_model_math_matches_cpp():
    """The model's split-math mirror must equal the C++ schedule.h
    aligned variants the engines actually use (keeps the model honest
    if the C++ math ever changes)."""
    import gloo_amd as ga

    for n in [1, 7, 999_999, 1_000_000, 5_000_000, 100_000_000]:
        for parts in [1, 2, 3, 4, 8]:
            for A in [1, 2, 4, 8]:
                for j in range(parts):
                    assert subspan_of_a(0, n, j, parts, A) == \
                        ga._C.subspan_of_a(0, n, j, parts, A), \
                        (n, parts, A, j)
