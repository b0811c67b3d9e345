"""Randomized transport protocol fuzz: random message sizes straddling
the eager threshold, random slot reuse, random post orderings and
delays across 3 ranks. Per-(pair, slot) FIFO and payload integrity are
asserted for every message. Seeded for reproducibility; set
GLOO_AMD_FUZZ_SEED / GLOO_AMD_FUZZ_MSGS to explore more."""
import os
import random
import time

import numpy as np
import pytest

import gloo_amd as ga

SIZES = [0, 1, 3, 64, 1000, 2047, 8191, 8192, 8193, 16384, 65536]
SLOTS = [1, 2, 3]


def _payload(msg_id, n):
    return (np.arange(n, dtype=np.int64) * 7 + msg_id).astype(np.int32)


def _script(seed, size, n_msgs):
    """One global message list all ranks derive identically."""
    rng = random.Random(seed)
    msgs = []
    for i in range(n_msgs):
        src = rng.randrange(size)
        dst = rng.choice([r for r in range(size) if r != src])
        msgs.append({
            "id": i,
            "src": src,
            "dst": dst,
            "slot": rng.choice(SLOTS),
            "n": rng.choice(SIZES),
        })
    return msgs


@pytest.mark.parametrize("seed", [11, 23])
def test_protocol_fuzz(spawn_threads, seed):
    seed = int(os.environ.get("GLOO_AMD_FUZZ_SEED", seed))
    n_msgs = int(os.environ.get("GLOO_AMD_FUZZ_MSGS", 120))
    size = 3
    msgs = _script(seed, size, n_msgs)

    def fn(ctx, rank, _):
        rng = random.Random(seed * 1000 + rank)
        sends = [m for m in msgs if m["src"] == rank]
        recvs = [m for m in msgs if m["dst"] == rank]
        # Keep per-(src, slot) FIFO for recvs and per-(dst, slot) FIFO
        # for sends (the protocol's ordering unit), but shuffle across
        # different queues to randomize interleaving.
        rng.shuffle(sends)
        order = {}
        for m in msgs:
            order.setdefault((m["src"], m["dst"], m["slot"]), []).append(m)
        for q in order.values():
            q.sort(key=lambda m: m["id"])
        sends_sorted = []
        seen = {}
        for m in sends:
            k = (m["src"], m["dst"], m["slot"])
            idx = seen.get(k, 0)
            seen[k] = idx + 1
            sends_sorted.append(order[k][idx])
        rng2 = random.Random(seed * 77 + rank)
        recvs_shuffled = [m for m in recvs]
        rng2.shuffle(recvs_shuffled)
        recvs_sorted = []
        seen = {}
        for m in recvs_shuffled:
            k = (m["src"], m["dst"], m["slot"])
            idx = seen.get(k, 0)
            seen[k] = idx + 1
            recvs_sorted.append(order[k][idx])

        # Interleave send and recv work on one thread with random delays.
        pend = []  # (ubuf, np_buf, msg, is_recv)
        si, ri = 0, 0
        while si < len(sends_sorted) or ri < len(recvs_sorted):
            do_send = si < len(sends_sorted) and (
                ri >= len(recvs_sorted) or rng.random() < 0.5)
            if do_send:
                m = sends_sorted[si]
                si += 1
                buf = _payload(m["id"], m["n"])
                ub = ctx.create_unbound_buffer(
                    buf.ctypes.data if m["n"] else 0, buf.nbytes)
                ub.send(m["dst"], slot=m["slot"])
                pend.append((ub, buf, m, False))
            else:
                m = recvs_sorted[ri]
                ri += 1
                out = np.full(m["n"], -1, dtype=np.int32)
                ub = ctx.create_unbound_buffer(
                    out.ctypes.data if m["n"] else 0, out.nbytes)
                ub.recv(m["src"], slot=m["slot"])
                pend.append((ub, out, m, True))
            if rng.random() < 0.1:
                time.sleep(rng.random() * 0.002)
        for ub, buf, m, is_recv in pend:
            if is_recv:
                ok, src = ub.wait_recv()
                assert ok and src == m["src"], m
                assert np.array_equal(buf, _payload(m["id"], m["n"])), m
            else:
                ub.wait_send()
        pend.clear()
        return True

    spawn_threads(size, fn)
