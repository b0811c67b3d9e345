"""TCP transport tests: tagged send/recv, recv-from-any, bound buffers,
timeouts, aborts (reference parity: gloo/test/send_recv_test.cc,
transport_test.cc, buffer_test.cc)."""
import numpy as np
import pytest

import gloo_amd as ga


def test_send_recv_basic(spawn_threads):
    def fn(ctx, rank, size):
        n = 1000
        buf = np.full(n, rank, dtype=np.float32)
        ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
        peer = (rank + 1) % size
        src = (rank - 1) % size
        out = np.zeros(n, dtype=np.float32)
        ub_out = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
        ub_out.recv(src, slot=7)
        ub.send(peer, slot=7)
        ok, got_src = ub_out.wait_recv()
        assert ok and got_src == src
        ub.wait_send()
        assert np.all(out == src)
        return True

    assert all(spawn_threads(4, fn))


def test_send_recv_offsets(spawn_threads):
    def fn(ctx, rank, size):
        buf = np.arange(100, dtype=np.int64) * (rank + 1)
        ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
        if rank == 0:
            ub.send(1, slot=3, offset=80, nbytes=160)  # elements 10..30
        elif rank == 1:
            out = np.zeros(100, dtype=np.int64)
            ub2 = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
            ub2.recv(0, slot=3, offset=0, nbytes=160)
            ub2.wait_recv()
            assert np.all(out[:20] == np.arange(10, 30, dtype=np.int64))
        if rank == 0:
            ub.wait_send()
        return True

    spawn_threads(2, fn)


def test_send_to_self(spawn_threads):
    def fn(ctx, rank, size):
        a = np.arange(50, dtype=np.float64)
        b = np.zeros(50, dtype=np.float64)
        ua = ctx.create_unbound_buffer(a.ctypes.data, a.nbytes)
        ub = ctx.create_unbound_buffer(b.ctypes.data, b.nbytes)
        ua.send(rank, slot=11)
        ub.recv(rank, slot=11)
        ub.wait_recv()
        ua.wait_send()
        assert np.all(b == a)
        return True

    spawn_threads(2, fn)


def test_recv_from_any(spawn_threads):
    """Rank 0 posts recv-from-any; every other rank sends once."""

    def fn(ctx, rank, size):
        n = 16
        if rank == 0:
            out = np.zeros(n, dtype=np.float32)
            seen = []
            for _ in range(size - 1):
                ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
                ub.recv_any(list(range(1, size)), slot=21)
                ok, src = ub.wait_recv()
                assert ok
                assert np.all(out == src)
                seen.append(src)
            assert sorted(seen) == list(range(1, size))
        else:
            buf = np.full(n, rank, dtype=np.float32)
            ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
            ub.send(0, slot=21)
            ub.wait_send()
        return True

    spawn_threads(4, fn)


def test_recv_from_any_subset_ordering(spawn_threads):
    """Sends race ahead of the posted any-recv (SEND_READY tally path)."""

    def fn(ctx, rank, size):
        import time

        n = 8
        if rank == 0:
            time.sleep(0.3)  # let SEND_READYs arrive first
            out = np.zeros(n, dtype=np.float32)
            got = []
            for _ in range(2):
                ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
                ub.recv_any([1, 2], slot=33)
                ok, src = ub.wait_recv()
                assert ok
                got.append(src)
            assert sorted(got) == [1, 2]
        elif rank in (1, 2):
            buf = np.full(n, rank, dtype=np.float32)
            ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
            ub.send(0, slot=33)
            ub.wait_send()
        return True

    spawn_threads(3, fn)


def test_mixed_targeted_and_any_recv(spawn_threads):
    """A targeted recv must not steal/strand tally entries for any-recv."""

    def fn(ctx, rank, size):
        import time

        n = 4
        if rank == 0:
            time.sleep(0.2)
            a = np.zeros(n, dtype=np.float32)
            b = np.zeros(n, dtype=np.float32)
            ua = ctx.create_unbound_buffer(a.ctypes.data, a.nbytes)
            ua.recv(1, slot=5)  # targeted at rank 1's first send
            ua.wait_recv()
            assert np.all(a == 1)
            ub = ctx.create_unbound_buffer(b.ctypes.data, b.nbytes)
            ub.recv_any([1, 2], slot=5)
            ok, src = ub.wait_recv()
            assert ok and src == 2 and np.all(b == 2)
        elif rank == 1:
            buf = np.full(n, 1, dtype=np.float32)
            ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
            ub.send(0, slot=5)
            ub.wait_send()
        elif rank == 2:
            buf = np.full(n, 2, dtype=np.float32)
            ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
            ub.send(0, slot=5)
            ub.wait_send()
        return True

    spawn_threads(3, fn)


def test_many_concurrent_slots(spawn_threads):
    def fn(ctx, rank, size):
        n = 64
        bufs = []
        peer = (rank + 1) % size
        src = (rank - 1) % size
        for s in range(32):
            out = np.zeros(n, dtype=np.int32)
            ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
            ub.recv(src, slot=100 + s)
            bufs.append((ub, out))
        sends = []
        for s in range(32):
            data = np.full(n, s, dtype=np.int32)
            ub = ctx.create_unbound_buffer(data.ctypes.data, data.nbytes)
            ub.send(peer, slot=100 + s)
            sends.append((ub, data))
        for s, (ub, out) in enumerate(bufs):
            ub.wait_recv()
            assert np.all(out == s)
        for ub, _ in sends:
            ub.wait_send()
        return True

    spawn_threads(3, fn)


def test_bound_buffers(spawn_threads):
    """Legacy one-sided-write style buffers with remote offset."""

    def fn(ctx, rank, size):
        n = 100
        src = np.arange(n, dtype=np.float32) + rank * 1000
        dst = np.zeros(n, dtype=np.float32)
        peer = (rank + 1) % size
        left = (rank - 1) % size
        sb = ctx.get_pair(peer).create_send_buffer(42, src.ctypes.data, src.nbytes)
        rb = ctx.get_pair(left).create_recv_buffer(42, dst.ctypes.data, dst.nbytes)
        # send elements [10, 35) to remote offset 40 bytes (element 10)
        sb.send(offset=40, length=100, roffset=40)
        rb.wait_recv()
        sb.wait_send()
        assert np.all(dst[10:35] == np.arange(10, 35, dtype=np.float32) + left * 1000)
        assert np.all(dst[:10] == 0) and np.all(dst[35:] == 0)
        return True

    spawn_threads(3, fn)


def test_bound_buffer_early_data(spawn_threads):
    """One-sided write that lands before createRecvBuffer is called."""

    def fn(ctx, rank, size):
        import time

        n = 16
        if rank == 0:
            src = np.full(n, 7.0, dtype=np.float32)
            sb = ctx.get_pair(1).create_send_buffer(9, src.ctypes.data, src.nbytes)
            sb.send()
            sb.wait_send()
        else:
            time.sleep(0.3)  # data arrives before registration
            dst = np.zeros(n, dtype=np.float32)
            rb = ctx.get_pair(0).create_recv_buffer(9, dst.ctypes.data, dst.nbytes)
            rb.wait_recv()
            assert np.all(dst == 7.0)
        return True

    spawn_threads(2, fn)


def test_recv_timeout(spawn_threads):
    def fn(ctx, rank, size):
        import time

        buf = np.zeros(4, dtype=np.float32)
        ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
        if rank == 0:
            ub.recv(1, slot=77)
            with pytest.raises(ga.TimeoutError):
                ub.wait_recv(timeout_ms=200)
        else:
            time.sleep(1.0)  # stay alive so EOF doesn't beat the timeout
        return True

    spawn_threads(2, fn)


def test_abort_wait(spawn_threads):
    def fn(ctx, rank, size):
        import threading as th
        import time

        buf = np.zeros(4, dtype=np.float32)
        ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
        if rank == 0:
            ub.recv(1, slot=78)
            t = th.Timer(0.2, ub.abort_wait_recv)
            t.start()
            ok, _ = ub.wait_recv(timeout_ms=10000)
            assert not ok  # aborted
            t.join()
        else:
            time.sleep(1.0)  # keep the pair open while rank 0 waits
        return True

    spawn_threads(2, fn)


def test_big_transfer(spawn_threads):
    """Multi-MB payload exercises partial read/write state machines."""

    def fn(ctx, rank, size):
        n = 3_000_000
        if rank == 0:
            data = np.arange(n, dtype=np.float32)
            ub = ctx.create_unbound_buffer(data.ctypes.data, data.nbytes)
            ub.send(1, slot=88)
            ub.wait_send()
        else:
            out = np.zeros(n, dtype=np.float32)
            ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
            ub.recv(0, slot=88)
            ub.wait_recv()
            assert np.array_equal(out, np.arange(n, dtype=np.float32))
        return True

    spawn_threads(2, fn)


def _eager_enabled():
    import os
    v = os.environ.get("GLOO_AMD_EAGER_MAX")
    return v is None or int(v) > 0


def test_eager_sends_before_recvs(spawn_threads):
    """Small sends complete and arrive before any recv is posted; the
    receiver's stash must deliver them in FIFO order."""
    if not _eager_enabled():
        pytest.skip("wait_send-before-recv requires eager sends")
    import threading as th

    barrier = th.Barrier(2)

    def fn(ctx, rank, size):
        n_msgs = 16
        if rank == 0:
            bufs = [np.full(64, i, dtype=np.float32) for i in range(n_msgs)]
            ubs = [ctx.create_unbound_buffer(b.ctypes.data, b.nbytes)
                   for b in bufs]
            for i, ub in enumerate(ubs):
                ub.send(1, slot=21)
            for ub in ubs:
                ub.wait_send()
            barrier.wait()  # receiver posts recvs only after all sends done
        else:
            barrier.wait()
            for i in range(n_msgs):
                out = np.zeros(64, dtype=np.float32)
                ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
                ub.recv(0, slot=21)
                ub.wait_recv()
                assert np.all(out == i), (i, out[0])
        return True

    spawn_threads(2, fn)


def test_eager_mixed_sizes_fifo(spawn_threads):
    """Small (eager-eligible) and large (rendezvous) sends interleaved on
    one slot must be received in send order."""

    def fn(ctx, rank, size):
        sizes = [16, 100_000, 32, 200_000, 8, 64]  # elements (f32)
        if rank == 0:
            bufs = [np.full(s, i + 1.0, dtype=np.float32)
                    for i, s in enumerate(sizes)]
            ubs = [ctx.create_unbound_buffer(b.ctypes.data, b.nbytes)
                   for b in bufs]
            for ub in ubs:
                ub.send(1, slot=22)
            for ub in ubs:
                ub.wait_send()
        else:
            for i, s in enumerate(sizes):
                out = np.zeros(s, dtype=np.float32)
                ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
                ub.recv(0, slot=22)
                ub.wait_recv()
                assert np.all(out == i + 1.0), (i, out[:4])
        return True

    spawn_threads(2, fn)


def test_eager_recv_from_any_stashed(spawn_threads):
    """recv-from-any must find payloads that arrived eagerly before the
    any-recv was registered."""
    import time

    def fn(ctx, rank, size):
        if rank in (0, 1):
            val = np.full(32, rank + 5.0, dtype=np.float32)
            ub = ctx.create_unbound_buffer(val.ctypes.data, val.nbytes)
            ub.send(2, slot=23)
            ub.wait_send()
        else:
            time.sleep(0.05)  # let both eager payloads arrive and stash
            seen = set()
            for _ in range(2):
                out = np.zeros(32, dtype=np.float32)
                ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
                ub.recv_any([0, 1], slot=23)
                ok, src = ub.wait_recv()
                assert ok and np.all(out == src + 5.0)
                seen.add(src)
            assert seen == {0, 1}
        return True

    spawn_threads(3, fn)


def test_eager_zero_length(spawn_threads):
    """Zero-byte eager sends (pure notifications) before and after the
    recv post."""
    if not _eager_enabled():
        pytest.skip("wait_send-before-recv requires eager sends")
    import threading as th

    barrier = th.Barrier(2)

    def fn(ctx, rank, size):
        b = np.zeros(1, dtype=np.float32)
        ub = ctx.create_unbound_buffer(b.ctypes.data, b.nbytes)
        if rank == 0:
            ub.send(1, slot=24, nbytes=0)
            ub.wait_send()
            barrier.wait()
        else:
            barrier.wait()  # payload already stashed
            ub.recv(0, slot=24, nbytes=0)
            ok, src = ub.wait_recv()
            assert ok and src == 0
        return True

    spawn_threads(2, fn)


def test_connect_rendezvous_timeout():
    """Bootstrap with an absent peer fails within the context timeout
    (reference tcp_test.cc connect-timeout unit)."""
    import time

    store = ga.HashStore()
    dev = ga.create_tcp_device()
    ctx = ga.Context(0, 2)
    ctx.set_timeout(500)
    t0 = time.monotonic()
    with pytest.raises(ga.GlooAmdError):
        ctx.connect_full_mesh(store, dev)
    assert time.monotonic() - t0 < 10.0


def test_unbound_buffer_lifetime(spawn_threads):
    """Destroying an unbound buffer with an unmatched posted op must
    drop it cleanly (reference memory_test.cc semantics)."""

    def fn(ctx, rank, size):
        n = 256
        # recv that will never match: destroyed before any send exists
        orphan = np.zeros(n, dtype=np.float32)
        ub = ctx.create_unbound_buffer(orphan.ctypes.data, orphan.nbytes)
        ub.recv((rank + 1) % size, slot=77)
        del ub  # queued op dropped; no crash, no spurious delivery later
        # the slot remains usable for fresh matched traffic
        x = np.full(n, float(rank), dtype=np.float32)
        y = np.zeros(n, dtype=np.float32)
        us = ctx.create_unbound_buffer(x.ctypes.data, x.nbytes)
        ur = ctx.create_unbound_buffer(y.ctypes.data, y.nbytes)
        ur.recv((rank + 1) % size, slot=78)
        us.send((rank + 1) % size, slot=78)
        ur.wait_recv()
        us.wait_send()
        assert np.all(y == float((rank + 1) % size))
        return True

    spawn_threads(2, fn)


def test_concurrent_fat_sends_all_pairs(spawn_threads):
    """Exercise the unlocked-writev tx drain (r02): every rank pushes an
    8MB payload to every peer CONCURRENTLY from separate threads while
    receiving, so several pairs of one context flush at once. With the
    old single-lock flush this serialized; either way the data must
    land intact (per-pair tx-busy drain, deque-stable TxOp fronts)."""
    import threading

    def fn(ctx, rank, size):
        n = 2_000_000  # 8 MB fp32
        sends = []
        recvs = []
        for peer in range(size):
            if peer == rank:
                continue
            x = np.full(n, float(rank * 10 + peer), dtype=np.float32)
            y = np.zeros(n, dtype=np.float32)
            us = ctx.create_unbound_buffer(x.ctypes.data, x.nbytes)
            ur = ctx.create_unbound_buffer(y.ctypes.data, y.nbytes)
            ur.recv(peer, slot=500 + rank)  # slot keyed by SENDER
            sends.append((us, peer, x))
            recvs.append((ur, peer, y))
        # concurrent posts from separate threads -> tx-busy contention
        ths = []
        for us, peer, _x in sends:
            ths.append(threading.Thread(
                target=lambda us=us, peer=peer: us.send(
                    peer, 500 + peer)))
        for t in ths:
            t.start()
        for t in ths:
            t.join()
        for ur, peer, y in recvs:
            ur.wait_recv()
            assert np.all(y == float(peer * 10 + rank)), (rank, peer)
        for us, _p, _x in sends:
            us.wait_send()
        return True

    spawn_threads(3, fn)
