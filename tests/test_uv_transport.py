"""libuv-driven transport (reference gloo/transport/uv parity: same
wire protocol over a libuv event loop)."""
import numpy as np

import gloo_amd as ga


def spawn_uv(size, fn):
    import threading

    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            dev = ga.create_tcp_device(use_libuv=True)
            ctx = ga.Context(rank, size)
            ctx.connect_full_mesh(store, dev)
            fn(ctx, rank, size)
            try:
                ga.barrier(ctx, tag=0xFFFF2)
            except ga.GlooAmdError:
                pass
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,)) for r in range(size)]
    [t.start() for t in ths]
    [t.join(60) for t in ths]
    assert not errors, errors[0]


def test_uv_allreduce():
    def fn(ctx, rank, size):
        x = np.arange(5000, dtype=np.float32) + rank
        ga.allreduce(ctx, [x.ctypes.data], x.size)
        expected = sum(np.arange(5000, dtype=np.float32) + r
                       for r in range(size))
        assert np.allclose(x, expected)

    spawn_uv(3, fn)


def test_uv_big_transfer():
    def fn(ctx, rank, size):
        n = 2_000_000
        if rank == 0:
            data = np.arange(n, dtype=np.float32)
            ub = ctx.create_unbound_buffer(data.ctypes.data, data.nbytes)
            ub.send(1, slot=4)
            ub.wait_send()
        else:
            out = np.zeros(n, dtype=np.float32)
            ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
            ub.recv(0, slot=4)
            ub.wait_recv()
            assert np.array_equal(out, np.arange(n, dtype=np.float32))

    spawn_uv(2, fn)


def test_uv_legacy_algorithm():
    def fn(ctx, rank, size):
        x = np.full(1000, float(rank + 1), dtype=np.float32)
        algo = ga._C.create_algorithm("allreduce_ring_chunked", ctx,
                                      [x.ctypes.data], 1000)
        algo.run()
        assert np.allclose(x, 1.0 + 2.0 + 3.0)

    spawn_uv(3, fn)


def test_uv_recv_from_any():
    def fn(ctx, rank, size):
        n = 16
        if rank == 0:
            out = np.zeros(n, dtype=np.float32)
            seen = []
            for _ in range(size - 1):
                ub = ctx.create_unbound_buffer(out.ctypes.data, out.nbytes)
                ub.recv_any(list(range(1, size)), slot=6)
                ok, src = ub.wait_recv()
                assert ok and np.all(out == src)
                seen.append(src)
            assert sorted(seen) == [1, 2]
        else:
            buf = np.full(n, rank, dtype=np.float32)
            ub = ctx.create_unbound_buffer(buf.ctypes.data, buf.nbytes)
            ub.send(0, slot=6)
            ub.wait_send()

    spawn_uv(3, fn)
