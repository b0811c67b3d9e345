"""v2 CPU collective tests with closed-form fixtures.

Mirrors the reference test strategy (gloo/test/allreduce_test.cc sweeps
over ranks x elements x algorithm; fixture pattern src[j] = j*stride+rank
so results are analytically checkable — gloo/test/base_test.h:194-262).
"""
import numpy as np
import pytest

import gloo_amd as ga


def fixture(rank, n, dtype=np.float32):
    return (np.arange(n, dtype=np.float64) * 1.0 + rank).astype(dtype)


@pytest.mark.parametrize("size", [1, 2, 3, 4])
@pytest.mark.parametrize("elements", [0, 1, 7, 1000, 100_000])
@pytest.mark.parametrize("algorithm", ["ring", "bcube"])
def test_allreduce_sum(spawn_threads, size, elements, algorithm):
    def fn(ctx, rank, _):
        x = fixture(rank, elements)
        ga.allreduce(
            ctx, [x.ctypes.data], x.size, ga.DType.f32, ga.ReduceOp.sum,
            algorithm=algorithm,
        )
        expected = sum(fixture(r, elements) for r in range(size))
        assert np.allclose(x, expected)
        return True

    assert all(spawn_threads(size, fn))


@pytest.mark.parametrize("size", [4])
def test_allreduce_bcube_power(spawn_threads, size):
    """size = base^k so the true bcube path runs."""

    def fn(ctx, rank, _):
        x = fixture(rank, 10_001)
        ga.allreduce(
            ctx, [x.ctypes.data], x.size, ga.DType.f32, ga.ReduceOp.sum,
            algorithm="bcube",
        )
        expected = sum(fixture(r, 10_001) for r in range(size))
        assert np.allclose(x, expected)
        return True

    spawn_threads(size, fn, base=2)


@pytest.mark.parametrize("size,base", [
    (6, 2),   # 2*3: repeated base factors + remainder group
    (5, 2),   # prime: one direct-exchange step
    (12, 2),  # 2*2*3
    (9, 3),   # base 3 perfect power
    (6, 3),   # 3*2
])
def test_allreduce_bcube_factorized(spawn_threads, size, base):
    """True base-B grouped exchange at non-perfect-power sizes
    (reference computeGroupSizePerStep, gloo/allreduce.cc:397-408)."""

    def fn(ctx, rank, _):
        x = fixture(rank, 10_001)
        ga.allreduce(
            ctx, [x.ctypes.data], x.size, ga.DType.f32, ga.ReduceOp.sum,
            algorithm="bcube",
        )
        expected = sum(fixture(r, 10_001) for r in range(size))
        assert np.allclose(x, expected)
        return True

    spawn_threads(size, fn, base=base)


@pytest.mark.parametrize("op,npop", [
    (ga.ReduceOp.sum, np.add),
    (ga.ReduceOp.product, np.multiply),
    (ga.ReduceOp.min, np.minimum),
    (ga.ReduceOp.max, np.maximum),
])
def test_allreduce_ops(spawn_threads, op, npop):
    size = 3

    def fn(ctx, rank, _):
        x = (np.arange(100, dtype=np.float32) % 5) + rank + 1
        expected = x.copy() * 0 + ((np.arange(100, dtype=np.float32) % 5) + 1)
        acc = None
        for r in range(size):
            v = (np.arange(100, dtype=np.float32) % 5) + r + 1
            acc = v if acc is None else npop(acc, v)
        ga.allreduce(ctx, [x.ctypes.data], x.size, ga.DType.f32, op)
        assert np.allclose(x, acc)
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("dtype,gatype", [
    (np.float64, ga.DType.f64),
    (np.int32, ga.DType.i32),
    (np.int64, ga.DType.i64),
    (np.int8, ga.DType.i8),
])
def test_allreduce_dtypes(spawn_threads, dtype, gatype):
    size = 2

    def fn(ctx, rank, _):
        x = (np.arange(64) % 7 + rank).astype(dtype)
        expected = sum((np.arange(64) % 7 + r).astype(np.int64) for r in range(size))
        ga.allreduce(ctx, [x.ctypes.data], x.size, gatype, ga.ReduceOp.sum)
        assert np.all(x.astype(np.int64) == expected.astype(np.int64))
        return True

    spawn_threads(size, fn)


def test_allreduce_fp16_bf16(spawn_threads):
    import torch

    size = 2

    def fn(ctx, rank, _):
        for tdt, gdt in [(torch.float16, ga.DType.f16), (torch.bfloat16, ga.DType.bf16)]:
            x = (torch.arange(128, dtype=torch.float32) % 9 + rank).to(tdt)
            ref = sum(
                ((torch.arange(128, dtype=torch.float32) % 9 + r).to(tdt)).float()
                for r in range(size)
            )
            ga.allreduce(ctx, [x.data_ptr()], x.numel(), gdt, ga.ReduceOp.sum)
            assert torch.allclose(x.float(), ref, atol=0.5)
        return True

    spawn_threads(size, fn)


def test_allreduce_multi_segment(spawn_threads):
    """Tiny maxSegmentSize forces the many-segment ring path
    (reference: allreduce_test.cc:330-360 shrinking maxSegmentSize)."""
    size = 3

    def fn(ctx, rank, _):
        x = fixture(rank, 5000)
        ga.allreduce(
            ctx, [x.ctypes.data], x.size, ga.DType.f32, ga.ReduceOp.sum,
            max_segment_size=256,
        )
        assert np.allclose(x, sum(fixture(r, 5000) for r in range(size)))
        return True

    spawn_threads(size, fn)


def test_allreduce_separate_input(spawn_threads):
    size = 2

    def fn(ctx, rank, _):
        inp = fixture(rank, 500)
        out = np.zeros(500, dtype=np.float32)
        ga.allreduce(
            ctx, [out.ctypes.data], 500, ga.DType.f32, ga.ReduceOp.sum,
            inputs=[inp.ctypes.data],
        )
        assert np.allclose(out, sum(fixture(r, 500) for r in range(size)))
        assert np.allclose(inp, fixture(rank, 500))  # input preserved
        return True

    spawn_threads(size, fn)


def test_allreduce_multiple_io(spawn_threads):
    """Multiple input and output pointers per rank."""
    size = 2

    def fn(ctx, rank, _):
        a = fixture(rank * 2, 300)
        b = fixture(rank * 2 + 1, 300)
        o1 = np.zeros(300, dtype=np.float32)
        o2 = np.zeros(300, dtype=np.float32)
        ga.allreduce(
            ctx, [o1.ctypes.data, o2.ctypes.data], 300, ga.DType.f32,
            ga.ReduceOp.sum, inputs=[a.ctypes.data, b.ctypes.data],
        )
        expected = sum(fixture(k, 300) for k in range(2 * size))
        assert np.allclose(o1, expected)
        assert np.allclose(o2, expected)
        return True

    spawn_threads(size, fn)


def test_concurrent_allreduce_tags(spawn_threads):
    """Two concurrent collectives on one context distinguished by tag."""
    size = 2

    def fn(ctx, rank, _):
        import threading as th

        xs = [fixture(rank + 10 * t, 2000) for t in range(2)]
        expected = [
            sum(fixture(r + 10 * t, 2000) for r in range(size)) for t in range(2)
        ]
        ths = [
            th.Thread(
                target=ga.allreduce,
                args=(ctx, [xs[t].ctypes.data], 2000),
                kwargs=dict(dtype=ga.DType.f32, op=ga.ReduceOp.sum, tag=t + 1),
            )
            for t in range(2)
        ]
        [t.start() for t in ths]
        [t.join() for t in ths]
        for t in range(2):
            assert np.allclose(xs[t], expected[t])
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [1, 2, 4])
@pytest.mark.parametrize("elements", [1, 100, 10_000])
def test_allgather(spawn_threads, size, elements):
    def fn(ctx, rank, _):
        inp = fixture(rank, elements)
        out = np.zeros(elements * size, dtype=np.float32)
        ga.allgather(ctx, out.ctypes.data, inp.ctypes.data, elements)
        for r in range(size):
            assert np.allclose(out[r * elements:(r + 1) * elements], fixture(r, elements))
        return True

    spawn_threads(size, fn)


def test_allgatherv(spawn_threads):
    size = 3
    counts = [5, 0, 17]

    def fn(ctx, rank, _):
        inp = fixture(rank, counts[rank])
        out = np.zeros(sum(counts), dtype=np.float32)
        ga.allgatherv(ctx, out.ctypes.data, inp.ctypes.data, counts)
        off = 0
        for r in range(size):
            assert np.allclose(out[off:off + counts[r]], fixture(r, counts[r]))
            off += counts[r]
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [1, 2, 4])
def test_alltoall(spawn_threads, size):
    n = 32

    def fn(ctx, rank, _):
        inp = np.concatenate([fixture(rank * size + d, n) for d in range(size)])
        out = np.zeros(n * size, dtype=np.float32)
        ga.alltoall(ctx, out.ctypes.data, inp.ctypes.data, n)
        for s in range(size):
            assert np.allclose(out[s * n:(s + 1) * n], fixture(s * size + rank, n))
        return True

    spawn_threads(size, fn)


def test_alltoallv(spawn_threads):
    size = 3

    def fn(ctx, rank, _):
        # rank r sends (r+1)*(d+1) elements to rank d
        in_counts = [(rank + 1) * (d + 1) for d in range(size)]
        out_counts = [(s + 1) * (rank + 1) for s in range(size)]
        inp = np.concatenate(
            [np.full((rank + 1) * (d + 1), rank * 10 + d, dtype=np.float32)
             for d in range(size)])
        out = np.zeros(sum(out_counts), dtype=np.float32)
        ga.alltoallv(ctx, out.ctypes.data, inp.ctypes.data, in_counts, out_counts)
        off = 0
        for s in range(size):
            assert np.allclose(out[off:off + out_counts[s]], s * 10 + rank)
            off += out_counts[s]
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [2, 5])
def test_barrier(spawn_threads, size):
    import time

    arrived = []

    def fn(ctx, rank, _):
        time.sleep(0.05 * rank)
        arrived.append(rank)
        ga.barrier(ctx)
        assert len(arrived) == size  # nobody passes until all arrive
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [2, 4, 5])
@pytest.mark.parametrize("root", [0, 1])
def test_broadcast(spawn_threads, size, root):
    n = 4321

    def fn(ctx, rank, _):
        out = fixture(rank, n) if rank == root else np.zeros(n, dtype=np.float32)
        ga.broadcast(ctx, out.ctypes.data, 0, n, ga.DType.f32, root=root)
        assert np.allclose(out, fixture(root, n))
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("root", [0, 2])
def test_gather(spawn_threads, root):
    size, n = 3, 77

    def fn(ctx, rank, _):
        inp = fixture(rank, n)
        out = np.zeros(n * size, dtype=np.float32) if rank == root else np.zeros(0)
        ga.gather(ctx, out.ctypes.data if rank == root else 0,
                  inp.ctypes.data, n, root=root)
        if rank == root:
            for r in range(size):
                assert np.allclose(out[r * n:(r + 1) * n], fixture(r, n))
        return True

    spawn_threads(size, fn)


def test_gatherv(spawn_threads):
    size = 3
    counts = [3, 9, 1]

    def fn(ctx, rank, _):
        inp = fixture(rank, counts[rank])
        out = np.zeros(sum(counts), dtype=np.float32)
        ga.gatherv(ctx, out.ctypes.data, inp.ctypes.data, counts, root=0)
        if rank == 0:
            off = 0
            for r in range(size):
                assert np.allclose(out[off:off + counts[r]], fixture(r, counts[r]))
                off += counts[r]
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("root", [0, 1])
def test_scatter(spawn_threads, root):
    size, n = 3, 55

    def fn(ctx, rank, _):
        inp = (np.concatenate([fixture(d, n) for d in range(size)])
               if rank == root else np.zeros(0))
        out = np.zeros(n, dtype=np.float32)
        ga.scatter(ctx, out.ctypes.data,
                   inp.ctypes.data if rank == root else 0, n, root=root)
        assert np.allclose(out, fixture(rank, n))
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("root", [0, 2])
@pytest.mark.parametrize("elements", [1, 999, 50_000])
def test_reduce(spawn_threads, root, elements):
    size = 3

    def fn(ctx, rank, _):
        inp = fixture(rank, elements)
        out = np.zeros(elements, dtype=np.float32)
        ga.reduce(ctx, out.ctypes.data, inp.ctypes.data, elements,
                  ga.DType.f32, ga.ReduceOp.sum, root=root)
        if rank == root:
            assert np.allclose(out, sum(fixture(r, elements) for r in range(size)))
        assert np.allclose(inp, fixture(rank, elements))
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [1, 2, 4])
def test_reduce_scatter(spawn_threads, size):
    n = 128  # per-rank block

    def fn(ctx, rank, _):
        inp = fixture(rank, n * size)
        out = np.zeros(n, dtype=np.float32)
        ga.reduce_scatter(ctx, out.ctypes.data, inp.ctypes.data, n)
        total = sum(fixture(r, n * size) for r in range(size))
        assert np.allclose(out, total[rank * n:(rank + 1) * n])
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [6, 8])
def test_allreduce_wide(spawn_threads, size):
    """Wider rank counts (node-scale)."""

    def fn(ctx, rank, _):
        x = fixture(rank, 20_000)
        ga.allreduce(ctx, [x.ctypes.data], x.size, ga.DType.f32,
                     ga.ReduceOp.sum)
        assert np.allclose(x, sum(fixture(r, 20_000) for r in range(size)))
        return True

    spawn_threads(size, fn)


def test_allreduce_bcube_8(spawn_threads):
    def fn(ctx, rank, _):
        x = fixture(rank, 9_999)
        ga.allreduce(ctx, [x.ctypes.data], x.size, ga.DType.f32,
                     ga.ReduceOp.sum, algorithm="bcube")
        assert np.allclose(x, sum(fixture(r, 9_999) for r in range(8)))
        return True

    spawn_threads(8, fn, base=2)


def test_alltoall_8(spawn_threads):
    size, n = 8, 64

    def fn(ctx, rank, _):
        inp = np.concatenate([fixture(rank * size + d, n) for d in range(size)])
        out = np.zeros(n * size, dtype=np.float32)
        ga.alltoall(ctx, out.ctypes.data, inp.ctypes.data, n)
        for s in range(size):
            assert np.allclose(out[s * n:(s + 1) * n], fixture(s * size + rank, n))
        return True

    spawn_threads(size, fn)


def test_allreduce_per_op_timeout(spawn_threads):
    """Per-op timeout override (reference allreduce_test.cc:386 parity):
    a rank alone in the collective times out after ~timeout_ms, not the
    context default."""
    import time

    def fn(ctx, rank, _):
        if rank == 0:
            x = fixture(0, 1000)
            t0 = time.monotonic()
            with pytest.raises(ga.TimeoutError):
                ga.allreduce(ctx, [x.ctypes.data], 1000, ga.DType.f32,
                             ga.ReduceOp.sum, timeout_ms=200)
            assert time.monotonic() - t0 < 5.0  # not the 30s default
        else:
            time.sleep(1.0)  # never joins in time; ctx gets poisoned
        return True

    spawn_threads(2, fn)


def test_all_collectives_torture_8rank(spawn_threads):
    """Integration: every v2 collective interleaved for many rounds on
    one 8-rank context (slot/tag hygiene across op types)."""
    size, n, iters = 8, 512, 8

    def fn(ctx, rank, _):
        for it in range(iters):
            x = fixture(rank + it, n)
            ga.allreduce(ctx, [x.ctypes.data], n, ga.DType.f32,
                         ga.ReduceOp.sum, tag=1)
            assert np.allclose(
                x, sum(fixture(r + it, n) for r in range(size)))

            inp = fixture(rank, n)
            out = np.zeros(n * size, dtype=np.float32)
            ga.allgather(ctx, out.ctypes.data, inp.ctypes.data, n, tag=2)
            assert np.allclose(out[rank * n:(rank + 1) * n], inp)

            b = fixture(it, n) if rank == it % size else np.zeros(
                n, dtype=np.float32)
            ga.broadcast(ctx, b.ctypes.data, 0, n, ga.DType.f32,
                         root=it % size, tag=3)
            assert np.allclose(b, fixture(it, n))

            a2a_in = np.concatenate(
                [fixture(rank * size + d, 16) for d in range(size)])
            a2a_out = np.zeros(16 * size, dtype=np.float32)
            ga.alltoall(ctx, a2a_out.ctypes.data, a2a_in.ctypes.data, 16,
                        tag=4)
            assert np.allclose(a2a_out[:16], fixture(rank, 16))

            r = fixture(rank, n)
            ga.reduce(ctx, r.ctypes.data, r.ctypes.data, n, ga.DType.f32,
                      ga.ReduceOp.max, root=0, tag=5)
            if rank == 0:
                assert np.allclose(r, fixture(size - 1, n))

            ga.barrier(ctx, tag=6)
        return True

    spawn_threads(size, fn)


def test_allreduce_16rank(spawn_threads):
    """Rank-count headroom beyond one node (reference sweeps to 16/32):
    16 thread-ranks, full mesh = 120 pairs in one process."""

    def fn(ctx, rank, _):
        x = fixture(rank, 5000)
        ga.allreduce(ctx, [x.ctypes.data], 5000, ga.DType.f32,
                     ga.ReduceOp.sum)
        assert np.allclose(x, sum(fixture(r, 5000) for r in range(16)))
        return True

    spawn_threads(16, fn)


def test_sustained_concurrent_streams(spawn_threads):
    """Sustained load: 4 ranks x 3 concurrent collective streams x many
    iterations (regression net for protocol liveness under contention;
    all ranks run identical iteration counts — collectives are
    collective)."""
    size = 4
    plan = [(1, 64, 120), (2, 3000, 80), (3, 100_000, 25)]

    def fn(ctx, rank, _):
        import threading as th

        errs = []

        def stream(tag, n, iters):
            try:
                for it in range(iters):
                    x = (np.arange(n, dtype=np.float64) + rank + it).astype(
                        np.float32)
                    ga.allreduce(ctx, [x.ctypes.data], n, ga.DType.f32,
                                 ga.ReduceOp.sum, tag=tag)
                    if it % 17 == 0:
                        ref = sum(
                            (np.arange(n, dtype=np.float64) + r + it).astype(
                                np.float32) for r in range(size))
                        assert np.allclose(x, ref), (tag, it)
            except Exception:  # noqa: BLE001
                import traceback

                errs.append(traceback.format_exc())

        ths = [th.Thread(target=stream, args=p) for p in plan]
        [t.start() for t in ths]
        [t.join() for t in ths]
        assert not errs, errs[0]
        return True

    spawn_threads(size, fn)
