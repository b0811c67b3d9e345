"""Legacy Algorithm class tests (reference parity: gloo/test/
allreduce_test.cc legacy sweeps, reduce_scatter_test, etc.).
Run twice each to exercise the notification/reuse protocol.
"""
import numpy as np
import pytest

import gloo_amd as ga


def fixture(rank, n):
    return (np.arange(n, dtype=np.float64) + rank).astype(np.float32)


ALLREDUCE_ALGOS = [
    "allreduce_ring",
    "allreduce_ring_chunked",
    "allreduce_halving_doubling",
    "allreduce_bcube",
]


@pytest.mark.parametrize("algo", ALLREDUCE_ALGOS)
@pytest.mark.parametrize("size", [1, 2, 3, 4])
@pytest.mark.parametrize("elements", [1, 100, 10_000])
def test_legacy_allreduce(spawn_threads, algo, size, elements):
    # (bcube now factorizes any size; HD uses binary blocks for non-pow2)

    def fn(ctx, rank, _):
        x = fixture(rank, elements)
        a = ga._C.create_algorithm(algo, ctx, [x.ctypes.data], elements)
        for it in range(2):
            np.copyto(x, fixture(rank, elements))
            a.run()
            expected = sum(fixture(r, elements) for r in range(size))
            assert np.allclose(x, expected), (algo, it)
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [2, 3])
def test_legacy_allreduce_multi_ptr(spawn_threads, size):
    def fn(ctx, rank, _):
        n = 500
        a_ = fixture(rank * 2, n)
        b_ = fixture(rank * 2 + 1, n)
        algo = ga._C.create_algorithm(
            "allreduce_ring", ctx, [a_.ctypes.data, b_.ctypes.data], n)
        algo.run()
        expected = sum(fixture(k, n) for k in range(2 * size))
        assert np.allclose(a_, expected) and np.allclose(b_, expected)
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [2, 4])
def test_legacy_allgather_ring(spawn_threads, size):
    def fn(ctx, rank, _):
        n = 77
        inp = fixture(rank, n)
        out = np.zeros(n * size, dtype=np.float32)
        algo = ga._C.create_algorithm(
            "allgather_ring", ctx, [inp.ctypes.data, out.ctypes.data], n)
        for _ in range(2):
            algo.run()
            for r in range(size):
                assert np.allclose(out[r * n:(r + 1) * n], fixture(r, n))
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("name", ["barrier_all_to_all", "barrier_all_to_one"])
def test_legacy_barriers(spawn_threads, name):
    import time

    size = 4
    arrived = []

    def fn(ctx, rank, _):
        algo = ga._C.create_algorithm(name, ctx)
        time.sleep(0.03 * rank)
        arrived.append(rank)
        algo.run()
        assert len(arrived) == size
        algo.run()  # reusable
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("root", [0, 2])
def test_legacy_broadcast_one_to_all(spawn_threads, root):
    size = 3

    def fn(ctx, rank, _):
        n = 1000
        x = fixture(rank, n) if rank == root else np.zeros(n, dtype=np.float32)
        algo = ga._C.create_algorithm(
            "broadcast_one_to_all", ctx, [x.ctypes.data], n, root=root)
        for _ in range(2):
            algo.run()
            assert np.allclose(x, fixture(root, n))
        return True

    spawn_threads(size, fn)


def test_legacy_pairwise_exchange(spawn_threads):
    size = 4

    def fn(ctx, rank, _):
        algo = ga._C.create_algorithm(
            "pairwise_exchange", ctx, bytes=1 << 16, steps=2)
        algo.run()
        algo.run()
        return True

    spawn_threads(size, fn)


@pytest.mark.parametrize("size", [2, 3, 4])
def test_legacy_reduce_scatter_hd(spawn_threads, size):
    def fn(ctx, rank, _):
        n = 1000
        # uneven distribution
        base = n // size
        recv = [base + (1 if r < n % size else 0) for r in range(size)]
        x = fixture(rank, n)
        algo = ga._C.create_algorithm(
            "reduce_scatter_halving_doubling", ctx, [x.ctypes.data], n,
            recv_elems=recv)
        algo.run()
        total = sum(fixture(r, n) for r in range(size))
        off = sum(recv[:rank])
        assert np.allclose(x[:recv[rank]], total[off:off + recv[rank]])
        return True

    spawn_threads(size, fn)


def test_legacy_allreduce_local(spawn_threads):
    def fn(ctx, rank, _):
        n = 100
        a_ = fixture(0, n)
        b_ = fixture(1, n)
        algo = ga._C.create_algorithm(
            "allreduce_local", ctx, [a_.ctypes.data, b_.ctypes.data], n)
        algo.run()
        assert np.allclose(a_, fixture(0, n) + fixture(1, n))
        assert np.allclose(b_, a_)
        return True

    spawn_threads(1, fn)


def test_legacy_allreduce_hd_8(spawn_threads):
    def fn(ctx, rank, _):
        x = fixture(rank, 12_345)
        a = ga._C.create_algorithm("allreduce_halving_doubling", ctx,
                                   [x.ctypes.data], 12_345)
        a.run()
        assert np.allclose(x, sum(fixture(r, 12_345) for r in range(8)))
        return True

    spawn_threads(8, fn)


def test_legacy_allreduce_hd_6(spawn_threads):
    """Non-power-of-2: pre/post folding path."""

    def fn(ctx, rank, _):
        x = fixture(rank, 5_000)
        a = ga._C.create_algorithm("allreduce_halving_doubling", ctx,
                                   [x.ctypes.data], 5_000)
        a.run()
        assert np.allclose(x, sum(fixture(r, 5_000) for r in range(6)))
        return True

    spawn_threads(6, fn)


@pytest.mark.parametrize("size", [5, 6, 7, 11])
@pytest.mark.parametrize("elements", [1, 7, 10_000, 100_003])
def test_legacy_hd_binary_blocks(spawn_threads, size, elements):
    """Non-power-of-2 halving-doubling via binary blocks (reference
    allreduce_halving_doubling.h:38-64 scheme): P decomposes into
    power-of-2 blocks (11 -> 8+2+1) that cascade their reduce-scattered
    segments instead of folding whole buffers into partners."""

    def fn(ctx, rank, _):
        x = fixture(rank, elements)
        a = ga._C.create_algorithm(
            "allreduce_halving_doubling", ctx, [x.ctypes.data], elements)
        for it in range(3):  # repeat runs on one instance
            np.copyto(x, fixture(rank, elements))
            a.run()
            expected = sum(fixture(r, elements) for r in range(size))
            assert np.allclose(x, expected), (size, elements, it)
        return True

    spawn_threads(size, fn)
