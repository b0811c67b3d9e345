"""torch.distributed ProcessGroup backend tests (CPU tensors).

Direct-construction tests run world=2/4 in threads; the subprocess test
goes through init_process_group("glooamd") + DDP (see pg_worker.py).
"""
import os
import subprocess
import sys
import threading

import pytest
import torch
import torch.distributed as dist

import gloo_amd as ga
from gloo_amd.pg import ProcessGroupGlooAmd


def spawn_pg(size, fn):
    store = ga.HashStore()
    errors = []
    results = [None] * size

    def worker(rank):
        try:
            pg = ProcessGroupGlooAmd(store, rank, size)
            results[rank] = fn(pg, rank, size)
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,)) for r in range(size)]
    [t.start() for t in ths]
    [t.join(60) for t in ths]
    assert not errors, errors[0]
    return results


def test_pg_allreduce():
    def fn(pg, rank, size):
        t = torch.arange(1000, dtype=torch.float32) + rank
        pg.allreduce([t]).wait()
        expected = sum(torch.arange(1000, dtype=torch.float32) + r
                       for r in range(size))
        assert torch.allclose(t, expected)
        return True

    spawn_pg(4, fn)


def test_pg_allreduce_ops():
    def fn(pg, rank, size):
        opts = dist.AllreduceOptions()
        opts.reduceOp = dist.ReduceOp.MAX
        t = torch.tensor([float(rank), -float(rank)])
        pg.allreduce([t], opts).wait()
        assert torch.allclose(t, torch.tensor([float(size - 1), 0.0]))

        opts.reduceOp = dist.ReduceOp.AVG
        t = torch.full((10,), float(rank))
        pg.allreduce([t], opts).wait()
        assert torch.allclose(t, torch.full((10,), (size - 1) / 2))
        return True

    spawn_pg(2, fn)


def test_pg_broadcast():
    def fn(pg, rank, size):
        opts = dist.BroadcastOptions()
        opts.rootRank = 1
        t = (torch.arange(64, dtype=torch.float64) if rank == 1
             else torch.zeros(64, dtype=torch.float64))
        pg.broadcast([t], opts).wait()
        assert torch.allclose(t, torch.arange(64, dtype=torch.float64))
        return True

    spawn_pg(3, fn)


def test_pg_allgather():
    def fn(pg, rank, size):
        inp = torch.full((8,), float(rank))
        outs = [[torch.zeros(8) for _ in range(size)]]
        pg.allgather(outs, [inp]).wait()
        for r in range(size):
            assert torch.allclose(outs[0][r], torch.full((8,), float(r)))
        # _allgather_base
        flat = torch.zeros(8 * size)
        pg._allgather_base(flat, inp).wait()
        for r in range(size):
            assert torch.allclose(flat[r * 8:(r + 1) * 8],
                                  torch.full((8,), float(r)))
        return True

    spawn_pg(3, fn)


def test_pg_reduce_scatter_base():
    def fn(pg, rank, size):
        inp = torch.arange(size * 4, dtype=torch.float32) + rank
        out = torch.zeros(4)
        pg._reduce_scatter_base(out, inp).wait()
        total = sum(torch.arange(size * 4, dtype=torch.float32) + r
                    for r in range(size))
        assert torch.allclose(out, total[rank * 4:(rank + 1) * 4])
        return True

    spawn_pg(4, fn)


def test_pg_alltoall_base():
    def fn(pg, rank, size):
        inp = torch.arange(size * 3, dtype=torch.float32) + 100 * rank
        out = torch.zeros(size * 3)
        pg.alltoall_base(out, inp, [], []).wait()
        for s in range(size):
            expect = torch.arange(rank * 3, rank * 3 + 3,
                                  dtype=torch.float32) + 100 * s
            assert torch.allclose(out[s * 3:(s + 1) * 3], expect)
        return True

    spawn_pg(3, fn)


def test_pg_gather_scatter():
    def fn(pg, rank, size):
        opts = dist.GatherOptions()
        opts.rootRank = 0
        inp = torch.full((5,), float(rank))
        outs = [[torch.zeros(5) for _ in range(size)]] if rank == 0 else []
        pg.gather(outs, [inp], opts).wait()
        if rank == 0:
            for r in range(size):
                assert torch.allclose(outs[0][r], torch.full((5,), float(r)))

        sopts = dist.ScatterOptions()
        sopts.rootRank = 0
        sout = torch.zeros(5)
        sins = ([[torch.full((5,), float(r) + 10) for r in range(size)]]
                if rank == 0 else [])
        pg.scatter([sout], sins, sopts).wait()
        assert torch.allclose(sout, torch.full((5,), float(rank) + 10))
        return True

    spawn_pg(3, fn)


def test_pg_send_recv_barrier():
    def fn(pg, rank, size):
        if rank == 0:
            t = torch.arange(32, dtype=torch.int64)
            pg.send([t], 1, tag=5).wait()
        elif rank == 1:
            t = torch.zeros(32, dtype=torch.int64)
            pg.recv([t], 0, tag=5).wait()
            assert torch.equal(t, torch.arange(32, dtype=torch.int64))
        pg.barrier().wait()
        return True

    spawn_pg(2, fn)


def test_pg_reduce():
    def fn(pg, rank, size):
        opts = dist.ReduceOptions()
        opts.rootRank = 1
        opts.reduceOp = dist.ReduceOp.SUM
        t = torch.full((100,), float(rank + 1))
        pg.reduce([t], opts).wait()
        if rank == 1:
            assert torch.allclose(t, torch.full((100,), float(
                sum(range(1, size + 1)))))
        return True

    spawn_pg(3, fn)


def test_init_process_group_ddp():
    """Full stack: init_process_group('glooamd') + DDP training step in
    subprocesses."""
    worker = os.path.join(os.path.dirname(__file__), "pg_worker.py")
    import socket

    # reserve a free port for torch's TCPStore (closing is racy but far
    # better than blind randints on a shared CI box)
    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        port = _s.getsockname()[1]
    procs = [
        subprocess.Popen(
            [sys.executable, worker, str(r), "2", str(port)],
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        )
        for r in range(2)
    ]
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
    assert all(p.returncode == 0 for p in procs), "\n".join(outs)
    assert all("DDP-OK" in o for o in outs), "\n".join(outs)


def test_pg_noncontiguous_tensors():
    """Collectives and p2p on views/transposes (staged automatically)."""

    def fn(pg, rank, size):
        m = torch.arange(16, dtype=torch.float32).reshape(4, 4) + rank
        col = m[:, 1]  # stride-4 view
        pg.allreduce([col]).wait()
        expect = (torch.arange(16, dtype=torch.float32).reshape(4, 4)[:, 1]
                  * size + sum(range(size)))
        assert torch.allclose(col, expect), col
        tr = (torch.arange(9, dtype=torch.float32).reshape(3, 3) * 0 + rank).T
        if rank == 0:
            pg.send([torch.full((3, 3), 7.0).T.contiguous()], 1, tag=3).wait()
        elif rank == 1:
            pg.recv([tr], 0, tag=3).wait()
            assert torch.allclose(tr, torch.full((3, 3), 7.0)), tr
        return True

    spawn_pg(2, fn)


def test_pg_lifecycle_reuse():
    """Create, use, destroy and re-create process groups in one process
    (device/loop/context teardown hygiene)."""

    def fn(pg, rank, size):
        t = torch.full((256,), float(rank + 1))
        pg.allreduce([t]).wait()
        assert torch.allclose(t, torch.full((256,), float(sum(range(1, size + 1)))))
        return True

    for _ in range(3):
        spawn_pg(2, fn)


def test_pg_monitored_barrier():
    def fn(pg, rank, size):
        import time

        time.sleep(0.02 * rank)  # staggered arrival
        pg.monitored_barrier(timeout=timedelta(seconds=10))
        return True

    from datetime import timedelta  # noqa: F811
    spawn_pg(3, fn)


def test_pg_monitored_barrier_reports_missing():
    """Rank 0 names the rank that never arrived."""
    from datetime import timedelta

    def fn(pg, rank, size):
        if rank == 2:
            return True  # never joins
        if rank == 0:
            try:
                # wait_all_ranks so a slow-but-alive rank 1 cannot mask
                # the genuinely missing rank 2 in the report
                pg.monitored_barrier(timeout=timedelta(milliseconds=800),
                                     wait_all_ranks=True)
                raise AssertionError("expected monitored_barrier to fail")
            except RuntimeError as e:
                assert "2" in str(e), e
        else:
            try:
                pg.monitored_barrier(timeout=timedelta(seconds=5))
            except Exception:
                pass  # context poisoned by rank 0's timeout is fine
        return True

    spawn_pg(3, fn)


def test_pg_monitored_barrier_keeps_context_usable():
    """A monitored_barrier timeout must not poison the context: the probe
    uses try_wait_recv (non-poisoning), so after the missing-rank report
    the same process group still completes collectives."""
    from datetime import timedelta

    def fn(pg, rank, size):
        if rank == 0:
            try:
                pg.monitored_barrier(timeout=timedelta(milliseconds=500),
                                     wait_all_ranks=True)
                raise AssertionError("expected monitored_barrier to fail")
            except RuntimeError as e:
                assert "1" in str(e), e
        # rank 1 never joined the monitored_barrier; both ranks now run a
        # normal collective, which only works if rank 0's context survived
        t = torch.full((512,), float(rank + 1))
        pg.allreduce([t]).wait()
        assert torch.allclose(t, torch.full((512,), 3.0))
        return True

    spawn_pg(2, fn)


def test_pg_over_uds(monkeypatch):
    """GLOO_AMD_UDS=1 routes the PG control plane over unix sockets."""
    monkeypatch.setenv("GLOO_AMD_UDS", "1")

    def fn(pg, rank, size):
        t = torch.full((2048,), float(rank + 1))
        pg.allreduce([t]).wait()
        assert torch.allclose(t, torch.full((2048,), 3.0))
        return True

    spawn_pg(2, fn)
