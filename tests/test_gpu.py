"""GPU tests (MI355X): kernel numerics vs plain PyTorch fp32 reference,
xGMI/IPC device allreduce (same-process and cross-process), halving
doubling, broadcast, host-staged path.

All multi-rank device tests here run 2 ranks on ONE GPU (the CI box has
a single MI355X): that exercises the full flag/doorbell/IPC protocol;
only the link bandwidth differs from the 8-GPU topology.
"""
import os
import subprocess
import sys

import numpy as np
import pytest

import gloo_amd as ga

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

GPU = torch.cuda.is_available() if hasattr(torch, "cuda") else False
if not GPU:
    pytest.skip("no GPU", allow_module_level=True)


def _dev(t):
    """Pinned stream-local H2D (see _host)."""
    out = torch.empty(t.shape, dtype=t.dtype, device="cuda")
    out.copy_(t.pin_memory())
    return out


def _host(t):
    """Pinned stream-local D2H for threaded multi-rank tests: pageable
    .cpu() has device-wide sync semantics and can deadlock with the peer
    rank's in-flight doorbell wait kernels on a shared GPU+process."""
    out = torch.empty(t.shape, dtype=t.dtype, pin_memory=True)
    out.copy_(t)
    return out


DTYPES = [
    (torch.float32, ga.DType.f32, 1e-6),
    (torch.float64, ga.DType.f64, 1e-12),
    (torch.float16, ga.DType.f16, 2e-3),
    (torch.bfloat16, ga.DType.bf16, 2e-2),
    (torch.int32, ga.DType.i32, 0),
    (torch.int64, ga.DType.i64, 0),
    (torch.int8, ga.DType.i8, 0),
    (torch.uint8, ga.DType.u8, 0),
]

OPS = [
    (ga.ReduceOp.sum, lambda a, b: a + b),
    (ga.ReduceOp.product, lambda a, b: a * b),
    (ga.ReduceOp.min, torch.minimum),
    (ga.ReduceOp.max, torch.maximum),
]


@pytest.mark.parametrize("tdt,gdt,tol", DTYPES)
@pytest.mark.parametrize("gop,ref", OPS)
def test_reduce2_numerics(tdt, gdt, tol, gop, ref):
    """HIP kernel vs plain PyTorch fp32 reference of the same op."""
    torch.manual_seed(42)
    n = 1_000_003  # odd: exercises the vector tail path
    if tdt.is_floating_point:
        a = (torch.rand(n, dtype=torch.float32) * 4 + 0.5).to(tdt).cuda()
        b = (torch.rand(n, dtype=torch.float32) * 4 + 0.5).to(tdt).cuda()
        expect = ref(a.float(), b.float())
    else:
        a = torch.randint(1, 7, (n,), dtype=tdt).cuda()
        b = torch.randint(1, 7, (n,), dtype=tdt).cuda()
        expect = ref(a.long(), b.long())
    dst = torch.empty_like(a)
    ga._C.hip_reduce2(dst.data_ptr(), a.data_ptr(), b.data_ptr(), n, gdt, gop)
    torch.cuda.synchronize()
    if tdt.is_floating_point:
        got = dst.float().cpu()
        assert torch.allclose(got, expect.cpu(), rtol=tol, atol=tol * 8), (
            (got - expect.cpu()).abs().max()
        )
    else:
        assert torch.equal(dst.long().cpu(), expect.cpu())


def test_reduce2_unaligned():
    """Offset slices force the scalar fallback path."""
    n = 4097
    base = torch.rand(n + 1, dtype=torch.float32).cuda()
    a = base[1:]  # 4-byte aligned but not 16
    b = torch.rand(n, dtype=torch.float32).cuda()
    dst = torch.empty(n, dtype=torch.float32).cuda()
    ga._C.hip_reduce2(
        dst.data_ptr(), a.data_ptr(), b.data_ptr(), n, ga.DType.f32,
        ga.ReduceOp.sum)
    torch.cuda.synchronize()
    assert torch.allclose(dst, a + b)


def test_allreduce_local_multi_ptr():
    n = 500_000
    ts = [torch.rand(n, dtype=torch.float32).cuda() for _ in range(4)]
    expect = sum(t.float() for t in ts)
    ga._C.hip_allreduce_local([t.data_ptr() for t in ts], n, ga.DType.f32,
                              ga.ReduceOp.sum)
    torch.cuda.synchronize()
    for t in ts:
        assert torch.allclose(t, expect, rtol=1e-5, atol=1e-4)


def _two_rank_device_test(algo_name, elements, dtype=torch.float32,
                          chunked=True, inbox_cap=0):
    """Two threads, one GPU each rank (device 0), full device protocol."""
    import threading

    store = ga.HashStore()
    errors = []
    results = {}

    def worker(rank):
        try:
            dev = ga.create_tcp_device()
            ctx = ga.Context(rank, 2)
            ctx.connect_full_mesh(store, dev)
            ctx.set_timeout(60000)
            torch.cuda.set_device(0)
            g = torch.Generator(device="cpu").manual_seed(rank)
            x = _dev(torch.rand(elements, generator=g,
                               dtype=torch.float32).to(dtype))
            ref_inputs = [
                torch.rand(elements,
                           generator=torch.Generator("cpu").manual_seed(r),
                           dtype=torch.float32).to(dtype).float()
                for r in range(2)
            ]
            expect = ref_inputs[0] + ref_inputs[1]
            if algo_name == "ring":
                algo = ga._C.HipAllreduceRing(ctx, 0, chunked, inbox_cap)
            else:
                algo = ga._C.HipAllreduceHalvingDoubling(ctx, 0, inbox_cap)
            gdt = ga.dtype_from_torch(dtype)
            algo.run(x.data_ptr(), elements, gdt, ga.ReduceOp.sum)
            # NOTE: no torch.cuda.synchronize() in threaded 2-ranks-on-one-
            # GPU tests: a device-wide sync would also wait on the OTHER
            # rank's in-flight doorbell wait kernels (deadlock). run() is
            # blocking (its streams are synced).
            got = _host(x).float()
            tol = 2e-2 if dtype in (torch.bfloat16, torch.float16) else 1e-4
            assert torch.allclose(got, expect, rtol=tol, atol=tol), (
                rank, (got - expect).abs().max().item())
            # run twice: cross-run seq/flag reuse (barrier keeps the other
            # rank's torch ops out of our spin window on a shared GPU)
            ga.barrier(ctx, tag=901)
            algo.run(x.data_ptr(), elements, gdt, ga.ReduceOp.sum)
            results[rank] = True
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,), daemon=True)
           for r in range(2)]
    [t.start() for t in ths]
    [t.join(100) for t in ths]
    assert not errors, errors[0]
    assert not any(t.is_alive() for t in ths), "worker hung"
    assert results == {0: True, 1: True}


def test_hip_allreduce_ring_chunked_2rank():
    _two_rank_device_test("ring", 2_000_000)


def test_hip_allreduce_ring_unchunked_2rank():
    _two_rank_device_test("ring", 600_000, chunked=False)


def test_hip_allreduce_ring_many_segments():
    # tiny inboxes force a deep segmented pipeline
    _two_rank_device_test("ring", 1_000_000, inbox_cap=64 * 1024)


def test_hip_allreduce_ring_bf16():
    _two_rank_device_test("ring", 1_000_000, dtype=torch.bfloat16)


def test_hip_allreduce_hd_2rank():
    _two_rank_device_test("hd", 2_000_000)


def test_hip_allreduce_hd_many_chunks():
    _two_rank_device_test("hd", 3_000_000, inbox_cap=128 * 1024)


def test_hip_broadcast_one_to_all():
    import threading

    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            dev = ga.create_tcp_device()
            ctx = ga.Context(rank, 2)
            ctx.connect_full_mesh(store, dev)
            ctx.set_timeout(60000)
            torch.cuda.set_device(0)
            n = 1_000_000
            if rank == 0:
                x = torch.arange(n, dtype=torch.float32, device="cuda")
            else:
                x = torch.zeros(n, dtype=torch.float32, device="cuda")
            algo = ga._C.HipBroadcastOneToAll(ctx, 0, 0)
            for it in range(2):
                algo.run(x.data_ptr(), x.numel() * 4)
                ga.barrier(ctx, tag=950 + it)  # see gpu_bcast_stress note
                assert torch.equal(
                    _host(x), torch.arange(n, dtype=torch.float32))
                ga.barrier(ctx, tag=960 + it)
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,), daemon=True)
           for r in range(2)]
    [t.start() for t in ths]
    [t.join(100) for t in ths]
    assert not errors, errors[0]


def _two_rank_generic(builder_and_check):
    import threading

    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            dev = ga.create_tcp_device()
            ctx = ga.Context(rank, 2)
            ctx.connect_full_mesh(store, dev)
            ctx.set_timeout(60000)
            torch.cuda.set_device(0)
            builder_and_check(ctx, rank)
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,), daemon=True)
           for r in range(2)]
    [t.start() for t in ths]
    [t.join(100) for t in ths]
    assert not errors, errors[0]
    assert not any(t.is_alive() for t in ths), "worker hung"


def test_hip_allreduce_direct():
    def fn(ctx, rank):
        n = 5_000_001  # odd block tails + >8MiB blocks -> chunk pipeline
        g = torch.Generator("cpu").manual_seed(rank)
        x = _dev(torch.rand(n, generator=g))
        ref = sum(
            torch.rand(n, generator=torch.Generator("cpu").manual_seed(r))
            for r in range(2))
        algo = ga._C.HipAllreduceDirect(ctx, 0)
        y = x.clone()
        for it in range(3):
            y.copy_(x)
            algo.run(y.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
            assert torch.allclose(_host(y), ref, atol=1e-5)
            ga.barrier(ctx, tag=911 + it)

    _two_rank_generic(fn)


def test_hip_allgather_ring():
    def fn(ctx, rank):
        n = 1_000_000
        inp = torch.full((n,), float(rank + 1), device="cuda")
        out = torch.zeros(2 * n, device="cuda")
        algo = ga._C.HipAllgatherRing(ctx, 0)
        for it in range(2):
            algo.run(inp.data_ptr(), out.data_ptr(), n, 4)
            assert torch.all(out[:n] == 1.0) and torch.all(out[n:] == 2.0)
            ga.barrier(ctx, tag=902 + it)

    _two_rank_generic(fn)


def test_hip_reduce_scatter_ring():
    def fn(ctx, rank):
        n = 500_000
        g = torch.Generator("cpu").manual_seed(rank)
        inp = _dev(torch.rand(2 * n, generator=g))
        out = torch.zeros(n, device="cuda")
        ref = sum(
            torch.rand(2 * n, generator=torch.Generator("cpu").manual_seed(r))
            for r in range(2))
        algo = ga._C.HipReduceScatterRing(ctx, 0)
        for it in range(2):
            algo.run(inp.data_ptr(), out.data_ptr(), n, ga.DType.f32,
                     ga.ReduceOp.sum)
            expect = ref[rank * n:(rank + 1) * n]
            assert torch.allclose(_host(out), expect, atol=1e-5)
            ga.barrier(ctx, tag=905 + it)

    _two_rank_generic(fn)


def test_hip_alltoall():
    def fn(ctx, rank):
        n = 300_000
        inp = torch.cat([
            torch.full((n,), float(rank * 2 + d), device="cuda")
            for d in range(2)
        ])
        out = torch.zeros(2 * n, device="cuda")
        algo = ga._C.HipAlltoall(ctx, 0)
        for it in range(2):
            algo.run(inp.data_ptr(), out.data_ptr(), n, 4)
            for s in range(2):
                assert torch.all(out[s * n:(s + 1) * n] == s * 2 + rank)
            ga.barrier(ctx, tag=908 + it)

    _two_rank_generic(fn)


def _n_rank_generic(size, builder_and_check, timeout=120):
    import threading

    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            dev = ga.create_tcp_device()
            ctx = ga.Context(rank, size)
            ctx.connect_full_mesh(store, dev)
            ctx.set_timeout(60000)
            torch.cuda.set_device(0)
            builder_and_check(ctx, rank)
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,), daemon=True)
           for r in range(size)]
    [t.start() for t in ths]
    [t.join(timeout) for t in ths]
    assert not errors, errors[0]
    assert not any(t.is_alive() for t in ths), "worker hung"


def test_hip_allreduce_hd_3rank_nonpow2():
    """Non-power-of-2 halving-doubling: rank 2 folds into rank 0 before
    and after a 2-rank exchange (device folding)."""
    def fn(ctx, rank):
        n = 1_500_000
        g = torch.Generator("cpu").manual_seed(rank)
        x = _dev(torch.rand(n, generator=g))
        ref = sum(
            torch.rand(n, generator=torch.Generator("cpu").manual_seed(r))
            for r in range(3))
        algo = ga._C.HipAllreduceHalvingDoubling(ctx, 0)
        y = x.clone()
        for it in range(3):  # cross-run fold seq/flag reuse
            y.copy_(x)
            algo.run(y.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
            assert torch.allclose(_host(y), ref, atol=1e-4), (
                rank, it, (_host(y) - ref).abs().max().item())
            ga.barrier(ctx, tag=921 + it)

    _n_rank_generic(3, fn)


def test_hip_allreduce_hd_3rank_small_inbox():
    """Non-pow2 fold with a chunked fold pipeline (tiny inbox)."""
    def fn(ctx, rank):
        n = 800_000
        g = torch.Generator("cpu").manual_seed(rank)
        x = _dev(torch.rand(n, generator=g))
        ref = sum(
            torch.rand(n, generator=torch.Generator("cpu").manual_seed(r))
            for r in range(3))
        algo = ga._C.HipAllreduceHalvingDoubling(ctx, 0, 256 * 1024)
        algo.run(x.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
        assert torch.allclose(_host(x), ref, atol=1e-4)
        ga.barrier(ctx, tag=931)

    _n_rank_generic(3, fn)


def test_hip_allreduce_ring_multi_input():
    """run_multi: fused local reduceN as the copy-in stage, result
    broadcast to every caller pointer (reference --inputs parity)."""
    def fn(ctx, rank):
        n = 600_000
        k = 3
        ts = [
            _dev(torch.rand(
                n, generator=torch.Generator("cpu").manual_seed(rank * 10 + i)))
            for i in range(k)
        ]
        ref = sum(
            torch.rand(n, generator=torch.Generator("cpu").manual_seed(r * 10 + i))
            for r in range(2) for i in range(k))
        algo = ga._C.HipAllreduceRing(ctx, 0)
        algo.run_multi([t.data_ptr() for t in ts], n, ga.DType.f32,
                       ga.ReduceOp.sum)
        for t in ts:
            assert torch.allclose(_host(t), ref, atol=1e-4), (
                rank, (_host(t) - ref).abs().max().item())
        ga.barrier(ctx, tag=941)

    _two_rank_generic(fn)


def test_hip_allreduce_direct_multi_input():
    def fn(ctx, rank):
        n = 400_000
        k = 2
        ts = [
            _dev(torch.rand(
                n, generator=torch.Generator("cpu").manual_seed(rank * 10 + i)))
            for i in range(k)
        ]
        ref = sum(
            torch.rand(n, generator=torch.Generator("cpu").manual_seed(r * 10 + i))
            for r in range(2) for i in range(k))
        algo = ga._C.HipAllreduceDirect(ctx, 0)
        algo.run_multi([t.data_ptr() for t in ts], n, ga.DType.f32,
                       ga.ReduceOp.sum)
        for t in ts:
            assert torch.allclose(_host(t), ref, atol=1e-4)
        ga.barrier(ctx, tag=945)

    _two_rank_generic(fn)


def test_hip_allreduce_ring_graph_replay_soak():
    """Long same-shape run chain: first run eager, second captures the
    hipGraph, 18 more replay it — cross-run seq/inbox gating must hold
    over a long monotonic doorbell chain."""
    def fn(ctx, rank):
        n = 1_000_000
        g = torch.Generator("cpu").manual_seed(rank)
        x = _dev(torch.rand(n, generator=g))
        ref = sum(
            torch.rand(n, generator=torch.Generator("cpu").manual_seed(r))
            for r in range(2))
        algo = ga._C.HipAllreduceRing(ctx, 0)
        y = x.clone()
        for it in range(20):
            y.copy_(x)
            algo.run(y.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
            assert torch.allclose(_host(y), ref, atol=1e-4), (rank, it)
            ga.barrier(ctx, tag=981)

    _two_rank_generic(fn)


def test_hip_allreduce_ring_alternating_shapes():
    """Shape switches: a switch forces an eager run (exact cross-run
    gates), repeats replay the cached graph; eager and graph runs must
    interleave with exact doorbell bookkeeping."""
    def fn(ctx, rank):
        algo = ga._C.HipAllreduceRing(ctx, 0)
        shapes = [1_000_000, 700_000, 1_000_000, 1_000_000,
                  700_000, 700_000, 1_000_000]
        for it, n in enumerate(shapes):
            g = torch.Generator("cpu").manual_seed(rank * 100 + it)
            x = _dev(torch.rand(n, generator=g))
            ref = sum(
                torch.rand(n, generator=torch.Generator(
                    "cpu").manual_seed(r * 100 + it))
                for r in range(2))
            algo.run(x.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
            assert torch.allclose(_host(x), ref, atol=1e-4), (rank, it, n)
            ga.barrier(ctx, tag=985)

    _two_rank_generic(fn)


def test_hip_p2p_bidirectional_chunked():
    """HipP2P: both ranks post a send AND a recv before flushing either
    (batch_isend_irecv pattern); 20MB payloads exercise the multi-chunk
    double-buffered lane protocol."""
    def fn(ctx, rank):
        n = 5_000_000  # 20 MB > 2 x 4MB lane slots -> needs acks
        peer = 1 - rank
        src = _dev(torch.full((n,), float(rank + 1)))
        dst = torch.zeros(n, device="cuda")
        eng = ga._C.HipP2P(ctx, 0)
        for it in range(2):  # cross-run lane seq reuse
            eng.post_send(peer, src.data_ptr(), n * 4)
            eng.post_recv(peer, dst.data_ptr(), n * 4)
            eng.flush_sends()
            eng.flush_recvs()
            assert torch.all(_host(dst) == peer + 1), (rank, it)
            ga.barrier(ctx, tag=971 + it)

    _two_rank_generic(fn)


def test_pg_cuda_collectives():
    """ProcessGroup with CUDA tensors: device-native paths."""
    from gloo_amd.pg import ProcessGroupGlooAmd

    import threading

    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            torch.cuda.set_device(0)
            pg = ProcessGroupGlooAmd(store, rank, 2)
            t = torch.full((700_000,), float(rank + 1), device="cuda")
            pg.allreduce([t]).wait()
            assert torch.all(t == 3.0)

            inp = torch.full((100_000,), float(rank), device="cuda")
            out = torch.zeros(200_000, device="cuda")
            pg._allgather_base(out, inp).wait()
            assert torch.all(out[:100_000] == 0) and torch.all(
                out[100_000:] == 1)

            rs_in = torch.arange(200_000, dtype=torch.float32,
                                 device="cuda") + rank
            rs_out = torch.zeros(100_000, device="cuda")
            pg._reduce_scatter_base(rs_out, rs_in).wait()
            expect = (torch.arange(200_000, dtype=torch.float32) * 2 +
                      1)[rank * 100_000:(rank + 1) * 100_000]
            assert torch.allclose(_host(rs_out), expect)

            a2a_in = torch.cat([
                torch.full((50_000,), float(rank * 2 + d), device="cuda")
                for d in range(2)])
            a2a_out = torch.zeros(100_000, device="cuda")
            pg.alltoall_base(a2a_out, a2a_in, [], []).wait()
            for s in range(2):
                assert torch.all(
                    a2a_out[s * 50_000:(s + 1) * 50_000] == s * 2 + rank)

            b = (torch.arange(50_000, dtype=torch.float32, device="cuda")
                 if rank == 0 else torch.zeros(50_000, device="cuda"))
            pg.broadcast([b]).wait()
            assert torch.allclose(
                _host(b), torch.arange(50_000, dtype=torch.float32))

            # host-staged CUDA paths: reduce, gather, scatter, p2p
            red = torch.full((30_000,), float(rank + 1), device="cuda")
            import torch.distributed as dist

            class _RO:
                rootRank = 0
                reduceOp = dist.ReduceOp.SUM
            pg.reduce([red], _RO()).wait()
            if rank == 0:
                assert torch.all(red == 3.0)

            g_in = torch.full((10_000,), float(rank + 5), device="cuda")
            if rank == 0:
                gouts = [[torch.zeros(10_000, device="cuda")
                          for _ in range(2)]]
                pg.gather(gouts, [g_in]).wait()
                assert torch.all(gouts[0][0] == 5) and torch.all(
                    gouts[0][1] == 6)
            else:
                pg.gather([], [g_in]).wait()

            s_out = torch.zeros(10_000, device="cuda")
            if rank == 0:
                s_ins = [[torch.full((10_000,), float(10 + d),
                                     device="cuda") for d in range(2)]]
                pg.scatter([s_out], s_ins).wait()
            else:
                pg.scatter([s_out], []).wait()
            assert torch.all(s_out == 10 + rank)

            if rank == 0:
                pg.send([torch.full((20_000,), 9.0, device="cuda")],
                        1, tag=4).wait()
            else:
                pr = torch.zeros(20_000, device="cuda")
                pg.recv([pr], 0, tag=4).wait()
                assert torch.all(pr == 9.0)
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(traceback.format_exc())

    ths = [threading.Thread(target=worker, args=(r,), daemon=True)
           for r in range(2)]
    [t.start() for t in ths]
    [t.join(100) for t in ths]
    assert not errors, errors[0]


def test_host_staged_small():
    """Below the on-device threshold the ring takes the host-staged path."""
    _two_rank_device_test("ring", 1000)  # 4 KB << 256 KB threshold


def test_cross_process_ipc_8rank():
    """Eight processes on one GPU (timeshared): the full node-scale
    multi-ring (strides 1,3,5,7) and P=8 direct engine over IPC — the
    same schedules the driver's 8-GPU scaling run executes."""
    worker = os.path.join(os.path.dirname(__file__), "ipc_worker.py")
    tmp = "/tmp/ga_ipc8_%d" % os.getpid()
    os.makedirs(tmp, exist_ok=True)
    env = dict(os.environ)
    env["HSA_ENABLE_IPC_MODE_LEGACY"] = "0"
    procs = [
        subprocess.Popen(
            [sys.executable, worker, str(r), "8", tmp],
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, env=env)
        for r in range(8)
    ]
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=300)
        outs.append(out.decode())
    assert all(p.returncode == 0 for p in procs), "\n".join(outs)
    assert all("IPC-OK" in o for o in outs), "\n".join(outs)


def test_cross_process_ipc_4rank():
    """Four processes on one GPU: exercises the multi-ring (stride 1,3)
    chunked allreduce over IPC."""
    worker = os.path.join(os.path.dirname(__file__), "ipc_worker.py")
    tmp = "/tmp/ga_ipc4_%d" % os.getpid()
    os.makedirs(tmp, exist_ok=True)
    env = dict(os.environ)
    env["HSA_ENABLE_IPC_MODE_LEGACY"] = "0"
    procs = [
        subprocess.Popen(
            [sys.executable, worker, str(r), "4", tmp],
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, env=env)
        for r in range(4)
    ]
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=300)
        outs.append(out.decode())
    assert all(p.returncode == 0 for p in procs), "\n".join(outs)
    assert all("IPC-OK" in o for o in outs), "\n".join(outs)


def test_cross_process_ipc():
    """Two real processes on one GPU: hipIpcMemHandle + doorbells."""
    worker = os.path.join(os.path.dirname(__file__), "ipc_worker.py")
    tmp = "/tmp/ga_ipc_test_%d" % os.getpid()
    os.makedirs(tmp, exist_ok=True)
    env = dict(os.environ)
    env["HSA_ENABLE_IPC_MODE_LEGACY"] = "0"
    procs = [
        subprocess.Popen(
            [sys.executable, worker, str(r), "2", tmp],
            stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT,
            env=env,
        )
        for r in range(2)
    ]
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=300)
        outs.append(out.decode())
    assert all(p.returncode == 0 for p in procs), "\n".join(outs)
    assert all("IPC-OK" in o for o in outs), "\n".join(outs)
