#!/usr/bin/env python3
"""Flagship benchmark: hip_allreduce_ring_chunked, fp32, 5M elements
(the largest row of the reference's published latency table,
BASELINE.md). One "step" = one in-place allreduce of the buffer across
all N GPUs over xGMI/IPC (host-staged TCP path on CPU-only machines).

Launched directly (N=1) or via torch.distributed.run with one rank per
GPU (reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT).

Prints ONE JSON line on rank 0:
  value  = aggregate reduced elements/s over the whole job (N * elements
           / mean step time), weak scaling (per-GPU buffer fixed).
  extras = p50/p99 step latency in us + derived bus bandwidth, for
           comparison against the reference table (vs_baseline = ref p50
           10608 us at 5M fp32 / our p50).
"""
import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

REF_P50_US_5M = 10608.0  # allreduce_ring_chunked 5M fp32 p50


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--elements", type=int, default=5_000_000)
    ap.add_argument("--algorithm", default="ring_chunked",
                    choices=["ring", "ring_chunked", "halving_doubling", "direct"])
    ap.add_argument("--dtype", default="f32", choices=["f32", "bf16", "f16"])
    ap.add_argument("--sweep", action="store_true",
                    help="also print the reference element sweep to stderr")
    args = ap.parse_args()

    import gloo_amd as ga

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    master_addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
    master_port = int(os.environ.get("MASTER_PORT", "29500")) + 1

    have_gpu = False
    torch = None
    try:
        import torch  # noqa: F811

        have_gpu = torch.cuda.is_available()
    except ImportError:
        pass

    if world > 1:
        store = ga.TcpStore(master_addr, master_port, is_server=(rank == 0))
    else:
        store = ga.HashStore()
    # CPU-only fallback is single-node by construction: use the
    # unix-socket transport (GPU runs keep TCP for the control plane).
    use_uds = (not have_gpu) and master_addr in ("127.0.0.1", "localhost")
    dev = ga.create_tcp_device(
        master_addr if world > 1 else "", use_uds=use_uds)
    ctx = ga.Context(rank, world)
    ctx.connect_full_mesh(store, dev)
    ctx.set_timeout(120000)

    tmap = {"f32": "float32", "bf16": "bfloat16", "f16": "float16"}
    gmap = {"f32": ga.DType.f32, "bf16": ga.DType.bf16, "f16": ga.DType.f16}
    gdt = gmap[args.dtype]
    esize = ga.dtype_size(gdt)

    if have_gpu:
        # one GPU per rank; wrap on boxes with fewer GPUs than ranks
        local_rank = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(local_rank)
        tdt = getattr(torch, tmap[args.dtype])
        buf = torch.rand(args.elements, dtype=torch.float32,
                         device="cuda").to(tdt)
        if world == 1:
            # Single GPU: the ring degenerates; bench the local multi-
            # pointer allreduce instead (gloo AllreduceLocal semantics:
            # fused 2-input reduction kernel + broadcast copy, HBM-bound).
            buf2 = torch.rand(args.elements, dtype=torch.float32,
                              device="cuda").to(tdt)

            def step(n=args.elements, ptr=None):
                ga._C.hip_allreduce_local(
                    [ptr if ptr is not None else buf.data_ptr(),
                     buf2.data_ptr()],
                    n, gdt, ga.ReduceOp.sum, local_rank)
        else:
            if args.algorithm == "direct" and world <= 8:
                algo = ga._C.HipAllreduceDirect(ctx, local_rank)
            elif (args.algorithm == "halving_doubling"
                    and world & (world - 1) == 0):
                algo = ga._C.HipAllreduceHalvingDoubling(ctx, local_rank)
            else:
                algo = ga._C.HipAllreduceRing(
                    ctx, local_rank, chunked=(args.algorithm != "ring"))

            def step(n=args.elements, ptr=None):
                algo.run(ptr if ptr is not None else buf.data_ptr(), n, gdt,
                         ga.ReduceOp.sum)

        def sync():
            torch.cuda.synchronize()
    else:
        import numpy as np

        buf = np.random.rand(args.elements).astype(np.float32)
        assert args.dtype == "f32", "CPU fallback benches fp32 only"

        def step(n=args.elements, ptr=None):
            ga.allreduce(ctx, [ptr if ptr is not None else buf.ctypes.data],
                         n, ga.DType.f32, ga.ReduceOp.sum)

        def sync():
            pass

    def barrier():
        ga.barrier(ctx, tag=ctx.next_slot())

    # warmup
    for _ in range(args.warmup):
        step()
    sync()
    barrier()
    sync()

    # timed region: exactly K steps
    lat = []
    t0 = time.perf_counter()
    for _ in range(args.steps):
        s = time.perf_counter()
        step()
        sync()
        lat.append(time.perf_counter() - s)
    sync()
    barrier()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks
    import numpy as np

    emax = np.array([elapsed], dtype=np.float64)
    ga.allreduce(ctx, [emax.ctypes.data], 1, ga.DType.f64, ga.ReduceOp.max,
                 tag=ctx.next_slot())
    elapsed = float(emax[0])

    # Supplementary: time the fully-connected direct engine too (the
    # fastest allreduce on all-to-all xGMI); reported in config, the
    # headline metric stays the named ring_chunked.
    direct_p50_us = None
    if have_gpu and world > 1 and world <= 8 \
            and args.algorithm == "ring_chunked":
        dalgo = ga._C.HipAllreduceDirect(ctx, local_rank)
        for _ in range(max(3, args.warmup // 2)):
            dalgo.run(buf.data_ptr(), args.elements, gdt, ga.ReduceOp.sum)
        dlat = []
        for _ in range(args.steps):
            s = time.perf_counter()
            dalgo.run(buf.data_ptr(), args.elements, gdt, ga.ReduceOp.sum)
            sync()
            dlat.append(time.perf_counter() - s)
        direct_p50_us = round(statistics.median(dlat) * 1e6, 1)

    ms_per_step = elapsed / args.steps * 1000.0
    p50_us = statistics.median(lat) * 1e6
    p99_us = sorted(lat)[max(0, int(len(lat) * 0.99) - 1)] * 1e6
    nbytes = args.elements * esize
    # ring allreduce wire bytes per rank ~= 2 * S * (P-1)/P
    bus_gbps = (2.0 * nbytes * max(world - 1, 1) / max(world, 1)) / (
        elapsed / args.steps) / 1e9
    value = world * args.elements / (elapsed / args.steps)

    if args.sweep:  # collective: every rank steps, rank 0 reports
        _sweep(rank, step, sync, args)

    if rank == 0:
        out = {
            "metric": "allreduce_ring_chunked p50 us + bus GB/s, fp32 "
                      "element sweep, 1/2/4/8 GPUs",
            "value": round(value / 1e6, 3),
            "unit": "Melem_reduced/s_aggregate",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(REF_P50_US_5M / p50_us, 2)
            if args.elements == 5_000_000 and args.dtype == "f32" else None,
            "dtype": args.dtype if args.dtype != "f32" else "fp32",
            "data": "synthetic",
            "config": {
                "collective": args.algorithm,
                "elements": args.elements,
                "payload_mb": round(nbytes / 1e6, 1),
                "transport": ("xgmi-ipc" if have_gpu else
                              ("uds-local" if use_uds else "tcp-localhost")),
                "p50_us": round(p50_us, 1),
                "p99_us": round(p99_us, 1),
                "bus_GBps": round(bus_gbps, 2),
                "direct_p50_us": direct_p50_us,
                "parallelism": f"ring{world}",
            },
        }
        print(json.dumps(out), flush=True)


def _sweep(rank, step_fn, sync, args):
    """Reference README element sweep (1 .. 5M fp32) -> stderr."""
    if rank == 0:
        print("# elements p50_us p99_us samples", file=sys.stderr)
    for n in [1, 10, 100, 1000, 10000, 100000, 1000000, 2000000, 5000000]:
        if n > args.elements:
            break
        iters = max(5, min(200, int(2e6 / max(n, 1000))))
        lat = []
        for _ in range(iters):
            s = time.perf_counter()
            step_fn(n=n)
            sync()
            lat.append((time.perf_counter() - s) * 1e6)
        lat.sort()
        if rank == 0:
            print(
                f"{n} {lat[len(lat)//2]:.1f} "
                f"{lat[int(len(lat)*0.99)-1]:.1f} {len(lat)}",
                file=sys.stderr)


if __name__ == "__main__":
    main()
