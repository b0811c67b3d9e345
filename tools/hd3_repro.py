#!/usr/bin/env python3
"""Repro loop for the 3-rank non-pow2 HD fold deadlock: run the
threaded 3-rank HD allreduce many times with a short timeout and
GLOO_AMD_FLAG_DEBUG so a hang dumps every rank's doorbell flags.

Flag layout at P=3 (T=1): 0,1=fDATA 2,3=fACK 4=fAGD 5,6=fFOLD
7,8=fFACK 9=fPOST 10=fPACK.
"""
import faulthandler
import os
import sys
import threading
import time

os.environ.setdefault("GLOO_AMD_FLAG_DEBUG", "1")
sys.path.insert(0, "/root/repo")

import torch  # noqa: E402
import gloo_amd as ga  # noqa: E402

faulthandler.dump_traceback_later(75, repeat=True)
torch.cuda.set_device(0)


SIZE = int(os.environ.get("REPRO_SIZE", "3"))
ENGINE = os.environ.get("REPRO_ENGINE", "hd")


def one_round(ri):
    store = ga.HashStore()
    errors = []

    def worker(rank):
        try:
            dev = ga.create_tcp_device()
            ctx = ga.Context(rank, SIZE)
            ctx.connect_full_mesh(store, dev)
            ctx.set_timeout(20000)
            n = 1_500_000
            g = torch.Generator("cpu").manual_seed(rank)
            x = torch.rand(n, generator=g).cuda()
            ref = sum(
                torch.rand(n, generator=torch.Generator("cpu").manual_seed(r))
                for r in range(SIZE))
            if ENGINE == "hd":
                algo = ga._C.HipAllreduceHalvingDoubling(ctx, 0)
            elif ENGINE == "ring":
                algo = ga._C.HipAllreduceRing(ctx, 0)
            else:
                algo = ga._C.HipAllreduceDirect(ctx, 0)
            y = x.clone()
            for it in range(5):
                y.copy_(x)
                algo.run(y.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
                got = torch.empty(n, pin_memory=True)
                got.copy_(y)
                assert torch.allclose(got, ref, atol=1e-4), (rank, it)
                ga.barrier(ctx, tag=921 + it)
        except Exception:  # noqa: BLE001
            import traceback

            errors.append(f"[round {ri} rank {rank}]\n"
                          + traceback.format_exc())

    t0 = time.time()
    ths = [threading.Thread(target=worker, args=(r,), daemon=True)
           for r in range(SIZE)]
    [t.start() for t in ths]
    [t.join(90) for t in ths]
    alive = [t.is_alive() for t in ths]
    dt = time.time() - t0
    if errors or any(alive):
        print(f"ROUND {ri} FAILED alive={alive} dt={dt:.1f}s")
        for e in errors:
            print(e)
        return False
    print(f"round {ri} ok dt={dt:.1f}s", flush=True)
    return True


fails = 0
for ri in range(int(sys.argv[1]) if len(sys.argv) > 1 else 15):
    if not one_round(ri):
        fails += 1
        if fails >= 2:
            break
print("DONE fails=", fails)
