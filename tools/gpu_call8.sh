#!/bin/bash
# Round-2 GPU call 8: full suite with the new soak tests + a 4-rank
# cross-process mixed-collective stress loop.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

timeout 1500 python -m pytest tests -m gpu -x -q \
  > gpurun_out/pytest_gpu8.log 2>&1
echo "PYTEST_EXIT=$?"

# 4-rank cross-process mixed stress: 60 iterations of ring + direct +
# allgather + reduce_scatter with verification (ipc_worker-style inline)
cat > /tmp/stress4.py <<'EOF'
import os, sys, torch
sys.path.insert(0, "/root/repo")
import gloo_amd as ga
rank, size = int(sys.argv[1]), int(sys.argv[2])
store = ga.FileStore(sys.argv[3])
dev = ga.create_tcp_device()
ctx = ga.Context(rank, size)
ctx.connect_full_mesh(store, dev)
ctx.set_timeout(120000)
torch.cuda.set_device(0)
os.environ.setdefault("GLOO_AMD_NUM_RINGS", "1")
ring = ga._C.HipAllreduceRing(ctx, 0)
direct = ga._C.HipAllreduceDirect(ctx, 0)
ag = ga._C.HipAllgatherRing(ctx, 0)
n = 400_000
for it in range(60):
    g = torch.Generator("cpu").manual_seed(rank * 1000 + it)
    x = torch.rand(n, generator=g).cuda()
    ref = sum(torch.rand(n, generator=torch.Generator("cpu").manual_seed(
        r * 1000 + it)) for r in range(size))
    algo = ring if it % 2 == 0 else direct
    algo.run(x.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
    got = torch.empty(n, pin_memory=True); got.copy_(x)
    assert torch.allclose(got, ref, atol=1e-4), (rank, it)
    if it % 10 == 0:
        inp = torch.full((50_000,), float(rank + it), device="cuda")
        out = torch.zeros(50_000 * size, device="cuda")
        ag.run(inp.data_ptr(), out.data_ptr(), 50_000, 4)
        h = torch.empty(50_000 * size, pin_memory=True); h.copy_(out)
        for s in range(size):
            assert torch.all(h[s*50_000:(s+1)*50_000] == s + it), (rank, it, s)
    ga.barrier(ctx, tag=700 + it)
print("STRESS-OK", rank)
EOF
TMP=/tmp/ga_stress4_$$
mkdir -p $TMP
for r in 0 1 2 3; do
  timeout 900 python /tmp/stress4.py $r 4 $TMP \
    > gpurun_out/stress4_r$r.log 2>&1 &
done
wait
grep -l "STRESS-OK" gpurun_out/stress4_r*.log | wc -l
tail -n 3 gpurun_out/pytest_gpu8.log
tail -n 2 gpurun_out/stress4_r0.log gpurun_out/stress4_r3.log
