#!/bin/bash
# Round-2 GPU call 3: graph-mode validation + rocprof evidence.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

# 1) full GPU suite (graph mode ON by default: the 2nd run of each
#    repeated-shape test goes through capture+replay)
export GLOO_AMD_LOG_LEVEL=INFO
timeout 1200 python -m pytest tests -m gpu -x -q \
  > gpurun_out/pytest_gpu3.log 2>&1
echo "PYTEST_EXIT=$?"
grep -c "captured ring schedule graph" gpurun_out/pytest_gpu3.log

# 2) bench N=2: graph vs eager
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29500 \
  bench.py --gpus 2 --steps 30 --warmup 10 \
  > gpurun_out/bench_n2_graph.json 2> gpurun_out/bench_n2_graph.log
echo "GRAPH_EXIT=$?"
GLOO_AMD_GRAPH=0 timeout 420 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29501 \
  bench.py --gpus 2 --steps 30 --warmup 10 \
  > gpurun_out/bench_n2_eager.json 2> gpurun_out/bench_n2_eager.log
echo "EAGER_EXIT=$?"
# deep-segment case where enqueue cost dominates: small inbox via env
GLOO_AMD_NUM_RINGS=1 timeout 420 python -m torch.distributed.run \
  --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 \
  --master-port 29502 bench.py --gpus 2 --steps 30 --warmup 10 \
  > gpurun_out/bench_n2_r1.json 2> gpurun_out/bench_n2_r1.log
echo "R1_EXIT=$?"

# 3) rocprof: kernel-level evidence for reduce2/reduceN rooflines
#    (fp32 + bf16) and the local fused allreduce
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_kernels \
  -- python -c "
import torch, gloo_amd as ga
torch.cuda.set_device(0)
n = 64_000_000
for dt, gdt in [(torch.float32, ga.DType.f32), (torch.bfloat16, ga.DType.bf16), (torch.float16, ga.DType.f16)]:
    a = torch.rand(n, dtype=torch.float32, device='cuda').to(dt)
    b = torch.rand(n, dtype=torch.float32, device='cuda').to(dt)
    d = torch.empty_like(a)
    for _ in range(20):
        ga._C.hip_reduce2(d.data_ptr(), a.data_ptr(), b.data_ptr(), n, gdt, ga.ReduceOp.sum)
    torch.cuda.synchronize()
ts = [torch.rand(n//4, dtype=torch.float32, device='cuda') for _ in range(8)]
for _ in range(20):
    ga._C.hip_allreduce_local([t.data_ptr() for t in ts], n//4, ga.DType.f32, ga.ReduceOp.sum, 0)
torch.cuda.synchronize()
print('KERNELS-DONE')
" > gpurun_out/prof_kernels.log 2>&1
echo "PROF_EXIT=$?"
ls gpurun_out/prof_kernels* 2>/dev/null | head -5
tail -n 3 gpurun_out/bench_n2_graph.json gpurun_out/bench_n2_eager.json \
  gpurun_out/bench_n2_r1.json
grep -h "captured ring" gpurun_out/bench_n2_graph.log | head -2
tail -n 4 gpurun_out/pytest_gpu3.log
