#!/bin/bash
# Round-2 GPU call 1: full GPU suite + driver-style bench preflight.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

# 1) full GPU test suite (new: non-pow2 HD, multi-input engines)
timeout 900 python -m pytest tests -m gpu -x -q \
  > gpurun_out/pytest_gpu.log 2>&1
echo "PYTEST_EXIT=$?"

# 2) bench exactly as the driver runs it: N=1 then torchrun 2/4/8
#    (ranks timeshare the one GPU; protocol identical to 8 GPUs)
export GLOO_AMD_LOG_LEVEL=INFO
timeout 300 python bench.py --steps 20 --warmup 5 \
  > gpurun_out/bench_n1.json 2> gpurun_out/bench_n1.log
echo "N1_EXIT=$?"
for N in 2 4 8; do
  timeout 420 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node $N --master-addr 127.0.0.1 --master-port 29500 \
    bench.py --gpus $N --steps 10 --warmup 3 \
    > gpurun_out/bench_n$N.json 2> gpurun_out/bench_n$N.log
  echo "N${N}_EXIT=$?"
done

# 3) BASELINE configs 3-5 smoke: HD (incl. non-pow2 3 ranks), ops, bf16
timeout 300 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 3 --master-addr 127.0.0.1 --master-port 29502 \
  bench.py --gpus 3 --steps 5 --warmup 2 --algorithm halving_doubling \
  > gpurun_out/bench_hd3.json 2> gpurun_out/bench_hd3.log
echo "HD3_EXIT=$?"
for OP in allgather reduce_scatter alltoall; do
  timeout 300 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29503 \
    bench.py --gpus 2 --steps 5 --warmup 2 --op $OP --elements 2000000 \
    > gpurun_out/bench_$OP.json 2> gpurun_out/bench_$OP.log
  echo "${OP}_EXIT=$?"
done
timeout 300 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29504 \
  bench.py --gpus 2 --steps 5 --warmup 2 --dtype bf16 --elements 20000000 \
  > gpurun_out/bench_bf16.json 2> gpurun_out/bench_bf16.log
echo "BF16_EXIT=$?"

tail -n 3 gpurun_out/bench_n*.json gpurun_out/bench_hd3.json \
  gpurun_out/bench_allgather.json gpurun_out/bench_reduce_scatter.json \
  gpurun_out/bench_alltoall.json gpurun_out/bench_bf16.json
grep -h "doorbell path" gpurun_out/*.log | head -2
tail -5 gpurun_out/pytest_gpu.log
