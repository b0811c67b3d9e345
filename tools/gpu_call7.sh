#!/bin/bash
# Round-2 GPU call 7: SDMA-copy traces (memory-copy domain) for the
# direct and ring engines -> overlap evidence.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd /root/repo

timeout 600 rocprofv3 --kernel-trace --memory-copy-trace --stats \
  -d gpurun_out/prof_d7 \
  -- python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29500 \
  bench.py --gpus 2 --steps 8 --warmup 3 --algorithm direct \
  --elements 100000000 > gpurun_out/d7.json 2> gpurun_out/d7.log
echo "D7_EXIT=$?"
python3 tools/summarize_prof.py gpurun_out/prof_d7 \
  > gpurun_out/prof_d7_summary.txt 2>&1
rm -rf gpurun_out/prof_d7

timeout 600 rocprofv3 --kernel-trace --memory-copy-trace --stats \
  -d gpurun_out/prof_r7 \
  -- python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29501 \
  bench.py --gpus 2 --steps 8 --warmup 3 \
  --elements 100000000 > gpurun_out/r7.json 2> gpurun_out/r7.log
echo "R7_EXIT=$?"
python3 tools/summarize_prof.py gpurun_out/prof_r7 \
  > gpurun_out/prof_r7_summary.txt 2>&1
rm -rf gpurun_out/prof_r7

cat gpurun_out/prof_d7_summary.txt | head -22
cat gpurun_out/prof_r7_summary.txt | head -22
