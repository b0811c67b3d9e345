#!/bin/bash
# Round-2 GPU call 5: aligned-split validation + re-profile.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd /root/repo

timeout 1200 python -m pytest tests -m gpu -x -q \
  > gpurun_out/pytest_gpu5.log 2>&1
echo "PYTEST_EXIT=$?"

for CASE in "ring_chunked 100000000" "direct 100000000" \
            "halving_doubling 100000000" "ring_chunked 5000000"; do
  set -- $CASE
  timeout 420 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29500 \
    bench.py --gpus 2 --steps 10 --warmup 3 --algorithm $1 \
    --elements $2 > gpurun_out/a5_$1_$2.json 2> gpurun_out/a5_$1_$2.log
  echo "$1_$2_EXIT=$?"
done

# rocprof re-trace of the ring at 1e8: expect zero scalar kernels
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_ring5 \
  -- python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29501 \
  bench.py --gpus 2 --steps 8 --warmup 3 --elements 100000000 \
  > gpurun_out/ring5_1e8.json 2> gpurun_out/ring5_1e8.log
echo "PROF_EXIT=$?"
python3 tools/summarize_prof.py gpurun_out/prof_ring5 \
  > gpurun_out/prof_ring5_summary.txt 2>&1
rm -rf gpurun_out/prof_ring5

# README element sweep at N=2 (graph-cached shapes)
timeout 500 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29502 \
  bench.py --gpus 2 --steps 10 --warmup 3 --sweep \
  > gpurun_out/sweep5.json 2> gpurun_out/sweep5.log
echo "SWEEP_EXIT=$?"

tail -n 2 gpurun_out/a5_*.json
cat gpurun_out/prof_ring5_summary.txt | head -16
grep -A11 "# elements" gpurun_out/sweep5.log | head -12
tail -n 3 gpurun_out/pytest_gpu5.log
