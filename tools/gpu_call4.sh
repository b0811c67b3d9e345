#!/bin/bash
# Round-2 GPU call 4: SDMA-overlap trace of the direct + ring engines,
# per-algorithm element sweeps (crossover table), bf16 1e8 row.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd /root/repo

# 1) rocprof trace of a 2-rank direct allreduce at 1e8 fp32 (400MB:
#    deep chunk pipeline) — overlap evidence
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_direct \
  -- python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29500 \
  bench.py --gpus 2 --steps 10 --warmup 3 --algorithm direct \
  --elements 100000000 > gpurun_out/direct_1e8.json \
  2> gpurun_out/direct_1e8.log
echo "DIRECT_EXIT=$?"

# 2) same for graph-replayed ring at 1e8
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_ring \
  -- python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29501 \
  bench.py --gpus 2 --steps 10 --warmup 3 \
  --elements 100000000 > gpurun_out/ring_1e8.json \
  2> gpurun_out/ring_1e8.log
echo "RING_EXIT=$?"

# 3) crossover: per-algorithm element sweep at N=2
for ALGO in ring_chunked direct halving_doubling; do
  timeout 500 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29502 \
    bench.py --gpus 2 --steps 10 --warmup 3 --algorithm $ALGO --sweep \
    > gpurun_out/sweep_$ALGO.json 2> gpurun_out/sweep_$ALGO.log
  echo "${ALGO}_EXIT=$?"
done

# 4) BASELINE config 5: bf16 1e8 elements
timeout 500 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29503 \
  bench.py --gpus 2 --steps 10 --warmup 3 --dtype bf16 \
  --elements 100000000 > gpurun_out/bf16_1e8.json \
  2> gpurun_out/bf16_1e8.log
echo "BF16_EXIT=$?"

python3 tools/summarize_prof.py gpurun_out/prof_direct \
  > gpurun_out/prof_direct_summary.txt 2>&1
python3 tools/summarize_prof.py gpurun_out/prof_ring \
  > gpurun_out/prof_ring_summary.txt 2>&1
# keep dbs off the 64MB merge budget
rm -rf gpurun_out/prof_direct gpurun_out/prof_ring
tail -n 6 gpurun_out/prof_direct_summary.txt
tail -n 6 gpurun_out/prof_ring_summary.txt
tail -n 2 gpurun_out/direct_1e8.json gpurun_out/ring_1e8.json \
  gpurun_out/bf16_1e8.json
grep -A11 "# elements" gpurun_out/sweep_ring_chunked.log | head -12
