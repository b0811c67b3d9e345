#!/bin/bash
# Round-2 GPU call 6: UDS control-plane validation on hardware, direct
# engine overlap trace, 100K-element anomaly isolation.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd /root/repo

# 1) UDS control plane with the GPU data plane (single-node always)
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29500 \
  bench.py --gpus 2 --steps 20 --warmup 5 --uds 1 \
  > gpurun_out/uds_gpu.json 2> gpurun_out/uds_gpu.log
echo "UDS_EXIT=$?"

# 2) direct engine at 1e8: blit-copy/reduce overlap evidence
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_d6 \
  -- python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29501 \
  bench.py --gpus 2 --steps 8 --warmup 3 --algorithm direct \
  --elements 100000000 > gpurun_out/d6.json 2> gpurun_out/d6.log
echo "D6_EXIT=$?"
python3 tools/summarize_prof.py gpurun_out/prof_d6 \
  > gpurun_out/prof_d6_summary.txt 2>&1
rm -rf gpurun_out/prof_d6

# 3) 100K anomaly: stream-ops off / graphs off isolation
for MODE in "GLOO_AMD_STREAM_OPS=0" "GLOO_AMD_GRAPH=0" ""; do
  env $MODE timeout 420 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29502 \
    bench.py --gpus 2 --steps 5 --warmup 2 --elements 100000 \
    > "gpurun_out/e100k_${MODE:-default}.json" \
    2> "gpurun_out/e100k_${MODE:-default}.log"
  echo "100K_${MODE:-default}_EXIT=$?"
done

tail -n 2 gpurun_out/uds_gpu.json gpurun_out/d6.json \
  gpurun_out/e100k_*.json
cat gpurun_out/prof_d6_summary.txt | head -20
