#!/bin/bash
# Round-2 GPU call 2: diagnose the timeshared N=4/8 collapse.
# Hypothesis: hipStreamWaitValue packets halt HW queues; 8 streams alias
# onto GPU_MAX_HW_QUEUES=4 -> convoy. Try kernel doorbells / more HW
# queues / fewer rings.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

run() { # name N extra-env...
  local name=$1 N=$2; shift 2
  env "$@" timeout 420 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node $N --master-addr 127.0.0.1 --master-port 29500 \
    bench.py --gpus $N --steps 10 --warmup 3 \
    > gpurun_out/tune_$name.json 2> gpurun_out/tune_$name.log
  echo "${name}_EXIT=$?"
}

run kq8 8 GLOO_AMD_STREAM_OPS=0 GLOO_AMD_NUM_RINGS=4
run kq8r2 8 GLOO_AMD_STREAM_OPS=0 GLOO_AMD_NUM_RINGS=2
run kq8r1 8 GLOO_AMD_STREAM_OPS=0 GLOO_AMD_NUM_RINGS=1
run sq8h8 8 GPU_MAX_HW_QUEUES=8 GLOO_AMD_NUM_RINGS=4
run sq8r1 8 GLOO_AMD_NUM_RINGS=1
run kq4 4 GLOO_AMD_STREAM_OPS=0
run sq4h8 4 GPU_MAX_HW_QUEUES=8

for f in gpurun_out/tune_*.json; do
  echo "== $f"; cat $f | python -c "
import json,sys
try:
  d=json.load(sys.stdin); c=d['config']
  print(d['n_gpus'], c['p50_us'], 'us p50, direct', c['direct_p50_us'])
except Exception as e: print('parse fail', e)"
done
