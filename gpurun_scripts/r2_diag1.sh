#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m pytest tests/test_gpu.py -q -p no:cacheprovider \
  -k "copy_engine or combine or allreduce_single or stream_engine" \
  --timeout=120 --timeout-method=thread --tb=short -rf \
  > gpurun_out/diag1.log 2>&1
echo RC=$?
tail -60 gpurun_out/diag1.log
