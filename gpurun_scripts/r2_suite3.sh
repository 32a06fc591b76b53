#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
for i in 1 2 3; do
  timeout 900 python -m pytest tests/ -q -m gpu -p no:cacheprovider \
    > gpurun_out/r2s3_$i.log 2>&1
  echo "SUITE$i RC=$?"
  tail -2 gpurun_out/r2s3_$i.log
done
grep -h "re-import" gpurun_out/r2s3_*.log | head
grep -h -A12 "engine flow dump" gpurun_out/r2s3_*.log | head -40
