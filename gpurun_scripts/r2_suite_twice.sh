#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
for i in 1 2; do
  timeout 900 python -m pytest tests/ -q -m gpu -p no:cacheprovider \
    > gpurun_out/r2_suite$i.log 2>&1
  echo "SUITE$i RC=$?" >> gpurun_out/r2_suite$i.log
  tail -4 gpurun_out/r2_suite$i.log
done
grep -h -A14 "engine flow dump" gpurun_out/r2_suite1.log gpurun_out/r2_suite2.log | head -90
