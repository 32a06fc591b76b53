#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
: > gpurun_out/r2_flaky.log
for i in $(seq 1 10); do
  timeout 240 python -m pytest \
    "tests/test_gpu.py::test_two_ranks_one_gpu[allgather_rs]" \
    "tests/test_gpu.py::test_two_ranks_extra[subcomm]" \
    -q -p no:cacheprovider >> gpurun_out/r2_flaky.log 2>&1
  echo "ITER$i RC=$?" >> gpurun_out/r2_flaky.log
done
grep -E "RC=|failed|passed" gpurun_out/r2_flaky.log | tail -25
grep -A14 "engine flow dump" gpurun_out/r2_flaky.log | head -80
