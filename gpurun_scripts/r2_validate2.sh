#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 1200 python -m pytest tests/ -q -m gpu -p no:cacheprovider \
  > gpurun_out/r2v2_pytest.log 2>&1
echo "PYTEST_RC=$?" | tee -a gpurun_out/r2v2_pytest.log
timeout 600 python bench.py --steps 10 --warmup 3 --sweep \
  > gpurun_out/r2v2_bench.log 2>&1
echo "BENCH_RC=$?" >> gpurun_out/r2v2_bench.log
tail -16 gpurun_out/r2v2_bench.log
tail -6 gpurun_out/r2v2_pytest.log
