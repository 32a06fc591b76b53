#!/bin/bash
# Round-2 first GPU validation: doorbell-visibility fix + full suite (no -x)
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
# 1) full GPU suite WITHOUT -x
timeout 900 python -m pytest tests/ -q -m gpu -p no:cacheprovider \
  > gpurun_out/r2_pytest_gpu.log 2>&1
echo "PYTEST_RC=$?" | tee -a gpurun_out/r2_pytest_gpu.log
# 2) repro loop of the round-1 failing test (5x)
for i in 1 2 3 4 5; do
  timeout 240 python -m pytest "tests/test_gpu.py::test_two_ranks_one_gpu[allgather_rs]" \
    -q -p no:cacheprovider >> gpurun_out/r2_coll2_loop.log 2>&1
  echo "ITER$i RC=$?" >> gpurun_out/r2_coll2_loop.log
done
# 3) P=1 bench sweep (perf after fix)
timeout 600 python bench.py --steps 10 --warmup 3 --sweep \
  > gpurun_out/r2_bench_sweep.log 2>&1
echo "BENCH_RC=$?" >> gpurun_out/r2_bench_sweep.log
tail -3 gpurun_out/r2_bench_sweep.log
tail -20 gpurun_out/r2_pytest_gpu.log
grep RC= gpurun_out/r2_coll2_loop.log
