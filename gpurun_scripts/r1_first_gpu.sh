#!/bin/bash
# Round-1 first GPU validation: tests, smoke, bench, rocprof.
cd /root/repo
mkdir -p gpurun_out
echo "=== rocm-smi ===" > gpurun_out/env.log
rocm-smi --showproductname >> gpurun_out/env.log 2>&1

timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit $?" >> gpurun_out/pytest_gpu.log

timeout 150 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke.log 2>&1
echo "smoke exit $?" >> gpurun_out/smoke.log

timeout 240 python bench.py --steps 10 --warmup 3 --sweep --out-csv gpurun_out/sweep1.csv > gpurun_out/bench1.log 2>&1
echo "bench exit $?" >> gpurun_out/bench1.log

export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -- python /root/repo/bench.py --steps 5 --warmup 2 --bytes 268435456 > /root/repo/gpurun_out/prof_bench.log 2>&1
echo "rocprof exit $?" >> /root/repo/gpurun_out/prof_bench.log
