#!/bin/bash
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit $?" >> gpurun_out/pytest_gpu.log

timeout 300 python bench.py --steps 10 --warmup 3 --sweep --out-csv gpurun_out/sweep1.csv > gpurun_out/bench1.log 2>&1
echo "bench exit $?" >> gpurun_out/bench1.log

export TMPDIR=/tmp
cd /tmp
timeout 240 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -- python /root/repo/bench.py --steps 5 --warmup 2 > /root/repo/gpurun_out/prof_bench.log 2>&1
echo "rocprof exit $?" >> /root/repo/gpurun_out/prof_bench.log
