#!/bin/bash
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit $?" >> gpurun_out/pytest_gpu.log
for W in 64 128 256; do
  ACCL_ENGINE_WGS=$W timeout 200 python bench.py --steps 10 --warmup 3 > gpurun_out/bench_w$W.log 2>&1
  echo "exit $?" >> gpurun_out/bench_w$W.log
done
timeout 300 python bench.py --steps 10 --warmup 3 --sweep --out-csv gpurun_out/sweep2.csv > gpurun_out/bench2.log 2>&1
echo "bench exit $?" >> gpurun_out/bench2.log
