#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
for i in 1 2; do
  timeout 600 python -m pytest tests/ -q -m gpu -p no:cacheprovider --timeout=300 \
    > gpurun_out/r2s4_$i.log 2>&1
  echo "SUITE$i RC=$?"
  tail -2 gpurun_out/r2s4_$i.log
done
# latency check with the device-resident ring
timeout 300 python bench.py --steps 10 --warmup 3 --sweep --bytes $((16<<20)) > gpurun_out/r2s4_lat.log 2>&1
grep -E "^4096|^16384|^65536" gpurun_out/r2s4_lat.log
ACCL_NO_DEV_RING=1 timeout 300 python bench.py --steps 10 --warmup 3 --sweep --bytes $((16<<20)) > gpurun_out/r2s4_lat_nodev.log 2>&1
grep -E "^4096|^16384|^65536" gpurun_out/r2s4_lat_nodev.log
grep -h "re-import" gpurun_out/r2s4_*.log | head -3
