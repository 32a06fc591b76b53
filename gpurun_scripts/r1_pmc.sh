#!/bin/bash
cd /root/repo; mkdir -p gpurun_out
export TMPDIR=/tmp
cd /tmp
rocprofv3 --list-avail > /root/repo/gpurun_out/counters.txt 2>&1
cd /root/repo
# pick an HBM-traffic counter set that exists on this chip
if grep -q "FETCH_SIZE" gpurun_out/counters.txt; then
  SET="FETCH_SIZE WRITE_SIZE L2CacheHit MemUnitBusy MemUnitStalled"
elif grep -q "TCC_EA0_RDREQ_sum" gpurun_out/counters.txt; then
  SET="TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum TCC_HIT_sum TCC_MISS_sum"
else
  SET="SQ_WAVES SQ_BUSY_CYCLES"
fi
echo "PMC SET: $SET" > gpurun_out/pmc_run.log
cd /tmp
timeout 240 rocprofv3 --pmc $SET -d /root/repo/gpurun_out/pmc -- python /root/repo/gpurun_scripts/diag_copy.py >> /root/repo/gpurun_out/pmc_run.log 2>&1
echo "pmc exit $?" >> /root/repo/gpurun_out/pmc_run.log
