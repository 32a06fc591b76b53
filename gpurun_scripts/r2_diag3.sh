#!/bin/bash
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONPATH=/root/repo
run() { echo "=== $1 ==="; timeout 120 env $2 python gpurun_scripts/r2_diag2.py 2>&1 | grep -vE "amdgpu.ids" | head -12; echo "RC=$?"; }
run baseline ""
run no_dev_ring "ACCL_NO_DEV_RING=1"
run wgs64 "ACCL_ENGINE_WGS=64"
run no_dev_ring_wgs64 "ACCL_NO_DEV_RING=1 ACCL_ENGINE_WGS=64"
