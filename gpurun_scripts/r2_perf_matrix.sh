#!/bin/bash
# Perf matrix: tile size x WG count x acquire fence, P=1 copy path
set -e
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
out=gpurun_out/r2_perf_matrix.csv
echo "tile_kb,wgs,no_acq,bytes,usec,engine_usec,busbw" > $out
run_one() {  # tile wgs noacq bytes
  local env_pre=""
  [ "$1" != 0 ] && env_pre="ACCL_TILE_KB=$1"
  [ "$3" = 1 ] && env_pre="$env_pre ACCL_NO_ACQ=1"
  r=$(env $env_pre ACCL_ENGINE_WGS=$2 timeout 240 python bench.py --steps 8 --warmup 3 --bytes $4 2>/dev/null | tail -1)
  us=$(python -c "import json,sys; d=json.loads('''$r'''); print(f\"{d['ms_per_step']*1000:.1f},{d['value']:.1f}\")")
  echo "$1,$2,$3,$4,$us" | tee -a $out
}
for tile in 0 512 1024 2048; do
  for b in 67108864 268435456 1073741824; do
    run_one $tile 640 0 $b
  done
done
for wgs in 320 960 1280; do
  run_one 0 $wgs 0 268435456
done
run_one 0 640 1 268435456
run_one 1024 640 1 268435456
run_one 1024 960 0 268435456
run_one 1024 960 0 1073741824
cat $out
