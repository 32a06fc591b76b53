#!/bin/bash
# CI mirror of the reference workflow (reference: .github/workflows/
# build-and-test.yml — build emulator + tests, then 1-rank and 2-rank runs).
set -e
cd "$(dirname "$0")/.."
echo "== build (hipcc, gfx950) =="
python -m accl_amd.build
echo "== unit/emulator suite =="
python -m pytest tests -q -m "not gpu"
echo "== 1-rank example =="
WORLD_SIZE=1 RANK=0 python examples/allreduce.py
echo "== 2-rank bench (torchrun, emulator) =="
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29617 \
  bench.py --backend emu --steps 3 --warmup 1 --bytes 1048576 | tail -1
echo "CI OK"
