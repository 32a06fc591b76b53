#!/bin/bash
# Round-2 opening validation: full GPU state of the round-1 deliverable in
# one call. Budget ~8 min of box time.
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/r2_pytest.log 2>&1
echo "pytest exit $?" >> gpurun_out/r2_pytest.log
timeout 120 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/r2_smoke.log 2>&1
timeout 150 python bench.py --steps 10 --warmup 3 --sweep --out-csv gpurun_out/r2_sweep.csv > gpurun_out/r2_bench.log 2>&1
timeout 100 python gpurun_scripts/diag_copy.py > gpurun_out/r2_diag.log 2>&1
tail -2 gpurun_out/r2_pytest.log; tail -1 gpurun_out/r2_smoke.log
grep -h value gpurun_out/r2_bench.log | head -c 200
