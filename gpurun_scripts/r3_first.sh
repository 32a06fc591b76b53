#!/bin/bash
# Round-3 opener: one call gives full correctness + the key perf A/Bs.
# Usage: /usr/local/graft/bin/gpurun --timeout 2100 -- 'bash gpurun_scripts/r3_first.sh'
set -x
mkdir -p gpurun_out
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
# 1) full suite (NO -x; per-test pytest-timeout bounds each test)
timeout 700 python -m pytest tests/ -q -m gpu -p no:cacheprovider \
  > gpurun_out/r3_pytest.log 2>&1
echo "PYTEST_RC=$?"
tail -3 gpurun_out/r3_pytest.log
# 2) headline + sweep
timeout 420 python bench.py --steps 10 --warmup 3 --sweep > gpurun_out/r3_bench.log 2>&1
grep -E "^[0-9]+," gpurun_out/r3_bench.log
# 3) mid-size latency A/B: inline cutoff 32(def)/64/128 KiB at 64/128/256 KiB msgs
for kb in 64 128; do
  ACCL_INLINE_KB=$kb timeout 240 python bench.py --steps 10 --warmup 3 --sweep --bytes $((16<<20)) \
    > gpurun_out/r3_inline$kb.log 2>&1
  echo "INLINE_KB=$kb:"; grep -E "^65536|^262144|^1048576" gpurun_out/r3_inline$kb.log
done
# 4) 2-proc direct + parked + drain-regression paths once more on fresh silicon
timeout 400 python -m pytest "tests/test_gpu.py::test_two_ranks_direct_paths" \
  "tests/test_gpu.py::test_two_ranks_parked_interleave" \
  "tests/test_gpu.py::test_two_ranks_drain_regressions" \
  "tests/test_gpu.py::test_four_ranks_nary_direct" -q -p no:cacheprovider \
  > gpurun_out/r3_direct.log 2>&1
echo "DIRECT_RC=$?"; tail -2 gpurun_out/r3_direct.log
# 5) protocol fuzz on the GPU engine: 2 procs / 1 GPU, small-slot geometry
#    (the late-round-2 deadlock surface — seeds from test_protocol_fuzz_tiny)
timeout 400 python - <<'EOF' > gpurun_out/r3_fuzz.log 2>&1
import sys
sys.path.insert(0, "/root/repo"); sys.path.insert(0, "/root/repo/tests")
from emu_util import run_ranks
import test_emulator as T
SMALLG = {"n_slots": 4, "slot_bytes": 4096, "timeout_us": 20_000_000}
for seed in (11, 23, 100, 161):
    run_ranks(lambda a, r, n: T._fuzz(a, r, n, seed), 2, backend="gpu",
              opts=SMALLG, timeout=120)
    print("gpu fuzz seed", seed, "ok", flush=True)
EOF
echo "FUZZ_RC=$?"; tail -5 gpurun_out/r3_fuzz.log
