#!/bin/bash
# Bisect the pytest-context GPU mystery (late round 2): same engine ops with
# progressively more torch involvement, all in fresh processes.
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
run_py() { echo "=== $1 ==="; timeout 90 python -c "$2" 2>&1 | grep -vE "amdgpu.ids" | tail -4; }
BODY='
import numpy as np, accl_amd as A
a = A.ACCL(nranks=1, rank=0, backend="gpu", heap_bytes=2<<30)
cnt = 1<<20
s1, s2, d = (a.create_buffer(cnt, A.DataType.float32) for _ in range(3))
x = np.random.default_rng(1).standard_normal(cnt, dtype=np.float32)
y = np.random.default_rng(2).standard_normal(cnt, dtype=np.float32)
s1.write(x); s2.write(y)
a.combine(cnt, A.ReduceFunction.SUM, s1, s2, d)
out = np.zeros(cnt, np.float32); d.read(out.view(np.int8))
print("combine maxerr:", np.abs(out-(x+y)).max())
'
run_py "a_no_torch" "$BODY
print('A OK')"
run_py "b_torch_imported" "import torch
$BODY
print('B OK')"
run_py "c_cuda_avail" "import torch
torch.cuda.is_available()
$BODY
print('C OK')"
run_py "d_cuda_tensor_before" "import torch
t = torch.ones(1000, device='cuda'); torch.cuda.current_stream().synchronize()
$BODY
print('D OK')"
run_py "e_cuda_tensor_during" "import torch
$BODY
t = (torch.from_numpy(x).cuda() + torch.from_numpy(y).cuda()).cpu().numpy()
print('torch ref maxerr:', np.abs(t-(x+y)).max())
print('E OK')"
run_py "f_pytest_context" "import pytest, torch
torch.cuda.is_available()
$BODY
t2 = torch.from_numpy(x).cuda(); torch.cuda.current_stream().synchronize()
print('F OK')"
