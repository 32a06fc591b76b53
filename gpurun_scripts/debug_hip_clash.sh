#!/bin/bash
# Diagnose hipSetDevice failure when torch and accl_amd._core coexist.
cd /root/repo
mkdir -p gpurun_out
exec > gpurun_out/hipclash.log 2>&1

echo "=== case 1: _core only ==="
timeout 60 python - <<'EOF'
from accl_amd import _core as c
a = c.create_gpu(1, 0, 0, 1 << 28, 0, {})
a.connect([a.local_blob()])
print("case1 OK")
a.deinit()
EOF
echo "case1 exit $?"

echo "=== case 2: torch first, then _core ==="
timeout 60 python - <<'EOF'
import torch
print("torch cuda avail:", torch.cuda.is_available())
x = torch.ones(4, device="cuda")
print("torch tensor ok", x.sum().item())
from accl_amd import _core as c
a = c.create_gpu(1, 0, 0, 1 << 28, 0, {})
a.connect([a.local_blob()])
print("case2 OK")
a.deinit()
EOF
echo "case2 exit $?"

echo "=== case 3: _core first, then torch ==="
timeout 60 python - <<'EOF'
from accl_amd import _core as c
a = c.create_gpu(1, 0, 0, 1 << 28, 0, {})
import torch
print("torch avail:", torch.cuda.is_available())
a.connect([a.local_blob()])
print("case3 OK")
a.deinit()
EOF
echo "case3 exit $?"

echo "=== libs loaded when torch imported ==="
timeout 60 python - <<'EOF'
import torch
torch.cuda.is_available()
for line in open('/proc/self/maps'):
    if any(k in line for k in ('amdhip', 'hsa-runtime', 'rocr')):
        print(line.split()[-1])
EOF

echo "=== lib deps of _core ==="
ldd accl_amd/_core*.so | grep -Ei 'hip|hsa'
echo "=== torch lib dir ==="
python -c "import torch, os; print(os.path.dirname(torch.__file__))"
ls /usr/local/lib/python3.10/dist-packages/torch/lib/ 2>/dev/null | grep -Ei 'hip|hsa' | head

echo "=== case 4: torch + LD_PRELOAD our hip ==="
LD_PRELOAD=/opt/rocm/lib/libamdhip64.so timeout 60 python - <<'EOF'
import torch
print("torch avail:", torch.cuda.is_available())
from accl_amd import _core as c
a = c.create_gpu(1, 0, 0, 1 << 28, 0, {})
a.connect([a.local_blob()])
print("case4 OK")
a.deinit()
EOF
echo "case4 exit $?"
