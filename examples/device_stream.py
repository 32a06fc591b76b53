#!/usr/bin/env python3
"""Device-initiated streaming demo (vadd_put): a HIP kernel computes x+1 and
pushes the result into a peer's stream ring from INSIDE the kernel; the
consumer pops it with pop_stream. Requires a GPU (single rank: self-loop)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np

import accl_amd as A
import accl_amd._core as core

if not A.ACCL._has_gpu():
    print("device_stream demo needs an MI355X (device-initiated push runs "
          "inside a HIP kernel) — skipping")
    sys.exit(0)

a = A.ACCL(nranks=1, rank=0, backend="gpu")
try:
    n = 10000
    src = a.create_buffer(n, A.DataType.float32)
    x = np.random.default_rng(0).standard_normal(n, dtype=np.float32)
    src.write(x)
    core.demo_vadd_put(a._a, src, n, 0, 7, 1.0)
    got, buf = np.zeros(0, np.float32), np.zeros(n, np.float32)
    while got.size < n:
        nb, tag = a.pop_stream(0, buf)
        got = np.concatenate([got, buf[:nb // 4]])
    assert np.allclose(got, x + 1.0)
    print("device-initiated vadd_put stream OK")
finally:
    a.close()
