#!/usr/bin/env python3
"""Minimal accl_amd usage: P-rank fp32 allreduce.

Run one process per rank (any bootstrap that gives RANK/WORLD_SIZE works):
  WORLD_SIZE=2 RANK=0 python examples/allreduce.py &
  WORLD_SIZE=2 RANK=1 python examples/allreduce.py
On GPUs, add LOCAL_RANK per device and initialize torch.distributed (gloo)
first so the arena IPC handles can be exchanged; on the CPU emulator the
rendezvous is implicit (shared-memory names are deterministic).
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np

import accl_amd as A

world = int(os.environ.get("WORLD_SIZE", "1"))
rank = int(os.environ.get("RANK", "0"))

a = A.ACCL(nranks=world, rank=rank, backend="auto", job="example")
try:
    n = 1 << 20
    src = a.create_buffer(n, A.DataType.float32)
    dst = a.create_buffer(n, A.DataType.float32)
    src.write(np.full(n, float(rank + 1), np.float32))
    a.allreduce(src, dst, n, A.ReduceFunction.SUM)
    out = np.zeros(n, np.float32)
    dst.read(out)
    expect = world * (world + 1) / 2
    assert np.allclose(out, expect), out[:4]
    print(f"rank {rank}: allreduce OK (sum = {out[0]})")
    print(a.dump_engine_status().strip())
finally:
    a.close()
