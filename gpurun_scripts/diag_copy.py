"""Diagnose P=1 copy throughput: torch baseline vs engine, wave histogram."""
import sys
import time

sys.path.insert(0, "/root/repo")
import numpy as np  # noqa: E402
import torch  # noqa: E402

import accl_amd as A  # noqa: E402


def t_copy(x, y, iters=10):
    for _ in range(3):
        y.copy_(x)
    torch.cuda.current_stream().synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        y.copy_(x)
    torch.cuda.current_stream().synchronize()
    return iters * x.numel() * x.element_size() / (time.perf_counter() - t0) / 1e9


cnt = 64 << 20  # 256 MiB fp32
x = torch.randn(cnt, device="cuda")
y = torch.empty_like(x)
print(f"torch d2d copy (no engine): {t_copy(x, y):.0f} GB/s", flush=True)

a = A.ACCL(nranks=1, rank=0, backend="gpu", heap_bytes=4 << 30)
print(f"torch d2d copy (engine idle): {t_copy(x, y):.0f} GB/s", flush=True)

s = a.create_buffer(cnt, A.DataType.float32, device_only=True)
d = a.create_buffer(cnt, A.DataType.float32, device_only=True)
base = np.array(a._a.debug_wave_tiles(), np.int64)

r = a.copy(s, d, cnt, from_device=True, to_device=True)  # warm
dur = []
host = []
for _ in range(5):
    t0 = time.perf_counter()
    r = a.copy(s, d, cnt, from_device=True, to_device=True)
    host.append(time.perf_counter() - t0)
    dur.append(r.duration_us())
gb = cnt * 4 / 1e9
print(f"accl copy 256MB: host {min(host)*1e6:.0f} us ({gb/min(host):.0f} GB/s), "
      f"device {min(dur):.0f} us ({gb/(min(dur)/1e6):.0f} GB/s)", flush=True)

wt = np.array(a._a.debug_wave_tiles(), np.int64) - base
nz = wt[wt > 0]
print(f"wave_tiles: active_waves={len(nz)} total_tiles={nz.sum()} "
      f"min={nz.min() if len(nz) else 0} max={nz.max() if len(nz) else 0} "
      f"first16={wt[:16].tolist()}", flush=True)

# small-message latency: 4KB copy
s2 = a.create_buffer(1024, A.DataType.float32, device_only=True)
d2 = a.create_buffer(1024, A.DataType.float32, device_only=True)
r = a.copy(s2, d2, 1024, from_device=True, to_device=True)
ts = []
for _ in range(50):
    t0 = time.perf_counter()
    r = a.copy(s2, d2, 1024, from_device=True, to_device=True)
    ts.append(time.perf_counter() - t0)
print(f"accl copy 4KB: host {min(ts)*1e6:.1f} us, device {r.duration_us():.1f} us",
      flush=True)
tl = a._a.debug_timeline()
print(f"timeline ticks(10ns): inline_wait {tl[6]-tl[5]} "
      f"op->mk_local {tl[9]-tl[8]} mk->flows {tl[10]-tl[9]} "
      f"flows->submit {tl[5]-tl[10]} inline->subret {tl[13]-tl[6]} "
      f"subret->alldone {tl[14]-tl[13]} alldone->ret {tl[11]-tl[14]}",
      flush=True)
a.close()
