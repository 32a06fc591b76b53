import faulthandler, sys, numpy as np
faulthandler.enable()
sys.stdout.reconfigure(line_buffering=True)
import accl_amd as A
DT = A.DataType; RF = A.ReduceFunction
print("creating ACCL...", flush=True)
a = A.ACCL(nranks=1, rank=0, backend="gpu", job="diag2", heap_bytes=2 << 30)
print("engine up; copy test", flush=True)
cnt = 1 << 20
s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
x = np.random.default_rng(1).standard_normal(cnt, dtype=np.float32)
s.write(x)
a.copy(s, d, cnt)
out = np.zeros(cnt, np.float32); d.read(out.view(np.int8))
print("copy ok:", np.array_equal(out, x), flush=True)
s2 = a.create_buffer(cnt, DT.float32)
y = np.random.default_rng(2).standard_normal(cnt, dtype=np.float32)
s2.write(y)
print("combine SUM...", flush=True)
try:
    a.combine(cnt, RF.SUM, s, s2, d)
    d.read(out.view(np.int8))
    err = np.abs(out - (x + y)).max()
    print("combine done, max err:", err, flush=True)
except Exception as e:
    print("combine FAILED:", e, flush=True)
    print("timeline:", a._a.debug_timeline(), flush=True)
print("small combine (inline)...", flush=True)
s3, s4, d3 = (a.create_buffer(1024, DT.float32) for _ in range(3))
s3.write(x[:1024]); s4.write(y[:1024])
try:
    a.combine(1024, RF.SUM, s3, s4, d3)
    o3 = np.zeros(1024, np.float32); d3.read(o3.view(np.int8))
    print("inline combine ok:", np.allclose(o3, x[:1024] + y[:1024]), flush=True)
except Exception as e:
    print("inline combine FAILED:", e, flush=True)
print("bf16 combine...", flush=True)
import torch
sb1, sb2, db = (a.create_buffer(8192, DT.bfloat16) for _ in range(3))
t1 = torch.randn(8192).to(torch.bfloat16); t2 = torch.randn(8192).to(torch.bfloat16)
sb1.write(t1.view(torch.int8).numpy()); sb2.write(t2.view(torch.int8).numpy())
try:
    a.combine(8192, RF.SUM, sb1, sb2, db)
    ob = np.zeros(8192, np.uint16); db.read(ob.view(np.int8))
    ref = (t1.float() + t2.float()).to(torch.bfloat16)
    got = torch.from_numpy(ob).view(torch.bfloat16)
    print("bf16 combine maxerr:", (got.float() - ref.float()).abs().max().item(), flush=True)
except Exception as e:
    print("bf16 combine FAILED:", e, flush=True)
print("allreduce P=1 4MB...", flush=True)
try:
    a.allreduce(s, d, cnt, RF.SUM)
    print("allreduce ok", flush=True)
except Exception as e:
    print("allreduce FAILED:", e, flush=True)
a.close()
print("ALL DONE", flush=True)
