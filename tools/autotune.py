#!/usr/bin/env python3
"""Segment/slot autotuner (SURVEY §7 stage 5: segment-size autotuning).

Sweeps eager slot geometry (slot_bytes x n_slots) over a message-size grid
for allreduce, on THE CURRENT hardware (GPU engine, or the emulator off-GPU
for harness testing), and writes the best configuration per size bracket to
a JSON tuning table:

  python tools/autotune.py --out accl_tuning.json [--ranks 2] [--quick]

Consume it at construction:

  opts = accl_amd.load_tuning("accl_tuning.json", message_bytes=hint)
  a = accl_amd.ACCL(..., opts=opts)
"""
import argparse
import itertools
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def worker(rank, world, cfg, sizes, steps, q, backend):
    import numpy as np

    import accl_amd as A
    a = A.ACCL(nranks=world, rank=rank, backend=backend,
               job=f"tune{os.getppid()}_{cfg['slot_bytes']}_{cfg['n_slots']}",
               opts={"slot_bytes": cfg["slot_bytes"],
                     "n_slots": cfg["n_slots"],
                     "timeout_us": 30_000_000})
    try:
        out = {}
        for nbytes in sizes:
            n = nbytes // 4
            s = a.create_buffer(n, A.DataType.float32)
            d = a.create_buffer(n, A.DataType.float32)
            s.write(np.ones(n, np.float32))
            for _ in range(2):
                a.allreduce(s, d, n, A.ReduceFunction.SUM)
            a.barrier()
            t0 = time.perf_counter()
            for _ in range(steps):
                a.allreduce(s, d, n, A.ReduceFunction.SUM)
            a.barrier()
            out[nbytes] = (time.perf_counter() - t0) / steps
        if rank == 0:
            q.put(out)
    finally:
        a.close()


def measure(cfg, sizes, steps, world, backend):
    import multiprocessing as mp
    ctx = mp.get_context("spawn" if backend == "gpu" else "fork")
    q = ctx.Queue()
    ps = [ctx.Process(target=worker,
                      args=(r, world, cfg, sizes, steps, q, backend))
          for r in range(world)]
    for p in ps:
        p.start()
    try:
        res = q.get(timeout=300)
    finally:
        for p in ps:
            p.join(timeout=20)
            if p.is_alive():
                p.terminate()
    return res


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="accl_tuning.json")
    ap.add_argument("--ranks", type=int, default=2)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--backend", default="auto")
    args = ap.parse_args()

    import accl_amd as A
    backend = args.backend
    if backend == "auto":
        backend = "gpu" if A.ACCL._has_gpu() else "emu"

    sizes = [1 << 14, 1 << 18, 1 << 22] if args.quick else \
        [1 << 12, 1 << 14, 1 << 16, 1 << 18, 1 << 20, 1 << 22, 1 << 24]
    slot_bytes = [1 << 16, 1 << 20] if args.quick else \
        [1 << 16, 1 << 18, 1 << 20, 1 << 22]
    n_slots = [4, 8] if args.quick else [4, 8, 16]

    results = []
    for sb, ns in itertools.product(slot_bytes, n_slots):
        cfg = {"slot_bytes": sb, "n_slots": ns}
        try:
            times = measure(cfg, sizes, args.steps, args.ranks, backend)
        except Exception as e:  # config infeasible on this box
            print(f"cfg {cfg}: failed ({e})", file=sys.stderr)
            continue
        results.append((cfg, times))
        print(f"slot={sb} n={ns}: " +
              " ".join(f"{b}B:{t*1e6:.0f}us" for b, t in sorted(times.items())))

    table = {}
    for nbytes in sizes:
        best = min(results, key=lambda r: r[1][nbytes])
        table[str(nbytes)] = {**best[0],
                              "usec": round(best[1][nbytes] * 1e6, 1)}
    meta = {"backend": backend, "ranks": args.ranks,
            "collective": "allreduce", "table": table}
    with open(args.out, "w") as f:
        json.dump(meta, f, indent=1)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
