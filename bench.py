#!/usr/bin/env python3
"""accl_amd flagship benchmark — all-reduce bus bandwidth (BASELINE.json).

One "step" = one blocking fp32 allreduce of the flagship message size
(default 256 MiB) through the persistent GPU engine over xGMI.

Run:  python bench.py --gpus N --steps K --warmup W
The driver launches N>1 as one rank per GPU via torch.distributed.run;
torch.distributed (gloo) is used ONLY to bootstrap (IPC handle allgather)
and to reduce timings across ranks — the data plane is accl_amd's own.

Bus bandwidth = 2(P-1)/P * bytes / t  (the standard allreduce factor;
reference metric protocol: BASELINE.md, test/host/Coyote/test.cpp:523-534
computes Gbit/s the same way for its collectives).
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

BASELINE_GBPS = 12.5  # reference anchor: 100 Gbps line rate


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=int(os.environ.get("WORLD_SIZE", "1")))
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--bytes", type=int, default=256 << 20)
    p.add_argument("--sweep", action="store_true",
                   help="also print the 4KB..1GB busbw curve (rank 0)")
    p.add_argument("--collective", default="allreduce",
                   choices=["allreduce", "bcast", "scatter", "gather",
                            "allgather", "reduce_scatter", "alltoall",
                            "reduce", "sendrecv"],
                   help="BASELINE configs 2-4: which collective to time")
    p.add_argument("--dtype", default="fp32", choices=["fp32", "bf16", "fp16"],
                   help="BASELINE config 4: bf16 grad/param patterns")
    p.add_argument("--overlap-gemm", action="store_true",
                   help="BASELINE config 5: run a co-resident torch GEMM "
                        "stream while timing the collective")
    p.add_argument("--backend", default="auto", choices=["auto", "gpu", "emu"])
    p.add_argument("--out-csv", default=None)
    return p.parse_args()


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    rank = int(os.environ.get("RANK", "0"))

    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        if not dist.is_initialized():
            dist.init_process_group("gloo", rank=rank, world_size=world)

    import accl_amd as A
    import numpy as np

    backend = args.backend
    if backend == "auto":
        backend = "gpu" if A.ACCL._has_gpu() else "emu"
    slot_mb = 4 if backend == "gpu" else 1
    # size the arena heap for src+dst (+sweep buffers) on BOTH backends —
    # the emulator arena is plain shm, so large default-bytes runs work too
    heap = max(4 * args.bytes + (64 << 20), 1 << 30)
    if args.sweep:
        heap += 2 << 30  # dedicated 1 GiB sweep src+dst (full-range sweep)
    a = A.ACCL(nranks=world, rank=rank, backend=backend,
               heap_bytes=heap,
               opts={"slot_bytes": slot_mb << 20, "n_slots": 8})

    DT = {"fp32": A.DataType.float32, "bf16": A.DataType.bfloat16,
          "fp16": A.DataType.float16}[args.dtype]
    esz = 4 if args.dtype == "fp32" else 2
    count = args.bytes // esz
    # collectives with per-rank chunking need count divisible by world
    count -= count % max(world, 1)
    # host-mirror buffers + hipMemcpy fill: torch-CUDA-free hot path (no
    # kernel launches competing with the persistent engine; the
    # overlap-gemm config is the one deliberate co-resident-compute case)
    src = a.create_buffer(count, DT)
    dst = a.create_buffer(count, DT)
    fill = np.random.default_rng(0).standard_normal(
        min(count, 1 << 22), dtype=np.float32)
    if esz == 2:
        fill = fill.astype(np.float16)
    reps = (count + fill.size - 1) // fill.size
    big = np.tile(fill, reps)[:count]
    src.write(np.ascontiguousarray(big).view(np.int8))

    RF = A.ReduceFunction.SUM
    per = count // max(world, 1)

    def one_allreduce(s=src, d=dst, n=count):
        a.allreduce(s, d, n, RF, from_device=True, to_device=True)

    def one_bcast():
        a.bcast(src, count, 0, from_device=True, to_device=True)

    def one_scatter():
        a.scatter(src, dst, per, 0, from_device=True, to_device=True)

    def one_gather():
        a.gather(src, dst, per, 0, from_device=True, to_device=True)

    def one_allgather():
        a.allgather(src, dst, per, from_device=True, to_device=True)

    def one_reduce_scatter():
        a.reduce_scatter(src, dst, per, RF, from_device=True, to_device=True)

    def one_reduce():
        a.reduce(src, dst, count, 0, RF, from_device=True, to_device=True)

    def one_alltoall():
        a.alltoall(src, dst, per, from_device=True, to_device=True)

    def one_sendrecv():
        # disjoint pairs (0,1)(2,3)...; odd world: last rank self-copies
        if world == 1 or (world % 2 == 1 and rank == world - 1):
            a.copy(src, dst, count, from_device=True, to_device=True)
        elif rank % 2 == 0:
            peer = rank + 1
            a.send(src, count, dst=peer, tag=1, from_device=True)
            a.recv(dst, count, src=peer, tag=2, to_device=True)
        else:
            peer = rank - 1
            a.recv(dst, count, src=peer, tag=1, to_device=True)
            a.send(src, count, dst=peer, tag=2, from_device=True)

    ONE = {"allreduce": one_allreduce, "bcast": one_bcast,
           "scatter": one_scatter, "gather": one_gather,
           "allgather": one_allgather, "reduce_scatter": one_reduce_scatter,
           "reduce": one_reduce, "alltoall": one_alltoall,
           "sendrecv": one_sendrecv}
    one_op = ONE[args.collective]

    overlap_stop = None
    if args.overlap_gemm and backend == "gpu":
        # BASELINE config 5: a co-resident GEMM stream keeps MFMA busy while
        # the engine's mover fleet (10 of 16 wave slots per CU) moves data
        import threading
        import torch
        overlap_stop = threading.Event()
        gemm_stats = {"n": 0}

        def gemm_loop():
            st = torch.cuda.Stream()
            with torch.cuda.stream(st):
                m = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
                while not overlap_stop.is_set():
                    m = m @ m
                    m = m / m.norm()
                    st.synchronize()
                    gemm_stats["n"] += 1

        gemm_thread = threading.Thread(target=gemm_loop, daemon=True)
        gemm_thread.start()

    def timed(fn, steps, warmup):
        for _ in range(warmup):
            fn()
        a.barrier()
        t0 = time.perf_counter()
        for _ in range(steps):
            fn()
        a.barrier()
        t1 = time.perf_counter()
        el = (t1 - t0) / steps
        if dist is not None:
            import torch
            t = torch.tensor([el], dtype=torch.float64)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            el = float(t.item())
        return el

    def busbw(nbytes, sec, P):
        factor = 2.0 * (P - 1) / P if P > 1 else 1.0
        return factor * nbytes / sec / 1e9

    el = timed(one_op, args.steps, args.warmup)
    if overlap_stop is not None:
        overlap_stop.set()
        gemm_thread.join(timeout=10)
    # bus-bandwidth accounting per collective (NCCL-tests conventions; the
    # reference computes Gbit/s the same way, test/host/Coyote/test.cpp:523)
    nbytes = count * esz
    P = world
    factor = {
        "allreduce": 2.0 * (P - 1) / P if P > 1 else 1.0,
        "reduce_scatter": (P - 1) / P if P > 1 else 1.0,
        "allgather": (P - 1) / P if P > 1 else 1.0,
        "alltoall": (P - 1) / P if P > 1 else 1.0,
        "bcast": 1.0, "scatter": 1.0 / P, "gather": 1.0 / P,
        "reduce": 1.0, "sendrecv": 1.0,
    }[args.collective]
    value = factor * nbytes / el / 1e9

    sweep_rows = []
    if args.sweep:
        # reference protocol: count sweep per collective with device-side
        # duration recorded to CSV (bench.cpp:25-61 + fixture CSV). The
        # sweep always covers the full 4 KB..1 GiB range: allocate dedicated
        # sweep buffers when --bytes is smaller than the 1 GiB endpoint.
        sweep_count = (1 << 30) // esz
        sweep_count -= sweep_count % max(world, 1)
        if sweep_count > count:
            sw_src = a.create_buffer(sweep_count, DT)
            sw_dst = a.create_buffer(sweep_count, DT)
            reps2 = (sweep_count + big.size - 1) // big.size
            sw_src.write(np.ascontiguousarray(
                np.tile(big, reps2)[:sweep_count]).view(np.int8))
        else:
            sw_src, sw_dst = src, dst
            sweep_count = count
        sz = 4096
        while sz <= (1 << 30):
            c = (sz // esz) - ((sz // esz) % max(world, 1))
            if 0 < c <= sweep_count:
                s2, d2 = sw_src.slice(0, c), sw_dst.slice(0, c)
                dev_us = [0.0]

                def one(s=s2, d=d2, c=c):
                    r = a.allreduce(s, d, c, A.ReduceFunction.SUM,
                                    from_device=True, to_device=True)
                    dev_us[0] = r.duration_us()

                e = timed(one, max(3, min(20, (1 << 26) // sz)), 2)
                sweep_rows.append((sz, e * 1e6, dev_us[0],
                                   busbw(sz, e, world)))
            sz *= 4
        if rank == 0:
            lines = ["bytes,usec,engine_usec,busbw_GBps"] + [
                f"{b},{u:.2f},{du:.2f},{g:.2f}" for b, u, du, g in sweep_rows]
            csv = "\n".join(lines)
            print(csv, file=sys.stderr)
            if args.out_csv:
                with open(args.out_csv, "w") as f:
                    f.write(csv + "\n")

    if rank == 0:
        out = {
            "metric": f"{args.collective}_busbw_GBps",
            "value": round(value, 3),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(el * 1e3, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(value / BASELINE_GBPS, 3),
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": args.collective,
                "collective": args.collective,
                "message_bytes": count * esz,
                "overlap_gemm": bool(args.overlap_gemm),
                "global_batch": 1,
                "seq_len": args.bytes // 4,
                "parallelism": f"fullmesh_p{world}",
                "backend": backend,
                "metric_desc": (f"{args.collective} bus bandwidth (GB/s), "
                                f"{args.dtype}, per BASELINE.json; "
                                "vs_baseline anchors the reference's "
                                "100 Gbps (12.5 GB/s) line rate"),
            },
        }
        print(json.dumps(out), flush=True)
    a.close()
    if dist is not None:
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    try:
        main()
    except Exception:
        # post-mortem detail for multi-rank failures (driver captures stderr)
        import traceback
        rank = os.environ.get("RANK", "?")
        print(f"[bench rank {rank}] FAILED:", file=sys.stderr, flush=True)
        traceback.print_exc()
        raise
