"""Multi-process harness for emulator (and single-GPU) collective tests.

Mirrors the reference's MPI-launched gtest + emulator-process model
(reference: test/host/xrt/src/test.cpp run under mpirun with --startemu,
test/host/xrt/src/utility.cpp:25-70): here each rank is a forked process
running one scenario function; shm arenas replace the ZMQ ethernet.
"""
import multiprocessing as mp
import os
import traceback

import numpy as np

_JOB_COUNTER = 0


def fresh_job():
    global _JOB_COUNTER
    _JOB_COUNTER += 1
    return f"t{os.getpid()}_{_JOB_COUNTER}"


def _worker(fn, rank, nranks, job, opts, q, backend):
    try:
        if backend == "gpu":
            # N engines share one test GPU: shrink each grid so all stay
            # resident together (full-size grids could not co-schedule)
            os.environ.setdefault("ACCL_ENGINE_WGS", str(max(16, 128 // nranks)))
        import accl_amd as A
        a = A.ACCL(nranks=nranks, rank=rank, backend=backend, job=job, **(
            {"opts": opts} if opts else {}))
        try:
            fn(a, rank, nranks)
        finally:
            a.close()
        q.put((rank, None))
    except Exception:
        q.put((rank, traceback.format_exc()))


def run_ranks(fn, nranks, opts=None, timeout=120, backend="emu"):
    """Run `fn(accl, rank, nranks)` on every rank; raise on any failure."""
    # fork is fine for the CPU emulator; GPU children must be spawned (a
    # forked child inherits the parent's initialized HIP runtime, which does
    # not survive fork)
    ctx = mp.get_context("spawn" if backend == "gpu" else "fork")
    q = ctx.Queue()
    job = fresh_job()
    ps = [ctx.Process(target=_worker, args=(fn, r, nranks, job, opts, q, backend))
          for r in range(nranks)]
    for p in ps:
        p.start()
    errs = []
    try:
        for _ in range(nranks):
            rank, err = q.get(timeout=timeout)
            if err:
                errs.append(f"rank {rank}:\n{err}")
    finally:
        for p in ps:
            p.join(timeout=10)
        for p in ps:
            if p.is_alive():
                # SIGINT first: KeyboardInterrupt unwinds the worker so
                # a.close() halts the persistent engine kernels — a hard
                # terminate would leave them running on the GPU and poison
                # every later test on the box
                import os as _os
                import signal as _signal
                try:
                    _os.kill(p.pid, _signal.SIGINT)
                except OSError:
                    pass
        for p in ps:
            if p.is_alive():
                p.join(timeout=20)
            if p.is_alive():
                p.terminate()
                errs.append(f"rank {p.pid} hung; terminated")
    if errs:
        raise AssertionError("\n".join(errs))


def rd(buf, count, dtype=np.float32):
    out = np.zeros(count, dtype)
    buf.read(out)
    return out


def pattern(count, rank, dtype=np.float32, seed=0):
    # deterministic per-rank data, bounded magnitude (safe for f16 sums)
    return (((np.arange(count) * 7 + rank * 13 + seed) % 61) - 30).astype(dtype)
