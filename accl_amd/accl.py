"""High-level Python API (the PyACCL-equivalent surface).

Construction mirrors the reference driver's bring-up: process launch and
rank discovery are the caller's job (the reference uses MPI only for that —
test/host/xrt/include/fixture.hpp:127); here the bootstrap allgather of the
arena IPC handles runs over torch.distributed when it is initialized, and is
computed locally for the emulator (shm names are deterministic).
"""
import json
import os

from . import _core

DataType = _core.DataType
ReduceFunction = _core.ReduceFunction


def generate_ranks(config_file=None, local=True, world_size=1,
                   start_port=5500, rxbuf_size=1 << 20):
    """Reference-compatible rank map (reference: accl_network_utils
    generate_ranks, driver/utils/accl_network_utils/include/
    accl_network_utils.hpp:41-53; config JSON = {"ips": [...]}).

    On MI355X the transport is xGMI peer memory, so ip/port are carried for
    config-format compatibility but do not open sockets; max_segment_size
    maps to the eager slot size (the reference's rxbuf_size)."""
    if config_file is not None:
        with open(config_file) as f:
            ips = json.load(f)["ips"]
    else:
        ips = ["127.0.0.1" if local else f"10.10.10.{i + 1}"
               for i in range(world_size)]
    return [{"ip": ip, "port": start_port + i, "session_id": i,
             "max_segment_size": rxbuf_size} for i, ip in enumerate(ips)]


def load_tuning(path, message_bytes=None):
    """Read a tools/autotune.py table; returns the opts dict for the size
    bracket closest to message_bytes (largest entry if None)."""
    with open(path) as f:
        meta = json.load(f)
    table = meta["table"]
    keys = sorted(int(k) for k in table)
    if message_bytes is None:
        pick = keys[-1]
    else:
        pick = min(keys, key=lambda k: abs(k - message_bytes))
    e = table[str(pick)]
    return {"slot_bytes": int(e["slot_bytes"]), "n_slots": int(e["n_slots"])}


def emu_job_name(seed=None):
    """A job name every rank can derive identically (pass a shared seed)."""
    seed = seed if seed is not None else os.environ.get("ACCL_EMU_JOB", "accljob")
    return f"accl_{seed}"


_TORCH_DT = None


def _torch_dtype_map():
    global _TORCH_DT
    if _TORCH_DT is None:
        import torch
        _TORCH_DT = {
            DataType.float32: torch.float32,
            DataType.float64: torch.float64,
            DataType.float16: torch.float16,
            DataType.bfloat16: torch.bfloat16,
            DataType.int32: torch.int32,
            DataType.int64: torch.int64,
            DataType.int8: torch.int8,
        }
    return _TORCH_DT


class ACCL:
    """One rank of the collective engine.

    backend: "emu" (CPU emulator over shm), "gpu" (MI355X persistent engine),
    or "auto" (gpu if a HIP device is visible, else emu).
    """

    def __init__(self, nranks=None, rank=None, backend="auto", job=None,
                 device=None, heap_bytes=None, bootstrap="auto", ranks=None,
                 opts=None, **extra):
        # accept both ACCL(opts={...}) and ACCL(slot_bytes=..., ...)
        opts = dict(opts or {})
        opts.update(extra)
        if ranks is not None:
            # reference-style rank map (generate_ranks): world size + eager
            # slot size come from it (reference: ACCL ctor takes the rank
            # vector, driver/xrt/include/accl.hpp:57)
            nranks = len(ranks)
            # opts itself is the kwargs dict make_config reads (top-level key)
            opts.setdefault(
                "slot_bytes", int(ranks[0].get("max_segment_size", 1 << 20)))
        if nranks is None:
            nranks = int(os.environ.get("WORLD_SIZE", "1"))
        if rank is None:
            rank = int(os.environ.get("RANK", "0"))
        if backend == "auto":
            backend = "gpu" if self._has_gpu() else "emu"
        self.backend_name = backend
        if backend == "emu":
            hb = heap_bytes or (256 << 20)
            self._a = _core.create_emu(nranks, rank, emu_job_name(job), hb, opts)
        elif backend == "gpu":
            if device is None:
                device = int(os.environ.get("LOCAL_RANK", "0"))
            hb = heap_bytes or (8 << 30)
            self._a = _core.create_gpu(nranks, rank, device, hb,
                                       int(os.environ.get("ACCL_ENGINE_WGS", "0")),
                                       opts)
        else:
            raise ValueError(f"unknown backend {backend!r}")
        self.device_index = device or 0
        self._connect(bootstrap, nranks, rank, job)

    @staticmethod
    def _has_gpu():
        try:
            import torch
            return torch.cuda.is_available()
        except Exception:
            return False

    def _connect(self, bootstrap, nranks, rank, job):
        blob = self._a.local_blob()
        if nranks == 1:
            self._a.connect([blob])
            return
        if callable(bootstrap):
            # caller-supplied allgather: fn(my_blob, nranks, rank) -> blobs
            self._a.connect(list(bootstrap(blob, nranks, rank)))
            return
        if self.backend_name == "emu":
            # shm names are deterministic — compute peers' blobs locally
            blobs = [f"/{emu_job_name(job)}_r{r}".encode() for r in range(nranks)]
            blobs[rank] = blob
            self._a.connect(blobs)
            return
        if bootstrap == "auto":
            try:
                import torch.distributed as dist
                bootstrap = "torch" if dist.is_initialized() else "file"
            except Exception:
                bootstrap = "file"
        if bootstrap == "torch":
            import torch.distributed as dist
            if not dist.is_initialized():
                raise RuntimeError(
                    "accl_amd: torch.distributed must be initialized (gloo is "
                    "fine) to bootstrap a multi-rank GPU engine")
            objs = [None] * nranks
            dist.all_gather_object(objs, bytes(blob))
            self._a.connect(list(objs))
        elif bootstrap == "file":
            # filesystem rendezvous under a job-derived directory (single node)
            import tempfile
            import time
            d = os.path.join(tempfile.gettempdir(),
                             f"accl_bootstrap_{emu_job_name(job)}")
            os.makedirs(d, exist_ok=True)
            with open(os.path.join(d, f"r{rank}.tmp"), "wb") as f:
                f.write(bytes(blob))
            os.rename(os.path.join(d, f"r{rank}.tmp"),
                      os.path.join(d, f"r{rank}.blob"))
            blobs = []
            deadline = time.time() + 60
            for r in range(nranks):
                p = os.path.join(d, f"r{r}.blob")
                while not os.path.exists(p):
                    if time.time() > deadline:
                        raise RuntimeError(f"bootstrap: rank {r} blob missing")
                    time.sleep(0.01)
                with open(p, "rb") as f:
                    blobs.append(f.read())
            self._a.connect(blobs)
        else:
            raise ValueError(f"unknown bootstrap {bootstrap!r}")

    # ---------------- buffers ----------------
    def create_buffer(self, count, dtype=DataType.float32, device_only=False):
        return self._a.create_buffer(count, dtype, device_only)

    def tensor(self, buf):
        """Zero-copy torch view of a buffer (device tensor on the gpu
        backend, cpu tensor view is not supported for emu — use read/write)."""
        import torch
        cap = self._a.buffer_dlpack(buf, self.device_index)
        return torch.from_dlpack(cap)

    def reduce_from_stream(self, lane, count, root, func, comm=0,
                           dst_stream=None, dtype=DataType.float32):
        """Reduce where this rank's operand comes from stream ring `lane`
        (reference: reduce stream2mem/stream2stream, test.cpp matrix) —
        composition of copy_from_stream + reduce (+ stream_put of the root
        result when dst_stream is given). Returns the result buffer on the
        root, else None."""
        staged = self.create_buffer(count, dtype)
        self.copy_from_stream(lane, staged, count)
        out = self.create_buffer(count, dtype)
        self.reduce(staged, out, count, root, func, comm=comm)
        if self._a.rank == root and dst_stream is not None:
            self.stream_put(out, count, dst=dst_stream[0],
                            tag=dst_stream[1], comm=comm)
        return out if self._a.rank == root else None

    def buffer_like(self, tensor, device_only=True):
        dt = {v: k for k, v in _torch_dtype_map().items()}[tensor.dtype]
        return self.create_buffer(tensor.numel(), dt, device_only)

    # ---------------- properties ----------------
    @property
    def rank(self):
        return self._a.rank

    @property
    def nranks(self):
        return self._a.nranks

    # ---------------- ops (delegate) ----------------
    def __getattr__(self, name):
        if name in ("copy", "put", "combine", "send", "recv", "bcast", "scatter",
                    "gather", "allgather", "reduce", "allreduce",
                    "reduce_scatter", "alltoall", "barrier", "nop",
                    "stream_put", "pop_stream", "stream_ready", "push_stream",
                    "copy_from_stream", "send_from_stream", "alive",
                    "soft_reset",
                    "info", "set_timeout_ms", "set_max_eager_size",
                    "set_max_rendezvous_size", "set_tuning",
                    "dump_communicator", "dump_eager_rx_buffers",
                    "dump_streams", "dump_engine_status", "dump_rendezvous",
                    "create_communicator", "split_communicator",
                    "free_request", "deinit"):
            return getattr(self._a, name)
        raise AttributeError(name)

    def close(self):
        self._a.deinit()
