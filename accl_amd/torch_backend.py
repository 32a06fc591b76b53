"""torch.distributed backend over the accl_amd engine.

Registers backend "accl": ``dist.init_process_group("accl")`` gives torch
DDP/FSDP-style collectives (allreduce, broadcast, all_gather(+_base),
reduce_scatter(_base), alltoall_base, send/recv, barrier) running on the
persistent MI355X engine (or the CPU emulator off-GPU). This EXCEEDS the
reference's surface (PyACCL has no torch integration) and is the "switch
from the reference" path for training users.

Tensors are staged through persistent arena buffers (zero-copy views via
DLPack on GPU, host shadows on the emulator); the one-time arena-IPC
bootstrap rides the c10d Store that init_process_group already provides.
"""
import numpy as np
import torch
import torch.distributed as dist

from . import _core
from .accl import ACCL, DataType, _torch_dtype_map


class _Work(dist.Work):
    """Completion wrapper: blocking ops pass req=None (already complete —
    is_completed() True is then truthful); async ops carry the engine
    Request and report real completion state."""

    def __init__(self, result=None, req=None):
        super().__init__()
        self._result = result if result is not None else []
        self._req = req

    def is_completed(self):
        return self._req is None or self._req.test()

    def is_success(self):
        return self._req is None or not self._req.test() or \
            self._req.retcode() == 0

    def wait(self, timeout=None):
        if self._req is not None:
            e = self._req.wait()
            if e:
                raise RuntimeError(f"accl work failed: {_core.error_to_string(e)}")
            self._req = None
        return True

    def exception(self):
        return None

    def get_future(self):
        # DDP's reducer consumes the bucket tensors through this future
        self.wait()
        fut = torch.futures.Future()
        fut.set_result(self._result)
        return fut


_T2DT = None


def _t2dt(dtype):
    global _T2DT
    if _T2DT is None:
        _T2DT = {v: k for k, v in _torch_dtype_map().items()}
    return _T2DT[dtype]


def _op_name(op):
    # opts.reduceOp is a ReduceOp instance; .op is the RedOpType enum
    rt = op.op if hasattr(op, "op") else op
    return getattr(rt, "name", str(rt))


def _is_avg(op):
    return _op_name(op) == "AVG"


def _red(op):
    from accl_amd import ReduceFunction
    name = _op_name(op)
    if name in ("SUM", "AVG"):
        return ReduceFunction.SUM
    if name == "MAX":
        return ReduceFunction.MAX
    raise NotImplementedError(f"accl backend: ReduceOp {name}")


class AcclProcessGroup(dist.ProcessGroup):
    def __init__(self, store, rank, size, device_hint=None):
        super().__init__(rank, size)
        backend = "gpu" if (device_hint == "cuda" or
                            (device_hint is None and ACCL._has_gpu())) else "emu"
        self._gpu = backend == "gpu"

        def store_allgather(blob, nranks, my_rank):
            store.set(f"accl_blob_{my_rank}", bytes(blob).hex())
            out = []
            for r in range(nranks):
                out.append(bytes.fromhex(
                    store.get(f"accl_blob_{r}").decode()))
            return out

        self._a = ACCL(nranks=size, rank=rank, backend=backend,
                       job=f"tpg{size}", bootstrap=store_allgather)
        self._bufs = {}  # (count, DataType) -> (buffer, torch_view_or_None)

    # ---------------- zero-copy path (GPU) ----------------
    # Contiguous CUDA tensors go to the engine as RAW device pointers: the
    # eager schedules only touch local src/dst through the mover fleet
    # (peers communicate via arena slots), so no staging copy or stream
    # round-trip is needed. The producing stream is synchronized first so
    # the engine never reads half-written operands.
    def _rawable(self, *tensors):
        return self._gpu and all(
            t.is_cuda and t.is_contiguous() for t in tensors)

    def _raw(self, op, count, src, dst, func=0, root=0, tag=0):
        torch.cuda.current_stream().synchronize()
        _core.call_raw(self._a._a, int(op), count, root=root, tag=tag,
                       function=int(func), addr0=src.data_ptr(),
                       addr2=dst.data_ptr(),
                       dtype=int(_t2dt(dst.dtype)),
                       wire_dtype=int(_t2dt(dst.dtype)))

    # ---------------- staging ----------------
    def _buf(self, count, tdt, which):
        # round capacity up to the next power of two so DDP's many bucket
        # shapes share buffers, and keep a bounded LRU (arena heap blocks
        # are freed when the Buffer is dropped)
        cap = 1 << max(0, (count - 1).bit_length())
        key = (which, cap, tdt)
        hit = self._bufs.pop(key, None)
        if hit is not None:
            self._bufs[key] = hit  # re-insert: most recently used
            return hit
        b = self._a.create_buffer(cap, _t2dt(tdt), device_only=self._gpu)
        view = self._a.tensor(b) if self._gpu else None
        while len(self._bufs) >= 16:
            self._bufs.pop(next(iter(self._bufs)))
        self._bufs[key] = (b, view)
        return b, view

    def _upload(self, t, count, which):
        b, view = self._buf(count, t.dtype, which)
        flat = t.reshape(-1)
        if self._gpu:
            view[:flat.numel()].copy_(flat)
            torch.cuda.current_stream().synchronize()
        else:
            arr = flat.detach().numpy() if t.dtype != torch.bfloat16 \
                else flat.detach().view(torch.int16).numpy()
            b.write(np.ascontiguousarray(arr).view(np.int8))
        return b, view

    def _download(self, t, b, view, count=None):
        flat = t.reshape(-1)
        n = count if count is not None else flat.numel()
        if self._gpu:
            flat[:n].copy_(view[:n])
            torch.cuda.current_stream().synchronize()
        else:
            nb = n * flat.element_size()
            raw = np.zeros(nb, np.int8)
            b.read(raw)
            src = torch.from_numpy(raw).view(
                torch.int16 if t.dtype == torch.bfloat16 else t.dtype)
            if t.dtype == torch.bfloat16:
                src = src.view(torch.bfloat16)
            flat[:n].copy_(src[:n])

    # ---------------- collectives ----------------
    def allreduce(self, tensors, opts=None):
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        for t in tensors:
            n = t.numel()
            if self._rawable(t):
                self._raw(_core.Op.allreduce, n, t, t, func=int(_red(op)))
            else:
                s, _ = self._upload(t, n, "ar_s")
                d, dv = self._buf(n, t.dtype, "ar_d")
                self._a.allreduce(s, d, n, _red(op), from_device=True,
                                  to_device=True)
                self._download(t, d, dv)
            if _is_avg(op):
                t.div_(self.size())
        return _Work(list(tensors))

    def broadcast(self, tensors, opts=None):
        root = opts.rootRank if opts is not None else 0
        for t in tensors:
            n = t.numel()
            if self._rawable(t):
                self._raw(_core.Op.bcast, n, t, t, root=root)
            else:
                b, bv = self._upload(t, n, "bc")
                self._a.bcast(b, n, root, from_device=True, to_device=True)
                self._download(t, b, bv)
        return _Work()

    def _allgather_base(self, output, input, opts=None):
        n = input.numel()
        if self._rawable(output, input):
            self._raw(_core.Op.allgather, n, input, output)
            return _Work()
        s, _ = self._upload(input, n, "ag_s")
        d, dv = self._buf(n * self.size(), input.dtype, "ag_d")
        self._a.allgather(s, d, n, from_device=True, to_device=True)
        self._download(output, d, dv, n * self.size())
        return _Work()

    def allgather(self, output_lists, input_list, opts=None):
        for outs, inp in zip(output_lists, input_list):
            n = inp.numel()
            s, _ = self._upload(inp, n, "ag_s")
            d, dv = self._buf(n * self.size(), inp.dtype, "ag_d")
            self._a.allgather(s, d, n, from_device=True, to_device=True)
            gathered = torch.empty(n * self.size(), dtype=inp.dtype)
            if self._gpu:
                gathered = gathered.cuda()
            self._download(gathered, d, dv)
            for r, o in enumerate(outs):
                o.reshape(-1).copy_(gathered[r * n:(r + 1) * n])
        return _Work()

    def _reduce_scatter_base(self, output, input, opts=None):
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        per = output.numel()
        if self._rawable(output, input):
            self._raw(_core.Op.reduce_scatter, per, input, output,
                      func=int(_red(op)))
        else:
            s, _ = self._upload(input, per * self.size(), "rs_s")
            d, dv = self._buf(per, input.dtype, "rs_d")
            self._a.reduce_scatter(s, d, per, _red(op), from_device=True,
                                   to_device=True)
            self._download(output, d, dv)
        if _is_avg(op):
            output.div_(self.size())
        return _Work()

    def reduce_scatter(self, outputs, input_lists, opts=None):
        for out, ins in zip(outputs, input_lists):
            flat = torch.cat([t.reshape(-1) for t in ins])
            self._reduce_scatter_base(out, flat, opts)
        return _Work()

    def reduce(self, tensors, opts=None):
        root = opts.rootRank if opts is not None else 0
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        for t in tensors:
            n = t.numel()
            s, _ = self._upload(t, n, "rd_s")
            d, dv = self._buf(n, t.dtype, "rd_d")
            self._a.reduce(s, d, n, root, _red(op), from_device=True,
                           to_device=True)
            if self.rank() == root:
                self._download(t, d, dv)
                if _is_avg(op):
                    t.div_(self.size())
        return _Work(list(tensors))

    def gather(self, output_lists, input_list, opts=None):
        root = opts.rootRank if opts is not None else 0
        for i, inp in enumerate(input_list):
            n = inp.numel()
            s, _ = self._upload(inp, n, "ga_s")
            d, dv = self._buf(n * self.size(), inp.dtype, "ga_d")
            self._a.gather(s, d, n, root, from_device=True, to_device=True)
            if self.rank() == root:
                gathered = torch.empty(n * self.size(), dtype=inp.dtype)
                if self._gpu:
                    gathered = gathered.cuda()
                self._download(gathered, d, dv)
                for r, o in enumerate(output_lists[i]):
                    o.reshape(-1).copy_(gathered[r * n:(r + 1) * n])
        return _Work()

    def scatter(self, output_list, input_lists, opts=None):
        root = opts.rootRank if opts is not None else 0
        for i, out in enumerate(output_list):
            n = out.numel()
            if self.rank() == root:
                flat = torch.cat([t.reshape(-1) for t in input_lists[i]])
                s, _ = self._upload(flat, n * self.size(), "sc_s")
            else:
                s, _ = self._buf(n, out.dtype, "sc_leaf")  # unused by leaves
            d, dv = self._buf(n, out.dtype, "sc_d")
            self._a.scatter(s, d, n, root, from_device=True, to_device=True)
            self._download(out, d, dv)
        return _Work()

    def alltoall_base(self, output, input, out_sizes, in_sizes, opts=None):
        if (out_sizes and len(set(out_sizes)) > 1) or \
                (in_sizes and len(set(in_sizes)) > 1):
            # uneven splits (MoE-style alltoallv): decompose into tagged
            # async send/recv pairs — the engine's parking + out-of-order
            # matching make the concurrent posting safe
            return self._alltoallv(output, input, out_sizes, in_sizes)
        per = input.numel() // self.size()
        if self._rawable(output, input):
            self._raw(_core.Op.alltoall, per, input, output)
            return _Work()
        s, _ = self._upload(input, per * self.size(), "a2a_s")
        d, dv = self._buf(per * self.size(), input.dtype, "a2a_d")
        self._a.alltoall(s, d, per, from_device=True, to_device=True)
        self._download(output, d, dv)
        return _Work()

    def _alltoallv(self, output, input, out_sizes, in_sizes):
        P, me = self.size(), self.rank()
        in_off = [0]
        for v in in_sizes:
            in_off.append(in_off[-1] + int(v))
        out_off = [0]
        for v in out_sizes:
            out_off.append(out_off[-1] + int(v))
        iflat, oflat = input.reshape(-1), output.reshape(-1)
        # local block
        oflat[out_off[me]:out_off[me + 1]].copy_(
            iflat[in_off[me]:in_off[me + 1]])
        reqs, keep = [], []
        base_tag = 0x2A2A0000  # private tag block, peer-disambiguated
        for p in range(P):
            if p == me:
                continue
            n_in = int(in_sizes[p])
            if n_in:
                sb, _ = self._buf(n_in, input.dtype, f"a2av_s{p}")
                self._upload_slice(iflat[in_off[p]:in_off[p] + n_in], sb)
                reqs.append((self._a.send(sb, n_in, dst=p,
                                          tag=base_tag + me,
                                          from_device=self._gpu,
                                          run_async=True), None, None, sb))
            n_out = int(out_sizes[p])
            if n_out:
                db, dv = self._buf(n_out, output.dtype, f"a2av_r{p}")
                reqs.append((self._a.recv(db, n_out, src=p,
                                          tag=base_tag + p,
                                          to_device=self._gpu,
                                          run_async=True),
                             (oflat, out_off[p], n_out, db, dv), None, db))
        for r, dl, _, buf in reqs:
            e = r.wait()
            if e:
                raise RuntimeError(
                    f"alltoallv failed: {_core.error_to_string(e)}")
            if dl is not None:
                oflat, off, n_out, db, dv = dl
                self._download_slice(oflat[off:off + n_out], db, dv)
        return _Work()

    def _upload_slice(self, flat, b):
        if self._gpu:
            v = self._a.tensor(b)
            v[:flat.numel()].copy_(flat)
            torch.cuda.current_stream().synchronize()
        else:
            arr = flat.detach().numpy() if flat.dtype != torch.bfloat16 \
                else flat.detach().view(torch.int16).numpy()
            b.write(np.ascontiguousarray(arr).view(np.int8))

    def _download_slice(self, flat, b, view=None):
        n = flat.numel()
        if self._gpu:
            v = view if view is not None else self._a.tensor(b)
            flat.copy_(v[:n])
            torch.cuda.current_stream().synchronize()
        else:
            raw = np.zeros(n * flat.element_size(), np.int8)
            b.read(raw)
            src = torch.from_numpy(raw).view(
                torch.int16 if flat.dtype == torch.bfloat16 else flat.dtype)
            if flat.dtype == torch.bfloat16:
                src = src.view(torch.bfloat16)
            flat.copy_(src[:n])

    def send(self, tensors, dst, tag):
        for t in tensors:
            n = t.numel()
            b, _ = self._upload(t, n, "sr")
            self._a.send(b, n, dst=dst, tag=tag, from_device=True)
        return _Work()

    def recv(self, tensors, src, tag):
        for t in tensors:
            n = t.numel()
            b, bv = self._buf(n, t.dtype, "rr")
            self._a.recv(b, n, src=src, tag=tag, to_device=True)
            self._download(t, b, bv)
        return _Work()

    def barrier(self, opts=None):
        self._a.barrier()
        return _Work()

    def getBackendName(self):
        return "accl"


def _create(store, rank, size, timeout=None, **kwargs):
    # c10d may hand us a PrefixStore wrapping the rendezvous store
    return AcclProcessGroup(store, rank, size)


def register():
    if "accl" not in dist.Backend.backend_list:
        dist.Backend.register_backend("accl", _create,
                                      devices=["cpu", "cuda"])


register()
