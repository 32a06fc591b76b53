"""torch.distributed backend "accl" over the engine (emulator on CPU).

Exceeds the reference surface: torch DDP users can switch to accl_amd with
one init_process_group call. GPU path shares this code with DLPack staging.
"""
import multiprocessing as mp
import os

import numpy as np
import pytest
import torch


def _rank_main(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        import accl_amd.torch_backend  # noqa: F401  (registers "accl")
        dist.init_process_group("accl", rank=rank, world_size=world)

        t = torch.full((1000,), float(rank + 1))
        dist.all_reduce(t)
        assert torch.allclose(t, torch.full((1000,), 3.0)), t[:4]

        b = torch.full((64,), float(rank * 7 + 1))
        dist.broadcast(b, src=1)
        assert torch.allclose(b, torch.full((64,), 8.0))

        out = torch.zeros(2 * 128)
        inp = torch.full((128,), float(rank + 5))
        dist.all_gather_into_tensor(out, inp)
        assert out[:128].eq(5.0).all() and out[128:].eq(6.0).all()

        rs_in = torch.arange(256, dtype=torch.float32) + rank
        rs_out = torch.zeros(128)
        dist.reduce_scatter_tensor(rs_out, rs_in)
        exp = (torch.arange(256, dtype=torch.float32)[rank * 128:(rank + 1) * 128]
               * 2 + 1)
        assert torch.allclose(rs_out, exp), (rs_out[:4], exp[:4])

        if rank == 0:
            dist.send(torch.full((32,), 9.0), dst=1, tag=3)
        else:
            r = torch.zeros(32)
            dist.recv(r, src=0, tag=3)
            assert r.eq(9.0).all()

        dist.barrier()

        # rooted collectives: reduce / gather / scatter
        rt = torch.full((96,), float(rank + 1))
        dist.reduce(rt, dst=1)
        if rank == 1:
            assert torch.allclose(rt, torch.full((96,), 3.0)), rt[:4]
        g_in = torch.full((48,), float(10 + rank))
        g_out = [torch.zeros(48) for _ in range(world)] if rank == 0 else None
        dist.gather(g_in, g_out, dst=0)
        if rank == 0:
            for r in range(world):
                assert g_out[r].eq(float(10 + r)).all(), (r, g_out[r][:3])
        sc_out = torch.zeros(40)
        sc_in = ([torch.full((40,), float(20 + r)) for r in range(world)]
                 if rank == 1 else None)
        dist.scatter(sc_out, sc_in, src=1)
        assert sc_out.eq(float(20 + rank)).all(), sc_out[:3]

        # uneven all_to_all_single (MoE-style alltoallv): rank r sends
        # (p+1)*10*(r+1) elems to p; receives (r+1)*10*(p+1) from p
        in_sizes = [(p + 1) * 10 * (rank + 1) for p in range(world)]
        out_sizes = [(rank + 1) * 10 * (p + 1) for p in range(world)]
        a2a_in = torch.cat([
            torch.full((in_sizes[p],), float(100 * rank + p))
            for p in range(world)])
        a2a_out = torch.zeros(sum(out_sizes))
        dist.all_to_all_single(a2a_out, a2a_in, out_sizes, in_sizes)
        off = 0
        for p in range(world):
            seg = a2a_out[off:off + out_sizes[p]]
            assert seg.eq(float(100 * p + rank)).all(), (rank, p, seg[:3])
            off += out_sizes[p]

        # one real DDP training step over the backend
        from torch.nn.parallel import DistributedDataParallel as DDP
        torch.manual_seed(7)
        model = torch.nn.Linear(16, 4)
        ddp = DDP(model)
        x = torch.randn(8, 16) * (rank + 1)
        loss = ddp(x).pow(2).mean()
        loss.backward()
        g0 = model.weight.grad.clone()
        # grads must be identical (averaged) on every rank
        gather = [torch.zeros_like(g0) for _ in range(world)]
        dist.all_gather(gather, g0)
        assert torch.allclose(gather[0], gather[1], atol=1e-6)

        dist.destroy_process_group()
        q.put((rank, None))
    except Exception:
        import traceback
        q.put((rank, traceback.format_exc()))


def test_torch_backend_collectives_and_ddp():
    ctx = mp.get_context("fork")
    q = ctx.Queue()
    port = 29650 + os.getpid() % 200
    ps = [ctx.Process(target=_rank_main, args=(r, 2, port, q)) for r in range(2)]
    for p in ps:
        p.start()
    errs = []
    try:
        for _ in range(2):
            rank, err = q.get(timeout=180)
            if err:
                errs.append(f"rank {rank}:\n{err}")
    finally:
        for p in ps:
            p.join(timeout=15)
            if p.is_alive():
                p.terminate()
                errs.append("rank hung")
    assert not errs, "\n".join(errs)


# ---------------- GPU engine variant (driver runs with -m gpu) ----------
def _gpu_rank(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ.setdefault("ACCL_ENGINE_WGS", "64")  # 2 engines, 1 GPU
        import torch.distributed as dist

        import accl_amd.torch_backend  # noqa: F401
        dist.init_process_group("accl", rank=rank, world_size=world)
        t = torch.full((4096,), float(rank + 1), device="cuda")
        dist.all_reduce(t)
        assert torch.allclose(t.cpu(), torch.full((4096,), 3.0))
        b = torch.full((256,), float(rank), device="cuda")
        dist.broadcast(b, src=0)
        assert b.cpu().eq(0.0).all()
        out = torch.zeros(2 * 128, device="cuda")
        dist.all_gather_into_tensor(out, torch.full((128,), float(rank + 5),
                                                    device="cuda"))
        assert out[:128].cpu().eq(5.0).all() and out[128:].cpu().eq(6.0).all()
        rt = torch.full((96,), float(rank + 1), device="cuda")
        dist.reduce(rt, dst=1)
        if rank == 1:
            assert rt.cpu().eq(3.0).all()
        if rank == 0:
            dist.send(torch.full((64,), 7.0, device="cuda"), dst=1, tag=9)
        else:
            rbuf = torch.zeros(64, device="cuda")
            dist.recv(rbuf, src=0, tag=9)
            assert rbuf.cpu().eq(7.0).all()
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, None))
    except Exception:
        import traceback
        q.put((rank, traceback.format_exc()))


@pytest.mark.gpu
def test_torch_backend_gpu():
    try:
        if not torch.cuda.is_available():
            pytest.skip("no HIP GPU")
    except Exception:
        pytest.skip("no HIP GPU")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29700 + os.getpid() % 200
    ps = [ctx.Process(target=_gpu_rank, args=(r, 2, port, q)) for r in range(2)]
    for p in ps:
        p.start()
    errs = []
    try:
        for _ in range(2):
            rank, err = q.get(timeout=180)
            if err:
                errs.append(f"rank {rank}:\n{err}")
    finally:
        for p in ps:
            p.join(timeout=15)
            if p.is_alive():
                p.terminate()
                errs.append("rank hung")
    assert not errs, "\n".join(errs)
