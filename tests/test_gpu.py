"""GPU-engine tests (run on an MI355X via gpurun; every test is @gpu).

The 2-rank tests run two processes sharing one GPU — same IPC/arena/engine
protocol as the multi-GPU case, with xGMI replaced by local HBM; the 8-GPU
path is exercised by the driver's round-end scaling bench."""
import numpy as np
import pytest

import accl_amd as A
from emu_util import pattern, rd, run_ranks

DT = A.DataType
RF = A.ReduceFunction

gpu = pytest.mark.gpu


def _has_gpu():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


pytestmark = [gpu, pytest.mark.skipif(not _has_gpu(), reason="no HIP GPU")]


@pytest.fixture(scope="module")
def acc1():
    a = A.ACCL(nranks=1, rank=0, backend="gpu", job="g1", heap_bytes=2 << 30)
    yield a
    a.close()


def test_copy_engine(acc1):
    a = acc1
    for cnt in (17, 4096, 1 << 20):
        s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
        x = np.random.default_rng(cnt).standard_normal(cnt, dtype=np.float32)
        s.write(x)
        a.copy(s, d, cnt)
        assert np.array_equal(rd(d, cnt), x)


def test_combine_numerics_vs_torch(acc1):
    """Engine reduce kernel vs a plain PyTorch fp32 reference."""
    import torch
    a = acc1
    cnt = 1 << 20
    s1, s2, d = (a.create_buffer(cnt, DT.float32) for _ in range(3))
    x = np.random.default_rng(1).standard_normal(cnt, dtype=np.float32)
    y = np.random.default_rng(2).standard_normal(cnt, dtype=np.float32)
    s1.write(x); s2.write(y)
    a.combine(cnt, RF.SUM, s1, s2, d)
    ref = (torch.from_numpy(x) + torch.from_numpy(y)).numpy()
    assert np.allclose(rd(d, cnt), ref, atol=0), "fp32 add must be exact"
    a.combine(cnt, RF.MAX, s1, s2, d)
    ref = torch.maximum(torch.from_numpy(x), torch.from_numpy(y)).numpy()
    assert np.array_equal(rd(d, cnt), ref)


def test_combine_dtypes(acc1):
    import torch
    a = acc1
    cnt = 8192
    for dt, tdt, tol in ((DT.bfloat16, torch.bfloat16, 0.0),
                         (DT.float16, torch.float16, 0.0),
                         (DT.float64, torch.float64, 0.0),
                         (DT.int32, torch.int32, 0.0)):
        s1, s2, d = (a.create_buffer(cnt, dt) for _ in range(3))
        t1 = torch.randn(cnt, dtype=torch.float32).to(tdt) if tdt.is_floating_point \
            else torch.randint(-1000, 1000, (cnt,), dtype=tdt)
        t2 = torch.randn(cnt, dtype=torch.float32).to(tdt) if tdt.is_floating_point \
            else torch.randint(-1000, 1000, (cnt,), dtype=tdt)
        s1.write(t1.view(torch.int8).numpy() if tdt == torch.bfloat16 else t1.numpy())
        s2.write(t2.view(torch.int8).numpy() if tdt == torch.bfloat16 else t2.numpy())
        a.combine(cnt, RF.SUM, s1, s2, d)
        ref = (t1.float() + t2.float()).to(tdt)
        got = np.zeros(cnt * t1.element_size(), np.int8)
        d.read(got.view(np.int8))
        got_t = torch.from_numpy(got).view(tdt)
        fr = ref.float()
        fg = got_t.float()
        assert torch.allclose(fg, fr, atol=float(tol), rtol=1e-2), str(dt)


def test_allreduce_single(acc1):
    a = acc1
    cnt = 1 << 22
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
    x = np.random.default_rng(3).standard_normal(cnt, dtype=np.float32)
    s.write(x)
    a.allreduce(s, d, cnt, RF.SUM)
    assert np.array_equal(rd(d, cnt), x)


# ---------------- 2 processes, 1 GPU: full protocol over IPC ----------------
def _ar2(a, rank, n):
    for cnt in (1000, 1 << 20):
        s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
        s.write(pattern(cnt, rank, seed=cnt))
        a.allreduce(s, d, cnt, RF.SUM)
        exp = np.stack([pattern(cnt, r, seed=cnt) for r in range(n)]).sum(0)
        assert np.allclose(rd(d, cnt), exp), f"cnt={cnt}"


def _sendrecv2(a, rank, n):
    cnt = 200_000
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
    s.write(pattern(cnt, rank))
    if rank == 0:
        a.send(s, cnt, dst=1, tag=5)
        a.recv(d, cnt, src=1, tag=6)
        assert np.array_equal(rd(d, cnt), pattern(cnt, 1))
    else:
        a.recv(d, cnt, src=0, tag=5)
        assert np.array_equal(rd(d, cnt), pattern(cnt, 0))
        a.send(s, cnt, dst=0, tag=6)
    a.barrier()


def _coll2(a, rank, n):
    cnt = 50_000
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt * n, DT.float32)
    s.write(pattern(cnt, rank, seed=5))
    a.allgather(s, d, cnt)
    exp = np.concatenate([pattern(cnt, r, seed=5) for r in range(n)])
    assert np.array_equal(rd(d, cnt * n), exp)
    s2 = a.create_buffer(cnt * n, DT.float32)
    d2 = a.create_buffer(cnt, DT.float32)
    s2.write(np.concatenate([pattern(cnt, 100 * rank + j, seed=8)
                             for j in range(n)]))
    a.reduce_scatter(s2, d2, cnt, RF.SUM)
    exp = np.stack([pattern(cnt, 100 * r + rank, seed=8)
                    for r in range(n)]).sum(0)
    assert np.allclose(rd(d2, cnt), exp)


@pytest.mark.parametrize("fn", [_ar2, _sendrecv2, _coll2],
                         ids=["allreduce", "sendrecv", "allgather_rs"])
def test_two_ranks_one_gpu(fn):
    run_ranks(fn, 2, backend="gpu", timeout=180)


# ---------------- streaming surface on the GPU engine ----------------
def test_stream_engine_loopback(acc1):
    """Engine-driven stream_put to self + host pop_stream."""
    a = acc1
    cnt = 5000
    s = a.create_buffer(cnt, DT.float32)
    x = np.random.default_rng(4).standard_normal(cnt, dtype=np.float32)
    s.write(x)
    a.stream_put(s, cnt, dst=0, tag=17)
    out = np.zeros(cnt, np.float32)
    nb, tag = a.pop_stream(0, out)
    assert nb == cnt * 4 and tag == 17
    assert np.array_equal(out, x)


def test_device_vadd_put(acc1):
    """Device-initiated stream: vadd_put kernel pushes from INSIDE a HIP
    kernel (reference: kernels/plugins/vadd_put/vadd_put.cpp demo)."""
    import accl_amd._core as core
    a = acc1
    cnt = 20000  # spans multiple 32KB segments
    s = a.create_buffer(cnt, DT.float32)
    x = np.random.default_rng(5).standard_normal(cnt, dtype=np.float32)
    s.write(x)
    core.demo_vadd_put(a._a, s, cnt, 0, 23, 1.0)
    got = np.zeros(0, np.float32)
    buf = np.zeros(cnt, np.float32)
    while got.size < cnt:
        nb, tag = a.pop_stream(0, buf)
        assert tag == 23 and nb > 0
        got = np.concatenate([got, buf[:nb // 4]])
    assert np.allclose(got, x + 1.0)


def _stream2(a, rank, n):
    cnt = 9000
    s = a.create_buffer(cnt, DT.float32)
    if rank == 0:
        s.write(pattern(cnt, 2))
        a.stream_put(s, cnt, dst=1, tag=3)
    else:
        out = np.zeros(cnt, np.float32)
        nb, tag = a.pop_stream(0, out)
        assert nb == cnt * 4 and tag == 3
        assert np.array_equal(out, pattern(cnt, 2))
    a.barrier()


def test_two_ranks_stream():
    run_ranks(_stream2, 2, backend="gpu", timeout=180)


def _subcomm2(a, rank, n):
    """Subgroup communicator on the GPU engine (reference multicomm tests,
    test/host/xrt/src/test.cpp:756-833)."""
    sub = a.split_communicator([0, 1])
    cnt = 10000
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
    s.write(pattern(cnt, rank, seed=11))
    a.allreduce(s, d, cnt, RF.SUM, comm=sub)
    exp = np.stack([pattern(cnt, r, seed=11) for r in range(n)]).sum(0)
    assert np.allclose(rd(d, cnt), exp)
    a.barrier(comm=sub)


def _compressed2(a, rank, n):
    """f32 data, f16 wire dtype (reference compression lane tests)."""
    cnt = 4096
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
    x = pattern(cnt, rank, seed=13) / 64.0
    s.write(x)
    a.allreduce(s, d, cnt, RF.SUM, compress_dtype=DT.float16)
    exp = np.stack([pattern(cnt, r, seed=13) / 64.0 for r in range(n)]).sum(0)
    assert np.allclose(rd(d, cnt), exp, atol=0.05), "f16-wire allreduce"


@pytest.mark.parametrize("fn", [_subcomm2, _compressed2],
                         ids=["subcomm", "compressed"])
def test_two_ranks_extra(fn):
    run_ranks(fn, 2, backend="gpu", timeout=180)


def test_device_call_collective(acc1):
    """A HIP kernel issues the engine's own stream_put through the
    device-call ring (reference: ACCLCommand via client_arbiter,
    accl_hls.h:134-188 + client_arbiter.cpp:21-51)."""
    import accl_amd._core as core
    a = acc1
    cnt = 3000
    s = a.create_buffer(cnt, DT.float32)
    scratch = a.create_buffer(cnt, DT.float32, device_only=True)
    x = np.random.default_rng(8).standard_normal(cnt, dtype=np.float32)
    s.write(x)
    core.demo_vadd_devicecall(a._a, s, scratch, cnt, 0, 31, 2.5)
    out = np.zeros(cnt, np.float32)
    nb, tag = a.pop_stream(0, out)
    assert nb == cnt * 4 and tag == 31
    assert np.allclose(out, x + 2.5)


# ---------------- round-2: barrier-free sequences + stress ----------------
def _b2b2(a, rank, n):
    """Barrier-free multi-collective chains (round-1 fresh-box failure
    shape) — a rank running ahead must not wedge its peer."""
    for it in range(3):
        cnt = 50_000 if it == 0 else 7000
        s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt * n, DT.float32)
        s.write(pattern(cnt, rank, seed=90 + it))
        a.allgather(s, d, cnt)
        exp = np.concatenate([pattern(cnt, r, seed=90 + it) for r in range(n)])
        assert np.array_equal(rd(d, cnt * n), exp)
        s2 = a.create_buffer(cnt * n, DT.float32)
        d2 = a.create_buffer(cnt, DT.float32)
        s2.write(np.concatenate([pattern(cnt, 100 * rank + j, seed=95 + it)
                                 for j in range(n)]))
        a.reduce_scatter(s2, d2, cnt, RF.SUM)
        exp = np.stack([pattern(cnt, 100 * r + rank, seed=95 + it)
                        for r in range(n)]).sum(0)
        assert np.allclose(rd(d2, cnt), exp)
        d3 = a.create_buffer(cnt, DT.float32)
        a.allreduce(s, d3, cnt, RF.SUM)
        exp = np.stack([pattern(cnt, r, seed=90 + it) for r in range(n)]).sum(0)
        assert np.allclose(rd(d3, cnt), exp)


def test_two_ranks_back_to_back():
    run_ranks(_b2b2, 2, backend="gpu", timeout=240)


def _stress2(a, rank, n):
    """Randomized op mix: sizes spanning inline (<32 KB) and fleet moves,
    bursts of small ops followed by large ones (the mover-ring recycle
    path), mixed tags."""
    rng = np.random.default_rng(123)  # same sequence on all ranks
    for it in range(30):
        op = rng.integers(0, 4)
        cnt = int(rng.integers(64, 300_000))
        if op == 0:
            s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
            s.write(pattern(cnt, rank, seed=it))
            a.allreduce(s, d, cnt, RF.SUM)
            exp = np.stack([pattern(cnt, r, seed=it) for r in range(n)]).sum(0)
            assert np.allclose(rd(d, cnt), exp), f"it={it} allreduce cnt={cnt}"
        elif op == 1:
            s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt * n, DT.float32)
            s.write(pattern(cnt, rank, seed=it))
            a.allgather(s, d, cnt)
            exp = np.concatenate([pattern(cnt, r, seed=it) for r in range(n)])
            assert np.array_equal(rd(d, cnt * n), exp), f"it={it} allgather"
        elif op == 2:
            tag = int(rng.integers(1, 1000))
            s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
            s.write(pattern(cnt, rank, seed=it))
            if rank == 0:
                a.send(s, cnt, dst=1, tag=tag)
            elif rank == 1:
                a.recv(d, cnt, src=0, tag=tag)
                assert np.array_equal(rd(d, cnt), pattern(cnt, 0, seed=it))
        else:
            b = a.create_buffer(cnt, DT.float32)
            if rank == 0:
                b.write(pattern(cnt, 55, seed=it))
            a.bcast(b, cnt, 0)
            assert np.array_equal(rd(b, cnt), pattern(cnt, 55, seed=it))
    a.barrier()


def test_two_ranks_gpu_stress():
    run_ranks(_stress2, 2, backend="gpu", timeout=420)


def test_inline_burst_then_fleet(acc1):
    """>MOVE_RING consecutive sub-32KB moves (inline path, no fleet
    doorbell) followed by a large fleet move: the mover cursors must step
    over the recycled slots instead of wedging (round-1 advisor finding)."""
    a = acc1
    small = a.create_buffer(256, DT.float32)
    sd = a.create_buffer(256, DT.float32)
    x = np.arange(256, dtype=np.float32)
    small.write(x)
    for _ in range(80):  # > MOVE_RING=64 inline submits
        a.copy(small, sd, 256)
    cnt = 1 << 22  # 16 MB fleet move
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
    y = np.random.default_rng(9).standard_normal(cnt, dtype=np.float32)
    s.write(y)
    a.copy(s, d, cnt)
    assert np.array_equal(rd(d, cnt), y)
    assert np.array_equal(rd(sd, 256), x)


DIRECT_GPU = {"max_eager": 4096, "n_slots": 4, "slot_bytes": 4096,
              "timeout_us": 20_000_000}


def _direct2(a, rank, n):
    """All collectives through the address-exchange direct path on the GPU
    engine (max_eager forced tiny; windowed n-ary fan-in for reduce/RS)."""
    cnt = 6000
    s = a.create_buffer(cnt * n, DT.float32)
    d = a.create_buffer(cnt, DT.float32)
    s.write(np.concatenate([pattern(cnt, rank * 10 + j, seed=131)
                            for j in range(n)]))
    a.scatter(s, d, cnt, 0)
    assert np.array_equal(rd(d, cnt), pattern(cnt, rank, seed=131))
    g = a.create_buffer(cnt * n, DT.float32)
    a.gather(d, g, cnt, 0)
    if rank == 0:
        exp = np.concatenate([pattern(cnt, r, seed=131) for r in range(n)])
        assert np.array_equal(rd(g, cnt * n), exp)
    a.barrier()
    a.alltoall(s, g, cnt)
    exp = np.concatenate([pattern(cnt, r * 10 + rank, seed=131)
                          for r in range(n)])
    assert np.array_equal(rd(g, cnt * n), exp)
    a.reduce_scatter(s, d, cnt, RF.SUM)
    exp = np.stack([pattern(cnt, r * 10 + rank, seed=131)
                    for r in range(n)]).sum(0)
    assert np.allclose(rd(d, cnt), exp)
    dr = a.create_buffer(cnt * n, DT.float32)
    a.reduce(s, dr, cnt * n, 0, RF.SUM)
    if rank == 0:
        exp = np.stack([np.concatenate([pattern(cnt, r * 10 + j, seed=131)
                                        for j in range(n)])
                        for r in range(n)]).sum(0)
        assert np.allclose(rd(dr, cnt * n), exp)
    a.barrier()


def test_two_ranks_direct_paths():
    run_ranks(_direct2, 2, backend="gpu", opts=DIRECT_GPU, timeout=240)


def _parked2(a, rank, n):
    """Parked recv + interleaved traffic + reversed-tag matching on the GPU
    engine (multi-call interleaving / CMD_CALL_RETRY analogue)."""
    import time
    cnt = 5000
    if rank == 0:
        d = a.create_buffer(cnt, DT.float32)
        req = a.recv(d, cnt, src=1, tag=5, run_async=True)
        t0 = time.monotonic()
        for i in range(6):
            s1 = a.create_buffer(2000, DT.float32)
            d1 = a.create_buffer(2000, DT.float32)
            x = pattern(2000, i)
            s1.write(x)
            a.copy(s1, d1, 2000)
            assert np.array_equal(rd(d1, 2000), x)
        assert time.monotonic() - t0 < 2.0, "copies blocked behind parked recv"
        assert req.wait() == 0
        assert np.array_equal(rd(d, cnt), pattern(cnt, 9, seed=3))
        # reversed tags
        d2, d1b = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
        r2 = a.recv(d2, cnt, src=1, tag=12, run_async=True)
        r1 = a.recv(d1b, cnt, src=1, tag=11, run_async=True)
        assert r1.wait() == 0 and r2.wait() == 0
        assert np.array_equal(rd(d1b, cnt), pattern(cnt, 11, seed=4))
        assert np.array_equal(rd(d2, cnt), pattern(cnt, 12, seed=4))
    else:
        time.sleep(2.2)
        s = a.create_buffer(cnt, DT.float32)
        s.write(pattern(cnt, 9, seed=3))
        a.send(s, cnt, dst=0, tag=5)
        s1, s2 = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
        s1.write(pattern(cnt, 11, seed=4))
        s2.write(pattern(cnt, 12, seed=4))
        a.send(s1, cnt, dst=0, tag=11)
        a.send(s2, cnt, dst=0, tag=12)
    a.barrier()


def test_two_ranks_parked_interleave():
    run_ranks(_parked2, 2, backend="gpu", timeout=240)


SMALL_GPU = {"n_slots": 4, "slot_bytes": 4096, "timeout_us": 20_000_000}


def _drain2(a, rank, n):
    """Cross-rank park-deadlock regressions on the GPU engine under small
    slots (same scenarios as the emulator's test_head_to_head_async and
    test_parked_recv_pool_overflow: flow-free parked-send pushes and
    drain-to-destination for parked recvs)."""
    cnt = 9_000
    other = 1 - rank
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
    s.write(pattern(cnt, rank, seed=2))
    sreq = a.send(s, cnt, dst=other, tag=7, run_async=True)
    rreq = a.recv(d, cnt, src=other, tag=7, run_async=True)
    a.barrier()
    assert sreq.wait() == 0 and rreq.wait() == 0
    assert np.array_equal(rd(d, cnt), pattern(cnt, other, seed=2))
    # parked recv whose message overflows the unexpected pool while the
    # engine grinds through collectives
    big = 12_000
    if rank == 1:
        db = a.create_buffer(big, DT.float32)
        req = a.recv(db, big, src=0, tag=88, run_async=True)
    else:
        sv = a.create_buffer(big, DT.float32)
        sv.write(pattern(big, 4, seed=5))
        sq = a.send(sv, big, dst=1, tag=88, run_async=True)
    s2, d2 = a.create_buffer(600, DT.float32), a.create_buffer(600, DT.float32)
    s2.write(pattern(600, rank, seed=11))
    for _ in range(8):
        a.allreduce(s2, d2, 600, RF.SUM)
    exp = np.stack([pattern(600, r, seed=11) for r in range(n)]).sum(0)
    assert np.allclose(rd(d2, 600), exp)
    if rank == 1:
        assert req.wait() == 0
        assert np.array_equal(rd(db, big), pattern(big, 4, seed=5))
    else:
        assert sq.wait() == 0
    a.barrier()


def test_two_ranks_drain_regressions():
    run_ranks(_drain2, 2, backend="gpu", opts=SMALL_GPU, timeout=240)


def _compressed_large(a, rank, n):
    """Fleet-path compression tiles: 4 MB f32 message on an f16 wire
    (vectorized cast tx, fused cast+reduce rx, up-cast phase 2)."""
    cnt = 1 << 20
    s, d = a.create_buffer(cnt, DT.float32), a.create_buffer(cnt, DT.float32)
    x = (pattern(cnt, rank, seed=41) / 8.0).astype(np.float32)
    s.write(x)
    a.allreduce(s, d, cnt, RF.SUM, compress_dtype=DT.float16)
    exp = np.stack([pattern(cnt, r, seed=41) / 8.0 for r in range(n)]).sum(0)
    assert np.allclose(rd(d, cnt), exp, atol=0.02), "f16-wire fleet allreduce"


def test_two_ranks_compressed_large():
    run_ranks(_compressed_large, 2, backend="gpu", timeout=240)


def _nary4(a, rank, n):
    """4-rank direct reduce/reduce_scatter: n-ary (nsrc=4) vectorized
    fan-in tiles vs numpy reference."""
    cnt = 40_000
    s = a.create_buffer(cnt * n, DT.float32)
    d = a.create_buffer(cnt, DT.float32)
    s.write(np.concatenate([pattern(cnt, rank * 7 + j, seed=55)
                            for j in range(n)]))
    a.reduce_scatter(s, d, cnt, RF.SUM)
    exp = np.stack([pattern(cnt, r * 7 + rank, seed=55)
                    for r in range(n)]).sum(0)
    assert np.allclose(rd(d, cnt), exp), "4-rank direct RS"
    dr = a.create_buffer(cnt * n, DT.float32)
    a.reduce(s, dr, cnt * n, 2, RF.SUM)
    if rank == 2:
        exp = np.stack([np.concatenate([pattern(cnt, r * 7 + j, seed=55)
                                        for j in range(n)])
                        for r in range(n)]).sum(0)
        assert np.allclose(rd(dr, cnt * n), exp), "4-rank direct reduce"
    a.barrier()
    # bf16 n-ary fan-in (config-4 dtype)
    import torch
    sb = a.create_buffer(cnt * n, DT.bfloat16)
    db = a.create_buffer(cnt, DT.bfloat16)
    tb = torch.from_numpy(np.concatenate(
        [pattern(cnt, rank * 7 + j, seed=56) / 16 for j in range(n)])) \
        .to(torch.bfloat16)
    sb.write(tb.view(torch.int8).numpy())
    a.reduce_scatter(sb, db, cnt, RF.SUM)
    allb = [torch.from_numpy(np.concatenate(
        [pattern(cnt, r * 7 + j, seed=56) / 16 for j in range(n)]))
        .to(torch.bfloat16) for r in range(n)]
    accs = [b[rank * cnt:(rank + 1) * cnt] for b in allb]
    # fp32 reference with tolerance (bf16 fold order differs by at most
    # a few ulps across schedules)
    ref32 = accs[rank].float()
    for p in range(n):
        if p != rank:
            ref32 = ref32 + accs[p].float()
    got = torch.from_numpy(rd(db, cnt, np.uint16)).view(torch.bfloat16)
    assert torch.allclose(got.float(), ref32, atol=0.6, rtol=0.02), \
        "bf16 n-ary RS numerics"
    a.barrier()


def test_four_ranks_nary_direct():
    run_ranks(_nary4, 4, backend="gpu", opts=DIRECT_GPU, timeout=300)


# torch-CUDA interop LAST: a torch-runtime failure beside the persistent
# engine must not mask the engine tests under -x
def test_dlpack_tensor_view(acc1):
    import torch
    a = acc1
    cnt = 4096
    b = a.create_buffer(cnt, DT.float32, device_only=True)
    t = a.tensor(b)
    assert t.is_cuda and t.numel() == cnt
    t.fill_(3.0)
    # device-wide sync would hang on the persistent engine kernel
    torch.cuda.current_stream().synchronize()
    d = a.create_buffer(cnt, DT.float32)
    a.copy(b, d, cnt, from_device=True)
    assert np.allclose(rd(d, cnt), 3.0)


def _put2(a, rank, n):
    """One-sided xGMI put (reference copy_p2p) on the GPU engine."""
    cnt = 30_000
    d = a.create_buffer(cnt, DT.float32)  # same alloc order -> same offset
    if rank == 0:
        s = a.create_buffer(cnt, DT.float32)
        s.write(pattern(cnt, 42, seed=6))
        a.put(s, cnt, 1, d.arena_offset)
    a.barrier()
    if rank == 1:
        assert np.array_equal(rd(d, cnt), pattern(cnt, 42, seed=6))
    a.barrier()


def test_two_ranks_one_sided_put():
    run_ranks(_put2, 2, backend="gpu", timeout=180)
