#!/usr/bin/env python3
"""End-to-end torch DDP training over the accl_amd backend.

Run (CPU emulator):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 examples/ddp_train.py
On MI355X GPUs the same command trains with gradients reduced by the
persistent engine over xGMI (one rank per GPU).
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

import accl_amd.torch_backend  # noqa: F401  (registers "accl")


def main():
    dist.init_process_group("accl")
    rank = dist.get_rank()
    use_gpu = torch.cuda.is_available()
    dev = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0))) \
        if use_gpu else torch.device("cpu")

    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(),
        torch.nn.Linear(128, 10)).to(dev)
    ddp = DDP(model)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.05)

    torch.manual_seed(100 + rank)  # different data per rank
    for step in range(5):
        x = torch.randn(32, 64, device=dev)
        y = torch.randint(0, 10, (32,), device=dev)
        loss = torch.nn.functional.cross_entropy(ddp(x), y)
        opt.zero_grad()
        loss.backward()   # grads all-reduced by the accl engine
        opt.step()
        if rank == 0:
            print(f"step {step}: loss {loss.item():.4f}", flush=True)

    # verify replicas stayed in sync
    w = model[0].weight.detach().reshape(-1)[:1000].contiguous().cpu()
    ws = [torch.zeros_like(w) for _ in range(dist.get_world_size())]
    dist.all_gather(ws, w)
    assert all(torch.allclose(ws[0], wi, atol=1e-6) for wi in ws)
    if rank == 0:
        print("DDP training over accl_amd OK (replicas in sync)")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
