"""Per-pair xGMI link probe: measures D2D copy bandwidth for every GPU
pair (hipMemcpyPeer via torch) and prints a matrix — the first thing to
run on a fresh multi-GPU node (feeds tools/autotune.py slot-geometry
choices). Reference analogue: accl_network_utils link bring-up."""
import json
import sys

import torch


def probe(nbytes=256 << 20, iters=5):
    n = torch.cuda.device_count()
    out = {"n_gpus": n, "bytes": nbytes, "pairs": {}}
    bufs = []
    for d in range(n):
        with torch.cuda.device(d):
            bufs.append(torch.empty(nbytes, dtype=torch.uint8, device=f"cuda:{d}"))
    for i in range(n):
        for j in range(n):
            if i == j:
                continue
            torch.cuda.synchronize(i)
            torch.cuda.synchronize(j)
            import time
            with torch.cuda.device(i):
                bufs[j].copy_(bufs[i])  # warm + map
                torch.cuda.synchronize(i)
                t0 = time.perf_counter()
                for _ in range(iters):
                    bufs[j].copy_(bufs[i])
                torch.cuda.synchronize(i)
                dt = (time.perf_counter() - t0) / iters
            out["pairs"][f"{i}->{j}"] = round(nbytes / dt / 1e9, 1)
    return out


if __name__ == "__main__":
    nb = int(sys.argv[1]) if len(sys.argv) > 1 else 256 << 20
    print(json.dumps(probe(nb), indent=1))
