#!/usr/bin/env python3
"""Randomized protocol fuzz campaign driver.

Replays the late-round-2 validation campaigns: both fuzz generations
(mixed-op scripts and permuted-posting-order bursts) across world sizes
and ring geometries from production-like down to deliberately hostile
(2-slot/1KB eager rings, 2-deep rendezvous rings).

Usage:
  python tools/fuzz_campaign.py                    # 200 runs, seeds 0..
  python tools/fuzz_campaign.py --seeds 1000 2500  # the mega-soak range
  python tools/fuzz_campaign.py --gpu              # GPU engine (on a box)

Every failure prints seed/P/geometry so it can be locked into
tests/test_emulator.py::test_protocol_fuzz_tiny.
"""
import argparse
import random
import sys
import time
import pathlib

ROOT = pathlib.Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))
sys.path.insert(0, str(ROOT / "tests"))

from emu_util import run_ranks  # noqa: E402
import test_emulator as T  # noqa: E402

GEOS = [
    ("SMALL", T.SMALL),
    ("DIRECT", T.DIRECT),
    ("TINY", {"n_slots": 2, "slot_bytes": 1024, "timeout_us": 20_000_000}),
    ("TINYD", {"n_slots": 2, "slot_bytes": 1024, "max_eager": 1024,
               "n_rndzv": 2, "timeout_us": 20_000_000}),
    ("MID", {"n_slots": 8, "slot_bytes": 16384, "timeout_us": 20_000_000}),
    ("MIDD", {"n_slots": 8, "slot_bytes": 16384, "max_eager": 16384,
              "n_rndzv": 4, "timeout_us": 20_000_000}),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seeds", nargs=2, type=int, default=[0, 200])
    ap.add_argument("--rng", type=int, default=778)
    ap.add_argument("--gpu", action="store_true",
                    help="run against the GPU engine (procs share one GPU)")
    args = ap.parse_args()

    rng = random.Random(args.rng)
    t0 = time.time()
    n_ok, fails = 0, []
    backend = "gpu" if args.gpu else None
    for seed in range(args.seeds[0], args.seeds[1]):
        P = rng.choice([2, 2, 3, 3, 4, 4, 6, 8] if not args.gpu
                       else [2, 2, 2, 3])
        name, opts = rng.choice(GEOS)
        gen = rng.choice([T._fuzz, T._fuzz, T._fuzz2])
        # on rendezvous-heavy geometries, half the runs also cap the
        # window size so multi-window paths (posting rotation, drain
        # window consumption) get fuzzed
        wind = 8192 if (name.endswith("D") and seed % 2) else 0

        def body(a, r, nn, gen=gen, seed=seed, wind=wind):
            if wind:
                a.set_max_rendezvous_size(wind)
            gen(a, r, nn, seed)

        try:
            if backend:
                run_ranks(body, P, backend=backend, opts=opts, timeout=240)
            else:
                run_ranks(body, P, opts=opts, timeout=240)
            n_ok += 1
        except Exception as e:  # noqa: BLE001 — report and continue
            fails.append((gen.__name__, seed, P, name, str(e)[:400]))
            print(f"FAIL {gen.__name__} seed={seed} P={P} {name}", flush=True)
        if seed % 50 == 49:
            print(f"... seed {seed}: {n_ok} ok, {len(fails)} fails, "
                  f"{time.time() - t0:.0f}s", flush=True)
    print(f"campaign: {n_ok} ok, {len(fails)} fails in {time.time()-t0:.0f}s")
    for f in fails[:10]:
        print(f)
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
