#!/usr/bin/env python3
"""Randomized GPU-vs-oracle parity fuzzing.

Draws random run-set shapes (run count, entry counts incl. empty runs,
key/value size distributions incl. ragged and zero-length, timestamp
ranges incl. negatives and collisions, overlap patterns) and checks the
HIP engine's output is bit-identical to the CPU oracle for both
keep_tombstones settings. Heavier than the pytest suite — run via gpurun:

    python tools/fuzz_parity.py --trials 150 --seed 1
"""
import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import oracle  # noqa: E402
import dbeel_amd  # noqa: E402
from dbeel_amd.format import Entry, build_run  # noqa: E402


def random_runs(rng):
    n_runs = int(rng.integers(1, 17))
    mode = rng.integers(0, 4)
    # shared key pool so overlaps (incl. >2-way) occur
    pool_size = int(rng.integers(5, 4000))
    if mode == 0:  # fixed short keys
        klen = int(rng.integers(1, 17))
        pool = {bytes(rng.integers(0, 256, klen, dtype=np.uint8))
                for _ in range(pool_size)}
    elif mode == 1:  # ragged keys incl. empty + prefixes
        pool = {b"", b"\x00"}
        for _ in range(pool_size):
            L = int(rng.integers(0, 129))
            k = bytes(rng.integers(0, 256, L, dtype=np.uint8))
            pool.add(k)
            if L > 1 and rng.random() < 0.2:
                pool.add(k[: L // 2])  # prefix pairs
    elif mode == 2:  # low-entropy keys (many shared prefixes)
        pool = {bytes(rng.integers(0, 3, int(rng.integers(1, 12)),
                                   dtype=np.uint8))
                for _ in range(pool_size)}
    else:  # long keys past every staged-suffix aux tier
        pool = {bytes(rng.integers(0, 4, int(rng.integers(41, 129)),
                                   dtype=np.uint8))
                for _ in range(pool_size)}
    pool = list(pool)

    ts_mode = rng.integers(0, 3)
    runs = []
    for r in range(n_runs):
        n = int(rng.integers(0, 600))
        if rng.random() < 0.1:
            n = 0  # empty run
        keys = sorted({pool[int(rng.integers(0, len(pool)))]
                       for _ in range(n)})
        ents = []
        for i, k in enumerate(keys):
            vmode = rng.random()
            if vmode < 0.25:
                data = b""  # tombstone
            else:
                data = bytes(rng.integers(0, 256,
                                          int(rng.integers(1, 300)),
                                          dtype=np.uint8))
            if ts_mode == 0:
                ts = (r << 40) + i  # monotone per run
            elif ts_mode == 1:
                ts = int(rng.integers(-100, 100))  # heavy collisions
            else:
                # full i128 range: compose from 128 random bits
                bits = rng.integers(0, 256, 16, dtype=np.uint8).tobytes()
                ts = int.from_bytes(bits, "little", signed=True)
            ents.append(Entry(k, data, ts))
        runs.append(build_run(ents))
    return runs


def random_scan_filter(rng):
    kw = {}
    if rng.random() < 0.5:
        kw["start_key"] = bytes(rng.integers(0, 256,
                                             int(rng.integers(0, 8)),
                                             dtype=np.uint8))
    if rng.random() < 0.5:
        kw["end_key"] = bytes(rng.integers(0, 256,
                                           int(rng.integers(0, 8)),
                                           dtype=np.uint8))
    if rng.random() < 0.7:
        n = int(rng.integers(1, 4))
        kw["hash_ranges"] = [
            (int(rng.integers(0, 2**32)), int(rng.integers(0, 2**32)))
            for _ in range(n)
        ]
    return kw


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=100)
    ap.add_argument("--seed", type=int, default=1)
    ap.add_argument("--device", type=int, default=0)
    ap.add_argument("--scan", action="store_true",
                    help="also fuzz dbeel_gpu_scan vs the AsyncIter model")
    args = ap.parse_args()

    if args.scan:
        sys.path.insert(0, os.path.join(
            os.path.dirname(os.path.abspath(__file__)), "..", "tests"))
        from pymm3 import scan_model

        from dbeel_amd.engine import scan as gpu_scan

    failures = 0
    for t in range(args.trials):
        rng = np.random.default_rng(args.seed * 1_000_003 + t)
        runs = random_runs(rng)
        for keep in (True, False):
            od, oi, on = oracle.compact(runs, keep_tombstones=keep)
            gd, gi, gn = dbeel_amd.compact(runs, keep_tombstones=keep,
                                           device=args.device)
            if (gd, gi, gn) != (od, oi, on):
                failures += 1
                print(f"TRIAL {t} keep={keep} MISMATCH: "
                      f"n {gn} vs {on}, index {gi == oi}, data {gd == od}")
        if args.scan:
            kw = random_scan_filter(rng)
            sd, si, sn = gpu_scan(runs, device=args.device, **kw)
            md, mi, mn = scan_model(runs, **kw)
            if (sd, si, sn) != (md, mi, mn):
                failures += 1
                print(f"TRIAL {t} SCAN MISMATCH {kw}: n {sn} vs {mn}")
        if (t + 1) % 25 == 0:
            print(f"{t + 1}/{args.trials} trials OK")
    if failures:
        print(f"FUZZ FAILED: {failures} mismatches")
        sys.exit(1)
    print(f"FUZZ OK: {args.trials} trials x 2 settings, all bit-identical")


if __name__ == "__main__":
    main()
