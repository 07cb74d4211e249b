"""Generate golden fixtures for the compaction oracle/GPU parity tests.

Expected outputs are computed by tests/pymerge.py — an INDEPENDENT
pure-Python restatement of lsm_tree.rs:950-1170 — so the C oracle
(oracle/compact_oracle.c) is pinned by cross-agreement of two separately
written restatements plus the reference's own semantic test scenario
(tests/test_oracle.py::test_get_after_compaction_scenario).

Run from the repo root:  python tests/golden/make_golden.py
Fixtures are committed; regenerate only when a case is added.
"""
from __future__ import annotations

import os
import sys
import zlib

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from dbeel_amd.format import Entry, build_run  # noqa: E402
from pymerge import merge  # noqa: E402

OUT_DIR = os.path.dirname(os.path.abspath(__file__))


def e(key: bytes, data: bytes, ts: int) -> Entry:
    return Entry(key, data, ts)


def sorted_run(entries):
    return build_run(sorted(entries, key=lambda x: x.key))


def case_basic(rng):
    # 3 runs, overlapping keys, later runs have newer timestamps, some
    # tombstones superseding old values.
    runs = []
    keys = [bytes([k]) * 4 for k in range(20)]
    runs.append(sorted_run([e(k, b"r0-" + k, 100 + i) for i, k in enumerate(keys[:12])]))
    runs.append(
        sorted_run(
            [e(k, b"", 200 + i) if i % 3 == 0 else e(k, b"r1-" + k, 200 + i)
             for i, k in enumerate(keys[6:16])]
        )
    )
    runs.append(sorted_run([e(k, b"r2-" + k, 300 + i) for i, k in enumerate(keys[10:20])]))
    return runs


def case_tie_ts(rng):
    # Same key, same timestamp in different runs: higher run index wins
    # (lsm_tree.rs:58-65 index tie-break).
    k = b"samekey"
    return [
        sorted_run([e(k, b"run0", 42), e(b"z0", b"v", 1)]),
        sorted_run([e(k, b"run1", 42), e(b"z1", b"v", 2)]),
        sorted_run([e(k, b"run2", 42), e(b"z2", b"v", 3)]),
    ]


def case_all_tombstones(rng):
    keys = [bytes([i, i]) for i in range(10)]
    return [
        sorted_run([e(k, b"old", i) for i, k in enumerate(keys)]),
        sorted_run([e(k, b"", 100 + i) for i, k in enumerate(keys)]),
    ]


def case_empty_run(rng):
    return [
        sorted_run([e(b"a", b"1", 1), e(b"b", b"2", 2)]),
        (b"", b""),
        sorted_run([e(b"b", b"3", 3), e(b"c", b"4", 4)]),
    ]


def case_single_run(rng):
    return [
        sorted_run(
            [e(bytes([i]), b"" if i % 4 == 0 else bytes([i]) * 3, i)
             for i in range(16)]
        )
    ]


def case_ragged(rng):
    # Variable-length keys incl. empty key, prefix pairs ([1] < [1,0]),
    # long keys up to 128 B; variable value sizes incl. large-ish.
    def rk(n):
        return bytes(rng.integers(0, 256, n, dtype=np.uint8))

    pool = [b"", b"\x01", b"\x01\x00", b"\x01\x00\x00", b"\xff" * 128]
    pool += [rk(int(rng.integers(1, 129))) for _ in range(60)]
    pool = sorted(set(pool))
    runs = []
    for r in range(4):
        chosen = sorted(
            {pool[i] for i in rng.choice(len(pool), 30, replace=False)}
        )
        ents = []
        for i, k in enumerate(chosen):
            dn = int(rng.integers(0, 5))
            data = b"" if dn == 0 else bytes(rng.integers(0, 256, dn * 97, dtype=np.uint8))
            ents.append(e(k, data, (r << 40) + i))
        runs.append(build_run(ents))
    return runs


def case_neg_ts(rng):
    # Negative i128 timestamps order below positive ones.
    k = b"kk"
    return [
        sorted_run([e(k, b"neg", -(10**18)), e(b"m", b"x", -5)]),
        sorted_run([e(k, b"pos", 7), e(b"n", b"y", -(1 << 80))]),
    ]


def case_sixteen_runs(rng):
    runs = []
    for r in range(16):
        keys = sorted({bytes(rng.integers(0, 256, 3, dtype=np.uint8)) for _ in range(25)})
        runs.append(
            build_run([e(k, bytes([r]) * (1 + (i % 7)), (r << 32) + i)
                       for i, k in enumerate(keys)])
        )
    return runs


def case_disjoint(rng):
    # No key overlap at all: output = ordered concatenation-by-key.
    return [
        sorted_run([e(bytes([2 * i]), bytes([i]), i) for i in range(10)]),
        sorted_run([e(bytes([2 * i + 1]), bytes([i]), 100 + i) for i in range(10)]),
    ]


def case_ts_shuffle(rng):
    # Winner is max (timestamp, run index), NOT the highest run: give some
    # keys their newest timestamp in an OLDER run.
    keys = [bytes([i, 7]) for i in range(30)]
    runs = []
    ts = rng.permutation(90).reshape(3, 30)
    for r in range(3):
        runs.append(
            sorted_run(
                [e(k, bytes([r]) + k, int(ts[r][i]) - 40)
                 for i, k in enumerate(keys)]
            )
        )
    return runs


CASES = {
    "basic": case_basic,
    "ts_shuffle": case_ts_shuffle,
    "tie_ts": case_tie_ts,
    "all_tombstones": case_all_tombstones,
    "empty_run": case_empty_run,
    "single_run": case_single_run,
    "ragged": case_ragged,
    "neg_ts": case_neg_ts,
    "sixteen_runs": case_sixteen_runs,
    "disjoint": case_disjoint,
}


def main():
    for name, fn in CASES.items():
        rng = np.random.default_rng(0xDBEE1 ^ zlib.crc32(name.encode()))
        runs = fn(rng)
        exp_keep = merge(runs, keep_tombstones=True)
        exp_drop = merge(runs, keep_tombstones=False)
        arrays = {"n_runs": np.array([len(runs)])}
        for i, (d, x) in enumerate(runs):
            arrays[f"run{i}_data"] = np.frombuffer(bytes(d), dtype=np.uint8)
            arrays[f"run{i}_index"] = np.frombuffer(bytes(x), dtype=np.uint8)
        arrays["keep_data"] = np.frombuffer(exp_keep[0], dtype=np.uint8)
        arrays["keep_index"] = np.frombuffer(exp_keep[1], dtype=np.uint8)
        arrays["drop_data"] = np.frombuffer(exp_drop[0], dtype=np.uint8)
        arrays["drop_index"] = np.frombuffer(exp_drop[1], dtype=np.uint8)
        path = os.path.join(OUT_DIR, f"{name}.npz")
        np.savez_compressed(path, **arrays)
        print(f"{name}: {len(runs)} runs, keep={len(exp_keep[1])//16} "
              f"drop={len(exp_drop[1])//16} survivors -> {path}")


if __name__ == "__main__":
    main()
