"""Adversarial parity cases targeting the GPU implementation's own seams.

These are deliberately constructed worst cases for the merge-path corank
design (dbeel_amd/csrc/dbeel_gpu.hip) and for the oracle's heap order —
not random fuzz. Each generator returns a list of runs [(data, index)]
(each run sorted ascending by unique keys, the dbeel flush invariant,
lsm_tree.rs:925-946).

Used by:
  - tests/test_adversarial.py  (CPU: oracle vs the independent heapq-based
    pymerge restatement, bit-exact; plus mutation-sensitivity checks)
  - tests/test_gpu_parity.py::test_adversarial_cases_gpu (GPU vs oracle)
"""
from __future__ import annotations

import numpy as np

from dbeel_amd.format import Entry, build_run

# the corank kernel's window geometry (dbeel_gpu.hip)
CORANK_BLOCK_POS = 4096
CORANK_STEPS = 8


def _run(entries):
    entries = sorted(entries, key=lambda e: e.key)
    keys = [e.key for e in entries]
    assert len(set(keys)) == len(keys), "runs must have unique keys"
    return build_run(entries)


def case_window_straddle():
    """Equal-key pairs straddling the corank kernel's 4096-position windows
    and its 8-position per-thread sub-windows.

    Run B holds one extra key smaller than everything else, so in the
    merged order every duplicate pair (a_i, b_i) sits at positions
    (2i+1, 2i+2) — odd offsets — and therefore straddles every 8-position
    sub-window boundary and the 4096 window boundaries at i=2047, 4095...
    Timestamps alternate which run wins so the supersession flag must be
    read across the boundary in both directions.
    """
    n = 6200  # > one full 4096 window of pairs
    rng = np.random.default_rng(0xA11CE)
    keys = sorted(
        {bytes(rng.integers(0, 256, 16, dtype=np.uint8)) for _ in range(n)}
    )
    run_a, run_b = [], []
    for i, k in enumerate(keys):
        # alternate winner: even i -> run A newer, odd i -> run B newer
        ts_a = 1000 + i * 4 + (2 if i % 2 == 0 else 0)
        ts_b = 1000 + i * 4 + (2 if i % 2 == 1 else 0)
        val_a = b"A" * (17 + (i % 3))
        val_b = b"" if i % 7 == 0 else b"B" * (23 + (i % 5))
        run_a.append(Entry(k, val_a, ts_a))
        run_b.append(Entry(k, val_b, ts_b))
    run_b.append(Entry(b"\x00" * 16, b"tiny", 1))  # the odd-offset shim
    return [_run(run_a), _run(run_b)]


def case_all_runs_same_keys(n_runs=64, ts_mode="asc"):
    """The same key set present in every one of n_runs runs (the MAX_RUNS
    worst case): every entry is in an equal-key group of size n_runs, so
    every crossrank pair carries supersession decisions.

    ts_mode:
      "asc"  — timestamps rise with run index (winner = last run)
      "equal"— ALL timestamps equal: winner decided purely by the run-index
               tie-break (lsm_tree.rs:58-65) — the mutation-sensitive case
      "desc" — timestamps fall with run index: timestamp beats run index,
               winner = run 0 (the inverted-order trap)
    """
    rng = np.random.default_rng(0xBEEF + n_runs)
    n_keys = 300
    keys = sorted(
        {bytes(rng.integers(0, 256, 16, dtype=np.uint8))
         for _ in range(n_keys)}
    )
    runs = []
    for r in range(n_runs):
        if ts_mode == "asc":
            ts = 10_000 + r
        elif ts_mode == "equal":
            ts = 42
        else:
            ts = 10_000 - r
        ents = [
            Entry(k, b"" if (r + i) % 9 == 0 else bytes([r]) * (5 + i % 4),
                  ts)
            for i, k in enumerate(keys)
        ]
        runs.append(_run(ents))
    return runs


def case_aux_boundary(klen):
    """Keys of exactly klen bytes at the aux staging boundaries: ties that
    are only resolvable by the LAST key byte (beyond the staged prefix for
    boundary lengths), prefix-of-each-other families, and exact duplicates
    across runs resolved by timestamp."""
    rng = np.random.default_rng(klen * 7919)
    base = [bytes(rng.integers(0, 256, klen, dtype=np.uint8))
            for _ in range(40)]
    run_a, run_b = [], []
    seen_a, seen_b = set(), set()

    def add(run, seen, k, v, ts):
        if k not in seen:
            seen.add(k)
            run.append(Entry(k, v, ts))

    for i, b in enumerate(base):
        # tie through klen-1 bytes, differ at the last byte
        k1 = b[:-1] + bytes([10])
        k2 = b[:-1] + bytes([200])
        add(run_a, seen_a, k1, b"a1", 100 + i)
        add(run_b, seen_b, k2, b"b2", 100 + i)
        # exact duplicate across runs: ts decides (run B newer)
        add(run_a, seen_a, b, b"old", 50 + i)
        add(run_b, seen_b, b, b"new" * (1 + i % 3), 60 + i)
        # prefix family: key that is a strict prefix of another
        if klen > 2:
            add(run_a, seen_a, b[: klen - 1], b"prefix", 70 + i)
            add(run_b, seen_b, b[: klen - 1] + b"\x00", b"padded", 70 + i)
    return [_run(run_a), _run(run_b)]


def case_long_key_ties():
    """Duplicate long keys (64-128 B) tied through the first 48+ bytes:
    forces the staged-prefix comparison to fall back to the full key bytes
    in the input blob, including exact duplicates of 128-B keys resolved
    by timestamp and near-duplicates differing only at byte 100."""
    rng = np.random.default_rng(0x10A6)
    run_a, run_b, run_c = [], [], []
    for i in range(60):
        stem = bytes(rng.integers(0, 256, 96, dtype=np.uint8))
        k_full = stem + bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        k_diff = bytearray(k_full)
        k_diff[100] ^= 0xFF  # differs deep past any staged prefix
        k_diff = bytes(k_diff)
        run_a.append(Entry(k_full, b"full-a", 200 + i))
        run_b.append(Entry(k_full, b"full-b", 300 + i))  # exact dup, b newer
        run_b.append(Entry(k_diff, b"deep-diff", 200 + i))
        run_c.append(Entry(stem + b"\x00" * 8, b"", 250 + i))  # tombstone
        run_c.append(Entry(stem[:64], b"short-stem", 250 + i))
    return [_run(run_a), _run(run_b), _run(run_c)]


def case_prefix_pad_zero():
    """Keys where zero bytes collide with zero-PADDING in any staged fixed
    prefix: families like "ab", "ab\\0", "ab\\0\\0", "ab\\x01" spread
    across runs with duplicates. Lexicographic order must hold exactly
    (a zero-extended key sorts AFTER its prefix)."""
    fams = []
    for stem in (b"ab", b"q\x00", b"\x00", b"seven88", b"eight889"):
        fams += [stem, stem + b"\x00", stem + b"\x00\x00", stem + b"\x01",
                 stem + b"\x00\x01"]
    fams = sorted(set(fams))
    run_a = [Entry(k, b"A", 10 + i) for i, k in enumerate(fams)]
    run_b = [Entry(k, b"" if i % 3 == 0 else b"B", 20 + i)
             for i, k in enumerate(fams)]  # dups, B newer, some tombstones
    run_c = [Entry(k + b"\xff", b"C", 5) for i, k in enumerate(fams)]
    return [_run(run_a), _run(run_b), _run(run_c)]


def case_ts_extremes():
    """Equal keys with i128 timestamp extremes across runs: both halves of
    the i128 comparison must be signed/unsigned-correct. Includes negative
    vs positive, extremes, and values differing only in the low half."""
    ts_vals = [
        -(2**127), -(2**64), -(2**63) - 1, -1, 0, 1, 2**63, 2**64 + 5,
        2**127 - 1,
    ]
    runs = []
    for r in range(3):
        ents = []
        for i, t in enumerate(ts_vals):
            k = b"tskey%02d" % i
            # rotate which run gets which extreme
            tt = ts_vals[(i + r) % len(ts_vals)]
            ents.append(Entry(k, bytes([r + 1]) * 3, tt))
        runs.append(_run(ents))
    return runs


def all_cases():
    """Named adversarial cases: [(name, runs)]."""
    cases = [
        ("window_straddle", case_window_straddle()),
        ("all64_ts_asc", case_all_runs_same_keys(64, "asc")),
        ("all64_ts_equal", case_all_runs_same_keys(64, "equal")),
        ("all8_ts_desc", case_all_runs_same_keys(8, "desc")),
        ("long_key_ties", case_long_key_ties()),
        ("prefix_pad_zero", case_prefix_pad_zero()),
        ("ts_extremes", case_ts_extremes()),
    ]
    for klen in (15, 16, 17, 31, 32, 33, 39, 40, 41, 47, 48, 49):
        cases.append((f"aux_boundary_{klen}", case_aux_boundary(klen)))
    return cases
